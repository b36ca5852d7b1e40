"""The hand-encoded TensorBoard event file: record framing (length +
masked crc32c) and protobuf fields decode back correctly."""

import glob
import os
import struct
import tempfile

from acco_amd.utils.tb_writer import (EventFileWriter, _crc32c, _masked_crc)


def test_crc32c_known_vectors():
    # RFC 3720 test vector: 32 bytes of zeros -> 0x8A9136AA
    assert _crc32c(b"\x00" * 32) == 0x8A9136AA
    assert _crc32c(b"123456789") == 0xE3069283


def _read_records(path):
    recs = []
    with open(path, "rb") as f:
        while True:
            hdr = f.read(8)
            if len(hdr) < 8:
                break
            (length,) = struct.unpack("<Q", hdr)
            (crc_hdr,) = struct.unpack("<I", f.read(4))
            assert crc_hdr == _masked_crc(hdr)
            data = f.read(length)
            (crc_data,) = struct.unpack("<I", f.read(4))
            assert crc_data == _masked_crc(data)
            recs.append(data)
    return recs


def test_event_file_roundtrip():
    with tempfile.TemporaryDirectory() as d:
        w = EventFileWriter(d)
        w.add_scalar("loss/train", 2.5, 7)
        w.add_scalar("lr", 1e-4, 8)
        w.close()
        files = glob.glob(os.path.join(d, "events.out.tfevents.*"))
        assert len(files) == 1
        recs = _read_records(files[0])
        assert len(recs) == 3                       # file_version + 2 scalars
        assert b"brain.Event:2" in recs[0]
        assert b"loss/train" in recs[1]
        # simple_value 2.5 as little-endian f32 appears in the record
        assert struct.pack("<f", 2.5) in recs[1]
        assert b"lr" in recs[2]
