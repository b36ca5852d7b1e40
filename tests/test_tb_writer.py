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


def _parse_fields(buf):
    """Decode one protobuf message level into {field_num: [payload, ...]}."""
    out = {}
    i = 0
    while i < len(buf):
        key = 0
        shift = 0
        while True:
            b = buf[i]
            i += 1
            key |= (b & 0x7F) << shift
            shift += 7
            if not b & 0x80:
                break
        num, wire = key >> 3, key & 7
        if wire == 0:                       # varint
            v = 0
            shift = 0
            while True:
                b = buf[i]
                i += 1
                v |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
            payload = v
        elif wire == 1:                     # fixed64
            payload = buf[i:i + 8]
            i += 8
        elif wire == 2:                     # length-delimited
            ln = 0
            shift = 0
            while True:
                b = buf[i]
                i += 1
                ln |= (b & 0x7F) << shift
                shift += 7
                if not b & 0x80:
                    break
            payload = buf[i:i + ln]
            i += ln
        elif wire == 5:                     # fixed32
            payload = buf[i:i + 4]
            i += 4
        else:
            raise AssertionError(f"unexpected wire type {wire}")
        out.setdefault(num, []).append(payload)
    return out


def test_histogram_roundtrip():
    vals = [0.0, 0.5, 1.0, 1.0, 2.0, -1.0]
    with tempfile.TemporaryDirectory() as d:
        w = EventFileWriter(d)
        w.add_histogram("grads/embed", vals, step=3, bins=4)
        w.close()
        recs = _read_records(glob.glob(os.path.join(d, "events.*"))[0])
        ev = _parse_fields(recs[1])
        assert ev[2] == [3]                              # step
        summ = _parse_fields(ev[5][0])
        val = _parse_fields(summ[1][0])
        assert val[1][0] == b"grads/embed"               # tag
        histo = _parse_fields(val[7][0])                 # Value.histo
        unpack = lambda b: struct.unpack("<d", b)[0]
        assert unpack(histo[1][0]) == -1.0               # min
        assert unpack(histo[2][0]) == 2.0                # max
        assert unpack(histo[3][0]) == len(vals)          # num
        assert unpack(histo[4][0]) == sum(vals)          # sum
        assert unpack(histo[5][0]) == sum(v * v for v in vals)
        limits = struct.unpack("<4d", histo[6][0])       # packed doubles
        counts = struct.unpack("<4d", histo[7][0])
        assert limits[-1] == 2.0
        assert sum(counts) == len(vals)
        # -1.0 and 0.0 land in bin 0 ([-1, -0.25)) and bin 1
        assert counts[0] == 1 and counts[-1] == 1        # min / max singleton


def test_histogram_degenerate():
    with tempfile.TemporaryDirectory() as d:
        w = EventFileWriter(d)
        w.add_histogram("const", [3.0, 3.0], step=0)     # zero-width range
        w.add_histogram("empty", [], step=0)             # dropped silently
        w.close()
        recs = _read_records(glob.glob(os.path.join(d, "events.*"))[0])
        assert len(recs) == 2                            # file_version + const
