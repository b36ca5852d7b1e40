"""Minimal TensorBoard event-file writer (no tensorboard package needed).

The reference logs scalars to TensorBoard (utils/logs_utils.py:187-224);
this environment has no tensorboard install, so we hand-encode the event
protobuf (Event{wall_time=1, step=2, summary=5{value=1{tag=1,
simple_value=2}}}) and the TFRecord framing (length + masked crc32c).
Files are readable by any standard TensorBoard."""

from __future__ import annotations

import os
import socket
import struct
import time
from typing import Optional

_CRC_TABLE = []


def _make_table():
    poly = 0x82F63B78          # Castagnoli, reflected
    for n in range(256):
        c = n
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        _CRC_TABLE.append(c)


_make_table()


def _crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = _CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


def _varint(n: int) -> bytes:
    out = b""
    while True:
        b7 = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b7 | 0x80])
        else:
            out += bytes([b7])
            return out


def _field(num: int, wire: int) -> bytes:
    return _varint((num << 3) | wire)


def _encode_event(wall_time: float, step: int, tag: Optional[str] = None,
                  value: Optional[float] = None,
                  file_version: Optional[str] = None) -> bytes:
    ev = _field(1, 1) + struct.pack("<d", wall_time)      # wall_time: double
    ev += _field(2, 0) + _varint(step & 0xFFFFFFFFFFFFFFFF)  # step: int64
    if file_version is not None:
        fv = file_version.encode()
        ev += _field(3, 2) + _varint(len(fv)) + fv
    if tag is not None:
        t = tag.encode()
        val = (_field(1, 2) + _varint(len(t)) + t +
               _field(2, 5) + struct.pack("<f", value))     # simple_value
        summ = _field(1, 2) + _varint(len(val)) + val       # Summary.value
        ev += _field(5, 2) + _varint(len(summ)) + summ      # Event.summary
    return ev


class EventFileWriter:
    def __init__(self, log_dir: str):
        os.makedirs(log_dir, exist_ok=True)
        fname = (f"events.out.tfevents.{int(time.time())}."
                 f"{socket.gethostname()}")
        self._f = open(os.path.join(log_dir, fname), "ab")
        self._write_record(_encode_event(time.time(), 0,
                                         file_version="brain.Event:2"))

    def _write_record(self, data: bytes) -> None:
        hdr = struct.pack("<Q", len(data))
        self._f.write(hdr)
        self._f.write(struct.pack("<I", _masked_crc(hdr)))
        self._f.write(data)
        self._f.write(struct.pack("<I", _masked_crc(data)))
        self._f.flush()

    def add_scalar(self, tag: str, value: float, step: int) -> None:
        self._write_record(_encode_event(time.time(), step, tag,
                                         float(value)))

    def add_histogram(self, tag: str, values, step: int,
                      bins: int = 30) -> None:
        """Summary.Value.histo (HistogramProto) — param/grad distribution
        logging the reference gets from SummaryWriter; linear bins."""
        vals = [float(v) for v in values]
        if not vals:
            return
        lo, hi = min(vals), max(vals)
        if hi == lo:
            hi = lo + 1e-12
        width = (hi - lo) / bins
        counts = [0.0] * bins
        for v in vals:
            idx = int((v - lo) / width)
            counts[min(idx, bins - 1)] += 1.0
        limits = [lo + width * (i + 1) for i in range(bins)]
        histo = (_field(1, 1) + struct.pack("<d", lo)
                 + _field(2, 1) + struct.pack("<d", hi)
                 + _field(3, 1) + struct.pack("<d", float(len(vals)))
                 + _field(4, 1) + struct.pack("<d", sum(vals))
                 + _field(5, 1) + struct.pack("<d", sum(v * v for v in vals)))
        for num, seq in ((6, limits), (7, counts)):     # packed doubles
            body = b"".join(struct.pack("<d", x) for x in seq)
            histo += _field(num, 2) + _varint(len(body)) + body
        t = tag.encode()
        val = (_field(1, 2) + _varint(len(t)) + t
               + _field(7, 2) + _varint(len(histo)) + histo)  # Value.histo
        summ = _field(1, 2) + _varint(len(val)) + val
        ev = (_field(1, 1) + struct.pack("<d", time.time())
              + _field(2, 0) + _varint(step & 0xFFFFFFFFFFFFFFFF)
              + _field(5, 2) + _varint(len(summ)) + summ)
        self._write_record(ev)

    def close(self) -> None:
        self._f.close()
