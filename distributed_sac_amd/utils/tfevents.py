"""Minimal TensorBoard-compatible tfevents writer (no tensorboard dep).

The reference logs through torch's SummaryWriter (logger.py per variant);
this image has no tensorboard package, so we hand-encode the tfevents
format: length-framed records with masked CRC32C, each record an ``Event``
protobuf carrying ``Summary/simple_value`` scalars.  Readable by standard
TensorBoard.

Proto schema (field numbers from tensorflow/core/util/event.proto):
  Event: 1=wall_time(double) 2=step(int64) 5=summary(Summary)
  Summary: 1=repeated Value;  Value: 1=tag(string) 2=simple_value(float)
"""

from __future__ import annotations

import os
import struct
import time
from typing import Optional

_CRC_TABLE = []


def _build_table():
    poly = 0x82F63B78  # CRC32C (Castagnoli), reflected
    for n in range(256):
        c = n
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        _CRC_TABLE.append(c)


_build_table()


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
            return out + bytes([b7])


def _tag(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _encode_event(wall_time: float, step: int, tag: Optional[str] = None,
                  value: Optional[float] = None,
                  file_version: Optional[str] = None) -> bytes:
    ev = _tag(1, 1) + struct.pack("<d", wall_time)
    if step:
        ev += _tag(2, 0) + _varint(step & 0xFFFFFFFFFFFFFFFF)
    if file_version is not None:
        fv = file_version.encode()
        ev += _tag(3, 2) + _varint(len(fv)) + fv
    if tag is not None:
        tag_b = tag.encode()
        val = (_tag(1, 2) + _varint(len(tag_b)) + tag_b
               + _tag(2, 5) + struct.pack("<f", float(value)))
        summary = _tag(1, 2) + _varint(len(val)) + val
        ev += _tag(5, 2) + _varint(len(summary)) + summary
    return ev


class TFEventWriter:
    """Append-only scalar event writer, one file per directory."""

    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        fname = "events.out.tfevents.%d.dsac" % int(time.time())
        self._f = open(os.path.join(logdir, fname), "ab")
        self._write_record(_encode_event(time.time(), 0,
                                         file_version="brain.Event:2"))

    def _write_record(self, data: bytes) -> None:
        header = struct.pack("<Q", len(data))
        self._f.write(header)
        self._f.write(struct.pack("<I", _masked_crc(header)))
        self._f.write(data)
        self._f.write(struct.pack("<I", _masked_crc(data)))

    def add_scalar(self, tag: str, value: float, step: int,
                   wall_time: Optional[float] = None) -> None:
        self._write_record(_encode_event(wall_time or time.time(), step,
                                         tag, value))

    def flush(self) -> None:
        self._f.flush()

    def close(self) -> None:
        self._f.flush()
        self._f.close()
