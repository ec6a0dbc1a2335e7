"""Dependency-free TensorBoard event-file READER (scalar records only).

Counterpart of utils/tfevents.py's writer: walks the TFRecord framing
(length + masked-CRC32C header, payload, payload CRC) and hand-decodes
the protobuf Event/Summary messages far enough to recover
(wall_time, step, tag, simple_value) tuples — no tensorboard install
needed (the build image has none, like the reference's plot scripts
assume one).  Used by the learning-curve analysis and tools/.
"""

from __future__ import annotations

import os
import struct
from typing import Dict, Iterator, List, Tuple


def _read_varint(buf: bytes, i: int) -> Tuple[int, int]:
    out = shift = 0
    while True:
        b = buf[i]
        i += 1
        out |= (b & 0x7F) << shift
        if not b & 0x80:
            return out, i
        shift += 7


def _fields(buf: bytes) -> Iterator[Tuple[int, int, bytes]]:
    """(field_number, wire_type, raw_value) over one protobuf message."""
    i = 0
    n = len(buf)
    while i < n:
        key, i = _read_varint(buf, i)
        fn, wt = key >> 3, key & 7
        if wt == 0:            # varint
            v, i = _read_varint(buf, i)
            yield fn, wt, v
        elif wt == 1:          # 64-bit
            yield fn, wt, buf[i:i + 8]
            i += 8
        elif wt == 2:          # length-delimited
            ln, i = _read_varint(buf, i)
            yield fn, wt, buf[i:i + ln]
            i += ln
        elif wt == 5:          # 32-bit
            yield fn, wt, buf[i:i + 4]
            i += 4
        else:
            raise ValueError(f"wire type {wt}")


def iter_records(path: str) -> Iterator[bytes]:
    data = open(path, "rb").read()
    off = 0
    while off + 12 <= len(data):
        (length,) = struct.unpack("<Q", data[off:off + 8])
        payload = data[off + 12:off + 12 + length]
        off += 16 + length
        yield payload


def read_scalars(path: str) -> List[Tuple[float, int, str, float]]:
    """[(wall_time, step, tag, value)] from one events file."""
    out = []
    for rec in iter_records(path):
        wall = 0.0
        step = 0
        vals = []
        for fn, wt, v in _fields(rec):          # Event
            if fn == 1 and wt == 1:             # wall_time double
                wall = struct.unpack("<d", v)[0]
            elif fn == 2 and wt == 0:           # step
                step = v
            elif fn == 5 and wt == 2:           # summary
                for fn2, wt2, v2 in _fields(v):     # Summary
                    if fn2 == 1 and wt2 == 2:       # Summary.Value
                        tag = None
                        val = None
                        for fn3, wt3, v3 in _fields(v2):
                            if fn3 == 1 and wt3 == 2:
                                tag = v3.decode("utf-8", "replace")
                            elif fn3 == 2 and wt3 == 5:   # simple_value
                                val = struct.unpack("<f", v3)[0]
                        if tag is not None and val is not None:
                            vals.append((tag, val))
        for tag, val in vals:
            out.append((wall, step, tag, val))
    return out


def read_scalars_dir(logdir: str) -> Dict[str, List[Tuple[int, float]]]:
    """tag -> [(step, value)] merged over every events file in a dir."""
    series: Dict[str, List[Tuple[int, float]]] = {}
    for f in sorted(os.listdir(logdir)):
        if "tfevents" not in f:
            continue
        for _w, step, tag, val in read_scalars(os.path.join(logdir, f)):
            series.setdefault(tag, []).append((step, val))
    for v in series.values():
        v.sort()
    return series
