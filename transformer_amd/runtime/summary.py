"""Minimal TensorBoard-compatible scalar summary writer (C15).

The reference writes per-epoch scalar summaries with tf.summary to
`logs/gradient_tape/<timestamp>/{train,test}` (reference train.py:75-76,
200-206).  Neither tensorflow nor the tensorboard package is available here,
so this module hand-encodes the Event protobuf + TFRecord framing (varint
proto encoding + masked CRC32C) — the files load in stock TensorBoard.
"""

from __future__ import annotations

import os
import socket
import struct
import time

# -- CRC32C (Castagnoli), table-driven ---------------------------------------
_POLY = 0x82F63B78
_TABLE = []
for _i in range(256):
    _c = _i
    for _ in range(8):
        _c = (_c >> 1) ^ (_POLY if _c & 1 else 0)
    _TABLE.append(_c)


def _crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = (_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)) & 0xFFFFFFFF
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return ((crc >> 15 | crc << 17) + 0xA282EAD8) & 0xFFFFFFFF


# -- protobuf primitives -----------------------------------------------------

def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        out.append(b | (0x80 if n else 0))
        if not n:
            return bytes(out)


def _field_bytes(num: int, payload: bytes) -> bytes:
    return _varint((num << 3) | 2) + _varint(len(payload)) + payload


def _event(wall_time: float, step: int | None = None,
           file_version: str | None = None, summary: bytes | None = None) -> bytes:
    out = bytearray()
    out += b"\x09" + struct.pack("<d", wall_time)          # 1: wall_time
    if step is not None:
        out += b"\x10" + _varint(step & 0xFFFFFFFFFFFFFFFF)  # 2: step
    if file_version is not None:
        out += _field_bytes(3, file_version.encode())
    if summary is not None:
        out += _field_bytes(5, summary)
    return bytes(out)


def _scalar_summary(tag: str, value: float) -> bytes:
    val = _field_bytes(1, tag.encode()) + b"\x15" + struct.pack("<f", value)
    return _field_bytes(1, val)  # Summary.value (repeated field 1)


class SummaryWriter:
    """Append-only scalar event writer; API mirrors the subset used by the
    training loop (add_scalar / flush / close)."""

    def __init__(self, log_dir: str):
        os.makedirs(log_dir, exist_ok=True)
        fname = f"events.out.tfevents.{int(time.time())}.{socket.gethostname()}"
        self._f = open(os.path.join(log_dir, fname), "ab")
        self._write(_event(time.time(), file_version="brain.Event:2"))

    def _write(self, record: bytes):
        hdr = struct.pack("<Q", len(record))
        self._f.write(hdr + struct.pack("<I", _masked_crc(hdr)) + record
                      + struct.pack("<I", _masked_crc(record)))

    def add_scalar(self, tag: str, value: float, step: int):
        self._write(_event(time.time(), step=step,
                           summary=_scalar_summary(tag, float(value))))
        self._f.flush()

    def flush(self):
        self._f.flush()

    def close(self):
        self._f.close()
