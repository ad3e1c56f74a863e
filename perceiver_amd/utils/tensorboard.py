"""Self-contained TensorBoard scalar logging.

This environment ships no ``tensorboard`` package, so this module writes the
event-file format directly: a TFRecord stream (length + masked CRC32C
framing) of hand-encoded ``Event`` protobuf messages carrying scalar
summaries. Files land under ``<logdir>/events.out.tfevents.<ts>.<host>`` and
load in stock TensorBoard — restoring the reference's Lightning
TensorBoardLogger behavior (scalars per step) without any dependency.

Only scalars are supported; richer summaries stay in the Trainer's JSONL.
"""
from __future__ import annotations

import os
import socket
import struct
import time
from typing import Optional

# ---------------------------------------------------------------- CRC32C
_CRC_TABLE = []


def _build_table():
    poly = 0x82F63B78  # Castagnoli, reflected
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
    return ((((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF)


# ------------------------------------------------------- protobuf encoding
def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _tag(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _pb_string(field: int, s: bytes) -> bytes:
    return _tag(field, 2) + _varint(len(s)) + s


def _pb_double(field: int, v: float) -> bytes:
    return _tag(field, 1) + struct.pack("<d", v)


def _pb_float(field: int, v: float) -> bytes:
    return _tag(field, 5) + struct.pack("<f", v)


def _pb_int(field: int, v: int) -> bytes:
    return _tag(field, 0) + _varint(v)


def _event(wall_time: float, step: int, *, file_version: Optional[str] = None,
           scalar: Optional[tuple] = None) -> bytes:
    body = _pb_double(1, wall_time) + _pb_int(2, step)
    if file_version is not None:
        body += _pb_string(3, file_version.encode())
    if scalar is not None:
        tag_name, value = scalar
        value_msg = _pb_string(1, tag_name.encode()) + _pb_float(2, float(value))
        summary = _pb_string(1, value_msg)  # Summary.value (repeated)
        body += _pb_string(5, summary)      # Event.summary
    return body


class ScalarWriter:
    """Append-only TensorBoard scalar event writer."""

    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        name = f"events.out.tfevents.{int(time.time())}.{socket.gethostname()}"
        self._f = open(os.path.join(logdir, name), "ab")
        self._record(_event(time.time(), 0, file_version="brain.Event:2"))

    def _record(self, payload: bytes):
        header = struct.pack("<Q", len(payload))
        self._f.write(header)
        self._f.write(struct.pack("<I", _masked_crc(header)))
        self._f.write(payload)
        self._f.write(struct.pack("<I", _masked_crc(payload)))

    def add_scalar(self, tag: str, value: float, step: int):
        self._record(_event(time.time(), int(step), scalar=(tag, value)))

    def flush(self):
        self._f.flush()

    def close(self):
        self._f.close()
