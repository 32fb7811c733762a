"""Minimal TensorBoard event-file writer (no tensorboard package needed).

The reference writes TF summary events (distributed_train.py:382-390,
nn_eval.py:107-110); this environment has no `tensorboard` module, so this
hand-encodes the Event/Summary protobuf wire format and TFRecord framing
(length + masked crc32c) directly — readable by any standard TensorBoard.

Only scalar summaries are supported (all the reference used).
"""

from __future__ import annotations

import os
import socket
import struct
import time

# ---- crc32c (Castagnoli), table-driven ------------------------------------
_POLY = 0x82F63B78
_TABLE = []
for _i in range(256):
    _c = _i
    for _ in range(8):
        _c = (_c >> 1) ^ _POLY if _c & 1 else _c >> 1
    _TABLE.append(_c)


def _crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = _TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ---- protobuf wire helpers -------------------------------------------------

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


def _field_varint(num: int, val: int) -> bytes:
    return _varint((num << 3) | 0) + _varint(val)


def _field_double(num: int, val: float) -> bytes:
    return _varint((num << 3) | 1) + struct.pack("<d", val)


def _field_float(num: int, val: float) -> bytes:
    return _varint((num << 3) | 5) + struct.pack("<f", val)


def _field_bytes(num: int, val: bytes) -> bytes:
    return _varint((num << 3) | 2) + _varint(len(val)) + val


def _event(wall_time: float, step: int | None = None,
           file_version: str | None = None, summary: bytes | None = None) -> bytes:
    out = _field_double(1, wall_time)
    if step is not None:
        out += _field_varint(2, step)
    if file_version is not None:
        out += _field_bytes(3, file_version.encode())
    if summary is not None:
        out += _field_bytes(5, summary)
    return out


def _scalar_summary(tag: str, value: float) -> bytes:
    v = _field_bytes(1, tag.encode()) + _field_float(2, float(value))
    return _field_bytes(1, v)


class EventWriter:
    """Drop-in minimal SummaryWriter: add_scalar(tag, value, step), close()."""

    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        fname = (f"events.out.tfevents.{int(time.time())}."
                 f"{socket.gethostname()}.{os.getpid()}.v2")
        self._f = open(os.path.join(logdir, fname), "ab")
        self._write_record(_event(time.time(), file_version="brain.Event:2"))

    def _write_record(self, data: bytes):
        hdr = struct.pack("<Q", len(data))
        self._f.write(hdr)
        self._f.write(struct.pack("<I", _masked_crc(hdr)))
        self._f.write(data)
        self._f.write(struct.pack("<I", _masked_crc(data)))
        self._f.flush()

    def add_scalar(self, tag: str, value: float, step: int):
        self._write_record(_event(time.time(), step=int(step),
                                  summary=_scalar_summary(tag, value)))

    def close(self):
        self._f.close()


def make_writer(logdir: str):
    """torch.utils.tensorboard when available, else the minimal writer."""
    try:
        from torch.utils.tensorboard import SummaryWriter
        return SummaryWriter(logdir)
    except Exception:
        return EventWriter(logdir)
