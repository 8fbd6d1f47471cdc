"""Minimal TensorBoard event-file writer (no tensorboard package needed).

The reference's default logger is TensorBoard (sheeprl/utils/logger.py:12,
configs/logger/tensorboard.yaml); this image ships neither tensorboard nor
tensorflow, so this module hand-encodes the two layers the format needs:

* the TFRecord framing: ``len(8B LE) | masked_crc32c(len) | payload |
  masked_crc32c(payload)`` with the Castagnoli CRC and TensorFlow's mask;
* the protobuf wire encoding of ``Event{wall_time=1, step=2,
  file_version=3, summary=5}`` and ``Summary{value{tag=1, simple_value=2}}``.

Files are named ``events.out.tfevents.<ts>.<host>`` and load in stock
TensorBoard.  Scalars only — exactly what the metric aggregators emit.
"""

from __future__ import annotations

import os
import socket
import struct
import time
from typing import Dict, Optional

_CRC_TABLE = []


def _crc32c_table() -> list:
    global _CRC_TABLE
    if not _CRC_TABLE:
        poly = 0x82F63B78  # Castagnoli, reflected
        table = []
        for n in range(256):
            c = n
            for _ in range(8):
                c = (c >> 1) ^ poly if c & 1 else c >> 1
            table.append(c)
        _CRC_TABLE = table
    return _CRC_TABLE


def _crc32c(data: bytes) -> int:
    table = _crc32c_table()
    crc = 0xFFFFFFFF
    for b in data:
        crc = table[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return ((crc >> 15) | (crc << 17)) + 0xA282EAD8 & 0xFFFFFFFF


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


def _field(num: int, wire: int) -> bytes:
    return _varint((num << 3) | wire)


def _len_delim(num: int, payload: bytes) -> bytes:
    return _field(num, 2) + _varint(len(payload)) + payload


def _double(num: int, v: float) -> bytes:
    return _field(num, 1) + struct.pack("<d", v)


def _float(num: int, v: float) -> bytes:
    return _field(num, 5) + struct.pack("<f", v)


def _int64(num: int, v: int) -> bytes:
    return _field(num, 0) + _varint(v & 0xFFFFFFFFFFFFFFFF)


def _event(wall_time: float, step: Optional[int] = None, file_version: Optional[str] = None,
           summary: Optional[bytes] = None) -> bytes:
    out = _double(1, wall_time)
    if step is not None:
        out += _int64(2, step)
    if file_version is not None:
        out += _len_delim(3, file_version.encode())
    if summary is not None:
        out += _len_delim(5, summary)
    return out


def _scalar_summary(tag: str, value: float) -> bytes:
    val = _len_delim(1, tag.encode()) + _float(2, float(value))
    return _len_delim(1, val)


class TensorBoardWriter:
    """Append-only tfevents writer for scalar metrics."""

    def __init__(self, log_dir: str) -> None:
        os.makedirs(log_dir, exist_ok=True)
        host = socket.gethostname() or "localhost"
        self._path = os.path.join(log_dir, f"events.out.tfevents.{int(time.time())}.{host}")
        self._fh = open(self._path, "ab")
        self._write_record(_event(time.time(), file_version="brain.Event:2"))

    def _write_record(self, payload: bytes) -> None:
        header = struct.pack("<Q", len(payload))
        self._fh.write(header)
        self._fh.write(struct.pack("<I", _masked_crc(header)))
        self._fh.write(payload)
        self._fh.write(struct.pack("<I", _masked_crc(payload)))

    def add_scalar(self, tag: str, value: float, step: int = 0) -> None:
        self._write_record(_event(time.time(), step=step, summary=_scalar_summary(tag, value)))

    def add_scalars(self, metrics: Dict[str, float], step: int = 0) -> None:
        for k, v in metrics.items():
            self.add_scalar(k, v, step)
        self.flush()

    def flush(self) -> None:
        self._fh.flush()

    def close(self) -> None:
        self._fh.flush()
        self._fh.close()
