"""Minimal TensorBoard event-file writer — capability C16.

Writes ``events.out.tfevents.*`` files readable by TensorBoard: the
TFRecord framing (u64 length + masked-crc32c of the length + payload +
masked-crc32c of the payload) around hand-encoded ``Event`` protos
(wall_time=1 double, step=2 int64, file_version=3 string,
summary=5 { value=1 { tag=1 string, simple_value=2 float } }).

No TensorFlow / protobuf dependency; CRC comes from the native bundle
codec (csrc/bundle masked_crc32c).
"""

from __future__ import annotations

import os
import socket
import struct
import time
from typing import Optional

import numpy as np

from rocps import _bundle as _C

__all__ = ["EventFileWriter"]


def _varint(v: int) -> bytes:
    out = b""
    while v >= 0x80:
        out += bytes([v & 0x7F | 0x80])
        v >>= 7
    return out + bytes([v])


def _tag(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _crc(data: bytes) -> int:
    return _C.masked_crc32c(np.frombuffer(data, dtype=np.uint8))


class EventFileWriter:
    def __init__(self, logdir: str, filename_suffix: str = ""):
        os.makedirs(logdir, exist_ok=True)
        fname = (f"events.out.tfevents.{int(time.time())}."
                 f"{socket.gethostname()}.{os.getpid()}{filename_suffix}")
        self.path = os.path.join(logdir, fname)
        self._f = open(self.path, "ab")
        self._write_event(self._event(wall_time=time.time(),
                                      file_version="brain.Event:2"))

    # -- proto encoding -----------------------------------------------------
    @staticmethod
    def _event(wall_time: float, step: Optional[int] = None,
               file_version: Optional[str] = None,
               summary: Optional[bytes] = None) -> bytes:
        out = _tag(1, 1) + struct.pack("<d", wall_time)
        if step is not None:
            out += _tag(2, 0) + _varint(step)
        if file_version is not None:
            fv = file_version.encode()
            out += _tag(3, 2) + _varint(len(fv)) + fv
        if summary is not None:
            out += _tag(5, 2) + _varint(len(summary)) + summary
        return out

    @staticmethod
    def _scalar_summary(tag: str, value: float) -> bytes:
        t = tag.encode()
        val = (_tag(1, 2) + _varint(len(t)) + t            # Value.tag
               + _tag(2, 5) + struct.pack("<f", value))    # simple_value
        return _tag(1, 2) + _varint(len(val)) + val        # Summary.value

    def _write_event(self, ev: bytes) -> None:
        header = struct.pack("<Q", len(ev))
        self._f.write(header)
        self._f.write(struct.pack("<I", _crc(header)))
        self._f.write(ev)
        self._f.write(struct.pack("<I", _crc(ev)))

    # -- public API ---------------------------------------------------------
    def add_scalar(self, tag: str, value: float, step: int) -> None:
        self._write_event(self._event(
            wall_time=time.time(), step=int(step),
            summary=self._scalar_summary(tag, float(value))))

    def flush(self) -> None:
        self._f.flush()

    def close(self) -> None:
        self._f.flush()
        self._f.close()
