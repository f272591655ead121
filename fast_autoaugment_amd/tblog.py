"""Dependency-free TensorBoard scalar logging.

The reference logs per-split scalars through tensorboardX
(reference train.py:176-181, metrics.py:88-93). This module writes real
TensorBoard event files (TFRecord framing + the Event/Summary protobuf wire
format encoded by hand) so `tensorboard --logdir logs/` works against our
runs without tensorboard/tensorboardX being importable at train time.

Wire format notes (stable since TF 1.x):
- record  = uint64 len | uint32 masked_crc32c(len) | payload | masked_crc32c(payload)
- payload = Event proto: wall_time (field 1, double), step (field 2, int64),
  file_version (field 3, string, first record only) or summary (field 5).
- Summary = repeated Value (field 1); Value = tag (field 1, string) +
  simple_value (field 2, float).
"""
from __future__ import annotations

import os
import socket
import struct
import time
from typing import List, Tuple

# ----------------------------------------------------------------- crc32c
# Castagnoli CRC-32 (reflected poly 0x82F63B78), table-driven.
_CRC_TABLE = []
for _i in range(256):
    _c = _i
    for _ in range(8):
        _c = (_c >> 1) ^ 0x82F63B78 if _c & 1 else _c >> 1
    _CRC_TABLE.append(_c)


def _crc32c(data: bytes) -> int:
    crc = 0xFFFFFFFF
    for b in data:
        crc = _CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = _crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ------------------------------------------------------------- protobuf enc
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


def _field_bytes(num: int, payload: bytes) -> bytes:
    return _varint((num << 3) | 2) + _varint(len(payload)) + payload


def _field_double(num: int, v: float) -> bytes:
    return _varint((num << 3) | 1) + struct.pack("<d", v)


def _field_float(num: int, v: float) -> bytes:
    return _varint((num << 3) | 5) + struct.pack("<f", v)


def _field_varint(num: int, v: int) -> bytes:
    return _varint(num << 3) + _varint(v & 0xFFFFFFFFFFFFFFFF)


def _event(wall_time: float, step: int = 0, file_version: str = None,
           summary: bytes = None) -> bytes:
    out = _field_double(1, wall_time)
    if step:
        out += _field_varint(2, step)
    if file_version is not None:
        out += _field_bytes(3, file_version.encode())
    if summary is not None:
        out += _field_bytes(5, summary)
    return out


def _scalar_summary(tag: str, value: float) -> bytes:
    val = _field_bytes(1, tag.encode()) + _field_float(2, float(value))
    return _field_bytes(1, val)


class SummaryWriter:
    """Minimal tensorboardX-compatible scalar writer (add_scalar/flush/close)."""

    def __init__(self, log_dir: str):
        self.log_dir = log_dir
        os.makedirs(log_dir, exist_ok=True)
        fname = "events.out.tfevents.%010d.%s" % (time.time(), socket.gethostname())
        self._f = open(os.path.join(log_dir, fname), "wb")
        self._write(_event(time.time(), file_version="brain.Event:2"))
        self.flush()

    def _write(self, payload: bytes):
        hdr = struct.pack("<Q", len(payload))
        self._f.write(hdr + struct.pack("<I", _masked_crc(hdr))
                      + payload + struct.pack("<I", _masked_crc(payload)))

    def add_scalar(self, tag: str, value, global_step: int = 0):
        try:
            value = float(value)
        except (TypeError, ValueError):
            return
        self._write(_event(time.time(), step=int(global_step),
                           summary=_scalar_summary(tag, value)))

    def flush(self):
        self._f.flush()

    def close(self):
        if not self._f.closed:
            self.flush()
            self._f.close()

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


# ---------------------------------------------------------------- reading
# Decoder used by tests (round-trips framing, CRCs and the scalar fields).
def read_scalars(path: str) -> List[Tuple[int, str, float]]:
    """Parse an event file -> [(step, tag, value)]. Validates record CRCs."""
    out = []
    with open(path, "rb") as f:
        data = f.read()
    pos = 0
    while pos < len(data):
        (length,) = struct.unpack_from("<Q", data, pos)
        (hcrc,) = struct.unpack_from("<I", data, pos + 8)
        assert hcrc == _masked_crc(data[pos:pos + 8]), "length crc mismatch"
        payload = data[pos + 12:pos + 12 + length]
        (pcrc,) = struct.unpack_from("<I", data, pos + 12 + length)
        assert pcrc == _masked_crc(payload), "payload crc mismatch"
        pos += 12 + length + 4
        out.extend(_parse_event(payload))
    return out


def _read_varint(buf: bytes, pos: int):
    n = shift = 0
    while True:
        b = buf[pos]
        pos += 1
        n |= (b & 0x7F) << shift
        if not b & 0x80:
            return n, pos
        shift += 7


def _parse_fields(buf: bytes):
    pos = 0
    while pos < len(buf):
        key, pos = _read_varint(buf, pos)
        num, wire = key >> 3, key & 7
        if wire == 0:
            v, pos = _read_varint(buf, pos)
        elif wire == 1:
            v = buf[pos:pos + 8]
            pos += 8
        elif wire == 2:
            ln, pos = _read_varint(buf, pos)
            v = buf[pos:pos + ln]
            pos += ln
        elif wire == 5:
            v = buf[pos:pos + 4]
            pos += 4
        else:
            raise ValueError(f"wire type {wire}")
        yield num, wire, v


def _parse_event(payload: bytes):
    step, summary = 0, None
    for num, wire, v in _parse_fields(payload):
        if num == 2 and wire == 0:
            step = v
        elif num == 5 and wire == 2:
            summary = v
    if summary is None:
        return []
    out = []
    for num, wire, v in _parse_fields(summary):
        if num == 1 and wire == 2:  # Summary.Value
            tag, val = None, None
            for n2, w2, v2 in _parse_fields(v):
                if n2 == 1 and w2 == 2:
                    tag = v2.decode()
                elif n2 == 2 and w2 == 5:
                    (val,) = struct.unpack("<f", v2)
            if tag is not None and val is not None:
                out.append((step, tag, val))
    return out
