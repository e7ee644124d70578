"""Minimal TensorBoard event-file writer (no tensorboard package needed).

Backs the reference's two-writer Summary layout (reference utils.py:21-24;
tags written via utils.py:29-37): one events.out.tfevents.* file per
logdir. Writes standard TFRecord-framed Event protobufs that TensorBoard
reads:
scalars and (PNG-encoded) images. Protobuf messages are hand-encoded on the
wire (the schema is tiny and frozen):

    Event      { 1: double wall_time; 2: int64 step; 3: string file_version;
                 5: Summary summary }
    Summary    { repeated 1: Value }
    Value      { 1: string tag; 2: float simple_value; 4: Image image }
    Image      { 1: int32 height; 2: int32 width; 3: int32 colorspace;
                 4: bytes encoded_image_string }

TFRecord framing: u64 length, u32 masked-crc32c(length), payload,
u32 masked-crc32c(payload).
"""

from __future__ import annotations

import os
import struct
import time


def _crc32c_table():
    poly = 0x82F63B78
    table = []
    for i in range(256):
        c = i
        for _ in range(8):
            c = (c >> 1) ^ poly if c & 1 else c >> 1
        table.append(c)
    return table


_TABLE = _crc32c_table()


def crc32c(data: bytes) -> int:
    c = 0xFFFFFFFF
    for b in data:
        c = _TABLE[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    c = crc32c(data)
    return (((c >> 15) | (c << 17)) + 0xA282EAD8) & 0xFFFFFFFF


# ---- protobuf wire encoding helpers ----

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


def _key(field: int, wire: int) -> bytes:
    return _varint((field << 3) | wire)


def _pb_bytes(field: int, data: bytes) -> bytes:
    return _key(field, 2) + _varint(len(data)) + data


def _pb_string(field: int, s: str) -> bytes:
    return _pb_bytes(field, s.encode("utf-8"))


def _pb_double(field: int, v: float) -> bytes:
    return _key(field, 1) + struct.pack("<d", v)


def _pb_float(field: int, v: float) -> bytes:
    return _key(field, 5) + struct.pack("<f", v)


def _pb_varint_field(field: int, v: int) -> bytes:
    return _key(field, 0) + _varint(v)


def encode_scalar_event(tag: str, value: float, step: int, wall_time: float) -> bytes:
    val = _pb_string(1, tag) + _pb_float(2, float(value))
    summary = _pb_bytes(1, val)
    return (_pb_double(1, wall_time) + _pb_varint_field(2, step)
            + _pb_bytes(5, summary))


def encode_image_event(tag: str, png: bytes, h: int, w: int, step: int,
                       wall_time: float, colorspace: int = 4) -> bytes:
    img = (_pb_varint_field(1, h) + _pb_varint_field(2, w)
           + _pb_varint_field(3, colorspace) + _pb_bytes(4, png))
    val = _pb_string(1, tag) + _pb_bytes(4, img)
    summary = _pb_bytes(1, val)
    return (_pb_double(1, wall_time) + _pb_varint_field(2, step)
            + _pb_bytes(5, summary))


class EventWriter:
    """One events.out.tfevents.* file in ``logdir``."""

    def __init__(self, logdir: str):
        os.makedirs(logdir, exist_ok=True)
        fname = f"events.out.tfevents.{int(time.time())}.cyclegan_amd"
        self._f = open(os.path.join(logdir, fname), "ab")
        first = _pb_double(1, time.time()) + _pb_string(3, "brain.Event:2")
        self._write_record(first)

    def _write_record(self, payload: bytes):
        hdr = struct.pack("<Q", len(payload))
        self._f.write(hdr)
        self._f.write(struct.pack("<I", _masked_crc(hdr)))
        self._f.write(payload)
        self._f.write(struct.pack("<I", _masked_crc(payload)))
        self._f.flush()

    def scalar(self, tag: str, value: float, step: int):
        self._write_record(encode_scalar_event(tag, value, step, time.time()))

    def image_png(self, tag: str, png: bytes, h: int, w: int, step: int):
        self._write_record(encode_image_event(tag, png, h, w, step, time.time()))

    def close(self):
        self._f.close()
