"""Minimal protobuf wire codec (hand-rolled, no codegen).

Used by the gRPC surface (servers/grpc_server.py) to encode/decode the
greptime.v1 messages without grpcio-tools (not installed in this image).
Wire format per the protobuf spec: varint (wire 0), 64-bit (wire 1),
length-delimited (wire 2), 32-bit (wire 5).
"""

from __future__ import annotations

import struct


def encode_varint(v: int) -> bytes:
    out = bytearray()
    v &= (1 << 64) - 1
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def decode_varint(buf: bytes, off: int) -> tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[off]
        off += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, off
        shift += 7
        if shift > 70:
            raise ValueError("varint too long")


def zigzag(v: int) -> int:
    return (v << 1) ^ (v >> 63)


def unzigzag(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


class Writer:
    """Append-only message writer."""

    def __init__(self):
        self.parts: list[bytes] = []

    def varint(self, field: int, v: int) -> "Writer":
        self.parts.append(encode_varint(field << 3 | 0))
        self.parts.append(encode_varint(int(v)))
        return self

    def bool(self, field: int, v: bool) -> "Writer":
        return self.varint(field, 1 if v else 0)

    def f64(self, field: int, v: float) -> "Writer":
        self.parts.append(encode_varint(field << 3 | 1))
        self.parts.append(struct.pack("<d", float(v)))
        return self

    def f32(self, field: int, v: float) -> "Writer":
        self.parts.append(encode_varint(field << 3 | 5))
        self.parts.append(struct.pack("<f", float(v)))
        return self

    def bytes(self, field: int, b: bytes) -> "Writer":
        self.parts.append(encode_varint(field << 3 | 2))
        self.parts.append(encode_varint(len(b)))
        self.parts.append(b)
        return self

    def string(self, field: int, s: str) -> "Writer":
        return self.bytes(field, s.encode())

    def msg(self, field: int, sub: "Writer") -> "Writer":
        return self.bytes(field, sub.build())

    def build(self) -> bytes:
        return b"".join(self.parts)


def fields(buf: bytes):
    """Yield (field_no, wire_type, value). value: int for wire 0,
    bytes for wire 2, 8/4-byte raw bytes for wire 1/5."""
    off = 0
    n = len(buf)
    while off < n:
        key, off = decode_varint(buf, off)
        field, wire = key >> 3, key & 7
        if wire == 0:
            v, off = decode_varint(buf, off)
        elif wire == 1:
            v = buf[off:off + 8]
            off += 8
        elif wire == 2:
            ln, off = decode_varint(buf, off)
            v = buf[off:off + ln]
            off += ln
        elif wire == 5:
            v = buf[off:off + 4]
            off += 4
        else:
            raise ValueError(f"unsupported wire type {wire}")
        yield field, wire, v


def as_f64(v) -> float:
    return struct.unpack("<d", v)[0]


def as_f32(v) -> float:
    return struct.unpack("<f", v)[0]


def as_i64(v: int) -> int:
    """Two's-complement interpretation of a varint as int64."""
    return v - (1 << 64) if v >= (1 << 63) else v
