"""Minimal proto3 wire-format primitives.

This framework keeps the reference's wire format (``protobufs/service.proto``
and ``protobufs/npproto/ndarray.proto`` in michaelosthege/pytensor-federated)
bit-compatible so that betterproto/grpclib clients of the reference can talk
to our workers.  The message set is tiny (4 messages, 3 RPCs), so instead of
depending on protoc codegen we implement the proto3 encoding directly:

* varint          (wire type 0)  -- int32/int64 fields
* 64-bit          (wire type 1)  -- unused here
* length-delim.   (wire type 2)  -- bytes / string / embedded message / packed
* 32-bit          (wire type 5)  -- float fields

Negative int64 values (e.g. numpy strides of reversed views) encode as
10-byte two's-complement varints, exactly like protoc output.  Decoders
accept fields in any order, unknown fields, and both packed and unpacked
repeated scalars, per the proto3 spec.
"""
from __future__ import annotations

import struct
from typing import List, Tuple

__all__ = [
    "encode_varint",
    "decode_varint",
    "encode_tag",
    "decode_tag",
    "encode_len_delimited",
    "encode_int64_field",
    "encode_packed_int64",
    "encode_float_field",
    "decode_fields",
    "skip_field",
    "int64_from_uint",
]

_MASK64 = (1 << 64) - 1


def encode_varint(value: int) -> bytes:
    """Encode a non-negative (already masked) integer as a varint."""
    value &= _MASK64
    out = bytearray()
    while True:
        b = value & 0x7F
        value >>= 7
        if value:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def decode_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    """Decode a varint at ``pos``; returns (value, new_pos)."""
    result = 0
    shift = 0
    while True:
        try:
            b = buf[pos]
        except IndexError:
            raise ValueError("truncated varint") from None
        result |= (b & 0x7F) << shift
        pos += 1
        if not b & 0x80:
            return result & _MASK64, pos
        shift += 7
        if shift >= 70:
            raise ValueError("varint too long")


def int64_from_uint(value: int) -> int:
    """Reinterpret an unsigned varint value as a signed int64."""
    if value >= 1 << 63:
        value -= 1 << 64
    return value


def encode_tag(field_number: int, wire_type: int) -> bytes:
    return encode_varint((field_number << 3) | wire_type)


def decode_tag(buf: bytes, pos: int) -> Tuple[int, int, int]:
    """Returns (field_number, wire_type, new_pos)."""
    key, pos = decode_varint(buf, pos)
    return key >> 3, key & 0x7, pos


def encode_len_delimited(field_number: int, payload: bytes) -> bytes:
    return encode_tag(field_number, 2) + encode_varint(len(payload)) + payload


def encode_int64_field(field_number: int, value: int) -> bytes:
    """A single (non-packed) int64/int32 varint field."""
    return encode_tag(field_number, 0) + encode_varint(value)


def encode_packed_int64(field_number: int, values) -> bytes:
    """Packed repeated int64 -- proto3's default for repeated scalars."""
    payload = b"".join(encode_varint(v) for v in values)
    return encode_len_delimited(field_number, payload)


def encode_float_field(field_number: int, value: float) -> bytes:
    return encode_tag(field_number, 5) + struct.pack("<f", value)


def skip_field(buf: bytes, pos: int, wire_type: int) -> int:
    if wire_type == 0:
        _, pos = decode_varint(buf, pos)
        return pos
    if wire_type == 1:
        return pos + 8
    if wire_type == 2:
        size, pos = decode_varint(buf, pos)
        return pos + size
    if wire_type == 5:
        return pos + 4
    raise ValueError(f"unsupported wire type {wire_type}")


def decode_fields(buf: bytes) -> List[Tuple[int, int, object]]:
    """Split a message into (field_number, wire_type, raw_value) triples.

    raw_value is: int for wire types 0, a memoryview for wire type 2, and
    raw 4/8-byte bytes for the fixed types.  Helper for the hand-written
    message classes; unknown fields are returned too (callers ignore them).
    """
    view = memoryview(buf)
    fields: List[Tuple[int, int, object]] = []
    pos = 0
    n = len(buf)
    while pos < n:
        fnum, wtype, pos = decode_tag(buf, pos)
        if wtype == 0:
            value, pos = decode_varint(buf, pos)
            fields.append((fnum, wtype, value))
        elif wtype == 2:
            size, pos = decode_varint(buf, pos)
            if pos + size > n:
                raise ValueError("truncated length-delimited field")
            fields.append((fnum, wtype, view[pos : pos + size]))
            pos += size
        elif wtype == 5:
            if pos + 4 > n:
                raise ValueError("truncated fixed32 field")
            fields.append((fnum, wtype, bytes(view[pos : pos + 4])))
            pos += 4
        elif wtype == 1:
            if pos + 8 > n:
                raise ValueError("truncated fixed64 field")
            fields.append((fnum, wtype, bytes(view[pos : pos + 8])))
            pos += 8
        else:
            raise ValueError(f"unsupported wire type {wtype}")
    return fields


def decode_packed_int64(raw: memoryview) -> List[int]:
    buf = bytes(raw)
    out: List[int] = []
    pos = 0
    while pos < len(buf):
        v, pos = decode_varint(buf, pos)
        out.append(int64_from_uint(v))
    return out
