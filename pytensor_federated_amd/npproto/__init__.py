"""Wire format for one ndarray (``npproto.ndarray`` message).

Mirrors the reference's betterproto-generated ``npproto.Ndarray``
(reference: pytensor_federated/npproto/__init__.py:12-23; schema
protobufs/npproto/ndarray.proto:7-12) with a hand-written proto3 codec:

    message ndarray {
        bytes data = 1;
        string dtype = 2;
        repeated int64 shape = 3;
        repeated int64 strides = 4;
    }

API kept betterproto-compatible where the reference relied on it:
``bytes(msg)`` serializes, ``Ndarray().parse(blob)`` deserializes.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List

from ..proto_wire import (
    decode_fields,
    decode_packed_int64,
    encode_len_delimited,
    encode_packed_int64,
    int64_from_uint,
)

__all__ = ["Ndarray"]


@dataclass(eq=False, repr=False)
class Ndarray:
    """Represents a NumPy array of arbitrary shape or dtype.

    The array must support the buffer protocol.
    """

    data: bytes = b""
    dtype: str = ""
    shape: List[int] = field(default_factory=list)
    strides: List[int] = field(default_factory=list)

    # -- proto3 codec -------------------------------------------------
    def SerializeToString(self) -> bytes:
        parts = []
        if self.data:
            parts.append(encode_len_delimited(1, bytes(self.data)))
        if self.dtype:
            parts.append(encode_len_delimited(2, self.dtype.encode("utf-8")))
        if self.shape:
            parts.append(encode_packed_int64(3, self.shape))
        if self.strides:
            parts.append(encode_packed_int64(4, self.strides))
        return b"".join(parts)

    def __bytes__(self) -> bytes:
        return self.SerializeToString()

    def parse(self, blob: bytes) -> "Ndarray":
        data = b""
        dtype = ""
        shape: List[int] = []
        strides: List[int] = []
        for fnum, wtype, raw in decode_fields(blob):
            if fnum == 1 and wtype == 2:
                data = bytes(raw)
            elif fnum == 2 and wtype == 2:
                dtype = bytes(raw).decode("utf-8")
            elif fnum == 3:
                if wtype == 2:
                    shape.extend(decode_packed_int64(raw))
                elif wtype == 0:
                    shape.append(int64_from_uint(raw))
            elif fnum == 4:
                if wtype == 2:
                    strides.extend(decode_packed_int64(raw))
                elif wtype == 0:
                    strides.append(int64_from_uint(raw))
        self.data = data
        self.dtype = dtype
        self.shape = shape
        self.strides = strides
        return self

    @classmethod
    def FromString(cls, blob: bytes) -> "Ndarray":
        return cls().parse(blob)

    def __repr__(self) -> str:
        return f"Ndarray(dtype={self.dtype!r}, shape={self.shape}, nbytes={len(self.data)})"
