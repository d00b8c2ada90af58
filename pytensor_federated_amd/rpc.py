"""RPC message types and service routes for ``ArraysToArraysService``.

Wire-compatible with the reference's betterproto-generated module
(reference: pytensor_federated/rpc.py:31-187; schema
protobufs/service.proto:6-41), re-implemented with the hand-written proto3
codec in :mod:`pytensor_federated_amd.proto_wire`:

    message InputArrays  { repeated npproto.ndarray items = 1; string uuid = 2; }
    message OutputArrays { repeated npproto.ndarray items = 1; string uuid = 2; }
    message GetLoadParams {}
    message GetLoadResult { int32 n_clients = 1; float percent_cpu = 2; float percent_ram = 3; }

    service ArraysToArraysService {
        rpc Evaluate(InputArrays) returns (OutputArrays);
        rpc EvaluateStream(stream InputArrays) returns (stream OutputArrays);
        rpc GetLoad(GetLoadParams) returns (GetLoadResult);
    }

Route strings are identical to grpclib's (no proto package ->
``/ArraysToArraysService/<Method>``), so reference clients interoperate.
"""
from __future__ import annotations

import struct
from dataclasses import dataclass, field
from typing import List

from .npproto import Ndarray
from .proto_wire import (
    decode_fields,
    encode_float_field,
    encode_int64_field,
    encode_len_delimited,
    int64_from_uint,
)

__all__ = [
    "InputArrays",
    "OutputArrays",
    "GetLoadParams",
    "GetLoadResult",
    "SERVICE_NAME",
    "ROUTE_EVALUATE",
    "ROUTE_EVALUATE_STREAM",
    "ROUTE_GET_LOAD",
]

SERVICE_NAME = "ArraysToArraysService"
ROUTE_EVALUATE = f"/{SERVICE_NAME}/Evaluate"
ROUTE_EVALUATE_STREAM = f"/{SERVICE_NAME}/EvaluateStream"
ROUTE_GET_LOAD = f"/{SERVICE_NAME}/GetLoad"


@dataclass(eq=False, repr=False)
class InputArrays:
    """Input type message of the ArraysToArraysService."""

    items: List[Ndarray] = field(default_factory=list)
    uuid: str = ""

    def SerializeToString(self) -> bytes:
        parts = [encode_len_delimited(1, item.SerializeToString()) for item in self.items]
        if self.uuid:
            parts.append(encode_len_delimited(2, self.uuid.encode("utf-8")))
        return b"".join(parts)

    def __bytes__(self) -> bytes:
        return self.SerializeToString()

    def parse(self, blob: bytes) -> "InputArrays":
        self.items = []
        self.uuid = ""
        for fnum, wtype, raw in decode_fields(blob):
            if fnum == 1 and wtype == 2:
                self.items.append(Ndarray().parse(bytes(raw)))
            elif fnum == 2 and wtype == 2:
                self.uuid = bytes(raw).decode("utf-8")
        return self

    @classmethod
    def FromString(cls, blob: bytes) -> "InputArrays":
        return cls().parse(blob)

    def __repr__(self) -> str:
        return f"InputArrays(items={self.items!r}, uuid={self.uuid!r})"


@dataclass(eq=False, repr=False)
class OutputArrays:
    """Output type message of the ArraysToArraysService.

    ``uuid`` echoes the uuid of the corresponding :class:`InputArrays`.
    """

    items: List[Ndarray] = field(default_factory=list)
    uuid: str = ""

    SerializeToString = InputArrays.SerializeToString
    __bytes__ = InputArrays.__bytes__

    def parse(self, blob: bytes) -> "OutputArrays":
        self.items = []
        self.uuid = ""
        for fnum, wtype, raw in decode_fields(blob):
            if fnum == 1 and wtype == 2:
                self.items.append(Ndarray().parse(bytes(raw)))
            elif fnum == 2 and wtype == 2:
                self.uuid = bytes(raw).decode("utf-8")
        return self

    @classmethod
    def FromString(cls, blob: bytes) -> "OutputArrays":
        return cls().parse(blob)

    def __repr__(self) -> str:
        return f"OutputArrays(items={self.items!r}, uuid={self.uuid!r})"


@dataclass(eq=False)
class GetLoadParams:
    """Input message for a GetLoad query (empty)."""

    def SerializeToString(self) -> bytes:
        return b""

    def __bytes__(self) -> bytes:
        return b""

    def parse(self, blob: bytes) -> "GetLoadParams":
        return self

    @classmethod
    def FromString(cls, blob: bytes) -> "GetLoadParams":
        return cls()


@dataclass(eq=False)
class GetLoadResult:
    """Result message of a GetLoad query."""

    n_clients: int = 0
    percent_cpu: float = 0.0
    percent_ram: float = 0.0

    def SerializeToString(self) -> bytes:
        parts = []
        if self.n_clients:
            parts.append(encode_int64_field(1, self.n_clients))
        if self.percent_cpu:
            parts.append(encode_float_field(2, self.percent_cpu))
        if self.percent_ram:
            parts.append(encode_float_field(3, self.percent_ram))
        return b"".join(parts)

    def __bytes__(self) -> bytes:
        return self.SerializeToString()

    def parse(self, blob: bytes) -> "GetLoadResult":
        for fnum, wtype, raw in decode_fields(blob):
            if fnum == 1 and wtype == 0:
                self.n_clients = int(int64_from_uint(raw))
            elif fnum == 2 and wtype == 5:
                self.percent_cpu = struct.unpack("<f", raw)[0]
            elif fnum == 3 and wtype == 5:
                self.percent_ram = struct.unpack("<f", raw)[0]
        return self

    @classmethod
    def FromString(cls, blob: bytes) -> "GetLoadResult":
        return cls().parse(blob)
