"""Low-latency raw-asyncio transport ("fast" transport).

gRPC's C-core has a measured ~1-2 ms per-call floor inside this ROCm
container (thread-handoff bound; a raw asyncio TCP roundtrip is 84 us on
the same loopback).  This module keeps the PAYLOAD wire format identical
(the same ``InputArrays``/``OutputArrays``/``GetLoadResult`` protobuf
bytes) but frames it over a persistent raw TCP connection:

    handshake:  b"FEDS1"
    frame:      [1B type][4B little-endian length][payload]

Types: 0x01 Evaluate, 0x02 GetLoad; responses 0x81/0x82; 0xFF error (UTF-8
message).  Semantically the persistent connection IS the reference's
bidirectional EvaluateStream (one send + one receive per evaluation,
``n_clients`` counted while connected -- reference service.py:104-112).

Off-node/third-party clients keep using the gRPC endpoint; this transport
is the framework's own client<->worker fast path (select it with
``ArraysToArraysServiceClient(..., transport="fast")``).
"""
from __future__ import annotations

import asyncio
import logging
from typing import Optional

from .rpc import GetLoadResult, InputArrays, OutputArrays

_log = logging.getLogger(__file__)

__all__ = ["start_fast_server_async", "FastStream", "fast_get_load"]

MAGIC = b"FEDS1"
T_EVAL = 0x01
T_LOAD = 0x02
T_EVAL_R = 0x81
T_LOAD_R = 0x82
T_ERR = 0xFF

#: Upper bound on one frame's payload (default 256 MiB).  The length header
#: is an untrusted 32-bit value; without this cap a single hostile frame
#: forces a ~4 GiB allocation in ``readexactly``.  Override via
#: ``FED_FASTSOCK_MAX_FRAME`` (bytes) for genuinely larger arrays.
import os as _os

MAX_FRAME_BYTES = int(_os.environ.get("FED_FASTSOCK_MAX_FRAME", 256 * 1024 * 1024))


class FrameTooLargeError(ConnectionError):
    """A frame header announced a payload above :data:`MAX_FRAME_BYTES`."""


def _frame(frame_type: int, payload: bytes) -> bytes:
    if len(payload) > MAX_FRAME_BYTES:
        raise FrameTooLargeError(
            f"refusing to send {len(payload)}-byte frame (cap {MAX_FRAME_BYTES})"
        )
    return bytes([frame_type]) + len(payload).to_bytes(4, "little") + payload


async def _read_frame(reader: asyncio.StreamReader):
    hdr = await reader.readexactly(5)
    length = int.from_bytes(hdr[1:5], "little")
    if length > MAX_FRAME_BYTES:
        raise FrameTooLargeError(
            f"peer announced {length}-byte frame (cap {MAX_FRAME_BYTES})"
        )
    payload = await reader.readexactly(length) if length else b""
    return hdr[0], payload


async def start_fast_server_async(service, bind: str, port: int):
    """Serve an ``ArraysToArraysService`` over the fast transport."""
    from .service import _run_compute_func

    async def handle(reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        try:
            magic = await reader.readexactly(5)
            if magic != MAGIC:
                writer.close()
                return
        except (asyncio.IncompleteReadError, ConnectionError):
            return
        service._n_clients += 1
        _log.info("fast client connected. Now serving %i clients.", service._n_clients)
        # Per-connection export region (requests on one connection are
        # sequential, so reset-between-requests is race-free here).
        exporter = service._new_exporter()
        try:
            while True:
                ftype, payload = await _read_frame(reader)
                if ftype == T_EVAL:
                    try:
                        if exporter is not None:
                            exporter.reset()
                        out = _run_compute_func(
                            InputArrays.FromString(payload), service._compute_func,
                            exporter,
                        )
                        writer.write(_frame(T_EVAL_R, out.SerializeToString()))
                    except Exception as ex:  # surface compute errors to client
                        writer.write(_frame(T_ERR, str(ex).encode("utf-8")))
                elif ftype == T_LOAD:
                    writer.write(_frame(T_LOAD_R, service.determine_load().SerializeToString()))
                else:
                    writer.write(_frame(T_ERR, f"unknown frame type {ftype}".encode()))
                await writer.drain()
        except (asyncio.IncompleteReadError, ConnectionError, asyncio.CancelledError):
            pass
        finally:
            service._n_clients -= 1
            _log.info("fast client disconnected. Now serving %i clients.", service._n_clients)
            try:
                writer.close()
            except Exception:
                pass

    server = await asyncio.start_server(handle, bind, port)
    _log.info("Serving fast transport on %s:%i", bind, port)
    return server


class FastStream:
    """Client side of one persistent fast connection.

    Duck-types the grpc.aio stream the client code drives
    (``write(InputArrays)`` / ``read() -> OutputArrays``).
    """

    def __init__(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        self._reader = reader
        self._writer = writer

    @staticmethod
    async def connect(host: str, port: int) -> "FastStream":
        reader, writer = await asyncio.open_connection(host, port)
        writer.write(MAGIC)
        await writer.drain()
        return FastStream(reader, writer)

    async def write(self, input_arrays: InputArrays) -> None:
        try:
            self._writer.write(_frame(T_EVAL, input_arrays.SerializeToString()))
            await self._writer.drain()
        except (ConnectionError, OSError) as ex:
            raise ConnectionError(f"fast stream write failed: {ex}") from ex

    async def read(self) -> OutputArrays:
        try:
            ftype, payload = await _read_frame(self._reader)
        except (asyncio.IncompleteReadError, ConnectionError, OSError) as ex:
            raise ConnectionError(f"fast stream closed: {ex}") from ex
        if ftype == T_ERR:
            raise RuntimeError(f"remote evaluation failed: {payload.decode('utf-8')}")
        if ftype != T_EVAL_R:
            raise ConnectionError(f"unexpected frame type {ftype}")
        return OutputArrays.FromString(payload)

    async def get_load(self) -> GetLoadResult:
        self._writer.write(_frame(T_LOAD, b""))
        await self._writer.drain()
        ftype, payload = await _read_frame(self._reader)
        if ftype != T_LOAD_R:
            raise ConnectionError(f"unexpected frame type {ftype}")
        return GetLoadResult.FromString(payload)

    def cancel(self) -> None:
        self.close()

    def close(self) -> None:
        try:
            self._writer.close()
        except Exception:
            pass


async def fast_get_load(host: str, port: int, timeout: float = 5) -> Optional[GetLoadResult]:
    """Load probe over the fast transport; None on refuse/timeout."""
    try:
        stream = await asyncio.wait_for(FastStream.connect(host, port), timeout)
        try:
            return await asyncio.wait_for(stream.get_load(), timeout)
        finally:
            stream.close()
    except (ConnectionError, OSError, asyncio.TimeoutError):
        return None
