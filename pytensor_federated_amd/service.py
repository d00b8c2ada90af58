"""gRPC service + client for ``ArraysToArraysService``.

Behavior parity with the reference's grpclib/betterproto implementation
(reference: pytensor_federated/service.py:45-423), rebuilt on ``grpc.aio``
(grpcio is the gRPC stack available in the ROCm image; the wire format is
fixed by service.proto, so reference clients interoperate):

* server: uuid echo, ``n_clients`` counting around the bidirectional
  stream, psutil-backed ``GetLoad`` primed at init (service.py:75-115);
* client: persistent bidirectional stream (the hot path), pickle-safe
  connection cache keyed by (id, pid, thread) so clients can cross
  ``multiprocessing`` boundaries (service.py:214-275), balanced connect
  with de-sync sleep + argmin(n_clients) over live servers
  (service.py:239-263), retry/failover on stream termination
  (service.py:408-416).

On an MI355X worker node this gRPC edge is only the *off-node* interface:
the 8 GPUs of one node exchange per-shard [logp, grads] via RCCL over xGMI
(see pytensor_federated_amd.parallel), never through protobuf.
"""
from __future__ import annotations

import asyncio
import logging
import os
import random
import threading
import uuid as uuid_module
from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np


from .npproto.utils import ndarray_from_numpy, ndarray_to_numpy
from .rpc import (
    GetLoadParams,
    GetLoadResult,
    InputArrays,
    OutputArrays,
    ROUTE_EVALUATE,
    ROUTE_EVALUATE_STREAM,
    ROUTE_GET_LOAD,
    SERVICE_NAME,
)
from .signatures import ComputeFunc
from .utils import argmin_none_or_func, get_useful_event_loop

_log = logging.getLogger(__file__)

__all__ = [
    "ArraysToArraysService",
    "ArraysToArraysServiceClient",
    "ClientPrivates",
    "get_load_async",
    "get_loads_async",
    "thread_pid_id",
    "serve_compute_func",
    "start_server_async",
]

#: range of the random de-synchronization sleep before balanced connect
#: (reference service.py:250). Tests shrink this.
_BALANCE_DESYNC_RANGE = (0.2, 2.0)


def _run_compute_func(
    input_arrays: InputArrays, compute_func: ComputeFunc, exporter=None
) -> OutputArrays:
    """Decode inputs, run the compute function, encode outputs, echo the uuid.

    Parity: reference service.py:45-72.  Decoding is a zero-copy numpy view
    over the message bytes; encoding copies (the protobuf owns its bytes).

    Device arrays: items with dtype ``hipipc/...`` (npproto.device) decode
    to CUDA torch tensors via the dmabuf IPC handle -- HBM -> HBM, no host
    staging.  With an ``exporter`` set, torch CUDA outputs are encoded the
    same way (on-node replies never round-trip through host bytes).
    """
    inputs = []
    for item in input_arrays.items:
        if item.dtype.startswith("hipipc/"):
            from .npproto.device import device_ndarray_to_torch

            inputs.append(device_ndarray_to_torch(item))
        else:
            inputs.append(ndarray_to_numpy(item))
    outputs = compute_func(*inputs)
    items = []
    for o in outputs:
        is_cuda_tensor = (
            type(o).__module__ == "torch" and getattr(o, "is_cuda", False)
        )
        if is_cuda_tensor and exporter is not None:
            items.append(exporter.export(o))
        elif is_cuda_tensor:
            from .npproto.utils import ndarray_from_torch

            items.append(ndarray_from_torch(o))
        else:
            items.append(ndarray_from_numpy(np.asarray(o)))
    return OutputArrays(items=items, uuid=input_arrays.uuid)


class ArraysToArraysService:
    """The server-side service wrapping one :class:`ComputeFunc`.

    Parity: reference service.py:75-115.  ``get_load`` additionally reports
    GPU utilization/VRAM when running on a ROCm device (the MI355X analog of
    the reference's CPU/RAM telemetry) -- folded into the same two floats so
    the message stays wire-compatible.
    """

    def __init__(
        self,
        compute_func: ComputeFunc,
        *,
        report_gpu_load: bool = False,
        device_arrays: bool = False,
    ) -> None:
        """``device_arrays=True``: reply tensors that live on a ROCm device
        are shipped as dmabuf IPC handles (HBM->HBM for same-node clients)
        instead of host byte copies."""
        import psutil

        self._compute_func = compute_func
        self._n_clients = 0
        self._report_gpu_load = report_gpu_load
        self._device_arrays = device_arrays
        # Unary device-array replies rotate through a small ring of export
        # regions so a region is only reused after _UNARY_RING-1 intervening
        # requests (the protocol has no consumption ack; see evaluate()).
        self._unary_exporters = None
        self._unary_idx = 0
        # Prime psutil's CPU monitoring so the first GetLoad is meaningful.
        psutil.getloadavg()

    _UNARY_RING = 4

    def _new_exporter(self):
        """Fresh device-array exporter, or None when device_arrays is off.

        Streamed connections each own one exporter: requests on one stream
        are strictly sequential and the client D2D-copies reply k before
        sending k+1, so resetting between requests is race-free *per
        connection* (a single shared exporter was not -- two concurrent
        streams could overwrite each other's regions mid-read)."""
        if not self._device_arrays:
            return None
        from .npproto.device import DeviceArrayExporter

        return DeviceArrayExporter()

    # -- telemetry ----------------------------------------------------
    @property
    def n_clients(self) -> int:
        return self._n_clients

    def determine_load(self) -> GetLoadResult:
        import psutil

        if self._report_gpu_load:
            gpu = _gpu_load_percent()
            if gpu is not None:
                return GetLoadResult(n_clients=self._n_clients, percent_cpu=gpu[0], percent_ram=gpu[1])
        loadavg_1min = psutil.getloadavg()[0]
        percent_cpu = loadavg_1min / psutil.cpu_count() * 100
        percent_ram = psutil.virtual_memory().percent
        return GetLoadResult(
            n_clients=self._n_clients,
            percent_cpu=percent_cpu,
            percent_ram=percent_ram,
        )

    # -- RPC handlers (grpc.aio behavior functions) --------------------
    async def evaluate(self, input_arrays: InputArrays, context=None) -> OutputArrays:
        exporter = None
        if self._device_arrays:
            if self._unary_exporters is None:
                self._unary_exporters = [
                    self._new_exporter() for _ in range(self._UNARY_RING)
                ]
            exporter = self._unary_exporters[self._unary_idx % self._UNARY_RING]
            self._unary_idx += 1
            exporter.reset()
        return _run_compute_func(input_arrays, self._compute_func, exporter)

    async def evaluate_stream(self, request_iterator, context=None):
        exporter = self._new_exporter()  # per-connection; see _new_exporter
        self._n_clients += 1
        _log.info("A client started a stream. Now serving %i clients.", self._n_clients)
        try:
            async for input_arrays in request_iterator:
                if exporter is not None:
                    exporter.reset()
                yield _run_compute_func(input_arrays, self._compute_func, exporter)
        finally:
            self._n_clients -= 1
            _log.info("A client ended a stream. Now serving %i clients.", self._n_clients)

    async def get_load(self, get_load_params: GetLoadParams, context=None) -> GetLoadResult:
        return self.determine_load()

    # -- server wiring -------------------------------------------------
    def rpc_handlers(self):
        import grpc

        return grpc.method_handlers_generic_handler(
            SERVICE_NAME,
            {
                "Evaluate": grpc.unary_unary_rpc_method_handler(
                    self.evaluate,
                    request_deserializer=InputArrays.FromString,
                    response_serializer=OutputArrays.SerializeToString,
                ),
                "EvaluateStream": grpc.stream_stream_rpc_method_handler(
                    self.evaluate_stream,
                    request_deserializer=InputArrays.FromString,
                    response_serializer=OutputArrays.SerializeToString,
                ),
                "GetLoad": grpc.unary_unary_rpc_method_handler(
                    self.get_load,
                    request_deserializer=GetLoadParams.FromString,
                    response_serializer=GetLoadResult.SerializeToString,
                ),
            },
        )


def _gpu_load_percent() -> Optional[Tuple[float, float]]:
    """(GPU busy %, HBM used %) -- the MI355X analog of the reference's
    CPU/RAM telemetry behind the same GetLoad message (service.py:88-96).

    Prefers amdsmi (the ROCm system-management library); falls back to
    torch's utilization counters; None when no GPU is present.
    """
    try:
        import amdsmi

        amdsmi.amdsmi_init()
        try:
            handles = amdsmi.amdsmi_get_processor_handles()
            if not handles:
                return None
            h = handles[0]
            busy = float(amdsmi.amdsmi_get_gpu_activity(h)["gfx_activity"])
            vram = amdsmi.amdsmi_get_gpu_vram_usage(h)
            vram_pct = vram["vram_used"] / max(1, vram["vram_total"]) * 100.0
            return busy, vram_pct
        finally:
            amdsmi.amdsmi_shut_down()
    except Exception:
        pass
    try:
        import torch

        if not torch.cuda.is_available():
            return None
        free, total = torch.cuda.mem_get_info()
        vram_pct = (1.0 - free / total) * 100.0
        busy = float(torch.cuda.utilization())
        return busy, vram_pct
    except Exception:
        return None


async def start_server_async(service: ArraysToArraysService, bind: str, port: int):
    """Create and start a ``grpc.aio`` server for one service; returns it."""
    import grpc.aio

    server = grpc.aio.server()
    server.add_generic_rpc_handlers((service.rpc_handlers(),))
    server.add_insecure_port(f"{bind}:{port}")
    await server.start()
    _log.info("Serving %s on %s:%i", SERVICE_NAME, bind, port)
    return server


def serve_compute_func(
    compute_func: ComputeFunc, bind: str, port: int, fast_port: Optional[int] = None
) -> None:
    """Blocking convenience: serve one compute function forever.

    ``fast_port`` additionally serves the same function over the low-latency
    fast transport (see ``fastsock``); both transports share the service
    object, so ``n_clients`` telemetry covers both.
    """

    async def _main():
        service = ArraysToArraysService(compute_func)
        server = await start_server_async(service, bind, port)
        if fast_port is not None:
            from .fastsock import start_fast_server_async

            await start_fast_server_async(service, bind, fast_port)
        await server.wait_for_termination()

    asyncio.run(_main())


# ---------------------------------------------------------------------------
# Client side
# ---------------------------------------------------------------------------


async def get_load_async(
    host: str, port: int, timeout: float = 5, transport: str = "grpc"
) -> Optional[GetLoadResult]:
    """Query one server's load; ``None`` if it refuses or times out.

    Parity: reference service.py:161-186.
    """
    if transport == "fast":
        from .fastsock import fast_get_load

        return await fast_get_load(host, port, timeout)
    import grpc
    import grpc.aio

    try:
        async with grpc.aio.insecure_channel(f"{host}:{port}") as channel:
            call = channel.unary_unary(
                ROUTE_GET_LOAD,
                request_serializer=GetLoadParams.SerializeToString,
                response_deserializer=GetLoadResult.FromString,
            )
            return await call(GetLoadParams(), timeout=timeout)
    except (grpc.RpcError, asyncio.TimeoutError, ConnectionError, OSError) as ex:
        _log.debug("GetLoad from %s:%i failed: %s", host, port, ex)
        return None


async def get_loads_async(
    hosts_and_ports: Sequence[Tuple[str, int]],
    timeout: float = 5,
    transport: str = "grpc",
) -> List[Optional[GetLoadResult]]:
    """Concurrently query the load of all servers (reference service.py:189-211)."""
    return list(
        await asyncio.gather(
            *(get_load_async(h, p, timeout=timeout, transport=transport) for h, p in hosts_and_ports)
        )
    )


class ClientPrivates:
    """Bundles the unpicklable connection state of one client.

    Held in the module-global ``_privates`` dict keyed by
    :func:`thread_pid_id`, never on the client object itself, so clients
    survive pickling into ``multiprocessing`` workers (reference
    service.py:214-275).
    """

    def __init__(self, channel, stream, host: str, port: int) -> None:
        self.channel = channel
        self.stream = stream  # grpc.aio StreamStreamCall or None
        self.host = host
        self.port = port
        self.lock = asyncio.Lock()

    @staticmethod
    async def connect(host: str, port: int, transport: str = "grpc") -> "ClientPrivates":
        """Open a channel + persistent bidirectional stream to one server."""
        if transport == "fast":
            from .fastsock import FastStream

            stream = await FastStream.connect(host, port)
            _log.info("Opened fast stream to %s:%s.", host, port)
            return ClientPrivates(None, stream, host, port)
        import grpc.aio

        channel = grpc.aio.insecure_channel(f"{host}:{port}")
        stream = channel.stream_stream(
            ROUTE_EVALUATE_STREAM,
            request_serializer=InputArrays.SerializeToString,
            response_deserializer=OutputArrays.FromString,
        )()
        _log.info("Opened channel and stream to %s:%s.", host, port)
        return ClientPrivates(channel, stream, host, port)

    @staticmethod
    async def connect_balanced(
        hosts_and_ports: Sequence[Tuple[str, int]],
        timeout: float = 5,
        transport: str = "grpc",
    ) -> "ClientPrivates":
        """Connect to the least-busy of several servers.

        Shuffle -> random de-sync sleep -> concurrent load probes -> argmin
        over ``n_clients`` of the live servers (reference service.py:239-263).
        Raises ``TimeoutError`` if none responded.
        """
        # Thread-safe RNG: seed per (pid, thread) to de-correlate forks.
        rng = random.Random(f"{os.getpid()}-{threading.get_ident()}-{random.random()}")
        hap = list(hosts_and_ports)
        rng.shuffle(hap)
        await asyncio.sleep(rng.uniform(*_BALANCE_DESYNC_RANGE))
        loads = await get_loads_async(hap, timeout=timeout, transport=transport)
        idx = argmin_none_or_func(loads, lambda load: load.n_clients)
        if idx is None:
            raise TimeoutError(
                f"None of {len(hap)} servers responded to the load request: {hap}"
            )
        host, port = hap[idx]
        return await ClientPrivates.connect(host, port, transport=transport)


#: module-global connection cache; see :class:`ClientPrivates`.  Bounded:
#: entries beyond _PRIVATES_MAX are evicted oldest-first with their channel
#: closed, so long-lived processes churning clients/threads cannot leak
#: connections (the common cids -- one per live (client, pid, thread) --
#: stay far below the cap).
_privates: Dict[str, ClientPrivates] = {}
_PRIVATES_MAX = 256


async def _evict_privates_lru() -> None:
    while len(_privates) > _PRIVATES_MAX:
        cid, privates = next(iter(_privates.items()))
        del _privates[cid]
        try:
            if privates.stream is not None:
                privates.stream.cancel()
            if privates.channel is not None:
                await privates.channel.close()
        except Exception:
            pass
        _log.info("Evicted idle connection %s (cache > %i).", cid, _PRIVATES_MAX)
        _client_exporters.pop(cid, None)


#: client-side device-array exporters, keyed like ``_privates`` so each
#: (client, pid, thread) owns its region (an exporter on the client object
#: itself would be shared across threads -- an overwrite race -- and would
#: break pickling into multiprocessing workers).
_client_exporters: Dict[str, object] = {}


def _after_fork_in_child() -> None:
    """Give forked children a fresh event loop.

    A forked child inherits the parent's loop object (shared self-pipe fds,
    possibly held locks) -- unusable.  Connections re-establish lazily via
    the pid-keyed ``_privates`` cache, which is why clients survive
    ``multiprocessing`` with both fork and spawn (PyMC chain workers)."""
    try:
        asyncio.set_event_loop(asyncio.new_event_loop())
    except Exception:
        pass


if hasattr(os, "register_at_fork"):
    os.register_at_fork(after_in_child=_after_fork_in_child)


def thread_pid_id(obj: object) -> str:
    """Identifier unique to (object, process, thread) (reference service.py:273-275)."""
    return f"{id(obj)}-{os.getpid()}-{threading.get_ident()}"


async def _streamed_evaluate(stream, input_arrays: InputArrays) -> OutputArrays:
    """One send + one receive on the persistent stream (the hot path).

    Stream end is detected structurally: grpc.aio's ``read()`` returns the
    ``grpc.aio.EOF`` sentinel (checked by identity) once the server closes;
    anything that is not a deserialized ``OutputArrays`` message likewise
    means the stream no longer carries replies.
    """
    import grpc.aio

    await stream.write(input_arrays)
    output = await stream.read()
    if output is None or output is grpc.aio.EOF or not isinstance(output, OutputArrays):
        raise ConnectionError("Bidirectional stream was closed by the server.")
    return output


async def _connect_evaluate_async(
    client: "ArraysToArraysServiceClient",
    input_arrays: InputArrays,
    use_stream: bool,
) -> OutputArrays:
    """Connect (or reuse the cached connection) and evaluate once.

    Parity: reference service.py:278-323.
    """
    cid = thread_pid_id(client)
    privates = _privates.get(cid)
    if privates is None:
        if client._hosts_and_ports:
            privates = await ClientPrivates.connect_balanced(
                client._hosts_and_ports, transport=client._transport
            )
        else:
            privates = await ClientPrivates.connect(
                client._host, client._port, transport=client._transport
            )
        _privates[cid] = privates
        await _evict_privates_lru()

    if client._transport == "fast":
        # the persistent fast connection IS the stream; unary == stream
        async with privates.lock:
            output = await _streamed_evaluate(privates.stream, input_arrays)
        if output.uuid != input_arrays.uuid:
            raise ValueError(
                f"Response uuid {output.uuid} does not match request uuid {input_arrays.uuid}."
            )
        return output

    if use_stream:
        async with privates.lock:
            output = await _streamed_evaluate(privates.stream, input_arrays)
    else:
        call = privates.channel.unary_unary(
            ROUTE_EVALUATE,
            request_serializer=InputArrays.SerializeToString,
            response_deserializer=OutputArrays.FromString,
        )
        output = await call(input_arrays)
    if output.uuid != input_arrays.uuid:
        raise ValueError(
            f"Response uuid {output.uuid} does not match request uuid {input_arrays.uuid}."
        )
    return output


class ArraysToArraysServiceClient:
    """Client facade for the ArraysToArraysService.

    Can be pickled & shared across processes/threads; each (process, thread)
    lazily opens its own connection.  With ``hosts_and_ports`` the connect is
    load-balanced and evaluation retries fail over to surviving servers.

    Parity: reference service.py:326-423.
    """

    def __init__(
        self,
        host: str = None,
        port: int = None,
        *,
        hosts_and_ports: Sequence[Tuple[str, int]] = None,
        use_stream: bool = True,
        retries: int = 2,
        transport: str = "grpc",
        device_arrays: bool = False,
    ) -> None:
        """
        Parameters
        ----------
        host : str
            IP address or host name of the remote gRPC server.
        port : int
            Port of the remote gRPC server.
        hosts_and_ports : list of (host, port) tuples, optional
            Takes precedence over ``host``/``port``; enables balancing+failover.
        use_stream : bool
            Evaluate over the persistent bidirectional stream (much faster
            than unary RPCs) -- the default.
        retries : int
            Failed evaluations are retried this many times, re-balancing to a
            live server after a broken stream.
        transport : str
            "grpc" (default; wire-compatible with reference clients/servers)
            or "fast" (this framework's raw-asyncio framing -- same protobuf
            payloads, ~10-20x lower per-call latency than grpc's C-core in
            containerized deployments).
        device_arrays : bool
            Ship torch CUDA tensor inputs as dmabuf IPC handles (HBM->HBM,
            same-node worker) instead of host byte copies, and return CUDA
            tensors for device-array replies.
        """
        if hosts_and_ports is None and (host is None or port is None):
            raise ValueError("Provide either host+port or hosts_and_ports.")
        self._host = host
        self._port = port
        self._hosts_and_ports = list(hosts_and_ports) if hosts_and_ports else None
        self._use_stream = use_stream
        self._retries = retries
        self._transport = transport
        self._device_arrays = device_arrays

    def __del__(self):
        cid = thread_pid_id(self)
        _client_exporters.pop(cid, None)
        privates = _privates.pop(cid, None)
        if privates is None:
            return
        try:
            loop = asyncio.get_event_loop_policy().get_event_loop()
            if loop.is_closed():
                return

            async def _close():
                try:
                    if privates.stream is not None:
                        privates.stream.cancel()
                finally:
                    if privates.channel is not None:
                        await privates.channel.close()

            if loop.is_running():
                loop.create_task(_close())
            else:
                loop.run_until_complete(_close())
            _log.info("Closed channel to %s:%s.", privates.host, privates.port)
        except Exception:
            pass

    def __call__(self, *inputs: Sequence[np.ndarray]) -> List[np.ndarray]:
        return self.evaluate(*inputs)

    def evaluate(self, *inputs: Sequence[np.ndarray], **kwargs) -> List[np.ndarray]:
        """Synchronous evaluation (drives the event loop)."""
        loop = get_useful_event_loop()
        return loop.run_until_complete(self.evaluate_async(*inputs, **kwargs))

    async def evaluate_async(
        self,
        *inputs: Sequence[np.ndarray],
        use_stream: bool = None,
        retries: int = None,
    ) -> List[np.ndarray]:
        """Evaluate remotely; retries fail over to a live server.

        Parity: reference service.py:376-423 (including the retry loop that
        closes a broken channel so the next attempt re-balances).
        """
        import grpc

        if use_stream is None:
            use_stream = self._use_stream
        if retries is None:
            retries = self._retries

        cid = thread_pid_id(self)
        exporter = _client_exporters.get(cid)
        items = []
        for i in inputs:
            if (
                self._device_arrays
                and type(i).__module__ == "torch"
                and getattr(i, "is_cuda", False)
            ):
                if exporter is None:
                    from .npproto.device import DeviceArrayExporter

                    exporter = DeviceArrayExporter()
                    _client_exporters[cid] = exporter
                items.append(exporter.export(i))
            else:
                items.append(ndarray_from_numpy(np.asarray(i)))
        input_arrays = InputArrays(items=items, uuid=str(uuid_module.uuid4()))
        last_error: Optional[BaseException] = None
        for attempt in range(retries + 1):
            try:
                output = await _connect_evaluate_async(self, input_arrays, use_stream)
                decoded = []
                for item in output.items:
                    if item.dtype.startswith("hipipc/"):
                        from .npproto.device import device_ndarray_to_torch

                        decoded.append(device_ndarray_to_torch(item))
                    else:
                        decoded.append(ndarray_to_numpy(item))
                if exporter is not None:
                    exporter.reset()  # server consumed the request arrays
                return decoded
            except (grpc.RpcError, ConnectionError, OSError) as ex:
                last_error = ex
                _log.warning(
                    "Evaluation attempt %i failed (%s). Closing the broken channel.",
                    attempt,
                    type(ex).__name__,
                )
                privates = _privates.pop(cid, None)
                if privates is not None:
                    try:
                        if privates.stream is not None:
                            privates.stream.cancel()
                        if privates.channel is not None:
                            await privates.channel.close()
                    except Exception:
                        pass
        raise last_error
