"""Fast-transport tests: same behaviors as the gRPC edge, lower latency."""
import asyncio
import multiprocessing
import socket
import time

import numpy as np
import pytest

import pytensor_federated_amd.service as service_mod
from pytensor_federated_amd.service import (
    ArraysToArraysService,
    ArraysToArraysServiceClient,
    _privates,
    get_load_async,
    thread_pid_id,
)

FAST_PORTS = (9561, 9562)


def _serve_fast(port: int):
    import asyncio

    from pytensor_federated_amd.fastsock import start_fast_server_async
    from pytensor_federated_amd.service import ArraysToArraysService

    async def main():
        service = ArraysToArraysService(lambda a, b: [a * b])
        server = await start_fast_server_async(service, "127.0.0.1", port)
        async with server:
            await server.serve_forever()

    asyncio.run(main())


def _wait_tcp(port, timeout=30.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=1):
                return
        except OSError:
            time.sleep(0.1)
    raise TimeoutError(f"port {port} never opened")


@pytest.fixture(scope="module")
def fast_servers():
    ctx = multiprocessing.get_context("spawn")
    procs = {p: ctx.Process(target=_serve_fast, args=(p,), daemon=True) for p in FAST_PORTS}
    for proc in procs.values():
        proc.start()
    try:
        for port in FAST_PORTS:
            _wait_tcp(port)
        yield procs
    finally:
        for proc in procs.values():
            if proc.is_alive():
                proc.terminate()
        for proc in procs.values():
            proc.join(timeout=10)


@pytest.mark.timeout(120)
def test_fast_evaluate_roundtrip(fast_servers):
    client = ArraysToArraysServiceClient("127.0.0.1", FAST_PORTS[0], transport="fast")
    a, b = np.array([1.0, 4.0]), np.array([2.0, 3.0])
    for _ in range(5):
        (out,) = client.evaluate(a, b)
    np.testing.assert_array_equal(out, a * b)
    del client


@pytest.mark.timeout(120)
def test_fast_get_load_and_n_clients(fast_servers):
    load = asyncio.run(get_load_async("127.0.0.1", FAST_PORTS[0], transport="fast"))
    assert load is not None
    client = ArraysToArraysServiceClient("127.0.0.1", FAST_PORTS[0], transport="fast")
    client.evaluate(np.array(2.0), np.array(3.0))  # opens persistent connection
    load = asyncio.run(get_load_async("127.0.0.1", FAST_PORTS[0], transport="fast"))
    assert load.n_clients >= 1
    del client


@pytest.mark.timeout(120)
def test_fast_remote_error_propagates(fast_servers):
    client = ArraysToArraysServiceClient("127.0.0.1", FAST_PORTS[0], transport="fast", retries=0)
    with pytest.raises(RuntimeError, match="remote evaluation failed"):
        client.evaluate(np.array(2.0))  # arity error on the worker
    del client


FAILOVER_FAST_PORTS = (9563, 9564)


@pytest.mark.timeout(180)
def test_fast_failover(monkeypatch):
    # own servers: this test kills one of them
    monkeypatch.setattr(service_mod, "_BALANCE_DESYNC_RANGE", (0.0, 0.01))
    ctx = multiprocessing.get_context("spawn")
    procs = {
        p: ctx.Process(target=_serve_fast, args=(p,), daemon=True) for p in FAILOVER_FAST_PORTS
    }
    for proc in procs.values():
        proc.start()
    try:
        for port in FAILOVER_FAST_PORTS:
            _wait_tcp(port)
        hap = [("127.0.0.1", p) for p in FAILOVER_FAST_PORTS]
        client = ArraysToArraysServiceClient(hosts_and_ports=hap, transport="fast", retries=2)
        a, b = np.array(2.0), np.array(5.0)
        (out,) = client.evaluate(a, b)
        np.testing.assert_array_equal(out, np.array(10.0))
        connected_port = _privates[thread_pid_id(client)].port
        procs[connected_port].terminate()
        procs[connected_port].join()
        (out,) = client.evaluate(a, b)
        np.testing.assert_array_equal(out, np.array(10.0))
        assert _privates[thread_pid_id(client)].port != connected_port
        del client
    finally:
        for proc in procs.values():
            if proc.is_alive():
                proc.terminate()
        for proc in procs.values():
            proc.join(timeout=10)


@pytest.mark.timeout(120)
def test_fast_latency_beats_grpc_floor(fast_servers):
    """The point of the transport: well under gRPC's ~1-2 ms container floor.

    Scheduling noise on a shared CPU box can stall cross-process ping-pong
    for a while (observed: multi-ms per call for seconds at a time while
    raw echoes stayed at ~100 us), so assert on the BEST batch out of
    several, with one full remeasure before declaring failure."""
    client = ArraysToArraysServiceClient("127.0.0.1", FAST_PORTS[1], transport="fast")

    async def run():
        a, b = np.array(2.0), np.array(3.0)
        for _ in range(20):
            await client.evaluate_async(a, b)
        best = float("inf")
        for _ in range(5):
            t0 = time.perf_counter()
            for _ in range(60):
                await client.evaluate_async(a, b)
            best = min(best, (time.perf_counter() - t0) / 60)
        return best

    per_call = asyncio.run(run())
    if per_call >= 0.002:  # transient stall: one remeasure after a pause
        time.sleep(2.0)
        per_call = asyncio.run(run())
    assert per_call < 0.002, f"fast transport too slow: {per_call * 1e6:.0f} us/call"
    del client


@pytest.mark.timeout(120)
def test_fast_server_survives_malformed_frames(fast_servers):
    """Garbage frames must not kill the worker: bad magic drops the
    connection; unknown frame types get T_ERR; the server keeps serving."""
    import asyncio as aio

    from pytensor_federated_amd.fastsock import MAGIC, _frame

    async def run():
        # 1. wrong magic -> connection closed, server alive
        r, w = await aio.open_connection("127.0.0.1", FAST_PORTS[0])
        w.write(b"BOGUS")
        await w.drain()
        data = await r.read(64)
        assert data == b""  # closed
        w.close()

        # 2. unknown frame type -> T_ERR reply
        r, w = await aio.open_connection("127.0.0.1", FAST_PORTS[0])
        w.write(MAGIC + _frame(0x7C, b"???"))
        await w.drain()
        hdr = await r.readexactly(5)
        assert hdr[0] == 0xFF
        w.close()

        # 3. truncated protobuf payload -> T_ERR (compute error), not a crash
        r, w = await aio.open_connection("127.0.0.1", FAST_PORTS[0])
        w.write(MAGIC + _frame(0x01, b"\xff\xff\xff"))
        await w.drain()
        hdr = await r.readexactly(5)
        assert hdr[0] == 0xFF
        w.close()

    asyncio.run(run())
    # server still healthy
    client = ArraysToArraysServiceClient("127.0.0.1", FAST_PORTS[0], transport="fast")
    (out,) = client.evaluate(np.array(2.0), np.array(3.0))
    np.testing.assert_array_equal(out, np.array(6.0))
    del client


@pytest.mark.timeout(180)
def test_fast_server_fuzz_random_frames(fast_servers):
    """Robustness: hundreds of random (type, length, payload) frames must
    never kill the worker -- every reply is a well-formed frame (or the
    connection closes), and real evaluations keep working afterwards."""
    import asyncio as aio
    import random

    from pytensor_federated_amd.fastsock import MAGIC, _frame

    rng = random.Random(91)

    async def run():
        for _ in range(40):
            r, w = await aio.open_connection("127.0.0.1", FAST_PORTS[0])
            w.write(MAGIC)
            try:
                for _ in range(rng.randint(1, 8)):
                    ftype = rng.randrange(256)
                    payload = bytes(rng.randrange(256) for _ in range(rng.randint(0, 64)))
                    w.write(_frame(ftype, payload))
                    await w.drain()
                    hdr = await aio.wait_for(r.readexactly(5), timeout=10)
                    length = int.from_bytes(hdr[1:5], "little")
                    assert length < 1 << 20
                    if length:
                        await aio.wait_for(r.readexactly(length), timeout=10)
            except (aio.IncompleteReadError, ConnectionError):
                pass  # server may drop the connection; must not die
            finally:
                w.close()

    asyncio.run(run())
    client = ArraysToArraysServiceClient("127.0.0.1", FAST_PORTS[0], transport="fast")
    (out,) = client.evaluate(np.array(3.0), np.array(4.0))
    np.testing.assert_array_equal(out, np.array(12.0))
    del client
