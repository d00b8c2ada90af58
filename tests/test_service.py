"""Service/transport tests: unit (no network) + multi-process integration.

Mirrors the reference's strategy (test_service.py): real gRPC servers as
localhost subprocesses, ``terminate()`` as the fault injector, pickling the
client into pool workers.

Fork note: grpc.aio starts a process-global poller thread on first use, so
``fork()`` AFTER the parent has evaluated is not supported (use spawn /
forkserver there).  The supported fork pattern -- parent builds the client,
workers fork BEFORE the first evaluation and connect lazily via the
pid-keyed connection cache (exactly PyMC's ``pm.sample(cores=N)`` flow) --
is covered hermetically in ``test_fork_pool_before_first_use``.
"""
import asyncio
import multiprocessing
import socket
import subprocess
import sys
import time
from unittest import mock

import numpy as np
import pytest

import pytensor_federated_amd.service as service_mod
from pytensor_federated_amd.npproto.utils import ndarray_from_numpy, ndarray_to_numpy
from pytensor_federated_amd.rpc import InputArrays
from pytensor_federated_amd.service import (
    ArraysToArraysService,
    ArraysToArraysServiceClient,
    _privates,
    _run_compute_func,
    get_loads_async,
    thread_pid_id,
)

TEST_PORTS = (9499, 9500, 9501)
DEAD_PORT = 9502
FAILOVER_PORTS = (9521, 9522)


def product_func(a, b):
    return [a * b]


# ---------------------------------------------------------------------------
# unit: no network
# ---------------------------------------------------------------------------


class TestRunComputeFunc:
    def test_message_in_message_out(self):
        a, b = np.array([1.0, 2.0, 3.0]), np.array([2.0, 0.5, 1.0])
        inp = InputArrays(items=[ndarray_from_numpy(a), ndarray_from_numpy(b)], uuid="id-1")
        out = _run_compute_func(inp, product_func)
        assert out.uuid == "id-1"
        np.testing.assert_array_equal(ndarray_to_numpy(out.items[0]), a * b)

    def test_multiple_outputs(self):
        inp = InputArrays(items=[ndarray_from_numpy(np.array(2.0))], uuid="x")
        out = _run_compute_func(inp, lambda v: [v + 1, v * 3])
        assert len(out.items) == 2


class TestDetermineLoad:
    def test_load_math(self):
        svc = ArraysToArraysService(product_func)
        with mock.patch("psutil.getloadavg", return_value=(2.0, 0.0, 0.0)), mock.patch(
            "psutil.cpu_count", return_value=8
        ):
            load = svc.determine_load()
        assert load.percent_cpu == pytest.approx(25.0)
        assert load.n_clients == 0

    def test_stream_counts_clients(self):
        svc = ArraysToArraysService(product_func)

        async def run():
            async def one_request():
                yield InputArrays(
                    items=[ndarray_from_numpy(np.array(2.0)), ndarray_from_numpy(np.array(3.0))],
                    uuid="u",
                )

            agen = svc.evaluate_stream(one_request())
            out = await agen.__anext__()
            assert svc.n_clients == 1
            assert out.uuid == "u"
            with pytest.raises(StopAsyncIteration):
                await agen.__anext__()
            assert svc.n_clients == 0

        asyncio.run(run())


# ---------------------------------------------------------------------------
# integration: real gRPC servers in subprocesses
# ---------------------------------------------------------------------------


def _serve(port: int, delay: float = 0.0):
    """Subprocess entry: serve product_func on a port (top-level, picklable)."""
    import time as _time

    from pytensor_federated_amd.service import serve_compute_func

    def fn(a, b):
        if delay:
            _time.sleep(delay)
        return [a * b]

    serve_compute_func(fn, "127.0.0.1", port)


def _wait_for_server(port: int, timeout: float = 30.0):
    """TCP-connect readiness probe (deliberately grpc-free)."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=1):
                return
        except OSError:
            time.sleep(0.1)
    raise TimeoutError(f"Server on port {port} did not come up.")


def _spawn_servers(ports):
    ctx = multiprocessing.get_context("spawn")
    procs = {p: ctx.Process(target=_serve, args=(p,), daemon=True) for p in ports}
    for proc in procs.values():
        proc.start()
    for port in ports:
        _wait_for_server(port)
    return procs


def _kill_servers(procs):
    for proc in procs.values():
        if proc.is_alive():
            proc.terminate()
    for proc in procs.values():
        proc.join(timeout=10)


@pytest.fixture()
def fast_desync(monkeypatch):
    monkeypatch.setattr(service_mod, "_BALANCE_DESYNC_RANGE", (0.0, 0.01))


@pytest.fixture(scope="module")
def servers():
    procs = _spawn_servers(TEST_PORTS)
    try:
        yield procs
    finally:
        _kill_servers(procs)


@pytest.mark.timeout(120)
def test_evaluate_stream_and_unary(servers):
    client = ArraysToArraysServiceClient("127.0.0.1", TEST_PORTS[0])
    a, b = np.array([1.0, 4.0]), np.array([2.0, 3.0])
    (out,) = client.evaluate(a, b)
    np.testing.assert_array_equal(out, a * b)
    (out,) = client.evaluate(a, b, use_stream=False)
    np.testing.assert_array_equal(out, a * b)
    # repeated calls reuse the persistent stream
    for _ in range(5):
        (out,) = client.evaluate(a, b)
    np.testing.assert_array_equal(out, a * b)
    del client


@pytest.mark.timeout(120)
def test_get_loads_ignores_dead_server(servers):
    loads = asyncio.run(
        get_loads_async([("127.0.0.1", p) for p in (*TEST_PORTS, DEAD_PORT)], timeout=3)
    )
    assert all(load is not None for load in loads[:3])
    assert loads[3] is None


@pytest.mark.timeout(180)
def test_balanced_connect_picks_idle_server(servers, fast_desync):
    hap = [("127.0.0.1", p) for p in TEST_PORTS]
    a, b = np.array(2.0), np.array(3.0)
    clients = []
    used_ports = []
    # each successive client must avoid servers already holding open streams
    for _ in range(3):
        c = ArraysToArraysServiceClient(hosts_and_ports=hap)
        c.evaluate(a, b)
        clients.append(c)
        used_ports.append(_privates[thread_pid_id(c)].port)
    assert sorted(used_ports) == sorted(TEST_PORTS), used_ports
    del clients


@pytest.mark.timeout(180)
def test_client_pickles_into_spawn_pool(servers):
    client = ArraysToArraysServiceClient("127.0.0.1", TEST_PORTS[1])
    ctx = multiprocessing.get_context("spawn")
    with ctx.Pool(2) as pool:
        results = pool.map(_eval_with_client, [(client, i) for i in range(4)])
        bound = pool.map(_BoundEval(client), list(range(4)))
    for i, r in zip(range(4), results):
        np.testing.assert_array_equal(r, np.array(2.0 * i))
    for i, r in zip(range(4), bound):
        np.testing.assert_array_equal(r, np.array(2.0 * i))
    del client


def _eval_with_client(args):
    client, i = args
    (out,) = client.evaluate(np.array(2.0), np.array(float(i)))
    return out


class _BoundEval:
    def __init__(self, client):
        self.client = client

    def __call__(self, i):
        (out,) = self.client.evaluate(np.array(2.0), np.array(float(i)))
        return out


_FORK_SCRIPT = r"""
import multiprocessing, sys
import numpy as np
from pytensor_federated_amd.service import ArraysToArraysServiceClient

def eval_one(args):
    client, i = args
    (out,) = client.evaluate(np.array(2.0), np.array(float(i)))
    return out

if __name__ == "__main__":
    port = int(sys.argv[1])
    # parent builds the client but never evaluates -> grpc untouched pre-fork
    client = ArraysToArraysServiceClient("127.0.0.1", port)
    ctx = multiprocessing.get_context("fork")
    with ctx.Pool(2) as pool:
        results = pool.map(eval_one, [(client, i) for i in range(4)])
    for i, r in zip(range(4), results):
        np.testing.assert_array_equal(r, np.array(2.0 * i))
    print("FORK-POOL-OK")
"""


@pytest.mark.timeout(180)
def test_fork_pool_before_first_use(servers, tmp_path):
    """The PyMC ``cores=N`` pattern: client pickled into fork workers that
    connect lazily (parent has not used gRPC before the fork)."""
    script = tmp_path / "fork_pool.py"
    script.write_text(_FORK_SCRIPT)
    import os
    from pathlib import Path

    repo_root = str(Path(__file__).resolve().parent.parent)
    env = dict(os.environ, PYTHONPATH=repo_root + os.pathsep + os.environ.get("PYTHONPATH", ""))
    proc = subprocess.run(
        [sys.executable, str(script), str(TEST_PORTS[2])],
        capture_output=True,
        text=True,
        timeout=150,
        cwd=repo_root,
        env=env,
    )
    assert proc.returncode == 0, proc.stderr
    assert "FORK-POOL-OK" in proc.stdout


@pytest.mark.timeout(240)
def test_failover_then_total_outage(fast_desync):
    procs = _spawn_servers(FAILOVER_PORTS)
    try:
        hap = [("127.0.0.1", p) for p in FAILOVER_PORTS]
        client = ArraysToArraysServiceClient(hosts_and_ports=hap, retries=2)
        a, b = np.array(2.0), np.array(5.0)
        (out,) = client.evaluate(a, b)
        np.testing.assert_array_equal(out, np.array(10.0))
        # kill the connected server -> next evaluate must fail over
        connected_port = _privates[thread_pid_id(client)].port
        procs[connected_port].terminate()
        procs[connected_port].join()
        (out,) = client.evaluate(a, b)
        np.testing.assert_array_equal(out, np.array(10.0))
        surviving_port = _privates[thread_pid_id(client)].port
        assert surviving_port != connected_port
        # kill the survivor too -> total outage
        procs[surviving_port].terminate()
        procs[surviving_port].join()
        with pytest.raises(TimeoutError, match="servers responded"):
            client.evaluate(a, b)
        del client
    finally:
        _kill_servers(procs)
