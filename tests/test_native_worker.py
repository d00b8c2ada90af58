"""Native C++ worker daemon: Python client <-> fed_worker binary (GPU)."""
import os
import struct
import subprocess
import socket
import time
from pathlib import Path

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

REPO = Path(__file__).resolve().parent.parent
WORKER = REPO / "pytensor_federated_amd" / "ops" / "fed_worker"
LIB = REPO / "pytensor_federated_amd" / "ops" / "libfedops_gfx950.so"
PORT = 9601


def _wait_tcp(port, timeout=60.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=1):
                return
        except OSError:
            time.sleep(0.1)
    raise TimeoutError(f"port {port} never opened")


@pytest.mark.timeout(300)
def test_cpp_worker_serves_kernel_evals(tmp_path):
    from pytensor_federated_amd.common import LogpGradServiceClient
    from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset

    assert WORKER.exists(), "fed_worker binary not built"
    x, y = generate_linear_dataset(100_000, seed=51)
    shard = tmp_path / "shard.bin"
    with open(shard, "wb") as f:
        f.write(struct.pack("<q", len(x)))
        f.write(np.asarray(x, dtype=np.float64).tobytes())
        f.write(np.asarray(y, dtype=np.float64).tobytes())

    env = dict(os.environ, FEDOPS_LIB=str(LIB))
    proc = subprocess.Popen(
        [str(WORKER), "--port", str(PORT), "--data", str(shard), "--sigma", "0.4",
         "--dtype", "bf16"],
        env=env,
        stderr=subprocess.PIPE,
    )
    try:
        _wait_tcp(PORT)
        # GetLoad telemetry from the C++ worker (before the client opens its
        # persistent connection: asyncio.run uses its own event loop)
        import asyncio

        from pytensor_federated_amd.service import get_load_async

        load = asyncio.run(get_load_async("127.0.0.1", PORT, transport="fast"))
        assert load is not None
        assert 0.0 <= load.percent_cpu <= 100.0
        # hipMemGetInfo granularity can round a small shard to 0% on 288 GB
        assert 0.0 <= load.percent_ram <= 100.0

        client = LogpGradServiceClient("127.0.0.1", PORT, transport="fast")
        logp, (ga, gb) = client.evaluate(1.5, 0.5)

        import torch

        ref_model = GaussianLinearModel(
            x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16, use_kernels=True
        )
        logp_ref, (ga_ref, gb_ref) = ref_model(1.5, 0.5)
        np.testing.assert_allclose(float(logp), float(logp_ref), rtol=1e-9)
        np.testing.assert_allclose(float(ga), float(ga_ref), rtol=1e-7)
        np.testing.assert_allclose(float(gb), float(gb_ref), rtol=1e-7)

        # latency: native worker round trip over loopback
        for _ in range(20):
            client.evaluate(1.5, 0.5)
        t0 = time.perf_counter()
        n = 200
        for _ in range(n):
            client.evaluate(1.5, 0.5)
        per_call = (time.perf_counter() - t0) / n
        print(f"native worker: {per_call * 1e6:.0f} us/call")
        assert per_call < 0.01
        del client
    finally:
        proc.terminate()
        proc.wait(timeout=10)


@pytest.mark.timeout(300)
def test_cpp_worker_serves_ode(tmp_path):
    from pytensor_federated_amd.common import LogpGradServiceClient
    from pytensor_federated_amd.models import (
        ODEModel,
        generate_ode_dataset,
        lotka_volterra_rhs,
    )

    import torch

    u0, obs_idx, y = generate_ode_dataset(
        n_experiments=64, n_obs=20, n_steps=100, sigma=0.1, seed=53
    )
    shard = tmp_path / "ode.bin"
    with open(shard, "wb") as f:
        f.write(struct.pack("<qqqdd", u0.shape[0], 100, len(obs_idx), 10.0 / 100, 0.1))
        f.write(np.asarray(u0, dtype=np.float64).tobytes())
        f.write(np.asarray(obs_idx, dtype=np.int64).tobytes())
        f.write(np.asarray(y, dtype=np.float64).tobytes())

    env = dict(os.environ, FEDOPS_LIB=str(LIB))
    proc = subprocess.Popen(
        [str(WORKER), "--port", str(PORT + 4), "--data", str(shard), "--model", "ode"],
        env=env, stderr=subprocess.PIPE,
    )
    try:
        _wait_tcp(PORT + 4)
        client = LogpGradServiceClient("127.0.0.1", PORT + 4, transport="fast")
        theta = np.array([0.8, 0.3, 0.6, 0.2])
        logp, (grad,) = client.evaluate(theta)
        ref_model = ODEModel(
            lotka_volterra_rhs, u0, 0.0, 10.0, 100, obs_idx, y, sigma=0.1,
            device="cuda:0", use_kernels=True,
        )
        logp_ref, (g_ref,) = ref_model.logp_grad(torch.as_tensor(theta))
        np.testing.assert_allclose(float(logp), float(logp_ref), rtol=1e-10)
        np.testing.assert_allclose(grad, g_ref.cpu().numpy(), rtol=1e-9)
        # second theta exercises the theta re-upload path
        theta2 = np.array([0.9, 0.25, 0.55, 0.22])
        logp2, (grad2,) = client.evaluate(theta2)
        logp2_ref, (g2_ref,) = ref_model.logp_grad(torch.as_tensor(theta2))
        np.testing.assert_allclose(float(logp2), float(logp2_ref), rtol=1e-10)
        np.testing.assert_allclose(grad2, g2_ref.cpu().numpy(), rtol=1e-9)
        del client
    finally:
        proc.terminate()
        proc.wait(timeout=10)


@pytest.mark.timeout(300)
def test_cpp_worker_serves_logistic(tmp_path):
    from pytensor_federated_amd.common import LogpGradServiceClient
    from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset

    import torch

    X, y, beta0 = generate_logistic_dataset(20_000, 512, seed=52)
    shard = tmp_path / "logit.bin"
    with open(shard, "wb") as f:
        f.write(struct.pack("<qq", X.shape[0], X.shape[1]))
        f.write(np.asarray(X, dtype=np.float64).tobytes())
        f.write(np.asarray(y, dtype=np.float64).tobytes())

    env = dict(os.environ, FEDOPS_LIB=str(LIB))
    proc = subprocess.Popen(
        [str(WORKER), "--port", str(PORT + 2), "--data", str(shard),
         "--model", "logistic", "--dtype", "bf16"],
        env=env, stderr=subprocess.PIPE,
    )
    try:
        _wait_tcp(PORT + 2)
        client = LogpGradServiceClient("127.0.0.1", PORT + 2, transport="fast")
        logp, (grad,) = client.evaluate(np.asarray(beta0, dtype=np.float64))
        ref_model = LogisticGLMModel(
            X, y, device="cuda:0", dtype=torch.bfloat16, use_kernels=True
        )
        # the worker quantizes beta to f32 inside the kernel wrapper; so does ours
        logp_ref, (g_ref,) = ref_model.logp_grad(
            torch.as_tensor(beta0, dtype=torch.float32)
        )
        np.testing.assert_allclose(float(logp), float(logp_ref), rtol=1e-6)
        np.testing.assert_allclose(
            grad, g_ref.cpu().numpy(), rtol=1e-5, atol=1e-4 * float(abs(g_ref).max())
        )
        del client
    finally:
        proc.terminate()
        proc.wait(timeout=10)


@pytest.mark.timeout(300)
def test_cpp_worker_serves_grpc(tmp_path):
    """A reference-style gRPC client evaluates against the NATIVE worker --
    no Python sidecar.  Covers all three RPC routes of service.proto over
    the worker's libnghttp2 HTTP/2 edge: EvaluateStream (the hot path),
    unary Evaluate, and GetLoad (reference service.py:75-115, README.md:35)."""
    from pytensor_federated_amd.common import LogpGradServiceClient
    from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    assert WORKER.exists(), "fed_worker binary not built"
    x, y = generate_linear_dataset(100_000, seed=54)
    shard = tmp_path / "shard.bin"
    with open(shard, "wb") as f:
        f.write(struct.pack("<q", len(x)))
        f.write(np.asarray(x, dtype=np.float64).tobytes())
        f.write(np.asarray(y, dtype=np.float64).tobytes())

    env = dict(os.environ, FEDOPS_LIB=str(LIB))
    grpc_port = PORT + 6
    proc = subprocess.Popen(
        [str(WORKER), "--port", str(PORT + 7), "--grpc-port", str(grpc_port),
         "--data", str(shard), "--sigma", "0.4", "--dtype", "bf16"],
        env=env,
        stderr=subprocess.PIPE,
    )
    try:
        _wait_tcp(grpc_port)

        import asyncio

        from pytensor_federated_amd.service import get_load_async

        # GetLoad over real gRPC
        load = asyncio.run(get_load_async("127.0.0.1", grpc_port, transport="grpc"))
        assert load is not None
        assert 0.0 <= load.percent_cpu <= 100.0
        assert 0.0 <= load.percent_ram <= 100.0

        import torch

        ref_model = GaussianLinearModel(
            x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16, use_kernels=True
        )
        logp_ref, (ga_ref, gb_ref) = ref_model(1.5, 0.5)

        # hot path: persistent bidirectional EvaluateStream
        client = LogpGradServiceClient("127.0.0.1", grpc_port, transport="grpc")
        logp, (ga, gb) = client.evaluate(1.5, 0.5)
        np.testing.assert_allclose(float(logp), float(logp_ref), rtol=1e-9)
        np.testing.assert_allclose(float(ga), float(ga_ref), rtol=1e-7)
        np.testing.assert_allclose(float(gb), float(gb_ref), rtol=1e-7)
        # several messages on ONE stream (the reference's per-eval roundtrip)
        for i in range(10):
            logp_i, _ = client.evaluate(1.5 + 0.01 * i, 0.5)
            assert np.isfinite(float(logp_i))
        del client

        # unary Evaluate on a fresh channel
        unary = ArraysToArraysServiceClient(
            "127.0.0.1", grpc_port, use_stream=False, transport="grpc"
        )
        outs = unary.evaluate(np.float64(1.5), np.float64(0.5))
        assert len(outs) == 3
        np.testing.assert_allclose(float(outs[0]), float(logp_ref), rtol=1e-9)
        del unary
    finally:
        proc.terminate()
        proc.wait(timeout=10)


@pytest.mark.timeout(300)
def test_cpp_worker_concurrent_clients(tmp_path):
    """Concurrent clients evaluate DIFFERENT thetas against one worker; each
    must get its own correct result (per-connection EvalCtx: private stream
    + buffers -- regression for the round-1 global-eval-mutex design)."""
    import threading

    from pytensor_federated_amd.common import LogpGradServiceClient
    from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset

    x, y = generate_linear_dataset(500_000, seed=55)
    shard = tmp_path / "shard.bin"
    with open(shard, "wb") as f:
        f.write(struct.pack("<q", len(x)))
        f.write(np.asarray(x, dtype=np.float64).tobytes())
        f.write(np.asarray(y, dtype=np.float64).tobytes())

    env = dict(os.environ, FEDOPS_LIB=str(LIB))
    port = PORT + 9
    proc = subprocess.Popen(
        [str(WORKER), "--port", str(port), "--data", str(shard), "--sigma", "0.4",
         "--dtype", "bf16"],
        env=env, stderr=subprocess.PIPE,
    )
    try:
        _wait_tcp(port)
        import torch

        ref_model = GaussianLinearModel(
            x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16, use_kernels=True
        )
        thetas = [(1.5 + 0.1 * i, 0.5 - 0.02 * i) for i in range(4)]
        refs = [ref_model(a, b) for a, b in thetas]
        errors = []

        def run_client(idx):
            try:
                client = LogpGradServiceClient("127.0.0.1", port, transport="fast")
                a, b = thetas[idx]
                for _ in range(50):
                    logp, (ga, gb) = client.evaluate(a, b)
                    logp_ref, (ga_ref, gb_ref) = refs[idx]
                    np.testing.assert_allclose(float(logp), float(logp_ref), rtol=1e-9)
                    np.testing.assert_allclose(float(ga), float(ga_ref), rtol=1e-7)
                del client
            except Exception as ex:  # surface into the main thread
                errors.append((idx, repr(ex)))

        threads = [threading.Thread(target=run_client, args=(i,)) for i in range(4)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=120)
        assert not errors, errors
    finally:
        proc.terminate()
        proc.wait(timeout=10)
