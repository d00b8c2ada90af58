"""Native worker gRPC edge, CPU-testable (echo mode, no kernels).

The fed_worker binary's HTTP/2+HPACK face (libnghttp2) must interoperate
with stock grpcio clients.  ``--model echo`` serves [sum of scalar inputs]
without touching the GPU, so the whole transport stack -- connection
preface, SETTINGS, HEADERS routing, DATA framing, stream replies, trailers,
GetLoad -- runs in the plain CPU suite.  The GPU suite's
``test_cpp_worker_serves_grpc`` covers the same edge over real kernels.
"""
import asyncio
import os
import socket
import subprocess
import time
from pathlib import Path

import numpy as np
import pytest

REPO = Path(__file__).resolve().parent.parent
WORKER = REPO / "pytensor_federated_amd" / "ops" / "fed_worker"
PORT = 9641


def _wait_tcp(port, timeout=60.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=1):
                return
        except OSError:
            time.sleep(0.1)
    raise TimeoutError(f"port {port} never opened")


@pytest.fixture(scope="module")
def echo_worker():
    if not WORKER.exists():
        pytest.skip("fed_worker binary not built")
    proc = subprocess.Popen(
        [str(WORKER), "--port", str(PORT), "--grpc-port", str(PORT + 1),
         "--model", "echo"],
        stderr=subprocess.PIPE,
    )
    try:
        _wait_tcp(PORT + 1)
        yield proc
    finally:
        proc.terminate()
        proc.wait(timeout=10)


@pytest.mark.timeout(120)
def test_grpc_stream_evaluate(echo_worker):
    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    client = ArraysToArraysServiceClient("127.0.0.1", PORT + 1, transport="grpc")
    outs = client.evaluate(np.float64(2.0), np.float64(3.5))
    assert len(outs) == 1
    np.testing.assert_allclose(float(outs[0]), 5.5)
    # many messages on one persistent stream
    for i in range(20):
        outs = client.evaluate(np.float64(i), np.float64(1.0))
        np.testing.assert_allclose(float(outs[0]), i + 1.0)
    del client


@pytest.mark.timeout(120)
def test_grpc_unary_evaluate(echo_worker):
    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    client = ArraysToArraysServiceClient(
        "127.0.0.1", PORT + 1, use_stream=False, transport="grpc"
    )
    outs = client.evaluate(np.float64(4.0), np.float64(0.25))
    np.testing.assert_allclose(float(outs[0]), 4.25)
    del client


@pytest.mark.timeout(120)
def test_grpc_get_load(echo_worker):
    from pytensor_federated_amd.service import get_load_async

    load = asyncio.run(get_load_async("127.0.0.1", PORT + 1, transport="grpc"))
    assert load is not None
    assert load.n_clients >= 0
    assert 0.0 <= load.percent_cpu <= 100.0


@pytest.mark.timeout(120)
def test_grpc_unknown_route_is_clean_error(echo_worker):
    import grpc

    with grpc.insecure_channel(f"127.0.0.1:{PORT + 1}") as channel:
        call = channel.unary_unary(
            "/ArraysToArraysService/DoesNotExist",
            request_serializer=lambda m: m,
            response_deserializer=lambda b: b,
        )
        with pytest.raises(grpc.RpcError) as exc:
            call(b"")
        assert exc.value.code() == grpc.StatusCode.UNIMPLEMENTED


@pytest.mark.timeout(120)
def test_fast_transport_echo_still_serves(echo_worker):
    """echo mode serves both edges; FEDS1 stays the low-latency path."""
    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    client = ArraysToArraysServiceClient("127.0.0.1", PORT, transport="fast")
    outs = client.evaluate(np.float64(7.0), np.float64(-2.0))
    np.testing.assert_allclose(float(outs[0]), 5.0)
    del client


@pytest.mark.timeout(120)
def test_grpc_malformed_payload_is_clean_error(echo_worker):
    """Garbage bytes in the gRPC message must come back as a clean INTERNAL
    status (the worker's proto parser rejects them), not a crash or hang."""
    import grpc

    with grpc.insecure_channel(f"127.0.0.1:{PORT + 1}") as channel:
        call = channel.unary_unary(
            "/ArraysToArraysService/Evaluate",
            request_serializer=lambda m: m,
            response_deserializer=lambda b: b,
        )
        with pytest.raises(grpc.RpcError) as exc:
            call(b"\xff\xfe\xfd garbage that is not a protobuf \x00\x01")
        assert exc.value.code() in (
            grpc.StatusCode.INTERNAL,
            grpc.StatusCode.INVALID_ARGUMENT,
        )
    # the worker must still serve correct requests afterwards
    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    client = ArraysToArraysServiceClient("127.0.0.1", PORT + 1, transport="grpc")
    outs = client.evaluate(np.float64(1.0), np.float64(2.0))
    np.testing.assert_allclose(float(outs[0]), 3.0)
    del client
