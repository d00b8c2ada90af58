#!/usr/bin/env python3
"""Native C++ worker: per-call latency over its TWO protocol edges.

The worker (ops/csrc/fed_worker.cpp) serves the same shard over the FEDS1
fast transport and, since round 2, real gRPC (HTTP/2 + HPACK via
libnghttp2).  This measures the per-evaluation round trip of each edge
with the standard Python clients -- the gRPC number is what a
reference-style (grpclib/grpcio) off-node client sees with no Python
sidecar in front of the native worker.

    python benchmarks/bench_worker_grpc.py --calls 1000        (GPU box)
    python benchmarks/bench_worker_grpc.py --echo --calls 2000 (any box)
"""
import argparse
import json
import os
import socket
import struct
import subprocess
import sys
import tempfile
import time
from pathlib import Path

import numpy as np

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))

WORKER = REPO / "pytensor_federated_amd" / "ops" / "fed_worker"
LIB = REPO / "pytensor_federated_amd" / "ops" / "libfedops_gfx950.so"


def _wait_tcp(port, timeout=60.0):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", port), timeout=1):
                return
        except OSError:
            time.sleep(0.1)
    raise TimeoutError(f"port {port} never opened")


def _client_proc(cport, kwargs, n, ready, go):
    """Top-level (picklable) client driver for the multi-process row.

    Imports + connects, signals ready, then waits for the collective start
    so process startup stays outside the timed window."""
    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    client = ArraysToArraysServiceClient("127.0.0.1", cport, **kwargs)
    a, b = np.float64(1.5), np.float64(0.5)
    client.evaluate(a, b)  # connect + warm
    ready.release()
    go.acquire()
    for _ in range(n):
        client.evaluate(a, b)
    del client


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--calls", type=int, default=1000)
    parser.add_argument("--warmup", type=int, default=100)
    parser.add_argument("--rows", type=int, default=100_000)
    parser.add_argument("--port", type=int, default=9651)
    parser.add_argument("--echo", action="store_true",
                        help="GPU-less transport-only mode (--model echo)")
    parser.add_argument("--clients", type=int, default=1,
                        help="concurrent clients (per edge) for the throughput row")
    args = parser.parse_args()

    from pytensor_federated_amd.service import ArraysToArraysServiceClient

    port, gport = args.port, args.port + 1
    cmd = [str(WORKER), "--port", str(port), "--grpc-port", str(gport)]
    tmp = None
    if args.echo:
        cmd += ["--model", "echo"]
    else:
        from pytensor_federated_amd.models import generate_linear_dataset

        x, y = generate_linear_dataset(args.rows, seed=60)
        tmp = tempfile.NamedTemporaryFile(suffix=".bin", delete=False)
        tmp.write(struct.pack("<q", len(x)))
        tmp.write(np.asarray(x, dtype=np.float64).tobytes())
        tmp.write(np.asarray(y, dtype=np.float64).tobytes())
        tmp.close()
        cmd += ["--data", tmp.name, "--sigma", "0.4", "--dtype", "bf16"]

    env = dict(os.environ, FEDOPS_LIB=str(LIB))
    proc = subprocess.Popen(cmd, env=env, stderr=subprocess.DEVNULL)
    try:
        _wait_tcp(gport)
        result = {}

        def one_client(cport, kwargs, n):
            client = ArraysToArraysServiceClient("127.0.0.1", cport, **kwargs)
            a, b = np.float64(1.5), np.float64(0.5)
            for _ in range(n):
                client.evaluate(a, b)
            del client

        if args.clients > 1:
            # concurrent-client throughput: one client PROCESS each (GIL-free
            # drivers); per-connection EvalCtx on the worker lets their
            # evaluations overlap on private HIP streams
            import multiprocessing

            ctx = multiprocessing.get_context("spawn")
            for name, kwargs in [("grpc_stream", dict(transport="grpc")),
                                 ("fast", dict(transport="fast"))]:
                cport = gport if name.startswith("grpc") else port
                one_client(cport, kwargs, args.warmup)  # server warm
                ready = ctx.Semaphore(0)
                go = ctx.Semaphore(0)
                procs = [
                    ctx.Process(target=_client_proc,
                                args=(cport, kwargs, args.calls, ready, go))
                    for _ in range(args.clients)
                ]
                for p in procs:
                    p.start()
                for _ in procs:
                    ready.acquire()  # all imported + connected
                t0 = time.perf_counter()
                for _ in procs:
                    go.release()
                for p in procs:
                    p.join()
                wall = time.perf_counter() - t0
                result[f"{name}_x{args.clients}"] = {
                    "aggregate_calls_per_s": args.calls * args.clients / wall,
                    "us_per_call_per_client": wall / args.calls * 1e6,
                }
        for name, kwargs in [
            ("grpc_stream", dict(transport="grpc", use_stream=True)),
            ("grpc_unary", dict(transport="grpc", use_stream=False)),
            ("fast", dict(transport="fast")),
        ]:
            cport = gport if name.startswith("grpc") else port
            client = ArraysToArraysServiceClient("127.0.0.1", cport, **kwargs)
            a, b = np.float64(1.5), np.float64(0.5)
            for _ in range(args.warmup):
                client.evaluate(a, b)
            t0 = time.perf_counter()
            for _ in range(args.calls):
                client.evaluate(a, b)
            per = (time.perf_counter() - t0) / args.calls
            result[name] = {"us_per_call": per * 1e6, "calls_per_s": 1.0 / per}
            del client
        print(json.dumps({
            "metric": "native-worker eval round trip by protocol edge",
            "config": {
                "mode": "echo" if args.echo else "gaussian_linear bf16 shard",
                "rows": None if args.echo else args.rows,
                "calls": args.calls,
            },
            **result,
        }))
    finally:
        proc.terminate()
        proc.wait(timeout=10)
        if tmp is not None:
            os.unlink(tmp.name)


if __name__ == "__main__":
    main()
