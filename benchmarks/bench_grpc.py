#!/usr/bin/env python3
"""BASELINE config 1: logp+grad calls/sec over loopback gRPC on CPU.

The reference's demo setup (demo_node.py linear model, N=10 rows,
bidirectional stream).  Establishes the comparison floor the reference
itself would set on this box (the reference cannot run here --
pytensor/grpclib are not installed -- so this measures OUR transport stack
on the same workload shape).

    python benchmarks/bench_grpc.py --calls 2000
"""
import argparse
import json
import multiprocessing
import socket
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def _serve(port: int, rows: int):
    from pytensor_federated_amd.common import wrap_logp_grad_func
    from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
    from pytensor_federated_amd.service import serve_compute_func

    x, y = generate_linear_dataset(rows, seed=0)
    model = GaussianLinearModel(x, y, sigma=0.4)
    serve_compute_func(
        wrap_logp_grad_func(model.as_logp_grad_func()), "127.0.0.1", port,
        fast_port=port + 1,
    )


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--calls", type=int, default=2000)
    parser.add_argument("--warmup", type=int, default=100)
    parser.add_argument("--rows", type=int, default=10)
    parser.add_argument("--port", type=int, default=9651)
    parser.add_argument("--transport", choices=["grpc", "fast"], default="grpc")
    parser.add_argument("--unary", action="store_true")
    args = parser.parse_args()

    ctx = multiprocessing.get_context("spawn")
    proc = ctx.Process(target=_serve, args=(args.port, args.rows), daemon=True)
    proc.start()
    deadline = time.time() + 30
    while time.time() < deadline:
        try:
            with socket.create_connection(("127.0.0.1", args.port), timeout=1):
                break
        except OSError:
            time.sleep(0.1)

    try:
        from pytensor_federated_amd.common import LogpGradServiceClient

        if args.transport == "fast":
            client = LogpGradServiceClient("127.0.0.1", args.port + 1, transport="fast")
        else:
            client = LogpGradServiceClient("127.0.0.1", args.port)
        use_stream = not args.unary
        for _ in range(args.warmup):
            client.evaluate(1.5, 0.5, use_stream=use_stream)
        t0 = time.perf_counter()
        for i in range(args.calls):
            client.evaluate(1.5 + 1e-4 * i, 0.5, use_stream=use_stream)
        elapsed = time.perf_counter() - t0
        print(
            json.dumps(
                {
                    "metric": "logp+grad calls/sec (loopback gRPC, CPU)",
                    "value": args.calls / elapsed,
                    "unit": "calls/s",
                    "ms_per_call": elapsed / args.calls * 1000,
                    "transport": args.transport + (" unary" if args.unary else " stream"),
                    "rows": args.rows,
                    "config": "BASELINE config 1",
                }
            )
        )
    finally:
        proc.terminate()
        proc.join(timeout=10)


if __name__ == "__main__":
    main()
