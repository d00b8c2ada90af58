#!/usr/bin/env python3
"""Federated worker demo: serve logp+grad of a private linear-model shard.

Parity with reference demo_node.py (LinearModelBlackbox + run_node_pool):
one gRPC server process per port, each owning a private synthetic dataset.
On a GPU box the shard lives in HBM and evaluates through the fused CDNA4
HIP kernel; on CPU it evaluates eagerly.

    python demo_node.py --bind 127.0.0.1 --ports 50000 50001 --delay 0.0
"""
import argparse
import multiprocessing


def run_node(bind: str, port: int, delay: float, device: str, rows: int, seed: int,
             fast_port: int = None):
    import torch

    from pytensor_federated_amd.common import wrap_logp_grad_func
    from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
    from pytensor_federated_amd.service import serve_compute_func

    x, y = generate_linear_dataset(rows, seed=seed)
    dtype = torch.bfloat16 if device.startswith("cuda") else torch.float64
    model = GaussianLinearModel(x, y, sigma=0.4, device=device, dtype=dtype, delay=delay or None)
    print(f"Serving linear-model shard ({rows} rows, seed {seed}) on {bind}:{port} "
          f"[{device}]" + (f" + fast:{fast_port}" if fast_port else ""))
    serve_compute_func(
        wrap_logp_grad_func(model.as_logp_grad_func()), bind, port, fast_port=fast_port
    )


def run_node_pool(bind: str, ports, delay: float, device: str, rows: int,
                  fast_offset: int = 0):
    """One server process per port (reference demo_node.py:98-121).

    ``fast_offset`` > 0 additionally serves the low-latency fast transport
    on port+offset for each worker."""
    ctx = multiprocessing.get_context("spawn")
    procs = [
        ctx.Process(
            target=run_node,
            args=(bind, port, delay, device, rows, i,
                  port + fast_offset if fast_offset else None),
            daemon=False,
        )
        for i, port in enumerate(ports)
    ]
    for p in procs:
        p.start()
    try:
        for p in procs:
            p.join()
    except KeyboardInterrupt:
        for p in procs:
            p.terminate()


def run_native_pool(ports, rows: int, grpc_offset: int):
    """One NATIVE worker daemon (ops/fed_worker, C++) per port: FEDS1 on the
    port, real gRPC on port+grpc_offset.  Zero Python on the serving path --
    the native analog of the reference's per-port worker pool."""
    import struct
    import subprocess
    import tempfile
    from pathlib import Path

    import numpy as np

    from pytensor_federated_amd.models import generate_linear_dataset
    from pytensor_federated_amd.ops.build import LIB_PATH, WORKER_BIN

    if not WORKER_BIN.exists():
        raise SystemExit("fed_worker not built: run python -m pytensor_federated_amd.ops.build")
    procs, files = [], []
    try:
        for i, port in enumerate(ports):
            x, y = generate_linear_dataset(rows, seed=i)
            tmp = tempfile.NamedTemporaryFile(suffix=".bin", delete=False)
            tmp.write(struct.pack("<q", len(x)))
            tmp.write(np.asarray(x, dtype=np.float64).tobytes())
            tmp.write(np.asarray(y, dtype=np.float64).tobytes())
            tmp.close()
            files.append(Path(tmp.name))
            import os

            env = dict(os.environ, FEDOPS_LIB=str(LIB_PATH))
            cmd = [str(WORKER_BIN), "--port", str(port), "--data", tmp.name,
                   "--sigma", "0.4", "--dtype", "bf16"]
            if grpc_offset:
                cmd += ["--grpc-port", str(port + grpc_offset)]
            print(f"native worker on {port}"
                  + (f" (gRPC {port + grpc_offset})" if grpc_offset else ""))
            procs.append(subprocess.Popen(cmd, env=env))
        for p in procs:
            p.wait()
    except KeyboardInterrupt:
        pass
    finally:
        for p in procs:
            p.terminate()
        for f in files:
            f.unlink(missing_ok=True)


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--bind", default="127.0.0.1")
    parser.add_argument("--ports", type=int, nargs="+", default=list(range(50000, 50015)))
    parser.add_argument("--delay", type=float, default=0.0)
    parser.add_argument("--device", default="cpu", help='"cpu" or "cuda:0"')
    parser.add_argument("--rows", type=int, default=10)
    parser.add_argument(
        "--fast-offset", type=int, default=0,
        help="also serve the fast transport on port+offset (0 = off)",
    )
    parser.add_argument(
        "--native", action="store_true",
        help="serve via the C++ fed_worker daemons instead of Python workers "
             "(GPU box; FEDS1 on each port, gRPC on port+grpc-offset)",
    )
    parser.add_argument("--grpc-offset", type=int, default=1000,
                        help="native mode: gRPC port offset (0 = FEDS1 only)")
    args, _ = parser.parse_known_args()
    if args.native:
        run_native_pool(args.ports, max(args.rows, 10), args.grpc_offset)
    else:
        run_node_pool(args.bind, args.ports, args.delay, args.device, args.rows,
                      args.fast_offset)
