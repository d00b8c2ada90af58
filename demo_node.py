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
    args, _ = parser.parse_known_args()
    run_node_pool(args.bind, args.ports, args.delay, args.device, args.rows,
                  args.fast_offset)
