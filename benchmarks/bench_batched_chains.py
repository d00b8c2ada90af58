#!/usr/bin/env python3
"""Multi-chain MCMC throughput: MFMA-batched 16-chain logistic evaluation.

Capability benchmark beyond the reference (whose multi-chain axis is
process-parallel PyMC chains): chain-evaluations/sec with all 16 chains
sharing one pass over X.

    python benchmarks/bench_batched_chains.py --rows 2000000
"""
import argparse
import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch

from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--rows", type=int, default=2_000_000)
    parser.add_argument("--features", type=int, default=1024)
    parser.add_argument("--steps", type=int, default=50)
    parser.add_argument("--warmup", type=int, default=10)
    args = parser.parse_args()
    assert torch.cuda.is_available(), "needs a ROCm GPU"

    X, y, _ = generate_logistic_dataset(args.rows, args.features, seed=70)
    m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
    theta1 = torch.randn(args.features, device="cuda:0") * 0.3
    theta16 = torch.randn(args.features, 16, device="cuda:0") * 0.3

    def timeit(fn, n):
        fn()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n

    t_single = timeit(lambda: m.logp_grad(theta1), args.steps)
    t_batch = timeit(lambda: m.logp_grad_batched(theta16), args.steps)
    print(json.dumps({
        "metric": "chain-evaluations/sec (16 lockstep chains, 1 GPU)",
        "value": 16.0 / t_batch,
        "single_chain_evals_per_s": 1.0 / t_single,
        "per_chain_speedup": t_single * 16 / t_batch,
        "ms_per_batched_step": t_batch * 1000,
        "config": {"rows": args.rows, "features": args.features, "dtype": "bf16",
                   "kernel": "k_logistic_glm_batched (MFMA)"},
    }))


if __name__ == "__main__":
    main()
