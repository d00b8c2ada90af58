#!/usr/bin/env python3
"""Multi-chain NUTS throughput: lockstep chains over ONE batched kernel.

C NUTS chains advance together; every leapfrog round is a single batched
adjoint-kernel sweep (``ODEModel.logp_grad_batched``), so whole-posterior
draws/sec scales with C at nearly constant kernel cost.  Compare against
``--chains 1`` (same driver, no amortization) for the lockstep speedup.

    python benchmarks/bench_nuts_batched.py --chains 16 --draws 200
"""
import argparse
import json
import sys
import time
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--chains", type=int, default=16)
    parser.add_argument("--experiments", type=int, default=256)
    parser.add_argument("--draws", type=int, default=200)
    parser.add_argument("--tune", type=int, default=400)
    parser.add_argument("--mass", default="diag", choices=["diag", "dense"])
    parser.add_argument("--adaptation", default="windowed", choices=["simple", "windowed"])
    args = parser.parse_args()

    import torch

    from pytensor_federated_amd.inference import sample_nuts_batched, split_rhat
    from pytensor_federated_amd.models import ODEModel
    from pytensor_federated_amd.models.ode import (
        generate_ode_dataset,
        lotka_volterra_rhs,
    )

    assert torch.cuda.is_available()
    theta_true = np.array([0.8, 0.3, 0.6, 0.2])
    u0, obs_idx, y_obs = generate_ode_dataset(
        n_experiments=args.experiments, n_obs=15, n_steps=40, t1=6.0, sigma=0.05
    )
    m = ODEModel(
        lotka_volterra_rhs, u0, 0.0, 6.0, 40, obs_idx, y_obs, 0.05, device="cuda:0"
    )

    def batched(theta):
        logp, G = m.logp_grad_batched(theta)
        return logp.cpu().numpy(), G.cpu().numpy()

    C = args.chains
    init = np.tile(theta_true[:, None], (1, C)) * (
        1 + 0.005 * np.random.RandomState(86).standard_normal((4, C))
    )
    # warm the kernel/workspaces before timing
    batched(init)
    t0 = time.perf_counter()
    chain, stats = sample_nuts_batched(
        batched, init, draws=args.draws, tune=args.tune, step_size=5e-4,
        seed=87, max_depth=8, mass=args.mass, adaptation=args.adaptation,
    )
    wall = time.perf_counter() - t0
    post_mean = chain.mean(axis=(0, 2))
    print(json.dumps({
        "metric": "NUTS posterior draws/sec (all chains, 1 GPU shard)",
        "value": args.draws * C / wall,
        "draws_per_sec_per_chain": args.draws / wall,
        "rounds_per_sec": stats["rounds"] / wall,
        "rounds": stats["rounds"],
        "leapfrogs": stats["leapfrogs"],
        "amortization": stats["leapfrogs"] / max(stats["rounds"], 1),
        "accept_stat_range": [float(min(stats["accept_stat"])),
                              float(max(stats["accept_stat"]))],
        "chain_rel_sd": [float(v) for v in
                         (chain.std(axis=0).mean(axis=1) / np.abs(post_mean))],
        "between_rel_sd": [float(v) for v in
                           (chain.mean(axis=0).std(axis=1) / np.abs(post_mean))],
        "step_size_range": [float(min(stats["step_sizes"])),
                            float(max(stats["step_sizes"]))],
        "posterior_mean": [float(v) for v in post_mean],
        "max_rel_err_vs_truth": float(np.max(np.abs(post_mean / theta_true - 1))),
        "split_rhat_max": float(max(
            split_rhat(chain[:, k, :].T) for k in range(4)
        )),
        "config": {"chains": C, "experiments": args.experiments,
                   "model": "lotka_volterra_ode_adjoint", "dtype": "f64",
                   "mass": args.mass, "adaptation": args.adaptation,
                   "kernel": "k_lv_forward_batched/k_lv_adjoint_batched"},
    }))


if __name__ == "__main__":
    main()
