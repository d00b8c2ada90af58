#!/usr/bin/env python3
"""Full-stack inference throughput: NUTS posterior draws/sec on a GPU shard.

The framework's end purpose: every NUTS leapfrog step is one fused
logp+grad call against the persistent eval-server kernel, so full Bayesian
posterior sampling over a 1e7-row private shard runs at interactive rates.

    python benchmarks/bench_nuts.py --rows 10000000 --draws 500
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
    parser.add_argument("--rows", type=int, default=10_000_000)
    parser.add_argument("--draws", type=int, default=500)
    parser.add_argument("--tune", type=int, default=300)
    args = parser.parse_args()

    import torch

    from pytensor_federated_amd.inference import sample_nuts, split_rhat, summary
    from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
    from pytensor_federated_amd.ops import PersistentLinearEngine

    assert torch.cuda.is_available()
    x, y = generate_linear_dataset(args.rows, seed=17)
    model = GaussianLinearModel(x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16)
    engine = PersistentLinearEngine(model._x, model._y, 0.4)
    n_evals = [0]

    def logp_grad(theta):
        n_evals[0] += 1
        logp, ga, gb = engine.logp_grad_sync(theta[0], theta[1])
        return np.asarray(logp), [np.array([ga, gb])]

    try:
        t0 = time.perf_counter()
        chain = sample_nuts(
            logp_grad, [np.array([1.0, 0.3])], draws=args.draws, tune=args.tune,
            step_size=0.001, seed=3,
        )
        wall = time.perf_counter() - t0
    finally:
        engine.close()
    samples = np.stack([d[0] for d in chain])
    print(json.dumps({
        "metric": "NUTS posterior draws/sec (1 GPU shard)",
        "value": args.draws / wall,
        "grad_evals_per_sec": n_evals[0] / wall,
        "n_grad_evals": n_evals[0],
        "posterior_mean": [float(v) for v in samples.mean(axis=0)],
        "posterior_sd": [float(v) for v in samples.std(axis=0)],
        "split_rhat": [float(split_rhat(samples[:, k])) for k in range(2)],
        "config": {"rows": args.rows, "dtype": "bf16", "model": "gaussian_linear",
                   "engine": "persistent-kernel"},
    }))


if __name__ == "__main__":
    main()
