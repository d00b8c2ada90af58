#!/usr/bin/env python3
"""Client demo: drive federated workers from a torch-side model.

Parity with reference demo_model.py: connects LogpGradServiceClients to the
worker pool (balanced), embeds them as a differentiable blackbox, finds the
MAP with gradient ascent and samples the posterior with a simple
random-walk Metropolis -- pytensor/pymc-free, but a PyMC user can do the
same via the optional wrapper_ops adapters.

    python demo_model.py --host 127.0.0.1 --ports 50000 50001 --parallel
"""
import argparse

import numpy as np
import torch


def run_model(host: str, ports, parallel: bool, map_steps: int, draws: int,
              sampler: str = "metropolis"):
    from pytensor_federated_amd.common import LogpGradServiceClient
    from pytensor_federated_amd.torch_ops import FederatedLogpGrad, LogpGradOp

    clients = [LogpGradServiceClient(host, p) for p in ports]
    if parallel:
        # concurrent fan-out over all shards (the reference's fused AsyncOps)
        logp_op = FederatedLogpGrad([c.evaluate_async for c in clients])
    else:
        ops = [LogpGradOp(c.evaluate) for c in clients]

        def logp_op(a, b):
            return sum(op(a, b) for op in ops)

    # ---- find_MAP: gradient ascent on the federated logp ----------------
    a = torch.tensor(0.0, requires_grad=True, dtype=torch.float64)
    b = torch.tensor(0.0, requires_grad=True, dtype=torch.float64)
    opt = torch.optim.Adam([a, b], lr=0.05)
    for step in range(map_steps):
        opt.zero_grad()
        loss = -logp_op(a, b)
        loss.backward()
        opt.step()
        if step % 20 == 0:
            print(f"  MAP step {step}: logp={-loss.item():.3f} a={a.item():.4f} b={b.item():.4f}")
    print(f"MAP estimate: intercept={a.item():.4f} slope={b.item():.4f}")

    if sampler == "nuts":
        # ---- posterior: NUTS (each leapfrog = one fused federated call) --
        from pytensor_federated_amd.inference import sample_nuts, summary

        def logp_grad(theta):
            ta = torch.tensor(theta[0], requires_grad=True, dtype=torch.float64)
            tb = torch.tensor(theta[1], requires_grad=True, dtype=torch.float64)
            lp = logp_op(ta, tb)
            lp.backward()
            return np.asarray(lp.detach().item()), [
                np.array([float(ta.grad), float(tb.grad)])
            ]

        draws_list = sample_nuts(
            logp_grad, [np.array([a.item(), b.item()])],
            draws=draws, tune=max(draws // 2, 25), step_size=0.01, seed=0,
        )
        chain = np.stack([d[0] for d in draws_list])
        print(f"NUTS posterior over {draws} draws:")
        print(summary({"intercept": chain[:, 0], "slope": chain[:, 1]}))
        return chain

    # ---- posterior: random-walk Metropolis ------------------------------
    rng = np.random.default_rng(0)
    theta = np.array([a.item(), b.item()])
    with torch.no_grad():
        cur_lp = float(logp_op(torch.tensor(theta[0]), torch.tensor(theta[1])))
    chain = []
    accepted = 0
    for _ in range(draws):
        prop = theta + rng.normal(scale=0.02, size=2)
        lp = float(logp_op(torch.tensor(prop[0]), torch.tensor(prop[1])))
        if np.log(rng.uniform()) < lp - cur_lp:
            theta, cur_lp = prop, lp
            accepted += 1
        chain.append(theta.copy())
    chain = np.asarray(chain)
    from pytensor_federated_amd.inference import summary

    print(f"Posterior over {draws} draws (accept {accepted / draws:.0%}):")
    print(summary({"intercept": chain[:, 0], "slope": chain[:, 1]}))
    return chain


if __name__ == "__main__":
    parser = argparse.ArgumentParser()
    parser.add_argument("--host", default="127.0.0.1")
    parser.add_argument("--ports", type=int, nargs="+", default=list(range(50000, 50015)))
    parser.add_argument("--parallel", action="store_true")
    parser.add_argument("--map-steps", type=int, default=100)
    parser.add_argument("--draws", type=int, default=200)
    parser.add_argument("--sampler", choices=["metropolis", "nuts"], default="metropolis")
    args, _ = parser.parse_known_args()
    run_model(args.host, args.ports, args.parallel, args.map_steps, args.draws,
              sampler=args.sampler)
