#!/usr/bin/env python3
"""Flagship benchmark: federated logp+grad calls/sec (BASELINE.json metric).

The JSON line's ``metric`` is "logp+grad calls/sec (whole node)" -- the
BASELINE.json metric string minus its trailing ", linear-regression demo
at 1/2/4/8 MI355X" clause, which describes the driver-run 1/2/4/8-GPU
scaling sweep rather than a single run; ``config`` names the model.

Measures the linear-regression demo config on 1..8 MI355X GPUs:
each rank (GPU) owns a private synthetic shard of 1e7 bf16 rows; one
"step" = one full federated logp+grad evaluation -- broadcast theta,
fused HIP kernel per shard, RCCL all-reduce of [logp, d/da, d/db], result
read back to the driver (what a PyMC client would receive).

Launch (the driver does this):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Rank 0 prints ONE JSON line with the whole-node call rate
(value = evals/sec * n_gpus: each federated evaluation is one worker-call
per GPU shard; weak scaling -- per-GPU rows fixed as N grows).
"""
import argparse
import json
import math
import os
import time

import numpy as np
import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=200)
    p.add_argument("--warmup", type=int, default=20)
    p.add_argument("--model", choices=["linear", "logistic", "ode"], default="linear")
    p.add_argument("--rows", type=int, default=None, help="rows per GPU shard")
    p.add_argument("--features", type=int, default=1024, help="logistic GLM features")
    p.add_argument("--dtype", choices=["bf16", "f32", "f64"], default="bf16")
    p.add_argument("--eager", action="store_true", help="disable HIP kernels (debug)")
    p.add_argument(
        "--no-readback",
        action="store_true",
        help="skip per-step host readback (measures enqueue throughput only; "
        "NOT the headline metric)",
    )
    return p.parse_known_args()[0]


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    distributed = world > 1

    have_gpu = torch.cuda.is_available()
    if have_gpu:
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)
    else:
        device = torch.device("cpu")

    dtype = {"bf16": torch.bfloat16, "f32": torch.float32, "f64": torch.float64}[args.dtype]
    if not have_gpu and dtype == torch.bfloat16:
        dtype = torch.float32  # CPU plumbing check only
    rows = args.rows
    if rows is None:
        rows = 10_000_000 if have_gpu else 100_000
        if args.model == "logistic":
            rows = 12_500_000 if have_gpu else 10_000

    if distributed:
        import torch.distributed as dist

        dist.init_process_group(backend="nccl" if have_gpu else "gloo")

    from pytensor_federated_amd.models import GaussianLinearModel, LogisticGLMModel
    from pytensor_federated_amd.parallel import FederatedShardEngine

    use_kernels = None if (have_gpu and not args.eager) else False

    # Private per-rank shard: synthetic, seeded by rank (federated = every
    # worker owns different data).  Generate on device to keep startup fast.
    gen = torch.Generator(device="cpu").manual_seed(1234 + rank)
    if args.model == "linear":
        x = torch.rand(rows, generator=gen) * 10.0
        noise = torch.randn(rows, generator=gen) * 0.4
        y = 1.5 + 0.5 * x + noise
        model = GaussianLinearModel(
            x, y, sigma=0.4, device=device, dtype=dtype, use_kernels=use_kernels
        )
        theta0 = np.array([1.5, 0.5])
        config_model = "gaussian_linear_regression"
        shape_cfg = {"rows_per_gpu": rows}
    elif args.model == "ode":
        from pytensor_federated_amd.models import ODEModel
        from pytensor_federated_amd.models.ode import (
            generate_ode_dataset,
            lotka_volterra_rhs,
        )

        n_exp = 1024 if have_gpu else 16
        n_steps = 50
        u0, obs_idx, y_obs = generate_ode_dataset(
            n_experiments=n_exp, n_obs=20, n_steps=n_steps, t1=8.0, sigma=0.1,
            seed=1234 + rank,
        )
        model = ODEModel(
            lotka_volterra_rhs, u0, 0.0, 8.0, n_steps, obs_idx, y_obs, 0.1,
            device=device, dtype=torch.float64,
        )
        theta0 = np.array([0.8, 0.3, 0.6, 0.2])
        config_model = "lotka_volterra_ode_adjoint"
        shape_cfg = {"experiments_per_gpu": n_exp, "rk4_steps": n_steps, "n_obs": 20}
    else:
        k = args.features
        X = torch.randn(rows, k, generator=gen) / math.sqrt(k)
        beta_true = torch.randn(k, generator=gen) * 0.5
        p = torch.sigmoid(X @ beta_true)
        y = (torch.rand(rows, generator=gen) < p).to(torch.float32)
        model = LogisticGLMModel(X, y, device=device, dtype=dtype, use_kernels=use_kernels)
        theta0 = beta_true.numpy()
        config_model = "logistic_glm"
        shape_cfg = {"rows_per_gpu": rows, "features": k}

    engine = FederatedShardEngine(model, use_distributed=distributed)

    readback = not args.no_readback
    a0, b0 = (float(theta0[0]), float(theta0[1])) if args.model == "linear" else (0.0, 0.0)
    theta_dev = torch.as_tensor(theta0, dtype=torch.float64, device=device)
    host_buf = None

    # path override for debugging/driver triage:
    #   PFA_BENCH_PATH=persistent|graphed|sync-native|engine
    path_override = os.environ.get("PFA_BENCH_PATH", "")

    # persistent eval-server path (single-GPU linear): resident kernel,
    # pinned request/result mailboxes -- no launch, no ramp per call
    persistent = None
    if (
        args.model == "linear"
        and have_gpu
        and not distributed
        and use_kernels is None
        and readback
        and path_override in ("", "persistent")
    ):
        try:
            from pytensor_federated_amd.ops import PersistentLinearEngine

            persistent = PersistentLinearEngine(model._x, model._y, model._sigma)
        except Exception as ex:
            print(f"# persistent path unavailable ({ex}); falling back", flush=True)
            persistent = None

    # graphed paths: hipGraph replay of [H2D theta -> model eval ->
    # RCCL all-reduce -> mailbox publish] per evaluation
    graphed = None
    if persistent is not None or path_override in ("sync-native", "engine"):
        pass
    elif args.model == "linear" and have_gpu and use_kernels is None and readback:
        try:
            from pytensor_federated_amd.parallel.graphed import GraphedLinearEngine

            # rank-safe: all ranks agree on the path (a one-sided capture
            # failure would desynchronize the collective sequence)
            graphed = GraphedLinearEngine.create_agreed(model, distributed=distributed)
            if graphed is None and rank == 0:
                print("# graphed path unavailable (agreed); engine fallback", flush=True)
        except Exception as ex:
            print(f"# graphed path unavailable ({ex}); falling back", flush=True)
            graphed = None
    elif args.model == "ode" and have_gpu and readback:
        try:
            from pytensor_federated_amd.parallel.graphed import GraphedLogpGradEngine

            graphed = GraphedLogpGradEngine.create_agreed(model, (4,), distributed=distributed)
            if graphed is None and rank == 0:
                print("# graphed ODE path unavailable (agreed); engine fallback", flush=True)
        except Exception as ex:
            print(f"# graphed ODE path unavailable ({ex}); falling back", flush=True)
            graphed = None

    # single-GPU linear serving path: one native sync call per evaluation
    fast_sync = (
        graphed is None
        and persistent is None
        and not distributed
        and readback
        and args.model == "linear"
        and hasattr(model, "logp_grad_sync")
        and path_override != "engine"
    )

    def one_step(t: int):
        nonlocal host_buf
        # every rank derives the same perturbed theta (the broadcast of theta
        # from the driver is folded into the all-reduce round trip below)
        scale = 1.0 + 0.001 * math.sin(t)
        if persistent is not None:
            return persistent.logp_grad_sync(a0 * scale, b0 * scale)
        if graphed is not None:
            if args.model == "linear":
                return graphed.logp_grad_sync(a0 * scale, b0 * scale)
            return graphed.logp_grad_sync(theta0 * scale)
        if fast_sync:
            return model.logp_grad_sync(a0 * scale, b0 * scale)
        if args.model == "linear":
            buf = engine.logp_grad_fused(a0 * scale, b0 * scale)
        else:
            buf = engine.logp_grad_fused(theta_dev * scale)
        if readback:
            # deliver the federated [logp, grads] to the driver like a real
            # client call would (forces the per-call sync)
            if host_buf is None:
                host_buf = torch.empty_like(buf, device="cpu", pin_memory=have_gpu)
            host_buf.copy_(buf)
            if have_gpu:
                torch.cuda.synchronize()
            return host_buf
        return None

    # Per-call collective cost (multi-rank only): mean wall time of one
    # eager all_reduce of the 3..K-double [logp, grads] buffer, measured
    # outside the bench loop so SCALE runs carry the latency diagnostic
    # whichever path (graphed or engine) the steps take.
    allreduce_us = None
    if distributed:
        import torch.distributed as dist

        probe = torch.zeros(
            3 if args.model == "linear" else int(np.prod(theta0.shape)) + 1,
            dtype=torch.float64,
            device=device,
        )
        for _ in range(20):  # warm the communicator/algorithm choice
            dist.all_reduce(probe)
        if have_gpu:
            torch.cuda.synchronize()
        dist.barrier()
        tar0 = time.perf_counter()
        reps = 200
        for _ in range(reps):
            dist.all_reduce(probe)
        if have_gpu:
            torch.cuda.synchronize()
        allreduce_us = (time.perf_counter() - tar0) / reps * 1e6

    def barrier_sync():
        if distributed:
            import torch.distributed as dist

            dist.barrier()
        if have_gpu and persistent is None:
            # (the persistent eval server answers synchronously; a device-wide
            # synchronize would instead wait out its idle lifetime)
            torch.cuda.synchronize()

    for t in range(args.warmup):
        one_step(t)
    barrier_sync()
    t0 = time.perf_counter()
    for t in range(args.steps):
        one_step(t)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks
    if distributed:
        import torch.distributed as dist

        e = torch.tensor([elapsed], dtype=torch.float64, device=device if have_gpu else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e[0])

    n_gpus = world if distributed else 1
    evals_per_s = args.steps / elapsed
    value = evals_per_s * n_gpus  # whole-node worker-call rate
    if rank == 0:
        result = {
            "metric": "logp+grad calls/sec (whole node)",
            "value": value,
            "unit": "calls/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "f64" if args.model == "ode" else (args.dtype if have_gpu else "f32"),
            "data": "synthetic",
            "config": {
                "model": config_model,
                "parallelism": (
                    f"federated shard-dp x{n_gpus} (RCCL all-reduce)"
                    if n_gpus > 1
                    else "federated shard x1 (single shard, no collective)"
                ),
                "allreduce_us": allreduce_us,
                "per_step": (
                    "theta -> fused logp+grad kernel per shard -> "
                    "all_reduce([logp,*grads]) -> host readback"
                    if n_gpus > 1
                    else "theta -> fused logp+grad kernel -> host readback"
                ),
                "readback": readback,
                "device": str(device),
                "kernels": bool(use_kernels is None),
                "path": (
                    "persistent-kernel" if persistent is not None
                    else "hipgraph-replay" if graphed is not None
                    else "sync-native" if fast_sync
                    else "engine"
                ),
                **shape_cfg,
            },
        }
        print(json.dumps(result), flush=True)

    if distributed:
        import torch.distributed as dist

        dist.destroy_process_group()


if __name__ == "__main__":
    main()
