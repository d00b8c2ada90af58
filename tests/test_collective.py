"""Multi-rank shard engine tests on the gloo backend (CPU, world_size 2).

The RCCL/xGMI path on the GPU node is the same code with backend "nccl";
correctness-by-construction is what these tests pin down: the all-reduced
``[logp, *grads]`` equals the whole-data evaluation exactly.
"""
import multiprocessing
import os

import numpy as np
import pytest
import torch

from pytensor_federated_amd.models import (
    GaussianLinearModel,
    LogisticGLMModel,
    generate_linear_dataset,
    generate_logistic_dataset,
)
from pytensor_federated_amd.parallel import MultiShardDispatcher, shard_slice


def test_shard_slice_partitions():
    n, w = 101, 8
    slices = [shard_slice(n, r, w) for r in range(w)]
    covered = []
    for s in slices:
        covered.extend(range(n)[s])
    assert covered == list(range(n))


def _rank_entry(rank, world_size, port, result_queue):
    import torch.distributed as dist

    from pytensor_federated_amd.parallel import FederatedShardEngine

    dist.init_process_group(
        backend="gloo",
        init_method=f"tcp://127.0.0.1:{port}",
        rank=rank,
        world_size=world_size,
    )
    try:
        x, y = generate_linear_dataset(101, seed=7)
        s = shard_slice(101, rank, world_size)
        model = GaussianLinearModel(x[s], y[s], sigma=0.4)
        engine = FederatedShardEngine(model)
        logp, grads = engine(1.1, 0.6)
        result_queue.put((rank, float(logp), [float(g) for g in grads]))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_federated_allreduce_equals_whole_data():
    world_size = 2
    port = 29571
    ctx = multiprocessing.get_context("spawn")
    queue = ctx.Queue()
    procs = [
        ctx.Process(target=_rank_entry, args=(r, world_size, port, queue), daemon=True)
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world_size):
        rank, logp, grads = queue.get(timeout=240)
        results[rank] = (logp, grads)
    for p in procs:
        p.join(timeout=30)

    x, y = generate_linear_dataset(101, seed=7)
    whole = GaussianLinearModel(x, y, sigma=0.4)
    logp_ref, grads_ref = whole(1.1, 0.6)
    for rank in range(world_size):
        logp, grads = results[rank]
        np.testing.assert_allclose(logp, logp_ref, rtol=1e-12)
        np.testing.assert_allclose(grads, [float(g) for g in grads_ref], rtol=1e-10)


def test_engine_without_distributed_is_local():
    from pytensor_federated_amd.parallel import FederatedShardEngine

    x, y = generate_linear_dataset(50, seed=8)
    model = GaussianLinearModel(x, y, sigma=0.4)
    engine = FederatedShardEngine(model, use_distributed=False)
    logp, grads = engine(0.9, 0.3)
    logp_ref, grads_ref = model(0.9, 0.3)
    np.testing.assert_allclose(logp, logp_ref, rtol=1e-14)
    np.testing.assert_allclose(grads[0], grads_ref[0], rtol=1e-14)


class TestMultiShardDispatcher:
    def test_sum_equals_whole_linear(self):
        x, y = generate_linear_dataset(80, seed=9)
        shards = [
            GaussianLinearModel(x[:40], y[:40], sigma=0.4),
            GaussianLinearModel(x[40:], y[40:], sigma=0.4),
        ]
        disp = MultiShardDispatcher(shards)
        logp, grads = disp(1.0, 0.5)
        whole = GaussianLinearModel(x, y, sigma=0.4)
        logp_ref, grads_ref = whole(1.0, 0.5)
        np.testing.assert_allclose(logp, logp_ref, rtol=1e-12)
        for g, gr in zip(grads, grads_ref):
            np.testing.assert_allclose(g, gr, rtol=1e-10)

    def test_sum_equals_whole_logistic(self):
        X, y, beta0 = generate_logistic_dataset(90, 8, seed=10)
        shards = [LogisticGLMModel(X[:45], y[:45]), LogisticGLMModel(X[45:], y[45:])]
        disp = MultiShardDispatcher(shards)
        logp, grads = disp(beta0)
        whole = LogisticGLMModel(X, y)
        logp_ref, grads_ref = whole(beta0)
        np.testing.assert_allclose(logp, logp_ref, rtol=1e-12)
        np.testing.assert_allclose(grads[0], grads_ref[0], rtol=1e-10)

    def test_empty_raises(self):
        with pytest.raises(ValueError):
            MultiShardDispatcher([])


def _agreed_rank_entry(rank, world_size, port, result_queue):
    """Rank entry: attempt graphed create_agreed on CPU (capture must fail on
    every rank) and report that the agreed protocol returns None without
    deadlocking the collective sequence."""
    import torch.distributed as dist

    from pytensor_federated_amd.parallel.graphed import GraphedLinearEngine

    dist.init_process_group(
        backend="gloo",
        init_method=f"tcp://127.0.0.1:{port}",
        rank=rank,
        world_size=world_size,
    )
    try:
        x, y = generate_linear_dataset(64, seed=3)
        s = shard_slice(64, rank, world_size)
        model = GaussianLinearModel(x[s], y[s], sigma=0.4)
        engine = GraphedLinearEngine.create_agreed(model, distributed=True)
        # after the agreed outcome the communicator must still be usable
        t = torch.ones(1, dtype=torch.float64)
        dist.all_reduce(t)
        result_queue.put((rank, engine is None, float(t[0])))
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_create_agreed_degrades_uniformly_across_ranks():
    """Multi-rank de-risk for the hipgraph path: when capture fails (here: no
    GPU on any rank), EVERY rank gets None back and the process group remains
    usable -- the property that keeps an 8-GPU job from deadlocking when one
    rank's capture fails."""
    world_size = 2
    port = 29572
    ctx = multiprocessing.get_context("spawn")
    queue = ctx.Queue()
    procs = [
        ctx.Process(
            target=_agreed_rank_entry, args=(r, world_size, port, queue), daemon=True
        )
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world_size):
        rank, engine_is_none, reduced = queue.get(timeout=240)
        results[rank] = (engine_is_none, reduced)
    for p in procs:
        p.join(timeout=30)
    for rank in range(world_size):
        engine_is_none, reduced = results[rank]
        assert engine_is_none  # uniform fallback on every rank
        assert reduced == world_size  # communicator still aligned afterwards
