"""hipGraph-replayed federated evaluation (the lowest-latency multi-GPU path).

One captured graph per rank holds the whole per-call pipeline:

    H2D copy(theta_pinned -> theta_dev)
    -> fused gaussian logp+grad kernel (theta read from device memory)
    -> RCCL all_reduce([logp, ga, gb]) over xGMI          (multi-rank only)
    -> publish kernel (device epoch++ -> pinned mailbox {results, seq})

This is the latency tier below the reference's hot path (its persistent
EvaluateStream, reference service.py:150-158, still pays a Python+gRPC
round trip per call): per evaluation the host writes two doubles into
pinned memory, replays the graph, and spin-reads the mailbox seq -- no per-step torch dispatch, no
stream sync call, no collective setup.  RCCL collectives are capturable on
ROCm through torch's ProcessGroupNCCL, so the all-reduce rides inside the
same replay.
"""
from __future__ import annotations

import time
from typing import Tuple

import numpy as np
import torch

__all__ = ["GraphedLinearEngine", "GraphedLogpGradEngine"]


class GraphedLogpGradEngine:
    """hipGraph-replay any torch model's ``logp_grad(theta)`` evaluation.

    Captures the model's whole forward+adjoint computation (an ODE model's
    RK4 sweep + discrete-adjoint backward is ~2500 small kernels -- replay
    removes every per-kernel Python dispatch and launch), the optional RCCL
    all-reduce, and the mailbox publish.  Requires a model whose
    ``logp_grad`` is shape-static and sync-free (ODEModel qualifies).
    """

    @staticmethod
    def create_agreed(model, theta_shape, distributed: bool = False, group=None):
        """Rank-safe construction (see GraphedLinearEngine.create_agreed)."""
        if not distributed:
            try:
                return GraphedLogpGradEngine(model, theta_shape, distributed=False, group=group)
            except Exception:
                return None
        import torch.distributed as dist

        device = model.device
        probe = torch.zeros(1, dtype=torch.float64, device=device)
        dist.all_reduce(probe, group=group)  # aligned communicator warm-up
        engine = None
        try:
            engine = GraphedLogpGradEngine(
                model, theta_shape, distributed=True, group=group, skip_comm_warmup=True
            )
            ok = 1.0
        except Exception:
            ok = 0.0
        flag = torch.tensor([ok], dtype=torch.float64, device=device)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN, group=group)
        return engine if float(flag[0]) >= 1.0 else None

    def __init__(self, model, theta_shape, distributed: bool = False, group=None,
                 skip_comm_warmup: bool = False) -> None:
        from ..ops import alloc_mailbox, publish_result

        self.model = model
        self._distributed = distributed
        self._group = group
        device = model.device
        n_theta = int(np.prod(theta_shape))
        self.theta_pinned = torch.zeros(n_theta, dtype=torch.float64, pin_memory=True)
        self._theta_np = self.theta_pinned.numpy()
        self.theta_dev = torch.zeros(theta_shape, dtype=torch.float64, device=device)

        logp, grads = model.logp_grad(self.theta_dev)
        n_out = 1 + sum(g.numel() for g in grads)
        self.buf = torch.zeros(n_out, dtype=torch.float64, device=device)
        self._grad_shapes = [g.shape for g in grads]
        self.mailbox = alloc_mailbox(n_out)
        self._seq_view = self.mailbox[n_out:].view(np.uint64)
        self.epoch_dev = torch.zeros(1, dtype=torch.int64, device=device)
        self._n_out = n_out

        def body(include_comm: bool = True):
            self.theta_dev.copy_(
                self.theta_pinned.reshape(self.theta_dev.shape), non_blocking=True
            )
            logp, grads = model.logp_grad(self.theta_dev)
            self.buf[0] = logp
            off = 1
            for g in grads:
                n = g.numel()
                self.buf[off : off + n] = g.reshape(-1).to(torch.float64)
                off += n
            if self._distributed and include_comm:
                import torch.distributed as dist

                dist.all_reduce(self.buf, op=dist.ReduceOp.SUM, group=self._group)
            publish_result(self.buf, self.mailbox, self.epoch_dev)

        # warmup; no collectives here under create_agreed (a rank whose
        # construction fails before them would desync the job)
        body(include_comm=not skip_comm_warmup)
        torch.cuda.synchronize()
        self._expected = int(self._seq_view[0])
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            body()

    def logp_grad_sync(self, theta) -> Tuple[float, np.ndarray]:
        self._theta_np[:] = np.asarray(theta, dtype=np.float64).reshape(-1)
        self.graph.replay()
        self._expected += 1
        seq = self._seq_view
        deadline = time.perf_counter() + 30.0
        while int(seq[0]) != self._expected:
            if time.perf_counter() > deadline:
                torch.cuda.synchronize()
                if int(seq[0]) == self._expected:
                    break
                raise RuntimeError("graphed eval timed out")
        return float(self.mailbox[0]), self.mailbox[1 : self._n_out].copy()

    def __call__(self, theta):
        logp, grad_flat = self.logp_grad_sync(theta)
        out, off = [], 0
        for shape in self._grad_shapes:
            n = int(np.prod(shape)) if len(shape) else 1
            out.append(grad_flat[off : off + n].reshape(shape))
            off += n
        return np.asarray(logp), out

    def as_logp_grad_func(self):
        return self.__call__


class GraphedLinearEngine:
    """Graph-replayed evaluator for a GaussianLinearModel shard."""

    @staticmethod
    def create_agreed(model, distributed: bool = False, group=None):
        """Rank-safe construction for multi-rank use.

        A graph capture that fails on SOME ranks while others proceed would
        desynchronize the collective sequence (the engine's warmup runs one
        all-reduce) and deadlock the job.  Protocol: every rank first runs
        one ALIGNED eager all-reduce (communicator init), then attempts the
        capture, then all ranks agree (MIN-reduce of a success flag);
        returns the engine only if EVERY rank captured, else None on all
        ranks -- the caller falls back to the eager engine path uniformly.
        """
        if not distributed:
            try:
                return GraphedLinearEngine(model, distributed=False, group=group)
            except Exception:
                return None
        import torch.distributed as dist

        device = model._x.device
        # aligned communicator warm-up (1 collective on every rank)
        probe = torch.zeros(3, dtype=torch.float64, device=device)
        dist.all_reduce(probe, group=group)
        engine = None
        try:
            engine = GraphedLinearEngine(
                model, distributed=True, group=group, skip_comm_warmup=True
            )
            ok = 1.0
        except Exception:
            engine = None
            ok = 0.0
        flag = torch.tensor([ok], dtype=torch.float64, device=device)
        dist.all_reduce(flag, op=dist.ReduceOp.MIN, group=group)
        if float(flag[0]) < 1.0:
            return None
        return engine

    def __init__(self, model, distributed: bool = False, group=None,
                 skip_comm_warmup: bool = False) -> None:
        from ..ops import (
            alloc_mailbox,
            gaussian_linear_launch_theta,
            gaussian_workspace,
            publish_result,
        )

        if not model._x.is_cuda:
            raise ValueError("GraphedLinearEngine needs a GPU-resident model.")
        self.model = model
        self._distributed = distributed
        self._group = group
        device = model._x.device
        self.theta_pinned = torch.zeros(2, dtype=torch.float64, pin_memory=True)
        self._theta_np = self.theta_pinned.numpy()
        self.theta_dev = torch.zeros(2, dtype=torch.float64, device=device)
        self.buf = torch.zeros(3, dtype=torch.float64, device=device)
        self.ws = gaussian_workspace(device)
        self.mailbox = alloc_mailbox(3)
        self._seq_view = self.mailbox[3:].view(np.uint64)
        self.epoch_dev = torch.zeros(1, dtype=torch.int64, device=device)
        self._expected = 0

        def body(include_comm: bool = True):
            self.theta_dev.copy_(self.theta_pinned, non_blocking=True)
            gaussian_linear_launch_theta(
                model._x, model._y, self.theta_dev, model._sigma, self.buf, self.ws
            )
            if self._distributed and include_comm:
                import torch.distributed as dist

                dist.all_reduce(self.buf, op=dist.ReduceOp.SUM, group=self._group)
            publish_result(self.buf, self.mailbox, self.epoch_dev)

        # warmup; under create_agreed the communicator was already warmed by
        # the ALIGNED probe, and this warmup must not issue collectives (a
        # rank whose construction fails before them would desync the job)
        body(include_comm=not skip_comm_warmup)
        torch.cuda.synchronize()
        self._expected = int(self._seq_view[0])

        self.graph = torch.cuda.CUDAGraph()
        # capture on a side stream per torch's capture contract
        with torch.cuda.graph(self.graph):
            body()

    def logp_grad_sync(self, intercept: float, slope: float) -> Tuple[float, float, float]:
        self._theta_np[0] = intercept
        self._theta_np[1] = slope
        self.graph.replay()
        self._expected += 1
        seq = self._seq_view
        deadline = time.perf_counter() + 5.0
        while int(seq[0]) != self._expected:
            if time.perf_counter() > deadline:
                torch.cuda.synchronize()
                if int(seq[0]) == self._expected:
                    break
                raise RuntimeError(
                    f"graphed eval timed out (seq {int(seq[0])} != {self._expected})"
                )
        return float(self.mailbox[0]), float(self.mailbox[1]), float(self.mailbox[2])

    def __call__(self, *params):
        logp, ga, gb = self.logp_grad_sync(float(params[0]), float(params[1]))
        return np.asarray(logp), [np.asarray(ga), np.asarray(gb)]

    def as_logp_grad_func(self):
        return self.__call__
