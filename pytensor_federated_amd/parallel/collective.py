"""Federated shard engine: N ranks, N private shards, one all-reduce.

The MI355X-native replacement for the reference's on-node fan-out
(N gRPC streams + graph-level sum of ``pm.Potential`` terms,
reference op_async.py:107-132 + demo_model.py:28-36): each of the node's
8 GPUs is one federated shard (one process per GPU, ``torch.distributed``
with the nccl backend = RCCL over xGMI); every evaluation computes the
shard's ``[logp, *grads]`` into ONE fused fp64 device buffer and a single
``all_reduce(SUM)`` produces the exact federated total on every rank
(logp and grads both distribute over data shards).

The payload is tiny (3 doubles for the linear demo, 1+K for the GLM), so
the binding constraint is latency, not bandwidth: one persistent
communicator, one pre-registered buffer, one collective per evaluation.

Works identically with the gloo backend on CPU (how the multi-rank path is
tested without GPUs) -- only the buffer's device changes.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

__all__ = ["FederatedShardEngine", "init_process_group_from_env", "shard_slice"]


def shard_slice(n: int, rank: int, world_size: int) -> slice:
    """Contiguous shard boundaries: rows [rank*n//W, (rank+1)*n//W)."""
    return slice(rank * n // world_size, (rank + 1) * n // world_size)


def init_process_group_from_env(backend: Optional[str] = None):
    """Initialize torch.distributed from torchrun env vars (idempotent)."""
    import torch.distributed as dist

    if dist.is_initialized():
        return dist.group.WORLD
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend)
    if backend == "nccl":
        torch.cuda.set_device(int(torch.distributed.get_rank() % torch.cuda.device_count()))
    return dist.group.WORLD


class FederatedShardEngine:
    """Owns one shard's model + the fused [logp, *grads] all-reduce buffer."""

    def __init__(self, model, group=None, use_distributed: Optional[bool] = None) -> None:
        """
        Parameters
        ----------
        model : LogpGradModel
            This rank's model over its private data shard.
        group : torch.distributed process group, optional
            Defaults to WORLD when torch.distributed is initialized.
        use_distributed : bool, optional
            Force-enable/disable the all-reduce (default: auto-detect).
        """
        import torch.distributed as dist

        self.model = model
        self._group = group
        if use_distributed is None:
            use_distributed = dist.is_available() and dist.is_initialized()
        self._distributed = use_distributed
        self._buf: Optional[torch.Tensor] = None
        self._grad_shapes: Optional[List[torch.Size]] = None
        self._grad_views: Optional[List[torch.Tensor]] = None

    # -- helpers --------------------------------------------------------
    def _init_buffer(self, logp: torch.Tensor, grads: Sequence[torch.Tensor]) -> None:
        total = 1 + sum(g.numel() for g in grads)
        self._buf = torch.empty(total, dtype=torch.float64, device=logp.device)
        self._grad_shapes = [g.shape for g in grads]
        self._grad_views = []
        off = 1
        for shape in self._grad_shapes:
            n = int(np.prod(shape)) if len(shape) else 1
            self._grad_views.append(self._buf[off : off + n].reshape(shape))
            off += n

    # -- evaluation -----------------------------------------------------
    def logp_grad(self, *params) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        """Shard-local eval + all-reduce; returns federated (logp, grads).

        The model writes its fused [logp, *grads] directly into the
        persistent all-reduce buffer when it supports ``out=`` (the HIP
        kernels do), so the hot path is: one kernel launch -> one
        ``all_reduce`` -> views.  No per-call allocations or packing.
        """
        import torch.distributed as dist

        if self._buf is None:
            logp, grads = self.model.logp_grad(*params)
            self._init_buffer(logp, grads)
            self._buf[0] = logp.to(torch.float64)
            for view, g in zip(self._grad_views, grads):
                view.copy_(g.reshape(view.shape).to(torch.float64))
        else:
            try:
                logp, grads = self.model.logp_grad(*params, out=self._buf)
                aliased = logp.data_ptr() == self._buf.data_ptr()
            except TypeError:  # model without out= support
                logp, grads = self.model.logp_grad(*params)
                aliased = False
            if not aliased:
                self._buf[0] = logp.to(torch.float64)
                for view, g in zip(self._grad_views, grads):
                    view.copy_(g.reshape(view.shape).to(torch.float64))
        if self._distributed:
            dist.all_reduce(self._buf, op=dist.ReduceOp.SUM, group=self._group)
        return self._buf[0], list(self._grad_views)

    def logp_grad_fused(self, *params) -> torch.Tensor:
        """Hot-path variant returning the raw fused fp64 buffer
        ``[logp, *grads]`` (still on device, no host sync)."""
        self.logp_grad(*params)
        return self._buf

    # -- numpy edge (what rank 0's gRPC service serves) ------------------
    def __call__(self, *params) -> Tuple[np.ndarray, List[np.ndarray]]:
        tparams = [torch.as_tensor(np.asarray(p, dtype=np.float64)) for p in params]
        logp, grads = self.logp_grad(*tparams)
        return (
            np.asarray(logp.detach().cpu().numpy()),
            [np.asarray(g.detach().cpu().numpy()) for g in grads],
        )

    def as_logp_grad_func(self):
        return self.__call__
