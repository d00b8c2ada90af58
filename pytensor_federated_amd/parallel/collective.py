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

    # -- helpers --------------------------------------------------------
    def _ensure_buffer(self, logp: torch.Tensor, grads: Sequence[torch.Tensor]) -> torch.Tensor:
        if self._buf is None:
            total = 1 + sum(g.numel() for g in grads)
            self._buf = torch.empty(total, dtype=torch.float64, device=logp.device)
            self._grad_shapes = [g.shape for g in grads]
        return self._buf

    # -- evaluation -----------------------------------------------------
    def logp_grad(self, *params) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        """Shard-local eval + all-reduce; returns federated (logp, grads)."""
        import torch.distributed as dist

        logp, grads = self.model.logp_grad(*params)
        buf = self._ensure_buffer(logp, grads)
        # Fuse into the persistent buffer.  When the model's kernel already
        # wrote views of one contiguous fp64 buffer this is a device-side
        # copy of a few doubles; eager paths pay one small pack.
        buf[0] = logp.to(torch.float64)
        off = 1
        for g in grads:
            n = g.numel()
            buf[off : off + n] = g.reshape(-1).to(torch.float64)
            off += n
        if self._distributed:
            dist.all_reduce(buf, op=dist.ReduceOp.SUM, group=self._group)
        logp_total = buf[0]
        out_grads = []
        off = 1
        for shape in self._grad_shapes:
            n = int(np.prod(shape)) if len(shape) else 1
            out_grads.append(buf[off : off + n].reshape(shape))
            off += n
        return logp_total, out_grads

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
