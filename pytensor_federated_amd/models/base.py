"""Model-side contract: differentiable blackboxes served by workers.

The reference's worker wraps a PyTensor-compiled ``[params] ->
[logp, *grads]`` function (reference demo_node.py:30-54).  Here a model is a
small class owning its private data shard (resident in HBM on a GPU
worker), exposing:

* ``logp_grad(*params) -> (logp, [grads])`` on torch tensors -- the hot
  path: eager torch everywhere, or a fused CDNA4 HIP kernel when the shard
  lives on a ROCm device and the extension is built;
* ``__call__`` / ``as_logp_grad_func()`` -- the numpy edge used by the gRPC
  service (``wrap_logp_grad_func`` layout ``[logp, *grads]``).
"""
from __future__ import annotations

import time
from typing import List, Optional, Tuple

import numpy as np
import torch

__all__ = ["LogpGradModel"]


class LogpGradModel:
    """Base class for logp+grad models with private data."""

    #: names of the scalar/vector parameters, in call order
    param_names: Tuple[str, ...] = ()

    def __init__(self, *, delay: Optional[float] = None) -> None:
        # Optional artificial delay, parity with the reference's demo worker
        # (demo_node.py:45-54) -- used by load-balancing demos/tests.
        self._delay = delay

    # -- to implement ---------------------------------------------------
    def logp_grad(self, *params: torch.Tensor) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        raise NotImplementedError

    # -- provided -------------------------------------------------------
    def __call__(self, *params) -> Tuple[np.ndarray, List[np.ndarray]]:
        """Numpy-edge evaluation (the LogpGradFunc signature)."""
        if self._delay is not None:
            t_start = time.perf_counter()
        # copy: wire-decoded arrays are read-only zero-copy views
        tparams = [torch.from_numpy(np.array(p, dtype=np.float64)) for p in params]
        logp, grads = self.logp_grad(*tparams)
        result = (
            np.asarray(logp.detach().cpu().double().numpy()),
            [np.asarray(g.detach().cpu().double().numpy()) for g in grads],
        )
        if self._delay is not None:
            remaining = self._delay - (time.perf_counter() - t_start)
            if remaining > 0:
                time.sleep(remaining)
        return result

    def as_logp_grad_func(self):
        return self.__call__

    def as_logp_func(self):
        def logp_func(*params):
            logp, _ = self(*params)
            return logp

        return logp_func
