"""Gaussian linear-regression logp+grad model.

Parity target: the reference's ``LinearModelBlackbox`` (demo_node.py:30-54):
``pred = intercept + x*slope``; ``logp = sum(log N(y | pred, sigma))``;
gradients w.r.t. (intercept, slope).  Closed form used here (the reference
gets the same from PyTensor autodiff):

    r        = y - (a + b*x)
    logp     = -N/2 * log(2*pi*sigma^2) - sum(r^2) / (2*sigma^2)
    dlogp/da = sum(r) / sigma^2
    dlogp/db = sum(r*x) / sigma^2

Compute paths:
* CPU / opt-out: eager torch, fp32+ accumulation (fp64 data -> fp64 math);
* MI355X: one fused CDNA4 HIP kernel (``ops.gaussian_linear_logp_grad``)
  computing all three reductions in a single pass over x,y -- the op is
  HBM-bandwidth-bound, so one pass is the speed-of-light shape.
"""
from __future__ import annotations

import math
from typing import List, Optional, Tuple

import numpy as np
import torch

from .base import LogpGradModel

__all__ = ["GaussianLinearModel", "generate_linear_dataset"]


def generate_linear_dataset(
    n_rows: int = 10,
    *,
    intercept: float = 1.5,
    slope: float = 0.5,
    sigma: float = 0.4,
    x_max: float = 10.0,
    seed: int = 0,
) -> Tuple[np.ndarray, np.ndarray]:
    """Synthetic dataset matching the reference demo's shape.

    (reference demo_node.py:58-61: ``x = linspace(0, 10, 10)``,
    ``y ~ N(1.5 + 0.5*x, 0.4^2)``)
    """
    rng = np.random.RandomState(seed)
    x = np.linspace(0, x_max, n_rows)
    y = rng.normal(loc=intercept + slope * x, scale=sigma)
    return x, y


class GaussianLinearModel(LogpGradModel):
    """Federated worker model: Gaussian linear regression on a private shard."""

    param_names = ("intercept", "slope")

    def __init__(
        self,
        x,
        y,
        sigma: float,
        *,
        device=None,
        dtype: torch.dtype = None,
        use_kernels: Optional[bool] = None,
        delay: Optional[float] = None,
    ) -> None:
        """
        Parameters
        ----------
        x, y : array-like
            The private data shard (stays resident on ``device``).
        sigma : float
            Fixed observation noise scale.
        device, dtype
            Where/how the shard is stored.  On an MI355X worker use
            ``device="cuda", dtype=torch.bfloat16`` -- 288 GB HBM3E holds
            ~7e10 bf16 rows per GPU.
        use_kernels : bool, optional
            ``None`` (default): fused HIP kernel on ROCm devices (loud error
            if the extension is missing), eager torch on CPU.
            ``False``: always eager torch.  ``True``: require the kernel.
        """
        super().__init__(delay=delay)
        x = torch.as_tensor(np.asarray(x)) if not isinstance(x, torch.Tensor) else x
        y = torch.as_tensor(np.asarray(y)) if not isinstance(y, torch.Tensor) else y
        if dtype is None:
            dtype = x.dtype if x.is_floating_point() else torch.float64
        if device is None:
            device = x.device
        self._x = x.to(device=device, dtype=dtype).contiguous()
        self._y = y.to(device=device, dtype=dtype).contiguous()
        self._sigma = float(sigma)
        self._use_kernels = use_kernels
        self._n = self._x.numel()
        self._ws = None  # per-model kernel workspace (slab + arrival ticket)
        if self._x.shape != self._y.shape or self._x.dim() != 1:
            raise ValueError("x and y must be equal-length 1-d arrays.")

    @property
    def device(self):
        return self._x.device

    @property
    def n_rows(self) -> int:
        return self._n

    def _kernel_path(self) -> bool:
        on_gpu = self._x.is_cuda
        if self._use_kernels is None:
            return on_gpu
        return self._use_kernels

    #: layout of the fused fp64 output buffer: [logp, dlogp/da, dlogp/db]
    fused_size = 3

    def logp_grad(
        self, intercept, slope, out: Optional[torch.Tensor] = None
    ) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        """``out`` (fp64[3], same device) receives the fused [logp, ga, gb]
        in place -- the zero-copy seam to the RCCL all-reduce buffer."""
        a = float(intercept)
        b = float(slope)
        if self._kernel_path():
            from ..ops import gaussian_linear_logp_grad

            logp, ga, gb = gaussian_linear_logp_grad(
                self._x, self._y, a, b, self._sigma, out=out, ws=self._kernel_ws()
            )
            return logp, [ga, gb]
        logp, grads = self._logp_grad_eager(a, b)
        if out is not None:
            out[0] = logp
            out[1] = grads[0]
            out[2] = grads[1]
            return out[0], [out[1], out[2]]
        return logp, grads

    def _kernel_ws(self):
        if self._ws is None:
            from ..ops import gaussian_workspace

            self._ws = gaussian_workspace(self._x.device)
        return self._ws

    def logp_grad_sync(self, intercept, slope) -> Tuple[float, float, float]:
        """Lowest-latency single-GPU path: one native call returning host
        floats (fused kernel + GPU-written pinned mailbox poll)."""
        if self._kernel_path():
            from ..ops import gaussian_linear_eval_sync

            return gaussian_linear_eval_sync(
                self._x, self._y, float(intercept), float(slope), self._sigma,
                ws=self._kernel_ws(),
            )
        logp, (ga, gb) = self._logp_grad_eager(float(intercept), float(slope))
        return float(logp), float(ga), float(gb)

    def _logp_grad_eager(self, a: float, b: float) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        x, y = self._x, self._y
        acc_dtype = torch.float64 if x.dtype == torch.float64 else torch.float32
        xf = x.to(acc_dtype)
        r = y.to(acc_dtype) - (a + b * xf)
        sig2 = self._sigma * self._sigma
        sum_r2 = torch.sum(r * r, dtype=torch.float64)
        sum_r = torch.sum(r, dtype=torch.float64)
        sum_rx = torch.sum(r * xf, dtype=torch.float64)
        logp = -0.5 * self._n * math.log(2.0 * math.pi * sig2) - sum_r2 / (2.0 * sig2)
        return logp, [sum_r / sig2, sum_rx / sig2]
