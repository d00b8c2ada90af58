"""Logistic-GLM logp+grad model (BASELINE.json config 4 family).

The GLM generalization of the reference's ComputeFunc contract
(signatures.py:27-33): parameters ``beta[K]``, private shard ``X[N,K]``,
``y[N] in {0,1}``:

    z    = X @ beta
    logp = sum( y*z - softplus(z) )          (numerically stable BCE)
    grad = X^T @ (y - sigmoid(z))

Compute paths:
* eager torch (CPU fallback / golden reference): two rocBLAS matmuls +
  elementwise;
* MI355X: ONE fused CDNA4 HIP kernel reading X exactly once -- per row-tile
  it computes z, sigmoid, the logp term and the rank-1 grad update while the
  tile is still in registers.  At N=1e8 x K=1024 bf16 the shard is 25.6 GB
  per GPU; the op is HBM-bound, so single-pass is ~2x over the eager
  two-matmul shape.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import numpy as np
import torch

from .base import LogpGradModel

__all__ = ["LogisticGLMModel", "generate_logistic_dataset"]


def generate_logistic_dataset(
    n_rows: int,
    n_features: int,
    *,
    seed: int = 0,
    beta_scale: float = 0.5,
) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """Synthetic (X, y, beta_true) with ~balanced classes."""
    rng = np.random.RandomState(seed)
    X = rng.standard_normal((n_rows, n_features)) / np.sqrt(n_features)
    beta = rng.standard_normal(n_features) * beta_scale
    p = 1.0 / (1.0 + np.exp(-(X @ beta)))
    y = (rng.uniform(size=n_rows) < p).astype(np.float64)
    return X, y, beta


class LogisticGLMModel(LogpGradModel):
    """Federated worker model: logistic regression on a private shard."""

    param_names = ("beta",)

    def __init__(
        self,
        X,
        y,
        *,
        device=None,
        dtype: torch.dtype = None,
        use_kernels: Optional[bool] = None,
        delay: Optional[float] = None,
    ) -> None:
        super().__init__(delay=delay)
        X = torch.as_tensor(np.asarray(X)) if not isinstance(X, torch.Tensor) else X
        y = torch.as_tensor(np.asarray(y)) if not isinstance(y, torch.Tensor) else y
        if dtype is None:
            dtype = X.dtype if X.is_floating_point() else torch.float64
        if device is None:
            device = X.device
        self._X = X.to(device=device, dtype=dtype).contiguous()
        self._y = y.to(device=device, dtype=dtype).contiguous()
        self._use_kernels = use_kernels
        if self._X.dim() != 2 or self._y.dim() != 1 or self._X.shape[0] != self._y.shape[0]:
            raise ValueError("X must be [N,K] and y [N].")
        self._n, self._k = self._X.shape
        self._k_pad = 0
        if (use_kernels is None or use_kernels) and self._X.is_cuda:
            # pad the feature dim to the fused kernel's lane-slice granule
            # (512 bf16 / 256 f32 per wave64 pass); zero columns change
            # nothing: z is unaffected and their gradient is exactly 0.
            granule = 512 if dtype == torch.bfloat16 else 256
            rem = self._k % granule
            if rem:
                self._k_pad = granule - rem
                pad = torch.zeros(
                    (self._n, self._k_pad), device=self._X.device, dtype=self._X.dtype
                )
                self._X = torch.cat([self._X, pad], dim=1).contiguous()

    @property
    def device(self):
        return self._X.device

    @property
    def n_rows(self) -> int:
        return int(self._n)

    @property
    def n_features(self) -> int:
        return int(self._k)

    @property
    def _k_eff(self) -> int:
        """Feature dim as the kernel sees it (incl. zero padding)."""
        return self._k + self._k_pad

    def _kernel_path(self) -> bool:
        want = self._X.is_cuda if self._use_kernels is None else self._use_kernels
        if want and self._use_kernels is None:
            from ..ops import logistic_kernel_supports

            if not logistic_kernel_supports(self._X.dtype, self._k_eff):
                import warnings

                warnings.warn(
                    f"fused logistic kernel not compiled for (dtype={self._X.dtype}, "
                    f"K={self._k_eff}); using the eager two-matmul path (~2x slower). "
                    f"Supported padded K: up to 2048 (bf16) / 1024 (f32).",
                    stacklevel=3,
                )
                return False
        return want

    @property
    def fused_size(self) -> int:
        """Layout of the fused fp64 output buffer: [logp, grad[K]]."""
        return 1 + int(self._k)

    def logp_grad(
        self, beta, out: Optional[torch.Tensor] = None
    ) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        beta = torch.as_tensor(beta)
        if beta.shape != (self._k,):
            raise ValueError(f"beta must have shape ({self._k},), got {tuple(beta.shape)}.")
        if self._kernel_path():
            from ..ops import logistic_glm_logp_grad

            if self._k_pad:
                beta = torch.cat(
                    [beta.to(torch.float32),
                     torch.zeros(self._k_pad, dtype=torch.float32, device=beta.device)]
                )
                # ``out`` (if given) is sized [1 + k]; the kernel writes
                # [1 + k_eff] -- evaluate into the kernel's own buffer and
                # copy the un-padded slice over.
                logp, grad = logistic_glm_logp_grad(self._X, self._y, beta)
                grad = grad[: self._k]
                if out is not None:
                    out[0] = logp
                    out[1:] = grad
                    return out[0], [out[1:]]
                return logp, [grad]
            logp, grad = logistic_glm_logp_grad(self._X, self._y, beta, out=out)
            return logp, [grad]
        logp, grads = self._logp_grad_eager(beta)
        if out is not None:
            out[0] = logp
            out[1:] = grads[0]
            return out[0], [out[1:]]
        return logp, grads

    def logp_grad_batched(self, theta) -> Tuple[torch.Tensor, torch.Tensor]:
        """Evaluate 16 chains at once: theta[K,16] -> (logp[16], G[K,16]).

        On an MI355X shard this runs the MFMA-batched kernel (one pass over
        X for all 16 proposal vectors); elsewhere an eager batched matmul.
        """
        theta = torch.as_tensor(theta)
        if theta.dim() != 2 or theta.shape[0] != self._k:
            raise ValueError(f"theta must be [{self._k}, B], got {tuple(theta.shape)}")
        if (
            self._kernel_path()
            and self._X.dtype == torch.bfloat16
            and theta.shape[1] == 16
            and self._k_eff in (512, 1024)
        ):
            from ..ops import logistic_glm_logp_grad_batched

            if self._k_pad:
                theta = torch.cat(
                    [theta.to(torch.float32),
                     torch.zeros(self._k_pad, theta.shape[1], dtype=torch.float32,
                                 device=theta.device)]
                )
            logp, G = logistic_glm_logp_grad_batched(self._X, self._y, theta)
            return logp, (G[: self._k] if self._k_pad else G)
        return self._logp_grad_batched_eager(theta)

    def _logp_grad_batched_eager(self, theta: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
        X, y = self._X[:, : self._k], self._y
        acc_dtype = torch.float64 if X.dtype == torch.float64 else torch.float32
        theta = theta.to(device=X.device, dtype=acc_dtype)
        if X.dtype == torch.bfloat16 and X.is_cuda and X.shape[0] > 1 << 20:
            # large GPU shard: chunk the rows so the f32 upcast of X stays
            # O(chunk) instead of materializing a full second copy of a
            # multi-GB shard per call (this is the fallback path for K or B
            # outside the MFMA kernel's coverage)
            B = theta.shape[1]
            logp = torch.zeros(B, dtype=torch.float64, device=X.device)
            G = torch.zeros((self._k, B), dtype=torch.float64, device=X.device)
            yf_all = y
            step = 1 << 21
            for s in range(0, X.shape[0], step):
                Xf = X[s : s + step].to(acc_dtype)
                yf = yf_all[s : s + step].to(acc_dtype)
                Z = Xf @ theta
                logp += torch.sum(
                    yf[:, None] * Z - torch.nn.functional.softplus(Z),
                    dim=0, dtype=torch.float64,
                )
                R = yf[:, None] - torch.sigmoid(Z)
                G += (Xf.t() @ R).to(torch.float64)
            return logp, G
        Xf = X.to(acc_dtype)
        yf = y.to(acc_dtype)
        Z = Xf @ theta
        logp = torch.sum(
            yf[:, None] * Z - torch.nn.functional.softplus(Z), dim=0, dtype=torch.float64
        )
        R = yf[:, None] - torch.sigmoid(Z)
        G = (Xf.t() @ R).to(torch.float64)
        return logp, G

    def _logp_grad_eager(self, beta: torch.Tensor) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        X, y = self._X[:, : self._k], self._y
        acc_dtype = torch.float64 if X.dtype == torch.float64 else torch.float32
        beta = beta.to(device=X.device, dtype=acc_dtype)
        Xf = X.to(acc_dtype)
        yf = y.to(acc_dtype)
        z = Xf @ beta
        logp = torch.sum(yf * z - torch.nn.functional.softplus(z), dtype=torch.float64)
        resid = yf - torch.sigmoid(z)
        grad = Xf.t() @ resid
        return logp, [grad.to(torch.float64)]
