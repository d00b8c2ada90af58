"""MAP estimation by Adam ascent on a LogpGradFunc.

The analog of the reference demo's ``pm.find_MAP`` step (reference
demo_model.py:38) driven through this framework's LogpGradFunc contract.
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import numpy as np

from ..signatures import LogpGradFunc

__all__ = ["find_map"]


def find_map(
    logp_grad_func: LogpGradFunc,
    init: Sequence[np.ndarray],
    *,
    steps: int = 200,
    lr: float = 0.05,
    beta1: float = 0.9,
    beta2: float = 0.999,
    eps: float = 1e-8,
    tol: float = 1e-8,
) -> Tuple[List[np.ndarray], float]:
    """Gradient-ascent MAP estimate; the gradients come from the (possibly
    remote, federated) logp+grad function itself -- no autodiff here.

    Returns (theta_hat, logp_at_theta_hat).
    """
    theta = [np.array(t, dtype=np.float64) for t in init]
    m = [np.zeros_like(t) for t in theta]
    v = [np.zeros_like(t) for t in theta]
    last_logp = -np.inf
    for t_step in range(1, steps + 1):
        logp, grads = logp_grad_func(*theta)
        logp = float(logp)
        for i, g in enumerate(grads):
            g = np.asarray(g, dtype=np.float64)
            m[i] = beta1 * m[i] + (1 - beta1) * g
            v[i] = beta2 * v[i] + (1 - beta2) * g * g
            mhat = m[i] / (1 - beta1**t_step)
            vhat = v[i] / (1 - beta2**t_step)
            theta[i] = theta[i] + lr * mhat / (np.sqrt(vhat) + eps)
        if abs(logp - last_logp) < tol * max(1.0, abs(logp)):
            break
        last_logp = logp
    logp, _ = logp_grad_func(*theta)
    return theta, float(logp)
