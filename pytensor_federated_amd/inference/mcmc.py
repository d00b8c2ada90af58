"""Random-walk Metropolis with tuned proposal scale.

Gradient-free: works against a plain LogpFunc (the reference's
non-differentiable ``LogpOp`` use case, wrapper_ops.py:44-81 +
test_wrapper_ops.py:291-317).
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np

from ..signatures import LogpFunc

__all__ = ["Metropolis", "sample_metropolis"]


class Metropolis:
    def __init__(
        self,
        logp_func: LogpFunc,
        init: Sequence[np.ndarray],
        *,
        scale: float = 0.1,
        seed: Optional[int] = None,
    ) -> None:
        self._logp = logp_func
        self.theta = [np.array(t, dtype=np.float64) for t in init]
        self._sizes = [t.size for t in self.theta]
        self.scale = float(scale)
        self.rng = np.random.default_rng(seed)
        self.current_logp = float(self._logp(*self.theta))
        self.n_accepted = 0
        self.n_steps = 0

    def step(self) -> List[np.ndarray]:
        prop = [
            t + self.rng.normal(scale=self.scale, size=t.shape) for t in self.theta
        ]
        lp = float(self._logp(*prop))
        self.n_steps += 1
        if np.log(self.rng.uniform()) < lp - self.current_logp:
            self.theta = prop
            self.current_logp = lp
            self.n_accepted += 1
        return [t.copy() for t in self.theta]

    def tune(self, interval_accept_rate: float) -> None:
        """PyMC-style proposal-scale tuning toward ~0.3 acceptance."""
        if interval_accept_rate < 0.05:
            self.scale *= 0.5
        elif interval_accept_rate < 0.2:
            self.scale *= 0.9
        elif interval_accept_rate > 0.95:
            self.scale *= 10.0
        elif interval_accept_rate > 0.75:
            self.scale *= 2.0
        elif interval_accept_rate > 0.5:
            self.scale *= 1.1


def sample_metropolis(
    logp_func: LogpFunc,
    init: Sequence[np.ndarray],
    *,
    draws: int = 1000,
    tune: int = 500,
    scale: float = 0.1,
    seed: Optional[int] = None,
    tune_interval: int = 100,
) -> List[np.ndarray]:
    """Returns a list of draws; each draw is the list of parameter arrays."""
    sampler = Metropolis(logp_func, init, scale=scale, seed=seed)
    accepted_at_interval = 0
    for i in range(tune):
        sampler.step()
        if (i + 1) % tune_interval == 0:
            rate = (sampler.n_accepted - accepted_at_interval) / tune_interval
            sampler.tune(rate)
            accepted_at_interval = sampler.n_accepted
    chain = [sampler.step() for _ in range(draws)]
    return chain
