"""No-U-Turn Sampler (Hoffman & Gelman 2014, Algorithm 6) over a LogpGradFunc.

The reference delegates posterior sampling to PyMC (reference
demo_model.py:38-44 runs ``pm.sample``); this sampler provides that
capability natively over the same fused-call contract.
Each leapfrog step costs exactly ONE fused logp+grad call -- which on this
framework is one HIP kernel + RCCL all-reduce (local engine) or one gRPC
round trip (remote workers).  Dual-averaging step-size adaptation toward a
target acceptance of 0.8, diagonal mass-matrix adaptation from the tuning
window.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np

from ..signatures import LogpGradFunc

__all__ = ["NUTS", "sample_nuts"]

_MAX_DEPTH = 10
_DELTA_MAX = 1000.0


class _Flat:
    """Flatten/unflatten between the multi-array theta and one vector."""

    def __init__(self, init: Sequence[np.ndarray]):
        self.shapes = [np.asarray(t, dtype=np.float64).shape for t in init]
        self.sizes = [int(np.prod(s)) if s else 1 for s in self.shapes]

    def flatten(self, arrays) -> np.ndarray:
        return np.concatenate([np.asarray(a, dtype=np.float64).reshape(-1) for a in arrays])

    def unflatten(self, vec: np.ndarray) -> List[np.ndarray]:
        out, off = [], 0
        for shape, size in zip(self.shapes, self.sizes):
            out.append(vec[off : off + size].reshape(shape))
            off += size
        return out


class NUTS:
    def __init__(
        self,
        logp_grad_func: LogpGradFunc,
        init: Sequence[np.ndarray],
        *,
        step_size: float = 0.1,
        target_accept: float = 0.8,
        seed: Optional[int] = None,
    ) -> None:
        self._func = logp_grad_func
        self._flat = _Flat(init)
        self.q = self._flat.flatten(init)
        self.rng = np.random.default_rng(seed)
        self.step_size = float(step_size)
        self.target_accept = float(target_accept)
        self.inv_mass = np.ones_like(self.q)
        # dual averaging state (Hoffman & Gelman sec 3.2)
        self._mu = np.log(10 * self.step_size)
        self._log_eps_bar = 0.0
        self._h_bar = 0.0
        self._t0 = 10.0
        self._gamma = 0.05
        self._kappa = 0.75
        self._adapt_count = 0
        self.n_divergent = 0
        # dense metric (M^-1 = Sigma, a posterior-covariance estimate);
        # None = diagonal metric via inv_mass
        self.mass_sigma = None
        self._mass_chol = None
        self._logp, self._grad = self._eval(self.q)

    def set_dense_mass(self, sigma: np.ndarray) -> None:
        self.mass_sigma = sigma
        self._mass_chol = np.linalg.cholesky(sigma)

    def _vel(self, p: np.ndarray) -> np.ndarray:
        if self.mass_sigma is None:
            return self.inv_mass * p
        return self.mass_sigma @ p

    def _sample_momentum(self) -> np.ndarray:
        z = self.rng.normal(size=self.q.shape)
        if self.mass_sigma is None:
            return z / np.sqrt(self.inv_mass)
        return np.linalg.solve(self._mass_chol.T, z)  # p ~ N(0, Sigma^-1)

    # -- model evaluation ------------------------------------------------
    def _eval(self, q: np.ndarray) -> Tuple[float, np.ndarray]:
        logp, grads = self._func(*self._flat.unflatten(q))
        return float(logp), self._flat.flatten(grads)

    def _leapfrog(self, q, p, grad, eps):
        p = p + 0.5 * eps * grad
        if self.mass_sigma is None:
            q = q + eps * self.inv_mass * p
        else:
            q = q + eps * (self.mass_sigma @ p)
        logp, grad = self._eval(q)
        p = p + 0.5 * eps * grad
        return q, p, logp, grad

    def _kinetic(self, p: np.ndarray) -> float:
        if self.mass_sigma is None:
            return 0.5 * float(np.sum(self.inv_mass * p * p))
        return 0.5 * float(p @ self._vel(p))

    # -- one NUTS transition ----------------------------------------------
    def step(self) -> List[np.ndarray]:
        q0, logp0, grad0 = self.q, self._logp, self._grad
        p0 = self._sample_momentum()
        joint0 = logp0 - self._kinetic(p0)
        log_u = joint0 + np.log(self.rng.uniform())

        q_minus = q_plus = q0
        p_minus = p_plus = p0
        grad_minus = grad_plus = grad0
        q_new, logp_new, grad_new = q0, logp0, grad0
        j, n, s = 0, 1, True
        alpha_sum, n_alpha = 0.0, 0

        while s and j < _MAX_DEPTH:
            v = 1 if self.rng.uniform() < 0.5 else -1
            if v == -1:
                (q_minus, p_minus, grad_minus, _, _, _, q_prop, logp_prop, grad_prop,
                 n_prime, s_prime, a, na) = self._build_tree(
                    q_minus, p_minus, grad_minus, log_u, v, j, joint0)
            else:
                (_, _, _, q_plus, p_plus, grad_plus, q_prop, logp_prop, grad_prop,
                 n_prime, s_prime, a, na) = self._build_tree(
                    q_plus, p_plus, grad_plus, log_u, v, j, joint0)
            if s_prime and self.rng.uniform() < n_prime / max(n, 1):
                q_new, logp_new, grad_new = q_prop, logp_prop, grad_prop
            n += n_prime
            alpha_sum += a
            n_alpha += na
            dq = q_plus - q_minus
            s = s_prime and (dq @ self._vel(p_minus) >= 0) and (
                dq @ self._vel(p_plus) >= 0
            )
            j += 1

        self.q, self._logp, self._grad = q_new, logp_new, grad_new
        self._last_accept_stat = alpha_sum / max(n_alpha, 1)
        return self._flat.unflatten(self.q.copy())

    def _build_tree(self, q, p, grad, log_u, v, j, joint0):
        if j == 0:
            q1, p1, logp1, grad1 = self._leapfrog(q, p, grad, v * self.step_size)
            joint = logp1 - self._kinetic(p1)
            if not np.isfinite(joint):
                joint = -np.inf  # out of support / overflow: reject cleanly
            n_prime = 1 if log_u <= joint else 0
            s_prime = log_u < joint + _DELTA_MAX
            if not s_prime:
                self.n_divergent += 1
            alpha = min(1.0, np.exp(min(joint - joint0, 0.0)))
            return (q1, p1, grad1, q1, p1, grad1, q1, logp1, grad1,
                    n_prime, s_prime, alpha, 1)
        (q_minus, p_minus, grad_minus, q_plus, p_plus, grad_plus, q_prop,
         logp_prop, grad_prop, n_prime, s_prime, a, na) = self._build_tree(
            q, p, grad, log_u, v, j - 1, joint0)
        if s_prime:
            if v == -1:
                (q_minus, p_minus, grad_minus, _, _, _, q_pp, logp_pp, grad_pp,
                 n_pp, s_pp, a2, na2) = self._build_tree(
                    q_minus, p_minus, grad_minus, log_u, v, j - 1, joint0)
            else:
                (_, _, _, q_plus, p_plus, grad_plus, q_pp, logp_pp, grad_pp,
                 n_pp, s_pp, a2, na2) = self._build_tree(
                    q_plus, p_plus, grad_plus, log_u, v, j - 1, joint0)
            if n_pp > 0 and self.rng.uniform() < n_pp / max(n_prime + n_pp, 1):
                q_prop, logp_prop, grad_prop = q_pp, logp_pp, grad_pp
            n_prime += n_pp
            a += a2
            na += na2
            dq = q_plus - q_minus
            s_prime = s_pp and (dq @ self._vel(p_minus) >= 0) and (
                dq @ self._vel(p_plus) >= 0
            )
        return (q_minus, p_minus, grad_minus, q_plus, p_plus, grad_plus,
                q_prop, logp_prop, grad_prop, n_prime, s_prime, a, na)

    # -- adaptation -------------------------------------------------------
    def adapt_step_size(self) -> None:
        self._adapt_count += 1
        m = self._adapt_count
        a_stat = getattr(self, "_last_accept_stat", 1.0)
        eta = 1.0 / (m + self._t0)
        self._h_bar = (1 - eta) * self._h_bar + eta * (self.target_accept - a_stat)
        log_eps = self._mu - np.sqrt(m) / self._gamma * self._h_bar
        # bound the early-iteration overshoot (dual averaging can run away
        # double-exponentially while the acceptance statistic saturates)
        log_eps = float(np.clip(log_eps, self._mu - 20.0, self._mu + 20.0))
        w = m ** (-self._kappa)
        self._log_eps_bar = w * log_eps + (1 - w) * self._log_eps_bar
        self.step_size = float(np.exp(log_eps))

    def freeze_step_size(self) -> None:
        self.step_size = float(np.exp(self._log_eps_bar))

    def reset_step_size_adaptation(self, step_size: float) -> None:
        """Restart dual averaging anchored at a fresh step size.

        MUST be called when the mass matrix changes: eps is measured in
        metric-whitened units, so a metric update invalidates both the
        current eps and the averaging anchor ``mu`` (Stan restarts its
        step-size adaptation at every metric-window boundary for the same
        reason)."""
        self.step_size = float(step_size)
        self._mu = np.log(10 * self.step_size)
        self._log_eps_bar = 0.0
        self._h_bar = 0.0
        self._adapt_count = 0


def sample_nuts(
    logp_grad_func: LogpGradFunc,
    init: Sequence[np.ndarray],
    *,
    draws: int = 1000,
    tune: int = 500,
    step_size: float = 0.1,
    target_accept: float = 0.8,
    seed: Optional[int] = None,
    adapt_mass: bool = True,
    mass: str = "diag",
) -> List[List[np.ndarray]]:
    """NUTS with dual-averaging step size + mass-matrix adaptation.

    ``mass="diag"`` (default) adapts a diagonal metric; ``mass="dense"``
    adapts a full covariance metric from the tuning window (required for
    strongly correlated posteriors).  Step-size adaptation restarts after
    the metric update (eps lives in metric-whitened units).

    Returns ``draws`` samples, each a list of parameter arrays.
    """
    sampler = NUTS(
        logp_grad_func, init, step_size=step_size, target_accept=target_accept, seed=seed
    )
    window: List[np.ndarray] = []
    for i in range(tune):
        sampler.step()
        sampler.adapt_step_size()
        if adapt_mass:
            window.append(sampler.q.copy())
            # one mass update mid-tune, then keep adapting step size
            if i == int(tune * 0.6) and len(window) > 10:
                w = np.stack(window[len(window) // 2 :])
                if mass == "dense":
                    n_w = w.shape[0]
                    cov = np.atleast_2d(np.cov(w.T))
                    lam = n_w / (n_w + 5.0)
                    diag = np.diag(np.maximum(np.diag(cov), 1e-12))
                    sampler.set_dense_mass(lam * cov + (1 - lam) * diag
                                           + 1e-12 * np.eye(cov.shape[0]))
                else:
                    var = np.var(w, axis=0)
                    sampler.inv_mass = np.maximum(var, 1e-10)
                window.clear()
                # eps units changed with the metric: restart dual averaging
                # at a conservative whitened-units step
                sampler.reset_step_size_adaptation(0.25)
    if tune > 0:
        # tune == 0 keeps the caller's step_size (freezing would reset it
        # to exp(log_eps_bar) == 1.0 with no adaptation history)
        sampler.freeze_step_size()
    return [sampler.step() for _ in range(draws)]
