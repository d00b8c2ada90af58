"""ODE logp model with adjoint gradients (BASELINE.json config 5).

``[theta] -> trajectories -> Gaussian log-likelihood`` with the gradient
computed by the discrete adjoint method: one RK4 forward sweep storing only
the step states, then a backward sweep propagating the adjoint state via
per-step vector-Jacobian products (torch.autograd re-derives each step's
local Jacobian action; memory is O(n_steps * state), never O(n_steps *
graph)).  The discrete adjoint is exact for the discretized system, so
federated workers (each owning its private experiments batch) return
gradients that sum exactly across shards -- same identity the linear/GLM
models exploit.  Like the reference's worker blackbox (reference
demo_node.py:30-43), the model is served behind the ComputeFunc contract;
only the compute inside changed.

Batching: each worker integrates a whole batch ``u[B, D]`` of experiments
in one vectorized sweep -- on an MI355X the RK4 right-hand side evaluates
as batched torch tensor ops on the GPU.
"""
from __future__ import annotations

import math
from typing import Callable, List, Optional, Tuple

import numpy as np
import torch

from .base import LogpGradModel

__all__ = ["ODEModel", "PolynomialRHS", "lotka_volterra_rhs", "generate_ode_dataset"]


def lotka_volterra_rhs(t, u: torch.Tensor, theta: torch.Tensor) -> torch.Tensor:
    """Classic 2-species Lotka-Volterra; u=[B,2], theta=[alpha,beta,gamma,delta]."""
    prey, pred = u[..., 0], u[..., 1]
    alpha, beta, gamma, delta = theta[0], theta[1], theta[2], theta[3]
    dprey = alpha * prey - beta * prey * pred
    dpred = delta * prey * pred - gamma * pred
    return torch.stack([dprey, dpred], dim=-1)


class PolynomialRHS:
    """Coefficient-table polynomial vector field:

        du_d/dt = sum_t  c_t * theta_{j_t} * prod_i u_i^{e_ti}

    (``j_t = -1`` drops the theta factor.)  One native CDNA4 kernel
    (ops/csrc/ode_poly.hip) interprets the table, so any model of this
    family -- Lotka-Volterra, SIR, oscillators, mass-action kinetics --
    gets the in-register forward+adjoint path WITHOUT recompilation;
    ``__call__`` evaluates the same field in torch for the eager/autograd
    path and the golden tests.  Limits for the native path: D<=4 states,
    P<=8 thetas, <=16 terms, integer exponents 0..15.
    """

    def __init__(self, terms, D: int, P: int):
        self.terms = [
            (int(d), int(j), float(c), tuple(int(v) for v in e))
            for (d, j, c, e) in terms
        ]
        self.D = int(D)
        self.P = int(P)
        for d, j, c, e in self.terms:
            if not (0 <= d < self.D) or not (-1 <= j < self.P):
                raise ValueError(f"bad term (d={d}, j={j}) for D={D}, P={P}")
            if len(e) > self.D:
                raise ValueError(f"exponent tuple {e} longer than D={D}")

    def __call__(self, t, u: torch.Tensor, theta: torch.Tensor) -> torch.Tensor:
        comps = []
        for d in range(self.D):
            acc = None
            for dt, j, c, e in self.terms:
                if dt != d:
                    continue
                mon = None
                for i, ei in enumerate(e):
                    if ei:
                        fac = u[..., i] ** ei if ei > 1 else u[..., i]
                        mon = fac if mon is None else mon * fac
                val = mon * c if mon is not None else u.new_full(u.shape[:-1], c)
                if j >= 0:
                    val = val * theta[j]
                acc = val if acc is None else acc + val
            comps.append(acc if acc is not None else u.new_zeros(u.shape[:-1]))
        return torch.stack(comps, dim=-1)

    def native_ok(self) -> bool:
        return (
            self.D <= 4
            and self.P <= 8
            and len(self.terms) <= 16
            and all(all(0 <= v <= 15 for v in e) for *_x, e in self.terms)
        )

    @classmethod
    def lotka_volterra(cls) -> "PolynomialRHS":
        """The LV field as a table (cross-checks the hand-derived kernel)."""
        return cls(
            terms=[
                (0, 0, 1.0, (1, 0)),   # +alpha * prey
                (0, 1, -1.0, (1, 1)),  # -beta  * prey * pred
                (1, 3, 1.0, (1, 1)),   # +delta * prey * pred
                (1, 2, -1.0, (0, 1)),  # -gamma * pred
            ],
            D=2,
            P=4,
        )

    @classmethod
    def sir(cls) -> "PolynomialRHS":
        """SIR epidemic model: dS=-b S I, dI=b S I - g I, dR=g I."""
        return cls(
            terms=[
                (0, 0, -1.0, (1, 1, 0)),
                (1, 0, 1.0, (1, 1, 0)),
                (1, 1, -1.0, (0, 1, 0)),
                (2, 1, 1.0, (0, 1, 0)),
            ],
            D=3,
            P=2,
        )


def _rk4_step(f, t, u, h, theta):
    k1 = f(t, u, theta)
    k2 = f(t + 0.5 * h, u + 0.5 * h * k1, theta)
    k3 = f(t + 0.5 * h, u + 0.5 * h * k2, theta)
    k4 = f(t + h, u + h * k3, theta)
    return u + (h / 6.0) * (k1 + 2 * k2 + 2 * k3 + k4)


class ODEModel(LogpGradModel):
    """Gaussian LL of observed ODE trajectories, adjoint gradient w.r.t. theta."""

    param_names = ("theta",)

    def __init__(
        self,
        f: Callable,
        u0,
        t0: float,
        t1: float,
        n_steps: int,
        obs_indices,
        y_obs,
        sigma: float,
        *,
        obs_components: Optional[list] = None,
        device=None,
        dtype: torch.dtype = torch.float64,
        delay: Optional[float] = None,
        use_kernels: Optional[bool] = None,
    ) -> None:
        """
        Parameters
        ----------
        f : callable(t, u, theta) -> du/dt
            Vectorized RHS over the experiments batch ``u[B, D]``.
        u0 : tensor [B, D]
            Initial states of this worker's private experiments.
        t0, t1, n_steps
            Fixed-step RK4 grid (h = (t1-t0)/n_steps).
        obs_indices : int array [n_obs]
            Step indices (0..n_steps) at which observations were taken.
        y_obs : tensor [n_obs, B, n_comp]
            Observed values (private data shard).
        sigma : float
            Observation noise scale.
        obs_components : list of int, optional
            Which state components are observed (default: all).
        """
        super().__init__(delay=delay)
        self.f = f
        u0 = torch.as_tensor(np.asarray(u0)) if not isinstance(u0, torch.Tensor) else u0
        y = torch.as_tensor(np.asarray(y_obs)) if not isinstance(y_obs, torch.Tensor) else y_obs
        if device is None:
            device = u0.device
        self._u0 = u0.to(device=device, dtype=dtype)
        self._y = y.to(device=device, dtype=dtype)
        self._t0 = float(t0)
        self._h = (float(t1) - float(t0)) / int(n_steps)
        self._n_steps = int(n_steps)
        self._obs_idx = [int(i) for i in obs_indices]
        if sorted(self._obs_idx) != self._obs_idx:
            raise ValueError("obs_indices must be sorted ascending.")
        self._obs_components = obs_components
        self._sigma = float(sigma)
        self._dtype = dtype
        self._use_kernels = use_kernels
        self._native_state = None  # lazy (obs_of_step, states_ws, out)

    def _native_kind(self):
        """Which native CDNA4 forward+adjoint path applies: the hand-derived
        Lotka-Volterra kernels ("lv"), the coefficient-table polynomial-RHS
        kernels ("poly", any PolynomialRHS within the table limits), or None
        (torch eager adjoint sweep)."""
        want = self._u0.is_cuda if self._use_kernels is None else self._use_kernels
        if not (
            want
            and self._u0.is_cuda
            and self._obs_components is None
            and self._dtype == torch.float64
        ):
            return None
        if self.f is lotka_volterra_rhs and self._u0.shape[-1] == 2:
            return "lv"
        if (
            isinstance(self.f, PolynomialRHS)
            and self.f.native_ok()
            and self._u0.shape[-1] == self.f.D
        ):
            return "poly"
        return None

    def _native_path(self) -> bool:
        return self._native_kind() is not None

    def _poly_native(self, theta_cp: torch.Tensor):
        """Generic polynomial-RHS native eval: theta_cp [C, P] -> out [C, 1+P]."""
        import math as _math

        from ..ops import ode_poly_logp_grad

        rhs: PolynomialRHS = self.f
        B, D = self._u0.shape
        C = theta_cp.shape[0]
        if self._native_state is None:
            obs_of_step = torch.full((self._n_steps + 1,), -1, dtype=torch.int32)
            for j, idx in enumerate(self._obs_idx):
                obs_of_step[idx] = j
            self._native_state = (obs_of_step.to(self._u0.device), None, None)
        obs_of_step, _, _ = self._native_state
        ws = getattr(self, "_poly_ws", None)
        need = C * (self._n_steps + 1) * B * D
        if ws is None or ws.numel() < need:
            ws = torch.empty(need, dtype=torch.float64, device=self._u0.device)
            self._poly_ws = ws
        out = ode_poly_logp_grad(
            rhs.terms, D, rhs.P, self._u0, self._y, obs_of_step,
            self._n_steps, self._h, self._sigma, theta_cp, ws,
        )
        n_vals = self._y.numel()
        logp_const = -0.5 * n_vals * _math.log(2.0 * _math.pi * self._sigma**2)
        return out, logp_const

    def _logp_grad_native(self, theta: torch.Tensor):
        import math as _math

        if self._native_kind() == "poly":
            theta_cp = theta.detach().to(torch.float64).reshape(1, self.f.P)
            out, logp_const = self._poly_native(theta_cp)
            return out[0, 0] + logp_const, [out[0, 1:]]

        from ..ops import ode_lv_logp_grad

        if self._native_state is None:
            obs_of_step = torch.full((self._n_steps + 1,), -1, dtype=torch.int32)
            for j, idx in enumerate(self._obs_idx):
                obs_of_step[idx] = j
            B = self._u0.shape[0]
            self._native_state = (
                obs_of_step.to(self._u0.device),
                torch.empty((self._n_steps + 1) * B * 2, dtype=torch.float64, device=self._u0.device),
                torch.empty(5, dtype=torch.float64, device=self._u0.device),
            )
        obs_of_step, states_ws, out = self._native_state
        ode_lv_logp_grad(
            self._u0, self._y, obs_of_step, self._n_steps, self._h, self._sigma,
            theta, states_ws, out=out,
        )
        n_vals = self._y.numel()
        logp_const = -0.5 * n_vals * _math.log(2.0 * _math.pi * self._sigma**2)
        return out[0] + logp_const, [out[1:5]]

    @property
    def device(self):
        return self._u0.device

    def _project(self, u: torch.Tensor) -> torch.Tensor:
        if self._obs_components is None:
            return u
        return u[..., self._obs_components]

    def _forward_states(self, theta: torch.Tensor) -> List[torch.Tensor]:
        states = [self._u0]
        u = self._u0
        with torch.no_grad():
            for k in range(self._n_steps):
                t = self._t0 + k * self._h
                u = _rk4_step(self.f, t, u, self._h, theta)
                states.append(u)
        return states

    def logp_grad_batched(self, theta_c) -> Tuple[torch.Tensor, torch.Tensor]:
        """Evaluate C chains at once: theta[4, C] -> (logp[C], G[4, C]).

        Native path integrates B*C (experiment, chain) trajectories in one
        kernel sweep -- the single-theta kernel at B=1024 occupies only 4
        CUs, so extra chains are nearly free.  Fallback loops chains.
        """
        import math as _math

        theta_c = torch.as_tensor(theta_c)
        n_par = self.f.P if isinstance(self.f, PolynomialRHS) else 4
        if theta_c.dim() != 2 or theta_c.shape[0] != n_par:
            raise ValueError(f"theta must be [{n_par}, C], got {tuple(theta_c.shape)}")
        C = theta_c.shape[1]
        if self._native_kind() == "poly":
            out, logp_const = self._poly_native(
                theta_c.t().contiguous().to(device=self._u0.device, dtype=torch.float64)
            )
            return out[:, 0] + logp_const, out[:, 1:].t().contiguous()
        if self._native_kind() == "lv":
            from ..ops import ode_lv_logp_grad_batched

            if self._native_state is None:
                self._logp_grad_native(torch.zeros(4, dtype=torch.float64))  # init
            obs_of_step, _, _ = self._native_state
            B = self._u0.shape[0]
            key = "_batched_ws"
            ws = getattr(self, key, None)
            if ws is None or ws.numel() < C * (self._n_steps + 1) * B * 2:
                ws = torch.empty(C * (self._n_steps + 1) * B * 2,
                                 dtype=torch.float64, device=self._u0.device)
                setattr(self, key, ws)
            res = ode_lv_logp_grad_batched(
                self._u0, self._y, obs_of_step, self._n_steps, self._h,
                self._sigma, theta_c.t().contiguous().to(torch.float64), ws,
            )
            n_vals = self._y.numel()
            logp_const = -0.5 * n_vals * _math.log(2.0 * _math.pi * self._sigma**2)
            return res[:, 0] + logp_const, res[:, 1:5].t().contiguous()
        logps = []
        grads = []
        for c in range(C):
            logp, (g,) = self.logp_grad(theta_c[:, c])
            logps.append(torch.tensor(float(logp), dtype=torch.float64))
            grads.append(torch.as_tensor(g, dtype=torch.float64).reshape(n_par))
        return torch.stack(logps), torch.stack(grads, dim=1)

    def logp_grad(self, theta) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        theta = torch.as_tensor(theta).to(device=self.device, dtype=self._dtype)
        if self._native_path():
            return self._logp_grad_native(theta)
        states = self._forward_states(theta)

        sig2 = self._sigma * self._sigma
        n_obs_values = 0
        logp = torch.zeros((), dtype=torch.float64, device=self.device)
        # dL/du at each observation (fed into the adjoint when the backward
        # sweep passes that step index)
        obs_grad = {}
        for j, idx in enumerate(self._obs_idx):
            pred = self._project(states[idx])
            resid = self._y[j] - pred
            n_obs_values += resid.numel()
            logp = logp - torch.sum(resid * resid, dtype=torch.float64) / (2.0 * sig2)
            g = torch.zeros_like(states[idx])
            if self._obs_components is None:
                g += resid / sig2
            else:
                g[..., self._obs_components] = resid / sig2
            obs_grad[idx] = g
        logp = logp - 0.5 * n_obs_values * math.log(2.0 * math.pi * sig2)

        # ---- discrete adjoint backward sweep -------------------------
        lam = torch.zeros_like(self._u0)
        g_theta = torch.zeros_like(theta)
        if self._n_steps in obs_grad:
            lam = lam + obs_grad[self._n_steps]
        for k in range(self._n_steps - 1, -1, -1):
            t = self._t0 + k * self._h
            u_k = states[k].detach().requires_grad_(True)
            theta_k = theta.detach().requires_grad_(True)
            with torch.enable_grad():
                u_next = _rk4_step(self.f, t, u_k, self._h, theta_k)
                gu, gth = torch.autograd.grad(
                    u_next, (u_k, theta_k), grad_outputs=lam, allow_unused=True
                )
            if gu is None:  # constant field: no state dependence
                gu = torch.zeros_like(u_k)
            if gth is None:  # theta-free field (e.g. fixed dynamics)
                gth = torch.zeros_like(theta_k)
            g_theta = g_theta + gth
            lam = gu
            if k in obs_grad:
                lam = lam + obs_grad[k]
        return logp, [g_theta.to(torch.float64)]


def generate_ode_dataset(
    n_experiments: int = 8,
    n_obs: int = 20,
    n_steps: int = 100,
    t1: float = 10.0,
    sigma: float = 0.1,
    theta=(0.8, 0.3, 0.6, 0.2),
    seed: int = 0,
):
    """Synthetic Lotka-Volterra observations for one worker's shard."""
    rng = np.random.RandomState(seed)
    u0 = 1.0 + rng.uniform(0.0, 1.0, size=(n_experiments, 2))
    theta_t = torch.as_tensor(np.asarray(theta, dtype=np.float64))
    u = torch.as_tensor(u0)
    h = t1 / n_steps
    obs_idx = np.linspace(1, n_steps, n_obs).astype(int).tolist()
    states = [u]
    for k in range(n_steps):
        u = _rk4_step(lotka_volterra_rhs, k * h, u, h, theta_t)
        states.append(u)
    y = np.stack([states[i].numpy() for i in obs_idx])
    y += rng.normal(scale=sigma, size=y.shape)
    return u0, obs_idx, y
