"""Generic polynomial-RHS ODE family: torch semantics + table validation (CPU).

The GPU counterparts (native kernel vs torch adjoint, poly-vs-LV kernel
equality) live in test_gpu.py; these pin the family's math on CPU.
"""
import numpy as np
import pytest
import torch

from pytensor_federated_amd.models import ODEModel, generate_ode_dataset
from pytensor_federated_amd.models.ode import (
    PolynomialRHS,
    _rk4_step,
    lotka_volterra_rhs,
)


class TestPolynomialRHS:
    def test_lv_table_matches_hand_rhs(self):
        rhs = PolynomialRHS.lotka_volterra()
        rng = np.random.default_rng(0)
        u = torch.as_tensor(rng.uniform(0.1, 3.0, size=(16, 2)))
        theta = torch.as_tensor([0.8, 0.3, 0.6, 0.2], dtype=torch.float64)
        np.testing.assert_allclose(
            rhs(0.0, u, theta).numpy(),
            lotka_volterra_rhs(0.0, u, theta).numpy(),
            rtol=1e-13,
            atol=1e-15,
        )

    def test_sir_conserves_population(self):
        rhs = PolynomialRHS.sir()
        u = torch.tensor([[0.9, 0.1, 0.0], [0.5, 0.3, 0.2]], dtype=torch.float64)
        theta = torch.tensor([1.5, 0.4], dtype=torch.float64)
        du = rhs(0.0, u, theta)
        np.testing.assert_allclose(du.sum(dim=-1).numpy(), 0.0, atol=1e-15)

    def test_autograd_flows_through_table_eval(self):
        rhs = PolynomialRHS.lotka_volterra()
        u = torch.tensor([[1.0, 2.0]], dtype=torch.float64, requires_grad=True)
        theta = torch.tensor([0.8, 0.3, 0.6, 0.2], dtype=torch.float64,
                             requires_grad=True)
        out = rhs(0.0, u, theta).sum()
        gu, gth = torch.autograd.grad(out, (u, theta))
        # d(du0+du1)/dalpha = prey
        np.testing.assert_allclose(float(gth[0]), 1.0)
        # d/dprey = alpha - beta*pred + delta*pred = 0.8 - 0.6 + 0.4
        np.testing.assert_allclose(float(gu[0, 0]), 0.8 - 0.3 * 2 + 0.2 * 2)

    def test_bad_tables_rejected(self):
        with pytest.raises(ValueError):
            PolynomialRHS([(2, 0, 1.0, (1, 0))], D=2, P=1)  # d out of range
        with pytest.raises(ValueError):
            PolynomialRHS([(0, 3, 1.0, (1,))], D=1, P=2)  # theta out of range
        with pytest.raises(ValueError):
            PolynomialRHS([(0, 0, 1.0, (1, 1, 1))], D=2, P=1)  # exponents too long

    def test_native_ok_limits(self):
        assert PolynomialRHS.lotka_volterra().native_ok()
        assert PolynomialRHS.sir().native_ok()


class TestPolyODEModelCPU:
    """Eager-path equivalence: a PolynomialRHS model is the SAME model."""

    def test_poly_lv_model_equals_hand_lv_model(self):
        u0, obs_idx, y = generate_ode_dataset(
            n_experiments=6, n_obs=10, n_steps=40, t1=6.0, sigma=0.1, seed=11
        )
        common = dict(
            u0=u0, t0=0.0, t1=6.0, n_steps=40, obs_indices=obs_idx, y_obs=y,
            sigma=0.1, use_kernels=False,
        )
        m_hand = ODEModel(lotka_volterra_rhs, **common)
        m_poly = ODEModel(PolynomialRHS.lotka_volterra(), **common)
        theta = torch.tensor([0.75, 0.32, 0.55, 0.21], dtype=torch.float64)
        logp_h, (g_h,) = m_hand.logp_grad(theta)
        logp_p, (g_p,) = m_poly.logp_grad(theta)
        np.testing.assert_allclose(float(logp_p), float(logp_h), rtol=1e-13)
        np.testing.assert_allclose(g_p.numpy(), g_h.numpy(), rtol=1e-11)

    def test_sir_adjoint_matches_full_autograd(self):
        """The discrete-adjoint gradient of a NON-LV family member equals
        differentiating straight through the whole unrolled RK4 graph."""
        rhs = PolynomialRHS.sir()
        rng = np.random.default_rng(5)
        B, n_steps = 4, 30
        u0 = np.stack(
            [0.8 + 0.2 * rng.random(B), 0.05 + 0.1 * rng.random(B),
             np.zeros(B)], axis=1
        )
        theta_true = torch.tensor([1.8, 0.5], dtype=torch.float64)
        # simulate observations
        u = torch.as_tensor(u0)
        h = 5.0 / n_steps
        states = [u]
        for k in range(n_steps):
            u = _rk4_step(rhs, k * h, u, h, theta_true)
            states.append(u)
        obs_idx = list(range(5, n_steps + 1, 5))
        y = np.stack([states[i].numpy() for i in obs_idx])
        y += rng.normal(scale=0.02, size=y.shape)

        model = ODEModel(
            rhs, u0, 0.0, 5.0, n_steps, obs_idx, y, sigma=0.02, use_kernels=False
        )
        theta = torch.tensor([1.6, 0.45], dtype=torch.float64)
        logp, (g,) = model.logp_grad(theta)

        # reference: autograd through the full unrolled integration
        th = theta.clone().requires_grad_(True)
        u = torch.as_tensor(u0)
        logp_ref = torch.zeros((), dtype=torch.float64)
        obs = {idx: j for j, idx in enumerate(obs_idx)}
        sig2 = 0.02**2
        yt = torch.as_tensor(y)
        for k in range(n_steps):
            u = _rk4_step(rhs, k * h, u, h, th)
            if k + 1 in obs:
                r = yt[obs[k + 1]] - u
                logp_ref = logp_ref - (r * r).sum() / (2 * sig2)
        logp_ref = logp_ref - 0.5 * yt.numel() * np.log(2 * np.pi * sig2)
        (g_ref,) = torch.autograd.grad(logp_ref, th)
        np.testing.assert_allclose(float(logp), float(logp_ref.detach()), rtol=1e-12)
        np.testing.assert_allclose(g.numpy(), g_ref.numpy(), rtol=1e-10)

    def test_batched_eager_loops_chains(self):
        rhs = PolynomialRHS.sir()
        rng = np.random.default_rng(6)
        u0 = np.stack([0.9 * np.ones(3), 0.1 * np.ones(3), np.zeros(3)], axis=1)
        theta_true = torch.tensor([1.5, 0.4], dtype=torch.float64)
        u = torch.as_tensor(u0)
        h = 0.2
        states = [u]
        for k in range(10):
            u = _rk4_step(rhs, k * h, u, h, theta_true)
            states.append(u)
        obs_idx = [5, 10]
        y = np.stack([states[i].numpy() for i in obs_idx])
        model = ODEModel(rhs, u0, 0.0, 2.0, 10, obs_idx, y, sigma=0.05,
                         use_kernels=False)
        theta_c = torch.tensor([[1.5, 1.4], [0.4, 0.35]], dtype=torch.float64)
        logps, G = model.logp_grad_batched(theta_c)
        for c in range(2):
            logp, (g,) = model.logp_grad(theta_c[:, c])
            np.testing.assert_allclose(float(logps[c]), float(logp), rtol=1e-12)
            np.testing.assert_allclose(G[:, c].numpy(), g.numpy(), rtol=1e-10)


class TestRandomTableProperty:
    """Property: for RANDOM coefficient tables, the discrete-adjoint gradient
    equals autograd through the fully unrolled integration."""

    def test_random_tables_adjoint_matches_autograd(self):
        rng = np.random.default_rng(99)
        for trial in range(6):
            D = int(rng.integers(1, 4))
            P = int(rng.integers(1, 4))
            T = int(rng.integers(1, 7))
            terms = []
            for _ in range(T):
                d = int(rng.integers(0, D))
                j = int(rng.integers(-1, P))
                c = float(rng.normal() * 0.5)
                e = tuple(int(v) for v in rng.integers(0, 3, size=D))
                terms.append((d, j, c, e))
            rhs = PolynomialRHS(terms, D=D, P=P)
            B, n_steps = 3, 12
            h = 0.05  # small step keeps random dynamics bounded
            u0 = rng.uniform(0.5, 1.5, size=(B, D))
            theta = torch.as_tensor(rng.uniform(0.2, 1.0, size=P))

            # observations from a perturbed-theta run
            u = torch.as_tensor(u0)
            states = [u]
            for k in range(n_steps):
                u = _rk4_step(rhs, k * h, u, h, theta)
                states.append(u)
            if not torch.isfinite(states[-1]).all():
                continue  # unstable random system; property vacuous
            obs_idx = [n_steps // 2, n_steps]
            y = np.stack([states[i].numpy() for i in obs_idx])
            y += rng.normal(scale=0.01, size=y.shape)

            model = ODEModel(rhs, u0, 0.0, n_steps * h, n_steps, obs_idx, y,
                             sigma=0.05, use_kernels=False)
            logp, (g,) = model.logp_grad(theta)

            th = theta.clone().requires_grad_(True)
            u = torch.as_tensor(u0)
            logp_ref = torch.zeros((), dtype=torch.float64)
            obs = {idx: jj for jj, idx in enumerate(obs_idx)}
            sig2 = 0.05**2
            yt = torch.as_tensor(y)
            for k in range(n_steps):
                u = _rk4_step(rhs, k * h, u, h, th)
                if k + 1 in obs:
                    r = yt[obs[k + 1]] - u
                    logp_ref = logp_ref - (r * r).sum() / (2 * sig2)
            logp_ref = logp_ref - 0.5 * yt.numel() * np.log(2 * np.pi * sig2)
            if logp_ref.requires_grad:
                (g_ref,) = torch.autograd.grad(logp_ref, th, allow_unused=True)
            else:  # table had no theta terms at all: gradient is zero
                g_ref = None
            if g_ref is None:
                g_ref = torch.zeros_like(th)
            np.testing.assert_allclose(float(logp), float(logp_ref.detach()), rtol=1e-11,
                                       err_msg=f"trial {trial} terms={terms}")
            np.testing.assert_allclose(
                g.numpy(), g_ref.numpy(), rtol=1e-8, atol=1e-10,
                err_msg=f"trial {trial} terms={terms}",
            )
