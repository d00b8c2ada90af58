"""Model-family numerics on CPU (golden-equivalence pattern).

GPU variants of the same checks (HIP kernel vs these references) live in
test_gpu.py.
"""
import math

import numpy as np
import pytest
import torch

from pytensor_federated_amd.models import (
    GaussianLinearModel,
    LogisticGLMModel,
    ODEModel,
    generate_linear_dataset,
    generate_logistic_dataset,
)
from pytensor_federated_amd.models.ode import generate_ode_dataset, lotka_volterra_rhs


def reference_linear_dataset():
    """The reference's anchor dataset (test_wrapper_ops.py:55-65)."""
    rng = np.random.RandomState(42)
    x = np.linspace(-3, 3, 15, dtype=float)
    y = rng.normal(2 * x + 0.5, scale=0.1)
    return x, y


class TestGaussianLinear:
    def test_reference_logp_anchor(self):
        # exact spot value from the reference test suite
        # (reference test_wrapper_ops.py:94)
        x, y = reference_linear_dataset()
        model = GaussianLinearModel(x, y, sigma=0.1)
        logp, grads = model(0.4, 1.2)
        np.testing.assert_allclose(logp, -1511.41423640139)
        assert len(grads) == 2

    def test_grads_match_finite_differences(self):
        x, y = generate_linear_dataset(50, seed=1)
        model = GaussianLinearModel(x, y, sigma=0.4)
        a, b = 1.1, 0.7
        logp, (ga, gb) = model(a, b)
        eps = 1e-6
        fd_a = (model(a + eps, b)[0] - logp) / eps
        fd_b = (model(a, b + eps)[0] - logp) / eps
        np.testing.assert_allclose(ga, fd_a, rtol=1e-3)
        np.testing.assert_allclose(gb, fd_b, rtol=1e-3)

    def test_grads_match_torch_autograd(self):
        x, y = generate_linear_dataset(100, seed=2)
        model = GaussianLinearModel(x, y, sigma=0.3)
        a = torch.tensor(0.9, dtype=torch.float64, requires_grad=True)
        b = torch.tensor(0.4, dtype=torch.float64, requires_grad=True)
        xt = torch.as_tensor(x)
        yt = torch.as_tensor(y)
        r = yt - (a + b * xt)
        logp_ref = -0.5 * len(x) * math.log(2 * math.pi * 0.09) - (r * r).sum() / (2 * 0.09)
        logp_ref.backward()
        logp, (ga, gb) = model(0.9, 0.4)
        np.testing.assert_allclose(logp, logp_ref.item(), rtol=1e-12)
        np.testing.assert_allclose(ga, a.grad.item(), rtol=1e-9)
        np.testing.assert_allclose(gb, b.grad.item(), rtol=1e-9)

    def test_shard_sum_equals_whole(self):
        # federated identity: sum of shard logps/grads == whole-data values
        x, y = generate_linear_dataset(101, seed=3)
        whole = GaussianLinearModel(x, y, sigma=0.4)
        logp_w, grads_w = whole(1.0, 0.5)
        # NOTE: the logp constant term is per-shard -n_s/2 log(2 pi s^2),
        # which sums to the whole-data constant -- exactness by construction.
        parts = [GaussianLinearModel(x[s], y[s], sigma=0.4) for s in (slice(0, 33), slice(33, 101))]
        logp_s = sum(p(1.0, 0.5)[0] for p in parts)
        np.testing.assert_allclose(logp_s, logp_w, rtol=1e-12)
        for k in range(2):
            np.testing.assert_allclose(
                sum(p(1.0, 0.5)[1][k] for p in parts), grads_w[k], rtol=1e-10
            )

    def test_delay_shim(self):
        import time

        x, y = generate_linear_dataset(10)
        model = GaussianLinearModel(x, y, sigma=0.4, delay=0.1)
        t0 = time.perf_counter()
        model(1.0, 0.5)
        assert time.perf_counter() - t0 >= 0.1


class TestLogisticGLM:
    def test_matches_torch_autograd(self):
        X, y, beta0 = generate_logistic_dataset(200, 16, seed=4)
        model = LogisticGLMModel(X, y)
        beta = torch.as_tensor(beta0).clone().requires_grad_(True)
        Xt, yt = torch.as_tensor(X), torch.as_tensor(y)
        z = Xt @ beta
        logp_ref = (yt * z - torch.nn.functional.softplus(z)).sum()
        logp_ref.backward()
        logp, (grad,) = model(beta0)
        np.testing.assert_allclose(logp, logp_ref.item(), rtol=1e-10)
        np.testing.assert_allclose(grad, beta.grad.numpy(), rtol=1e-8)

    def test_shard_sum_equals_whole(self):
        X, y, beta0 = generate_logistic_dataset(150, 8, seed=5)
        whole = LogisticGLMModel(X, y)
        logp_w, (grad_w,) = whole(beta0)
        parts = [LogisticGLMModel(X[:70], y[:70]), LogisticGLMModel(X[70:], y[70:])]
        logp_s = sum(p(beta0)[0] for p in parts)
        grad_s = sum(p(beta0)[1][0] for p in parts)
        np.testing.assert_allclose(logp_s, logp_w, rtol=1e-12)
        np.testing.assert_allclose(grad_s, grad_w, rtol=1e-10)

    def test_bad_beta_shape(self):
        X, y, _ = generate_logistic_dataset(10, 4)
        with pytest.raises(ValueError, match="beta must have shape"):
            LogisticGLMModel(X, y).logp_grad(torch.zeros(5))


class TestODE:
    def _model(self, sigma=0.1):
        u0, obs_idx, y = generate_ode_dataset(n_experiments=3, n_obs=8, n_steps=40, t1=6.0, sigma=sigma)
        return ODEModel(
            lotka_volterra_rhs, u0, 0.0, 6.0, 40, obs_idx, y, sigma
        )

    def test_adjoint_matches_backprop_through_solver(self):
        model = self._model()
        theta0 = np.array([0.8, 0.3, 0.6, 0.2])
        logp, (grad,) = model(theta0)

        # reference: direct autograd through the full integration graph
        from pytensor_federated_amd.models.ode import _rk4_step

        theta = torch.as_tensor(theta0).clone().requires_grad_(True)
        u = model._u0.clone()
        states = [u]
        for k in range(model._n_steps):
            u = _rk4_step(model.f, model._t0 + k * model._h, u, model._h, theta)
            states.append(u)
        sig2 = model._sigma ** 2
        logp_ref = torch.zeros((), dtype=torch.float64)
        n_vals = 0
        for j, idx in enumerate(model._obs_idx):
            r = model._y[j] - states[idx]
            n_vals += r.numel()
            logp_ref = logp_ref - (r * r).sum() / (2 * sig2)
        logp_ref = logp_ref - 0.5 * n_vals * math.log(2 * math.pi * sig2)
        logp_ref.backward()

        np.testing.assert_allclose(logp, logp_ref.item(), rtol=1e-10)
        np.testing.assert_allclose(grad, theta.grad.numpy(), rtol=1e-8, atol=1e-10)

    def test_adjoint_matches_finite_differences(self):
        model = self._model()
        theta0 = np.array([0.8, 0.3, 0.6, 0.2])
        logp, (grad,) = model(theta0)
        eps = 1e-5
        for i in range(4):
            hi, lo = theta0.copy(), theta0.copy()
            hi[i] += eps
            lo[i] -= eps
            fd = (model(hi)[0] - model(lo)[0]) / (2 * eps)
            np.testing.assert_allclose(grad[i], fd, rtol=1e-5, atol=1e-6)

    def test_shard_sum_equals_whole(self):
        sigma = 0.1
        u0, obs_idx, y = generate_ode_dataset(n_experiments=4, n_obs=6, n_steps=30, t1=5.0, sigma=sigma)
        theta0 = np.array([0.8, 0.3, 0.6, 0.2])
        whole = ODEModel(lotka_volterra_rhs, u0, 0.0, 5.0, 30, obs_idx, y, sigma)
        logp_w, (grad_w,) = whole(theta0)
        parts = [
            ODEModel(lotka_volterra_rhs, u0[:2], 0.0, 5.0, 30, obs_idx, y[:, :2], sigma),
            ODEModel(lotka_volterra_rhs, u0[2:], 0.0, 5.0, 30, obs_idx, y[:, 2:], sigma),
        ]
        logp_s = sum(p(theta0)[0] for p in parts)
        grad_s = sum(p(theta0)[1][0] for p in parts)
        np.testing.assert_allclose(logp_s, logp_w, rtol=1e-10)
        np.testing.assert_allclose(grad_s, grad_w, rtol=1e-8)
