"""Inference drivers: MAP, Metropolis, NUTS against closed-form posteriors."""
import numpy as np
import pytest

from pytensor_federated_amd.inference import find_map, sample_metropolis, sample_nuts
from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset


def gaussian_2d_logp_grad(mu, cov_inv):
    """N(mu, cov) as a LogpGradFunc over one 2-vector parameter."""

    def func(theta):
        d = np.asarray(theta, dtype=np.float64) - mu
        g = -cov_inv @ d
        return np.asarray(-0.5 * d @ cov_inv @ d), [g]

    return func


class TestFindMap:
    def test_quadratic_max(self):
        mu = np.array([1.0, -2.0])
        cov_inv = np.array([[2.0, 0.3], [0.3, 1.0]])
        theta, logp = find_map(
            gaussian_2d_logp_grad(mu, cov_inv), [np.zeros(2)], steps=800, lr=0.1
        )
        np.testing.assert_allclose(theta[0], mu, atol=1e-3)
        assert logp == pytest.approx(0.0, abs=1e-5)

    def test_linear_model_map_recovers_ols(self):
        x, y = generate_linear_dataset(200, seed=41)
        model = GaussianLinearModel(x, y, sigma=0.4)
        theta, _ = find_map(model.as_logp_grad_func(), [np.array(0.0), np.array(0.0)],
                            steps=2000, lr=0.05)
        # flat-prior MAP == OLS estimate
        A = np.stack([np.ones_like(x), x], axis=1)
        beta_ols, *_ = np.linalg.lstsq(A, y, rcond=None)
        np.testing.assert_allclose([float(theta[0]), float(theta[1])], beta_ols, atol=5e-3)


class TestMetropolis:
    def test_recovers_1d_gaussian(self):
        def logp(theta):
            t = float(theta)
            return np.asarray(-0.5 * (t - 3.0) ** 2 / 4.0)

        chain = sample_metropolis(
            logp, [np.array(0.0)], draws=4000, tune=1000, seed=0
        )
        samples = np.array([float(d[0]) for d in chain])
        assert abs(samples.mean() - 3.0) < 0.25
        assert abs(samples.std() - 2.0) < 0.35


class TestNUTS:
    def test_recovers_correlated_gaussian(self):
        mu = np.array([1.0, -1.0])
        cov = np.array([[1.0, 0.6], [0.6, 2.0]])
        cov_inv = np.linalg.inv(cov)
        chain = sample_nuts(
            gaussian_2d_logp_grad(mu, cov_inv),
            [np.zeros(2)],
            draws=1500,
            tune=600,
            seed=1,
        )
        samples = np.stack([d[0] for d in chain])
        np.testing.assert_allclose(samples.mean(axis=0), mu, atol=0.15)
        np.testing.assert_allclose(np.cov(samples.T), cov, atol=0.45)

    def test_linear_model_posterior_matches_conjugate(self):
        """Flat-prior linear regression: posterior = N(beta_hat, sig^2 (A^T A)^-1)."""
        x, y = generate_linear_dataset(120, seed=42)
        sigma = 0.4
        model = GaussianLinearModel(x, y, sigma=sigma)

        def logp_grad(theta):
            logp, grads = model.as_logp_grad_func()(theta[0], theta[1])
            return logp, [np.array([float(grads[0]), float(grads[1])])]

        chain = sample_nuts(logp_grad, [np.zeros(2)], draws=1200, tune=600, seed=2)
        samples = np.stack([d[0] for d in chain])

        A = np.stack([np.ones_like(x), x], axis=1)
        beta_hat, *_ = np.linalg.lstsq(A, y, rcond=None)
        post_cov = sigma**2 * np.linalg.inv(A.T @ A)
        post_sd = np.sqrt(np.diag(post_cov))
        for k in range(2):
            assert abs(samples.mean(axis=0)[k] - beta_hat[k]) < 6 * post_sd[k]
            assert abs(samples.std(axis=0)[k] - post_sd[k]) < 0.4 * post_sd[k]

    def test_dense_mass_sequential(self):
        """mass="dense" on a sharp rho=0.99 posterior (sequential sampler)."""
        mu = np.array([0.8, 0.3])
        sd = np.array([1e-3, 3e-4])
        R = np.array([[1.0, 0.99], [0.99, 1.0]])
        prec = np.linalg.inv(np.outer(sd, sd) * R)
        chain = sample_nuts(
            gaussian_2d_logp_grad(mu, prec),
            [mu * 1.005],
            draws=400, tune=300, step_size=5e-4, seed=23, mass="dense",
        )
        samples = np.stack([d[0] for d in chain])
        np.testing.assert_allclose(samples.mean(axis=0), mu, atol=5e-4)
        np.testing.assert_allclose(samples.std(axis=0), sd, rtol=0.3)

    def test_tune_zero_keeps_user_step_size(self):
        """tune=0 must not freeze: exp(log_eps_bar)=1.0 would silently
        replace the caller's step size with no adaptation history."""
        calls = [0]

        def logp_grad(theta):
            calls[0] += 1
            t = np.asarray(theta, dtype=np.float64)
            return np.asarray(-0.5 * t @ t), [-t]

        chain = sample_nuts(
            logp_grad, [np.zeros(2)], draws=20, tune=0, step_size=0.3, seed=24
        )
        assert len(chain) == 20
        assert calls[0] > 20  # it actually sampled

    def test_divergences_rare_on_gaussian(self):
        from pytensor_federated_amd.inference.nuts import NUTS

        sampler = NUTS(gaussian_2d_logp_grad(np.zeros(2), np.eye(2)), [np.zeros(2)], seed=3)
        for _ in range(50):
            sampler.step()
            sampler.adapt_step_size()
        assert sampler.n_divergent == 0


class TestNUTSBatched:
    def test_single_chain_bit_identical_to_sequential(self):
        """C == 1 lockstep NUTS must reproduce sample_nuts draw-for-draw:
        the generator rewrite changes scheduling, not the math or the RNG
        consumption order."""
        from pytensor_federated_amd.inference import sample_nuts_batched

        mu = np.array([1.0, -1.0])
        cov_inv = np.linalg.inv(np.array([[1.0, 0.6], [0.6, 2.0]]))
        single = gaussian_2d_logp_grad(mu, cov_inv)

        def batched(theta):
            logp, (g,) = single(theta[:, 0])
            return np.array([float(logp)]), np.asarray(g).reshape(2, 1)

        chain_ref = sample_nuts(single, [np.zeros(2)], draws=60, tune=40, seed=11)
        ref = np.stack([d[0] for d in chain_ref])
        chain, stats = sample_nuts_batched(
            batched, np.zeros((2, 1)), draws=60, tune=40, seed=11,
            adaptation="simple",  # sequential sample_nuts has no windowed mode
        )
        np.testing.assert_array_equal(chain[:, :, 0], ref)
        assert stats["chains"] == 1

    def test_recovers_gaussian_all_chains(self):
        from pytensor_federated_amd.inference import sample_nuts_batched
        from pytensor_federated_amd.inference.diagnostics import split_rhat

        mu = np.array([1.0, -1.0])
        cov = np.array([[1.0, 0.6], [0.6, 2.0]])
        cov_inv = np.linalg.inv(cov)

        def batched(theta):
            d = theta - mu[:, None]
            logp = -0.5 * np.einsum("kb,kj,jb->b", d, cov_inv, d)
            grad = -cov_inv @ d
            return logp, grad

        C = 6
        rng = np.random.RandomState(12)
        chain, stats = sample_nuts_batched(
            batched, rng.standard_normal((2, C)), draws=600, tune=400, seed=13
        )
        flat = chain.transpose(0, 2, 1).reshape(-1, 2)
        np.testing.assert_allclose(flat.mean(axis=0), mu, atol=0.15)
        np.testing.assert_allclose(np.cov(flat.T), cov, atol=0.45)
        for k in range(2):
            assert split_rhat(chain[:, k, :].T) < 1.05
        # lockstep amortization: C chains' leapfrogs shared far fewer
        # batched rounds than a sequential run would pay
        assert stats["leapfrogs"] > 2.0 * stats["rounds"]
        assert all(d == 0 for d in stats["divergences"])

    def test_dense_mass_on_correlated_posterior(self):
        """Dense (full-covariance) metric on a sharp rho=0.99 posterior:
        converged R-hat and materially higher ESS than the diagonal metric
        buys on the same budget."""
        from pytensor_federated_amd.inference import sample_nuts_batched
        from pytensor_federated_amd.inference.diagnostics import (
            effective_sample_size,
            split_rhat,
        )

        mu = np.array([0.8, 0.3, 0.6, 0.2])
        sd = np.array([1e-3, 3e-4, 1e-3, 3e-4])
        R = np.eye(4)
        R[0, 1] = R[1, 0] = 0.99
        R[2, 3] = R[3, 2] = 0.97
        prec = np.linalg.inv(np.outer(sd, sd) * R)

        def batched(theta):
            d = theta - mu[:, None]
            logp = -0.5 * np.einsum("kb,kj,jb->b", d, prec, d)
            return logp, -prec @ d

        C = 8
        init = np.tile(mu[:, None], (1, C)) * (
            1 + 0.005 * np.random.RandomState(86).standard_normal((4, C))
        )
        chain, stats = sample_nuts_batched(
            batched, init, draws=150, tune=300, step_size=5e-4, seed=87,
            max_depth=8, mass="dense",
        )
        assert max(split_rhat(chain[:, k, :].T) for k in range(4)) < 1.05
        assert min(
            effective_sample_size(chain[:, k, :].T.reshape(-1)) for k in range(4)
        ) > 300
        np.testing.assert_allclose(chain.mean(axis=(0, 2)), mu, atol=5e-4)

    def test_windowed_adaptation_schedule(self):
        """Expanding windows: step-size-only buffers at both ends, doubling
        metric windows in between, remainder absorbed by the last window."""
        from pytensor_federated_amd.inference.nuts_batched import metric_window_ends

        ends = metric_window_ends(400)
        assert ends == [85, 135, 360]
        assert all(e2 > e1 for e1, e2 in zip(ends, ends[1:]))
        # small tune still yields at least one update inside the buffers
        assert metric_window_ends(80) == [55]
        # term buffer is respected: last update leaves room to re-adapt eps
        assert max(metric_window_ends(1000)) <= 1000 - 100

    def test_windowed_adaptation_converges(self):
        from pytensor_federated_amd.inference import sample_nuts_batched
        from pytensor_federated_amd.inference.diagnostics import split_rhat

        mu = np.array([0.8, 0.3, 0.6, 0.2])
        sd = np.array([1e-3, 3e-4, 1e-3, 3e-4])
        R = np.eye(4)
        R[0, 1] = R[1, 0] = 0.99
        R[2, 3] = R[3, 2] = 0.97
        prec = np.linalg.inv(np.outer(sd, sd) * R)

        def batched(theta):
            d = theta - mu[:, None]
            return -0.5 * np.einsum("kb,kj,jb->b", d, prec, d), -prec @ d

        C = 8
        init = np.tile(mu[:, None], (1, C)) * (
            1 + 0.005 * np.random.RandomState(86).standard_normal((4, C))
        )
        chain, stats = sample_nuts_batched(
            batched, init, draws=150, tune=300, step_size=5e-4, seed=88,
            max_depth=8, mass="dense", adaptation="windowed",
        )
        assert max(split_rhat(chain[:, k, :].T) for k in range(4)) < 1.05
        np.testing.assert_allclose(chain.mean(axis=(0, 2)), mu, atol=5e-4)

    def test_batched_linear_model_matches_conjugate(self):
        """Lockstep chains over a model's python-level batched evaluator
        reproduce the closed-form flat-prior posterior."""
        from pytensor_federated_amd.inference import sample_nuts_batched

        x, y = generate_linear_dataset(120, seed=42)
        sigma = 0.4
        model = GaussianLinearModel(x, y, sigma=sigma)
        func = model.as_logp_grad_func()

        def batched(theta):
            logps, grads = [], []
            for c in range(theta.shape[1]):
                logp, (ga, gb) = func(theta[0, c], theta[1, c])
                logps.append(float(logp))
                grads.append([float(ga), float(gb)])
            return np.array(logps), np.array(grads).T

        chain, _ = sample_nuts_batched(
            batched, np.zeros((2, 4)), draws=400, tune=300, seed=14
        )
        flat = chain.transpose(0, 2, 1).reshape(-1, 2)
        A = np.stack([np.ones_like(x), x], axis=1)
        beta_hat, *_ = np.linalg.lstsq(A, y, rcond=None)
        post_sd = np.sqrt(np.diag(sigma**2 * np.linalg.inv(A.T @ A)))
        for k in range(2):
            assert abs(flat.mean(axis=0)[k] - beta_hat[k]) < 6 * post_sd[k]
            assert abs(flat.std(axis=0)[k] - post_sd[k]) < 0.4 * post_sd[k]


class TestConstrainedSupport:
    def test_nuts_rejects_out_of_support_cleanly(self):
        """Hard support boundary (logp = -inf for x <= 0): leapfrog steps
        that cross it must become clean rejections (non-finite Hamiltonian
        guard), never crashes or out-of-support draws."""
        def logp_grad(x):
            v = float(x[0])
            if v <= 0:
                return np.array(-np.inf), [np.zeros(1)]
            # Gamma(3, 1): logp = 2 log x - x
            return np.array(2 * np.log(v) - v), [np.array([2.0 / v - 1.0])]

        chain = sample_nuts(
            logp_grad, [np.array([0.1])], draws=800, tune=400, seed=21
        )
        samples = np.array([float(d[0][0]) for d in chain])
        assert np.all(samples > 0)
        # Gamma(3,1): mean 3, sd sqrt(3)
        assert abs(samples.mean() - 3.0) < 0.5
        assert abs(samples.std() - np.sqrt(3.0)) < 0.6

    def test_nuts_batched_rejects_out_of_support_cleanly(self):
        from pytensor_federated_amd.inference import sample_nuts_batched

        def batched(theta):
            x = theta[0]
            ok = x > 0
            logp = np.where(ok, 2 * np.log(np.where(ok, x, 1.0)) - x, -np.inf)
            grad = np.where(ok, 2.0 / np.where(ok, x, 1.0) - 1.0, 0.0)
            return logp, grad[None, :]

        chain, stats = sample_nuts_batched(
            batched, np.full((1, 4), 0.1), draws=500, tune=300, seed=22
        )
        assert np.all(chain > 0)
        pooled = chain.reshape(-1)
        assert abs(pooled.mean() - 3.0) < 0.5


class TestMALABatched:
    def test_recovers_gaussian_all_chains(self):
        from pytensor_federated_amd.inference import sample_mala_batched

        K, B = 3, 8
        rng = np.random.RandomState(7)
        Ainv = np.linalg.inv(np.diag([1.0, 2.0, 0.5]))
        mu = np.array([1.0, -1.0, 0.5])

        def batched(theta):
            d = theta - mu[:, None]
            logp = -0.5 * np.einsum("kb,kj,jb->b", d, Ainv, d)
            grad = -Ainv @ d
            return logp, grad

        chain, stats = sample_mala_batched(
            batched, rng.standard_normal((K, B)), draws=3000, tune=800,
            step_size=0.1, seed=8,
        )
        assert 0.3 < stats["accept_rate"] < 0.95
        flat = chain.transpose(0, 2, 1).reshape(-1, K)  # pool chains
        np.testing.assert_allclose(flat.mean(axis=0), mu, atol=0.15)
        np.testing.assert_allclose(flat.var(axis=0), [1.0, 2.0, 0.5], rtol=0.3)

    def test_batched_logistic_model_chains_move(self):
        from pytensor_federated_amd.inference import sample_mala_batched
        from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset

        X, y, beta_true = generate_logistic_dataset(400, 8, seed=9)
        model = LogisticGLMModel(X, y)

        def batched(theta):
            logp, G = model.logp_grad_batched(theta)
            return np.asarray(logp), np.asarray(G)

        init = np.zeros((8, 4))
        chain, stats = sample_mala_batched(
            batched, init, draws=800, tune=400, step_size=0.05, seed=10
        )
        post_mean = chain.mean(axis=(0, 2))
        # posterior mean correlates with the truth (weakly informative N=400)
        assert np.corrcoef(post_mean, beta_true)[0, 1] > 0.5


def _serve_linear_worker(port):
    from pytensor_federated_amd.common import wrap_logp_grad_func
    from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
    from pytensor_federated_amd.service import serve_compute_func

    x, y = generate_linear_dataset(80, seed=77)
    model = GaussianLinearModel(x, y, sigma=0.4)
    serve_compute_func(
        wrap_logp_grad_func(model.as_logp_grad_func()), "127.0.0.1", port,
        fast_port=port + 1,
    )


@pytest.mark.timeout(300)
def test_nuts_over_live_worker():
    """End-to-end: NUTS driver -> fast transport -> worker (remote grads).

    The analog of the reference's MCMC-through-live-gRPC test
    (test_wrapper_ops.py:291-317), with this framework's own sampler.
    """
    import multiprocessing
    import socket
    import time as _time

    from pytensor_federated_amd.common import LogpGradServiceClient
    from pytensor_federated_amd.inference import sample_nuts

    port = 9581
    ctx = multiprocessing.get_context("spawn")
    proc = ctx.Process(target=_serve_linear_worker, args=(port,), daemon=True)
    proc.start()
    try:
        deadline = _time.time() + 30
        while _time.time() < deadline:
            try:
                with socket.create_connection(("127.0.0.1", port + 1), timeout=1):
                    break
            except OSError:
                _time.sleep(0.1)
        client = LogpGradServiceClient("127.0.0.1", port + 1, transport="fast")

        def logp_grad(theta):
            logp, grads = client.evaluate(theta[0], theta[1])
            return logp, [np.array([float(grads[0]), float(grads[1])])]

        chain = sample_nuts(logp_grad, [np.zeros(2)], draws=300, tune=200, seed=5)
        samples = np.stack([d[0] for d in chain])
        # ground truth: intercept 1.5, slope 0.5
        assert abs(samples[:, 0].mean() - 1.5) < 0.5
        assert abs(samples[:, 1].mean() - 0.5) < 0.12
        del client
    finally:
        proc.terminate()
        proc.join(timeout=10)


class TestDiagnostics:
    def test_rhat_converged_vs_split_chains(self):
        from pytensor_federated_amd.inference import split_rhat

        rng = np.random.default_rng(11)
        good = rng.standard_normal((4, 2000))
        assert abs(split_rhat(good) - 1.0) < 0.02
        # chains stuck at different means -> R-hat far above 1
        bad = good + np.array([0.0, 0.0, 3.0, 3.0])[:, None]
        assert split_rhat(bad) > 1.5

    def test_ess_iid_vs_autocorrelated(self):
        from pytensor_federated_amd.inference import effective_sample_size

        rng = np.random.default_rng(12)
        iid = rng.standard_normal((2, 4000))
        ess_iid = effective_sample_size(iid)
        assert ess_iid > 0.6 * 8000
        # AR(1) with phi=0.95: ESS ~ N * (1-phi)/(1+phi) ~ 0.026 N
        phi = 0.95
        ar = np.empty((2, 4000))
        for c in range(2):
            e = rng.standard_normal(4000)
            ar[c, 0] = e[0]
            for t in range(1, 4000):
                ar[c, t] = phi * ar[c, t - 1] + np.sqrt(1 - phi**2) * e[t]
        ess_ar = effective_sample_size(ar)
        assert ess_ar < 0.15 * 8000
        assert ess_ar > 20

    def test_summary_formats(self):
        from pytensor_federated_amd.inference import summary

        rng = np.random.default_rng(13)
        text = summary({"intercept": rng.normal(1.5, 0.1, (2, 500)),
                        "slope": rng.normal(0.5, 0.05, (2, 500))})
        assert "intercept" in text and "r_hat" in text
        assert len(text.splitlines()) == 3


class TestLockstepEquivalenceProperty:
    def test_bit_identity_over_random_posteriors(self):
        """Property: for ANY 2-d Gaussian posterior and seed, C=1 lockstep
        NUTS is draw-for-draw identical to the sequential sampler (stronger
        form of the fixed-seed test above)."""
        from pytensor_federated_amd.inference import sample_nuts_batched

        rng = np.random.RandomState(31)
        for trial in range(5):
            mu = rng.standard_normal(2) * 2.0
            a = rng.standard_normal((2, 2))
            cov = a @ a.T + 0.3 * np.eye(2)
            cov_inv = np.linalg.inv(cov)
            single = gaussian_2d_logp_grad(mu, cov_inv)

            def batched(theta, _single=single):
                logp, (g,) = _single(theta[:, 0])
                return np.array([float(logp)]), np.asarray(g).reshape(2, 1)

            seed = int(rng.randint(0, 10_000))
            tune = int(rng.randint(10, 60))
            draws = int(rng.randint(10, 50))
            init = rng.standard_normal(2)
            ref = np.stack([
                d[0] for d in sample_nuts(single, [init.copy()],
                                          draws=draws, tune=tune, seed=seed)
            ])
            chain, _ = sample_nuts_batched(
                batched, init[:, None].copy(), draws=draws, tune=tune, seed=seed,
                adaptation="simple",  # bit-identity vs the sequential sampler
            )
            np.testing.assert_array_equal(chain[:, :, 0], ref,
                                          err_msg=f"trial {trial}")
