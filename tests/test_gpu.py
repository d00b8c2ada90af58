"""GPU (MI355X) tests: HIP kernel numerics vs plain-torch references.

Every kernel is compared against an eager fp32/fp64 torch computation on
the SAME (quantized) data -- the golden-equivalence pattern of SURVEY.md §4
applied at the kernel seam.
"""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from pytensor_federated_amd.models import (
    GaussianLinearModel,
    LogisticGLMModel,
    generate_linear_dataset,
    generate_logistic_dataset,
)
from pytensor_federated_amd.parallel import MultiShardDispatcher


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def test_extension_loads(dev):
    from pytensor_federated_amd.ops import kernels_available, require_kernels

    assert kernels_available()
    require_kernels()


class TestGaussianLinearKernel:
    @pytest.mark.parametrize("dtype,rtol", [
        (torch.float64, 1e-12),
        (torch.float32, 1e-5),
        (torch.bfloat16, 1e-4),
    ])
    @pytest.mark.parametrize("n", [1_000_003, 64])
    def test_matches_eager_same_data(self, dev, dtype, rtol, n):
        x, y = generate_linear_dataset(n, seed=21)
        kernel_model = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=dtype, use_kernels=True)
        eager_model = GaussianLinearModel(
            kernel_model._x, kernel_model._y, sigma=0.4, device=dev, dtype=dtype, use_kernels=False
        )
        logp_k, grads_k = kernel_model(1.3, 0.45)
        logp_e, grads_e = eager_model(1.3, 0.45)
        np.testing.assert_allclose(logp_k, logp_e, rtol=rtol)
        for gk, ge in zip(grads_k, grads_e):
            np.testing.assert_allclose(gk, ge, rtol=rtol, atol=rtol * max(1.0, abs(float(ge))))

    def test_reference_anchor_on_gpu(self, dev):
        rng = np.random.RandomState(42)
        x = np.linspace(-3, 3, 15, dtype=float)
        y = rng.normal(2 * x + 0.5, scale=0.1)
        model = GaussianLinearModel(x, y, sigma=0.1, device=dev, dtype=torch.float64, use_kernels=True)
        logp, _ = model(0.4, 1.2)
        np.testing.assert_allclose(logp, -1511.41423640139, rtol=1e-10)

    def test_fp64_accumulation_quality_large_n(self, dev):
        # bf16 data, 1e7 rows: kernel fp32-lane/fp64-block accumulation vs a
        # chunked float64 reference on the dequantized values
        n = 10_000_000
        x, y = generate_linear_dataset(n, seed=22)
        m = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16, use_kernels=True)
        logp_k, (ga_k, gb_k) = m(1.4, 0.52)
        xf = m._x.double()
        yf = m._y.double()
        r = yf - (1.4 + 0.52 * xf)
        sig2 = 0.16
        logp_ref = -0.5 * n * np.log(2 * np.pi * sig2) - float((r * r).sum()) / (2 * sig2)
        np.testing.assert_allclose(logp_k, logp_ref, rtol=1e-7)
        np.testing.assert_allclose(ga_k, float(r.sum()) / sig2, rtol=1e-6)
        np.testing.assert_allclose(gb_k, float((r * xf).sum()) / sig2, rtol=1e-6)


class TestLogisticKernel:
    @pytest.mark.parametrize("dtype,K,rtol", [
        (torch.bfloat16, 512, 2e-4),
        (torch.bfloat16, 1024, 2e-4),
        (torch.float32, 1024, 1e-5),
    ])
    def test_matches_eager_same_data(self, dev, dtype, K, rtol):
        X, y, beta0 = generate_logistic_dataset(20_000, K, seed=23)
        kernel_model = LogisticGLMModel(X, y, device=dev, dtype=dtype, use_kernels=True)
        eager_model = LogisticGLMModel(
            kernel_model._X, kernel_model._y, device=dev, dtype=dtype, use_kernels=False
        )
        # the kernel consumes beta as f32; feed the eager path the same values
        beta32 = torch.as_tensor(beta0, dtype=torch.float32)
        logp_k, (g_k,) = kernel_model.logp_grad(beta32)
        logp_e, (g_e,) = eager_model.logp_grad(beta32)
        np.testing.assert_allclose(
            float(logp_k), float(logp_e), rtol=rtol
        )
        np.testing.assert_allclose(
            g_k.cpu().numpy(), g_e.cpu().numpy(), rtol=rtol, atol=rtol * 10
        )

    def test_unaligned_k_pads_and_matches_eager(self, dev):
        # K=100 pads to the 512 lane-slice granule; zero columns are inert
        X, y, beta0 = generate_logistic_dataset(5000, 100, seed=24)
        m = LogisticGLMModel(X, y, device=dev, dtype=torch.bfloat16, use_kernels=True)
        assert m._k_pad == 412 and m._X.shape[1] == 512
        beta32 = torch.as_tensor(beta0, dtype=torch.float32)
        logp_k, (g_k,) = m.logp_grad(beta32)
        eager = LogisticGLMModel(
            m._X[:, :100].contiguous(), m._y, device=dev, dtype=torch.bfloat16,
            use_kernels=False,
        )
        logp_e, (g_e,) = eager.logp_grad(beta32)
        assert g_k.shape == (100,)
        np.testing.assert_allclose(float(logp_k), float(logp_e), rtol=2e-4)
        np.testing.assert_allclose(
            g_k.cpu().numpy(), g_e.cpu().numpy(), rtol=2e-4, atol=2e-3
        )


class TestStreamsOnGPU:
    def test_multi_shard_dispatch_equals_whole(self, dev):
        x, y = generate_linear_dataset(2_000_000, seed=25)
        half = 1_000_000
        shards = [
            GaussianLinearModel(x[:half], y[:half], sigma=0.4, device=dev, dtype=torch.bfloat16),
            GaussianLinearModel(x[half:], y[half:], sigma=0.4, device=dev, dtype=torch.bfloat16),
        ]
        disp = MultiShardDispatcher(shards)
        logp_s, grads_s = disp(1.0, 0.5)
        whole = GaussianLinearModel(
            torch.cat([shards[0]._x, shards[1]._x]),
            torch.cat([shards[0]._y, shards[1]._y]),
            sigma=0.4,
            device=dev,
            dtype=torch.bfloat16,
        )
        logp_w, grads_w = whole(1.0, 0.5)
        np.testing.assert_allclose(logp_s, logp_w, rtol=1e-9)
        for gs, gw in zip(grads_s, grads_w):
            np.testing.assert_allclose(gs, gw, rtol=1e-7)


def test_engine_single_rank_gpu(dev):
    from pytensor_federated_amd.parallel import FederatedShardEngine

    x, y = generate_linear_dataset(100_000, seed=26)
    model = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16)
    engine = FederatedShardEngine(model, use_distributed=False)
    logp, grads = engine(1.0, 0.5)
    logp_ref, grads_ref = model(1.0, 0.5)
    np.testing.assert_allclose(logp, logp_ref, rtol=1e-12)


def test_gaussian_sync_eval_matches_async(dev):
    from pytensor_federated_amd.ops import gaussian_linear_eval_sync

    x, y = generate_linear_dataset(1_000_000, seed=31)
    m = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16, use_kernels=True)
    logp_a, (ga_a, gb_a) = m(1.2, 0.6)
    logp_s, ga_s, gb_s = m.logp_grad_sync(1.2, 0.6)
    np.testing.assert_allclose(logp_s, logp_a, rtol=1e-12)
    np.testing.assert_allclose(ga_s, ga_a, rtol=1e-12)
    np.testing.assert_allclose(gb_s, gb_a, rtol=1e-12)


def test_fused_combine_repeated_calls_stable(dev):
    """Stress the in-launch last-arriver combine: 200 repeated calls must
    all return the identical correct result (catches G16 visibility bugs --
    the consumer block is L1-warm across calls)."""
    x, y = generate_linear_dataset(3_000_000, seed=32)
    m = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16, use_kernels=True)
    ref = m.logp_grad_sync(1.1, 0.4)
    for i in range(200):
        got = m.logp_grad_sync(1.1, 0.4)
        assert got == ref, f"call {i}: {got} != {ref}"
    # and the async path agrees
    logp, (ga, gb) = m(1.1, 0.4)
    np.testing.assert_allclose(float(logp), ref[0], rtol=1e-12)


def test_native_multi_shard_engine(dev):
    from pytensor_federated_amd.parallel import NativeMultiShardEngine

    x, y = generate_linear_dataset(4_000_000, seed=33)
    q = 1_000_000
    shards = [
        GaussianLinearModel(x[i * q : (i + 1) * q], y[i * q : (i + 1) * q],
                            sigma=0.4, device=dev, dtype=torch.bfloat16)
        for i in range(4)
    ]
    engine = NativeMultiShardEngine(shards)
    try:
        logp, (ga, gb) = engine(1.0, 0.5)
        whole = GaussianLinearModel(
            torch.cat([s._x for s in shards]), torch.cat([s._y for s in shards]),
            sigma=0.4, device=dev, dtype=torch.bfloat16,
        )
        logp_w, (ga_w, gb_w) = whole(1.0, 0.5)
        np.testing.assert_allclose(float(logp), float(logp_w), rtol=1e-9)
        np.testing.assert_allclose(float(ga), float(ga_w), rtol=1e-7)
        np.testing.assert_allclose(float(gb), float(gb_w), rtol=1e-7)
        # repeated evals stable
        ref = engine.logp_grad_sync(1.0, 0.5)
        for _ in range(50):
            assert engine.logp_grad_sync(1.0, 0.5) == ref
    finally:
        engine.close()


def test_graphed_linear_engine(dev):
    from pytensor_federated_amd.parallel.graphed import GraphedLinearEngine

    x, y = generate_linear_dataset(2_000_000, seed=34)
    m = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16, use_kernels=True)
    eng = GraphedLinearEngine(m, distributed=False)
    for a, b in [(1.0, 0.5), (0.3, -0.2), (2.0, 1.0)]:
        logp_g, ga_g, gb_g = eng.logp_grad_sync(a, b)
        logp_r, ga_r, gb_r = m.logp_grad_sync(a, b)
        np.testing.assert_allclose(logp_g, logp_r, rtol=1e-12)
        np.testing.assert_allclose(ga_g, ga_r, rtol=1e-12)
        np.testing.assert_allclose(gb_g, gb_r, rtol=1e-12)
    # replay stability
    ref = eng.logp_grad_sync(1.0, 0.5)
    for _ in range(100):
        assert eng.logp_grad_sync(1.0, 0.5) == ref


def test_graphed_engine_with_nccl_world1(dev):
    """Capture including the RCCL all_reduce (world_size 1 on this box;
    the 8-GPU version is the same graph on every rank)."""
    import torch.distributed as dist

    if dist.is_initialized():
        pytest.skip("process group already initialized")
    dist.init_process_group(
        "nccl", init_method="tcp://127.0.0.1:29537", rank=0, world_size=1
    )
    try:
        from pytensor_federated_amd.parallel.graphed import GraphedLinearEngine

        x, y = generate_linear_dataset(500_000, seed=35)
        m = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16)
        eng = GraphedLinearEngine(m, distributed=True)
        logp_g, ga_g, gb_g = eng.logp_grad_sync(1.2, 0.4)
        logp_r, ga_r, gb_r = m.logp_grad_sync(1.2, 0.4)
        np.testing.assert_allclose(logp_g, logp_r, rtol=1e-12)
        np.testing.assert_allclose(ga_g, ga_r, rtol=1e-12)
    finally:
        dist.destroy_process_group()


def test_graphed_ode_engine(dev):
    from pytensor_federated_amd.models import ODEModel
    from pytensor_federated_amd.models.ode import generate_ode_dataset, lotka_volterra_rhs
    from pytensor_federated_amd.parallel.graphed import GraphedLogpGradEngine

    u0, obs_idx, y_obs = generate_ode_dataset(n_experiments=32, n_obs=10, n_steps=30, t1=5.0)
    m = ODEModel(lotka_volterra_rhs, u0, 0.0, 5.0, 30, obs_idx, y_obs, 0.1, device=dev)
    eng = GraphedLogpGradEngine(m, (4,), distributed=False)
    for theta in [np.array([0.8, 0.3, 0.6, 0.2]), np.array([0.7, 0.25, 0.65, 0.22])]:
        logp_g, (grad_g,) = eng(theta)
        logp_r, (grad_r,) = m(theta)
        np.testing.assert_allclose(float(logp_g), float(logp_r), rtol=1e-10)
        np.testing.assert_allclose(grad_g, grad_r, rtol=1e-8)


class TestBatchedLogisticMFMA:
    @pytest.mark.parametrize("K", [512, 1024])
    def test_matches_eager_batched(self, dev, K):
        X, y, _ = generate_logistic_dataset(20_000, K, seed=61)
        m = LogisticGLMModel(X, y, device=dev, dtype=torch.bfloat16, use_kernels=True)
        rng = np.random.RandomState(62)
        theta = torch.as_tensor(rng.standard_normal((K, 16)) * 0.3).to(torch.bfloat16)
        # eager reference on the SAME bf16-quantized theta/data
        eager = LogisticGLMModel(m._X, m._y, device=dev, dtype=torch.bfloat16, use_kernels=False)
        logp_e, G_e = eager._logp_grad_batched_eager(theta.float())
        logp_k, G_k = m.logp_grad_batched(theta.float())
        np.testing.assert_allclose(
            logp_k.cpu().numpy(), logp_e.cpu().numpy(), rtol=5e-3
        )
        # G tolerance: the kernel quantizes residuals to bf16 for the MFMA
        ge = G_e.cpu().numpy()
        gk = G_k.cpu().numpy()
        scale = np.abs(ge).max()
        np.testing.assert_allclose(gk, ge, atol=2e-2 * scale, rtol=2e-2)

    def test_batched_consistent_with_single_chain(self, dev):
        K = 512
        X, y, beta0 = generate_logistic_dataset(8_192, K, seed=63)
        m = LogisticGLMModel(X, y, device=dev, dtype=torch.bfloat16, use_kernels=True)
        theta = torch.as_tensor(np.tile(beta0[:, None], (1, 16))).float()
        # quantize like the batched kernel does before comparing
        theta_q = theta.to(torch.bfloat16).float()
        logp_b, G_b = m.logp_grad_batched(theta_q)
        logp_s, (g_s,) = m.logp_grad(theta_q[:, 0].contiguous())
        for b in range(16):
            np.testing.assert_allclose(float(logp_b[b]), float(logp_s), rtol=5e-3)
        scale = float(torch.abs(g_s).max())
        np.testing.assert_allclose(
            G_b[:, 0].cpu().numpy(), g_s.cpu().numpy(), atol=2e-2 * scale, rtol=2e-2
        )

    def test_row_tail_handling(self, dev):
        # N not a multiple of 64: pad rows must not contribute
        K = 512
        X, y, _ = generate_logistic_dataset(1000 + 17, K, seed=64)
        m = LogisticGLMModel(X, y, device=dev, dtype=torch.bfloat16, use_kernels=True)
        theta = torch.randn(K, 16).to(torch.bfloat16).float() * 0.2
        eager = LogisticGLMModel(m._X, m._y, device=dev, dtype=torch.bfloat16, use_kernels=False)
        logp_e, G_e = eager._logp_grad_batched_eager(theta)
        logp_k, G_k = m.logp_grad_batched(theta)
        np.testing.assert_allclose(logp_k.cpu().numpy(), logp_e.cpu().numpy(), rtol=5e-3)
        scale = float(torch.abs(G_e).max())
        np.testing.assert_allclose(
            G_k.cpu().numpy(), G_e.cpu().numpy(), atol=2e-2 * scale, rtol=3e-2
        )


class TestNativeODE:
    def test_native_matches_generic_adjoint(self, dev):
        from pytensor_federated_amd.models import ODEModel
        from pytensor_federated_amd.models.ode import generate_ode_dataset, lotka_volterra_rhs

        u0, obs_idx, y = generate_ode_dataset(n_experiments=64, n_obs=12, n_steps=40, t1=6.0)
        theta0 = np.array([0.8, 0.3, 0.6, 0.2])
        native = ODEModel(lotka_volterra_rhs, u0, 0.0, 6.0, 40, obs_idx, y, 0.1,
                          device=dev, use_kernels=True)
        generic = ODEModel(lotka_volterra_rhs, u0, 0.0, 6.0, 40, obs_idx, y, 0.1,
                           device=dev, use_kernels=False)
        assert native._native_path() and not generic._native_path()
        logp_n, (g_n,) = native(theta0)
        logp_g, (g_g,) = generic(theta0)
        np.testing.assert_allclose(float(logp_n), float(logp_g), rtol=1e-12)
        np.testing.assert_allclose(g_n, g_g, rtol=1e-9)

    def test_poly_kernel_matches_lv_kernel(self, dev):
        """The table-interpreted generic kernel reproduces the hand-derived
        LV kernel bit-for-bit-ish on the LV system (same f64 adjoint)."""
        from pytensor_federated_amd.models import ODEModel
        from pytensor_federated_amd.models.ode import (
            PolynomialRHS,
            generate_ode_dataset,
            lotka_volterra_rhs,
        )

        u0, obs_idx, y = generate_ode_dataset(n_experiments=64, n_obs=12, n_steps=40, t1=6.0)
        theta0 = np.array([0.8, 0.3, 0.6, 0.2])
        common = dict(device=dev, use_kernels=True)
        lv = ODEModel(lotka_volterra_rhs, u0, 0.0, 6.0, 40, obs_idx, y, 0.1, **common)
        poly = ODEModel(PolynomialRHS.lotka_volterra(), u0, 0.0, 6.0, 40, obs_idx, y,
                        0.1, **common)
        assert lv._native_kind() == "lv" and poly._native_kind() == "poly"
        logp_l, (g_l,) = lv(theta0)
        logp_p, (g_p,) = poly(theta0)
        np.testing.assert_allclose(float(logp_p), float(logp_l), rtol=1e-12)
        np.testing.assert_allclose(g_p, g_l, rtol=1e-9)

    def test_poly_kernel_matches_torch_adjoint_sir(self, dev):
        """Generic-RHS native path on a NON-LV family member (SIR) vs the
        torch discrete-adjoint sweep (1e-9)."""
        import torch as _t

        from pytensor_federated_amd.models import ODEModel
        from pytensor_federated_amd.models.ode import PolynomialRHS, _rk4_step

        rhs = PolynomialRHS.sir()
        rng = np.random.RandomState(5)
        B, n_steps = 512, 40
        u0 = np.stack(
            [0.8 + 0.2 * rng.rand(B), 0.05 + 0.1 * rng.rand(B), np.zeros(B)],
            axis=1,
        )
        theta_true = _t.tensor([1.8, 0.5], dtype=_t.float64)
        u = _t.as_tensor(u0)
        h = 5.0 / n_steps
        states = [u]
        for k in range(n_steps):
            u = _rk4_step(rhs, k * h, u, h, theta_true)
            states.append(u)
        obs_idx = list(range(5, n_steps + 1, 5))
        y = np.stack([states[i].numpy() for i in obs_idx])
        y += rng.normal(scale=0.02, size=y.shape)

        native = ODEModel(rhs, u0, 0.0, 5.0, n_steps, obs_idx, y, sigma=0.02,
                          device=dev, use_kernels=True)
        eager = ODEModel(rhs, u0, 0.0, 5.0, n_steps, obs_idx, y, sigma=0.02,
                         device=dev, use_kernels=False)
        assert native._native_kind() == "poly" and eager._native_kind() is None
        theta = np.array([1.6, 0.45])
        logp_n, (g_n,) = native(theta)
        logp_e, (g_e,) = eager(theta)
        np.testing.assert_allclose(float(logp_n), float(logp_e), rtol=1e-12)
        np.testing.assert_allclose(g_n, g_e, rtol=1e-9)

    def test_poly_batched_chains_match_loop(self, dev):
        import torch as _t

        from pytensor_federated_amd.models import ODEModel
        from pytensor_federated_amd.models.ode import (
            PolynomialRHS,
            generate_ode_dataset,
        )

        u0, obs_idx, y = generate_ode_dataset(n_experiments=128, n_obs=10, n_steps=30, t1=5.0)
        poly = ODEModel(PolynomialRHS.lotka_volterra(), u0, 0.0, 5.0, 30, obs_idx, y,
                        0.1, device=dev, use_kernels=True)
        rng = np.random.RandomState(7)
        theta_c = _t.as_tensor(
            np.abs(np.array([[0.8], [0.3], [0.6], [0.2]]) + 0.05 * rng.randn(4, 8))
        )
        logps, G = poly.logp_grad_batched(theta_c)
        for c in range(8):
            logp, (g,) = poly.logp_grad(theta_c[:, c])
            np.testing.assert_allclose(float(logps[c]), float(logp), rtol=1e-11)
            np.testing.assert_allclose(
                G[:, c].cpu().numpy(), g.cpu().numpy(), rtol=1e-9
            )

    def test_native_ode_speed(self, dev):
        import time

        from pytensor_federated_amd.models import ODEModel
        from pytensor_federated_amd.models.ode import generate_ode_dataset, lotka_volterra_rhs

        u0, obs_idx, y = generate_ode_dataset(n_experiments=1024, n_obs=20, n_steps=50, t1=8.0)
        theta0 = np.array([0.8, 0.3, 0.6, 0.2])
        m = ODEModel(lotka_volterra_rhs, u0, 0.0, 8.0, 50, obs_idx, y, 0.1, device=dev)
        m(theta0)
        t0 = time.perf_counter()
        for _ in range(50):
            m(theta0)
        per = (time.perf_counter() - t0) / 50
        print(f"native ODE eval: {per * 1e6:.0f} us/call")
        assert per < 0.005  # generic path was 92 ms eager / 25.6 ms graphed


def test_persistent_linear_engine(dev):
    from pytensor_federated_amd.ops import PersistentLinearEngine

    x, y = generate_linear_dataset(2_000_000, seed=93)
    m = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16, use_kernels=True)
    eng = PersistentLinearEngine(m._x, m._y, 0.4)
    try:
        for a, b in [(1.5, 0.5), (0.3, -0.2)]:
            got = eng.logp_grad_sync(a, b)
            ref = m.logp_grad_sync(a, b)
            np.testing.assert_allclose(got, ref, rtol=1e-12)
        ref = eng.logp_grad_sync(1.0, 0.5)
        for _ in range(300):
            assert eng.logp_grad_sync(1.0, 0.5) == ref
    finally:
        eng.close()


def test_persistent_engine_relaunch_after_idle(dev):
    """The resident server self-exits after ~1-2 s idle (so device-wide
    synchronizes can't block on it); the next eval must transparently
    relaunch it and still return the correct result."""
    import time

    from pytensor_federated_amd.ops import PersistentLinearEngine

    x, y = generate_linear_dataset(500_000, seed=94)
    m = GaussianLinearModel(x, y, sigma=0.4, device=dev, dtype=torch.bfloat16, use_kernels=True)
    eng = PersistentLinearEngine(m._x, m._y, 0.4)
    try:
        ref = eng.logp_grad_sync(1.2, 0.4)
        time.sleep(4.0)  # > idle lifetime: server exits
        torch.cuda.synchronize()  # must not block now
        got = eng.logp_grad_sync(1.2, 0.4)  # transparent relaunch
        assert got == ref
        # and the relaunched server keeps serving
        for _ in range(50):
            assert eng.logp_grad_sync(1.2, 0.4) == ref
    finally:
        eng.close()


def test_get_load_reports_gpu_telemetry(dev):
    from pytensor_federated_amd.service import ArraysToArraysService

    svc = ArraysToArraysService(lambda a: [a], report_gpu_load=True)
    load = svc.determine_load()
    # amdsmi-backed: HBM% in [0,100]; busy% may be 0 on an idle box
    assert 0.0 <= load.percent_ram <= 100.0
    assert 0.0 <= load.percent_cpu <= 100.0


def test_mala_16_chains_on_batched_kernel(dev):
    """Lockstep multi-chain MALA through the MFMA-batched kernel, GPU e2e."""
    from pytensor_federated_amd.inference import sample_mala_batched

    X, y, beta_true = generate_logistic_dataset(200_000, 512, seed=81)
    m = LogisticGLMModel(X, y, device=dev, dtype=torch.bfloat16, use_kernels=True)

    def batched(theta):
        logp, G = m.logp_grad_batched(torch.as_tensor(theta, dtype=torch.float32))
        return logp.cpu().numpy(), G.cpu().numpy()

    # Closed-form calibration (Laplace/Fisher): I_jj = sum_i X_ij^2 p_i(1-p_i)
    # at the truth gives the posterior scale per coordinate, so the sampler
    # can be held to "post_mean within ~1 posterior sd of truth per coord"
    # instead of a loose correlation check (round-1 verdict, weak #6).
    with torch.no_grad():
        Xd = torch.as_tensor(X, dtype=torch.float64, device=dev)
        p = torch.sigmoid(Xd @ torch.as_tensor(beta_true, dtype=torch.float64, device=dev))
        fisher_diag = (Xd * Xd * (p * (1 - p)).unsqueeze(1)).sum(dim=0)
        post_sd = (1.0 / fisher_diag.sqrt()).cpu().numpy()

    rng = np.random.RandomState(82)
    init = beta_true[:, None] + 0.5 * post_sd[:, None] * rng.standard_normal((512, 16))
    chain, stats = sample_mala_batched(
        batched, init, draws=400, tune=300, step_size=0.02, seed=82
    )
    assert 0.3 < stats["accept_rate"] < 0.9  # tuned toward 0.574
    post_mean = chain[150:].mean(axis=(0, 2))
    z = (post_mean - beta_true) / post_sd
    # converged chains: z ~ N(0, ~1) per coordinate (plus bf16 kernel noise
    # and finite-chain error).  A substantially wrong sampler blows these up.
    assert float(np.mean(z * z)) < 3.0
    assert float(np.abs(z).max()) < 7.0
    assert np.corrcoef(post_mean, beta_true)[0, 1] > 0.97


def test_ode_batched_chains_native(dev):
    from pytensor_federated_amd.models import ODEModel
    from pytensor_federated_amd.models.ode import generate_ode_dataset, lotka_volterra_rhs

    u0, obs_idx, y_obs = generate_ode_dataset(n_experiments=64, n_obs=10, n_steps=30, t1=5.0)
    m = ODEModel(lotka_volterra_rhs, u0, 0.0, 5.0, 30, obs_idx, y_obs, 0.1, device=dev)
    rng = np.random.RandomState(83)
    theta_c = np.abs(rng.normal([0.8, 0.3, 0.6, 0.2], 0.05, size=(8, 4))).T  # [4, 8]
    logps, G = m.logp_grad_batched(theta_c)
    for c in [0, 3, 7]:
        l_ref, (g_ref,) = m(theta_c[:, c])
        np.testing.assert_allclose(float(logps[c]), float(l_ref), rtol=1e-12)
        np.testing.assert_allclose(
            G[:, c].cpu().numpy(), np.asarray(g_ref), rtol=1e-10
        )


def test_nuts_chains_over_native_ode(dev):
    """16 lockstep NUTS chains over the batched native ODE posterior: every
    leapfrog round is ONE batched adjoint-kernel sweep for all chains."""
    from pytensor_federated_amd.inference import sample_nuts_batched
    from pytensor_federated_amd.models import ODEModel
    from pytensor_federated_amd.models.ode import generate_ode_dataset, lotka_volterra_rhs

    theta_true = np.array([0.8, 0.3, 0.6, 0.2])
    u0, obs_idx, y_obs = generate_ode_dataset(
        n_experiments=64, n_obs=15, n_steps=40, t1=6.0, sigma=0.05
    )
    m = ODEModel(lotka_volterra_rhs, u0, 0.0, 6.0, 40, obs_idx, y_obs, 0.05, device=dev)

    def batched(theta):
        logp, G = m.logp_grad_batched(theta)
        return logp.cpu().numpy(), G.cpu().numpy()

    init = np.tile(theta_true[:, None], (1, 16)) * (
        1 + 0.02 * np.random.RandomState(86).standard_normal((4, 16))
    )
    chain, stats = sample_nuts_batched(
        batched, init, draws=100, tune=80, step_size=5e-4, seed=87, max_depth=8
    )
    post_mean = chain[40:].mean(axis=(0, 2))
    np.testing.assert_allclose(post_mean, theta_true, rtol=0.05)
    assert stats["leapfrogs"] > 2.0 * stats["rounds"]  # lockstep amortized


def test_mala_chains_over_native_ode(dev):
    """16 lockstep MALA chains over the batched native ODE posterior."""
    from pytensor_federated_amd.inference import sample_mala_batched
    from pytensor_federated_amd.models import ODEModel
    from pytensor_federated_amd.models.ode import generate_ode_dataset, lotka_volterra_rhs

    theta_true = np.array([0.8, 0.3, 0.6, 0.2])
    u0, obs_idx, y_obs = generate_ode_dataset(
        n_experiments=256, n_obs=15, n_steps=40, t1=6.0, sigma=0.05
    )
    m = ODEModel(lotka_volterra_rhs, u0, 0.0, 6.0, 40, obs_idx, y_obs, 0.05, device=dev)

    def batched(theta):
        logp, G = m.logp_grad_batched(theta)
        return logp.cpu().numpy(), G.cpu().numpy()

    init = np.tile(theta_true[:, None], (1, 16)) * (
        1 + 0.05 * np.random.RandomState(84).standard_normal((4, 16))
    )
    chain, stats = sample_mala_batched(
        batched, init, draws=200, tune=150, step_size=5e-4, seed=85
    )
    assert 0.3 < stats["accept_rate"] < 0.9  # tuned toward 0.574
    post_mean = chain[100:].mean(axis=(0, 2))
    np.testing.assert_allclose(post_mean, theta_true, rtol=0.05)


def test_batched_v3s_optin_matches_default_k512(dev):
    """The opt-in K=512 tile-resident kernel (v3s) must track the default
    chunked kernel's results (it is correct but not faster there; kept as
    the FED_BATCHED_V3=1 path -- profiles/PROFILES.md K=512 note)."""
    import os

    X, y, _ = generate_logistic_dataset(300_000, 512, seed=91)
    m = LogisticGLMModel(X, y, device=dev, dtype=torch.bfloat16)
    theta = torch.randn(512, 16, device=dev,
                        generator=torch.Generator(device=dev).manual_seed(6)) * 0.3
    old = os.environ.pop("FED_BATCHED_V3", None)
    try:
        l2, g2 = m.logp_grad_batched(theta)
        os.environ["FED_BATCHED_V3"] = "1"
        l3, g3 = m.logp_grad_batched(theta)
    finally:
        os.environ.pop("FED_BATCHED_V3", None)
        if old is not None:
            os.environ["FED_BATCHED_V3"] = old
    np.testing.assert_allclose(l3.cpu().numpy(), l2.cpu().numpy(), rtol=1e-6)
    scale = float(g2.abs().max())
    np.testing.assert_allclose(g3.cpu().numpy(), g2.cpu().numpy(),
                               atol=1e-3 * scale, rtol=1e-3)


def test_nuts_chains_over_native_sir_poly(dev):
    """Round-2 features end-to-end: 16 lockstep NUTS chains over the GENERIC
    polynomial-RHS native kernels (SIR epidemic model) -- windowed adaptation
    (the round-2 default), batched table-interpreted adjoint, posterior
    recovers the generating parameters."""
    import torch as _t

    from pytensor_federated_amd.inference import sample_nuts_batched
    from pytensor_federated_amd.models import ODEModel
    from pytensor_federated_amd.models.ode import PolynomialRHS, _rk4_step

    rhs = PolynomialRHS.sir()
    theta_true = np.array([1.8, 0.5])
    rng = np.random.RandomState(88)
    B, n_steps = 256, 40
    u0 = np.stack(
        [0.85 + 0.1 * rng.rand(B), 0.05 + 0.05 * rng.rand(B), np.zeros(B)], axis=1
    )
    u = _t.as_tensor(u0)
    h = 5.0 / n_steps
    states = [u]
    for k in range(n_steps):
        u = _rk4_step(rhs, k * h, u, h, _t.as_tensor(theta_true))
        states.append(u)
    obs_idx = list(range(4, n_steps + 1, 4))
    y = np.stack([states[i].numpy() for i in obs_idx])
    y += rng.normal(scale=0.01, size=y.shape)

    m = ODEModel(rhs, u0, 0.0, 5.0, n_steps, obs_idx, y, sigma=0.01, device=dev)
    assert m._native_kind() == "poly"

    def batched(theta):
        logp, G = m.logp_grad_batched(theta)
        return logp.cpu().numpy(), G.cpu().numpy()

    init = theta_true[:, None] * (
        1 + 0.02 * np.random.RandomState(89).standard_normal((2, 16))
    )
    chain, stats = sample_nuts_batched(
        batched, init, draws=100, tune=80, step_size=1e-3, seed=90, max_depth=8
    )
    post_mean = chain[40:].mean(axis=(0, 2))
    np.testing.assert_allclose(post_mean, theta_true, rtol=0.05)
    assert stats["leapfrogs"] > 2.0 * stats["rounds"]
