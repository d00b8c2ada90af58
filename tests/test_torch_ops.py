"""torch.autograd embedding contract (mirrors reference test_wrapper_ops.py).

Uses the reference's mock pattern: a dummy quadratic model with hand-derived
analytic gradients (reference test_wrapper_ops.py:34-45) -- no network.
"""
import numpy as np
import pytest
import torch

from pytensor_federated_amd.torch_ops import FederatedLogpGrad, LogpGradOp, LogpOp


def dummy_quadratic_model(a, b):
    """Sum of squared residuals with manual gradients (mock compute layer)."""
    rng = np.random.RandomState(42)
    x = np.array([1.0, 2.0, 3.0])
    y = rng.normal(2 * x**2 + 0.5, scale=0.1)
    pred = a * x**2 + b
    cost = np.asarray(np.sum((pred - y) ** 2))
    grads = [
        np.asarray(np.sum(2 * x**2 * (pred - y))),
        np.asarray(np.sum(2 * (pred - y))),
    ]
    return cost, grads


class _CountingClient:
    """Mock LogpGradFunc that counts calls (fused forward+grad check)."""

    def __init__(self):
        self.n_calls = 0

    def __call__(self, a, b):
        self.n_calls += 1
        return dummy_quadratic_model(a, b)


def test_forward_value():
    op = LogpGradOp(dummy_quadratic_model)
    out = op(2.0, 0.5)
    expected, _ = dummy_quadratic_model(2.0, 0.5)
    assert out.item() == pytest.approx(float(expected))


def test_backward_uses_remote_gradients():
    op = LogpGradOp(dummy_quadratic_model)
    a = torch.tensor(1.7, requires_grad=True, dtype=torch.float64)
    b = torch.tensor(0.3, requires_grad=True, dtype=torch.float64)
    logp = op(a, b)
    logp.backward()
    _, (ga, gb) = dummy_quadratic_model(1.7, 0.3)
    assert a.grad.item() == pytest.approx(float(ga))
    assert b.grad.item() == pytest.approx(float(gb))


def test_backward_scales_by_cotangent():
    op = LogpGradOp(dummy_quadratic_model)
    a = torch.tensor(1.7, requires_grad=True, dtype=torch.float64)
    b = torch.tensor(0.3, requires_grad=True, dtype=torch.float64)
    (3.0 * op(a, b)).backward()
    _, (ga, gb) = dummy_quadratic_model(1.7, 0.3)
    assert a.grad.item() == pytest.approx(3.0 * float(ga))
    assert b.grad.item() == pytest.approx(3.0 * float(gb))


def test_forward_and_grad_are_one_call():
    client = _CountingClient()
    op = LogpGradOp(client)
    a = torch.tensor(1.0, requires_grad=True, dtype=torch.float64)
    b = torch.tensor(0.0, requires_grad=True, dtype=torch.float64)
    logp = op(a, b)
    logp.backward()
    assert client.n_calls == 1  # fused semantics (reference wrapper_ops.py:119-132)


def test_accepts_raw_floats_and_arrays():
    op = LogpGradOp(dummy_quadratic_model)
    # regression analog of reference issue #24 (non-Variable inputs)
    out = op(2, np.array(0.5))
    assert out.shape == ()


def test_wrong_grad_count_raises():
    def bad(a):
        return np.asarray(1.0), [np.asarray(1.0), np.asarray(2.0)]

    with pytest.raises(ValueError, match="gradients for"):
        LogpGradOp(bad)(1.0)


def test_logp_op_no_grad():
    def logp_only(a):
        return np.asarray(float(a) ** 2)

    op = LogpOp(logp_only)
    out = op(torch.tensor(3.0))
    assert out.item() == pytest.approx(9.0)
    assert not out.requires_grad


def test_op_equality_by_func():
    op1 = LogpGradOp(dummy_quadratic_model)
    op2 = LogpGradOp(dummy_quadratic_model)
    assert op1 == op2 and hash(op1) == hash(op2)


def test_federated_sums_shards():
    async def shard1(a, b):
        logp, grads = dummy_quadratic_model(a, b)
        return logp, grads

    async def shard2(a, b):
        logp, grads = dummy_quadratic_model(a, b)
        return logp * 2.0, [g * 2.0 for g in grads]

    fed = FederatedLogpGrad([shard1, shard2])
    a = torch.tensor(1.5, requires_grad=True, dtype=torch.float64)
    b = torch.tensor(0.5, requires_grad=True, dtype=torch.float64)
    logp = fed(a, b)
    logp.backward()
    ref_logp, (ga, gb) = dummy_quadratic_model(1.5, 0.5)
    assert logp.item() == pytest.approx(3.0 * float(ref_logp))
    assert a.grad.item() == pytest.approx(3.0 * float(ga))
    assert b.grad.item() == pytest.approx(3.0 * float(gb))
