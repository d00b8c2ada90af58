"""torch.autograd-native graph embedding of remote/blackbox logp functions.

The reference embeds a ``LogpGradFunc`` into PyTensor graphs via
``LogpGradOp`` (reference wrapper_ops.py:84-146): forward+gradient are one
fused remote call, and ``grad()`` returns ``g_logp * grads``.  This module
is the MI355X-native analog on ``torch.autograd``: the blackbox's gradients
are fetched in the same call as the logp (stashed on the autograd context),
so a torch-side ``backward()`` costs no extra RPC -- exactly the fused
semantics the reference gets from CSE-merging the re-applied Op
(wrapper_ops.py:119-132).

PyTensor adapters with the reference's exact Op API live in
``wrapper_ops`` (optional import).
"""
from __future__ import annotations

from typing import Callable, Sequence

import numpy as np
import torch

from .op_async import gather_evaluate, run_coroutine_blocking
from .signatures import LogpFunc, LogpGradFunc

__all__ = ["LogpGradOp", "LogpOp", "federated_logp_grad", "FederatedLogpGrad"]


def _to_numpy(t) -> np.ndarray:
    if isinstance(t, torch.Tensor):
        t = t.detach()
        if t.device.type != "cpu":
            t = t.cpu()
        if t.dtype == torch.bfloat16:
            t = t.float()
        return t.numpy()
    return np.asarray(t)


class _LogpGradFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, func: LogpGradFunc, *inputs: torch.Tensor):
        np_inputs = [_to_numpy(t) for t in inputs]
        logp, grads = func(*np_inputs)
        if len(grads) != len(inputs):
            raise ValueError(f"Got {len(grads)} gradients for {len(inputs)} inputs.")
        ctx.grads = [
            # copy: wire-decoded grads are zero-copy READ-ONLY views over
            # the message bytes, which torch tensors must not alias
            torch.tensor(np.asarray(g), dtype=t.dtype if t.is_floating_point() else torch.float64).to(t.device)
            for g, t in zip(grads, inputs)
        ]
        return torch.as_tensor(float(np.asarray(logp)), dtype=torch.float64)

    @staticmethod
    def backward(ctx, g_logp):
        # d(logp)/d(input_i) scaled by the incoming cotangent
        # (reference wrapper_ops.py:132: ``[g_logp * g for g in gradients]``).
        return (None, *[g_logp.to(g.dtype) * g for g in ctx.grads])


class LogpGradOp:
    """Differentiable blackbox ``[*theta] -> logp`` backed by a LogpGradFunc.

    ``op = LogpGradOp(client.evaluate)``; ``logp = op(a, b)`` is a 0-d torch
    tensor through which ``logp.backward()`` flows using the remotely
    computed gradients.  Equality/hashing follow the wrapped function so
    identical ops merge in caches (parity: reference wrapper_ops.py:91
    ``_props`` equality).
    """

    def __init__(self, logp_grad_func: LogpGradFunc) -> None:
        self._logp_grad_func = logp_grad_func

    def __call__(self, *inputs) -> torch.Tensor:
        tensors = [
            t if isinstance(t, torch.Tensor) else torch.as_tensor(np.asarray(t, dtype=float))
            for t in inputs
        ]
        return _LogpGradFunction.apply(self._logp_grad_func, *tensors)

    def __eq__(self, other) -> bool:
        return (
            type(other) is type(self) and other._logp_grad_func == self._logp_grad_func
        )

    def __hash__(self) -> int:
        return hash((type(self), self._logp_grad_func))


class LogpOp:
    """Non-differentiable blackbox logp (parity: reference wrapper_ops.py:44-81).

    For Metropolis-style samplers that need no gradients.
    """

    def __init__(self, logp_func: LogpFunc) -> None:
        self._logp_func = logp_func

    def __call__(self, *inputs) -> torch.Tensor:
        np_inputs = [_to_numpy(t) for t in inputs]
        logp = self._logp_func(*np_inputs)
        return torch.as_tensor(np.asarray(logp), dtype=torch.float64)

    def __eq__(self, other) -> bool:
        return type(other) is type(self) and other._logp_func == self._logp_func

    def __hash__(self) -> int:
        return hash((type(self), self._logp_func))


class _FederatedLogpGradFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, async_funcs, *inputs: torch.Tensor):
        np_inputs = [_to_numpy(t) for t in inputs]
        results = run_coroutine_blocking(
            gather_evaluate(list(async_funcs), [np_inputs] * len(async_funcs))
        )
        total_logp = 0.0
        grads = [np.zeros_like(x, dtype=np.float64) for x in np_inputs]
        for logp, shard_grads in results:
            total_logp += float(logp)
            for g_acc, g in zip(grads, shard_grads):
                g_acc += np.asarray(g, dtype=np.float64)
        ctx.grads = [
            torch.as_tensor(g, dtype=t.dtype if t.is_floating_point() else torch.float64).to(t.device)
            for g, t in zip(grads, inputs)
        ]
        return torch.as_tensor(total_logp, dtype=torch.float64)

    @staticmethod
    def backward(ctx, g_logp):
        return (None, *[g_logp.to(g.dtype) * g for g in ctx.grads])


class FederatedLogpGrad:
    """Sum of N federated shard logps with concurrent fan-out.

    The user-facing analog of the reference's N ``pm.Potential`` shards +
    async fusion (demo_model.py:28-36 + op_async fan-out): all shard RPCs
    run concurrently on one event loop; logp and grads sum exactly because
    both distribute over data shards.  Each element of ``async_funcs`` is an
    async LogpGradFunc (e.g. ``LogpGradServiceClient.evaluate_async``).
    """

    def __init__(self, async_funcs: Sequence[Callable]) -> None:
        self._async_funcs = tuple(async_funcs)

    def __call__(self, *inputs) -> torch.Tensor:
        tensors = [
            t if isinstance(t, torch.Tensor) else torch.as_tensor(np.asarray(t, dtype=float))
            for t in inputs
        ]
        return _FederatedLogpGradFunction.apply(self._async_funcs, *tensors)


def federated_logp_grad(async_funcs: Sequence[Callable], *inputs) -> torch.Tensor:
    """Functional form of :class:`FederatedLogpGrad`."""
    return FederatedLogpGrad(async_funcs)(*inputs)
