"""Signature adapters and typed service clients.

Parity: reference common.py:12-161.  ``wrap_logp_grad_func`` defines the
wire layout ``[logp, dinput0, dinput1, ...]`` that the client-side
``LogpGradServiceClient`` and the graph Ops (torch_ops / wrapper_ops)
consume -- and that the RCCL shard-sum path reduces elementwise (the sum of
per-shard logps and grads is exact because both distribute over data shards).
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import numpy as np

from .service import ArraysToArraysServiceClient
from .signatures import ComputeFunc, LogpFunc, LogpGradFunc

__all__ = [
    "wrap_logp_func",
    "wrap_logp_grad_func",
    "LogpServiceClient",
    "LogpGradServiceClient",
]


def wrap_logp_func(logp_func: LogpFunc) -> ComputeFunc:
    """Adapt a LogpFunc to the generic ComputeFunc signature.

    Validates that the function returns a scalar ``()``-shaped ndarray
    (reference common.py:12-23).
    """

    def compute_func(*inputs: Sequence[np.ndarray]) -> List[np.ndarray]:
        logp = logp_func(*inputs)
        if not isinstance(logp, np.ndarray) or logp.shape != ():
            raise TypeError(
                f"The logp function must return a scalar ndarray, got {type(logp)} {getattr(logp, 'shape', None)}."
            )
        return [logp]

    return compute_func


def wrap_logp_grad_func(logp_grad_func: LogpGradFunc) -> ComputeFunc:
    """Adapt a LogpGradFunc to the generic ComputeFunc signature.

    Validates the (logp, grads) contract and flattens it to the wire layout
    ``[logp, *grads]`` with one gradient per input (reference common.py:26-49).
    """

    def compute_func(*inputs: Sequence[np.ndarray]) -> List[np.ndarray]:
        result = logp_grad_func(*inputs)
        if not (isinstance(result, tuple) and len(result) == 2):
            raise TypeError(
                f"The logp-grad function must return a (logp, gradients) tuple, got {type(result)}."
            )
        logp, gradients = result
        if not isinstance(logp, np.ndarray) or logp.shape != ():
            raise TypeError(
                f"The first return value must be a scalar ndarray, got {type(logp)}."
            )
        if len(gradients) != len(inputs):
            raise ValueError(
                f"Got {len(gradients)} gradients for {len(inputs)} inputs."
            )
        return [logp, *gradients]

    return compute_func


class LogpServiceClient:
    """Client for a worker serving a wrapped :class:`LogpFunc`.

    Parity: reference common.py:52-102.
    """

    def __init__(
        self,
        host: str = None,
        port: int = None,
        *,
        hosts_and_ports: Sequence[Tuple[str, int]] = None,
        **client_kwargs,
    ) -> None:
        self._client = ArraysToArraysServiceClient(
            host, port, hosts_and_ports=hosts_and_ports, **client_kwargs
        )

    def __call__(self, *inputs: Sequence[np.ndarray]) -> np.ndarray:
        return self.evaluate(*inputs)

    def evaluate(self, *inputs: Sequence[np.ndarray], use_stream: bool = True) -> np.ndarray:
        """Evaluate the remote logp (bidirectional stream by default)."""
        (logp,) = self._client.evaluate(*inputs, use_stream=use_stream)
        return logp

    async def evaluate_async(
        self, *inputs: Sequence[np.ndarray], use_stream: bool = True
    ) -> np.ndarray:
        (logp,) = await self._client.evaluate_async(*inputs, use_stream=use_stream)
        return logp


class LogpGradServiceClient:
    """Client for a worker serving a wrapped :class:`LogpGradFunc`.

    Unpacks the ``[logp, *gradients]`` wire layout back into the
    ``(logp, gradients)`` contract (reference common.py:105-161).
    """

    def __init__(
        self,
        host: str = None,
        port: int = None,
        *,
        hosts_and_ports: Sequence[Tuple[str, int]] = None,
        **client_kwargs,
    ) -> None:
        self._client = ArraysToArraysServiceClient(
            host, port, hosts_and_ports=hosts_and_ports, **client_kwargs
        )

    def __call__(
        self, *inputs: Sequence[np.ndarray]
    ) -> Tuple[np.ndarray, Sequence[np.ndarray]]:
        return self.evaluate(*inputs)

    def evaluate(
        self, *inputs: Sequence[np.ndarray], use_stream: bool = True
    ) -> Tuple[np.ndarray, Sequence[np.ndarray]]:
        logp, *gradients = self._client.evaluate(*inputs, use_stream=use_stream)
        return logp, gradients

    async def evaluate_async(
        self, *inputs: Sequence[np.ndarray], use_stream: bool = True
    ) -> Tuple[np.ndarray, Sequence[np.ndarray]]:
        logp, *gradients = await self._client.evaluate_async(*inputs, use_stream=use_stream)
        return logp, gradients
