"""Type and signature definitions (parity: reference signatures.py:8-33).

The three contract types of the framework.  Implementations may run on CPU
(numpy), on one MI355X (torch + HIP kernels), or across 8 GPUs (RCCL-summed
shards) -- the contract at this seam is always numpy arrays in/out, which is
what crosses the gRPC edge.
"""
from typing import Callable, Sequence, Tuple

import numpy as np

__all__ = ["ComputeFunc", "LogpFunc", "LogpGradFunc"]

ComputeFunc = Callable[
    [Sequence[np.ndarray]],  # arbitrary number of input arrays
    Sequence[np.ndarray],  # arbitrary number of output arrays
]
"""Generic compute function: multiple arrays in, multiple arrays out."""

LogpFunc = Callable[
    [Sequence[np.ndarray]],  # arbitrary number of input arrays
    np.ndarray,  # scalar log-p
]
"""Log-probability function without gradients (e.g. a log-likelihood)."""

LogpGradFunc = Callable[
    [Sequence[np.ndarray]],  # arbitrary number of input arrays
    Tuple[np.ndarray, Sequence[np.ndarray]],  # scalar log-p, grads w.r.t. each input
]
"""Log-probability function with gradients w.r.t. its inputs."""
