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
    [Sequence[np.ndarray]],  # any N parameter/data arrays
    Sequence[np.ndarray],  # any M result arrays
]
"""Generic compute function: multiple arrays in, multiple arrays out."""

LogpFunc = Callable[
    [Sequence[np.ndarray]],  # any N parameter arrays
    np.ndarray,  # 0-d log-probability
]
"""Log-probability function without gradients (e.g. a log-likelihood)."""

LogpGradFunc = Callable[
    [Sequence[np.ndarray]],  # any N parameter arrays
    Tuple[np.ndarray, Sequence[np.ndarray]],  # (0-d log-p, one grad per input)
]
"""Log-probability function with gradients w.r.t. its inputs."""
