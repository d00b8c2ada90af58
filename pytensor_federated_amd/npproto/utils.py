"""ndarray <-> ``Ndarray`` message converters.

Behavior parity with reference npproto/utils.py:9-24:

* serialize copies the buffer (``bytes(arr.data)``), preserving strides, so
  any buffer-protocol dtype round-trips (datetime64, unicode, sub-byte views);
* deserialize is a zero-copy ``numpy.ndarray`` view over the message bytes,
  honoring the transmitted strides.

Additions for the MI355X build: torch.Tensor converters.  torch has dtypes
numpy lacks (bfloat16); those are transmitted with dtype string
``"bfloat16"`` and raw little-endian payload, and decoded back to torch.
"""
from __future__ import annotations

import numpy

from . import Ndarray

__all__ = [
    "ndarray_from_numpy",
    "ndarray_to_numpy",
    "ndarray_from_torch",
    "ndarray_to_torch",
    "TORCH_ONLY_DTYPES",
]

#: dtype strings that numpy cannot represent; round-trip via torch only.
TORCH_ONLY_DTYPES = {"bfloat16", "float8_e4m3fn", "float8_e5m2"}


def ndarray_from_numpy(arr: numpy.ndarray) -> Ndarray:
    # ``tobytes()`` linearizes into C order (and, unlike the buffer protocol,
    # also handles datetime64/object dtypes on numpy>=2), so the transmitted
    # strides must be the C-contiguous ones -- normalizing here (the
    # reference serializes raw strides, which corrupts non-contiguous views;
    # contiguous arrays encode identically either way).
    if not arr.flags.c_contiguous:
        arr = numpy.ascontiguousarray(arr)
    return Ndarray(
        shape=list(arr.shape),
        dtype=str(arr.dtype),
        data=arr.tobytes(),
        strides=list(arr.strides),
    )


def ndarray_to_numpy(nda: Ndarray) -> numpy.ndarray:
    return numpy.ndarray(
        buffer=nda.data,
        shape=nda.shape,
        dtype=numpy.dtype(nda.dtype),
        strides=nda.strides,
    )


def ndarray_from_torch(tensor) -> Ndarray:
    """Serialize a torch.Tensor (moved to CPU, made contiguous if needed)."""
    import torch

    t = tensor.detach()
    if t.device.type != "cpu":
        t = t.cpu()
    if not t.is_contiguous():
        t = t.contiguous()
    dtype_name = str(t.dtype).replace("torch.", "")
    if dtype_name in TORCH_ONLY_DTYPES:
        raw = t.view(torch.uint8) if t.element_size() == 1 else t.view(torch.int16)
        np_arr = raw.numpy()
        itemsize = t.element_size()
        return Ndarray(
            shape=list(t.shape),
            dtype=dtype_name,
            data=np_arr.tobytes(),
            strides=[s * itemsize for s in t.stride()],
        )
    return ndarray_from_numpy(t.numpy())


def ndarray_to_torch(nda: Ndarray, device=None):
    """Deserialize to a torch.Tensor (copies; optionally onto a device)."""
    import torch

    if nda.dtype in TORCH_ONLY_DTYPES:
        torch_dtype = getattr(torch, nda.dtype)
        itemsize = torch.tensor([], dtype=torch_dtype).element_size()
        stor = numpy.frombuffer(bytearray(nda.data), dtype=numpy.uint8)
        t = torch.from_numpy(stor)
        if itemsize == 2:
            t = t.view(torch.int16)
        t = t.view(torch_dtype)
        strides = [s // itemsize for s in nda.strides]
        t = torch.as_strided(t, nda.shape, strides).clone()
    else:
        t = torch.from_numpy(ndarray_to_numpy(nda).copy())
    if device is not None:
        t = t.to(device)
    return t
