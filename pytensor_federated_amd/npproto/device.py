"""On-node device-buffer codec: HBM -> HBM array transport via dmabuf IPC.

The reference serializes every array through a host byte copy
(npproto/utils.py:13 ``bytes(arr.data)``).  For processes sharing one
MI355X node that is three copies (D2H, socket, H2D); this codec replaces
the payload with a 64-byte hipIpc handle + offset, so the receiver maps
the exporter's HBM region directly and does ONE device-to-device copy.

Wire format stays an ``Ndarray`` message (shape/strides as usual) with
``dtype = "hipipc/<real_dtype>"`` and ``data`` = the binary header::

    [64B hipIpcMemHandle][8B LE offset][8B LE nbytes][16B reserved]

Requires dmabuf IPC (``HSA_ENABLE_IPC_MODE_LEGACY=0``, the image default).
Use :class:`DeviceArrayExporter` on the sending side (it owns a dedicated
hipMalloc region -- IPC handles must reference allocation bases, which
torch's caching allocator does not expose portably) and
:func:`device_ndarray_to_torch` on the receiving side.
"""
from __future__ import annotations

import ctypes
import struct
from typing import Dict, Optional

from . import Ndarray

__all__ = [
    "DEVICE_DTYPE_PREFIX",
    "is_device_ndarray",
    "DeviceArrayExporter",
    "device_ndarray_to_torch",
    "close_imported_handles",
]

DEVICE_DTYPE_PREFIX = "hipipc/"
_HEADER = struct.Struct("<64sqq16x")


def _lib():
    from ..ops import require_kernels

    lib = require_kernels()
    if not hasattr(lib.fed_ipc_get_handle, "_configured"):
        lib.fed_device_alloc.restype = ctypes.c_void_p
        lib.fed_device_alloc.argtypes = [ctypes.c_longlong]
        lib.fed_device_free.restype = ctypes.c_int
        lib.fed_device_free.argtypes = [ctypes.c_void_p]
        lib.fed_d2d_copy.restype = ctypes.c_int
        lib.fed_d2d_copy.argtypes = [ctypes.c_void_p, ctypes.c_void_p, ctypes.c_longlong, ctypes.c_void_p]
        lib.fed_stream_sync.restype = ctypes.c_int
        lib.fed_stream_sync.argtypes = [ctypes.c_void_p]
        lib.fed_ipc_get_handle.restype = ctypes.c_int
        lib.fed_ipc_get_handle.argtypes = [ctypes.c_void_p, ctypes.c_char_p]
        lib.fed_ipc_open.restype = ctypes.c_int
        lib.fed_ipc_open.argtypes = [ctypes.c_char_p, ctypes.POINTER(ctypes.c_void_p)]
        lib.fed_ipc_close.restype = ctypes.c_int
        lib.fed_ipc_close.argtypes = [ctypes.c_void_p]
        lib.fed_ipc_get_handle._configured = True
    return lib


def is_device_ndarray(nda: Ndarray) -> bool:
    return nda.dtype.startswith(DEVICE_DTYPE_PREFIX)


class DeviceArrayExporter:
    """Owns a hipMalloc region whose IPC handle is shared with importers.

    ``export(tensor)`` copies the tensor device-to-device into the region
    (growing it when needed) and returns the handle-bearing ``Ndarray``.
    The region stays mapped on the receiving side, so repeated exports of
    equal-size arrays reuse the same remote mapping.
    """

    def __init__(self) -> None:
        self._lib = _lib()
        self._base: Optional[int] = None
        self._capacity = 0
        self._offset = 0
        self._handle = b""

    def _ensure(self, nbytes: int) -> None:
        if self._base is not None and self._capacity >= self._offset + nbytes:
            return
        # grow: new region (old one stays alive until exporter is dropped --
        # importers may still have it mapped)
        cap = max(1 << 20, 2 * (self._offset + nbytes))
        base = self._lib.fed_device_alloc(cap)
        if not base:
            raise RuntimeError("hipMalloc failed for IPC export region")
        handle = ctypes.create_string_buffer(64)
        rc = self._lib.fed_ipc_get_handle(base, handle)
        if rc != 0:
            raise RuntimeError(f"hipIpcGetMemHandle failed ({rc})")
        self._base = base
        self._capacity = cap
        self._offset = 0
        self._handle = handle.raw

    def reset(self) -> None:
        """Reuse the region from the start (previous exports invalidated)."""
        self._offset = 0

    def export(self, tensor) -> Ndarray:
        import torch

        assert isinstance(tensor, torch.Tensor) and tensor.is_cuda
        t = tensor.detach()
        if not t.is_contiguous():
            t = t.contiguous()
        nbytes = t.numel() * t.element_size()
        aligned = (nbytes + 255) & ~255
        self._ensure(aligned)
        dst = self._base + self._offset
        stream = torch.cuda.current_stream().cuda_stream
        rc = self._lib.fed_d2d_copy(dst, t.data_ptr(), nbytes, stream)
        if rc != 0:
            raise RuntimeError(f"device copy failed ({rc})")
        rc = self._lib.fed_stream_sync(stream)  # handle is ready for peers
        if rc != 0:
            raise RuntimeError(f"stream sync failed ({rc})")
        dtype_name = str(t.dtype).replace("torch.", "")
        nda = Ndarray(
            data=_HEADER.pack(self._handle, self._offset, nbytes),
            dtype=DEVICE_DTYPE_PREFIX + dtype_name,
            shape=list(t.shape),
            strides=[s * t.element_size() for s in t.stride()],
        )
        self._offset += aligned
        return nda


#: importer-side cache: handle bytes -> mapped base pointer.  Bounded: when
#: it exceeds _OPENED_MAX distinct exporter regions, the least recently used
#: mapping is closed (after a device sync so no in-flight copy still reads it).
_opened: Dict[bytes, int] = {}
_OPENED_MAX = 64


def _evict_opened_lru() -> None:
    import torch

    lib = _lib()
    while len(_opened) > _OPENED_MAX:
        handle, base = next(iter(_opened.items()))
        torch.cuda.synchronize()
        lib.fed_ipc_close(base)
        del _opened[handle]


def device_ndarray_to_torch(nda: Ndarray, device=None):
    """Map the exporter's region (cached) and D2D-copy into a fresh tensor.

    The wire header is untrusted: ``nbytes`` must equal the byte size implied
    by ``shape``/``dtype`` and ``offset`` must be non-negative, or the message
    is rejected before any device copy happens.
    """
    import torch

    if not is_device_ndarray(nda):
        raise ValueError("not a device ndarray")
    if len(nda.data) != _HEADER.size:
        raise ValueError(
            f"device ndarray header must be {_HEADER.size} bytes, got {len(nda.data)}"
        )
    handle, offset, nbytes = _HEADER.unpack(bytes(nda.data))
    dtype_name = nda.dtype[len(DEVICE_DTYPE_PREFIX) :]
    dtype = getattr(torch, dtype_name, None)
    if not isinstance(dtype, torch.dtype):
        raise ValueError(f"unknown device ndarray dtype {dtype_name!r}")
    itemsize = torch.empty(0, dtype=dtype).element_size()
    numel = 1
    for s in nda.shape:
        if s < 0:
            raise ValueError(f"negative dimension in shape {nda.shape}")
        numel *= s
    expected = numel * itemsize
    if nbytes != expected:
        raise ValueError(
            f"device ndarray nbytes={nbytes} does not match "
            f"shape {list(nda.shape)} x {dtype} = {expected} bytes"
        )
    if offset < 0:
        raise ValueError(f"negative device ndarray offset {offset}")
    lib = _lib()
    base = _opened.get(handle)
    if base is None:
        ptr = ctypes.c_void_p()
        rc = lib.fed_ipc_open(handle, ctypes.byref(ptr))
        if rc != 0:
            raise RuntimeError(f"hipIpcOpenMemHandle failed ({rc})")
        base = ptr.value
        _opened[handle] = base
        _evict_opened_lru()
    else:
        # refresh LRU position
        del _opened[handle]
        _opened[handle] = base
    if device is None:
        device = torch.device("cuda", torch.cuda.current_device())
    out = torch.empty(nda.shape, dtype=dtype, device=device)
    stream = torch.cuda.current_stream().cuda_stream
    rc = lib.fed_d2d_copy(out.data_ptr(), base + offset, nbytes, stream)
    if rc != 0:
        raise RuntimeError(f"device copy failed ({rc})")
    return out


def close_imported_handles() -> None:
    lib = _lib()
    for base in _opened.values():
        lib.fed_ipc_close(base)
    _opened.clear()
