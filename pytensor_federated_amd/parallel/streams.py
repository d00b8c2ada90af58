"""Multi-shard dispatch on ONE GPU via overlapped HIP streams.

The on-device analog of the reference's ``ParallelAsyncOp`` fan-out
(reference op_async.py:107-132: asyncio.gather over N RPC coroutines): N
shard models resident on one MI355X evaluate concurrently, each on its own
HIP stream; the consumer stream waits on per-shard events and sums the
fused ``[logp, *grads]`` buffers on device.  Kernel launches are async, so
the host issues all N launches back-to-back and the GPU overlaps them --
max-of-durations instead of sum, the same semantics the reference's tests
assert for async fusion (test_op_async.py:166-195).

On CPU the dispatcher degrades to sequential evaluation (used by the CPU
test suite for numerical equivalence).
"""
from __future__ import annotations

from typing import List, Sequence, Tuple

import torch

__all__ = ["MultiShardDispatcher", "NativeMultiShardEngine"]


class NativeMultiShardEngine:
    """N Gaussian-linear shards on one GPU, fanned out by the NATIVE engine.

    One ctypes call per federated evaluation: the C++/HIP engine launches
    every shard's fused kernel on its own HIP stream, event-gates a combine
    kernel, and the host polls the pinned result mailbox.  ~zero Python in
    the fan-out (the native successor of the asyncio gather fan-out,
    reference op_async.py:126-131).
    """

    def __init__(self, models: Sequence) -> None:
        import ctypes

        from ..ops import _DTYPE_CODE, require_kernels

        if not models:
            raise ValueError("Need at least one shard model.")
        for m in models:
            if not m._x.is_cuda:
                raise ValueError("NativeMultiShardEngine needs GPU-resident shards.")
        dtypes = {m._x.dtype for m in models}
        sigmas = {m._sigma for m in models}
        if len(dtypes) != 1 or len(sigmas) != 1:
            raise ValueError("All shards must share dtype and sigma.")
        self.models = list(models)  # keeps the shard tensors alive
        lib = require_kernels()
        lib.fed_linear_engine_create.restype = ctypes.c_void_p
        lib.fed_linear_engine_create.argtypes = [
            ctypes.c_int,
            ctypes.POINTER(ctypes.c_void_p),
            ctypes.POINTER(ctypes.c_void_p),
            ctypes.POINTER(ctypes.c_longlong),
            ctypes.c_double,
            ctypes.c_int,
        ]
        lib.fed_linear_engine_eval.restype = ctypes.c_int
        lib.fed_linear_engine_eval.argtypes = [
            ctypes.c_void_p, ctypes.c_double, ctypes.c_double,
            ctypes.POINTER(ctypes.c_double), ctypes.c_void_p,
        ]
        lib.fed_linear_engine_destroy.restype = ctypes.c_int
        lib.fed_linear_engine_destroy.argtypes = [ctypes.c_void_p]
        self._lib = lib
        n = len(models)
        xs = (ctypes.c_void_p * n)(*[m._x.data_ptr() for m in models])
        ys = (ctypes.c_void_p * n)(*[m._y.data_ptr() for m in models])
        ns = (ctypes.c_longlong * n)(*[m._x.numel() for m in models])
        self._handle = lib.fed_linear_engine_create(
            n, xs, ys, ns, float(models[0]._sigma), _DTYPE_CODE[models[0]._x.dtype]
        )
        if not self._handle:
            raise RuntimeError("fed_linear_engine_create failed")
        self._out = (ctypes.c_double * 3)()

    def logp_grad_sync(self, intercept: float, slope: float):
        """Returns (logp, d/da, d/db) host floats; blocks until complete."""
        sync = torch.cuda.current_stream().cuda_stream
        rc = self._lib.fed_linear_engine_eval(
            self._handle, float(intercept), float(slope), self._out, sync
        )
        if rc != 0:
            raise RuntimeError(f"fed_linear_engine_eval failed with code {rc}")
        return self._out[0], self._out[1], self._out[2]

    def __call__(self, *params):
        import numpy as np

        logp, ga, gb = self.logp_grad_sync(float(params[0]), float(params[1]))
        return np.asarray(logp), [np.asarray(ga), np.asarray(gb)]

    def as_logp_grad_func(self):
        return self.__call__

    def close(self) -> None:
        if getattr(self, "_handle", None):
            self._lib.fed_linear_engine_destroy(self._handle)
            self._handle = None

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class MultiShardDispatcher:
    def __init__(self, models: Sequence) -> None:
        if not models:
            raise ValueError("Need at least one shard model.")
        self.models = list(models)
        self._on_gpu = all(getattr(m, "device", torch.device("cpu")).type == "cuda" for m in self.models)
        if self._on_gpu:
            self._streams = [torch.cuda.Stream() for _ in self.models]
            self._events = [torch.cuda.Event() for _ in self.models]

    def logp_grad(self, *params) -> Tuple[torch.Tensor, List[torch.Tensor]]:
        """Sum of per-shard logps and grads, computed with overlapped streams."""
        results = []
        if self._on_gpu:
            main = torch.cuda.current_stream()
            for model, stream, event in zip(self.models, self._streams, self._events):
                stream.wait_stream(main)
                with torch.cuda.stream(stream):
                    results.append(model.logp_grad(*params))
                event.record(stream)
            for event in self._events:
                main.wait_event(event)
        else:
            for model in self.models:
                results.append(model.logp_grad(*params))
        logp = results[0][0].to(torch.float64).clone()
        grads = [g.to(torch.float64).clone() for g in results[0][1]]
        for shard_logp, shard_grads in results[1:]:
            logp += shard_logp.to(torch.float64)
            for acc, g in zip(grads, shard_grads):
                acc += g.to(torch.float64)
        return logp, grads

    def __call__(self, *params):
        import numpy as np

        tparams = [torch.as_tensor(np.asarray(p, dtype=np.float64)) for p in params]
        logp, grads = self.logp_grad(*tparams)
        return (
            np.asarray(logp.detach().cpu().numpy()),
            [np.asarray(g.detach().cpu().numpy()) for g in grads],
        )

    def as_logp_grad_func(self):
        return self.__call__
