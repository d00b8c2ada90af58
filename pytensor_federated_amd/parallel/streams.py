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

from typing import List, Optional, Sequence, Tuple

import torch

__all__ = ["MultiShardDispatcher"]


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
