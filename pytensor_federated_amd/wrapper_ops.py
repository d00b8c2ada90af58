"""Optional PyTensor adapters: the reference's Op API on top of this framework.

Only importable when ``pytensor`` is installed (gated in ``__init__`` like
the reference's optional L4/L5 import, reference __init__.py:1-12).  Gives
PyMC users the exact embedding API of the reference:

* ``LogpOp`` / ``LogpGradOp``  (reference wrapper_ops.py:44-146)
* ``AsyncLogpOp`` / ``AsyncLogpGradOp`` with coroutine clients
* ``ArraysToArraysOp`` / ``AsyncArraysToArraysOp`` (wrapper_ops.py:14-41)
* a ``fuse_asyncs`` rewrite that batches independent async applies into one
  concurrently-gathered apply (reference op_async.py:68-234)

The fan-out machinery itself is framework-owned (``op_async.gather_evaluate``);
these classes are thin shims, so the compute path (HIP kernels, RCCL shard
sum) is identical whether driven from PyTensor, torch, or raw numpy.

NOTE: pytensor itself is not installed in the ROCm image; the adapter
logic is tested against a faithful graph-API stub
(tests/test_wrapper_ops.py), and the torch-native equivalents in
``torch_ops`` are the production-tested path.
"""
from __future__ import annotations

import asyncio
from typing import List, Sequence, Union

import numpy as np

import pytensor.tensor as at
from pytensor.compile.ops import FromFunctionOp
from pytensor.graph.basic import Apply, Variable
from pytensor.graph.fg import FunctionGraph
from pytensor.graph.features import ReplaceValidate
from pytensor.graph.op import Op
from pytensor.graph.rewriting.basic import GraphRewriter

from .op_async import run_coroutine_blocking
from .signatures import LogpFunc, LogpGradFunc

__all__ = [
    "AsyncOp",
    "AsyncFromFunctionOp",
    "ParallelAsyncOp",
    "ArraysToArraysOp",
    "AsyncArraysToArraysOp",
    "LogpOp",
    "AsyncLogpOp",
    "LogpGradOp",
    "AsyncLogpGradOp",
    "find_parallelizable_applies",
    "parallelize_async_applies",
    "parallelize_all_async_applies",
    "AsyncFusionOptimizer",
]


class AsyncOp(Op):
    """An Op whose ``perform`` drives ``perform_async`` to completion.

    Parity: reference op_async.py:16-34.
    """

    def perform(self, node, inputs, output_storage, params=None):
        run_coroutine_blocking(self.perform_async(node, inputs, output_storage, params))

    async def perform_async(self, node, inputs, output_storage, params=None):
        raise NotImplementedError()


class AsyncFromFunctionOp(AsyncOp, FromFunctionOp):
    """FromFunctionOp wrapping a coroutine function (reference op_async.py:37-65)."""

    def __init__(self, fn, itypes, otypes, infer_shape=None):
        if not asyncio.iscoroutinefunction(fn):
            raise ValueError("`fn` must be a coroutine function.")
        super().__init__(fn, itypes, otypes, infer_shape)

    async def perform_async(self, node, inputs, output_storage, params=None):
        outs = await self._FromFunctionOp__fn(*inputs)
        if not isinstance(outs, (list, tuple)):
            outs = (outs,)
        for i, out in enumerate(outs):
            output_storage[i][0] = out


class ParallelAsyncOp(AsyncOp):
    """Fuses N independent AsyncOp applies into one concurrently-run apply.

    Parity: reference op_async.py:68-132.  ``perform_async`` slices inputs
    and output storage per child apply and gathers the child coroutines on
    one loop -- N concurrent RPC round trips.
    """

    def __init__(self, applies: Sequence[Apply]) -> None:
        self.applies = list(applies)
        for app in self.applies:
            if not isinstance(app.op, AsyncOp):
                raise ValueError(f"Apply {app} does not wrap an AsyncOp.")
        super().__init__()

    def make_node(self, *inputs: Variable) -> Apply:
        expected = [inp for app in self.applies for inp in app.inputs]
        if len(inputs) != len(expected):
            raise ValueError(
                f"ParallelAsyncOp expected {len(expected)} inputs, got {len(inputs)}."
            )
        outputs = [out.type() for app in self.applies for out in app.outputs]
        return Apply(self, list(inputs), outputs)

    async def perform_async(self, node, inputs, output_storage, params=None):
        coros = []
        i = o = 0
        for app in self.applies:
            ni, no = len(app.inputs), len(app.outputs)
            coros.append(
                app.op.perform_async(app, inputs[i : i + ni], output_storage[o : o + no])
            )
            i += ni
            o += no
        await asyncio.gather(*coros)


class ArraysToArraysOp(FromFunctionOp):
    """Generic graph embedding of a ComputeFunc (reference wrapper_ops.py:14-41)."""

    def __init__(self, compute_func, itypes, otypes, infer_shape=None):
        super().__init__(compute_func, itypes, otypes, infer_shape)

    def make_node(self, *inputs: Variable) -> Apply:
        return super().make_node(*[at.as_tensor_variable(i) for i in inputs])


class AsyncArraysToArraysOp(AsyncFromFunctionOp):
    def make_node(self, *inputs: Variable) -> Apply:
        return super().make_node(*[at.as_tensor_variable(i) for i in inputs])


class LogpOp(Op):
    """Wraps a LogpFunc; one scalar output, no gradient.

    Parity: reference wrapper_ops.py:44-81.
    """

    _props = ("_logp_func",)

    def __init__(self, logp_func: LogpFunc) -> None:
        self._logp_func = logp_func
        super().__init__()

    def make_node(self, *inputs: Union[Variable, int, float, np.ndarray]) -> Apply:
        inputs = [at.as_tensor_variable(i) for i in inputs]
        return Apply(self, inputs, [at.dscalar().type()])

    def perform(self, node, inputs, output_storage, params=None):
        output_storage[0][0] = np.asarray(self._logp_func(*inputs))


class AsyncLogpOp(AsyncOp, LogpOp):
    async def perform_async(self, node, inputs, output_storage, params=None):
        output_storage[0][0] = np.asarray(await self._logp_func(*inputs))


class LogpGradOp(Op):
    """The flagship Op: fused remote logp + gradients.

    Parity: reference wrapper_ops.py:84-146 -- outputs
    ``[scalar logp, grad_per_input...]``; ``grad()`` re-applies itself on the
    same inputs (merged with the forward node by CSE because ``_props`` makes
    equal-func Ops compare equal) and returns ``[g_logp * g for g in grads]``.
    """

    _props = ("_logp_grad_func",)

    def __init__(self, logp_grad_func: LogpGradFunc) -> None:
        self._logp_grad_func = logp_grad_func
        super().__init__()

    def make_node(self, *inputs: Union[Variable, int, float, np.ndarray]) -> Apply:
        inputs = [at.as_tensor_variable(i) for i in inputs]
        outputs = [at.dscalar().type()] + [i.type() for i in inputs]
        return Apply(self, inputs, outputs)

    def perform(self, node, inputs, output_storage, params=None):
        logp, gradients = self._logp_grad_func(*inputs)
        output_storage[0][0] = np.asarray(logp)
        for g, gradient in enumerate(gradients):
            output_storage[1 + g][0] = np.asarray(gradient)

    def grad(self, inputs: Sequence[Variable], output_grads: List[Variable]) -> List[Variable]:
        from pytensor.gradient import DisconnectedType

        g_logp, *g_grads = output_grads
        for i, g in enumerate(g_grads):
            if not isinstance(g.type, DisconnectedType):
                raise ValueError(f"Can't propagate gradients with respect to gradient output {i}.")
        _, *gradients = self(*inputs)
        return [g_logp * g for g in gradients]

    def connection_pattern(self, node):
        # logp depends on every input; the gradient outputs are terminal.
        n_in = len(node.inputs)
        return [[True] + [False] * n_in for _ in range(n_in)]


class AsyncLogpGradOp(AsyncOp, LogpGradOp):
    async def perform_async(self, node, inputs, output_storage, params=None):
        logp, gradients = await self._logp_grad_func(*inputs)
        output_storage[0][0] = np.asarray(logp)
        for g, gradient in enumerate(gradients):
            output_storage[1 + g][0] = np.asarray(gradient)


# -- graph rewrite: fuse independent async applies --------------------------


def _apply_depends_on(apply: Apply, others: Sequence[Apply]) -> bool:
    """True if ``apply`` (transitively) consumes any output of ``others``."""
    frontier = list(apply.inputs)
    seen = set()
    targets = {id(out) for other in others for out in other.outputs}
    while frontier:
        var = frontier.pop()
        if id(var) in seen:
            continue
        seen.add(id(var))
        if id(var) in targets:
            return True
        if var.owner is not None:
            frontier.extend(var.owner.inputs)
    return False


def find_parallelizable_applies(fg: FunctionGraph, op_cls: type = AsyncOp) -> List[Apply]:
    """Mutually independent AsyncOp applies (reference op_async.py:135-167)."""
    candidates: List[Apply] = []
    for apply in fg.toposort():
        if not isinstance(apply.op, op_cls):
            continue
        if isinstance(apply.op, ParallelAsyncOp):
            continue
        if not _apply_depends_on(apply, candidates):
            candidates.append(apply)
    return candidates if len(candidates) > 1 else []


def parallelize_async_applies(fg: FunctionGraph, applies: Sequence[Apply]) -> None:
    """Replace independent applies by one ParallelAsyncOp apply.

    Parity: reference op_async.py:170-195.
    """
    pop = ParallelAsyncOp(applies)
    flat_inputs = [inp for app in applies for inp in app.inputs]
    new_apply = pop.make_node(*flat_inputs)
    old_outputs = [out for app in applies for out in app.outputs]
    fg.replace_all_validate(
        list(zip(old_outputs, new_apply.outputs)), reason="parallelize_async_applies"
    )


def parallelize_all_async_applies(fg: FunctionGraph) -> None:
    """Repeat fusion until fixpoint -> layers of parallel groups.

    Parity: reference op_async.py:198-213.
    """
    while True:
        applies = find_parallelizable_applies(fg)
        if not applies:
            return
        parallelize_async_applies(fg, applies)


class AsyncFusionOptimizer(GraphRewriter):
    """Graph rewriter fusing independent async applies (op_async.py:216-224)."""

    def add_requirements(self, fgraph: FunctionGraph):
        fgraph.attach_feature(ReplaceValidate())

    def apply(self, fgraph: FunctionGraph):
        parallelize_all_async_applies(fgraph)


def register_fuse_asyncs() -> None:
    """Register the fusion rewrite into PyTensor's optdb as ``fuse_asyncs``.

    Same registration as reference op_async.py:227-234 (tag ``fast_run``,
    position 90) so every default-mode compile gets the fan-out fusion.
    Called at import when pytensor is present.
    """
    from pytensor.compile.mode import optdb

    try:
        optdb.register("fuse_asyncs", AsyncFusionOptimizer(), "fast_run", position=90)
    except Exception:
        pass  # already registered (module re-import)


register_fuse_asyncs()
