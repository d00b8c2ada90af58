"""Async execution engine with automatic fan-out fusion.

The reference implements this layer as PyTensor graph machinery: ``AsyncOp``
runs a coroutine from a sync ``perform`` (op_async.py:16-34), independent
async applies are fused into a ``ParallelAsyncOp`` whose RPCs run
concurrently via ``asyncio.gather`` (op_async.py:68-132), and a graph
rewrite finds independent layers until fixpoint (op_async.py:135-234).

This MI355X-native rebuild keeps the exact semantics but owns the graph:
:class:`AsyncTaskGraph` is a small DAG of async compute nodes; compiling it
performs the same independence analysis (:func:`fuse_parallel_layers` --
the analog of ``parallelize_all_async_applies``) and evaluation runs each
fused layer with one ``asyncio.gather`` -- N concurrent gRPC round trips,
or, on a GPU worker, N shard computations issued to N HIP streams (see
``parallel.streams``).  Optional PyTensor adapters that register the same
rewrite into PyTensor's optdb live in ``wrapper_ops``.
"""
from __future__ import annotations

import asyncio
import inspect
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple, Union

from .utils import get_useful_event_loop

__all__ = [
    "run_coroutine_blocking",
    "gather_evaluate",
    "AsyncComputeNode",
    "AsyncTaskGraph",
    "find_parallelizable_nodes",
    "fuse_parallel_layers",
]


def run_coroutine_blocking(coro):
    """Run a coroutine to completion from sync code, even under a running loop.

    The analog of ``AsyncOp.perform`` (reference op_async.py:16-34): uses
    :func:`get_useful_event_loop` so nested invocation (Jupyter, PyMC
    samplers) re-enters via nest_asyncio.
    """
    loop = get_useful_event_loop()
    return loop.run_until_complete(coro)


async def gather_evaluate(
    funcs: Sequence[Callable[..., Any]],
    inputs_per_func: Sequence[Sequence[Any]],
) -> List[Any]:
    """Run N async functions concurrently on one loop; returns their results.

    The fan-out primitive of the framework: the analog of
    ``ParallelAsyncOp.perform``'s gather (reference op_async.py:107-132).
    """
    if len(funcs) != len(inputs_per_func):
        raise ValueError("funcs and inputs_per_func must have equal length.")
    return list(
        await asyncio.gather(*(f(*args) for f, args in zip(funcs, inputs_per_func)))
    )


#: an input reference: graph input name, or (node_name, output_index)
InputRef = Union[str, Tuple[str, int]]


@dataclass
class AsyncComputeNode:
    """One node of an :class:`AsyncTaskGraph`.

    ``fn`` is an async (or sync) callable taking the resolved input arrays
    and returning a sequence of outputs.  ``is_async`` nodes are candidates
    for fan-out fusion (like the reference's ``AsyncOp`` applies); sync
    nodes act as barriers, like non-async Ops do in the reference's rewrite
    (op_async.py:135-167 only collects ``AsyncOp`` instances).
    """

    name: str
    fn: Callable[..., Any]
    inputs: List[InputRef] = field(default_factory=list)
    n_outputs: int = 1

    @property
    def is_async(self) -> bool:
        return inspect.iscoroutinefunction(self.fn)

    def depends_on(self, other: "AsyncComputeNode") -> bool:
        return any(isinstance(ref, tuple) and ref[0] == other.name for ref in self.inputs)


class AsyncTaskGraph:
    """A DAG of compute nodes with automatic parallel-layer fusion.

    Usage::

        g = AsyncTaskGraph()
        g.add_node("shard0", client0.evaluate_async, inputs=["theta"])
        g.add_node("shard1", client1.evaluate_async, inputs=["theta"])
        g.add_node("total", lambda a, b: [a[0] + b[0]],
                   inputs=[("shard0", 0), ("shard1", 0)])
        g.set_outputs([("total", 0)])
        result = g.evaluate(theta=theta)      # shard0 ∥ shard1, then total

    ``compile()`` groups mutually independent async nodes into layers
    (fixpoint, like ``parallelize_all_async_applies``, reference
    op_async.py:198-213); ``evaluate`` runs each layer concurrently.
    """

    def __init__(self) -> None:
        self._nodes: Dict[str, AsyncComputeNode] = {}
        self._outputs: List[InputRef] = []
        self._layers: Optional[List[List[str]]] = None

    # -- construction --------------------------------------------------
    def add_node(
        self,
        name: str,
        fn: Callable[..., Any],
        inputs: Sequence[InputRef] = (),
        n_outputs: int = 1,
    ) -> AsyncComputeNode:
        if name in self._nodes:
            raise ValueError(f"Duplicate node name {name!r}.")
        for ref in inputs:
            if isinstance(ref, tuple) and ref[0] not in self._nodes:
                raise ValueError(f"Node {name!r} references unknown node {ref[0]!r}.")
        node = AsyncComputeNode(name, fn, list(inputs), n_outputs)
        self._nodes[name] = node
        self._layers = None
        return node

    def set_outputs(self, outputs: Sequence[InputRef]) -> None:
        self._outputs = list(outputs)
        self._layers = None

    @property
    def nodes(self) -> Dict[str, AsyncComputeNode]:
        return dict(self._nodes)

    # -- analysis ------------------------------------------------------
    def toposort(self) -> List[str]:
        order: List[str] = []
        seen: Dict[str, int] = {}  # 0=visiting, 1=done

        def visit(name: str):
            state = seen.get(name)
            if state == 1:
                return
            if state == 0:
                raise ValueError(f"Cycle detected at node {name!r}.")
            seen[name] = 0
            for ref in self._nodes[name].inputs:
                if isinstance(ref, tuple):
                    visit(ref[0])
            seen[name] = 1
            order.append(name)

        for name in self._nodes:
            visit(name)
        return order

    def compile(self) -> List[List[str]]:
        """Compute the fused parallel layers (cached)."""
        if self._layers is None:
            self._layers = fuse_parallel_layers(self)
        return self._layers

    # -- execution -----------------------------------------------------
    async def evaluate_async(self, **graph_inputs) -> List[Any]:
        layers = self.compile()
        values: Dict[str, Any] = {}  # node name -> list of outputs

        def resolve(ref: InputRef) -> Any:
            if isinstance(ref, tuple):
                name, idx = ref
                return values[name][idx]
            try:
                return graph_inputs[ref]
            except KeyError:
                raise KeyError(f"Missing graph input {ref!r}.") from None

        for layer in layers:
            coros = []
            sync_results: List[Tuple[str, Any]] = []
            async_names: List[str] = []
            for name in layer:
                node = self._nodes[name]
                args = [resolve(ref) for ref in node.inputs]
                if node.is_async:
                    coros.append(node.fn(*args))
                    async_names.append(name)
                else:
                    sync_results.append((name, node.fn(*args)))
            if coros:
                results = await asyncio.gather(*coros)
                for name, res in zip(async_names, results):
                    values[name] = _as_output_list(res)
            for name, res in sync_results:
                values[name] = _as_output_list(res)
        return [resolve(ref) for ref in self._outputs]

    def evaluate(self, **graph_inputs) -> List[Any]:
        return run_coroutine_blocking(self.evaluate_async(**graph_inputs))

    __call__ = evaluate


def _as_output_list(res: Any) -> List[Any]:
    if isinstance(res, (list, tuple)):
        return list(res)
    return [res]


def find_parallelizable_nodes(graph: AsyncTaskGraph, done: set) -> List[str]:
    """Mutually independent async nodes whose inputs are all computed.

    The analog of ``find_parallelizable_applies`` (reference
    op_async.py:135-167): scan in topological order, collect async nodes
    that do not depend on any not-yet-computed node.
    """
    nodes = graph.nodes
    ready: List[str] = []
    for name in graph.toposort():
        if name in done:
            continue
        node = nodes[name]
        deps_done = all(
            (not isinstance(ref, tuple)) or ref[0] in done for ref in node.inputs
        )
        if deps_done and node.is_async:
            ready.append(name)
    return ready


def fuse_parallel_layers(graph: AsyncTaskGraph) -> List[List[str]]:
    """Group the graph into layers of concurrently-runnable nodes (fixpoint).

    The analog of ``parallelize_all_async_applies`` (reference
    op_async.py:198-213): each layer contains either one batch of mutually
    independent *async* nodes (executed with a single ``asyncio.gather``)
    or ready sync nodes.  Repeats until every node is placed.
    """
    nodes = graph.nodes
    done: set = set()
    layers: List[List[str]] = []
    order = graph.toposort()
    while len(done) < len(nodes):
        async_ready = find_parallelizable_nodes(graph, done)
        if async_ready:
            layers.append(async_ready)
            done.update(async_ready)
            continue
        # No async node ready: place the next ready sync nodes as one layer.
        sync_ready = []
        for name in order:
            if name in done:
                continue
            node = nodes[name]
            deps_done = all(
                (not isinstance(ref, tuple)) or ref[0] in done for ref in node.inputs
            )
            if deps_done and not node.is_async:
                sync_ready.append(name)
        if not sync_ready:
            raise ValueError("Graph has unsatisfiable dependencies (cycle?).")
        layers.append(sync_ready)
        done.update(sync_ready)
    return layers
