"""Async engine semantics: fan-out concurrency + layer fusion.

Mirrors the reference's delay-op tests (test_op_async.py:36-206): parallel
evaluation of independent delays takes ~max instead of ~sum, and the layer
analysis fuses a diamond graph into two parallel layers.
"""
import asyncio
import time

import pytest

from pytensor_federated_amd.op_async import (
    AsyncTaskGraph,
    fuse_parallel_layers,
    gather_evaluate,
    run_coroutine_blocking,
)


def make_delay(delay: float):
    async def fn(x):
        # busy-wait like the reference's _AsyncDelay for precise timings,
        # but asyncio.sleep suffices for coarse windows
        await asyncio.sleep(delay)
        return [x + delay]

    return fn


def test_gather_evaluate_is_concurrent():
    f1, f2 = make_delay(0.25), make_delay(0.25)
    t0 = time.perf_counter()
    results = run_coroutine_blocking(gather_evaluate([f1, f2], [[0.0], [1.0]]))
    wall = time.perf_counter() - t0
    assert results == [[0.25], [1.25]]
    assert wall < 0.4, f"expected ~max(0.25) not sum, took {wall:.3f}s"


def test_blocking_run_from_sync():
    async def coro():
        return "ok"

    assert run_coroutine_blocking(coro()) == "ok"


def test_graph_two_independent_nodes_parallel():
    g = AsyncTaskGraph()
    g.add_node("a", make_delay(0.3), inputs=["x"])
    g.add_node("b", make_delay(0.2), inputs=["x"])
    g.add_node("sum", lambda a, b: [a + b], inputs=[("a", 0), ("b", 0)])
    g.set_outputs([("sum", 0)])
    layers = g.compile()
    assert layers == [["a", "b"], ["sum"]]
    t0 = time.perf_counter()
    (out,) = g.evaluate(x=1.0)
    wall = time.perf_counter() - t0
    assert out == pytest.approx(2.5)
    assert wall < 0.45, f"parallel(0.3,0.2) must be < 0.45s, took {wall:.3f}s"


def test_graph_fixpoint_two_layers():
    # diamond: (a ∥ b) -> (c ∥ d): 4 delays of 0.2 run in 2 layers (~0.4s)
    g = AsyncTaskGraph()
    g.add_node("a", make_delay(0.2), inputs=["x"])
    g.add_node("b", make_delay(0.2), inputs=["x"])
    g.add_node("c", make_delay(0.2), inputs=[("a", 0)])
    g.add_node("d", make_delay(0.2), inputs=[("b", 0)])
    g.set_outputs([("c", 0), ("d", 0)])
    layers = g.compile()
    assert layers == [["a", "b"], ["c", "d"]]
    t0 = time.perf_counter()
    c, d = g.evaluate(x=0.0)
    wall = time.perf_counter() - t0
    assert c == pytest.approx(0.4) and d == pytest.approx(0.4)
    assert wall < 0.7, f"two fused layers of 0.2 must be < 0.7s, took {wall:.3f}s"


def test_sync_nodes_act_as_barriers():
    g = AsyncTaskGraph()
    g.add_node("a", make_delay(0.05), inputs=["x"])
    g.add_node("mid", lambda a: [a * 2], inputs=[("a", 0)])
    g.add_node("b", make_delay(0.05), inputs=[("mid", 0)])
    g.set_outputs([("b", 0)])
    assert g.compile() == [["a"], ["mid"], ["b"]]
    (out,) = g.evaluate(x=1.0)
    assert out == pytest.approx((1.05 * 2) + 0.05)


def test_cycle_detection():
    g = AsyncTaskGraph()
    g.add_node("a", make_delay(0.01), inputs=["x"])
    # manual cycle injection
    g._nodes["a"].inputs = [("a", 0)]
    with pytest.raises(ValueError, match="[Cc]ycle"):
        g.toposort()


def test_duplicate_and_unknown_node_errors():
    g = AsyncTaskGraph()
    g.add_node("a", make_delay(0.01), inputs=["x"])
    with pytest.raises(ValueError, match="Duplicate"):
        g.add_node("a", make_delay(0.01))
    with pytest.raises(ValueError, match="unknown node"):
        g.add_node("b", make_delay(0.01), inputs=[("zzz", 0)])


def test_missing_graph_input():
    g = AsyncTaskGraph()
    g.add_node("a", make_delay(0.01), inputs=["x"])
    g.set_outputs([("a", 0)])
    with pytest.raises(KeyError, match="Missing graph input"):
        g.evaluate(y=1.0)


def test_fuse_parallel_layers_exposed():
    g = AsyncTaskGraph()
    g.add_node("a", make_delay(0.01), inputs=["x"])
    g.add_node("b", make_delay(0.01), inputs=["x"])
    g.set_outputs([("a", 0), ("b", 0)])
    assert fuse_parallel_layers(g) == [["a", "b"]]
