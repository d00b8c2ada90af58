"""Structural tests for the PyTensor adapters against tests/pytensor_stub.

pytensor itself is not installable in the ROCm image, so these run
wrapper_ops against a minimal faithful stub of the graph API (real
toposort, real consumer rewiring) IN A SUBPROCESS -- verifying the
adapter's own logic: perform() storage layout, ParallelAsyncOp
input/output slicing, the dependence scan, fusion-to-fixpoint layering,
and the optdb registration.  The production-tested equivalents live in
torch_ops.py / op_async.py (see COVERAGE.md).
"""
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

SCRIPT = r"""
import sys
sys.path.insert(0, r"%(repo)s")
sys.path.insert(0, r"%(tests)s")
import pytensor_stub
pytensor_stub.install()

import asyncio
import numpy as np
from pytensor_stub import FunctionGraph, TensorType, Variable, optdb
import pytensor_federated_amd.wrapper_ops as W

# optdb registration happened at import
assert "fuse_asyncs" in optdb.registered, optdb.registered

dbl = TensorType("float64", ())

# ---- LogpGradOp.perform: [logp, grad0, grad1] storage layout ----------
def lgf(a, b):
    return np.asarray(-(a * a + b * b)), [np.asarray(-2 * a), np.asarray(-2 * b)]

op = W.LogpGradOp(lgf)
node = op.make_node(Variable(dbl), Variable(dbl))
assert len(node.outputs) == 3
storage = [[None], [None], [None]]
op.perform(node, [2.0, 3.0], storage)
assert float(storage[0][0]) == -13.0
assert float(storage[1][0]) == -4.0 and float(storage[2][0]) == -6.0

# equal-func ops compare equal -> CSE-mergeable (reference _props parity)
assert W.LogpGradOp(lgf)._logp_grad_func == op._logp_grad_func

# ---- AsyncFromFunctionOp: coroutine perform ---------------------------
calls = []
async def afn(x):
    calls.append(float(x))
    await asyncio.sleep(0.01)
    return np.asarray(x * 10.0)

aop = W.AsyncFromFunctionOp(afn, [dbl], [dbl])
anode = aop.make_node(Variable(dbl))
st = [[None]]
aop.perform(anode, [4.0], st)
assert float(st[0][0]) == 40.0 and calls == [4.0]

# rejects plain functions
try:
    W.AsyncFromFunctionOp(lambda x: x, [dbl], [dbl])
    raise AssertionError("expected ValueError")
except ValueError:
    pass

# ---- ParallelAsyncOp: slicing + concurrent gather ---------------------
import time
async def slow(x):
    await asyncio.sleep(0.15)
    return np.asarray(x + 1.0)

ops = [W.AsyncFromFunctionOp(slow, [dbl], [dbl]) for _ in range(3)]
applies = [o.make_node(Variable(dbl)) for o in ops]
pop = W.ParallelAsyncOp(applies)
pnode = pop.make_node(*[inp for app in applies for inp in app.inputs])
assert len(pnode.outputs) == 3
storage = [[None], [None], [None]]
t0 = time.perf_counter()
pop.perform(pnode, [1.0, 2.0, 3.0], storage)
wall = time.perf_counter() - t0
assert [float(s[0]) for s in storage] == [2.0, 3.0, 4.0]
assert wall < 0.40, f"3x sleep(0.15) must gather concurrently, took {wall:.3f}s"

# ---- fusion: two independent layers fuse to fixpoint ------------------
x = Variable(dbl)
l1 = [o.make_node(x) for o in (W.AsyncFromFunctionOp(slow, [dbl], [dbl]),
                               W.AsyncFromFunctionOp(slow, [dbl], [dbl]))]
# layer 2 consumes layer-1 outputs -> NOT parallelizable with layer 1
l2op = W.AsyncFromFunctionOp(slow, [dbl], [dbl])
l2 = l2op.make_node(l1[0].outputs[0])
fg = FunctionGraph([x], [l1[1].outputs[0], l2.outputs[0]])

found = W.find_parallelizable_applies(fg)
assert set(id(a) for a in found) == set(id(a) for a in l1), "layer 1 only"

W.parallelize_all_async_applies(fg)
# after fixpoint: layer 1 fused into one ParallelAsyncOp; l2 consumes it
tops = fg.toposort()
par = [a for a in tops if isinstance(a.op, W.ParallelAsyncOp)]
assert len(par) == 1 and len(par[0].outputs) == 2
l2_after = [a for a in tops if a.op is l2op][0]
assert l2_after.inputs[0].owner is par[0], "consumer rewired to fused node"
# second scan finds nothing new (fixpoint; single apply is not a group)
assert W.find_parallelizable_applies(fg) == []

print("WRAPPER_OPS_STUB_OK")
"""


def test_wrapper_ops_against_stub():
    proc = subprocess.run(
        [sys.executable, "-c", SCRIPT % {"repo": str(REPO), "tests": str(REPO / "tests")}],
        capture_output=True, text=True, timeout=240,
    )
    assert proc.returncode == 0, f"stdout:\n{proc.stdout}\nstderr:\n{proc.stderr}"
    assert "WRAPPER_OPS_STUB_OK" in proc.stdout
