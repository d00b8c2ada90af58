import sys, pathlib, time
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
from pytensor_federated_amd.ops import PersistentLinearEngine

x, y = generate_linear_dataset(1_000_000, seed=91)
m = GaussianLinearModel(x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16)
eng = PersistentLinearEngine(m._x, m._y, 0.4)
try:
    for a, b in [(1.5, 0.5), (0.3, -0.2), (2.0, 1.0)]:
        got = eng.logp_grad_sync(a, b)
        ref = m.logp_grad_sync(a, b)
        for g, r in zip(got, ref):
            assert abs(g - r) <= 1e-9 * max(1, abs(r)), (a, b, got, ref)
    print("correctness OK (1e6)")
    ref = eng.logp_grad_sync(1.0, 0.5)
    for i in range(2000):
        assert eng.logp_grad_sync(1.0, 0.5) == ref, i
    print("2000 repeated calls stable")
finally:
    eng.close()

x, y = generate_linear_dataset(10_000_000, seed=92)
m = GaussianLinearModel(x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16)
eng = PersistentLinearEngine(m._x, m._y, 0.4)
try:
    got = eng.logp_grad_sync(1.5, 0.5)
    ref = m.logp_grad_sync(1.5, 0.5)
    for g, r in zip(got, ref):
        assert abs(g - r) <= 1e-9 * max(1, abs(r)), (got, ref)
    print("correctness OK (1e7)")
    for n_it in (2000, 2000):
        t0 = time.perf_counter()
        for _ in range(n_it):
            eng.logp_grad_sync(1.5, 0.5)
        per = (time.perf_counter() - t0) / n_it
        print(f"persistent @1e7 bf16: {per*1e6:.1f} us/call = {1/per:.0f} calls/s")
finally:
    eng.close()
print("PROBE DONE")
