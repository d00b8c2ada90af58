import sys, pathlib, time, ctypes
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import numpy as np, torch
from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
from pytensor_federated_amd.ops import PersistentLinearEngine, require_kernels

lib = require_kernels()
lib.fed_gaussian_persistent_debug.restype = ctypes.c_int
lib.fed_gaussian_persistent_debug.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_double), ctypes.POINTER(ctypes.c_double)]

def dump(eng, label):
    req = (ctypes.c_double * 8)(); res = (ctypes.c_double * 8)()
    lib.fed_gaussian_persistent_debug(eng._handle, req, res)
    ru = np.frombuffer(bytes(bytearray(req)), dtype=np.float64)
    ri = np.frombuffer(bytes(bytearray(res)), dtype=np.float64)
    stamps = np.frombuffer(bytes(bytearray(res)), dtype=np.uint64)
    print(f"  [{label}] req={ru[:4]} res={ri[:4]} stamp={stamps[4]} lastseq={stamps[5]} resseq={stamps[3]}")

x, y = generate_linear_dataset(1_000_000, seed=91)
m = GaussianLinearModel(x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16)
eng = PersistentLinearEngine(m._x, m._y, 0.4)
try:
    time.sleep(1.0)
    dump(eng, "after launch")
    try:
        got = eng.logp_grad_sync(1.5, 0.5)
        print("first eval:", got)
    except RuntimeError as ex:
        print("first eval failed:", ex)
        dump(eng, "after fail")
        raise SystemExit(1)
    dump(eng, "after eval")
    ref = m.logp_grad_sync(1.5, 0.5)
    print("ref       :", ref)
    t0 = time.perf_counter()
    for _ in range(1000):
        eng.logp_grad_sync(1.5, 0.5)
    per = (time.perf_counter() - t0) / 1000
    print(f"persistent @1e6: {per*1e6:.1f} us/call")
finally:
    rc = eng._lib.fed_gaussian_persistent_stop(eng._handle); eng._handle = None
    print("stop rc:", rc)
