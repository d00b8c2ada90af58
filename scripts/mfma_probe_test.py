import sys, pathlib; sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import ctypes, numpy as np, torch
from pytensor_federated_amd.ops import require_kernels, _stream_ptr
lib = require_kernels()
lib.fed_mfma_probe.restype = ctypes.c_int
lib.fed_mfma_probe.argtypes = [ctypes.c_void_p]*3 + [ctypes.c_void_p]
rng = np.random.RandomState(0)
A = rng.uniform(-1, 1, (16, 32)).astype(np.float32)
B = rng.uniform(-1, 1, (32, 16)).astype(np.float32)
At = torch.as_tensor(A).to(torch.bfloat16).cuda()
Bt = torch.as_tensor(B).to(torch.bfloat16).cuda()
D = torch.zeros(16, 16, dtype=torch.float32, device="cuda")
rc = lib.fed_mfma_probe(At.data_ptr(), Bt.data_ptr(), D.data_ptr(), _stream_ptr())
assert rc == 0, rc
torch.cuda.synchronize()
ref = (At.float() @ Bt.float()).cpu().numpy()
got = D.cpu().numpy()
err = np.abs(got - ref).max()
print("max abs err:", err)
if err < 1e-3: print("MFMA FRAGMENT MAP OK")
else:
    print("MISMATCH; got[0,:4]", got[0,:4], "ref[0,:4]", ref[0,:4])
    print("transposed match?", np.abs(got.T - ref).max())
