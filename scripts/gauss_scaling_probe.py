import sys, pathlib, time
sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))
import torch
from pytensor_federated_amd.models import GaussianLinearModel

for n in [5_000_000, 10_000_000, 20_000_000, 40_000_000, 80_000_000]:
    x = (torch.rand(n, device="cuda") * 10).to(torch.bfloat16)
    y = (1.5 + 0.5 * x.float() + torch.randn(n, device="cuda") * 0.4).to(torch.bfloat16)
    m = GaussianLinearModel(x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16)
    m.logp_grad_sync(1.5, 0.5)
    t0 = time.perf_counter()
    iters = 300 if n <= 20_000_000 else 100
    for _ in range(iters):
        m.logp_grad_sync(1.5, 0.5)
    per = (time.perf_counter() - t0) / iters
    bw = 4 * n / per / 1e12
    print(f"N={n:>9}: {per*1e6:7.1f} us/call  eff-BW {bw:5.2f} TB/s")
    del x, y, m
    torch.cuda.empty_cache()
