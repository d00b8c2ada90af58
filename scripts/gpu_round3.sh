#!/bin/bash
# GPU call 3: validate slab-reduction kernels; re-profile; save committed profile summaries.
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
python -m pytensor_federated_amd.ops.build 2>&1 | tail -1
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee gpurun_out/pytest_gpu3.log
timeout 300 python bench.py --steps 1000 --warmup 100 > gpurun_out/bench3_n1.json 2> gpurun_out/bench3_n1.err
tail -1 gpurun_out/bench3_n1.json
timeout 300 python bench.py --model logistic --rows 12500000 --steps 60 --warmup 10 > gpurun_out/bench3_logistic.json 2> gpurun_out/bench3_logistic.err
tail -1 gpurun_out/bench3_logistic.json
timeout 300 python - <<'PY' 2>&1 | tee gpurun_out/latency_breakdown3.log
import time, torch
from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
from pytensor_federated_amd.parallel import FederatedShardEngine
from pytensor_federated_amd.ops import gaussian_linear_logp_grad
x, y = generate_linear_dataset(10_000_000, seed=1)
m = GaussianLinearModel(x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16)
e = FederatedShardEngine(m, use_distributed=False)
e.logp_grad_fused(1.5, 0.5); torch.cuda.synchronize()
out = torch.empty(3, dtype=torch.float64, device="cuda:0")
host = torch.empty(3, dtype=torch.float64, pin_memory=True)
def bench(label, fn, n=1000):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/n*1e6:.1f} us")
def kernel_sync():
    gaussian_linear_logp_grad(m._x, m._y, 1.5, 0.5, 0.4, out=out); torch.cuda.synchronize()
bench("kernel + sync", kernel_sync)
def kernel_readback():
    gaussian_linear_logp_grad(m._x, m._y, 1.5, 0.5, 0.4, out=out); host.copy_(out); torch.cuda.synchronize()
bench("kernel + pinned readback + sync", kernel_readback)
def engine_rb():
    host.copy_(e.logp_grad_fused(1.5, 0.5)); torch.cuda.synchronize()
bench("engine fused + readback + sync", engine_rb)
PY
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/prof3" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 200 --warmup 20 > "$GRAFT_REPO_ROOT/gpurun_out/rocprof3.log" 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/prof3_logistic" -- python "$GRAFT_REPO_ROOT/bench.py" --model logistic --rows 12500000 --steps 20 --warmup 3 >> "$GRAFT_REPO_ROOT/gpurun_out/rocprof3.log" 2>&1
for f in $(find "$GRAFT_REPO_ROOT/gpurun_out/prof3" "$GRAFT_REPO_ROOT/gpurun_out/prof3_logistic" -name "*kernel_stats*"); do echo "== $f"; head -6 "$f"; done
