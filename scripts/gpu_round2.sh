#!/bin/bash
# GPU call 2: validate fp64-accum fix + fused-path perf; kernel-level timings.
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
python -m pytensor_federated_amd.ops.build 2>&1 | tail -1
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -6 | tee gpurun_out/pytest_gpu2.log
timeout 300 python bench.py --steps 500 --warmup 50 > gpurun_out/bench2_n1.json 2> gpurun_out/bench2_n1.err
tail -1 gpurun_out/bench2_n1.json
timeout 300 python bench.py --steps 500 --warmup 50 --no-readback > gpurun_out/bench2_norb.json 2>&1
tail -1 gpurun_out/bench2_norb.json
timeout 300 python bench.py --model logistic --rows 12500000 --steps 40 --warmup 5 > gpurun_out/bench2_logistic.json 2> gpurun_out/bench2_logistic.err
tail -1 gpurun_out/bench2_logistic.json
# micro-breakdown: where does a call spend time?
timeout 300 python - <<'PY' 2>&1 | tee gpurun_out/latency_breakdown.log
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
def bench(label, fn, n=300):
    fn(); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    print(f"{label}: {(time.perf_counter()-t0)/n*1e6:.1f} us")
bench("raw ctypes kernel call (async)", lambda: gaussian_linear_logp_grad(m._x, m._y, 1.5, 0.5, 0.4, out=out))
def kernel_sync():
    gaussian_linear_logp_grad(m._x, m._y, 1.5, 0.5, 0.4, out=out); torch.cuda.synchronize()
bench("kernel + sync", kernel_sync)
def kernel_readback():
    gaussian_linear_logp_grad(m._x, m._y, 1.5, 0.5, 0.4, out=out); host.copy_(out); torch.cuda.synchronize()
bench("kernel + pinned readback + sync", kernel_readback)
bench("engine.logp_grad_fused (async)", lambda: e.logp_grad_fused(1.5, 0.5))
def engine_rb():
    host.copy_(e.logp_grad_fused(1.5, 0.5)); torch.cuda.synchronize()
bench("engine fused + readback + sync", engine_rb)
PY
# kernel-level stats as CSV
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/prof2" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 100 --warmup 10 > "$GRAFT_REPO_ROOT/gpurun_out/rocprof2.log" 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/prof2_logistic" -- python "$GRAFT_REPO_ROOT/bench.py" --model logistic --rows 12500000 --steps 20 --warmup 3 >> "$GRAFT_REPO_ROOT/gpurun_out/rocprof2.log" 2>&1
find "$GRAFT_REPO_ROOT/gpurun_out/prof2" "$GRAFT_REPO_ROOT/gpurun_out/prof2_logistic" -name "*stats*" | head
for f in $(find "$GRAFT_REPO_ROOT/gpurun_out/prof2" "$GRAFT_REPO_ROOT/gpurun_out/prof2_logistic" -name "*kernel_stats*"); do echo "== $f"; head -8 "$f"; done
