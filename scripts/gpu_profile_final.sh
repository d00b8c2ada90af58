#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
python -m pytensor_federated_amd.ops.build >/dev/null 2>&1
export TMPDIR=/tmp
cd /tmp
# flagship gaussian (graphed path)
timeout 200 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/proff_gauss" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 300 --warmup 30 > "$GRAFT_REPO_ROOT/gpurun_out/proff.log" 2>&1
# logistic single-chain
timeout 200 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/proff_logistic" -- python "$GRAFT_REPO_ROOT/bench.py" --model logistic --rows 12500000 --steps 20 --warmup 3 >> "$GRAFT_REPO_ROOT/gpurun_out/proff.log" 2>&1
# batched logistic + ODE graphed
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/proff_batched" -- python - >> "$GRAFT_REPO_ROOT/gpurun_out/proff.log" 2>&1 <<PY
import sys; sys.path.insert(0, "$GRAFT_REPO_ROOT")
import torch
from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset
X, y, _ = generate_logistic_dataset(2_000_000, 1024, seed=70)
m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
theta16 = torch.randn(1024, 16, device="cuda:0")*0.3
for _ in range(25): m.logp_grad_batched(theta16)
torch.cuda.synchronize()
PY
for f in $(find "$GRAFT_REPO_ROOT/gpurun_out/proff_gauss" "$GRAFT_REPO_ROOT/gpurun_out/proff_logistic" "$GRAFT_REPO_ROOT/gpurun_out/proff_batched" -name "*kernel_stats*"); do echo "== $f"; head -5 "$f"; done
# PMC counters for the two hot kernels (separate pass, counters only)
timeout 300 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_BUSY_CYCLES --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/proff_pmc" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 50 --warmup 5 >> "$GRAFT_REPO_ROOT/gpurun_out/proff.log" 2>&1 || true
