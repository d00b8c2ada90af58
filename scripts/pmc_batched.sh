#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
python -m pytensor_federated_amd.ops.build >/dev/null 2>&1
export TMPDIR=/tmp
cd /tmp
cat > /tmp/run_batched.py <<PY
import sys; sys.path.insert(0, "$GRAFT_REPO_ROOT")
import torch
from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset
X, y, _ = generate_logistic_dataset(2_000_000, 1024, seed=70)
m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
theta16 = torch.randn(1024, 16, device="cuda:0")*0.3
for _ in range(10): m.logp_grad_batched(theta16)
torch.cuda.synchronize()
PY
# counters-only runs (one set per pass; SQ slots limited)
timeout 200 rocprofv3 --pmc SQ_VALU_MFMA_BUSY_CYCLES,SQ_WAVE_CYCLES,SQ_LDS_BANK_CONFLICT,SQ_WAIT_ANY --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/pmc_batched" -- python /tmp/run_batched.py > "$GRAFT_REPO_ROOT/gpurun_out/pmc_batched.log" 2>&1
find "$GRAFT_REPO_ROOT/gpurun_out/pmc_batched" -name "*.csv" | head -3
for f in $(find "$GRAFT_REPO_ROOT/gpurun_out/pmc_batched" -name "*counter*" -o -name "*pmc*" | head -2); do echo "== $f"; head -20 "$f"; done
