#!/bin/bash
# Round-2 GPU call 1: validate suite + A/B the batched-logistic v2 kernel.
# Usage (via gpurun): bash scripts/gpu_r2_batched.sh
set -x
cd /root/repo
mkdir -p gpurun_out

# 1. full GPU test suite (catches regressions from grid/workspace changes)
timeout 600 python -m pytest tests -m gpu -x -q 2>&1 | tail -5 | tee gpurun_out/r2c1_pytest.log

# 2. A/B: v1 vs v2 batched kernel, 2e6x1024 (round-1 reference point)
timeout 300 bash -c 'FED_BATCHED_V1=1 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50' \
    > gpurun_out/r2c1_batched_v1.json 2>gpurun_out/r2c1_batched_v1.err
timeout 300 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 \
    > gpurun_out/r2c1_batched_v2.json 2>gpurun_out/r2c1_batched_v2.err

# 3. numerics cross-check v1 vs v2 at full config-4 shard size (1.25e7 rows)
timeout 600 python - > gpurun_out/r2c1_v1v2_check.json 2>gpurun_out/r2c1_v1v2_check.err <<'EOF'
import json, os, time, torch
from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset

rows = 12_500_000
X, y, _ = generate_logistic_dataset(rows, 1024, seed=70)
m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
theta16 = torch.randn(1024, 16, device="cuda:0", generator=torch.Generator(device="cuda:0").manual_seed(3)) * 0.3

def run(v1):
    os.environ.pop("FED_BATCHED_V1", None)
    if v1:
        os.environ["FED_BATCHED_V1"] = "1"
    logp, G = m.logp_grad_batched(theta16)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        logp, G = m.logp_grad_batched(theta16)
    torch.cuda.synchronize()
    return logp.clone(), G.clone(), (time.perf_counter() - t0) / 20

l1, g1, t1 = run(True)
l2, g2, t2 = run(False)
print(json.dumps({
    "rows": rows,
    "v1_ms": t1 * 1e3, "v2_ms": t2 * 1e3,
    "logp_max_rel": float(((l1 - l2).abs() / l1.abs()).max()),
    "grad_max_rel": float(((g1 - g2).abs() / (g1.abs() + 1e-6)).max()),
    "logp_v1_0": float(l1[0]), "logp_v2_0": float(l2[0]),
}))
EOF

# 4. rocprof kernel stats for the v2 kernel (counters-free run)
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c1_prof -- \
    python benchmarks/bench_batched_chains.py --rows 2000000 --steps 30 \
    > gpurun_out/r2c1_prof_run.log 2>&1 || true
# keep only the stats csv (trace files can be large)
find gpurun_out/r2c1_prof -name "*kernel_trace*" -delete 2>/dev/null || true

echo "=== results ==="
cat gpurun_out/r2c1_batched_v1.json gpurun_out/r2c1_batched_v2.json gpurun_out/r2c1_v1v2_check.json
tail -3 gpurun_out/r2c1_pytest.log
