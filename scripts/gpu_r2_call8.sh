#!/bin/bash
# Round-2 GPU call 8: v3 (glds tile-resident) numerics + timing.
set -x
cd /root/repo
mkdir -p gpurun_out
# numerics first, tightly timed out
timeout 300 bash -c "FED_BATCHED_V3=1 python -m pytest tests/test_gpu.py -q -k batched" 2>&1 | tail -3 | tee gpurun_out/r2c8_pytest.log
# direct v2-vs-v3 value check + timing
timeout 300 python - > gpurun_out/r2c8_v2v3.json 2>gpurun_out/r2c8_v2v3.err <<'PYEOF'
import json, os, time, torch
from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset
X, y, _ = generate_logistic_dataset(2_000_000, 1024, seed=70)
m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
theta16 = torch.randn(1024, 16, device="cuda:0", generator=torch.Generator(device="cuda:0").manual_seed(3)) * 0.3
def run(env):
    os.environ.pop("FED_BATCHED_V3", None)
    if env: os.environ["FED_BATCHED_V3"] = "1"
    logp, G = m.logp_grad_batched(theta16); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(40): logp, G = m.logp_grad_batched(theta16)
    torch.cuda.synchronize()
    return logp.clone(), G.clone(), (time.perf_counter()-t0)/40
l2, g2, t2 = run(False)
l3, g3, t3 = run(True)
print(json.dumps({
    "v2_ms": t2*1e3, "v3_ms": t3*1e3,
    "logp_max_rel": float(((l2-l3).abs()/l2.abs()).max()),
    "grad_max_rel": float(((g2-g3).abs()/(g2.abs()+1e-6)).max()),
    "logp_v2_0": float(l2[0]), "logp_v3_0": float(l3[0]),
}))
PYEOF
# config-4 shard timing
timeout 300 bash -c "FED_BATCHED_V3=1 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 30" \
    > gpurun_out/r2c8_v3_big.json 2>gpurun_out/r2c8_v3_big.err
echo "=== results ==="
tail -3 gpurun_out/r2c8_pytest.log
cat gpurun_out/r2c8_v2v3.json gpurun_out/r2c8_v3_big.json
