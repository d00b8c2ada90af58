#!/bin/bash
# Round-2 GPU call 12: v3s (K=512) validation + batched suite + timing.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 400 python -m pytest tests/test_gpu.py -q -k "batched or Logistic" 2>&1 | tail -2 | tee gpurun_out/r2c12_pytest.log
# K=512 A/B: v2 vs v3s at 2e6 rows (generate_logistic_dataset with 512 features)
timeout 300 python - > gpurun_out/r2c12_k512.json 2>gpurun_out/r2c12_k512.err <<'PYEOF'
import json, os, time, torch
from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset
X, y, _ = generate_logistic_dataset(2_000_000, 512, seed=71)
m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
th = torch.randn(512, 16, device="cuda:0", generator=torch.Generator(device="cuda:0").manual_seed(2)) * 0.3
def run(v3):
    os.environ["FED_BATCHED_V3"] = "1" if v3 else "0"
    logp, G = m.logp_grad_batched(th); torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50): logp, G = m.logp_grad_batched(th)
    torch.cuda.synchronize()
    return logp.clone(), G.clone(), (time.perf_counter()-t0)/50
l2, g2, t2 = run(False)
l3, g3, t3 = run(True)
# partial-tile sizes too
extra = {}
for rows in (97, 4093):
    Xp, yp, _ = generate_logistic_dataset(rows, 512, seed=9)
    mp = LogisticGLMModel(Xp, yp, device="cuda:0", dtype=torch.bfloat16)
    thp = torch.randn(512, 16, device="cuda:0", generator=torch.Generator(device="cuda:0").manual_seed(4)) * 0.3
    os.environ["FED_BATCHED_V3"] = "0"; l2p, g2p = mp.logp_grad_batched(thp)
    os.environ["FED_BATCHED_V3"] = "1"; l3p, g3p = mp.logp_grad_batched(thp)
    extra[rows] = {"logp_rel": float(((l2p-l3p).abs()/l2p.abs()).max()),
                   "grad_absmax": float((g2p-g3p).abs().max())}
print(json.dumps({"v2_ms": t2*1e3, "v3s_ms": t3*1e3,
                  "logp_max_rel": float(((l2-l3).abs()/l2.abs()).max()),
                  "grad_absmax": float((g2-g3).abs().max()),
                  "partial": extra}))
PYEOF
echo "=== results ==="
tail -2 gpurun_out/r2c12_pytest.log
cat gpurun_out/r2c12_k512.json
