#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 bash -c "FED_BATCHED_V3=1 python -m pytest tests/test_gpu.py -q -k batched" 2>&1 | tail -2 | tee gpurun_out/r2c10_pytest.log
timeout 300 bash -c "FED_BATCHED_V3=1 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 60" > gpurun_out/r2c10_v3_2e6.json 2>gpurun_out/r2c10.err
timeout 300 bash -c "FED_BATCHED_V3=1 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 40" > gpurun_out/r2c10_v3_big.json 2>>gpurun_out/r2c10.err
# odd row counts exercise the partial-tile path hard
timeout 300 python - > gpurun_out/r2c10_partial.json 2>gpurun_out/r2c10_partial.err <<'PYEOF'
import json, os, torch
from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset
out = {}
for rows in (97, 4093, 100003):
    X, y, _ = generate_logistic_dataset(rows, 1024, seed=5)
    m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
    th = torch.randn(1024, 16, device="cuda:0", generator=torch.Generator(device="cuda:0").manual_seed(1)) * 0.3
    os.environ.pop("FED_BATCHED_V3", None)
    l2, g2 = m.logp_grad_batched(th)
    os.environ["FED_BATCHED_V3"] = "1"
    l3, g3 = m.logp_grad_batched(th)
    out[rows] = {"logp_rel": float(((l2-l3).abs()/l2.abs()).max()),
                 "grad_absmax": float((g2-g3).abs().max())}
print(json.dumps(out))
PYEOF
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 400 bash -c "FED_BATCHED_V3=1 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT SQ_VALU_MFMA_BUSY -d gpurun_out/r2c10_pmc --output-format csv -- python benchmarks/bench_batched_chains.py --rows 2000000 --steps 10 --warmup 2" > gpurun_out/r2c10_pmc.log 2>&1 || true
echo "=== results ==="
tail -2 gpurun_out/r2c10_pytest.log
cat gpurun_out/r2c10_v3_2e6.json gpurun_out/r2c10_v3_big.json gpurun_out/r2c10_partial.json
