#!/usr/bin/env python3
"""Read the FED_V3_PROF cycle split of the v3 batched kernel.

Splits the kernel's wall time into: stage-gate [1] (h0 DMA wait + barrier),
stage-gate [4] (h1+y wait + barrier), the zc/R barrier region, the phase-B
h1 barrier, and the remainder (compute + epilogue).
"""
import json
import os
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
os.environ["FED_V3_PROF"] = "1"

import torch

from pytensor_federated_amd.models import LogisticGLMModel, generate_logistic_dataset
from pytensor_federated_amd.ops import _ws_cache

rows = int(sys.argv[1]) if len(sys.argv) > 1 else 2_000_000
X, y, _ = generate_logistic_dataset(rows, 1024, seed=70)
m = LogisticGLMModel(X, y, device="cuda:0", dtype=torch.bfloat16)
theta = torch.randn(1024, 16, device="cuda:0",
                    generator=torch.Generator(device="cuda:0").manual_seed(3)) * 0.3
for _ in range(3):
    m.logp_grad_batched(theta)
torch.cuda.synchronize()

ws = _ws_cache[(0, "logistic_batched1024")]
slab_cols = 16 + 1024 * 16
prof = ws[300 * slab_cols : 300 * slab_cols + 2 * 8 * 256].view(torch.int64).reshape(256, 8)
tot = prof[:, 4].double()
mask = tot > 0
p = prof[mask].double()
res = {
    "blocks": int(mask.sum()),
    "gate1_h0_pct": float((p[:, 0] / p[:, 4]).mean() * 100),
    "gate4_h1y_pct": float((p[:, 1] / p[:, 4]).mean() * 100),
    "zcR_barriers_pct": float((p[:, 2] / p[:, 4]).mean() * 100),
    "phaseB_h1_barrier_pct": float((p[:, 3] / p[:, 4]).mean() * 100),
    "phaseA_pct": float((p[:, 5] / p[:, 4]).mean() * 100),
    "phaseB_pct": float((p[:, 6] / p[:, 4]).mean() * 100),
}
res["accounted_wait_pct"] = sum(
    res[k] for k in ("gate1_h0_pct", "gate4_h1y_pct", "zcR_barriers_pct",
                     "phaseB_h1_barrier_pct"))
print(json.dumps(res))
