#!/bin/bash
# Connection-churn + sustained-concurrency stress on the native worker.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python - > gpurun_out/churn.json 2>gpurun_out/churn.err <<'PYEOF'
import json, os, struct, subprocess, socket, tempfile, time
import numpy as np
from pathlib import Path
from pytensor_federated_amd.models import GaussianLinearModel, generate_linear_dataset
from pytensor_federated_amd.service import ArraysToArraysServiceClient
import torch

REPO = Path.cwd()
WORKER = REPO / "pytensor_federated_amd" / "ops" / "fed_worker"
LIB = REPO / "pytensor_federated_amd" / "ops" / "libfedops_gfx950.so"
x, y = generate_linear_dataset(500_000, seed=77)
tmp = tempfile.NamedTemporaryFile(suffix=".bin", delete=False)
tmp.write(struct.pack("<q", len(x)))
tmp.write(np.asarray(x, dtype=np.float64).tobytes())
tmp.write(np.asarray(y, dtype=np.float64).tobytes())
tmp.close()
env = dict(os.environ, FEDOPS_LIB=str(LIB))
proc = subprocess.Popen([str(WORKER), "--port", "9701", "--grpc-port", "9702",
                         "--data", tmp.name, "--sigma", "0.4", "--dtype", "bf16"],
                        env=env, stderr=subprocess.DEVNULL)
def wait(p):
    for _ in range(300):
        try:
            socket.create_connection(("127.0.0.1", p), timeout=1).close(); return
        except OSError: time.sleep(0.1)
    raise TimeoutError
wait(9702)
ref = GaussianLinearModel(x, y, sigma=0.4, device="cuda:0", dtype=torch.bfloat16)
refs = {}
try:
    t0 = time.time()
    churns = 0
    evals = 0
    # 400 connect-evaluate-disconnect cycles alternating transports,
    # randomized thetas checked against the reference model
    rng = np.random.RandomState(3)
    while churns < 400 and time.time() - t0 < 600:
        tr, port = (("fast", 9701) if churns % 2 == 0 else ("grpc", 9702))
        c = ArraysToArraysServiceClient("127.0.0.1", port, transport=tr)
        a = float(1.5 + 0.2 * rng.randn()); b = float(0.5 + 0.1 * rng.randn())
        key = (round(a, 12), round(b, 12))
        if key not in refs:
            refs[key] = ref(a, b)
        for _ in range(5):
            logp, ga, gb = c.evaluate(a, b)
            lr, (gar, gbr) = refs[key]
            np.testing.assert_allclose(float(logp), float(lr), rtol=1e-9)
            np.testing.assert_allclose(float(ga), float(gar), rtol=1e-7)
            evals += 1
        del c
        churns += 1
    print(json.dumps({"churn_cycles": churns, "evals": evals,
                      "wall_s": time.time() - t0, "ok": True}))
finally:
    proc.terminate(); proc.wait(timeout=10); os.unlink(tmp.name)
PYEOF
echo "=== result ==="
cat gpurun_out/churn.json
tail -3 gpurun_out/churn.err
