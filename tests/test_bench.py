"""bench.py driver-contract tests (CPU): one JSON line, required fields."""
import json
import subprocess
import sys
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent

REQUIRED = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


@pytest.mark.timeout(300)
@pytest.mark.parametrize("model", ["linear", "logistic", "ode"])
def test_bench_json_contract(model):
    proc = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--model", model,
         "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=280, cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-500:]
    line = [l for l in proc.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert REQUIRED.issubset(d.keys()), REQUIRED - set(d.keys())
    assert d["metric"] == "logp+grad calls/sec (whole node)"
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["n_gpus"] == 1
    assert "model" in d["config"] and "parallelism" in d["config"]


@pytest.mark.timeout(300)
def test_bench_multirank_gloo():
    proc = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29671", str(REPO / "bench.py"),
         "--gpus", "2", "--steps", "3", "--warmup", "1"],
        capture_output=True, text=True, timeout=280, cwd=REPO,
    )
    assert proc.returncode == 0, proc.stderr[-500:]
    line = [l for l in proc.stdout.strip().splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_gpus"] == 2
    assert "x2" in d["config"]["parallelism"]
