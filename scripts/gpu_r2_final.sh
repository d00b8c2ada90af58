#!/bin/bash
# Final round-2 validation: full suite + flagship on the shipped defaults.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/final_pytest.log | tail -3
python -c "import __graft_entry__ as e; e.smoke()" 2>&1 | tail -1 | tee gpurun_out/final_smoke.log
timeout 600 python bench.py --steps 200000 --warmup 1000 > gpurun_out/final_bench.json 2>gpurun_out/final.err
timeout 300 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 > gpurun_out/final_batched.json 2>>gpurun_out/final.err
echo "=== results ==="
grep -E "passed|failed" gpurun_out/final_pytest.log | tail -1
cat gpurun_out/final_smoke.log gpurun_out/final_bench.json gpurun_out/final_batched.json
