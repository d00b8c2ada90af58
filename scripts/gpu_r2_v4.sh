#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 400 bash -c "FED_BATCHED_V4=1 python -m pytest tests/test_gpu.py -q -k batched" 2>&1 | tail -2 | tee gpurun_out/v4_pytest.log
for rep in 1 2 3; do
  for v4 in 1 0; do
    timeout 200 bash -c "FED_BATCHED_V4=$v4 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50" \
      > gpurun_out/v4_${v4}_r${rep}.json 2>>gpurun_out/v4.err
  done
done
echo "=== results ==="
tail -2 gpurun_out/v4_pytest.log
for rep in 1 2 3; do for v4 in 1 0; do
  echo -n "v4=$v4 rep=$rep: "; python -c "import json;print(json.load(open('gpurun_out/v4_${v4}_r${rep}.json'))['ms_per_batched_step'])" 2>/dev/null || echo ERR
done; done
