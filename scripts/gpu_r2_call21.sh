#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
# numerics under nt
timeout 400 python -m pytest tests/test_gpu.py -q -k batched 2>&1 | tail -2 | tee gpurun_out/r2c21_pytest.log
# same-box A/B nt vs cached, 3 reps, both sizes
for rep in 1 2 3; do
  for ntv in 1 0; do
    timeout 200 bash -c "FED_V3_NT=$ntv python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50" \
      > gpurun_out/r2c21_nt${ntv}_r${rep}.json 2>>gpurun_out/r2c21.err
  done
done
timeout 300 bash -c "FED_V3_NT=1 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 40" > gpurun_out/r2c21_big_nt.json 2>>gpurun_out/r2c21.err
echo "=== results ==="
tail -2 gpurun_out/r2c21_pytest.log
for rep in 1 2 3; do for ntv in 1 0; do
  echo -n "nt=$ntv rep=$rep: "; python -c "import json;print(json.load(open('gpurun_out/r2c21_nt${ntv}_r${rep}.json'))['ms_per_batched_step'])" 2>/dev/null || echo ERR
done; done
cat gpurun_out/r2c21_big_nt.json
