#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
for rep in 1 2 3; do
  for fm in 1 0; do
    timeout 300 bash -c "FED_PK_FENCE=$fm python bench.py --steps 100000 --warmup 1000" \
      > gpurun_out/r2c19_fm${fm}_r${rep}.json 2>>gpurun_out/r2c19.err
  done
done
# correctness under the drain publish: repeated-call stability + numerics
timeout 500 bash -c "FED_PK_FENCE=0 python -m pytest tests/test_gpu.py -q -k 'persistent or fused_combine'" 2>&1 | tail -2 | tee gpurun_out/r2c19_pytest.log
echo "=== results ==="
for rep in 1 2 3; do for fm in 1 0; do
  echo -n "fence=$fm rep=$rep: "; python -c "import json;print(json.load(open('gpurun_out/r2c19_fm${fm}_r${rep}.json'))['value'])" 2>/dev/null || echo ERR
done; done
tail -2 gpurun_out/r2c19_pytest.log
