#!/bin/bash
# Same-box A/B of the persistent-server poll protocols (box variance is
# +-8%, so cross-box comparisons of these ~1us effects are meaningless).
set -x
cd /root/repo
mkdir -p gpurun_out
for rep in 1 2 3; do
  for sl in 1 0; do
    timeout 300 bash -c "FED_PK_SEQLOCK=$sl python bench.py --steps 100000 --warmup 1000" \
      > gpurun_out/r2c17_sl${sl}_r${rep}.json 2>>gpurun_out/r2c17.err
  done
done
timeout 400 python -m pytest tests/test_gpu.py -q -k "persistent" 2>&1 | tail -2 | tee gpurun_out/r2c17_pytest.log
echo "=== results ==="
for rep in 1 2 3; do for sl in 1 0; do
  echo -n "sl=$sl rep=$rep: "; python -c "import json;print(json.load(open('gpurun_out/r2c17_sl${sl}_r${rep}.json'))['value'])" 2>/dev/null || echo ERR
done; done
tail -2 gpurun_out/r2c17_pytest.log
