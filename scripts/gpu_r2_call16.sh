#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
# persistent-path regression + stability (relaunch-after-idle incl.)
timeout 500 python -m pytest tests/test_gpu.py -q -k "persistent or fused_combine or gaussian" 2>&1 | tail -2 | tee gpurun_out/r2c16_pytest.log
# flagship before/after (fresh box; compare with 52.0k repro figure)
timeout 600 python bench.py --steps 200000 --warmup 1000 > gpurun_out/r2c16_bench.json 2>gpurun_out/r2c16.err
echo "=== results ==="
tail -2 gpurun_out/r2c16_pytest.log
cat gpurun_out/r2c16_bench.json
