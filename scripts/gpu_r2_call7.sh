#!/bin/bash
# Round-2 GPU call 7: grid-size A/B for phase-B L2 residency.
set -x
cd /root/repo
mkdir -p gpurun_out
for g in 256 320 384 448 512; do
  timeout 200 bash -c "FED_BATCHED_GRID=$g python benchmarks/bench_batched_chains.py --rows 2000000 --steps 40" \
      > gpurun_out/r2c7_grid$g.json 2>gpurun_out/r2c7_grid$g.err
done
timeout 200 bash -c "FED_BATCHED_GRID=256 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 30" \
    > gpurun_out/r2c7_grid256_big.json 2>gpurun_out/r2c7_big.err
timeout 600 bash -c "FED_BATCHED_GRID=256 python -m pytest tests/test_gpu.py -q -k batched" 2>&1 | tail -2 | tee gpurun_out/r2c7_pytest.log
echo "=== results ==="
for g in 256 320 384 448 512; do echo "grid $g:"; cat gpurun_out/r2c7_grid$g.json; done
cat gpurun_out/r2c7_grid256_big.json
