#!/bin/bash
# Round-2 GPU call 3: v2(=v1+swizzle) A/B + ODE poly kernel tests + full suite.
set -x
cd /root/repo
mkdir -p gpurun_out

# 1. full GPU suite (incl. new poly kernels + worker grpc)
timeout 900 python -m pytest tests -m gpu -x -q 2>&1 | tail -4 | tee gpurun_out/r2c3_pytest.log

# 2. A/B v1 vs v2 at 2e6 and at the full config-4 shard (1.25e7)
for rows in 2000000 12500000; do
  timeout 300 bash -c "FED_BATCHED_V1=1 python benchmarks/bench_batched_chains.py --rows $rows --steps 40" \
      > gpurun_out/r2c3_v1_${rows}.json 2>gpurun_out/r2c3_v1_${rows}.err
  timeout 300 python benchmarks/bench_batched_chains.py --rows $rows --steps 40 \
      > gpurun_out/r2c3_v2_${rows}.json 2>gpurun_out/r2c3_v2_${rows}.err
done

# 3. PMC for v2 (should show conflict cycles collapsed)
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT SQ_VALU_MFMA_BUSY \
    -d gpurun_out/r2c3_pmc --output-format csv -- \
    python benchmarks/bench_batched_chains.py --rows 2000000 --steps 10 --warmup 2 \
    > gpurun_out/r2c3_pmc_run.log 2>&1 || true

echo "=== results ==="
cat gpurun_out/r2c3_v1_2000000.json gpurun_out/r2c3_v2_2000000.json \
    gpurun_out/r2c3_v1_12500000.json gpurun_out/r2c3_v2_12500000.json
tail -4 gpurun_out/r2c3_pytest.log
