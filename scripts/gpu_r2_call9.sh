#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 300 bash -c "FED_BATCHED_V3=1 python -m pytest tests/test_gpu.py -q -k batched" 2>&1 | tail -2 | tee gpurun_out/r2c9_pytest.log
timeout 300 bash -c "FED_BATCHED_V3=1 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 60" > gpurun_out/r2c9_v3_2e6.json 2>gpurun_out/r2c9.err
timeout 300 bash -c "FED_BATCHED_V3=1 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 40" > gpurun_out/r2c9_v3_big.json 2>>gpurun_out/r2c9.err
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 400 bash -c "FED_BATCHED_V3=1 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT SQ_VALU_MFMA_BUSY -d gpurun_out/r2c9_pmc --output-format csv -- python benchmarks/bench_batched_chains.py --rows 2000000 --steps 10 --warmup 2" > gpurun_out/r2c9_pmc.log 2>&1 || true
echo "=== results ==="
tail -2 gpurun_out/r2c9_pytest.log
cat gpurun_out/r2c9_v3_2e6.json gpurun_out/r2c9_v3_big.json
