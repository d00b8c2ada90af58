#!/bin/bash
# Round-2 GPU call 2: batched v2 (theta back in LDS) A/B + native-worker gRPC.
set -x
cd /root/repo
mkdir -p gpurun_out

# 1. the new gRPC worker test + batched numerics tests
timeout 600 python -m pytest tests/test_native_worker.py tests/test_worker_grpc_cpu.py -x -q 2>&1 | tail -4 | tee gpurun_out/r2c2_pytest.log
timeout 600 python -m pytest tests/test_gpu.py -x -q -k "batched" 2>&1 | tail -3 | tee -a gpurun_out/r2c2_pytest.log

# 2. A/B v1 vs v2, 2e6 rows
timeout 300 bash -c 'FED_BATCHED_V1=1 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50' \
    > gpurun_out/r2c2_batched_v1.json 2>gpurun_out/r2c2_batched_v1.err
timeout 300 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 \
    > gpurun_out/r2c2_batched_v2.json 2>gpurun_out/r2c2_batched_v2.err

# 3. PMC counters for v2 (counters-only run per pool rules)
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_LDS_BANK_CONFLICT SQ_VALU_MFMA_BUSY \
    -d gpurun_out/r2c2_pmc --output-format csv -- \
    python benchmarks/bench_batched_chains.py --rows 2000000 --steps 10 --warmup 2 \
    > gpurun_out/r2c2_pmc_run.log 2>&1 || true

echo "=== results ==="
cat gpurun_out/r2c2_batched_v1.json gpurun_out/r2c2_batched_v2.json
tail -6 gpurun_out/r2c2_pytest.log
