#!/bin/bash
# Round-2 GPU call 11: FULL suite with v3 default + final benchmark sweep.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/r2c11_pytest.log | tail -3
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/r2c11_bench.json 2>gpurun_out/r2c11_bench.err
timeout 300 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 60 > gpurun_out/r2c11_batched_2e6.json 2>gpurun_out/r2c11.err
timeout 300 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 40 > gpurun_out/r2c11_batched_big.json 2>>gpurun_out/r2c11.err
timeout 600 python benchmarks/bench_nuts_batched.py --chains 16 --draws 300 --tune 400 --mass dense > gpurun_out/r2c11_nuts.json 2>gpurun_out/r2c11_nuts.err
# MALA over the (now v3) logistic batched kernel e2e
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c11_prof -- \
    python benchmarks/bench_batched_chains.py --rows 2000000 --steps 30 > gpurun_out/r2c11_prof.log 2>&1 || true
find gpurun_out/r2c11_prof -name "*kernel_trace*" -delete 2>/dev/null || true
echo "=== results ==="
tail -3 gpurun_out/r2c11_pytest.log
cat gpurun_out/r2c11_bench.json gpurun_out/r2c11_batched_2e6.json gpurun_out/r2c11_batched_big.json gpurun_out/r2c11_nuts.json
