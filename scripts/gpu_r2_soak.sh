#!/bin/bash
# Round-2 stability soak: long repeated runs on the final build.
set -x
cd /root/repo
mkdir -p gpurun_out
# 1. full suite x2 (catches order effects / leaks)
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -2 | tee gpurun_out/soak_pytest1.log
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -2 | tee gpurun_out/soak_pytest2.log
# 2. 1000-step batched soak (v3 stability over many tiles/prefetch cycles)
timeout 600 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 500 --warmup 20 > gpurun_out/soak_batched.json 2>gpurun_out/soak.err
# 3. flagship soak (persistent kernel, 300k evals)
timeout 600 python bench.py --steps 300000 --warmup 1000 > gpurun_out/soak_bench.json 2>>gpurun_out/soak.err
# 4. worker under 8 concurrent client processes, 2000 calls each
timeout 600 python benchmarks/bench_worker_grpc.py --calls 2000 --clients 8 > gpurun_out/soak_worker.json 2>>gpurun_out/soak.err
echo "=== results ==="
tail -2 gpurun_out/soak_pytest1.log gpurun_out/soak_pytest2.log
cat gpurun_out/soak_batched.json gpurun_out/soak_bench.json gpurun_out/soak_worker.json
