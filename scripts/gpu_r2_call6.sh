#!/bin/bash
# Round-2 GPU call 6: validate per-connection EvalCtx worker on GPU.
set -x
cd /root/repo
mkdir -p gpurun_out
timeout 600 python -m pytest tests/test_native_worker.py tests/test_worker_grpc_cpu.py tests/test_device_codec.py -q 2>&1 | tee gpurun_out/r2c6_pytest.log | tail -3
timeout 600 python benchmarks/bench_worker_grpc.py --calls 2000 --clients 4 \
    > gpurun_out/r2c6_worker_grpc.json 2>gpurun_out/r2c6_worker_grpc.err
echo "=== results ==="
tail -3 gpurun_out/r2c6_pytest.log
cat gpurun_out/r2c6_worker_grpc.json
