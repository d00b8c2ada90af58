#!/bin/bash
# Round-2 GPU call 5: validate the reverse-walk fix + windowed default,
# re-time the batched kernel, native-worker gRPC latency on GPU.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/r2c5_pytest_full.log | tail -3

timeout 300 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 \
    > gpurun_out/r2c5_batched_2e6.json 2>gpurun_out/r2c5_batched.err
timeout 300 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 40 \
    > gpurun_out/r2c5_batched_125e5.json 2>>gpurun_out/r2c5_batched.err

timeout 300 python benchmarks/bench_worker_grpc.py --calls 2000 \
    > gpurun_out/r2c5_worker_grpc.json 2>gpurun_out/r2c5_worker_grpc.err

# final stats CSV of the FIXED kernel for profiles/
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c5_prof -- \
    python benchmarks/bench_batched_chains.py --rows 2000000 --steps 30 \
    > gpurun_out/r2c5_prof_run.log 2>&1 || true
find gpurun_out/r2c5_prof -name "*kernel_trace*" -delete 2>/dev/null || true

echo "=== results ==="
tail -3 gpurun_out/r2c5_pytest_full.log
cat gpurun_out/r2c5_batched_2e6.json gpurun_out/r2c5_batched_125e5.json gpurun_out/r2c5_worker_grpc.json
