#!/bin/bash
# Round-1 first GPU validation: build, smoke, gpu tests, bench, rocprof stats.
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
python -c "import torch; print(torch.__version__, torch.cuda.is_available(), torch.cuda.get_device_name(0))" 2>&1 | tee gpurun_out/env.log
python -m pytensor_federated_amd.ops.build 2>&1 | tail -2
timeout 300 python __graft_entry__.py 2>&1 | tail -3 | tee gpurun_out/smoke.log
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -15 | tee gpurun_out/pytest_gpu.log
timeout 300 python bench.py --steps 300 --warmup 30 > gpurun_out/bench_n1.json 2> gpurun_out/bench_n1.err
tail -1 gpurun_out/bench_n1.json
timeout 300 python bench.py --model logistic --rows 2000000 --steps 30 --warmup 5 > gpurun_out/bench_logistic.json 2> gpurun_out/bench_logistic.err
tail -1 gpurun_out/bench_logistic.json
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 50 --warmup 5 > "$GRAFT_REPO_ROOT/gpurun_out/rocprof_bench.log" 2>&1
tail -25 "$GRAFT_REPO_ROOT/gpurun_out/rocprof_bench.log"
