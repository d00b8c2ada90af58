#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out
python -m pytensor_federated_amd.ops.build 2>&1 | tail -1
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -4 | tee gpurun_out/pytest_gpu4.log
timeout 300 python bench.py --steps 2000 --warmup 200 > gpurun_out/bench4_n1.json 2> gpurun_out/bench4_n1.err
tail -1 gpurun_out/bench4_n1.json
timeout 300 python bench.py --model logistic --rows 12500000 --steps 60 --warmup 10 > gpurun_out/bench4_logistic.json 2> gpurun_out/bench4_logistic.err
tail -1 gpurun_out/bench4_logistic.json
export TMPDIR=/tmp
cd /tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/prof4" -- python "$GRAFT_REPO_ROOT/bench.py" --steps 300 --warmup 30 > "$GRAFT_REPO_ROOT/gpurun_out/rocprof4.log" 2>&1
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv -d "$GRAFT_REPO_ROOT/gpurun_out/prof4_logistic" -- python "$GRAFT_REPO_ROOT/bench.py" --model logistic --rows 12500000 --steps 20 --warmup 3 >> "$GRAFT_REPO_ROOT/gpurun_out/rocprof4.log" 2>&1
for f in $(find "$GRAFT_REPO_ROOT/gpurun_out/prof4" "$GRAFT_REPO_ROOT/gpurun_out/prof4_logistic" -name "*kernel_stats*"); do echo "== $f"; head -5 "$f"; done
