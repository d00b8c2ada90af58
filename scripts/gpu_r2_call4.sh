#!/bin/bash
# Round-2 GPU call 4: full suite (complete log), flagship bench sanity,
# reverse-walk A/B, NUTS windowed-vs-simple, final v2 rocprof stats.
set -x
cd /root/repo
mkdir -p gpurun_out

# 1. full GPU suite -- complete output this time
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tee gpurun_out/r2c4_pytest_full.log | tail -5

# 2. flagship bench sanity (should be ~50k calls/s persistent path)
timeout 300 python bench.py --gpus 1 --steps 20 --warmup 5 > gpurun_out/r2c4_bench.json 2>gpurun_out/r2c4_bench.err

# 3. batched v2 with reverse phase-B walk (vs 1.109 forward)
timeout 300 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 \
    > gpurun_out/r2c4_v2rev_2e6.json 2>gpurun_out/r2c4_v2rev.err
timeout 300 python benchmarks/bench_batched_chains.py --rows 12500000 --steps 40 \
    > gpurun_out/r2c4_v2rev_125e5.json 2>>gpurun_out/r2c4_v2rev.err

# 4. NUTS windowed vs simple adaptation (promote-to-default decision)
timeout 600 python benchmarks/bench_nuts_batched.py --chains 16 --draws 300 --tune 400 --mass dense \
    > gpurun_out/r2c4_nuts_simple.json 2>gpurun_out/r2c4_nuts_simple.err
timeout 600 python benchmarks/bench_nuts_batched.py --chains 16 --draws 300 --tune 400 --mass dense --adaptation windowed \
    > gpurun_out/r2c4_nuts_windowed.json 2>gpurun_out/r2c4_nuts_windowed.err

# 5. final kernel stats CSV for profiles/
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/r2c4_prof -- \
    python benchmarks/bench_batched_chains.py --rows 2000000 --steps 30 \
    > gpurun_out/r2c4_prof_run.log 2>&1 || true
find gpurun_out/r2c4_prof -name "*kernel_trace*" -delete 2>/dev/null || true

echo "=== results ==="
tail -3 gpurun_out/r2c4_pytest_full.log
cat gpurun_out/r2c4_bench.json
cat gpurun_out/r2c4_v2rev_2e6.json gpurun_out/r2c4_v2rev_125e5.json
echo "--- nuts ---"
cat gpurun_out/r2c4_nuts_simple.json gpurun_out/r2c4_nuts_windowed.json
