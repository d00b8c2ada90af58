#!/bin/bash
# Reproduce every round-2 measured number on an MI355X box:
#   gpurun --timeout 1800 -- 'bash scripts/repro_r2.sh'
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
python -m pytensor_federated_amd.ops.build 2>&1 | tail -1
python -m pytest tests -m gpu -q 2>&1 | grep -E "passed|failed" | tail -1
# batched logistic (v4 tr_b16 default at K=1024: ~0.89 ms @2e6x1024; ~5.3 ms @config-4)
python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 | tail -1
python benchmarks/bench_batched_chains.py --rows 12500000 --steps 40 | tail -1
python benchmarks/bench_batched_chains.py --rows 2000000 --features 512 --steps 50 | tail -1
# ladder A/Bs (v1 / v2 / v3-scalar-image / v4-default)
FED_BATCHED_V1=1 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 | tail -1
FED_BATCHED_V3=0 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 | tail -1
FED_BATCHED_V4=0 python benchmarks/bench_batched_chains.py --rows 2000000 --steps 50 | tail -1
# windowed (default) vs simple NUTS adaptation
python benchmarks/bench_nuts_batched.py --chains 16 --draws 300 --tune 400 --mass dense | tail -1
python benchmarks/bench_nuts_batched.py --chains 16 --draws 300 --tune 400 --mass dense --adaptation simple | tail -1
# native-worker protocol edges (gRPC stream/unary vs FEDS1) + 4-client row
python benchmarks/bench_worker_grpc.py --calls 2000 --clients 4 | tail -1
# flagship (unchanged from round 1: ~50k calls/s persistent path)
python bench.py --steps 100000 --warmup 1000 | tail -1
