#!/bin/bash
# Reproduce every round-1 measured number on an MI355X box:
#   gpurun --timeout 1800 -- 'bash scripts/repro_r1.sh'
set -x
cd "${GRAFT_REPO_ROOT:-/root/repo}"
python -m pytensor_federated_amd.ops.build 2>&1 | tail -1
python __graft_entry__.py 2>&1 | tail -1                    # smoke
python -m pytest tests -m gpu -q 2>&1 | grep -E "passed|failed" | tail -1
# flagship + model configs
python bench.py --steps 100000 --warmup 1000 | tail -1                         # linear 52k
python bench.py --model logistic --rows 12500000 --steps 80 --warmup 10 | tail -1
python bench.py --model ode --steps 1000 --warmup 100 | tail -1
# batched chains + full-stack samplers
python benchmarks/bench_batched_chains.py --rows 2000000 | tail -1
python benchmarks/bench_nuts.py --draws 300 --tune 200 | tail -1
python benchmarks/bench_nuts_batched.py --chains 16 --draws 300 --mass dense | tail -1
python benchmarks/bench_device_transport.py | tail -1
