#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
python -m pytensor_federated_amd.ops.build 2>&1 | tail -1
timeout 240 python __graft_entry__.py 2>&1 | tail -1
timeout 500 python -m pytest tests -m gpu -q 2>&1 | grep -E "passed|failed" | tail -1
for args in "--steps 5000 --warmup 500" "--model logistic --rows 12500000 --steps 80 --warmup 10" "--model ode --steps 1000 --warmup 100"; do
  timeout 250 python bench.py $args 2>&1 | grep -v "^#" | tail -1 | python -c "import json,sys; d=json.loads(sys.stdin.read()); print(d['config']['model'], 'calls/s:', round(d['value'],1), 'ms/step:', round(d['ms_per_step'],4), d['config'].get('path',''))"
done
echo "=== 3x repeated soak (stability across engine restarts) ==="
for i in 1 2 3; do
  timeout 200 python bench.py --steps 200000 --warmup 1000 2>&1 | grep -v "^#" | tail -1 | python -c "import json,sys; d=json.loads(sys.stdin.read()); print('  soak', round(d['value']), 'calls/s')"
done
timeout 200 python benchmarks/bench_batched_chains.py --rows 2000000 2>&1 | tail -1 | python -c "import json,sys; d=json.loads(sys.stdin.read()); print('batched chains/s:', round(d['value']))"
