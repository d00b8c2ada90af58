#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT"
timeout 200 python -m pytest tests/test_native_worker.py -m gpu -q 2>&1 | tail -3
timeout 180 python __graft_entry__.py 2>&1 | tail -1
timeout 150 python bench.py --steps 100000 --warmup 1000 2>&1 | grep -v "^#" | tail -1
timeout 150 python bench.py --model ode --steps 1000 --warmup 100 2>&1 | grep -v "^#" | tail -1
