#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
timeout 900 python -m pytest tests/test_ops_gpu.py tests/test_gpu_e2e.py -m gpu -q 2>&1 | tail -2
timeout 900 python bench.py --gpus 1 --steps 3 --warmup 1 2>&1 | tail -1
