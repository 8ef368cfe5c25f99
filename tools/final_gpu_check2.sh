#!/bin/bash
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
{
  timeout 120 python -m pytest tests/test_ops_gpu.py tests/test_model_gpu.py -q 2>&1 | tail -2
  timeout 90 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')" 2>&1 | tail -1
  timeout 150 python bench.py --gpus 1 --steps 10 --warmup 3 2>&1 | tail -1
} > gpurun_out/final_check2.log 2>&1
tail -8 gpurun_out/final_check2.log
