#!/bin/bash
# Final fresh-box validation: smoke + default bench (the driver's round-end flow).
set -x
cd "$GRAFT_REPO_ROOT" || cd /root/repo
mkdir -p gpurun_out
{
  timeout 200 python -c "import __graft_entry__ as g; g.smoke(); print('SMOKE OK')"
  timeout 260 python bench.py --gpus 1 --steps 30 --warmup 5
} > gpurun_out/final_freshbox.log 2>&1
tail -25 gpurun_out/final_freshbox.log
