#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== full gpu suite ==="
  timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -3
  echo "=== smoke ==="
  timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tail -2
  echo "=== honest flagship bench (steps=10 x 4 blocks) ==="
  timeout 600 python bench.py --steps 10 --warmup 2 --blocks-per-step 4 2>&1 | tail -1
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check5.log 2>&1
tail -25 gpurun_out/r02_check5.log
