#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  for s in 2 3 4 6; do
    timeout 400 python bench.py --steps 6 --warmup 2 --blocks-per-step 4 --slots $s 2>&1 | tail -1 | python3 -c "import json,sys,os; d=json.load(sys.stdin); print('slots=$s', d['value'], 'Msps', d['ms_per_step']/4, 'ms/blk')"
  done
  echo "=== ALL DONE ==="
} > gpurun_out/r02_slots.log 2>&1
tail -10 gpurun_out/r02_slots.log
