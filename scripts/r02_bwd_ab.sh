#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  for v in 0 1 0 1; do
    SRTB_FFT_BWD32=$v timeout 500 python bench.py --steps 10 --warmup 2 2>&1 | tail -1 | python3 -c "import json,sys; d=json.load(sys.stdin); print('bwd32=$v', d['value'], 'Msps', round(d['ms_per_step']/8,2), 'ms/blk')"
  done
  echo "=== ALL DONE ==="
} > gpurun_out/r02_bwd_ab.log 2>&1
tail -8 gpurun_out/r02_bwd_ab.log
