#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== engine suites (bwd plan override) ==="
  timeout 900 python -m pytest tests/test_gpu_engine.py tests/test_gpu_fft.py -q 2>&1 | tail -2
  echo "=== flagship bench ==="
  timeout 600 python bench.py --steps 15 --warmup 3 2>&1 | tail -1
  echo "=== kernel stats (confirm preop back on pair32) ==="
  export TMPDIR=/tmp; cd /tmp
  timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /root/repo/gpurun_out/prof/final2 -- \
    python /root/repo/bench.py --steps 3 --warmup 1 --blocks-per-step 2 2>&1 | tail -1
  echo "=== ALL DONE ==="
} > /root/repo/gpurun_out/r02_check10.log 2>&1
tail -12 /root/repo/gpurun_out/r02_check10.log
