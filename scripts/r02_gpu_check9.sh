#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== fused ladder: engine + feature suites ==="
  timeout 900 python -m pytest tests/test_gpu_engine.py tests/test_gpu_features.py tests/test_gpu_main_app.py -q 2>&1 | tail -2
  echo "=== flagship bench with ladder ==="
  timeout 600 python bench.py --steps 15 --warmup 3 2>&1 | tail -1
  echo "=== 240 s soak (one-time pool step settles by t=90) ==="
  timeout 420 python scripts/soak.py --seconds 240 --slots 2 2>&1 | tail -14
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check9.log 2>&1
tail -30 gpurun_out/r02_check9.log
