#!/bin/bash
set -x
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== FULL gpu suite (shipped build) ==="
  timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -2
  echo "=== smoke ==="
  timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tail -1
  echo "=== flagship bench ==="
  timeout 600 python bench.py --steps 15 --warmup 3 2>&1 | tail -1
  echo "=== dual-pol bench ==="
  timeout 420 python benchmarks/dual_pol_bench.py --steps 4 2>&1 | tail -1
  echo "=== crab DM sweep ==="
  timeout 420 python benchmarks/crab_dm_sweep.py 2>&1 | tail -1
  echo "=== kernel stats (shipped defaults) ==="
  cd /tmp
  timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /root/repo/gpurun_out/prof/ship -- \
    python /root/repo/bench.py --steps 3 --warmup 1 --blocks-per-step 2 2>&1 | tail -1
  echo "=== ALL DONE ==="
} > /root/repo/gpurun_out/r02_final2.log 2>&1
tail -20 /root/repo/gpurun_out/r02_final2.log
