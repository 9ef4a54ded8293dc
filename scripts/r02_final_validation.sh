#!/bin/bash
set -x
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== FULL gpu suite ==="
  timeout 1500 python -m pytest tests -m gpu -q 2>&1 | tail -2
  echo "=== smoke ==="
  timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tail -1
  echo "=== flagship bench (driver-shaped: steps=15 warmup=3) ==="
  timeout 600 python bench.py --steps 15 --warmup 3 2>&1 | tail -1
  echo "=== 150 s engine soak (stability of the new FFT plans) ==="
  timeout 300 python scripts/soak.py --seconds 150 2>&1 | tail -5
  echo "=== torchrun world=2 sanity (gloo, shared GPU) ==="
  timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29421 bench.py --gpus 2 --steps 2 \
    --warmup 1 --backend gloo --input-count $((1<<26)) --channels 256 \
    --blocks-per-step 1 2>&1 | tail -1
  echo "=== rocprof kernel stats of the final config ==="
  cd /tmp
  timeout 600 rocprofv3 --kernel-trace --stats --output-format csv \
    -d /root/repo/gpurun_out/prof/final -- \
    python /root/repo/bench.py --steps 3 --warmup 1 --blocks-per-step 2 2>&1 | tail -2
  find /root/repo/gpurun_out/prof/final -name "*kernel_stats.csv" | head -2
  echo "=== ALL DONE ==="
} > /root/repo/gpurun_out/r02_final.log 2>&1
tail -30 /root/repo/gpurun_out/r02_final.log
