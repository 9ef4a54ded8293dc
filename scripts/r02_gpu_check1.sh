#!/bin/bash
# round-2 first GPU validation: tests + honest bench + world=2 path
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== pytest -m gpu ==="
  timeout 1200 python -m pytest tests -m gpu -q 2>&1 | tail -15
  echo "=== bench honest (small steps, full J1644 shape) ==="
  timeout 420 python bench.py --steps 3 --warmup 1 --blocks-per-step 2 2>&1 | tail -3
  echo "=== torchrun world=2 on one GPU (gloo collectives, shared GPU) ==="
  timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29411 bench.py --gpus 2 --steps 2 \
    --warmup 1 --backend gloo --n $((1<<26)) --channels 256 --slots 2 2>&1 | tail -4
  echo "=== torchrun world=2 nccl duplicate-device probe (expected to fail fast) ==="
  timeout 180 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29412 bench.py --gpus 2 --steps 1 \
    --warmup 0 --backend nccl --n $((1<<24)) --channels 256 --slots 2 2>&1 | tail -6
  echo "=== done rc=$? ==="
} > gpurun_out/r02_check1.log 2>&1
tail -40 gpurun_out/r02_check1.log
