#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== fft tuning sweep ==="
  timeout 1500 python scripts/r02_fft_tune.py
  echo "=== fixed UDP native test ==="
  timeout 300 python -m pytest tests/test_native_app.py::test_srtb_backend_udp_ingest_with_overlap -x -q 2>&1 | tail -3
  echo "=== torchrun world=2 on one GPU (gloo) ==="
  timeout 420 python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
    --master-addr 127.0.0.1 --master-port 29413 bench.py --gpus 2 --steps 2 \
    --warmup 1 --backend gloo --input-count $((1<<26)) --channels 256 --slots 2 \
    --blocks-per-step 1 2>&1 | tail -4
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check2.log 2>&1
tail -60 gpurun_out/r02_check2.log
