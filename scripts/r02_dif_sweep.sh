#!/bin/bash
set -x
export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== DIF F sweep ==="
  for f in 4 8 16 32; do
    SRTB_FFT_DIF_F=$f timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
C = native(); import torch; torch.cuda.set_device(0)
import numpy as np
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(4,1<<18))+1j*rng.normal(size=(4,1<<18))).astype(np.complex64)).cuda()
err=((C.native_fft(x,1)-torch.fft.ifft(x,dim=1)*(1<<18)).abs().max()).item()
print("F=%s err=%.2e fwd29=%.3f ms bwd18=%.3f ms" % (
  os.environ["SRTB_FFT_DIF_F"], err,
  C.bench_fft(1<<29,1,-1,20,"native"), C.bench_fft(1<<18,2048,1,20,"native")))
PY
  done
  echo "=== bench kernel stats (csv) ==="
  cd /tmp
  timeout 600 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/prof/bench2 -- \
    python /root/repo/bench.py --steps 3 --warmup 1 --blocks-per-step 2 2>&1 | tail -4
  ls /root/repo/gpurun_out/prof/bench2/* 2>/dev/null | head
  echo "=== ALL DONE ==="
} > gpurun_out/r02_dif_sweep.log 2>&1
tail -40 gpurun_out/r02_dif_sweep.log
