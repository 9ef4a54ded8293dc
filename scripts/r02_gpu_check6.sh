#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== wave kernel: correctness + timing vs stockham/rocfft ==="
  for ln in 256 512 1024; do
    SWEEP_LEN=$ln SRTB_FFT_WAVE=1 timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
LEN=int(os.environ["SWEEP_LEN"]); BATCH=(1<<27)//LEN
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(16,LEN))+1j*rng.normal(size=(16,LEN))).astype(np.complex64)).cuda()
for sign in (-1, 1):
    ref = torch.fft.fft(x,dim=1) if sign==-1 else torch.fft.ifft(x,dim=1)*LEN
    err=((C.native_fft(x,sign)-ref).abs().max()/ref.abs().max()).item()
    print(f"len={LEN} sign={sign} err={err:.2e}")
print(f"len={LEN} wave={C.bench_fft(LEN,BATCH,1,30,'native'):.3f} ms")
PY
    SWEEP_LEN=$ln timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
C = native(); import torch; torch.cuda.set_device(0)
LEN=int(os.environ["SWEEP_LEN"]); BATCH=(1<<27)//LEN
print(f"len={LEN} stockham={C.bench_fft(LEN,BATCH,1,30,'native'):.3f} ms rocfft={C.bench_fft(LEN,BATCH,1,30,'hipfft'):.3f} ms")
PY
  done
  echo "=== 2pol python-runner GPU test with full output ==="
  timeout 600 python -m pytest tests/test_gpu_main_app.py::test_main_gpu_2pol_fanout -rA -q 2>&1 | tail -30
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check6.log 2>&1
tail -55 gpurun_out/r02_check6.log
