#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== new default plans: fft+engine suites ==="
  timeout 900 python -m pytest tests/test_gpu_fft.py tests/test_gpu_engine.py -q 2>&1 | tail -2
  echo "=== E=32 wave (2048) opt-in ==="
  SRTB_FFT_WAVE=2 timeout 300 python - <<'PY'
import sys
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(16,2048))+1j*rng.normal(size=(16,2048))).astype(np.complex64)).cuda()
errs=[]
for sign in (-1,1):
    ref = torch.fft.fft(x,dim=1) if sign==-1 else torch.fft.ifft(x,dim=1)*2048
    errs.append(((C.native_fft(x,sign)-ref).abs().max()/ref.abs().max()).item())
print("wave32 len=2048 err=%.2e t=%.3f ms" % (max(errs), C.bench_fft(2048, (1<<27)//2048, 1, 30, "native")))
PY
  echo "=== crossover recheck 2^17 x 4096 + new flagship timings ==="
  timeout 300 python - <<'PY'
import sys
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
C = native(); import torch; torch.cuda.set_device(0)
print("bwd17 native=%.3f rocfft=%.3f" % (C.bench_fft(1<<17, 4096, 1, 20, "native"), C.bench_fft(1<<17, 4096, 1, 20, "hipfft")))
print("bwd16 native=%.3f rocfft=%.3f" % (C.bench_fft(1<<16, 8192, 1, 20, "native"), C.bench_fft(1<<16, 8192, 1, 20, "hipfft")))
print("fwd29 %.3f bwd18 %.3f" % (C.bench_fft(1<<29,1,-1,20,"native"), C.bench_fft(1<<18,2048,1,20,"native")))
PY
  echo "=== flagship bench ==="
  timeout 600 python bench.py --steps 10 --warmup 2 --blocks-per-step 4 2>&1 | tail -1
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check8.log 2>&1
tail -25 gpurun_out/r02_check8.log
