#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== wave default + recurrence: correctness + timing ==="
  for ln in 256 512 1024; do
    SWEEP_LEN=$ln timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
LEN=int(os.environ["SWEEP_LEN"]); BATCH=(1<<27)//LEN
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(16,LEN))+1j*rng.normal(size=(16,LEN))).astype(np.complex64)).cuda()
errs=[]
for sign in (-1,1):
    ref = torch.fft.fft(x,dim=1) if sign==-1 else torch.fft.ifft(x,dim=1)*LEN
    errs.append(((C.native_fft(x,sign)-ref).abs().max()/ref.abs().max()).item())
print(f"len={LEN} err={max(errs):.2e} wave_rec={C.bench_fft(LEN,BATCH,1,30,'native'):.3f} ms")
PY
  done
  echo "=== full gpu fft + engine suites ==="
  timeout 900 python -m pytest tests/test_gpu_fft.py tests/test_gpu_engine.py tests/test_gpu_kernels.py -q 2>&1 | tail -2
  echo "=== 2pol tests (race fixed) ==="
  timeout 600 python -m pytest tests/test_gpu_main_app.py::test_main_gpu_2pol_fanout tests/test_native_app.py -q -m gpu 2>&1 | tail -2
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check7.log 2>&1
tail -30 gpurun_out/r02_check7.log
