#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== stockham radix-16: correctness + timing ==="
  for ln in 256 512 1024 2048; do
    SWEEP_LEN=$ln timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
LEN=int(os.environ["SWEEP_LEN"]); BATCH=(1<<27)//LEN
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(8,LEN))+1j*rng.normal(size=(8,LEN))).astype(np.complex64)).cuda()
ref = torch.fft.ifft(x,dim=1)*LEN
err=((C.native_fft(x,1)-ref).abs().max()/ref.abs().max()).item()
t16 = C.bench_fft(LEN,BATCH,1,30,"native")
print(f"len={LEN} err={err:.2e} sp16={t16:.3f} ms")
PY
    SWEEP_LEN=$ln SRTB_FFT_SP16=0 timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
C = native(); import torch; torch.cuda.set_device(0)
LEN=int(os.environ["SWEEP_LEN"]); BATCH=(1<<27)//LEN
print(f"len={LEN} r4only={C.bench_fft(LEN,BATCH,1,30,'native'):.3f} ms")
PY
  done
  echo "=== full gpu fft tests ==="
  timeout 900 python -m pytest tests/test_gpu_fft.py -q 2>&1 | tail -2
  echo "=== UDP test (fixed) ==="
  timeout 300 python -m pytest tests/test_native_app.py::test_srtb_backend_udp_ingest_with_overlap -x -q 2>&1 | tail -3
  echo "=== 2pol + rccl native tests ==="
  timeout 600 python -m pytest tests/test_native_app.py -q -m gpu 2>&1 | tail -2
  echo "=== flagship bench sanity ==="
  timeout 420 python bench.py --steps 2 --warmup 1 --blocks-per-step 2 2>&1 | tail -1
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check4.log 2>&1
tail -60 gpurun_out/r02_check4.log
