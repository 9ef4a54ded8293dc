#!/bin/bash
# Probe a length-64 DIF final pass (3 radix-4 stages, store runs = F*8 B)
# via the factor-override env — no planner changes needed.
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== fwd 2^29 with final 64 ==="
  for df in 32 64 128; do
    SRTB_FFT_FACTORS="64,64,64,32,64" SRTB_FFT_DIF_F=$df timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(1,1<<24))+1j*rng.normal(size=(1,1<<24))).astype(np.complex64)).cuda()
# correctness at a reduced len with the same final (factors won't apply
# to 2^24; check the 2^29 plan only via bench numerics? use full len, small check)
x29 = torch.from_numpy((rng.normal(size=(1,1<<29))+1j*rng.normal(size=(1,1<<29))).astype(np.complex64)).cuda()
ref = torch.fft.fft(x29, dim=1)
err=((C.native_fft(x29,-1)-ref).abs().max()/ref.abs().max()).item()
print("F=%s err=%.2e fwd29=%.3f ms" % (os.environ["SRTB_FFT_DIF_F"], err,
      C.bench_fft(1<<29,1,-1,20,"native")))
PY
  done
  echo "=== bwd 2^18 x 2048 with final 64 ==="
  for df in 32 64 128; do
    SRTB_FFT_FACTORS="64,64,64" SRTB_FFT_DIF_F=$df timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(8,1<<18))+1j*rng.normal(size=(8,1<<18))).astype(np.complex64)).cuda()
ref = torch.fft.ifft(x, dim=1)*(1<<18)
err=((C.native_fft(x,1)-ref).abs().max()/ref.abs().max()).item()
print("F=%s err=%.2e bwd18=%.3f ms" % (os.environ["SRTB_FFT_DIF_F"], err,
      C.bench_fft(1<<18,2048,1,20,"native")))
PY
  done
  echo "=== ALL DONE ==="
} > gpurun_out/r02_dif64.log 2>&1
tail -20 gpurun_out/r02_dif64.log
