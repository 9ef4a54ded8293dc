#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
{
  echo "=== single-pass DIF measurement (new) vs stockham (old) vs rocfft ==="
  for ln in 1024 4096 256; do
    SWEEP_LEN=$ln SWEEP_BATCH=$(( (1<<27) / ln )) SWEEP_SIGN=1 timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
LEN=int(os.environ["SWEEP_LEN"]); BATCH=int(os.environ["SWEEP_BATCH"]); SIGN=int(os.environ["SWEEP_SIGN"])
rng = np.random.default_rng(1)
chk=min(BATCH, max(1,(1<<24)//LEN))
x = torch.from_numpy((rng.normal(size=(chk,LEN))+1j*rng.normal(size=(chk,LEN))).astype(np.complex64)).cuda()
out = C.native_fft(x, SIGN)
ref = torch.fft.ifft(x, dim=1)*LEN if SIGN==1 else torch.fft.fft(x,dim=1)
err=(out-ref).abs().max().item()/ref.abs().max().item()
t_dif = C.bench_fft(LEN,BATCH,SIGN,30,"native")
t_roc = C.bench_fft(LEN,BATCH,SIGN,30,"hipfft")
os.environ["SRTB_FFT_NOSP_DIF"]="1"
print(f"len={LEN} err={err:.2e} dif={t_dif:.3f} ms rocfft={t_roc:.3f} ms")
PY
    SWEEP_LEN=$ln SWEEP_BATCH=$(( (1<<27) / ln )) SRTB_FFT_NOSP_DIF=1 timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
C = native()
import torch
torch.cuda.set_device(0)
LEN=int(os.environ["SWEEP_LEN"]); BATCH=int(os.environ["SWEEP_BATCH"])
print(f"len={LEN} stockham={C.bench_fft(LEN,BATCH,1,30,'native'):.3f} ms")
PY
  done
  echo "=== odd-log2 lengths still stockham (correct + timed) ==="
  SWEEP_LEN=2048 SWEEP_BATCH=65536 SWEEP_SIGN=1 timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
rng=np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(8,2048))+1j*rng.normal(size=(8,2048))).astype(np.complex64)).cuda()
err=(C.native_fft(x,1)-torch.fft.ifft(x,dim=1)*2048).abs().max().item()
print("len=2048 err=%.2e t=%.3f" % (err, C.bench_fft(2048,65536,1,30,"native")))
PY
  echo "=== bwd swizzle re-measure 50 iters ==="
  for v in 0 1; do
    SRTB_FFT_SWIZZLE=$v timeout 300 python - <<'PY'
import sys, os
sys.path.insert(0, "/root/repo")
from srtb_amd.ops import native
C = native()
import torch; torch.cuda.set_device(0)
print("swz=%s bwd18=%.3f ms fwd29=%.3f ms" % (os.environ.get("SRTB_FFT_SWIZZLE"),
      C.bench_fft(1<<18, 2048, 1, 50, "native"), C.bench_fft(1<<29, 1, -1, 30, "native")))
PY
  done
  echo "=== gpu fft test suite ==="
  timeout 900 python -m pytest tests/test_gpu_fft.py -q 2>&1 | tail -3
  echo "=== UDP debug ==="
  bash scripts/r02_udp_debug.sh
  echo "=== ALL DONE ==="
} > gpurun_out/r02_check3.log 2>&1
tail -70 gpurun_out/r02_check3.log
