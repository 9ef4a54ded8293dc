#!/usr/bin/env python3
"""Sweep SRTB_FFT_FACTORS plan shapes for a given transform: correctness
against torch.fft at a reduced length, then isolated timing.  Run on a GPU
box, e.g.:

    SWEEP_LEN=$((1<<18)) SWEEP_BATCH=2048 SWEEP_SIGN=1 \
        python scripts/fft_factor_sweep.py "" "32,32,256" "64,16,256"
"""

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CHILD = '''
import sys, os
sys.path.insert(0, {root!r})
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
LEN = int(os.environ.get("SWEEP_LEN", str(1 << 29)))
BATCH = int(os.environ.get("SWEEP_BATCH", "1"))
SIGN = int(os.environ.get("SWEEP_SIGN", "-1"))
rng = np.random.default_rng(1)
# correctness at the SWEEP length itself when feasible (so the override
# plan is the one checked), on a reduced batch
chk_batch = min(BATCH, max(1, (1 << 24) // LEN))
x = torch.from_numpy((rng.normal(size=(chk_batch, LEN))
                      + 1j * rng.normal(size=(chk_batch, LEN))
                      ).astype(np.complex64)).cuda()
out = C.native_fft(x, SIGN)
ref = (torch.fft.fft(x, dim=1) if SIGN == -1
       else torch.fft.ifft(x, dim=1) * LEN)
err = (out - ref).abs().max().item() / ref.abs().max().item()
t = C.bench_fft(LEN, BATCH, SIGN, 15)
print("err=%.2e t=%.3f ms" % (err, t))
'''


def main():
    variants = sys.argv[1:] or [
        "", "64,64,64,8,256", "64,64,32,16,256",
        "64,32,32,32,256", "32,64,64,8,256", "8,64,64,64,256"]
    child = CHILD.format(root=ROOT)
    for fac in variants:
        env = dict(os.environ)
        env.pop("SRTB_FFT_FACTORS", None)
        if fac:
            env["SRTB_FFT_FACTORS"] = fac
        r = subprocess.run([sys.executable, "-c", child],
                           capture_output=True, text=True, env=env,
                           timeout=240)
        label = fac if fac else "default"
        out = r.stdout.strip() or r.stderr.strip()[-200:]
        print(f"factors [{label}]: {out}", flush=True)


if __name__ == "__main__":
    main()
