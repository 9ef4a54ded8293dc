#!/usr/bin/env python3
"""Sweep SRTB_FFT_FACTORS plan shapes for the forward 2^29 C2C: correctness
at 2^20 against torch.fft, then isolated timing.  Run on a GPU box."""

import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

CHILD = """
import sys
sys.path.insert(0, %r)
from srtb_amd.ops import native
import torch, numpy as np
C = native(); torch.cuda.set_device(0)
rng = np.random.default_rng(1)
x = torch.from_numpy((rng.normal(size=(1, 1 << 20))
                      + 1j * rng.normal(size=(1, 1 << 20))
                      ).astype(np.complex64)).cuda()
out = C.native_fft(x, -1)
ref = torch.fft.fft(x, dim=1)
err = (out - ref).abs().max().item() / ref.abs().max().item()
t = C.bench_fft(1 << 29, 1, -1, 15)
print("err=%%.2e fwd2^29=%%.3f ms" %% (err, t))
""" % ("ROOT_PLACEHOLDER",)


def main():
    variants = sys.argv[1:] or [
        "", "64,64,64,8,256", "64,64,32,16,256",
        "64,32,32,32,256", "32,64,64,8,256", "8,64,64,64,256"]
    child = CHILD.replace("ROOT_PLACEHOLDER", ROOT)
    for fac in variants:
        env = dict(os.environ)
        env.pop("SRTB_FFT_FACTORS", None)
        # the 2^20 correctness check uses its own default plan; only 2^29
        # uses 5 factors, so set the env for both (invalid for 2^20 -> the
        # override is ignored there by the product check)
        if fac:
            env["SRTB_FFT_FACTORS"] = fac
        r = subprocess.run([sys.executable, "-c", child],
                           capture_output=True, text=True, env=env,
                           timeout=180)
        label = fac if fac else "default"
        out = r.stdout.strip() or r.stderr.strip()[-200:]
        print(f"factors [{label}]: {out}", flush=True)


if __name__ == "__main__":
    main()
