#!/usr/bin/env python3
"""Round-2 FFT tuning measurements (run on a GPU box).

Sweeps the column-pass tuning experiments (SRTB_FFT_SWIZZLE / SRTB_FFT_NT)
on the two flagship shapes, measures the mid512 F=8 4-pass plan, soaks the
pair32 kernel, and baselines the single-pass (<=4096) kernel.
Each variant runs in a fresh process (env is read once per process).
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
LEN = int(os.environ["SWEEP_LEN"]); BATCH = int(os.environ["SWEEP_BATCH"])
SIGN = int(os.environ["SWEEP_SIGN"]); ITERS = int(os.environ.get("SWEEP_ITERS", "15"))
rng = np.random.default_rng(1)
chk_batch = min(BATCH, max(1, (1 << 24) // LEN))
x = torch.from_numpy((rng.normal(size=(chk_batch, LEN))
                      + 1j * rng.normal(size=(chk_batch, LEN))
                      ).astype(np.complex64)).cuda()
out = C.native_fft(x, SIGN)
ref = (torch.fft.fft(x, dim=1) if SIGN == -1 else torch.fft.ifft(x, dim=1) * LEN)
err = (out - ref).abs().max().item() / ref.abs().max().item()
backend = os.environ.get("SWEEP_BACKEND", "native")
t = C.bench_fft(LEN, BATCH, SIGN, ITERS, backend)
print("err=%.2e t=%.3f ms" % (err, t))
'''


def run(label, ln, batch, sign, env_extra, iters=15):
    env = dict(os.environ)
    for k in ("SRTB_FFT_FACTORS", "SRTB_FFT_SWIZZLE", "SRTB_FFT_NT",
              "SRTB_FFT_PAIR32", "SRTB_FFT_MIDF"):
        env.pop(k, None)
    env.update(env_extra)
    env["SWEEP_LEN"] = str(ln)
    env["SWEEP_BATCH"] = str(batch)
    env["SWEEP_SIGN"] = str(sign)
    env["SWEEP_ITERS"] = str(iters)
    r = subprocess.run([sys.executable, "-c", CHILD.format(root=ROOT)],
                       capture_output=True, text=True, env=env, timeout=600)
    out = r.stdout.strip() or r.stderr.strip()[-300:]
    print(f"[{label}] {out}", flush=True)


def main():
    combos = [("base", {}),
              ("swz", {"SRTB_FFT_SWIZZLE": "1"}),
              ("nt", {"SRTB_FFT_NT": "1"}),
              ("swz+nt", {"SRTB_FFT_SWIZZLE": "1", "SRTB_FFT_NT": "1"})]
    print("== fwd 2^29 C2C (greedy [64,64,64,8]x256) ==")
    for lb, e in combos:
        run(f"fwd29 {lb}", 1 << 29, 1, -1, e)
    print("== bwd 2^18 x 2048 ==")
    for lb, e in combos:
        run(f"bwd18 {lb}", 1 << 18, 2048, 1, e)
    print("== mid512 F=8 4-pass plan (2^29) ==")
    run("mid512 F=8", 1 << 29, 1, -1,
        {"SRTB_FFT_FACTORS": "64,64,512,256", "SRTB_FFT_MIDF": "8"})
    run("mid512 F=8 +swz+nt", 1 << 29, 1, -1,
        {"SRTB_FFT_FACTORS": "64,64,512,256", "SRTB_FFT_MIDF": "8",
         "SRTB_FFT_SWIZZLE": "1", "SRTB_FFT_NT": "1"})
    print("== pair32 soak (bwd 2^18 shape, 300 iters) ==")
    run("pair32 soak", 1 << 18, 2048, 1, {"SRTB_FFT_PAIR32": "1"}, iters=300)
    print("== single-pass small lengths (batched) ==")
    for ln in (1024, 2048, 4096):
        run(f"small {ln} native", ln, (1 << 27) // ln, 1, {})
        run(f"small {ln} rocfft", ln, (1 << 27) // ln, 1,
            {"SWEEP_BACKEND": "hipfft"})


if __name__ == "__main__":
    main()
