"""srtb-correlator equivalent: offline cross-correlation of two recordings.

Reference src/correlator.cpp:35-152: unpack both files → R2C FFT →
corr = scale * F1 * conj(F2) → backward C2C → |corr| → output file
(float32).  scale = 1/n to undo the unnormalized FFT round trip.

Usage:
  python -m srtb_amd.tools.correlator file1 file2 out.bin
      [--nbits -8] [--count 2**20] [--offset-bytes 0] [--device cpu|cuda]
"""

from __future__ import annotations

import argparse
import sys

import numpy as np

from .. import ref


def correlate_cpu(x1: np.ndarray, x2: np.ndarray) -> np.ndarray:
    n = x1.size
    f1 = np.fft.rfft(x1)[:-1].astype(np.complex64)
    f2 = np.fft.rfft(x2)[:-1].astype(np.complex64)
    corr = ref.correlate_spectra(f1, f2, 1.0 / n)
    nc = corr.size
    out = np.fft.ifft(corr) * nc
    return np.abs(out).astype(np.float32)


def correlate_gpu(x1: np.ndarray, x2: np.ndarray) -> np.ndarray:
    import torch
    from ..ops import native
    C = native()
    n = x1.size
    t1 = torch.from_numpy(x1).cuda()
    t2 = torch.from_numpy(x2).cuda()
    f1 = torch.fft.rfft(t1)[:-1].contiguous().to(torch.complex64)
    f2 = torch.fft.rfft(t2)[:-1].contiguous().to(torch.complex64)
    corr, _ = C.correlate(f1, f2, 1.0 / n)
    out = torch.fft.ifft(corr) * corr.numel()
    return out.abs().float().cpu().numpy()


def main(argv=None) -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("file1")
    ap.add_argument("file2")
    ap.add_argument("out")
    ap.add_argument("--nbits", type=int, default=-8)
    ap.add_argument("--count", type=str, default="2 ** 20")
    ap.add_argument("--offset-bytes", type=int, default=0)
    ap.add_argument("--device", default="auto")
    args = ap.parse_args(argv)

    from ..utils.expr import evaluate_int
    count = evaluate_int(args.count)
    bits = abs(args.nbits)
    nbytes = count * bits // 8

    sigs = []
    for path in (args.file1, args.file2):
        raw = np.fromfile(path, dtype=np.uint8,
                          count=nbytes, offset=args.offset_bytes)
        if raw.size < nbytes:
            raise SystemExit(f"{path}: too short")
        sigs.append(ref.unpack(raw, args.nbits))

    use_gpu = False
    if args.device != "cpu":
        try:
            import torch
            use_gpu = torch.cuda.is_available()
        except ImportError:
            pass
    out = (correlate_gpu if use_gpu else correlate_cpu)(*sigs)
    out.tofile(args.out)
    # positive-frequency-only correlation (reference keeps Nc bins): a time
    # shift of `lag` samples peaks at index (Nc - lag/2) mod Nc
    nc = out.size
    peak = int(np.argmax(out))
    lag = 2 * (nc - peak) if peak > nc // 2 else -2 * peak
    print(f"[correlator] wrote {args.out} ({out.size} float32); "
          f"peak index {peak} (~lag {lag} samples), value {out.max():.3e}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
