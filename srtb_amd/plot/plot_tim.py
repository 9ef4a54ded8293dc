"""Offline plot of a detection time series dump (.tim) — reference
src/plot_tim.py equivalent.  Falls back to a text summary without matplotlib.

Usage: python -m srtb_amd.plot.plot_tim dump.tim [out.png]
"""

from __future__ import annotations

import sys

import numpy as np


def main(argv=None) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    if not argv:
        print(__doc__)
        return 2
    path = argv[0]
    out = argv[1] if len(argv) > 1 else None
    ts = np.fromfile(path, dtype=np.float32)
    std = ts.std() or 1.0
    snr = (ts - ts.mean()) / std
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        plt.figure(figsize=(12, 4))
        plt.plot(snr)
        plt.xlabel("time bin")
        plt.ylabel("S/N")
        out = out or (path + ".png")
        plt.savefig(out, dpi=120, bbox_inches="tight")
        print(f"wrote {out}")
    except ImportError:
        peak = int(np.argmax(snr))
        print(f"{path}: n={ts.size} mean={ts.mean():.4e} std={std:.4e} "
              f"peak_snr={snr.max():.2f} at bin {peak}")
        for i in np.argsort(snr)[-10:][::-1]:
            print(f"  bin {int(i):8d}  snr {snr[i]:8.2f}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
