"""Offline plot of a detection spectrum dump (.npy) — reference
src/plot_spectrum.py equivalent.

Uses matplotlib when available; otherwise renders a PPM heatmap (no
dependency beyond numpy) so the utility works in minimal environments.

Usage: python -m srtb_amd.plot.plot_spectrum dump.npy [out.png|out.ppm]
"""

from __future__ import annotations

import sys

import numpy as np

from .. import ref


def main(argv=None) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    if not argv:
        print(__doc__)
        return 2
    path = argv[0]
    out = argv[1] if len(argv) > 1 else None
    wf = np.load(path)
    power = np.abs(wf) ** 2 if np.iscomplexobj(wf) else wf
    try:
        import matplotlib
        matplotlib.use("Agg")
        import matplotlib.pyplot as plt
        plt.figure(figsize=(12, 6))
        plt.imshow(np.log10(power + 1e-30), aspect="auto",
                   origin="lower", cmap="viridis")
        plt.xlabel("time bin")
        plt.ylabel("frequency channel")
        plt.colorbar(label="log10 power")
        out = out or (path + ".png")
        plt.savefig(out, dpi=120, bbox_inches="tight")
        print(f"wrote {out}")
    except ImportError:
        from ..main import write_ppm
        img = ref.normalize_by_mean(power)
        pix = ref.generate_pixmap(np.clip(img, 0, 1))
        out = out or (path + ".ppm")
        write_ppm(out, pix)
        print(f"matplotlib unavailable; wrote {out}")
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
