#!/usr/bin/env python3
"""Reproduce the reference's J1644-4559 correctness demo on synthetic data:
synthesize a 2-bit baseband recording (noise block + a block containing a
pulse dispersed at DM 478.80 over the 1405-1469 MHz reversed band), replay it
through the NATIVE srtb-backend executable with the shipped config, and plot
the dedispersed waterfall + time series of the detection.

Run on a GPU box:  python scripts/make_j1644_demo.py --out gpurun_out/demo
"""

import argparse
import glob
import os
import subprocess
import sys

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default="gpurun_out/demo")
    ap.add_argument("--n", type=int, default=2**26)
    ap.add_argument("--channels", type=int, default=2**11)
    args = ap.parse_args()
    os.makedirs(args.out, exist_ok=True)

    from srtb_amd.config import Config
    from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse

    cfg = Config()
    cfg.baseband_input_count = args.n
    cfg.spectrum_channel_count = args.channels
    cfg.baseband_input_bits = 2
    cfg.baseband_freq_low = 1437.0   # 1405 + 64/2
    cfg.baseband_bandwidth = -64.0
    cfg.baseband_sample_rate = 128e6
    cfg.dm = -478.80
    print("synthesizing dispersed pulse (DM -478.80, 2-bit)...", flush=True)
    rng = np.random.default_rng(1644)
    # the pulse must land in the VALID window of its replay block: with
    # overlap on, only the first (N - nsamps_reserved) samples of a block
    # survive truncation (~65% here), and block 2 starts at N - reserved
    from srtb_amd import ref as _ref
    reserved = _ref.nsamps_reserved(args.n, args.channels, 1437.0, -64.0,
                                    128e6, -478.80)
    t_pulse = 0.25 * (args.n - reserved) / cfg.baseband_sample_rate
    pulse_block = synthesize_dispersed_pulse(cfg, t_pulse, pulse_amp=1.0,
                                             noise_sigma=2.0, rng=rng)
    lv = rng.integers(0, 4, args.n, dtype=np.uint8).reshape(-1, 4)
    noise_block = ((lv[:, 0] << 6) | (lv[:, 1] << 4) | (lv[:, 2] << 2)
                   | lv[:, 3]).astype(np.uint8)
    rec = os.path.join(args.out, "j1644_synth.bin")
    # layout: noise block, then the pulse block aligned to replay block 2's
    # start (= N - reserved samples in)
    step_bytes = (args.n - reserved) * 2 // 8
    buf = np.concatenate([noise_block[:step_bytes], pulse_block])
    buf.tofile(rec)

    print("replaying through bin/srtb-backend ...", flush=True)
    out = subprocess.run(
        [os.path.join(ROOT, "bin", "srtb-backend"),
         "--config_file_name",
         os.path.join(ROOT, "configs", "srtb_config_1644-4559.cfg"),
         "--baseband_input_count", str(args.n),
         "--spectrum_channel_count", str(args.channels),
         "--baseband_reserve_sample", "1",
         "--input_file_path", rec,
         "--baseband_output_file_prefix", args.out + "/det_"],
        capture_output=True, text=True, timeout=600)
    print(out.stdout[-2000:])
    print(out.stderr[-2000:])
    assert out.returncode == 0

    npys = sorted(glob.glob(args.out + "/det_*.npy"))
    tims = sorted(glob.glob(args.out + "/det_*.tim"))
    print("products:", npys, tims)
    assert npys, "no detection waterfall written"

    import matplotlib
    matplotlib.use("Agg")
    import matplotlib.pyplot as plt

    wf = np.load(npys[0])
    power = np.abs(wf) ** 2
    fig, (ax1, ax2) = plt.subplots(
        2, 1, figsize=(12, 8), sharex=True,
        gridspec_kw={"height_ratios": [3, 1]})
    tsamp = 2 * args.channels / cfg.baseband_sample_rate * 1e3  # ms per bin
    extent = [0, power.shape[1] * tsamp, 1437.0 - 64.0, 1437.0]
    ax1.imshow(np.log10(power + 1e-30), aspect="auto", origin="lower",
               cmap="viridis", extent=extent)
    ax1.set_ylabel("frequency [MHz]")
    ax1.set_title("J1644-4559 synthetic replay: dedispersed waterfall "
                  "(DM 478.80, 2-bit baseband, native srtb-backend)")
    ts = np.fromfile(tims[0], dtype=np.float32)
    snr = (ts - ts.mean()) / ts.std()
    ax2.plot(np.arange(ts.size) * tsamp, snr, lw=0.5)
    ax2.set_xlabel("time [ms]")
    ax2.set_ylabel("S/N")
    png = os.path.join(args.out, "result_j1644_synth.png")
    plt.savefig(png, dpi=110, bbox_inches="tight")
    print("wrote", png, "peak S/N", float(snr.max()))


if __name__ == "__main__":
    main()
