#!/usr/bin/env python3
"""BASELINE.json config 5: Crab giant-pulse search — 16-bit baseband,
coherent DM-trial sweep + boxcar detection, DM trials sharded across GPUs.

Each rank uploads its synthetic 16-bit block once and re-runs the chain from
the R2C spectrum at its share of the DM trial list (the engine's per-submit
DM override recomputes the fp64 phase on the fly); per-trial peak SNRs are
all-gathered and rank 0 reports the best trial.

Usage: python benchmarks/crab_dm_sweep.py [--n 2**28] [--trials 16]
       torchrun --nproc-per-node 8 benchmarks/crab_dm_sweep.py
"""

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", type=int, default=2**28)
    ap.add_argument("--channels", type=int, default=2**11)
    ap.add_argument("--trials", type=int, default=16)
    ap.add_argument("--dm-center", type=float, default=56.77)  # Crab DM
    ap.add_argument("--dm-span", type=float, default=8.0)
    ap.add_argument("--inject", action="store_true",
                    help="inject a dispersed pulse at dm-center")
    args = ap.parse_args()

    import torch
    from srtb_amd.config import Config
    from srtb_amd.pipeline.gpu import DmTrialSweep
    from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local_rank)
    if world > 1:
        torch.distributed.init_process_group(backend="nccl")

    cfg = Config()
    cfg.baseband_input_count = args.n
    cfg.spectrum_channel_count = args.channels
    cfg.baseband_input_bits = -16
    cfg.baseband_freq_low = 1000.0
    cfg.baseband_bandwidth = 500.0
    cfg.baseband_sample_rate = 1e9
    cfg.dm = args.dm_center
    cfg.mitigate_rfi_average_method_threshold = 1e30
    cfg.mitigate_rfi_spectral_kurtosis_threshold = 1e30
    cfg.signal_detect_signal_noise_threshold = 8.0
    cfg.signal_detect_max_boxcar_length = 256

    if args.inject:
        cfg8 = Config()
        for k in ("baseband_input_count", "spectrum_channel_count",
                  "baseband_freq_low", "baseband_bandwidth",
                  "baseband_sample_rate", "dm"):
            setattr(cfg8, k, getattr(cfg, k))
        cfg8.baseband_input_bits = -8
        t = 0.4 * args.n / cfg.baseband_sample_rate
        raw8 = synthesize_dispersed_pulse(cfg8, t, pulse_amp=35.0,
                                          noise_sigma=2.0)
        raw = (raw8.view(np.int8).astype(np.int16) * 64).view(np.uint8)
    else:
        rng = np.random.default_rng(7)
        raw = rng.integers(0, 256, 2 * args.n, dtype=np.uint8)

    dms = np.linspace(args.dm_center - args.dm_span,
                      args.dm_center + args.dm_span, args.trials)
    if args.inject:
        dms[args.trials // 2] = args.dm_center  # ensure the true DM is a trial
    my_dms = [float(d) for i, d in enumerate(dms) if i % world == rank]

    sweep = DmTrialSweep(cfg, nsamps_reserved=0)
    # warmup one trial
    sweep.sweep(raw, my_dms[:1])
    if world > 1:
        torch.distributed.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    trials = sweep.sweep(raw, my_dms)
    torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0

    # gather (dm, peak_snr) across ranks
    local = torch.tensor([[t.dm, t.peak_snr] for t in trials],
                         dtype=torch.float64, device="cuda")
    if world > 1:
        sizes = [len(dms) // world + (1 if r < len(dms) % world else 0)
                 for r in range(world)]
        gathered = [torch.zeros(sz, 2, dtype=torch.float64, device="cuda")
                    for sz in sizes]
        torch.distributed.all_gather(gathered, local)
        allt = torch.cat(gathered, 0)
    else:
        allt = local
    if rank == 0:
        best = allt[allt[:, 1].argmax()]
        print(json.dumps({
            "metric": "DM trials/s (coherent, 16-bit baseband)",
            "value": round(len(dms) / elapsed, 2),
            "unit": "trials/s",
            "n_gpus": world,
            "trials": len(dms),
            "ms_per_trial": round(elapsed / max(len(my_dms), 1) * 1e3, 1),
            "samples_per_trial": args.n,
            "best_dm": round(float(best[0]), 3),
            "best_peak_snr": round(float(best[1]), 1),
            "higher_is_better": True,
            "scaling": "strong",
        }), flush=True)
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
