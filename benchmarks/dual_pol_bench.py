#!/usr/bin/env python3
"""BASELINE.json config 3: dual-polarization 8-bit ingest, one packet stream
per rank (2 streams on 2 MI355X), RCCL stream split over xGMI.

Each rank processes blocks of 2 * N interleaved int8 samples: one GPU unpack
fan-out into two polarization sample streams, then the full chain per pol
through two engines; detection counters are all-reduced over RCCL.

Usage: python benchmarks/dual_pol_bench.py [--steps K] [--warmup W] [--n N]
       torchrun --nproc-per-node 2 benchmarks/dual_pol_bench.py
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
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--n", type=int, default=2**28,
                    help="samples per polarization per block")
    ap.add_argument("--channels", type=int, default=2**11)
    args = ap.parse_args()

    import torch
    from srtb_amd.config import Config
    from srtb_amd.pipeline.gpu import DualPolPipeline

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    torch.cuda.set_device(local_rank)
    if world > 1:
        torch.distributed.init_process_group(backend="nccl")

    cfg = Config()
    cfg.baseband_input_count = args.n
    cfg.spectrum_channel_count = args.channels
    cfg.baseband_input_bits = -8
    cfg.baseband_freq_low = 1000.0
    cfg.baseband_bandwidth = 500.0
    cfg.baseband_sample_rate = 1e9
    cfg.dm = 56.77
    cfg.mitigate_rfi_average_method_threshold = 5.0
    cfg.mitigate_rfi_spectral_kurtosis_threshold = 1.05
    cfg.signal_detect_signal_noise_threshold = 8.0
    cfg.signal_detect_max_boxcar_length = 16

    pipe = DualPolPipeline(cfg, nsamps_reserved=0, kind="interleave")
    rng = np.random.default_rng(100 + rank)
    raw = rng.integers(0, 256, 2 * args.n, dtype=np.uint8)

    counts = torch.zeros(2, dtype=torch.int64, device="cuda")

    def agg(results):
        c = [sum(cnt for _, cnt in r["counts"]) for r in results]
        counts.copy_(torch.tensor(c, dtype=torch.int64))
        if world > 1:
            torch.distributed.all_reduce(counts)

    def run(k):
        # keep both engine slots in flight: block i+1's fan-out + chains
        # enqueue while block i drains
        inflight = []
        for _ in range(k):
            inflight.append(pipe.submit_block(raw))
            if len(inflight) >= 2:
                agg(pipe.wait_block(inflight.pop(0)))
        while inflight:
            agg(pipe.wait_block(inflight.pop(0)))

    run(args.warmup)
    if world > 1:
        torch.distributed.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    run(args.steps)
    torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    elapsed = time.perf_counter() - t0
    if world > 1:
        e = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        torch.distributed.all_reduce(e, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(e.item())

    total = 2.0 * args.n * args.steps * world  # both pols
    if rank == 0:
        print(json.dumps({
            "metric": "dual-pol baseband Msamples/s (8-bit, 2 pol/stream)",
            "value": round(total / elapsed / 1e6, 1),
            "unit": "Msamples/s",
            "n_gpus": world,
            "steps": args.steps,
            "ms_per_step": round(elapsed / args.steps * 1e3, 3),
            "higher_is_better": True,
            "scaling": "weak",
            "config": {"n_per_pol": args.n, "bits": -8,
                       "channels": args.channels,
                       "parallelism": f"stream-sharded dp{world} x 2 pol"},
        }), flush=True)
    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
