"""GPU-side orchestration helpers above the native engine.

- DualPolPipeline: one packet stream fanning out to 2 polarization sample
  streams (naocpsr_snap1 / gznupsr_a1 / byte-interleaved cpsr2), each run
  through the full chain (reference unpack_pipe.hpp:146-390 fan-out).
- DmTrialSweep: coherent DM-trial search over one block (Crab giant-pulse
  search config of BASELINE.json) using the engine's per-submit DM override.
"""

from __future__ import annotations

from dataclasses import dataclass

import numpy as np

from .. import ref
from ..config import Config


def _make_engine(C, cfg: Config, nsamps_reserved: int, nbits: int,
                 n: int | None = None, n_slots: int = 2):
    nc = (n or cfg.baseband_input_count) // 2
    ranges = ref.parse_rfi_freq_list(cfg.mitigate_rfi_freq_list)
    bins = ref.rfi_ranges_to_bins(cfg.baseband_freq_low,
                                  cfg.baseband_bandwidth, nc, ranges)
    return C.PipelineEngine(
        n=n or cfg.baseband_input_count, nbits=nbits,
        channels=cfg.spectrum_channel_count, freq_low=cfg.baseband_freq_low,
        bandwidth=cfg.baseband_bandwidth, sample_rate=cfg.baseband_sample_rate,
        dm=cfg.dm, rfi_threshold=cfg.mitigate_rfi_average_method_threshold,
        sk_threshold=cfg.mitigate_rfi_spectral_kurtosis_threshold,
        snr_threshold=cfg.signal_detect_signal_noise_threshold,
        max_boxcar=cfg.signal_detect_max_boxcar_length,
        nsamps_reserved=nsamps_reserved,
        zap_ranges=[[int(a), int(b)] for a, b in bins],
        use_phase_table=False, enable_rfi_s1=True, enable_sk=True,
        n_slots=n_slots)


class DualPolPipeline:
    """Process a 2-polarization packed stream: one unpack fan-out on the GPU,
    then the per-pol chain through one engine per polarization."""

    def __init__(self, cfg: Config, nsamps_reserved: int, kind: str):
        import torch
        from ..ops import native
        self.torch = torch
        self.C = native()
        self.kind = kind  # "interleave" | "naocpsr_snap1" | "gznupsr_a1"
        self.cfg = cfg
        # per-pol blocks have baseband_input_count samples each
        self.engines = [
            _make_engine(self.C, cfg, nsamps_reserved, nbits=-8),
            _make_engine(self.C, cfg, nsamps_reserved, nbits=-8),
        ]

    def submit_block(self, raw: np.ndarray) -> dict:
        """Fan one packed block out and enqueue both polarization chains;
        returns a handle for wait_block.  The handle owns the fan-out
        tensors: they must stay alive until the engines drained them (the
        caching allocator would otherwise recycle their memory under the
        engine streams)."""
        torch = self.torch
        n = self.cfg.baseband_input_count
        raw_t = torch.from_numpy(np.ascontiguousarray(raw)).cuda()
        if self.kind == "gznupsr_a1":
            pols = self.C.unpack_gznupsr_a1(raw_t, 2)
        else:
            pols = self.C.unpack_2pol(raw_t, self.kind)
        slots = []
        for eng, pol in zip(self.engines, pols):
            assert pol.numel() == n
            slots.append(eng.submit_samples(pol))
        return {"slots": slots, "pols": pols, "raw": raw_t}

    def wait_block(self, handle: dict) -> list[dict]:
        out = []
        for eng, slot in zip(self.engines, handle["slots"]):
            res = eng.wait(slot)
            res["slot"] = slot
            out.append(res)
        return out

    def process_block(self, raw: np.ndarray) -> list[dict]:
        """raw: packed bytes holding 2 * baseband_input_count int8 samples.
        Returns one result dict per polarization."""
        return self.wait_block(self.submit_block(raw))


@dataclass
class DmTrial:
    dm: float
    counts: list
    zero_count: int
    peak_snr: float


class DmTrialSweep:
    """Coherent DM-trial sweep over one baseband block.

    The block is uploaded once; each trial re-runs the chain from the R2C
    spectrum with a different dedispersion DM (on-the-fly fp64 phase).  The
    best trial is the one with the highest peak SNR in its time series.
    """

    def __init__(self, cfg: Config, nsamps_reserved: int = 0):
        import torch
        from ..ops import native
        self.torch = torch
        self.C = native()
        self.cfg = cfg
        self.eng = _make_engine(self.C, cfg, nsamps_reserved,
                                nbits=cfg.baseband_input_bits)

    def sweep(self, raw: np.ndarray, dms: list[float]) -> list[DmTrial]:
        torch = self.torch
        raw_t = torch.from_numpy(np.ascontiguousarray(raw)).cuda()
        trials = []
        # keep both engine slots in flight: trial i+1 uploads/computes while
        # trial i drains (the serial submit→wait loop left the GPU idle
        # during each host-side readback)
        inflight: list[tuple[float, int]] = []
        n_slots = self.eng.n_slots

        def drain_one():
            dm, slot = inflight.pop(0)
            res = self.eng.wait(slot)
            ts = self.eng.time_series(slot)
            std = float(ts.std())
            peak = float(ts.max()) / std if std > 0 else 0.0
            trials.append(DmTrial(dm=float(dm), counts=res["counts"],
                                  zero_count=res["zero_count"],
                                  peak_snr=peak))

        for dm in dms:
            inflight.append((float(dm),
                             self.eng.submit(raw_t, dm_override=float(dm))))
            if len(inflight) >= n_slots:
                drain_one()
        while inflight:
            drain_one()
        return trials

    def best(self, trials: list[DmTrial]) -> DmTrial:
        return max(trials, key=lambda t: t.peak_snr)
