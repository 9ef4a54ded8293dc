"""CPU (NumPy) pipeline — the plumbing/oracle path.

Composes the reference-semantics ops in :mod:`srtb_amd.ref` into the full
per-block chain of the reference's streaming data path
(reference src/main.cpp:167-228):

    unpack(+window) → R2C FFT (drop Nyquist) → RFI s1 (mean zap + normalize
    + manual zap) → coherent dedispersion → waterfall batched iFFT →
    RFI s2 (spectral kurtosis) → signal detect (row-sum, baseline, boxcar)

Used for: CPU-only plumbing runs (BASELINE.json config 1), numerics oracle for
the GPU engine, and the full-pipeline integration test with synthetic
dispersed pulses (which the reference lacks — SURVEY.md §4).
"""

from __future__ import annotations

import numpy as np

from .. import ref
from ..config import Config


class CpuPipeline:
    """Single-stream, single-block-at-a-time CPU pipeline."""

    def __init__(self, cfg: Config):
        self.cfg = cfg
        self.rfi_ranges = ref.parse_rfi_freq_list(cfg.mitigate_rfi_freq_list)
        self.window_kind = "rectangle"  # reference default_window

    def nsamps_reserved(self) -> int:
        c = self.cfg
        return ref.nsamps_reserved(
            c.baseband_input_count, c.spectrum_channel_count,
            c.baseband_freq_low, c.baseband_bandwidth, c.baseband_sample_rate,
            c.dm, c.baseband_reserve_sample)

    def process_block(self, raw: np.ndarray) -> dict:
        """raw: packed baseband bytes for one block (uint8 array).

        Returns dict with 'spectrum' (post-RFI-s1, pre-dedispersion is not
        kept), 'waterfall' [S][L] complex64, 'time_series', 'detections'.
        """
        c = self.cfg
        window = None
        if self.window_kind != "rectangle":
            window = ref.window_coefficients(self.window_kind, c.baseband_input_count)
        samples = ref.unpack(raw, c.baseband_input_bits, window)
        if samples.size != c.baseband_input_count:
            raise ValueError(
                f"block has {samples.size} samples, expected {c.baseband_input_count}")
        return self.process_samples(samples)

    def process_samples(self, samples: np.ndarray) -> dict:
        """Run the chain from already-unpacked float samples (the entry used
        by multi-polarization fan-out, reference unpack_pipe.hpp:146-390)."""
        c = self.cfg
        spec = ref.fft_r2c_drop_nyquist(samples)  # Nc bins
        spec = ref.rfi_mitigate_s1(spec, c.mitigate_rfi_average_method_threshold,
                                   c.spectrum_channel_count)
        if self.rfi_ranges:
            spec = ref.rfi_mitigate_manual(spec, c.baseband_freq_low,
                                           c.baseband_bandwidth, self.rfi_ranges)

        nc = spec.size
        f_min = c.baseband_freq_low
        f_c = f_min + c.baseband_bandwidth
        df = c.baseband_bandwidth / nc
        spec = ref.coherent_dedisperse(spec, f_min, f_c, df, c.dm)

        n_channels = min(c.spectrum_channel_count, nc)
        wf = ref.waterfall_ifft(spec, n_channels)  # [S][L]
        if self.window_kind != "rectangle":
            # K21: de-apply the FFT window after the backward waterfall FFT
            # (reference fft_pipe.hpp:350-358; rectangle default skips it)
            coef = ref.window_coefficients(self.window_kind, wf.shape[1])
            wf = wf / coef[None, :]
        wf = ref.rfi_mitigate_sk(wf, c.mitigate_rfi_spectral_kurtosis_threshold)

        det = ref.detect_signals(
            wf, self.nsamps_reserved(), c.signal_detect_signal_noise_threshold,
            c.signal_detect_channel_threshold, c.signal_detect_max_boxcar_length)
        return {
            "spectrum": spec,
            "waterfall": wf,
            "time_series": det["time_series"],
            "zero_count": det["zero_count"],
            "detections": det["detections"],
        }


def synthesize_dispersed_pulse(cfg: Config, pulse_t: float, pulse_amp: float,
                               noise_sigma: float = 1.0,
                               rng: np.random.Generator | None = None) -> np.ndarray:
    """Synthesize one block of 8-bit baseband containing Gaussian noise plus a
    dispersed impulse at time ``pulse_t`` (seconds into the block) with the
    config's DM — built in the frequency domain with the *inverse* of the
    dedispersion phase so the pipeline's dedispersion exactly re-aligns it.

    Returns a uint8/int8 packed byte array of the config's bit width.
    """
    rng = rng or np.random.default_rng(42)
    c = cfg
    n = c.baseband_input_count
    nc = n // 2
    # impulse in time domain (band-limited click): delta at pulse_t
    x = np.zeros(n, dtype=np.float64)
    idx = int(pulse_t * c.baseband_sample_rate)
    idx = max(0, min(n - 1, idx))
    # a short wideband pulse (few samples wide, smoothed)
    width = 32
    t = np.arange(-width, width + 1)
    x[np.clip(idx + t, 0, n - 1)] += np.exp(-0.5 * (t / (width / 4)) ** 2)

    # disperse it: multiply spectrum by conj(dedispersion factor)
    X = np.fft.rfft(x)
    f_min = c.baseband_freq_low
    f_c = f_min + c.baseband_bandwidth
    df = c.baseband_bandwidth / nc
    fac = ref.dedisp_phase_factors(nc, f_min, f_c, df, c.dm)
    X[:-1] *= np.conj(fac.astype(np.complex128))
    x_disp = np.fft.irfft(X, n)
    x_disp *= pulse_amp / max(np.abs(x_disp).max(), 1e-30)

    sig = x_disp + rng.normal(0, noise_sigma, n)
    bits = c.baseband_input_bits
    if bits == -8:
        q = np.clip(np.round(sig), -128, 127).astype(np.int8)
        return q.view(np.uint8)
    if bits == 8:
        q = np.clip(np.round(sig + 128), 0, 255).astype(np.uint8)
        return q
    if bits == 2:
        # 2-bit quantization around mean: levels 0..3, MSB-first packing
        thr = noise_sigma
        lv = np.digitize(sig, [-thr, 0, thr]).astype(np.uint8)  # 0..3
        lv = lv.reshape(-1, 4)
        packed = (lv[:, 0] << 6) | (lv[:, 1] << 4) | (lv[:, 2] << 2) | lv[:, 3]
        return packed.astype(np.uint8)
    if bits == 4:
        # 4-bit levels 0..15 centered on 8, step = noise_sigma/2
        lv = np.clip(np.round(sig / (noise_sigma / 2) + 8), 0, 15
                     ).astype(np.uint8)
        lv = lv.reshape(-1, 2)
        return ((lv[:, 0] << 4) | lv[:, 1]).astype(np.uint8)
    if bits == 1:
        # 1-bit sign quantization, MSB-first packing
        lv = (sig > 0).astype(np.uint8).reshape(-1, 8)
        packed = np.zeros(lv.shape[0], dtype=np.uint8)
        for j in range(8):
            packed |= lv[:, j] << (7 - j)
        return packed
    raise ValueError(f"unsupported bits {bits} for synthesis")
