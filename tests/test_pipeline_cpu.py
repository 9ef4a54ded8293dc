"""Full-pipeline integration test on synthetic dispersed pulses (CPU oracle).

This is the end-to-end validation the reference lacks: inject a
pulse dispersed at a known DM into noise, run the whole chain, and assert the
pulse is detected at the right time bin.
"""

import numpy as np
import pytest

from srtb_amd.config import Config
from srtb_amd.pipeline.cpu import CpuPipeline, synthesize_dispersed_pulse


def small_cfg(bits=-8, dm=60.0):
    c = Config()
    c.baseband_input_count = 1 << 18
    c.spectrum_channel_count = 1 << 6
    c.baseband_input_bits = bits
    c.baseband_freq_low = 1400.0
    c.baseband_bandwidth = 64.0
    c.baseband_sample_rate = 128e6
    c.dm = dm
    c.mitigate_rfi_average_method_threshold = 1e30  # keep all bins (pulse is wideband)
    c.mitigate_rfi_spectral_kurtosis_threshold = 1e30
    c.signal_detect_signal_noise_threshold = 6.0
    c.signal_detect_max_boxcar_length = 16
    c.baseband_reserve_sample = True
    return c


def test_pipeline_detects_dispersed_pulse():
    cfg = small_cfg()
    pipe = CpuPipeline(cfg)
    t_pulse = 0.4 * cfg.baseband_input_count / cfg.baseband_sample_rate
    raw = synthesize_dispersed_pulse(cfg, t_pulse, pulse_amp=40.0, noise_sigma=2.0)
    res = pipe.process_block(raw)
    assert len(res["detections"]) > 0, "injected pulse not detected"
    ts = res["time_series"]
    peak_bin = int(np.argmax(ts))
    # pulse lands at time bin t_pulse / (2S / fs)
    expect_bin = int(t_pulse * cfg.baseband_sample_rate) // (2 * cfg.spectrum_channel_count)
    assert abs(peak_bin - expect_bin) <= 2


def test_pipeline_quiet_on_pure_noise():
    cfg = small_cfg()
    cfg.signal_detect_signal_noise_threshold = 8.0
    pipe = CpuPipeline(cfg)
    rng = np.random.default_rng(0)
    raw = np.clip(np.round(rng.normal(0, 16, cfg.baseband_input_count)),
                  -128, 127).astype(np.int8).view(np.uint8)
    res = pipe.process_block(raw)
    assert len(res["detections"]) == 0


def test_pipeline_wrong_dm_misses_pulse():
    """Dedispersing at the wrong DM smears the pulse below threshold."""
    cfg = small_cfg(dm=60.0)
    t_pulse = 0.4 * cfg.baseband_input_count / cfg.baseband_sample_rate
    raw = synthesize_dispersed_pulse(cfg, t_pulse, pulse_amp=30.0, noise_sigma=2.0)
    good = CpuPipeline(cfg).process_block(raw)

    cfg_wrong = small_cfg(dm=0.0)
    wrong = CpuPipeline(cfg_wrong).process_block(raw)
    peak_good = np.max(good["time_series"]) / np.std(good["time_series"])
    peak_wrong = np.max(wrong["time_series"]) / np.std(wrong["time_series"])
    assert peak_good > 2 * peak_wrong


def test_pipeline_2bit_path():
    cfg = small_cfg(bits=2)
    pipe = CpuPipeline(cfg)
    t_pulse = 0.5 * cfg.baseband_input_count / cfg.baseband_sample_rate
    raw = synthesize_dispersed_pulse(cfg, t_pulse, pulse_amp=12.0, noise_sigma=2.0)
    assert raw.size == cfg.baseband_input_bytes
    res = pipe.process_block(raw)
    # 2-bit quantization loses SNR but the pulse must still be the peak
    expect_bin = int(t_pulse * cfg.baseband_sample_rate) // (2 * cfg.spectrum_channel_count)
    peak_bin = int(np.argmax(res["time_series"]))
    assert abs(peak_bin - expect_bin) <= 2


def test_pipeline_waterfall_shape():
    cfg = small_cfg()
    pipe = CpuPipeline(cfg)
    rng = np.random.default_rng(1)
    raw = rng.integers(0, 256, cfg.baseband_input_count, dtype=np.uint8)
    cfg.baseband_input_bits = 8
    res = pipe.process_block(raw)
    S = cfg.spectrum_channel_count
    L = cfg.nsamps_complex // S
    assert res["waterfall"].shape == (S, L)
    reserved_bins = pipe.nsamps_reserved() // S
    assert res["time_series"].size == L - reserved_bins
