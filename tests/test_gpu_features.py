"""GPU tests of DM-trial sweep, dual-pol fan-out, sample submission."""

import numpy as np
import pytest
import torch

from srtb_amd.config import Config
from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def C():
    from srtb_amd.ops import native
    torch.cuda.set_device(0)
    return native()


def small_cfg(bits=-8, dm=60.0, n=1 << 18):
    c = Config()
    c.baseband_input_count = n
    c.spectrum_channel_count = 1 << 6
    c.baseband_input_bits = bits
    c.baseband_freq_low = 1400.0
    c.baseband_bandwidth = 64.0
    c.baseband_sample_rate = 128e6
    c.dm = dm
    c.mitigate_rfi_average_method_threshold = 1e30
    c.mitigate_rfi_spectral_kurtosis_threshold = 1e30
    c.signal_detect_signal_noise_threshold = 6.0
    c.signal_detect_max_boxcar_length = 16
    return c


def test_dm_override_equals_configured(C):
    cfg = small_cfg(dm=25.0)
    from srtb_amd.pipeline.gpu import _make_engine
    eng_a = _make_engine(C, cfg, 0, nbits=-8)
    cfg_b = small_cfg(dm=99.0)
    eng_b = _make_engine(C, cfg_b, 0, nbits=-8)
    rng = np.random.default_rng(0)
    raw = torch.from_numpy(np.clip(np.round(
        rng.normal(0, 16, cfg.baseband_input_count)), -128, 127)
        .astype(np.int8).view(np.uint8).copy())
    sa = eng_a.submit(raw, dm_override=99.0)
    eng_a.wait(sa)
    sb = eng_b.submit(raw)
    eng_b.wait(sb)
    np.testing.assert_allclose(eng_a.time_series(sa).cpu().numpy(),
                               eng_b.time_series(sb).cpu().numpy(),
                               rtol=1e-5, atol=1e-3)


def test_dm_trial_sweep_finds_true_dm(C):
    from srtb_amd.pipeline.gpu import DmTrialSweep
    cfg = small_cfg(bits=-16, dm=0.0)  # Crab config uses 16-bit baseband
    true_dm = 56.77  # Crab nebula DM
    cfg_pulse = small_cfg(bits=-16, dm=true_dm)
    t = 0.4 * cfg.baseband_input_count / cfg.baseband_sample_rate
    # synthesize at 16 bits: reuse -8 synth then widen
    cfg8 = small_cfg(bits=-8, dm=true_dm)
    raw8 = synthesize_dispersed_pulse(cfg8, t, pulse_amp=35.0, noise_sigma=2.0)
    x16 = (raw8.view(np.int8).astype(np.int16) * 64)
    raw = x16.view(np.uint8)
    sweep = DmTrialSweep(cfg, nsamps_reserved=0)
    dms = [0.0, 20.0, 40.0, true_dm, 80.0, 120.0]
    trials = sweep.sweep(raw, dms)
    best = sweep.best(trials)
    assert best.dm == true_dm, [(t.dm, round(t.peak_snr, 1)) for t in trials]


def test_submit_samples_equals_raw(C):
    from srtb_amd.pipeline.gpu import _make_engine
    cfg = small_cfg()
    eng = _make_engine(C, cfg, 0, nbits=-8)
    rng = np.random.default_rng(1)
    raw_np = np.clip(np.round(rng.normal(0, 16, cfg.baseband_input_count)),
                     -128, 127).astype(np.int8)
    raw = torch.from_numpy(raw_np.view(np.uint8).copy())
    s1 = eng.submit(raw)
    eng.wait(s1)
    ts1 = eng.time_series(s1).cpu().numpy().copy()
    samples = torch.from_numpy(raw_np.astype(np.float32)).cuda()
    s2 = eng.submit_samples(samples)
    eng.wait(s2)
    ts2 = eng.time_series(s2).cpu().numpy()
    np.testing.assert_allclose(ts1, ts2, rtol=1e-6)


def test_dual_pol_pipeline(C):
    from srtb_amd.pipeline.gpu import DualPolPipeline
    cfg = small_cfg()
    cfg.baseband_input_count = 1 << 16
    # two pols with a pulse only in pol 1 (byte-interleaved, cpsr2-style)
    rng = np.random.default_rng(2)
    n = cfg.baseband_input_count
    p0 = np.clip(np.round(rng.normal(0, 8, n)), -128, 127).astype(np.int8)
    cfg_pulse = small_cfg(n=n)
    t = 0.5 * n / cfg.baseband_sample_rate
    p1 = synthesize_dispersed_pulse(cfg_pulse, t, pulse_amp=50.0,
                                    noise_sigma=2.0, rng=rng).view(np.int8)
    raw = np.empty(2 * n, dtype=np.int8)
    raw[0::2] = p0
    raw[1::2] = p1
    pipe = DualPolPipeline(cfg, 0, kind="interleave")
    results = pipe.process_block(raw.view(np.uint8))
    assert len(results) == 2
    c0 = sum(c for _, c in results[0]["counts"])
    c1 = sum(c for _, c in results[1]["counts"])
    assert c1 > 0, results
    assert c0 == 0, results
