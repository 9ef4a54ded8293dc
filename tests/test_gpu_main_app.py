"""GPU tests of the Python application runner and benchmark scripts."""

import glob
import json
import os
import subprocess
import sys

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_main_gpu_file_replay(tmp_path):
    """python -m srtb_amd.main with --device cuda detects and writes."""
    from srtb_amd.config import Config
    from srtb_amd.main import main
    from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse

    cfg = Config()
    cfg.baseband_input_count = 1 << 18
    cfg.spectrum_channel_count = 1 << 6
    cfg.baseband_input_bits = -8
    cfg.baseband_freq_low = 1400.0
    cfg.baseband_bandwidth = 64.0
    cfg.baseband_sample_rate = 128e6
    cfg.dm = 60.0
    t = 0.4 * cfg.baseband_input_count / cfg.baseband_sample_rate
    raw = synthesize_dispersed_pulse(cfg, t, pulse_amp=40.0, noise_sigma=2.0)
    rec = tmp_path / "rec.bin"
    raw.tofile(rec)
    cfg_file = tmp_path / "c.cfg"
    cfg_file.write_text(f"""
baseband_input_count = 2 ** 18
spectrum_channel_count = 2 ** 6
baseband_input_bits = -8
baseband_freq_low = 1400
baseband_bandwidth = 64
baseband_sample_rate = 128 * 1e6
dm = 60.0
baseband_reserve_sample = 0
mitigate_rfi_average_method_threshold = 1e30
mitigate_rfi_spectral_kurtosis_threshold = 1e30
signal_detect_signal_noise_threshold = 6
signal_detect_max_boxcar_length = 16
input_file_path = {rec}
baseband_output_file_prefix = {tmp_path}/g_
""")
    rc = main(["--config_file_name", str(cfg_file), "--waterfall-ppm", "1"])
    assert rc == 0
    assert glob.glob(str(tmp_path / "g_*.bin"))
    assert glob.glob(str(tmp_path / "g_*.npy"))
    assert glob.glob(str(tmp_path / "g_waterfall_*.ppm"))


def test_main_gpu_2pol_fanout(tmp_path):
    """python -m srtb_amd.main on a 2-pol naocpsr_snap1 recording runs the
    GpuMultiPolPipeline (fan-out unpack kernel + per-pol engines) and
    writes per-stream waterfalls."""
    from tests.test_main_app import make_2pol_recording
    from srtb_amd.main import main

    cfg, rec = make_2pol_recording(tmp_path)
    cfg_file = tmp_path / "p2g.cfg"
    cfg_file.write_text(f"""
baseband_format_type = naocpsr_snap1
baseband_input_count = 2 ** 16
spectrum_channel_count = 2 ** 6
baseband_input_bits = -8
baseband_freq_low = 1400
baseband_bandwidth = 64
baseband_sample_rate = 128 * 1e6
dm = 40.0
baseband_reserve_sample = 0
mitigate_rfi_average_method_threshold = 1e30
mitigate_rfi_spectral_kurtosis_threshold = 1e30
signal_detect_signal_noise_threshold = 6
signal_detect_max_boxcar_length = 16
input_file_path = {rec}
baseband_output_file_prefix = {tmp_path}/p2g_
""")
    rc = main(["--config_file_name", str(cfg_file)])
    assert rc == 0
    counter = 2 << 16
    npys = sorted(glob.glob(str(tmp_path / "p2g_*.npy")))
    assert str(tmp_path / f"p2g_{counter}.0.npy") in npys, npys
    assert str(tmp_path / f"p2g_{counter}.1.npy") in npys, npys


def run_script(path, *args, timeout=600):
    out = subprocess.run([sys.executable, path, *args], capture_output=True,
                         text=True, timeout=timeout, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    return json.loads(out.stdout.strip().splitlines()[-1])


def test_dual_pol_bench_single_gpu():
    d = run_script("benchmarks/dual_pol_bench.py", "--steps", "2",
                   "--warmup", "1", "--n", str(1 << 22))
    assert d["value"] > 0
    assert d["config"]["parallelism"].endswith("2 pol")


def test_crab_dm_sweep_finds_injected_dm():
    d = run_script("benchmarks/crab_dm_sweep.py", "--n", str(1 << 22),
                   "--trials", "8", "--inject", "--channels", str(1 << 8))
    assert abs(d["best_dm"] - 56.77) < 1e-6, d
    assert d["value"] > 0
