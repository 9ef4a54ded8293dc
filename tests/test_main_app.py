"""End-to-end test of the main application (file replay, CPU path)."""

import glob
import os

import numpy as np
import pytest

from srtb_amd.config import Config
from srtb_amd.main import main, write_ppm
from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse


def make_recording(tmp_path, n_blocks=2):
    cfg = Config()
    cfg.baseband_input_count = 1 << 17
    cfg.spectrum_channel_count = 1 << 6
    cfg.baseband_input_bits = -8
    cfg.baseband_freq_low = 1400.0
    cfg.baseband_bandwidth = 64.0
    cfg.baseband_sample_rate = 128e6
    cfg.dm = 40.0
    cfg.baseband_reserve_sample = False
    rng = np.random.default_rng(7)
    blocks = []
    for b in range(n_blocks):
        if b == 1:
            t = 0.5 * cfg.baseband_input_count / cfg.baseband_sample_rate
            raw = synthesize_dispersed_pulse(cfg, t, pulse_amp=40.0,
                                             noise_sigma=2.0, rng=rng)
        else:
            raw = np.clip(np.round(rng.normal(0, 2, cfg.baseband_input_count)),
                          -128, 127).astype(np.int8).view(np.uint8)
        blocks.append(raw)
    path = tmp_path / "recording.bin"
    np.concatenate(blocks).tofile(path)
    return cfg, str(path)


def test_main_file_replay_detects_and_writes(tmp_path):
    cfg, rec = make_recording(tmp_path)
    cfg_file = tmp_path / "test.cfg"
    cfg_file.write_text(f"""
baseband_input_count = 2 ** 17
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
baseband_output_file_prefix = {tmp_path}/out_
""")
    rc = main(["--config_file_name", str(cfg_file), "--device", "cpu"])
    assert rc == 0
    bins = glob.glob(str(tmp_path / "out_*.bin"))
    npys = glob.glob(str(tmp_path / "out_*.npy"))
    tims = glob.glob(str(tmp_path / "out_*.tim"))
    assert bins and npys and tims, "detection products missing"
    # the pulse block starts at sample 2^17 → counter in file names
    assert any("131072" in b for b in bins)
    wf = np.load(npys[0])
    assert wf.shape == (2**6, 2**17 // 2 // 2**6)


def test_main_quiet_recording_writes_nothing(tmp_path):
    cfg, rec = make_recording(tmp_path, n_blocks=1)  # block 0 = pure noise
    cfg_file = tmp_path / "q.cfg"
    cfg_file.write_text(f"""
baseband_input_count = 2 ** 17
spectrum_channel_count = 2 ** 6
baseband_input_bits = -8
baseband_freq_low = 1400
baseband_bandwidth = 64
baseband_sample_rate = 128 * 1e6
dm = 40.0
baseband_reserve_sample = 0
mitigate_rfi_average_method_threshold = 1e30
mitigate_rfi_spectral_kurtosis_threshold = 1e30
signal_detect_signal_noise_threshold = 10
input_file_path = {rec}
baseband_output_file_prefix = {tmp_path}/q_
""")
    rc = main(["--config_file_name", str(cfg_file), "--device", "cpu"])
    assert rc == 0
    assert not glob.glob(str(tmp_path / "q_*.bin"))


def make_2pol_recording(tmp_path, n_blocks=2):
    """naocpsr_snap1-style '1 1 2 2' int8 interleave with a dispersed pulse
    in BOTH polarizations of block 1."""
    cfg = Config()
    cfg.baseband_input_count = 1 << 16
    cfg.spectrum_channel_count = 1 << 6
    cfg.baseband_input_bits = -8
    cfg.baseband_freq_low = 1400.0
    cfg.baseband_bandwidth = 64.0
    cfg.baseband_sample_rate = 128e6
    cfg.dm = 40.0
    cfg.baseband_reserve_sample = False
    rng = np.random.default_rng(11)
    n = cfg.baseband_input_count
    blocks = []
    for b in range(n_blocks):
        pols = []
        for _pol in range(2):
            if b == 1:
                t = 0.5 * n / cfg.baseband_sample_rate
                raw = synthesize_dispersed_pulse(cfg, t, pulse_amp=40.0,
                                                 noise_sigma=2.0, rng=rng)
            else:
                raw = np.clip(np.round(rng.normal(0, 2, n)),
                              -128, 127).astype(np.int8).view(np.uint8)
            pols.append(raw.view(np.int8))
        # "1 1 2 2": 2 samples of pol0 then 2 of pol1, repeating
        inter = np.empty((n // 2, 4), dtype=np.int8)
        inter[:, 0:2] = pols[0].reshape(-1, 2)
        inter[:, 2:4] = pols[1].reshape(-1, 2)
        blocks.append(inter.reshape(-1).view(np.uint8))
    path = tmp_path / "rec2pol.bin"
    np.concatenate(blocks).tofile(path)
    return cfg, str(path)


def test_main_2pol_fanout_detects_both_pols(tmp_path):
    """A 2-pol naocpsr_snap1 recording fans out into two per-pol pipelines;
    the dispersed pulse is detected in BOTH pols and coincident products are
    written (reference unpack_pipe.hpp:146-390 + write_signal_pipe.hpp:81-140)."""
    cfg, rec = make_2pol_recording(tmp_path)
    cfg_file = tmp_path / "p2.cfg"
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
baseband_output_file_prefix = {tmp_path}/p2_
""")
    rc = main(["--config_file_name", str(cfg_file), "--device", "cpu"])
    assert rc == 0
    npys = sorted(glob.glob(str(tmp_path / "p2_*.npy")))
    # both polarizations of the pulse block wrote a waterfall:
    # ${prefix}${counter}.0.npy and .1.npy
    # the block counter is the RAW file sample index: block 1 of a 2-pol
    # interleaved file starts at raw sample 2 * 2^16
    counter = 2 << 16
    assert str(tmp_path / f"p2_{counter}.0.npy") in npys
    assert str(tmp_path / f"p2_{counter}.1.npy") in npys
    tims = glob.glob(str(tmp_path / "p2_*.tim"))
    assert tims, "no time-series product written"
    bins = glob.glob(str(tmp_path / "p2_*.bin"))
    assert any(str(counter) in b for b in bins)


def test_main_waterfall_ppm(tmp_path):
    cfg, rec = make_recording(tmp_path, n_blocks=1)
    cfg_file = tmp_path / "w.cfg"
    cfg_file.write_text(f"""
baseband_input_count = 2 ** 17
spectrum_channel_count = 2 ** 6
baseband_input_bits = -8
baseband_sample_rate = 128 * 1e6
baseband_freq_low = 1400
baseband_bandwidth = 64
dm = 0
baseband_reserve_sample = 0
input_file_path = {rec}
baseband_output_file_prefix = {tmp_path}/w_
gui_pixmap_width = 64
gui_pixmap_height = 32
""")
    rc = main(["--config_file_name", str(cfg_file), "--device", "cpu",
               "--waterfall-ppm", "1"])
    assert rc == 0
    ppms = glob.glob(str(tmp_path / "w_waterfall_*.ppm"))
    assert ppms
    with open(ppms[0], "rb") as f:
        assert f.readline().strip() == b"P6"
        assert f.readline().strip() == b"64 32"


def test_write_ppm_roundtrip(tmp_path):
    img = np.full((2, 3), 0xFF123456, dtype=np.uint32)
    p = str(tmp_path / "x.ppm")
    write_ppm(p, img)
    data = open(p, "rb").read()
    assert data.endswith(bytes([0x12, 0x34, 0x56]) * 6)


def test_main_baseband_write_all(tmp_path):
    cfg, rec = make_recording(tmp_path, n_blocks=2)
    cfg_file = tmp_path / "wa.cfg"
    cfg_file.write_text(f"""
baseband_input_count = 2 ** 17
spectrum_channel_count = 2 ** 6
baseband_input_bits = -8
baseband_sample_rate = 128 * 1e6
baseband_freq_low = 1400
baseband_bandwidth = 64
dm = 0
baseband_reserve_sample = 0
baseband_write_all = 1
signal_detect_signal_noise_threshold = 100
input_file_path = {rec}
baseband_output_file_prefix = {tmp_path}/wa_
""")
    rc = main(["--config_file_name", str(cfg_file), "--device", "cpu"])
    assert rc == 0
    data = np.fromfile(tmp_path / "wa_all_r0.bin", dtype=np.uint8)
    orig = np.fromfile(rec, dtype=np.uint8)
    np.testing.assert_array_equal(data, orig)


@pytest.mark.timeout(60)
def test_main_udp_two_endpoints(tmp_path):
    """One rank serving two UDP endpoints (reference N input pipes)."""
    import socket
    import struct
    import threading
    import time as _time
    from srtb_amd.main import main as srtb_main

    ports = [29931, 29932]
    payload = 4096
    block_samples = payload * 2

    def sender():
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        _time.sleep(0.8)
        for c in range(5):  # enough for 2 blocks per endpoint
            for p in ports:
                pkt = struct.pack("<Q", c) + bytes([c]) * payload
                s.sendto(pkt, ("127.0.0.1", p))
            _time.sleep(0.01)
        s.close()

    t = threading.Thread(target=sender)
    t.start()
    rc = srtb_main([
        "--baseband_format_type", "fastmb_roach2",
        "--baseband_input_count", str(block_samples),
        "--baseband_input_bits", "8",
        "--spectrum_channel_count", "16",
        "--baseband_reserve_sample", "0",
        "--udp_receiver_address", "127.0.0.1, 127.0.0.1",
        "--udp_receiver_port", f"{ports[0]}, {ports[1]}",
        "--baseband_output_file_prefix", str(tmp_path) + "/u_",
        "--device", "cpu", "--max-blocks", "4"])
    t.join()
    assert rc == 0


def test_main_with_shipped_j1644_config(tmp_path):
    """The shipped configs/srtb_config_1644-4559.cfg loads through the real
    path (BASELINE config 1 plumbing) with size overridden for CPU speed."""
    import os
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    rng = np.random.default_rng(3)
    n = 1 << 18
    raw = rng.integers(0, 256, n * 2 // 8, dtype=np.uint8)  # 2-bit packed
    rec = tmp_path / "j.bin"
    raw.tofile(rec)
    rc = main(["--config_file_name",
               os.path.join(root, "configs", "srtb_config_1644-4559.cfg"),
               "--baseband_input_count", str(n),
               "--spectrum_channel_count", "2 ** 6",
               "--input_file_path", str(rec),
               "--baseband_output_file_prefix", str(tmp_path) + "/j_",
               "--gui_enable", "0",
               "--device", "cpu"])
    assert rc == 0
