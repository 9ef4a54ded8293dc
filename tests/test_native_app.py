"""Tests of the native C++ application layer (bin/srtb-backend etc.)."""

import os
import socket
import struct
import subprocess
import threading
import time

import numpy as np
import pytest

from srtb_amd import ref

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BACKEND = os.path.join(ROOT, "bin", "srtb-backend")
RECEIVER = os.path.join(ROOT, "bin", "srtb-baseband-receiver")
CORRELATOR = os.path.join(ROOT, "bin", "srtb-correlator")

needs_bins = pytest.mark.skipif(not os.path.exists(BACKEND),
                                reason="native binaries not built")


@needs_bins
def test_dry_run_parses_reference_config(tmp_path):
    cfg = tmp_path / "c.cfg"
    cfg.write_text("""
baseband_input_count = 2 ** 20
spectrum_channel_count = 2 ** 6
baseband_input_bits = 2
dm = -478.80
baseband_freq_low = 1405 + (64 / 2)
baseband_bandwidth = -64
baseband_sample_rate = 128 * 1e6
mitigate_rfi_freq_list = 1418-1422
baseband_reserve_sample = 0
""")
    out = subprocess.run([BACKEND, "--dry-run", "--config_file_name",
                          str(cfg)], capture_output=True, text=True,
                         timeout=30)
    assert out.returncode == 0, out.stderr
    lines = dict(l.split(" = ", 1) for l in out.stdout.strip().splitlines()
                 if " = " in l)
    assert lines["baseband_input_count"] == "1048576"
    assert lines["baseband_freq_low"] == "1437"
    assert lines["baseband_input_bits"] == "2"
    assert float(lines["dm"]) == -478.8
    assert lines["nsamps_reserved"] == "0"


@needs_bins
def test_dry_run_cli_overrides_and_expressions(tmp_path):
    cfg = tmp_path / "c.cfg"
    cfg.write_text("dm = 100\n")
    out = subprocess.run(
        [BACKEND, "--dry-run", "--config_file_name", str(cfg),
         "--dm", "2 ** 3 + sqrt(16)", "--baseband_input_count=2**18"],
        capture_output=True, text=True, timeout=30)
    assert out.returncode == 0, out.stderr
    assert "dm = 12" in out.stdout  # cmd > cfg-file
    assert "baseband_input_count = 262144" in out.stdout


@needs_bins
def test_nsamps_reserved_matches_python_oracle(tmp_path):
    n, s = 2**25, 2**11
    out = subprocess.run(
        [BACKEND, "--dry-run", "--baseband_input_count", str(n),
         "--spectrum_channel_count", str(s), "--baseband_freq_low", "1437",
         "--baseband_bandwidth", "-64", "--baseband_sample_rate", "128e6",
         "--dm", "-478.8", "--baseband_reserve_sample", "1"],
        capture_output=True, text=True, timeout=30)
    assert out.returncode == 0, out.stderr
    got = int(out.stdout.strip().splitlines()[-1].split(" = ")[1])
    expect = ref.nsamps_reserved(n, s, 1437.0, -64.0, 128e6, -478.8)
    assert got == expect


PIPE_TEST = os.path.join(ROOT, "bin", "srtb-pipe-test")


@pytest.mark.skipif(not os.path.exists(PIPE_TEST),
                    reason="native binaries not built")
def test_native_pipe_framework():
    """Generic pipe framework (csrc/app/pipe.h): 3-stage bounded pipeline,
    composite pipes, loose (drop-under-load) push, stop semantics."""
    out = subprocess.run([PIPE_TEST], capture_output=True, text=True,
                         timeout=60)
    assert out.returncode == 0, (out.stdout, out.stderr)
    assert "PIPE TEST OK" in out.stdout


@needs_bins
def test_native_baseband_receiver_loopback(tmp_path):
    """recvmmsg path over 127.0.0.1 with counter gaps zero-filled."""
    port = 29901
    payload = 4096
    block_samples = payload * 4  # 4 packets per block, 8-bit

    def sender():
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        time.sleep(0.5)
        for c in [0, 1, 3, 4, 5, 6, 7, 8]:  # packet 2 lost
            pkt = struct.pack("<Q", c) + bytes([c + 1]) * payload
            s.sendto(pkt, ("127.0.0.1", port))
            time.sleep(0.005)
        s.close()

    t = threading.Thread(target=sender)
    t.start()
    out = subprocess.run(
        [RECEIVER, "--max-blocks", "2",
         "--baseband_format_type", "fastmb_roach2",
         "--baseband_input_count", str(block_samples),
         "--baseband_input_bits", "8",
         "--udp_receiver_address", "127.0.0.1",
         "--udp_receiver_port", str(port),
         "--baseband_output_file_prefix", str(tmp_path) + "/rec_"],
        capture_output=True, text=True, timeout=30)
    t.join()
    assert out.returncode == 0, out.stderr
    data = np.fromfile(tmp_path / "rec_recording.bin", dtype=np.uint8)
    assert data.size == 2 * block_samples
    # packet 0,1 present; packet 2 zero-filled; packet 3 present
    assert (data[:payload] == 1).all()
    assert (data[2 * payload:3 * payload] == 0).all()
    assert (data[3 * payload:4 * payload] == 4).all()


@needs_bins
def test_native_baseband_receiver_simple_headerless(tmp_path):
    """baseband_format_type=simple is a headerless sequential stream
    (reference backend_registry.hpp:36-39): bytes append in arrival order,
    no counter header is consumed."""
    port = 29903
    block_bytes = 4096 * 2

    def sender():
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        time.sleep(0.5)
        # arbitrary-size datagrams summing to 2 blocks + margin
        data = bytes(range(256)) * 70  # 17920 bytes
        for off in range(0, len(data), 1500):
            s.sendto(data[off:off + 1500], ("127.0.0.1", port))
            time.sleep(0.005)
        s.close()

    t = threading.Thread(target=sender)
    t.start()
    out = subprocess.run(
        [RECEIVER, "--max-blocks", "2",
         "--baseband_format_type", "simple",
         "--baseband_input_count", str(block_bytes),
         "--baseband_input_bits", "8",
         "--udp_receiver_address", "127.0.0.1",
         "--udp_receiver_port", str(port),
         "--baseband_output_file_prefix", str(tmp_path) + "/simp_"],
        capture_output=True, text=True, timeout=30)
    t.join()
    assert out.returncode == 0, out.stderr
    data = np.fromfile(tmp_path / "simp_recording.bin", dtype=np.uint8)
    assert data.size == 2 * block_bytes
    expect = np.frombuffer((bytes(range(256)) * 70)[:2 * block_bytes],
                           dtype=np.uint8)
    np.testing.assert_array_equal(data, expect)


# ---------------- GPU end-to-end runs of the native executables ----------------

@pytest.mark.gpu
def test_srtb_backend_file_replay_detects(tmp_path):
    from srtb_amd.config import Config
    from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse

    cfg = Config()
    cfg.baseband_input_count = 1 << 18
    cfg.spectrum_channel_count = 1 << 6
    cfg.baseband_input_bits = -8
    cfg.baseband_freq_low = 1400.0
    cfg.baseband_bandwidth = 64.0
    cfg.baseband_sample_rate = 128e6
    cfg.dm = 60.0
    rng = np.random.default_rng(0)
    noise = np.clip(np.round(rng.normal(0, 2, cfg.baseband_input_count)),
                    -128, 127).astype(np.int8).view(np.uint8)
    t_pulse = 0.4 * cfg.baseband_input_count / cfg.baseband_sample_rate
    pulse = synthesize_dispersed_pulse(cfg, t_pulse, pulse_amp=40.0,
                                       noise_sigma=2.0)
    rec = tmp_path / "rec.bin"
    np.concatenate([noise, pulse]).tofile(rec)

    out = subprocess.run(
        [BACKEND, "--input_file_path", str(rec),
         "--baseband_input_count", str(cfg.baseband_input_count),
         "--baseband_input_bits", "-8",
         "--spectrum_channel_count", str(cfg.spectrum_channel_count),
         "--baseband_freq_low", "1400", "--baseband_bandwidth", "64",
         "--baseband_sample_rate", "128e6", "--dm", "60.0",
         "--baseband_reserve_sample", "0",
         "--mitigate_rfi_average_method_threshold", "1e30",
         "--mitigate_rfi_spectral_kurtosis_threshold", "1e30",
         "--signal_detect_signal_noise_threshold", "6",
         "--signal_detect_max_boxcar_length", "16",
         "--baseband_output_file_prefix", str(tmp_path) + "/out_"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    assert "blocks=2 detections=1" in out.stdout, (out.stdout, out.stderr)
    import glob
    assert glob.glob(str(tmp_path / "out_*.bin"))
    npys = glob.glob(str(tmp_path / "out_*.npy"))
    assert npys
    wf = np.load(npys[0])
    assert wf.shape == (cfg.spectrum_channel_count,
                        cfg.baseband_input_count // 2 //
                        cfg.spectrum_channel_count)
    assert glob.glob(str(tmp_path / "out_*.tim"))


@pytest.mark.gpu
def test_srtb_backend_2pol_fanout(tmp_path):
    """naocpsr_snap1 file replay: one packed 2-pol stream fans out on the
    GPU into per-pol engines; the dispersed pulse present in both pols is
    detected in both and products for both streams are written."""
    from srtb_amd.config import Config
    from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse

    cfg = Config()
    cfg.baseband_input_count = 1 << 17
    cfg.spectrum_channel_count = 1 << 6
    cfg.baseband_input_bits = -8
    cfg.baseband_freq_low = 1400.0
    cfg.baseband_bandwidth = 64.0
    cfg.baseband_sample_rate = 128e6
    cfg.dm = 60.0
    rng = np.random.default_rng(5)
    n = cfg.baseband_input_count
    blocks = []
    for b in range(2):
        pols = []
        for _pol in range(2):
            if b == 1:
                t = 0.4 * n / cfg.baseband_sample_rate
                raw = synthesize_dispersed_pulse(cfg, t, pulse_amp=40.0,
                                                 noise_sigma=2.0, rng=rng)
            else:
                raw = np.clip(np.round(rng.normal(0, 2, n)), -128,
                              127).astype(np.int8).view(np.uint8)
            pols.append(raw.view(np.int8))
        inter = np.empty((n // 2, 4), dtype=np.int8)  # "1 1 2 2"
        inter[:, 0:2] = pols[0].reshape(-1, 2)
        inter[:, 2:4] = pols[1].reshape(-1, 2)
        blocks.append(inter.reshape(-1).view(np.uint8))
    rec = tmp_path / "rec2.bin"
    np.concatenate(blocks).tofile(rec)

    out = subprocess.run(
        [BACKEND, "--input_file_path", str(rec),
         "--baseband_format_type", "naocpsr_snap1",
         "--baseband_input_count", str(n),
         "--baseband_input_bits", "8",
         "--spectrum_channel_count", str(cfg.spectrum_channel_count),
         "--baseband_freq_low", "1400", "--baseband_bandwidth", "64",
         "--baseband_sample_rate", "128e6", "--dm", "60.0",
         "--baseband_reserve_sample", "0",
         "--mitigate_rfi_average_method_threshold", "1e30",
         "--mitigate_rfi_spectral_kurtosis_threshold", "1e30",
         "--signal_detect_signal_noise_threshold", "6",
         "--signal_detect_max_boxcar_length", "16",
         "--baseband_output_file_prefix", str(tmp_path) + "/p2_"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    assert "blocks=2 detections=1" in out.stdout, (out.stdout, out.stderr)
    import glob
    counter = 2 * n  # pulse block starts at raw sample 2*N
    npys = sorted(glob.glob(str(tmp_path / "p2_*.npy")))
    # both polarizations dumped a waterfall: .0.npy and .1.npy
    assert str(tmp_path / f"p2_{counter}.0.npy") in npys, npys
    assert str(tmp_path / f"p2_{counter}.1.npy") in npys, npys
    wf = np.load(npys[0])
    assert wf.shape == (cfg.spectrum_channel_count,
                        n // 2 // cfg.spectrum_channel_count)
    assert glob.glob(str(tmp_path / "p2_*.bin"))
    assert glob.glob(str(tmp_path / "p2_*.tim"))


@pytest.mark.gpu
@pytest.mark.timeout(120)
def test_srtb_backend_udp_ingest_with_overlap(tmp_path):
    """Native UDP ingest end-to-end on the GPU: counter-stamped packets with
    an injected gap; blocks keep the dedispersion-overlap tail of the
    previous block (UDP overlap reservation) and loss stats are logged."""
    import socket
    import threading
    import time as _time

    n = 1 << 16
    s_chan = 1 << 5
    payload = 4096
    port = 29955

    proc = subprocess.Popen(
        [BACKEND,
         "--baseband_format_type", "fastmb_roach2",
         "--baseband_input_count", str(n),
         "--baseband_input_bits", "8",
         "--spectrum_channel_count", str(s_chan),
         "--baseband_freq_low", "1400", "--baseband_bandwidth", "64",
         "--baseband_sample_rate", "128e6", "--dm", "0.5",
         "--baseband_reserve_sample", "1",
         "--mitigate_rfi_average_method_threshold", "1e30",
         "--mitigate_rfi_spectral_kurtosis_threshold", "1e30",
         "--signal_detect_signal_noise_threshold", "1e30",
         "--udp_receiver_address", "127.0.0.1",
         "--udp_receiver_port", str(port),
         "--baseband_output_file_prefix", str(tmp_path) + "/u_",
         "--max-blocks", "2"],
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)

    def sender():
        # stream packets continuously until the backend exits (engine init
        # on a fresh box can take tens of seconds before the socket binds);
        # drop packet 7 to exercise gap zero-fill + loss accounting
        sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        rng = np.random.default_rng(9)
        pay = rng.integers(0, 256, payload, dtype=np.uint8).tobytes()
        c = 0
        while proc.poll() is None and c < 500000:
            if c != 7:
                sock.sendto(c.to_bytes(8, "little") + pay, ("127.0.0.1", port))
            c += 1
            _time.sleep(0.0005)
        sock.close()

    t = threading.Thread(target=sender)
    t.start()
    try:
        out, _ = proc.communicate(timeout=90)
    except subprocess.TimeoutExpired:
        proc.kill()
        out, _ = proc.communicate(timeout=10)
        t.join()
        raise AssertionError(f"backend did not finish; output so far:\n{out}")
    finally:
        if proc.poll() is None:
            proc.kill()
        t.join()
    assert proc.returncode == 0, out
    assert "blocks=2" in out, out
    assert "loss_rate=" in out, out


@pytest.mark.gpu
def test_srtb_backend_rccl_comm_single_rank(tmp_path):
    """Native RCCL path: SRTB_FORCE_COMM=1 initializes a real (1-rank)
    RCCL communicator, runs the file replay sharded loop and all-reduces
    the end-of-run detection stats over it."""
    from srtb_amd.config import Config
    from srtb_amd.pipeline.cpu import synthesize_dispersed_pulse

    cfg = Config()
    cfg.baseband_input_count = 1 << 17
    cfg.spectrum_channel_count = 1 << 6
    cfg.baseband_input_bits = -8
    cfg.baseband_freq_low = 1400.0
    cfg.baseband_bandwidth = 64.0
    cfg.baseband_sample_rate = 128e6
    cfg.dm = 60.0
    t_pulse = 0.4 * cfg.baseband_input_count / cfg.baseband_sample_rate
    pulse = synthesize_dispersed_pulse(cfg, t_pulse, pulse_amp=40.0,
                                       noise_sigma=2.0)
    rec = tmp_path / "rec.bin"
    pulse.tofile(rec)
    env = dict(os.environ)
    env.update({"SRTB_FORCE_COMM": "1", "WORLD_SIZE": "1", "RANK": "0",
                "LOCAL_RANK": "0",
                "SRTB_RCCL_ID_FILE": str(tmp_path / "rccl_id")})
    out = subprocess.run(
        [BACKEND, "--input_file_path", str(rec),
         "--baseband_input_count", str(cfg.baseband_input_count),
         "--baseband_input_bits", "-8",
         "--spectrum_channel_count", str(cfg.spectrum_channel_count),
         "--baseband_freq_low", "1400", "--baseband_bandwidth", "64",
         "--baseband_sample_rate", "128e6", "--dm", "60.0",
         "--baseband_reserve_sample", "0",
         "--mitigate_rfi_average_method_threshold", "1e30",
         "--mitigate_rfi_spectral_kurtosis_threshold", "1e30",
         "--signal_detect_signal_noise_threshold", "6",
         "--signal_detect_max_boxcar_length", "16",
         "--baseband_output_file_prefix", str(tmp_path) + "/rc_"],
        capture_output=True, text=True, timeout=300, env=env)
    assert out.returncode == 0, out.stderr
    # the all-ranks line comes from the RCCL allreduce
    assert "world=1 blocks=1 detections=1 (all ranks)" in out.stdout, \
        (out.stdout, out.stderr)


@pytest.mark.gpu
def test_srtb_backend_write_all(tmp_path):
    """baseband_write_all=1 replaces the product tail with one rolling
    file holding every block minus the overlap (reference
    write_file_pipe.hpp:41-94)."""
    rng = np.random.default_rng(2)
    n = 1 << 17
    raw = np.clip(np.round(rng.normal(0, 8, 2 * n)), -128,
                  127).astype(np.int8).view(np.uint8)
    rec = tmp_path / "wa.bin"
    raw.tofile(rec)
    out = subprocess.run(
        [BACKEND, "--input_file_path", str(rec),
         "--baseband_input_count", str(n),
         "--baseband_input_bits", "-8",
         "--spectrum_channel_count", "64",
         "--baseband_freq_low", "1400", "--baseband_bandwidth", "64",
         "--baseband_sample_rate", "128e6", "--dm", "0",
         "--baseband_reserve_sample", "0", "--baseband_write_all", "1",
         "--mitigate_rfi_average_method_threshold", "1e30",
         "--mitigate_rfi_spectral_kurtosis_threshold", "1e30",
         "--signal_detect_signal_noise_threshold", "1e30",
         "--baseband_output_file_prefix", str(tmp_path) + "/wa_"],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    data = np.fromfile(tmp_path / "wa_all_r0.bin", dtype=np.uint8)
    np.testing.assert_array_equal(data, raw)
    # the product tail is replaced: no per-detection files
    import glob
    assert not glob.glob(str(tmp_path / "wa_*.npy"))


@pytest.mark.gpu
def test_srtb_correlator_native(tmp_path):
    rng = np.random.default_rng(1)
    n = 1 << 14
    sig = np.clip(np.round(rng.normal(0, 16, n)), -128, 127).astype(np.int8)
    f1, f2 = tmp_path / "a.bin", tmp_path / "b.bin"
    sig.tofile(f1)
    np.roll(sig, 8).tofile(f2)
    out = subprocess.run(
        [CORRELATOR, str(f1), str(f2), str(tmp_path / "c.bin"),
         "--nbits", "-8", "--count", str(n)],
        capture_output=True, text=True, timeout=300)
    assert out.returncode == 0, out.stderr
    corr = np.fromfile(tmp_path / "c.bin", dtype=np.float32)
    assert corr.size == n // 2
    assert int(np.argmax(corr)) == n // 2 - 4  # peak at Nc - lag/2
