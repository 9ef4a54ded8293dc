"""Tests of the live waterfall GUI server (reference gui/ Qt equivalent)."""

import glob
import json
import struct
import urllib.request

import numpy as np

from srtb_amd.gui import WaterfallServer, encode_bmp


def get(url: str) -> bytes:
    with urllib.request.urlopen(url, timeout=10) as r:
        return r.read()


def test_encode_bmp_layout():
    # 2x3 ARGB32: check header fields and bottom-up BGRA pixel bytes
    argb = np.array([[0xFF112233, 0xFF445566, 0xFF778899],
                     [0xFFAABBCC, 0xFFDDEEFF, 0xFF000000]], dtype=np.uint32)
    bmp = encode_bmp(argb)
    assert bmp[:2] == b"BM"
    size, _, _, off = struct.unpack_from("<IHHI", bmp, 2)
    assert size == len(bmp) and off == 54
    hsz, w, h, planes, bpp = struct.unpack_from("<IiiHH", bmp, 14)
    assert (hsz, w, h, planes, bpp) == (40, 3, 2, 1, 32)
    # bottom row first; ARGB32 LE = B,G,R,A in memory
    assert bmp[54:58] == bytes([0xCC, 0xBB, 0xAA, 0xFF])
    assert bmp[54 + 12:54 + 16] == bytes([0x33, 0x22, 0x11, 0xFF])


def test_server_serves_page_status_and_frames():
    with WaterfallServer(host="127.0.0.1", port=0) as gui:
        base = f"http://127.0.0.1:{gui.port}"
        page = get(base + "/")
        assert b"live waterfall" in page
        # no frames yet
        st = json.loads(get(base + "/status.json"))
        assert st["streams"] == []
        # push two streams (the reference opens one window per stream)
        f0 = np.full((8, 16), 0xFF102030, dtype=np.uint32)
        f1 = np.full((8, 16), 0xFF405060, dtype=np.uint32)
        gui.push_frame(0, f0)
        gui.push_frame(1, f1)
        gui.update_status(blocks=3, written=1)
        st = json.loads(get(base + "/status.json"))
        assert st["streams"] == [0, 1] and st["blocks"] == 3
        bmp = get(base + "/frame0.bmp")
        assert bmp == encode_bmp(f0)
        bmp1 = get(base + "/frame1.bmp")
        assert bmp1 == encode_bmp(f1)
        # live spectrum line view (reference spectrum.qml equivalent)
        gui.push_spectrum(0, np.linspace(0.0, 2.0, 4096))
        sj = json.loads(get(base + "/spectrum0.json"))
        assert len(sj["values"]) == 1024  # downsampled
        assert abs(sj["values"][0] - 0.000733) < 1e-3
        assert sj["values"][-1] > 1.9
        # latest-frame-wins (drop-under-load semantics)
        f0b = np.full((8, 16), 0xFF010203, dtype=np.uint32)
        gui.push_frame(0, f0b)
        assert get(base + "/frame0.bmp") == encode_bmp(f0b)
        # unknown stream → 404
        try:
            get(base + "/frame9.bmp")
            assert False, "expected 404"
        except urllib.error.HTTPError as e:
            assert e.code == 404


def test_main_gui_enable_live_frames(tmp_path):
    """File replay with gui_enable=1 serves live frames during the run
    (offscreen GUI smoke test; CPU path)."""
    import threading
    import urllib.error
    from tests.test_main_app import make_recording
    from srtb_amd.main import main

    cfg, rec = make_recording(tmp_path, n_blocks=2)
    cfg_file = tmp_path / "gui.cfg"
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
baseband_output_file_prefix = {tmp_path}/gui_
gui_enable = 1
gui_pixmap_width = 64
gui_pixmap_height = 32
""")
    captured = {}

    # grab a frame while the run is live: poll from a side thread
    def poller():
        import time
        base = "http://127.0.0.1:18265"
        for _ in range(600):
            try:
                st = json.loads(get(base + "/status.json"))
                if st.get("streams"):
                    captured["bmp"] = get(base + "/frame0.bmp")
                    captured["spec"] = get(base + "/spectrum0.json")
                    captured["status"] = st
                    return
            except (urllib.error.URLError, OSError):
                pass
            time.sleep(0.02)

    t = threading.Thread(target=poller)
    t.start()
    rc = main(["--config_file_name", str(cfg_file), "--device", "cpu",
               "--gui-port", "18265", "--gui-linger", "4"])
    t.join()
    assert rc == 0
    assert "bmp" in captured, "no live frame observed during the run"
    spec = json.loads(captured["spec"])
    assert len(spec["values"]) >= 32  # live spectrum line served too
    bmp = captured["bmp"]
    assert bmp[:2] == b"BM"
    _, w, h = struct.unpack_from("<Iii", bmp, 14)
    assert (w, h) == (64, 32)
