import textwrap

import pytest

from srtb_amd.config import Config, parse_args, parse_config_file


def test_defaults_match_reference():
    c = Config()
    # defaults from reference userspace/include/srtb/config.hpp:80-249
    assert c.baseband_input_count == 1 << 28
    assert c.baseband_input_bits == 8
    assert c.baseband_format_type == "simple"
    assert c.baseband_freq_low == 1000.0
    assert c.baseband_bandwidth == 500.0
    assert c.baseband_sample_rate == 1e9
    assert c.baseband_reserve_sample is True
    assert c.dm == 0
    assert c.mitigate_rfi_average_method_threshold == 10
    assert c.mitigate_rfi_spectral_kurtosis_threshold == 1.1
    assert c.spectrum_channel_count == 1 << 15
    assert c.signal_detect_signal_noise_threshold == 6
    assert c.signal_detect_channel_threshold == 0.9
    assert c.signal_detect_max_boxcar_length == 1024


def test_parse_j1644_style_config(tmp_path):
    cfg_text = textwrap.dedent("""\
        # example config file
        baseband_input_count = 2 ** 30
        spectrum_channel_count = 2 ** 11
        baseband_output_file_prefix = /dev/shm/
        log_level = 4
        mitigate_rfi_average_method_threshold = 1.5
        mitigate_rfi_spectral_kurtosis_threshold = 1.05
        signal_detect_signal_noise_threshold = 8
        signal_detect_max_boxcar_length = 256
        input_file_path = /tmp/buf3.bin
        baseband_input_bits = 2
        input_file_offset_bytes = 0
        dm = -478.80
        baseband_reserve_sample = 0
        baseband_freq_low = 1405 + (64 / 2)
        baseband_bandwidth = -64
        baseband_sample_rate = 128 * 1e6
        mitigate_rfi_freq_list = 1418-1422
    """)
    p = tmp_path / "srtb_config.cfg"
    p.write_text(cfg_text)
    c = parse_config_file(str(p))
    assert c.baseband_input_count == 2**30
    assert c.spectrum_channel_count == 2**11
    assert c.baseband_input_bits == 2
    assert c.dm == -478.80
    assert c.baseband_reserve_sample is False
    assert c.baseband_freq_low == 1437.0
    assert c.baseband_bandwidth == -64.0
    assert c.baseband_sample_rate == 128e6
    assert c.mitigate_rfi_freq_list == "1418-1422"
    assert c.signal_detect_max_boxcar_length == 256
    assert c.log_level == 4


def test_cli_overrides_config_file(tmp_path):
    p = tmp_path / "c.cfg"
    p.write_text("dm = 100\nbaseband_input_bits = 2\n")
    c = parse_args(["--config_file_name", str(p), "--dm", "200"])
    assert c.dm == 200.0  # cmd > cfg-file > default (reference README.md:146)
    assert c.baseband_input_bits == 2


def test_cli_equals_form_and_lists():
    c = parse_args(["--udp_receiver_port=12004,12005",
                    "--udp_receiver_address=10.0.1.2, 10.0.1.3",
                    "--baseband_input_count=2 ** 20"])
    assert c.udp_receiver_port == [12004, 12005]
    assert c.udp_receiver_address == ["10.0.1.2", "10.0.1.3"]
    assert c.baseband_input_count == 2**20


def test_unknown_key_rejected():
    c = Config()
    with pytest.raises(KeyError):
        c.assign("no_such_key", "1")


def test_derived_quantities():
    c = Config()
    c.assign("baseband_input_count", "2 ** 20")
    c.assign("spectrum_channel_count", "2 ** 8")
    assert c.nsamps_complex == 2**19
    assert c.waterfall_length == 2**11
    c.assign("baseband_input_bits", "2")
    assert c.baseband_input_bytes == 2**20 * 2 // 8
