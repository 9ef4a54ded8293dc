"""Runtime configuration — compatible with the reference's ``srtb_config.cfg``.

Key names, default values and the three-tier precedence (command line > config
file > defaults) mirror the reference (userspace/include/srtb/config.hpp:80-249,
userspace/include/srtb/program_options.hpp:34-309).  Numeric values are kept as
strings and evaluated as arithmetic expressions (``2 ** 30``) exactly like the
reference's exprgrammar path — see :mod:`srtb_amd.utils.expr`.

This module is host-side only (no GPU / torch imports).
"""

from __future__ import annotations

import dataclasses
import shlex
from dataclasses import dataclass, field
from typing import List

from .utils.expr import evaluate, evaluate_int

# key -> type tag used when assigning parsed strings.
# "int" / "real" go through the expression evaluator; "str" is verbatim;
# "bool" accepts 0/1 expressions; "strlist"/"intlist" are comma-separated.
_FIELD_KINDS = {
    "config_file_name": "str",
    "baseband_input_count": "int",
    "baseband_input_bits": "int",
    "baseband_format_type": "str",
    "baseband_freq_low": "real",
    "baseband_bandwidth": "real",
    "baseband_sample_rate": "real",
    "baseband_reserve_sample": "bool",
    "dm": "real",
    "udp_receiver_address": "strlist",
    "udp_receiver_port": "intlist",
    "udp_receiver_cpu_preferred": "intlist",
    "input_file_path": "str",
    "input_file_offset_bytes": "int",
    "baseband_output_file_prefix": "str",
    "baseband_write_all": "bool",
    "fft_fftw_wisdom_path": "str",
    "mitigate_rfi_average_method_threshold": "real",
    "mitigate_rfi_spectral_kurtosis_threshold": "real",
    "mitigate_rfi_freq_list": "str",
    "spectrum_sum_count": "int",
    "spectrum_channel_count": "int",
    "signal_detect_signal_noise_threshold": "real",
    "signal_detect_channel_threshold": "real",
    "signal_detect_max_boxcar_length": "int",
    "thread_query_work_wait_time": "int",
    "gui_enable": "bool",
    "gui_pixmap_width": "int",
    "gui_pixmap_height": "int",
    "log_level": "int",
}


@dataclass
class Config:
    """All runtime knobs, defaults identical to the reference's ``srtb::configs``."""

    config_file_name: str = "srtb_config.cfg"
    baseband_input_count: int = 1 << 28
    baseband_input_bits: int = 8
    baseband_format_type: str = "simple"
    baseband_freq_low: float = 1000.0
    baseband_bandwidth: float = 500.0
    baseband_sample_rate: float = 1000 * 1e6
    baseband_reserve_sample: bool = True
    dm: float = 0.0
    udp_receiver_address: List[str] = field(default_factory=lambda: ["10.0.1.2"])
    udp_receiver_port: List[int] = field(default_factory=lambda: [12004])
    udp_receiver_cpu_preferred: List[int] = field(default_factory=lambda: [0])
    input_file_path: str = ""
    input_file_offset_bytes: int = 0
    baseband_output_file_prefix: str = "srtb_baseband_output_"
    baseband_write_all: bool = False
    fft_fftw_wisdom_path: str = "srtb_fftw_wisdom.txt"
    mitigate_rfi_average_method_threshold: float = 10.0
    mitigate_rfi_spectral_kurtosis_threshold: float = 1.1
    mitigate_rfi_freq_list: str = ""
    spectrum_sum_count: int = 1
    spectrum_channel_count: int = 1 << 15
    signal_detect_signal_noise_threshold: float = 6.0
    signal_detect_channel_threshold: float = 0.9
    signal_detect_max_boxcar_length: int = 1024
    thread_query_work_wait_time: int = 1000
    gui_enable: bool = False
    gui_pixmap_width: int = 1920
    gui_pixmap_height: int = 1080
    log_level: int = 3

    # ---- assignment from strings (expression values) ----

    def assign(self, key: str, raw: str) -> None:
        kind = _FIELD_KINDS.get(key)
        if kind is None:
            raise KeyError(f"unknown config key: {key!r}")
        raw = raw.strip()
        if kind == "str":
            value = raw
        elif kind == "int":
            value = evaluate_int(raw)
        elif kind == "real":
            value = evaluate(raw)
        elif kind == "bool":
            value = bool(evaluate_int(raw))
        elif kind == "strlist":
            value = [s.strip() for s in raw.split(",") if s.strip()]
        elif kind == "intlist":
            value = [evaluate_int(s) for s in raw.split(",") if s.strip()]
        else:  # pragma: no cover
            raise AssertionError(kind)
        setattr(self, key, value)

    # ---- derived quantities ----

    @property
    def baseband_input_bytes(self) -> int:
        """Bytes per input block for this bit width (abs(bits) may be <8)."""
        bits = abs(self.baseband_input_bits)
        return self.baseband_input_count * bits // 8

    @property
    def nsamps_complex(self) -> int:
        return self.baseband_input_count // 2

    @property
    def waterfall_length(self) -> int:
        """Time bins per block in the waterfall (= Nc / spectrum_channel_count)."""
        return max(1, self.nsamps_complex // self.spectrum_channel_count)

    def copy(self) -> "Config":
        return dataclasses.replace(
            self,
            udp_receiver_address=list(self.udp_receiver_address),
            udp_receiver_port=list(self.udp_receiver_port),
            udp_receiver_cpu_preferred=list(self.udp_receiver_cpu_preferred),
        )


def parse_config_file(path: str, cfg: Config | None = None) -> Config:
    """Parse a ``srtb_config.cfg``-style file (``key = value``, ``#`` comments)."""
    cfg = cfg or Config()
    with open(path, "r") as f:
        for lineno, line in enumerate(f, 1):
            # strip comments ('#' starts a comment anywhere, like Boost.PO cfg files)
            line = line.split("#", 1)[0].strip()
            if not line:
                continue
            if "=" not in line:
                raise ValueError(f"{path}:{lineno}: expected 'key = value', got {line!r}")
            key, raw = line.split("=", 1)
            cfg.assign(key.strip(), raw)
    return cfg


def parse_args(argv: List[str], cfg: Config | None = None) -> Config:
    """Parse ``--key value`` / ``--key=value`` command-line options.

    Precedence matches the reference: command line overrides the config file
    (``--config_file_name`` is honoured first), which overrides defaults.
    """
    cfg = cfg or Config()
    # first pass: find config file option
    pairs = []
    it = iter(range(len(argv)))
    i = 0
    while i < len(argv):
        a = argv[i]
        if not a.startswith("--"):
            raise ValueError(f"unexpected positional argument {a!r}")
        a = a[2:]
        if "=" in a:
            key, raw = a.split("=", 1)
        else:
            if i + 1 >= len(argv):
                raise ValueError(f"missing value for --{a}")
            key, raw = a, argv[i + 1]
            i += 1
        pairs.append((key, raw))
        i += 1

    for key, raw in pairs:
        if key == "config_file_name":
            cfg.config_file_name = raw.strip()
            parse_config_file(cfg.config_file_name, cfg)
    for key, raw in pairs:
        if key != "config_file_name":
            cfg.assign(key, raw)
    return cfg


def load(argv: List[str] | None = None, config_file: str | None = None) -> Config:
    cfg = Config()
    if config_file is not None:
        parse_config_file(config_file, cfg)
    if argv:
        parse_args(argv, cfg)
    return cfg
