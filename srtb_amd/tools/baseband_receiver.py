"""srtb_baseband_receiver equivalent: record UDP baseband straight to disk.

Reference src/baseband_receiver.cpp:37-87: UDP receive → (cast) →
write-file composite pipe, record-only mode (no GPU processing).

Usage:
  python -m srtb_amd.tools.baseband_receiver --config_file_name cfg
      [--max-blocks N]
"""

from __future__ import annotations

import sys

import numpy as np

from ..config import parse_args
from ..io import backends as bk
from ..io.udp import BlockAssembler, UdpPacketProvider, run_receiver


def main(argv=None) -> int:
    argv = list(sys.argv[1:] if argv is None else argv)
    max_blocks = None
    cfg_argv = []
    i = 0
    while i < len(argv):
        a = argv[i]
        if a.startswith("--max-blocks"):
            max_blocks = int(a.split("=", 1)[1] if "=" in a else argv[i + 1])
            i += 1 if "=" in a else 2
        else:
            cfg_argv.append(a)
            i += 1
    cfg = parse_args(cfg_argv)
    backend = bk.get_backend(cfg.baseband_format_type)
    bits = abs(cfg.baseband_input_bits)
    block_bytes = cfg.baseband_input_count * bits // 8 * \
        bk.get_data_stream_count(cfg.baseband_format_type)
    assembler = BlockAssembler(backend, block_bytes)
    provider = UdpPacketProvider(cfg.udp_receiver_address[0],
                                 cfg.udp_receiver_port[0])
    out_path = cfg.baseband_output_file_prefix + "recording.bin"
    f = open(out_path, "ab")
    state = {"n": 0}

    def on_block(blk: np.ndarray, ts: int):
        f.write(blk.tobytes())
        f.flush()
        state["n"] += 1
        print(f"[baseband_receiver] block {state['n']} "
              f"(loss_rate={assembler.stats.loss_rate:.2e})")

    def stop():
        return max_blocks is not None and state["n"] >= max_blocks

    try:
        run_receiver(provider, assembler, on_block, stop)
    finally:
        f.close()
        provider.close()
    return 0


if __name__ == "__main__":
    raise SystemExit(main())
