"""Telescope packet-format registry.

Capability parity with reference userspace/include/srtb/io/backend_registry.hpp:36-181
and io/vdif_header.hpp:28-61: each backend describes its UDP packet layout
(header size, payload size, per-packet polarization interleave) and parses
(counter, timestamp) from a packet.
"""

from __future__ import annotations

import struct
from dataclasses import dataclass


@dataclass(frozen=True)
class VdifHeader:
    """VLBI VDIF data frame header (8 little-endian 32-bit words)."""
    seconds_from_ref_epoch: int
    legacy_mode: int
    invalid_data: int
    data_frame_count_in_second: int
    reference_epoch: int
    data_frame_length: int
    log2_channels: int
    vdif_version: int
    station_id: int
    thread_id: int
    bits_per_sample_minus_1: int
    data_type: int

    @classmethod
    def parse(cls, buf: bytes) -> "VdifHeader":
        w = struct.unpack_from("<8I", buf)
        return cls(
            seconds_from_ref_epoch=w[0] & 0x3FFFFFFF,
            legacy_mode=(w[0] >> 30) & 1,
            invalid_data=(w[0] >> 31) & 1,
            data_frame_count_in_second=w[1] & 0xFFFFFF,
            reference_epoch=(w[1] >> 24) & 0x3F,
            data_frame_length=w[2] & 0xFFFFFF,
            log2_channels=(w[2] >> 24) & 0x1F,
            vdif_version=(w[2] >> 29) & 0x7,
            station_id=w[3] & 0xFFFF,
            thread_id=(w[3] >> 16) & 0x3FF,
            bits_per_sample_minus_1=(w[3] >> 26) & 0x1F,
            data_type=(w[3] >> 31) & 1,
        )


class Backend:
    name = "simple"
    data_stream_count = 1
    packet_header_size = 0
    packet_payload_size = 0  # 0 = any

    @staticmethod
    def parse_packet(buf: bytes) -> tuple[int, int]:
        """Return (counter, timestamp)."""
        return 0, 0

    @staticmethod
    def packet_valid(buf: bytes) -> bool:
        """VDIF-framed formats override this with the invalid-data bit."""
        return True


class Simple(Backend):
    name = "simple"
    data_stream_count = 1


class FastmbRoach2(Backend):
    """ROACH2: uint64 LE counter + 4096 B int8 payload (packet 4104 B)."""
    name = "fastmb_roach2"
    data_stream_count = 1
    packet_header_size = 8
    packet_payload_size = 4104

    @staticmethod
    def parse_packet(buf: bytes) -> tuple[int, int]:
        counter = struct.unpack_from("<Q", buf)[0]
        return counter, counter


class NaocpsrSnap1(FastmbRoach2):
    """SNAP-1: same header; payload is 2-pol '1 1 2 2' int8 interleave."""
    name = "naocpsr_snap1"
    data_stream_count = 2


class GznupsrA1(Backend):
    """ZCU111: 32 B VDIF header + 32 B counter + 8192 B payload (8256 B);
    counter = VDIF words 6,7 as uint64 LE; payload 4-byte words cycling
    over 2 (v2) or 4 (v1) ADC streams."""
    name = "gznupsr_a1"
    data_stream_count = 2
    packet_header_size = 64
    packet_payload_size = 8256

    @staticmethod
    def parse_packet(buf: bytes) -> tuple[int, int]:
        w6, w7 = struct.unpack_from("<II", buf, 24)
        counter = w6 | (w7 << 32)
        return counter, counter

    @staticmethod
    def parse_vdif(buf: bytes) -> VdifHeader:
        return VdifHeader.parse(buf)

    @staticmethod
    def packet_valid(buf: bytes) -> bool:
        # VDIF invalid-data flag: bit 31 of word 0 (io/vdif_header.hpp:28-61)
        w0 = struct.unpack_from("<I", buf)[0]
        return (w0 >> 31) == 0


_BACKENDS = {b.name: b for b in (Simple, FastmbRoach2, NaocpsrSnap1, GznupsrA1)}
_ALIASES = {"naocpsr_roach2": "fastmb_roach2"}


def resolve_alias(name: str) -> str:
    return _ALIASES.get(name, name)


def get_backend(name: str) -> type[Backend]:
    n = resolve_alias(name)
    if n not in _BACKENDS:
        raise ValueError(f"unknown backend {name!r}")
    return _BACKENDS[n]


def get_data_stream_count(name: str) -> int:
    return get_backend(name).data_stream_count
