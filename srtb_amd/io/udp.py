"""UDP baseband ingest: packet providers and block-assembly workers.

Python implementation of the reference's receive path
(io/udp/udp_receiver.hpp:42-272, io/udp/recvmmsg_packet_provider.hpp): a
provider yields raw packets; the block worker places payloads at
(counter - begin) * payload_size inside a block buffer, zero-filling lost
packets and counting the loss rate.  The production-rate native path
(recvmmsg batching, pinned hugepage buffers, core pinning) lives in
csrc/app/udp_receiver.h and the srtb-backend / srtb-baseband-receiver tools; this module is
the protocol logic, unit-testable without sockets, plus a socket provider for
integration tests and moderate-rate use.
"""

from __future__ import annotations

import socket
from dataclasses import dataclass, field

import numpy as np

from .backends import Backend, get_backend


class UdpPacketProvider:
    """Blocking socket provider (reference recvfrom_packet_provider)."""

    def __init__(self, address: str, port: int, rcvbuf: int = 1 << 26):
        self.sock = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        self.sock.setsockopt(socket.SOL_SOCKET, socket.SO_RCVBUF, rcvbuf)
        self.sock.bind((address, port))

    def receive(self, max_size: int = 1 << 16) -> bytes:
        return self.sock.recv(max_size)

    def close(self):
        self.sock.close()


@dataclass
class LossStats:
    received: int = 0
    lost: int = 0
    out_of_order: int = 0
    wrong_size: int = 0
    duplicate: int = 0
    invalid: int = 0

    @property
    def loss_rate(self) -> float:
        total = self.received + self.lost
        return self.lost / total if total else 0.0


class BlockAssembler:
    """Assemble fixed-size baseband blocks from counter-stamped packets
    (reference udp_receive_block_worker, io/udp/udp_receiver.hpp:180-272).

    Packet counter c carries payload for byte range
    [(c - begin) * payload, ...); gaps are zero-filled, late/duplicate
    packets are dropped, wrong-size packets are skipped.
    """

    def __init__(self, backend: type[Backend], block_bytes: int):
        # 'simple' (reference backend_registry.hpp:36-39) is a headerless
        # linear sample stream: sequential append, no counter
        self.headerless = backend.packet_payload_size == 0
        self.backend = backend
        if self.headerless:
            self.payload = 0
            self.packets_per_block = 0
            self.fill_bytes = 0
            self.stream_offset = 0
            self._pending = b""
        else:
            assert backend.packet_payload_size > backend.packet_header_size > 0, \
                "BlockAssembler needs a counter-stamped backend format"
            self.payload = (backend.packet_payload_size -
                            backend.packet_header_size)
            if block_bytes % self.payload != 0:
                raise ValueError(
                    f"block_bytes {block_bytes} not a multiple of payload "
                    f"{self.payload}")
            self.packets_per_block = block_bytes // self.payload
        self.block_bytes = block_bytes
        self.begin_counter: int | None = None
        self.buf = np.zeros(block_bytes, dtype=np.uint8)
        self.filled = np.zeros(self.packets_per_block, dtype=bool)
        self.stats = LossStats()
        self.first_timestamp = 0
        # counter/timestamp identifying the most recently COMPLETED block
        # (reference block_first_counter — set when push() returns a block)
        self.last_block_counter = 0
        self.last_block_timestamp = 0

    def _reset(self, begin: int):
        self.begin_counter = begin
        self.buf[:] = 0
        self.filled[:] = False

    def push(self, packet: bytes) -> np.ndarray | None:
        """Feed one packet; returns a completed block or None.

        A block completes when a packet at/after the end arrives; missing
        packets stay zero (counted as lost).  When a block is returned,
        `last_block_counter`/`last_block_timestamp` identify THAT block
        (its begin counter — reference block_first_counter semantics).
        """
        if self.headerless:
            return self._push_headerless(packet)
        if len(packet) != self.backend.packet_payload_size:
            self.stats.wrong_size += 1
            return None
        if not self.backend.packet_valid(packet):
            self.stats.invalid += 1
            return None  # slot stays zero, counted as lost on block close
        counter, ts = self.backend.parse_packet(packet)
        if self.begin_counter is None:
            self._reset(counter)
            self.first_timestamp = ts
        idx = counter - self.begin_counter
        if idx < 0:
            self.stats.out_of_order += 1
            return None
        if idx >= self.packets_per_block:
            # complete current block (zero-fill the tail as lost); stamp it
            # with ITS begin counter, not the triggering packet's
            self.last_block_counter = self.begin_counter
            self.last_block_timestamp = self.first_timestamp
            out = self.finish()
            # advance begin by whole blocks so this packet lands in the new
            # block; fully-lost intermediate blocks are accounted as lost
            skip_blocks = idx // self.packets_per_block
            self.stats.lost += (skip_blocks - 1) * self.packets_per_block
            self._reset(self.begin_counter + skip_blocks * self.packets_per_block)
            self.first_timestamp = ts
            res = self.push(packet)
            assert res is None
            return out
        off = idx * self.payload
        if self.filled[idx]:
            self.stats.duplicate += 1
            return None
        self.buf[off:off + self.payload] = np.frombuffer(
            packet, dtype=np.uint8)[self.backend.packet_header_size:]
        self.filled[idx] = True
        self.stats.received += 1
        return None

    def _push_headerless(self, packet: bytes) -> np.ndarray | None:
        take = min(len(packet), self.block_bytes - self.fill_bytes)
        self.buf[self.fill_bytes:self.fill_bytes + take] = np.frombuffer(
            packet[:take], dtype=np.uint8)
        self.fill_bytes += take
        self.stats.received += 1
        if self.fill_bytes >= self.block_bytes:
            self.last_block_counter = self.stream_offset
            self.last_block_timestamp = self.stream_offset
            self.stream_offset += self.block_bytes
            out = self.buf.copy()
            self.buf[:] = 0
            self.fill_bytes = 0
            rem = packet[take:]
            if rem:
                r = self._push_headerless(rem)
                assert r is None
                self.stats.received -= 1  # same packet, counted once
            return out
        return None

    def finish(self) -> np.ndarray:
        """Close out the current block (zero-filled gaps counted as lost)."""
        self.stats.lost += int((~self.filled).sum())
        out = self.buf.copy()
        return out


def run_receiver(provider, assembler: BlockAssembler, on_block,
                 stop_flag) -> None:
    """Receive loop: provider → assembler → on_block(block, block_counter).

    The counter passed to on_block is the completed block's OWN begin
    counter (reference block_first_counter), not the stream's first one.
    """
    while not stop_flag():
        pkt = provider.receive()
        if not pkt:
            continue
        blk = assembler.push(pkt)
        if blk is not None:
            on_block(blk, assembler.last_block_timestamp)
