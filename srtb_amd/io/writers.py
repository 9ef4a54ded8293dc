"""Detection-product writers: .bin / .npy / .tim + sigproc filterbank header.

File naming and formats mirror the reference (write_signal_pipe.hpp:150-280):
  ${prefix}${counter}.bin        raw baseband bytes of the block
  ${prefix}${counter}.${i}.npy   complex64 waterfall, shape [n_channels, len]
                                 (i = first free index, for multiple pols)
  ${prefix}${counter}.${L}.tim   float32 time series at boxcar length L
The polarization-coincidence window logic of write_signal_pipe.hpp:81-140 is
reimplemented in SignalWriteScheduler.
"""

from __future__ import annotations

import os
import struct
from collections import deque
from dataclasses import dataclass, field

import numpy as np


def write_baseband_bin(prefix: str, counter: int, raw: np.ndarray) -> str:
    path = f"{prefix}{counter}.bin"
    with open(path, "wb") as f:
        f.write(np.ascontiguousarray(raw).tobytes())
        f.flush()
        os.fdatasync(f.fileno())  # reference fdatasyncs baseband dumps
    return path

def spectrum_npy_path(prefix: str, counter: int) -> str:
    i = 0
    while True:
        path = f"{prefix}{counter}.{i}.npy"
        if not os.path.exists(path):
            return path
        i += 1


def write_spectrum_npy(prefix: str, counter: int, waterfall: np.ndarray) -> str:
    """Write the waterfall as ${prefix}${counter}.${i}.npy with the first
    FREE index i — claimed with O_EXCL so concurrent writers (two pols of
    one block on the async pool) get distinct indices instead of both
    scanning to .0.npy."""
    i = 0
    while True:
        path = f"{prefix}{counter}.{i}.npy"
        try:
            with open(path, "xb") as f:
                np.save(f, np.asarray(waterfall, dtype=np.complex64))
            return path
        except FileExistsError:
            i += 1


def write_time_series_tim(prefix: str, counter: int, boxcar_length: int,
                          series: np.ndarray) -> str:
    path = f"{prefix}{counter}.{boxcar_length}.tim"
    with open(path, "wb") as f:
        f.write(np.asarray(series, dtype=np.float32).tobytes())
    return path


@dataclass
class BlockProducts:
    """Everything dumpable for one processed block of one data stream."""
    counter: int                    # udp packet counter or timestamp
    timestamp: int                  # ns-scale timestamp (for coincidence)
    raw: np.ndarray | None = None   # packed baseband bytes
    waterfall: np.ndarray | None = None  # [S][L] complex64
    time_series: list[tuple[int, np.ndarray]] = field(default_factory=list)
    # [(boxcar_length, series), ...]; empty = no detection


class SignalWriteScheduler:
    """Polarization-coincidence writer (reference write_signal_pipe.hpp:77-150).

    A block is written if it has detections, or (in real-time mode) if its
    timestamp falls within ±0.45 block of a recent positive from another
    stream.  Negative blocks are held back in a bounded queue until their
    coincidence window has safely passed.
    """

    def __init__(self, prefix: str, block_samples: int, sample_rate: float,
                 real_time: bool = True, max_pending: int = 8,
                 async_writes: bool = False):
        self.prefix = prefix
        self.real_time = real_time
        self.overlap_window_ns = 0.45 * 1e9 * block_samples / sample_rate
        self.recent_positive: deque[int] = deque()
        self.pending_negative: deque[BlockProducts] = deque()
        self.max_pending = max_pending
        self.written: list[str] = []
        # product writes off the pipeline thread (reference posts them to
        # asio thread_pools, write_signal_pipe.hpp:55-57); call close() (or
        # use as a context manager) to flush
        self._pool = None
        if async_writes:
            from concurrent.futures import ThreadPoolExecutor
            self._pool = ThreadPoolExecutor(max_workers=2,
                                            thread_name_prefix="srtb-writer")
            self._futures = []

    def _overlaps_positive(self, ts: int) -> bool:
        return any(abs(ts - t) < self.overlap_window_ns
                   for t in self.recent_positive)

    def _write(self, blk: BlockProducts) -> None:
        if self._pool is not None:
            # prune completed futures so day-long runs don't accumulate them
            if len(self._futures) > 64:
                self._futures = [f for f in self._futures if not f.done()]
            self._futures.append(self._pool.submit(self._write_sync, blk))
            return
        self._write_sync(blk)

    def _write_sync(self, blk: BlockProducts) -> None:
        if blk.raw is not None:
            self.written.append(
                write_baseband_bin(self.prefix, blk.counter, blk.raw))
        if blk.waterfall is not None:
            self.written.append(
                write_spectrum_npy(self.prefix, blk.counter, blk.waterfall))
        for L, series in blk.time_series:
            self.written.append(
                write_time_series_tim(self.prefix, blk.counter, L, series))

    def push(self, blk: BlockProducts) -> None:
        has_signal = len(blk.time_series) > 0
        # expire outdated positives (reference keeps 5 windows)
        while (self.real_time and self.recent_positive and
               blk.timestamp - self.recent_positive[0] >
               5 * self.overlap_window_ns):
            self.recent_positive.popleft()

        if has_signal:
            self.recent_positive.append(blk.timestamp)
            self._write(blk)
        elif self.real_time and self._overlaps_positive(blk.timestamp):
            self._write(blk)
        elif self.real_time:
            self.pending_negative.append(blk)
            while len(self.pending_negative) > self.max_pending:
                self.pending_negative.popleft()

        # re-check one held-back negative against updated positives
        if self.real_time and self.pending_negative:
            cand = self.pending_negative[0]
            if self._overlaps_positive(cand.timestamp):
                self.pending_negative.popleft()
                self._write(cand)
            elif (self.recent_positive and
                  cand.timestamp + self.overlap_window_ns <
                  self.recent_positive[-1] - 5 * self.overlap_window_ns):
                self.pending_negative.popleft()  # can never match anymore

    def close(self) -> None:
        """Flush and join the async write pool (no-op in sync mode)."""
        if self._pool is not None:
            for f in self._futures:
                f.result()
            self._pool.shutdown(wait=True)
            self._pool = None

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False


# ---------------------------------------------------------------------------
# sigproc filterbank header (reference io/sigproc_filterbank.hpp:30-73)
# ---------------------------------------------------------------------------


def _send(parts: list[bytes], value) -> None:
    if isinstance(value, str):
        b = value.encode()
        parts.append(struct.pack("<i", len(b)) + b)
    elif isinstance(value, int):
        parts.append(struct.pack("<i", value))
    elif isinstance(value, float):
        parts.append(struct.pack("<d", value))
    else:
        raise TypeError(type(value))


def to_sigproc_dms(x: float) -> float:
    """Convert degrees to sigproc ddmmss.s packed representation."""
    sign = -1.0 if x < 0 else 1.0
    xa = abs(x)
    d = int(xa)
    m = int((xa - d) * 60)
    s = ((xa - d) * 60 - m) * 60
    return sign * (d * 10000 + m * 100 + s)


def filterbank_header(*, telescope_id: int = 0, machine_id: int = 0,
                      data_type: int = 1, fch1: float, foff: float,
                      nchans: int, tsamp: float, tstart: float = 0.0,
                      nbits: int = 32, nifs: int = 1,
                      source_name: str = "srtb", src_raj: float = 0.0,
                      src_dej: float = 0.0) -> bytes:
    """Serialize a minimal sigproc filterbank header."""
    parts: list[bytes] = []
    _send(parts, "HEADER_START")
    for key, val in [
        ("telescope_id", telescope_id), ("machine_id", machine_id),
        ("data_type", data_type), ("fch1", float(fch1)),
        ("foff", float(foff)), ("nchans", nchans),
        ("tsamp", float(tsamp)), ("tstart", float(tstart)),
        ("nbits", nbits), ("nifs", nifs),
        ("src_raj", float(src_raj)), ("src_dej", float(src_dej)),
    ]:
        _send(parts, key)
        _send(parts, val)
    _send(parts, "source_name")
    _send(parts, source_name)
    _send(parts, "HEADER_END")
    return b"".join(parts)


def write_filterbank(path: str, header: bytes, data: np.ndarray) -> None:
    """Write a sigproc .fil file: header + [time][chan] float32 intensities."""
    with open(path, "wb") as f:
        f.write(header)
        f.write(np.asarray(data, dtype=np.float32).tobytes())
