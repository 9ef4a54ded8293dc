"""File-replay input with dedispersion-overlap seek-back.

Reference read_file_pipe (pipeline/read_file_pipe.hpp:58-126): each block
re-reads the last ``nsamps_reserved`` samples of the previous one so the
dedispersion edge region can be discarded downstream; an initial byte offset
skips recorded headers.
"""

from __future__ import annotations

import os
from typing import Iterator

import numpy as np


class FileBlockReader:
    """Yield fixed-size packed-baseband blocks from a recorded file."""

    def __init__(self, path: str, block_samples: int, nbits: int,
                 nsamps_reserved: int = 0, offset_bytes: int = 0):
        self.path = path
        self.block_samples = block_samples
        self.bits = abs(nbits)
        if (block_samples * self.bits) % 8 != 0:
            raise ValueError("block not byte-aligned")
        self.block_bytes = block_samples * self.bits // 8
        if (nsamps_reserved * self.bits) % 8 != 0:
            # round the overlap down to a whole byte (keeps alignment)
            nsamps_reserved -= nsamps_reserved % (8 // min(self.bits, 8))
        self.reserved_bytes = nsamps_reserved * self.bits // 8
        if self.reserved_bytes >= self.block_bytes:
            raise ValueError("overlap >= block")
        self.offset_bytes = offset_bytes
        self.file_size = os.path.getsize(path)

    def __iter__(self) -> Iterator[tuple[int, np.ndarray]]:
        """Yields (timestamp_sample_index, block_bytes_array)."""
        step = self.block_bytes - self.reserved_bytes
        pos = self.offset_bytes
        counter = 0
        with open(self.path, "rb") as f:
            while pos + self.block_bytes <= self.file_size:
                f.seek(pos)
                raw = np.frombuffer(f.read(self.block_bytes), dtype=np.uint8)
                sample_index = (pos - self.offset_bytes) * 8 // self.bits
                yield sample_index, raw
                pos += step
                counter += 1

    def n_blocks(self) -> int:
        step = self.block_bytes - self.reserved_bytes
        avail = self.file_size - self.offset_bytes
        if avail < self.block_bytes:
            return 0
        return 1 + (avail - self.block_bytes) // step
