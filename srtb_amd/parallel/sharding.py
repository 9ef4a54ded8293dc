"""Multi-GPU stream sharding over RCCL/xGMI (torch.distributed).

The reference is single-process/single-GPU; the MI355X-native
scale-out keeps each baseband stream's whole pipeline resident on one GPU
(xGMI traffic stays control-plane): streams (UDP endpoints / polarizations /
beams / DM trials) are sharded round-robin across ranks, config is broadcast
from rank 0, and small detection statistics are all-reduced per block.
Backend: "nccl" (= RCCL on ROCm) on GPU, "gloo" for CPU tests.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist

from ..config import Config


def init_distributed(backend: str | None = None,
                     timeout_s: float = 600.0) -> tuple[int, int, int]:
    """Initialize from torchrun env vars; returns (rank, world, local_rank).

    No-op (0, 1, 0) when WORLD_SIZE is absent or 1.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1, 0
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(
            backend=backend,
            timeout=datetime.timedelta(seconds=timeout_s))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return rank, world, local_rank


def shard_streams(n_streams: int, world_size: int, rank: int) -> list[int]:
    """Round-robin assignment of stream ids to this rank."""
    return [s for s in range(n_streams) if s % world_size == rank]


def broadcast_config(cfg: Config | None, src: int = 0) -> Config:
    """Broadcast the parsed config from rank ``src`` to all ranks."""
    if not (dist.is_available() and dist.is_initialized()):
        assert cfg is not None
        return cfg
    obj = [cfg]
    dist.broadcast_object_list(obj, src=src)
    assert obj[0] is not None
    return obj[0]


@dataclass
class DetectionStats:
    blocks: int = 0
    detections: int = 0       # blocks with >= 1 positive series
    zapped_channels: int = 0  # accumulated zero_count
    signal_counts: int = 0    # accumulated above-threshold sample counts


class DetectionAggregator:
    """All-reduce per-block detection statistics across ranks."""

    def __init__(self, device: torch.device | str = "cpu"):
        self.device = torch.device(device)
        self.local = DetectionStats()

    def update(self, zero_count: int, counts: list[tuple[int, int]]) -> None:
        self.local.blocks += 1
        total = sum(c for _, c in counts)
        self.local.signal_counts += total
        self.local.zapped_channels += zero_count
        if total > 0:
            self.local.detections += 1

    def reduce(self) -> DetectionStats:
        """Global stats over all ranks (identity when not distributed)."""
        if not (dist.is_available() and dist.is_initialized()):
            return self.local
        # the nccl(=RCCL) backend reduces GPU tensors only; gloo wants CPU
        device = self.device
        if dist.get_backend() == "nccl" and device.type != "cuda":
            device = torch.device("cuda")
        t = torch.tensor(
            [self.local.blocks, self.local.detections,
             self.local.zapped_channels, self.local.signal_counts],
            dtype=torch.int64, device=device)
        dist.all_reduce(t)
        v = t.tolist()
        return DetectionStats(blocks=v[0], detections=v[1],
                              zapped_channels=v[2], signal_counts=v[3])
