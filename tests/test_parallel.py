"""Multi-process (gloo, world_size=2) tests of the parallel layer — run on CPU."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from srtb_amd.parallel.sharding import shard_streams


def test_shard_streams():
    assert shard_streams(8, 2, 0) == [0, 2, 4, 6]
    assert shard_streams(8, 2, 1) == [1, 3, 5, 7]
    assert shard_streams(3, 8, 5) == []
    assert shard_streams(3, 1, 0) == [0, 1, 2]


def _dist_worker(rank, world, port, fn_name):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["WORLD_SIZE"] = str(world)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    from srtb_amd.parallel.sharding import (DetectionAggregator,
                                            broadcast_config,
                                            init_distributed)
    from srtb_amd.config import Config
    import torch.distributed as dist

    r, w, _ = init_distributed(backend="gloo")
    assert (r, w) == (rank, world)
    try:
        if fn_name == "broadcast":
            cfg = None
            if rank == 0:
                cfg = Config()
                cfg.dm = 478.8
                cfg.baseband_input_count = 2**20
            out = broadcast_config(cfg if rank == 0 else None)
            assert out.dm == 478.8
            assert out.baseband_input_count == 2**20
        elif fn_name == "aggregate":
            agg = DetectionAggregator()
            # rank 0 sees 2 blocks (1 detection), rank 1 sees 3 (0 detections)
            if rank == 0:
                agg.update(5, [(1, 3), (2, 0)])
                agg.update(0, [(1, 0)])
            else:
                for _ in range(3):
                    agg.update(1, [(1, 0)])
            stats = agg.reduce()
            assert stats.blocks == 5
            assert stats.detections == 1
            assert stats.signal_counts == 3
            assert stats.zapped_channels == 8
    finally:
        dist.destroy_process_group()


@pytest.mark.parametrize("fn", ["broadcast", "aggregate"])
def test_distributed_gloo_world2(fn):
    port = 29712 + hash(fn) % 500
    mp.spawn(_dist_worker, args=(2, port, fn), nprocs=2, join=True)
