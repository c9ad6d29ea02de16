"""Multi-process (world_size=2, gloo) coverage of bench.py's sharding +
reduction path — the series-sharded data distribution of SURVEY.md §8e
(independent series, no data-path collective; only the timing MAX and the
aggregate count cross ranks). Runs on CPU."""
import os

import numpy as np
import pytest

from bench import shard_range


def test_shard_range_partition():
    for nseries in (1, 7, 1000, 1_000_000):
        for world in (1, 2, 4, 8):
            ranges = [shard_range(nseries, world, r) for r in range(world)]
            # contiguous, disjoint, covering
            assert ranges[0][0] == 0
            assert ranges[-1][1] == nseries
            for (a, b), (c, d) in zip(ranges, ranges[1:]):
                assert b == c
            sizes = [b - a for a, b in ranges]
            assert max(sizes) - min(sizes) <= 1  # balanced


def _worker(rank, world, port, out):
    import torch
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    nseries, npts = 1000, 7
    lo, hi = shard_range(nseries, world, rank)
    local_pts = (hi - lo) * npts
    # whole-job aggregate = sum of per-rank points; elapsed = MAX over ranks
    t = torch.tensor([float(local_pts)])
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    e = torch.tensor([0.1 * (rank + 1)])
    dist.all_reduce(e, op=dist.ReduceOp.MAX)
    dist.barrier()
    out.put((rank, float(t.item()), float(e.item())))
    dist.destroy_process_group()


def test_two_rank_aggregate():
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    out = ctx.Queue()
    port = 29781
    ps = [ctx.Process(target=_worker, args=(r, 2, port, out)) for r in range(2)]
    for p in ps:
        p.start()
    results = [out.get(timeout=120) for _ in range(2)]
    for p in ps:
        p.join(timeout=120)
        assert p.exitcode == 0
    for _, total, elapsed in results:
        assert total == 1000 * 7
        assert elapsed == pytest.approx(0.2)
