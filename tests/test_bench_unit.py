"""Host-side unit tests for bench.py's sharding/config logic (the N>1 code
path the driver launches via torch.distributed.run — SURVEY.md §8e)."""
import importlib.util
import os
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
spec = importlib.util.spec_from_file_location("bench", os.path.join(REPO, "bench.py"))
bench = importlib.util.module_from_spec(spec)
spec.loader.exec_module(bench)


def test_shard_range_partitions_exactly():
    for n in (1, 7, 64, 1_000_000, 8_000_001):
        for world in (1, 2, 3, 8):
            got = [bench.shard_range(n, world, r) for r in range(world)]
            # contiguous, disjoint, covering
            assert got[0][0] == 0 and got[-1][1] == n
            for (a, b), (c, d) in zip(got, got[1:]):
                assert b == c and a <= b
            sizes = [hi - lo for lo, hi in got]
            # balanced within 1
            assert max(sizes) - min(sizes) <= 1


def test_shard_range_matches_reference_shard_model():
    """Contiguous equal ranges — the bench owns placement, mirroring the
    reference's shard-hash distribution (aggregator/sharding/hash.go:89)
    minus the hashing."""
    lo, hi = bench.shard_range(10, 4, 1)
    assert (lo, hi) == (3, 6)  # 3,3,2,2 with extras to low ranks
