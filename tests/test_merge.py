"""Replica-merge oracle vs the reference's own MultiReaderIterator tests
(dbnode/encoding/multi_reader_iterator_test.go:52-251, transcribed) plus
property checks. The default equal-timestamp strategy is IterateLastPushed
(iterators_types.go:41-56)."""
import numpy as np

import oracle

START = 1700000000 * 10**9
S = 10**9


def run_merge(replicas, stride=16):
    """replicas: list of [(ts, val), ...] per replica."""
    r = len(replicas)
    ts = np.zeros((r, 1, stride), np.int64)
    vals = np.zeros((r, 1, stride), np.float64)
    counts = np.zeros((r, 1), np.uint32)
    for i, rep in enumerate(replicas):
        for j, (t, v) in enumerate(rep):
            ts[i, 0, j] = t
            vals[i, 0, j] = v
        counts[i, 0] = len(rep)
    ot, ov, oc, oe = oracle.merge_batch(ts, vals, counts)
    n = int(oc[0])
    return list(zip(ot[0, :n].tolist(), ov[0, :n].tolist())), int(oe[0])


def test_merges_multi():
    """TestMultiReaderIteratorMergesMulti :52-76 (disjoint replicas)."""
    a = [(START + i * S, float(i)) for i in (1, 2, 3)]
    b = [(START + i * S, float(i)) for i in (4, 5, 6)]
    out, err = run_merge([a, b])
    assert err == 0
    assert out == a + b


def test_merges_empty():
    out, err = run_merge([[], []])
    assert err == 0 and out == []


def test_deduplicates_single():
    """:188-205 — duplicate ts within one replica collapses to the first."""
    vals = [(START + 1 * S, 1.0), (START + 2 * S, 2.0), (START + 2 * S, 2.5)]
    out, err = run_merge([vals])
    assert err == 0
    assert out == vals[:2]


def test_deduplicates_multi():
    """:207-228 — three identical replicas collapse to one copy."""
    vals = [(START + i * S, float(i)) for i in (1, 2, 3)]
    out, err = run_merge([vals, vals, vals])
    assert err == 0
    assert out == vals


def test_error_on_out_of_order():
    """:230-251 — a decreasing timestamp errors after emitting the prefix."""
    vals = [(START + 1 * S, 1.0), (START + 3 * S, 3.0), (START + 2 * S, 2.0)]
    out, err = run_merge([vals])
    assert err != 0
    assert out == vals[:2]


def test_last_pushed_wins_on_ties():
    """IterateLastPushed: for ties across replicas the LAST in values order
    wins — replica order initially, perturbed only by exhaustion swaps."""
    a = [(START + 1 * S, 10.0), (START + 2 * S, 20.0)]
    b = [(START + 1 * S, 11.0), (START + 2 * S, 21.0)]
    out, err = run_merge([a, b])
    assert err == 0
    assert out == [(START + 1 * S, 11.0), (START + 2 * S, 21.0)]


def test_interleaved_overlap():
    a = [(START + i * S, 100.0 + i) for i in (1, 3, 5, 7)]
    b = [(START + i * S, 200.0 + i) for i in (2, 3, 6, 7, 8)]
    out, err = run_merge([a, b])
    assert err == 0
    assert out == [(START + 1 * S, 101.0), (START + 2 * S, 202.0),
                   (START + 3 * S, 203.0), (START + 5 * S, 105.0),
                   (START + 6 * S, 206.0), (START + 7 * S, 207.0),
                   (START + 8 * S, 208.0)]


def test_merge_brute_force_random():
    """Random ragged replicas vs a brute-force model of the semantics."""
    rng = np.random.default_rng(5)
    for _ in range(200):
        r = int(rng.integers(1, 5))
        replicas = []
        for _ in range(r):
            n = int(rng.integers(0, 12))
            t = np.sort(rng.choice(np.arange(1, 15), size=n, replace=False)) \
                if n else np.array([], int)
            replicas.append([(START + int(x) * S, float(rng.integers(0, 100)))
                             for x in t])
        out, err = run_merge(replicas)
        assert err == 0
        # brute force: union of ts; winner per the values-order simulation is
        # checked only for increasing ts + correct ts set + value from SOME
        # replica holding that ts (the exact winner is pinned by the explicit
        # cases above)
        all_ts = sorted({t for rep in replicas for t, _ in rep})
        assert [t for t, _ in out] == all_ts
        for t, v in out:
            assert any((t, v) in [(x, y) for x, y in rep] or
                       any(x == t and y == v for x, y in rep)
                       for rep in replicas)
