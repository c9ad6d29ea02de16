"""GPU parity: the HIP kernels vs the oracle (C restatement pinned to the
reference's golden vectors). Bit-exact for decode/encode; exact (bit-equal
accumulation order) for the fused rollup. Runs on an MI355X via gpurun."""
import base64
import json
import os

import numpy as np
import pytest

import oracle

pytestmark = pytest.mark.gpu

GOLDEN = os.path.join(os.path.dirname(__file__), "golden")


@pytest.fixture(scope="module")
def torch():
    import torch as t
    if not t.cuda.is_available():
        pytest.skip("no GPU")
    t.cuda.set_device("cuda:0")
    return t


@pytest.fixture(scope="module")
def engine():
    from m3_amd import engine as e
    return e


def _to_dev(torch, arr, dtype):
    return torch.from_numpy(np.ascontiguousarray(arr).view(dtype)).to("cuda:0")


def gpu_decode(torch, engine, streams, npts_cap, int_optimized=True):
    from m3_amd.engine import pack_streams
    blob, offsets, lens = pack_streams(streams)
    n = len(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out_ts = torch.empty((n, npts_cap), dtype=torch.int64, device="cuda:0")
    out_vals = torch.empty((n, npts_cap), dtype=torch.float64, device="cuda:0")
    out_counts = torch.empty(n, dtype=torch.int32, device="cuda:0")
    out_errs = torch.empty(n, dtype=torch.int32, device="cuda:0")
    engine.decode_batch_dev(d_blob, d_off, d_lens, out_ts, out_vals,
                            out_counts, out_errs, int_optimized=int_optimized)
    torch.cuda.synchronize()
    return (out_ts.cpu().numpy(), out_vals.cpu().numpy(),
            out_counts.cpu().numpy(), out_errs.cpu().numpy())


def test_decode_production_streams(torch, engine):
    """The 10 embedded production streams (encoder_benchmark_test.go:36-47)
    + the regression stream decode bit-exactly vs the oracle."""
    with open(os.path.join(GOLDEN, "production_streams.json")) as f:
        ps = json.load(f)
    streams = [base64.b64decode(b) for b in ps["samples"]] + \
              [base64.b64decode(ps["regression"])]
    g_ts, g_vals, g_counts, g_errs = gpu_decode(torch, engine, streams, 800)
    assert np.all(g_errs == 0)
    for i, s in enumerate(streams):
        dec = oracle.decode_series(s, int_optimized=True)
        n = len(dec["ts"])
        assert g_counts[i] == n, i
        assert np.array_equal(g_ts[i, :n], dec["ts"]), i
        assert np.array_equal(g_vals[i, :n].view(np.uint64),
                              np.asarray(dec["vals"]).view(np.uint64)), i


@pytest.mark.parametrize("intopt", [True, False])
def test_decode_random_roundtrip_vs_oracle(torch, engine, intopt):
    """Oracle-encoded random distributions (all four §8d kinds + unit changes
    + annotations) decode bit-exactly on the GPU."""
    from m3_amd.workload import gen_chunk
    rng = np.random.default_rng(1234)
    nseries, npts = 512, 300
    ts, vals = gen_chunk(0, nseries, npts)
    # perturb cadence for irregular timestamps on half the series
    jitter = rng.integers(0, 5, (nseries // 2, npts)) * 10**9
    ts[: nseries // 2] += np.cumsum(jitter, axis=1)
    streams = []
    for i in range(nseries):
        streams.append(oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]),
                                            int_optimized=intopt))
    g_ts, g_vals, g_counts, g_errs = gpu_decode(torch, engine, streams, npts,
                                                int_optimized=intopt)
    assert np.all(g_errs == 0)
    assert np.all(g_counts == npts)
    assert np.array_equal(g_ts, ts)
    assert np.array_equal(g_vals.view(np.uint64), vals.view(np.uint64))


def test_decode_with_markers(torch, engine):
    """Streams with time-unit changes and annotations (markers mid-stream)."""
    rng = np.random.default_rng(7)
    n, npts = 64, 200
    streams, exp_ts, exp_vals = [], [], []
    for i in range(n):
        ts = 1427162462 * 10**9 + np.cumsum(rng.integers(1, 900, npts)) * 10**9
        vals = np.round(rng.random(npts) * 1e5, 3)
        units = np.full(npts, 1, np.uint8)
        units[0] = 2
        units[npts // 2] = 3
        anns = [b"foo" if j < 3 else (b"" if j != 50 else b"mid-annotation")
                for j in range(npts)]
        streams.append(oracle.encode_series(ts, vals, units=units, annotations=anns,
                                            start_ns=int(ts[0]) - 1))
        exp_ts.append(ts)
        exp_vals.append(vals)
    g_ts, g_vals, g_counts, g_errs = gpu_decode(torch, engine, streams, npts)
    assert np.all(g_errs == 0)
    assert np.all(g_counts == npts)
    assert np.array_equal(g_ts, np.stack(exp_ts))
    assert np.array_equal(np.stack(exp_vals), g_vals)


@pytest.mark.parametrize("intopt", [True, False])
def test_encode_bit_exact_vs_oracle(torch, engine, intopt):
    """GPU encoder output is byte-identical to the oracle encoder (which is
    byte-identical to the reference on its golden vectors)."""
    from m3_amd.workload import gen_chunk
    nseries, npts = 512, 300
    ts, vals = gen_chunk(0, nseries, npts)
    counts = np.full(nseries, npts, np.uint32)
    d_ts = torch.from_numpy(ts).to("cuda:0")
    d_vals = torch.from_numpy(vals).to("cuda:0")
    d_counts = torch.from_numpy(counts.astype(np.int32)).to("cuda:0")
    out_stride = (24 * npts + 32 + 7) & ~7
    d_out = torch.zeros((nseries, out_stride), dtype=torch.uint8, device="cuda:0")
    d_lens = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    d_errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.encode_batch_dev(d_ts, d_vals, d_counts, d_out, d_lens, d_errs,
                            int_optimized=intopt)
    torch.cuda.synchronize()
    errs = d_errs.cpu().numpy()
    assert np.all(errs == 0)
    lens = d_lens.cpu().numpy()
    rows = d_out.cpu().numpy()
    o_rows, o_lens = oracle.encode_batch(ts, vals, counts, int_optimized=intopt)
    assert np.array_equal(lens, o_lens)
    for i in range(nseries):
        assert bytes(rows[i, :lens[i]]) == bytes(o_rows[i, :o_lens[i]]), i


@pytest.mark.parametrize("metric,aggs", [
    ("counter", ["sum", "min", "max", "count", "mean", "sumsq", "stdev"]),
    ("gauge", ["last", "min", "max", "mean", "count", "sum", "sumsq", "stdev"]),
    ("timer", ["sum", "sumsq", "mean", "min", "max", "count", "stdev",
               "median", "p50", "p95", "p99"]),
])
def test_rollup_vs_oracle(torch, engine, metric, aggs):
    """Fused decode->rollup vs oracle full-CKMS rollup of the decoded data.
    Timer aggs are the m3 default timer set (types_options.go:125-143)."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(17)
    nseries, npts = 256, 240
    start = (1427162462 * 10**9 // (60 * 10**9)) * 60 * 10**9
    ts = start + np.arange(npts, dtype=np.int64) * 10 * 10**9
    ts = np.broadcast_to(ts, (nseries, npts)).copy()
    # irregular cadence for some series (gaps -> empty buckets)
    ts[:32] += np.cumsum(rng.integers(0, 3, (32, npts)), axis=1) * 10**9 * 60
    if metric == "counter":
        vals = rng.integers(-10**6, 10**6, (nseries, npts)).astype(np.float64)
    else:
        vals = np.round(rng.random((nseries, npts)) * 1e4, 4)
        if metric == "gauge":
            vals[0, ::7] = np.nan
    counts = np.full(nseries, npts, np.uint32)
    streams = [oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    window = 60 * 10**9
    base = (ts[:, 0] // window) * window
    nbuckets = int(((ts[:, -1] - base) // window).max()) + 1
    # oracle rollup over the decoded points
    mt = dict(counter=oracle.METRIC_COUNTER, gauge=oracle.METRIC_GAUGE,
              timer=oracle.METRIC_TIMER)[metric]
    o_out, o_wts = oracle.rollup_batch(ts, vals, counts, mt, window, nbuckets, aggs)
    # gpu fused rollup straight from the streams
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out = torch.empty((nseries, nbuckets, len(aggs)), dtype=torch.float64,
                      device="cuda:0")
    wts = torch.empty((nseries, nbuckets), dtype=torch.int64, device="cuda:0")
    errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.rollup_batch_dev(d_blob, d_off, d_lens, mt, window, nbuckets, aggs,
                            out, wts, errs)
    torch.cuda.synchronize()
    assert np.all(errs.cpu().numpy() == 0)
    assert np.array_equal(wts.cpu().numpy(), o_wts)
    g = out.cpu().numpy()
    eq = np.array_equal(g.view(np.uint64), o_out.view(np.uint64))
    if not eq:
        bad = np.argwhere(g.view(np.uint64) != o_out.view(np.uint64))
        raise AssertionError(f"{metric}: {len(bad)} mismatches, first {bad[0]}: "
                             f"gpu={g[tuple(bad[0])]} oracle={o_out[tuple(bad[0])]}")


def test_decode_error_reporting(torch, engine):
    """Truncated streams must flag per-series errors, not crash or fall back."""
    good = oracle.encode_series(
        [1427162462 * 10**9 + i * 10**9 for i in range(50)],
        [float(i) for i in range(50)], start_ns=1427162462 * 10**9)
    streams = [good, good[: len(good) // 2], b"\x00" * 8, good]
    g_ts, g_vals, g_counts, g_errs = gpu_decode_no_check(torch, engine, streams, 64)
    assert g_errs[0] == 0 and g_errs[3] == 0
    assert g_counts[0] == 50 and g_counts[3] == 50
    assert g_errs[1] != 0  # truncated mid-stream


def gpu_decode_no_check(torch, engine, streams, npts_cap):
    from m3_amd.engine import pack_streams
    blob, offsets, lens = pack_streams(streams)
    n = len(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out_ts = torch.zeros((n, npts_cap), dtype=torch.int64, device="cuda:0")
    out_vals = torch.zeros((n, npts_cap), dtype=torch.float64, device="cuda:0")
    out_counts = torch.empty(n, dtype=torch.int32, device="cuda:0")
    out_errs = torch.empty(n, dtype=torch.int32, device="cuda:0")
    engine.decode_batch_dev(d_blob, d_off, d_lens, out_ts, out_vals,
                            out_counts, out_errs)
    torch.cuda.synchronize()
    return (out_ts.cpu().numpy(), out_vals.cpu().numpy(),
            out_counts.cpu().numpy(), out_errs.cpu().numpy())


def test_smoke_entry(torch, engine):
    import __graft_entry__
    __graft_entry__.smoke()


def test_decode_perm_schedule_matches(torch, engine):
    """Length-sorted scheduling permutation is schedule-only: outputs are
    identical to the unpermuted decode."""
    from m3_amd.engine import pack_streams
    from m3_amd.workload import gen_chunk
    nseries, npts = 300, 200
    ts, vals = gen_chunk(0, nseries, npts)
    streams = [oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    outs = []
    for perm in (None, torch.argsort(d_lens).to(torch.int32)):
        out_ts = torch.zeros((nseries, npts), dtype=torch.int64, device="cuda:0")
        out_vals = torch.zeros((nseries, npts), dtype=torch.float64, device="cuda:0")
        out_counts = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
        out_errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
        engine.decode_batch_dev(d_blob, d_off, d_lens, out_ts, out_vals,
                                out_counts, out_errs, d_perm=perm)
        torch.cuda.synchronize()
        assert np.all(out_errs.cpu().numpy() == 0)
        outs.append((out_ts.cpu().numpy(), out_vals.cpu().numpy(),
                     out_counts.cpu().numpy()))
    assert np.array_equal(outs[0][0], outs[1][0])
    assert np.array_equal(outs[0][1].view(np.uint64), outs[1][1].view(np.uint64))
    assert np.array_equal(outs[0][2], outs[1][2])
    assert np.array_equal(outs[0][0], ts)


def test_rollup_deep_buckets_retry(torch, engine):
    """Timer buckets deeper than the per-lane staging (16) must transparently
    retry on the wave-per-series kernel — results still exact vs oracle."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(29)
    nseries, npts = 96, 300
    start = (1427162462 * 10**9 // (60 * 10**9)) * 60 * 10**9
    ts = start + np.arange(npts, dtype=np.int64) * 10**9  # 1s cadence: 60/bucket
    ts = np.broadcast_to(ts, (nseries, npts)).copy()
    ts[:16] = start + np.arange(npts, dtype=np.int64) * 10 * 10**9  # some shallow
    vals = np.round(rng.random((nseries, npts)) * 1e4, 4)
    counts = np.full(nseries, npts, np.uint32)
    aggs = ["sum", "min", "max", "median", "p95", "p99", "count"]
    window = 60 * 10**9
    base = (ts[:, 0] // window) * window
    nbuckets = int(((ts[:, -1] - base) // window).max()) + 1
    o_out, o_wts = oracle.rollup_batch(ts, vals, counts, oracle.METRIC_TIMER,
                                       window, nbuckets, aggs)
    streams = [oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out = torch.empty((nseries, nbuckets, len(aggs)), dtype=torch.float64,
                      device="cuda:0")
    wts = torch.empty((nseries, nbuckets), dtype=torch.int64, device="cuda:0")
    errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.rollup_batch_dev(d_blob, d_off, d_lens, engine.METRIC_TIMER,
                            window, nbuckets, aggs, out, wts, errs)
    torch.cuda.synchronize()
    assert np.all(errs.cpu().numpy() == 0)  # retried series cleared
    assert np.array_equal(wts.cpu().numpy(), o_wts)
    g = out.cpu().numpy()
    assert np.array_equal(g.view(np.uint64), o_out.view(np.uint64))


def test_full_pipeline_parity_large(torch, engine):
    """Full-size parity: GPU encode of 32k series x 1440 pts -> compact ->
    GPU decode, compared BIT-EXACTLY against the oracle decoding the same
    blob on the CPU. (Comparing against the generated inputs instead would
    be wrong: convertToIntFloat canonicalizes near-decimal floats — the
    reference-documented lossy conversion, m3tsz.go:72-77.)"""
    from m3_amd import workload
    nseries, npts = 32768, 1440
    d_blob, d_offsets, d_lens, enc_bytes = workload.encode_on_device(
        torch, nseries, npts, chunk=16384, device="cuda:0")
    out_ts = torch.empty((nseries, npts), dtype=torch.int64, device="cuda:0")
    out_vals = torch.empty((nseries, npts), dtype=torch.float64, device="cuda:0")
    out_counts = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    out_errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.decode_batch_dev(d_blob, d_offsets, d_lens, out_ts, out_vals,
                            out_counts, out_errs,
                            d_perm=torch.argsort(d_lens).to(torch.int32))
    torch.cuda.synchronize()
    assert int(out_errs.abs().sum().item()) == 0
    assert bool((out_counts == npts).all().item())
    g_ts = out_ts.cpu().numpy()
    g_vals = out_vals.cpu().numpy()
    assert np.all(np.diff(g_ts, axis=1) > 0)  # strictly monotonic everywhere
    blob = d_blob.cpu().numpy()
    offs = d_offsets.cpu().numpy().astype(np.uint64)
    o_ts, o_vals, o_counts = oracle.decode_batch(blob, offs, stride=npts)
    assert np.array_equal(o_counts, np.full(nseries, npts, np.uint32))
    assert np.array_equal(g_ts, o_ts)
    assert np.array_equal(g_vals.view(np.uint64), o_vals.view(np.uint64))


def test_decode_edge_cases(torch, engine):
    """Edge grammar paths on GPU: us/ns units (64-bit default DoD buckets),
    int-overflow values (sig==64 slow path), huge time jumps, float/int mode
    flapping, repeats — all bit-exact vs the oracle."""
    rng = np.random.default_rng(99)
    streams, exp = [], []
    start = 1427162462 * 10**9
    # us-unit and ns-unit streams (64-bit default bucket)
    for unit in (3, 4):
        ts = start + np.cumsum(rng.integers(1, 10**7, 100))  # ns steps
        if unit == 3:
            ts = (ts // 1000) * 1000  # us-aligned
        vals = rng.random(100) * 1e6
        units = np.full(100, unit, np.uint8)
        streams.append(oracle.encode_series(ts, vals, units=units,
                                            start_ns=start - 1))
        exp.append((ts, None))
    # overflow / giant-diff values (sig 64, float<->int flapping)
    li, ln = float(2**63 - 2), float(-(2**63) + 1)
    vals = np.array([li, 10, ln, 10, ln, li, -12, li, 14.5, li, ln,
                     12.34858499392, li, 0.0, -0.0, 1e13, 1e13 - 2, 5.0])
    ts = start + np.arange(len(vals)) * 10**9
    streams.append(oracle.encode_series(ts, vals, start_ns=start))
    exp.append((ts, None))
    # repeat-heavy + mode flapping
    vals = np.tile(np.array([7.0, 7.0, 7.0, 1.5e300, 1.5e300, 7.0]), 40)
    ts = start + np.arange(len(vals)) * 10**9
    streams.append(oracle.encode_series(ts, vals, start_ns=start))
    exp.append((ts, None))
    # huge forward/backward time jumps (32-bit s-unit default bucket limits)
    ts = start + np.cumsum(rng.integers(-2000, 500000, 200)) * 10**9
    ts = np.maximum.accumulate(ts - ts.min() + start)  # keep positive
    vals = rng.random(200)
    streams.append(oracle.encode_series(ts, vals, start_ns=int(ts[0])))
    exp.append((ts, None))

    g_ts, g_vals, g_counts, g_errs = gpu_decode(torch, engine, streams, 256)
    assert np.all(g_errs == 0)
    for i, s in enumerate(streams):
        dec = oracle.decode_series(s, int_optimized=True)
        n = len(dec["ts"])
        assert g_counts[i] == n, i
        assert np.array_equal(g_ts[i, :n], dec["ts"]), i
        assert np.array_equal(g_vals[i, :n].view(np.uint64),
                              np.asarray(dec["vals"]).view(np.uint64)), i


@pytest.mark.parametrize("metric,agg", [("gauge", "last"), ("timer", "p99"),
                                        ("counter", "sum")])
def test_aggregate_tiles_vs_oracle(torch, engine, metric, agg):
    """AggregateTiles-shaped pipeline (decode -> 1m rollup -> re-encode) is
    byte-identical to oracle rollup + oracle encode of the tile series,
    including series with empty tiles (gaps)."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(31)
    nseries, npts = 128, 240
    start = (1427162462 * 10**9 // (60 * 10**9)) * 60 * 10**9
    ts = start + np.arange(npts, dtype=np.int64) * 10 * 10**9
    ts = np.broadcast_to(ts, (nseries, npts)).copy()
    ts[:32] += np.cumsum(rng.integers(0, 3, (32, npts)), axis=1) * 10**9 * 60
    if metric == "counter":
        vals = rng.integers(-10**6, 10**6, (nseries, npts)).astype(np.float64)
    else:
        vals = np.round(rng.random((nseries, npts)) * 1e4, 3)
    counts = np.full(nseries, npts, np.uint32)
    streams = [oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    window = 60 * 10**9
    base = (ts[:, 0] // window) * window
    nbuckets = int(((ts[:, -1] - base) // window).max()) + 1
    mt = dict(counter=oracle.METRIC_COUNTER, gauge=oracle.METRIC_GAUGE,
              timer=oracle.METRIC_TIMER)[metric]
    # oracle expectation: rollup -> drop empty tiles -> encode per series
    o_out, o_wts = oracle.rollup_batch(ts, vals, counts, mt, window, nbuckets,
                                       [agg, "count"])
    exp_streams = []
    for i in range(nseries):
        sel = o_out[i, :, 1] > 0
        t_i = o_wts[i][sel]
        v_i = o_out[i, :, 0][sel]
        exp_streams.append(
            oracle.encode_series(t_i, v_i, start_ns=int(t_i[0])) if len(t_i)
            else b"")
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    mt_gpu = dict(counter=engine.METRIC_COUNTER, gauge=engine.METRIC_GAUGE,
                  timer=engine.METRIC_TIMER)[metric]
    tile_bytes, tile_lens, tile_counts = engine.aggregate_tiles_dev(
        torch, d_blob, d_off, d_lens, mt_gpu, window, nbuckets, agg)
    tl = tile_lens.cpu().numpy()
    tb = tile_bytes.cpu().numpy()
    for i in range(nseries):
        assert bytes(tb[i, :tl[i]]) == exp_streams[i], i


def test_merge_replicas_vs_oracle(torch, engine):
    """GPU replica merge (k_merge) vs the oracle MultiReaderIterator
    restatement: random ragged overlapping replicas, bit-exact (incl. the
    IterateLastPushed tie winner and the out-of-order error point)."""
    rng = np.random.default_rng(47)
    nseries, stride, R = 512, 64, 3
    START = 1700000000 * 10**9
    ts = np.zeros((R, nseries, stride), np.int64)
    vals = np.zeros((R, nseries, stride), np.float64)
    counts = np.zeros((R, nseries), np.uint32)
    for r in range(R):
        for i in range(nseries):
            n = int(rng.integers(0, stride))
            t = np.sort(rng.choice(np.arange(200), size=n, replace=False))
            ts[r, i, :n] = START + t * 10**9
            vals[r, i, :n] = np.round(rng.random(n) * 100, 2)
            counts[r, i] = n
    o_ts, o_vals, o_counts, o_errs = oracle.merge_batch(ts, vals, counts,
                                                        out_stride=stride * R)
    d_ts = torch.from_numpy(ts.reshape(R * nseries, stride)).to("cuda:0")
    d_vals = torch.from_numpy(vals.reshape(R * nseries, stride)).to("cuda:0")
    d_counts = torch.from_numpy(counts.reshape(-1).astype(np.int32)).to("cuda:0")
    out_ts = torch.zeros((nseries, stride * R), dtype=torch.int64, device="cuda:0")
    out_vals = torch.zeros((nseries, stride * R), dtype=torch.float64, device="cuda:0")
    out_counts = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    out_errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.merge_batch_dev(d_ts, d_vals, d_counts, R, out_ts, out_vals,
                           out_counts, out_errs)
    torch.cuda.synchronize()
    assert np.all(out_errs.cpu().numpy() == 0)
    assert np.array_equal(out_counts.cpu().numpy().astype(np.uint32), o_counts)
    g_ts = out_ts.cpu().numpy()
    g_vals = out_vals.cpu().numpy()
    for i in range(nseries):
        n = int(o_counts[i])
        assert np.array_equal(g_ts[i, :n], o_ts[i, :n]), i
        assert np.array_equal(g_vals[i, :n], o_vals[i, :n]), i


def test_merge_out_of_order_flag(torch, engine):
    """A replica with a decreasing timestamp flags err 100 after emitting
    the prefix — exactly like errOutOfOrderIterator."""
    START = 1700000000 * 10**9
    ts = np.zeros((1, 1, 8), np.int64)
    vals = np.zeros((1, 1, 8), np.float64)
    ts[0, 0, :3] = [START + 1, START + 3, START + 2]
    vals[0, 0, :3] = [1.0, 3.0, 2.0]
    counts = np.array([[3]], np.uint32)
    d_ts = torch.from_numpy(ts.reshape(1, 8)).to("cuda:0")
    d_vals = torch.from_numpy(vals.reshape(1, 8)).to("cuda:0")
    d_counts = torch.from_numpy(counts.reshape(-1).astype(np.int32)).to("cuda:0")
    out_ts = torch.zeros((1, 8), dtype=torch.int64, device="cuda:0")
    out_vals = torch.zeros((1, 8), dtype=torch.float64, device="cuda:0")
    out_counts = torch.empty(1, dtype=torch.int32, device="cuda:0")
    out_errs = torch.empty(1, dtype=torch.int32, device="cuda:0")
    engine.merge_batch_dev(d_ts, d_vals, d_counts, 1, out_ts, out_vals,
                           out_counts, out_errs)
    torch.cuda.synchronize()
    assert int(out_errs[0].item()) == 100
    assert int(out_counts[0].item()) == 2
    assert out_ts[0, :2].cpu().numpy().tolist() == [START + 1, START + 3]


def test_rollup_large_exact_windows(torch, engine):
    """Buckets up to ~300 values stay EXACT (the host-computed no-compression
    cap for the default timer quantile set is ~498): bit-equal to the oracle
    full-CKMS rollup via the wave-kernel retry path."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(53)
    nseries, npts = 64, 900
    window = 600 * 10**9  # 10m windows, 2s cadence -> 300 values/bucket
    start = (1427162462 * 10**9 // window) * window
    ts = start + np.arange(npts, dtype=np.int64) * 2 * 10**9
    ts = np.broadcast_to(ts, (nseries, npts)).copy()
    vals = np.round(rng.random((nseries, npts)) * 1e4, 4)
    counts = np.full(nseries, npts, np.uint32)
    aggs = ["median", "p95", "p99", "count"]
    base = (ts[:, 0] // window) * window
    nbuckets = int(((ts[:, -1] - base) // window).max()) + 1
    o_out, o_wts = oracle.rollup_batch(ts, vals, counts, oracle.METRIC_TIMER,
                                       window, nbuckets, aggs)
    streams = [oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out = torch.empty((nseries, nbuckets, len(aggs)), dtype=torch.float64,
                      device="cuda:0")
    wts = torch.empty((nseries, nbuckets), dtype=torch.int64, device="cuda:0")
    errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.rollup_batch_dev(d_blob, d_off, d_lens, engine.METRIC_TIMER,
                            window, nbuckets, aggs, out, wts, errs)
    torch.cuda.synchronize()
    assert np.all(errs.cpu().numpy() == 0)
    g = out.cpu().numpy()
    assert np.array_equal(g.view(np.uint64), o_out.view(np.uint64))


def test_rollup_deep_buckets_full_ckms(torch, engine):
    """Buckets beyond the exact-order-statistics cap run the real compressed
    CKMS on the GPU (k_rollup_ckms, third retry tier): bit-equal to the
    oracle's quantile/cm/stream.go:77-429 restatement, including compression
    merges, the 1024-value insert cadence, and the reference's stale-zero
    top-quantile regime (un-emitted computed[] stays 0.0 for some
    numValues). Depths straddle the insertAndCompressEvery boundary."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(61)
    window_s = 5040
    window = window_s * 10**9
    cadences = [8, 5, 4, 2, 1]     # depths 630, 1008, 1260, 2520, 5040
    aggs = ["median", "p95", "p99", "p9999", "min", "max", "mean", "count",
            "sum", "stdev"]
    start = (1427162462 * 10**9 // window) * window
    series = []
    for cad in cadences:
        depth = window_s // cad
        npts = 2 * depth           # two full buckets
        t = start + np.arange(npts, dtype=np.int64) * cad * 10**9
        for dist in range(5):
            if dist == 0:
                v = np.round(rng.random(npts) * 1e4, 4)
            elif dist == 1:
                v = np.round(np.sort(rng.random(npts)) * 1e3, 4)
            elif dist == 2:
                v = np.round(np.sort(rng.random(npts))[::-1] * 1e3, 4).copy()
            elif dist == 3:        # heavy ties
                v = np.round(rng.random(npts) * 10, 1)
            else:                  # constant
                v = np.full(npts, 42.5)
            series.append((t, v))
    nseries = len(series)
    width = max(len(t) for t, _ in series)
    ts = np.zeros((nseries, width), np.int64)
    vals = np.zeros((nseries, width), np.float64)
    counts = np.zeros(nseries, np.uint32)
    for i, (t, v) in enumerate(series):
        ts[i, :len(t)] = t
        vals[i, :len(t)] = v
        counts[i] = len(t)
    nbuckets = 2
    o_out, o_wts = oracle.rollup_batch(ts, vals, counts, oracle.METRIC_TIMER,
                                       window, nbuckets, aggs)
    streams = [oracle.encode_series(ts[i, :counts[i]], vals[i, :counts[i]],
                                    start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out = torch.empty((nseries, nbuckets, len(aggs)), dtype=torch.float64,
                      device="cuda:0")
    wts = torch.empty((nseries, nbuckets), dtype=torch.int64, device="cuda:0")
    errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.rollup_batch_dev(d_blob, d_off, d_lens, engine.METRIC_TIMER,
                            window, nbuckets, aggs, out, wts, errs)
    torch.cuda.synchronize()
    assert np.all(errs.cpu().numpy() == 0)
    g = out.cpu().numpy()
    gw = wts.cpu().numpy()
    assert np.array_equal(gw, o_wts)
    for i in range(nseries):
        assert np.array_equal(g[i].view(np.uint64), o_out[i].view(np.uint64)), \
            (i, g[i], o_out[i])


def test_rollup_deep_buckets_mixed_with_shallow(torch, engine):
    """A batch mixing shallow (per-lane tier), medium (wave tier) and deep
    (CKMS tier) series resolves each series on the right tier with every
    result bit-equal to the oracle."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(67)
    window_s = 1200
    window = window_s * 10**9
    start = (1427162462 * 10**9 // window) * window
    aggs = ["p95", "p99", "max", "count", "sum"]
    cadences = [120, 10, 1]        # depths 10 (lane), 120 (wave), 1200 (ckms)
    series = []
    for cad in cadences:
        for _ in range(6):
            npts = 3 * (window_s // cad)
            t = start + np.arange(npts, dtype=np.int64) * cad * 10**9
            v = np.round(rng.random(npts) * 1e3, 3)
            series.append((t, v))
    nseries = len(series)
    width = max(len(t) for t, _ in series)
    ts = np.zeros((nseries, width), np.int64)
    vals = np.zeros((nseries, width), np.float64)
    counts = np.zeros(nseries, np.uint32)
    for i, (t, v) in enumerate(series):
        ts[i, :len(t)] = t
        vals[i, :len(t)] = v
        counts[i] = len(t)
    nbuckets = 3
    o_out, o_wts = oracle.rollup_batch(ts, vals, counts, oracle.METRIC_TIMER,
                                       window, nbuckets, aggs)
    streams = [oracle.encode_series(ts[i, :counts[i]], vals[i, :counts[i]],
                                    start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out = torch.empty((nseries, nbuckets, len(aggs)), dtype=torch.float64,
                      device="cuda:0")
    wts = torch.empty((nseries, nbuckets), dtype=torch.int64, device="cuda:0")
    errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.rollup_batch_dev(d_blob, d_off, d_lens, engine.METRIC_TIMER,
                            window, nbuckets, aggs, out, wts, errs)
    torch.cuda.synchronize()
    assert np.all(errs.cpu().numpy() == 0)
    g = out.cpu().numpy()
    assert np.array_equal(g.view(np.uint64), o_out.view(np.uint64))
    assert np.array_equal(wts.cpu().numpy(), o_wts)


def test_fileset_ingest_end_to_end(torch, engine, tmp_path):
    """§8f row 1 end-to-end: oracle-written fileset volume -> native reader
    (validate digests/checksums, repack) -> HIP batch decode on the GPU ->
    bit-exact vs the original points."""
    from oracle import fileset_writer as fsw
    rng = np.random.default_rng(41)
    BLOCK_START = 1427162400 * 10**9
    series = []
    raw = {}
    for i in range(200):
        npts = int(rng.integers(2, 700))
        ts = BLOCK_START + np.cumsum(rng.integers(1, 30, npts)) * 10**9
        kind = i % 3
        if kind == 0:
            vals = np.round(rng.random(npts) * 100, 2)
        elif kind == 1:
            vals = np.cumsum(rng.integers(-1000, 1000, npts)).astype(float)
        else:
            vals = rng.random(npts) * 1e6
        blob = oracle.encode_series(ts, vals, start_ns=int(ts[0]))
        sid = f"ns.metric.{i:05d}".encode()
        series.append((sid, blob, None))
        raw[sid] = (ts, vals)
    fsw.write_volume(str(tmp_path), BLOCK_START, series)
    ids, d_ts, d_vals, d_counts, d_errs = engine.fileset_ingest_dev(
        torch, str(tmp_path), BLOCK_START)
    torch.cuda.synchronize()
    assert np.all(d_errs.cpu().numpy() == 0)
    g_ts = d_ts.cpu().numpy()
    g_vals = d_vals.cpu().numpy()
    g_counts = d_counts.cpu().numpy()
    assert len(ids) == len(series)
    # parity target is the ORACLE decode of the same packed volume: the
    # reference codec's int optimization canonicalizes some float inputs
    # (convertToIntFloat finds an exact multiplier, e.g. ...8835670002 ->
    # ...883567), so raw inputs are not always bit-identical after a
    # roundtrip — but GPU and oracle must agree exactly.
    from m3_amd.engine import FilesetVolume
    with FilesetVolume(str(tmp_path), BLOCK_START) as v:
        blob, offsets, lens = v.pack()
    o_off = np.concatenate([offsets, [np.uint64(len(blob))]])
    o_ts, o_vals, o_counts = oracle.decode_batch(blob, o_off,
                                                 stride=g_ts.shape[1])
    for i, sid in enumerate(ids):
        ts, vals = raw[sid]
        n = len(ts)
        assert g_counts[i] == n == o_counts[i], sid
        assert np.array_equal(g_ts[i, :n], ts), sid
        assert np.array_equal(g_vals[i, :n].view(np.uint64),
                              o_vals[i, :n].view(np.uint64)), sid
        # timestamps always survive exactly; values match input except
        # where the codec canonicalized (still equal as oracle output)
        assert np.array_equal(o_ts[i, :n], ts), sid


def test_commitlog_bootstrap_end_to_end(torch, engine, tmp_path):
    """Commitlog half of §8f row 1: oracle-written commit log -> native
    reader (chunk validation, metadata registration) -> grouped series ->
    HIP batch encode -> byte-exact vs oracle encode of the same points."""
    from oracle import commitlog_writer as clw
    rng = np.random.default_rng(71)
    START = 1427162462 * 10**9
    nseries, npts = 150, 200
    per = {}
    entries = []
    for i in range(nseries):
        ts = START + np.cumsum(rng.integers(1, 30, npts)) * 10**9
        vals = np.round(rng.random(npts) * 1e4, 3)
        per[i] = (ts, vals)
    for t in range(npts):          # interleaved, like live ingest
        for i in range(nseries):
            entries.append((i, f"boot.{i:04d}".encode(), i % 8,
                            int(per[i][0][t]), float(per[i][1][t]), 4,
                            None, None))
    path = tmp_path / "commitlog-0-0.db"
    clw.write_commitlog(str(path), entries)
    meta, d_bytes, d_lens, d_errs = engine.commitlog_bootstrap_dev(
        torch, str(path))
    torch.cuda.synchronize()
    assert len(meta) == nseries
    assert np.all(d_errs.cpu().numpy() == 0)
    g_lens = d_lens.cpu().numpy()
    g_rows = d_bytes.cpu().numpy()
    ts = np.stack([per[i][0] for i in range(nseries)])
    vals = np.stack([per[i][1] for i in range(nseries)])
    counts = np.full(nseries, npts, np.uint32)
    o_rows, o_lens = oracle.encode_batch(ts, vals, counts, int_optimized=True)
    assert np.array_equal(g_lens.astype(np.uint32), o_lens)
    for i in range(nseries):
        assert meta[i]["id"] == f"boot.{i:04d}".encode()
        assert bytes(g_rows[i, :g_lens[i]]) == bytes(o_rows[i, :o_lens[i]]), i


def test_regather_layout_pass(torch, engine):
    """k_regather physical reorder: dst row i holds src stream perm[i];
    decoding the repacked blob without a scheduling perm must equal the
    oracle decode of the original streams in perm order."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(79)
    nseries, npts = 300, 120
    START = 1427162462 * 10**9
    ts = START + np.cumsum(rng.integers(1, 60, (nseries, npts)), axis=1) * 10**9
    vals = np.round(rng.random((nseries, npts)) * 1e3, 3)
    streams = [oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    perm = torch.randperm(nseries, device="cuda:0").to(torch.int32)
    new_lens = d_lens[perm.long()]
    aligned = ((new_lens.to(torch.int64) + 15) // 16) * 16
    new_off = torch.zeros(nseries + 1, dtype=torch.int64, device="cuda:0")
    torch.cumsum(aligned, 0, out=new_off[1:])
    d_blob2 = torch.zeros(int(new_off[-1].item()), dtype=torch.uint8,
                          device="cuda:0")
    engine.regather_dev(d_blob, d_off, d_lens, perm, new_off[:-1].contiguous(),
                        d_blob2)
    out_ts = torch.zeros((nseries, npts), dtype=torch.int64, device="cuda:0")
    out_vals = torch.zeros((nseries, npts), dtype=torch.float64, device="cuda:0")
    out_counts = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    out_errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.decode_batch_dev(d_blob2, new_off[:-1].contiguous(),
                            new_lens.contiguous().to(torch.int32), out_ts,
                            out_vals, out_counts, out_errs)
    torch.cuda.synchronize()
    assert np.all(out_errs.cpu().numpy() == 0)
    p = perm.cpu().numpy()
    # pack_streams offsets already carry the n+1 end sentinel
    o_ts, o_vals, _ = oracle.decode_batch(blob, offsets, stride=npts)
    g_ts = out_ts.cpu().numpy()
    g_vals = out_vals.cpu().numpy()
    for i in range(nseries):
        s = p[i]
        assert np.array_equal(g_ts[i], o_ts[s, :npts]), i
        assert np.array_equal(g_vals[i].view(np.uint64),
                              o_vals[s, :npts].view(np.uint64)), i


def test_rollup_duplicate_quantile_aggs_gpu(torch, engine):
    """median + p50 share one quantile slot in the plan (sorted unique
    list); GPU bit-equal to oracle for the duplicated agg outputs."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(89)
    nseries, npts = 64, 90
    START = 1427162462 * 10**9
    window = npts * 10**9
    start = (START // window) * window
    ts = start + np.arange(npts, dtype=np.int64) * 10**9
    ts = np.broadcast_to(ts, (nseries, npts)).copy()
    vals = np.round(rng.random((nseries, npts)) * 100, 2)
    counts = np.full(nseries, npts, np.uint32)
    aggs = ["median", "p50", "p95", "count"]
    o_out, _ = oracle.rollup_batch(ts, vals, counts, oracle.METRIC_TIMER,
                                   window, 1, aggs)
    streams = [oracle.encode_series(ts[i], vals[i], start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out = torch.empty((nseries, 1, 4), dtype=torch.float64, device="cuda:0")
    wts = torch.empty((nseries, 1), dtype=torch.int64, device="cuda:0")
    errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.rollup_batch_dev(d_blob, d_off, d_lens, engine.METRIC_TIMER,
                            window, 1, aggs, out, wts, errs)
    torch.cuda.synchronize()
    assert np.all(errs.cpu().numpy() == 0)
    g = out.cpu().numpy()
    assert np.array_equal(g.view(np.uint64), o_out.view(np.uint64))
    assert np.array_equal(g[:, 0, 0], g[:, 0, 1])


def test_rollup_nondefault_ckms_options(torch, engine):
    """Configurable CKMS stream options (eps/insertAndCompressEvery, cm
    options.go:30-32) through the _opts ABI: non-default (eps=0.01,
    every=64) compresses far earlier — exact_cap shrinks and the deep
    CKMS tier engages at shallow depths; bit-equal to the oracle with the
    same options at depths straddling the insert cadence."""
    from m3_amd.engine import pack_streams
    rng = np.random.default_rng(97)
    eps, every = 0.01, 64
    window_s = 1040
    window = window_s * 10**9
    start = (1427162462 * 10**9 // window) * window
    aggs = ["median", "p95", "p99", "min", "max", "count", "sum"]
    series = []
    for cad in (26, 16, 13, 8, 2):   # depths 40, 65, 80, 130, 520
        depth = window_s // cad
        npts = 2 * depth
        t = start + np.arange(npts, dtype=np.int64) * cad * 10**9
        for dist in range(3):
            if dist == 0:
                v = np.round(rng.random(npts) * 1e3, 3)
            elif dist == 1:
                v = np.round(np.sort(rng.random(npts)) * 100, 4)
            else:
                v = np.round(rng.random(npts) * 5, 1)  # ties
            series.append((t, v))
    nseries = len(series)
    width = max(len(t) for t, _ in series)
    ts = np.zeros((nseries, width), np.int64)
    vals = np.zeros((nseries, width), np.float64)
    counts = np.zeros(nseries, np.uint32)
    for i, (t, v) in enumerate(series):
        ts[i, :len(t)] = t
        vals[i, :len(t)] = v
        counts[i] = len(t)
    o_out, o_wts = oracle.rollup_batch(ts, vals, counts, oracle.METRIC_TIMER,
                                       window, 2, aggs, eps=eps, every=every)
    streams = [oracle.encode_series(ts[i, :counts[i]], vals[i, :counts[i]],
                                    start_ns=int(ts[i, 0]))
               for i in range(nseries)]
    blob, offsets, lens = pack_streams(streams)
    d_blob = torch.from_numpy(blob).to("cuda:0")
    d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
    d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
    out = torch.empty((nseries, 2, len(aggs)), dtype=torch.float64,
                      device="cuda:0")
    wts = torch.empty((nseries, 2), dtype=torch.int64, device="cuda:0")
    errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
    engine.rollup_batch_dev(d_blob, d_off, d_lens, engine.METRIC_TIMER,
                            window, 2, aggs, out, wts, errs,
                            eps=eps, every=every)
    torch.cuda.synchronize()
    assert np.all(errs.cpu().numpy() == 0)
    g = out.cpu().numpy()
    for i in range(nseries):
        assert np.array_equal(g[i].view(np.uint64), o_out[i].view(np.uint64)), \
            (i, g[i], o_out[i])
    # invalid options are rejected, not approximated
    import pytest as _pytest
    with _pytest.raises(engine.M3GpuError):
        engine.rollup_batch_dev(d_blob, d_off, d_lens, engine.METRIC_TIMER,
                                window, 2, aggs, out, wts, errs, every=2000)


def test_decode_annotations(torch, engine):
    """Annotation-returning bulk decode (m3gpu_decode_batch_dev_ann) vs the
    oracle: the reference's roundtrip annotation shapes
    (roundtrip_test.go:135-148 — repeated "foo"/"bar" runs, dedupe via
    xxhash) plus long and binary annotations. Events must reproduce the
    sticky PrevAnt view (iterator.go:226-231) byte-for-byte."""
    rng = np.random.default_rng(1234)
    start = 1427162462 * 10**9
    streams, ann_truth = [], []
    cases = []
    # repeated foo/bar runs (reference roundtrip shapes)
    cases.append([b"foo"] * 10 + [b"bar"] * 10 + [b"foo"] * 5)
    # sparse: annotation only at points 0, 7, 8
    sparse = [None] * 25
    sparse[0], sparse[7], sparse[8] = b"first", b"mid", b"mid2"
    cases.append(sparse)
    # none at all
    cases.append([None] * 30)
    # binary + long annotations, changing every point
    cases.append([bytes(rng.integers(0, 256, 40, dtype=np.uint8).tolist())
                  for _ in range(12)])
    for anns in cases:
        n = len(anns)
        ts = start + np.arange(n) * 10**9
        vals = np.round(rng.random(n) * 100, 2)
        streams.append(oracle.encode_series(ts, vals, annotations=anns,
                                            start_ns=start))
        dec = oracle.decode_series(streams[-1], with_annotations=True)
        # oracle gives per-point NEW annotations; carry forward = PrevAnt
        cur, ref = None, []
        for a in dec["annotations"]:
            if a is not None:
                cur = a
            ref.append(cur)
        ann_truth.append((dec["ts"], dec["vals"], ref))

    from m3_amd.engine import (pack_streams, decode_batch_dev_ann,
                               decoded_ann_per_point)
    blob, offsets, lens = pack_streams(streams)
    dev = "cuda:0"
    d_blob = torch.from_numpy(blob).to(dev)
    d_off = torch.from_numpy(offsets[:-1].astype(np.int64)).to(dev)
    d_lens = torch.from_numpy(lens.astype(np.int32)).to(dev)
    ns, stride = len(streams), 64
    out_ts = torch.empty((ns, stride), dtype=torch.int64, device=dev)
    out_vals = torch.empty((ns, stride), dtype=torch.float64, device=dev)
    out_counts = torch.empty(ns, dtype=torch.int32, device=dev)
    out_errs = torch.empty(ns, dtype=torch.int32, device=dev)
    out_ann = torch.zeros((ns, 1024), dtype=torch.uint8, device=dev)
    decode_batch_dev_ann(d_blob, d_off, d_lens, out_ts, out_vals, out_counts,
                         out_errs, out_ann)
    torch.cuda.synchronize()
    assert torch.all(out_errs == 0).item()
    h_ann = out_ann.cpu().numpy()
    for i, (ref_ts, ref_vals, ref_anns) in enumerate(ann_truth):
        n = int(out_counts[i].item())
        assert n == len(ref_ts)
        assert np.array_equal(out_ts[i, :n].cpu().numpy(), ref_ts)
        assert np.array_equal(out_vals[i, :n].cpu().numpy().view(np.uint64),
                              np.asarray(ref_vals).view(np.uint64))
        got = decoded_ann_per_point(h_ann[i], n)
        assert got == ref_anns, f"series {i}"


def test_decode_annotations_region_overflow(torch, engine):
    """A region too small for a series' annotations flags CAPACITY on that
    series only; values still decode."""
    rng = np.random.default_rng(55)
    start = 1427162462 * 10**9
    anns = [bytes([65 + (i % 26)]) * 50 for i in range(20)]  # 1000B of anns
    ts = start + np.arange(20) * 10**9
    vals = rng.random(20)
    big = oracle.encode_series(ts, vals, annotations=anns, start_ns=start)
    small = oracle.encode_series(ts, vals, start_ns=start)
    from m3_amd.engine import pack_streams, decode_batch_dev_ann
    blob, offsets, lens = pack_streams([big, small])
    dev = "cuda:0"
    d_blob = torch.from_numpy(blob).to(dev)
    d_off = torch.from_numpy(offsets[:-1].astype(np.int64)).to(dev)
    d_lens = torch.from_numpy(lens.astype(np.int32)).to(dev)
    out_ts = torch.empty((2, 32), dtype=torch.int64, device=dev)
    out_vals = torch.empty((2, 32), dtype=torch.float64, device=dev)
    out_counts = torch.empty(2, dtype=torch.int32, device=dev)
    out_errs = torch.empty(2, dtype=torch.int32, device=dev)
    out_ann = torch.zeros((2, 128), dtype=torch.uint8, device=dev)  # too small
    decode_batch_dev_ann(d_blob, d_off, d_lens, out_ts, out_vals, out_counts,
                         out_errs, out_ann)
    torch.cuda.synchronize()
    assert int(out_errs[0].item()) == 6  # capacity
    assert int(out_errs[1].item()) == 0
    assert int(out_counts[0].item()) == 20  # values still decoded


def test_decode_annotations_host_form(torch, engine):
    """m3gpu_decode_batch_ann (host-pointer convenience form, the cgo
    surface) produces the same values + annotation events as the _dev
    form."""
    import ctypes
    from ctypes import POINTER as P, c_uint8, c_uint64, c_uint32, c_int64, \
        c_double, c_int32
    rng = np.random.default_rng(77)
    start = 1427162462 * 10**9
    anns = [b"x", None, b"yy", None, b"zzz"] * 4
    ts = start + np.arange(20) * 10**9
    vals = np.round(rng.random(20) * 10, 3)
    stream = oracle.encode_series(ts, vals, annotations=anns, start_ns=start)
    from m3_amd.engine import pack_streams, decoded_ann_per_point, lib
    blob, offsets, lens = pack_streams([stream])
    L = lib()
    L.m3gpu_decode_batch_ann.restype = ctypes.c_int
    L.m3gpu_decode_batch_ann.argtypes = [
        P(c_uint8), c_uint64, P(c_uint64), P(c_uint32), c_uint32,
        ctypes.c_int, c_uint8, P(c_int64), P(c_double), P(c_uint32),
        P(c_int32), c_uint32, P(c_uint8), c_uint32]
    stride, astride = 32, 256
    o_ts = np.zeros((1, stride), np.int64)
    o_vals = np.zeros((1, stride), np.float64)
    o_counts = np.zeros(1, np.uint32)
    o_errs = np.zeros(1, np.int32)
    o_ann = np.zeros((1, astride), np.uint8)
    rc = L.m3gpu_decode_batch_ann(
        blob.ctypes.data_as(P(c_uint8)), blob.nbytes,
        offsets.ctypes.data_as(P(c_uint64)), lens.ctypes.data_as(P(c_uint32)),
        1, 1, 1,
        o_ts.ctypes.data_as(P(c_int64)), o_vals.ctypes.data_as(P(c_double)),
        o_counts.ctypes.data_as(P(c_uint32)), o_errs.ctypes.data_as(P(c_int32)),
        stride, o_ann.ctypes.data_as(P(c_uint8)), astride)
    assert rc == 0 and o_errs[0] == 0 and o_counts[0] == 20
    assert np.array_equal(o_ts[0, :20], ts)
    got = decoded_ann_per_point(o_ann[0], 20)
    cur, ref = None, []
    for a in anns:
        if a is not None:
            cur = a
        ref.append(cur)
    assert got == ref
