#!/usr/bin/env python3
"""GPU parity soak (not part of the default suite): random parameterized
batches through decode / encode / rollup / merge, each compared bit-exactly
against the oracle. Run on an MI355X:  python tests/soak_gpu.py [seconds]
"""
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

import torch  # noqa: E402
import oracle  # noqa: E402
from m3_amd import engine  # noqa: E402
from m3_amd.engine import pack_streams  # noqa: E402

START = 1427162462 * 10**9


def random_batch(rng):
    nseries = int(rng.integers(16, 600))
    npts = int(rng.integers(1, 800))
    kind = rng.integers(0, 5, nseries)
    ts = np.empty((nseries, npts), np.int64)
    vals = np.empty((nseries, npts), np.float64)
    for i in range(nseries):
        step = rng.integers(1, 1200, npts)
        if rng.random() < 0.3:
            step = np.full(npts, 10)
        ts[i] = START + np.cumsum(step) * 10**9
        k = kind[i]
        if k == 0:
            vals[i] = np.cumsum(rng.integers(-10**5, 10**5, npts)).astype(float)
        elif k == 1:
            vals[i] = rng.integers(0, 10**7, npts) + rng.integers(0, 10**6, npts) * 1e-6
        elif k == 2:
            vals[i] = np.round(rng.random(npts) * 10, 1)
        elif k == 3:
            bits = rng.integers(0, 2**52, npts, dtype=np.uint64) | np.uint64(0x3FF0000000000000)
            vals[i] = bits.view(np.float64)
        else:  # repeats + mode flaps
            base = np.repeat(rng.random(max(1, npts // 5)) * 100, 5)[:npts]
            vals[i] = np.resize(base, npts)
    return ts, vals


def soak(budget_s=120):
    t0 = time.time()
    trial = 0
    rng = np.random.default_rng(int(os.environ.get("SOAK_SEED", "12345")))
    while time.time() - t0 < budget_s:
        trial += 1
        intopt = bool(rng.integers(0, 2))
        ts, vals = random_batch(rng)
        nseries, npts = ts.shape
        counts = np.full(nseries, npts, np.uint32)
        # oracle encode -> GPU decode
        o_rows, o_lens = oracle.encode_batch(ts, vals, counts, int_optimized=intopt)
        streams = [bytes(o_rows[i, :o_lens[i]]) for i in range(nseries)]
        blob, offsets, lens = pack_streams(streams)
        d_blob = torch.from_numpy(blob).to("cuda:0")
        d_off = torch.from_numpy(offsets.astype(np.int64)).to("cuda:0")
        d_lens = torch.from_numpy(lens.astype(np.int32)).to("cuda:0")
        out_ts = torch.zeros((nseries, npts), dtype=torch.int64, device="cuda:0")
        out_vals = torch.zeros((nseries, npts), dtype=torch.float64, device="cuda:0")
        out_counts = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
        out_errs = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
        perm = torch.argsort(d_lens).to(torch.int32) if rng.random() < 0.5 else None
        engine.decode_batch_dev(d_blob, d_off, d_lens, out_ts, out_vals,
                                out_counts, out_errs, int_optimized=intopt,
                                d_perm=perm)
        torch.cuda.synchronize()
        assert int(out_errs.abs().sum().item()) == 0, f"trial {trial} decode errs"
        o_ts, o_vals, o_counts = oracle.decode_batch(
            blob, offsets, int_optimized=intopt, stride=npts + 4)
        assert np.array_equal(out_counts.cpu().numpy(), o_counts.astype(np.int32)), trial
        g_ts, g_vals = out_ts.cpu().numpy(), out_vals.cpu().numpy()
        assert np.array_equal(g_ts, o_ts[:, :npts]), f"trial {trial} ts"
        assert np.array_equal(g_vals.view(np.uint64),
                              o_vals[:, :npts].view(np.uint64)), f"trial {trial} vals"
        # GPU encode of the canonical decode -> byte-exact vs oracle encode
        d_ts = torch.from_numpy(o_ts[:, :npts].copy()).to("cuda:0")
        d_vals = torch.from_numpy(o_vals[:, :npts].copy()).to("cuda:0")
        d_counts = torch.from_numpy(counts.astype(np.int32)).to("cuda:0")
        out_stride = (24 * npts + 32 + 15) & ~15
        d_out = torch.zeros((nseries, out_stride), dtype=torch.uint8, device="cuda:0")
        d_el = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
        d_ee = torch.empty(nseries, dtype=torch.int32, device="cuda:0")
        engine.encode_batch_dev(d_ts, d_vals, d_counts, d_out, d_el, d_ee,
                                int_optimized=intopt)
        torch.cuda.synchronize()
        assert int(d_ee.abs().sum().item()) == 0, trial
        o2_rows, o2_lens = oracle.encode_batch(o_ts[:, :npts], o_vals[:, :npts],
                                               counts, int_optimized=intopt)
        g_lens = d_el.cpu().numpy()
        assert np.array_equal(g_lens.astype(np.uint32), o2_lens), trial
        g_rows = d_out.cpu().numpy()
        for i in range(nseries):
            assert bytes(g_rows[i, :g_lens[i]]) == bytes(o2_rows[i, :o2_lens[i]]), \
                (trial, i)
        # randomized rollup parity (all three dispatch tiers incl. the
        # compressed-CKMS deep-bucket kernel) every few trials
        if trial % 3 == 0:
            metric = int(rng.integers(0, 3))
            window_s = int(rng.choice([60, 600, 1200, 3600]))
            window = window_s * 10**9
            start = (START // window) * window
            rn = int(rng.integers(8, 80))
            cad = int(rng.choice([1, 2, 5, 10, 60]))
            depth = window_s // cad
            rnp = min(int(rng.integers(1, 4)) * depth, 6000)
            rts = start + np.arange(rnp, dtype=np.int64) * cad * 10**9
            rts = np.broadcast_to(rts, (rn, rnp)).copy()
            rvals = np.round(rng.random((rn, rnp)) * 1e4, 3)
            if metric == 0:
                rvals = np.trunc(rvals)
            rcounts = np.full(rn, rnp, np.uint32)
            aggs = (["median", "p95", "p99", "p9999", "min", "max", "count",
                     "sum", "stdev"] if metric == 2 else
                    ["min", "max", "mean", "count", "sum", "sumsq", "stdev"])
            nbuckets = (rnp * cad + window_s - 1) // window_s
            o_out, o_wts = oracle.rollup_batch(rts, rvals, rcounts, metric,
                                               window, nbuckets, aggs)
            streams_r = [oracle.encode_series(rts[i], rvals[i],
                                              start_ns=int(rts[i, 0]))
                         for i in range(rn)]
            rblob, roff, rlens = pack_streams(streams_r)
            d_rblob = torch.from_numpy(rblob).to("cuda:0")
            d_roff = torch.from_numpy(roff.astype(np.int64)).to("cuda:0")
            d_rlens = torch.from_numpy(rlens.astype(np.int32)).to("cuda:0")
            r_out = torch.empty((rn, nbuckets, len(aggs)),
                                dtype=torch.float64, device="cuda:0")
            r_wts = torch.empty((rn, nbuckets), dtype=torch.int64,
                                device="cuda:0")
            r_errs = torch.empty(rn, dtype=torch.int32, device="cuda:0")
            engine.rollup_batch_dev(d_rblob, d_roff, d_rlens, metric, window,
                                    nbuckets, aggs, r_out, r_wts, r_errs)
            torch.cuda.synchronize()
            assert np.all(r_errs.cpu().numpy() == 0), (trial, "rollup errs")
            assert np.array_equal(r_out.cpu().numpy().view(np.uint64),
                                  o_out.view(np.uint64)), (trial, "rollup",
                                                           metric, depth)
        if trial % 10 == 0:
            print(f"  soak trial {trial} ok ({time.time()-t0:.0f}s)", flush=True)
    print(f"SOAK PASSED: {trial} random batches bit-exact", flush=True)


if __name__ == "__main__":
    soak(int(sys.argv[1]) if len(sys.argv) > 1 else 120)
