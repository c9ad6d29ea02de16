#!/usr/bin/env python3
"""bench.py — headline benchmark: batched M3TSZ decode on MI355X.

Measures BASELINE.json's metric ("datapoints/sec M3TSZ decode (1M series x
1440 pts) + HBM GB/s vs peak") on BASELINE.json configs[1] — the largest
single-GPU configuration — with synthetic data (m3_amd/workload.py shapes,
SURVEY.md §8d) encoded on-device by the product encoder. A "step" = one
batched decode of the full resident batch into SoA (ts, val) rows.

Multi-GPU (--gpus N, launched via torch.distributed.run): the series shard
trivially (independent series, zero exchange — SURVEY.md §8e): each rank
owns nseries/N series, weak scaling, no data-path collective; value is the
whole-job aggregate.

cpu_baseline: the oracle (C restatement of the reference decoder; the only
permitted oracle use in this file) timed on host cores over a bounded sample
of the same streams. roofline: HIP-event kernel timing of the dominant
(decode) kernel vs the 8 TB/s HBM3E peak.
"""
import argparse
import json
import os
import sys
import time

import numpy as np

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

HBM_PEAK_GBS = 8000.0  # 8 TB/s spec peak (MI355X_MICROARCH.md)


def shard_range(nseries, world, rank):
    """Contiguous series range owned by `rank` (equal counts; series are
    statically sharded — no exchange, mirroring the reference's shard-hash
    data distribution at aggregator/sharding/hash.go:89)."""
    per = nseries // world
    extra = nseries % world
    lo = rank * per + min(rank, extra)
    hi = lo + per + (1 if rank < extra else 0)
    return lo, hi


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--nseries", type=int, default=None,
                   help="TOTAL series. Default: 1M PER GPU (weak scaling, "
                        "BASELINE config 4: 8M across 8 GPUs); an explicit "
                        "value is a fixed total (strong scaling)")
    p.add_argument("--npts", type=int, default=1440)
    p.add_argument("--chunk", type=int, default=65536)
    p.add_argument("--cpu-sample-series", type=int, default=0,
                   help="series in the cpu_baseline sample (0 = auto-size to ~15s)")
    p.add_argument("--skip-cpu-baseline", action="store_true")
    p.add_argument("--no-sort", action="store_true",
                   help="disable length-sorted wave scheduling")
    p.add_argument("--layout", choices=["index", "sorted"], default=None,
                   help="physical blob layout: 'sorted' repacks streams in "
                        "schedule order before the timed region (free at "
                        "pack time in production; k_regather here). Default: "
                        "sorted for the synthetic mix (+4%%), index for "
                        "--data production (already length-homogeneous; "
                        "sorting only scrambles locality, -9%%)")
    p.add_argument("--data", choices=["synthetic", "production"],
                   default="synthetic",
                   help="production = the 10 real M3TSZ streams embedded in "
                        "the reference's benchmarks, replicated to nseries")
    p.add_argument("--mode", choices=["decode", "encode", "rollup", "tiles",
                                      "parse-only"],
                   default="decode",
                   help="decode = headline metric (BASELINE configs[1]); "
                        "encode/rollup = BASELINE configs[2]/[3]")
    return p.parse_args()


def main():
    args = parse_args()
    import torch
    import m3_amd
    from m3_amd import engine, workload

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    dist = None
    ndev = max(torch.cuda.device_count(), 1)
    backend = "nccl"
    if world > ndev:
        # oversubscription (more ranks than GPUs, e.g. a 2-rank rendezvous
        # test on a 1-GPU box): RCCL refuses duplicate devices, so the
        # barrier/MAX-reduce run over gloo; the compute still shares the GPU
        backend = "gloo"
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        dist.init_process_group(backend)
    device = f"cuda:{local_rank % ndev}"
    torch.cuda.set_device(device)

    if args.nseries is None:
        args.nseries = 1_000_000 * world  # weak scaling: 1M series per GPU
        args.scaling = "weak"
    else:
        args.scaling = "strong" if world > 1 else "weak"
    lo, hi = shard_range(args.nseries, world, rank)
    n_local = hi - lo
    t0 = time.time()
    if args.data == "production":
        d_blob, d_offsets, d_lens, enc_bytes, expected_counts = \
            build_production_batch(torch, n_local, device)
        args.npts = int(expected_counts.max())
    else:
        d_blob, d_offsets, d_lens, enc_bytes = workload.encode_on_device(
            torch, n_local, args.npts, chunk=args.chunk, device=device,
            rank_offset=lo, verbose=(rank == 0))
        expected_counts = None
    torch.cuda.synchronize()
    if rank == 0:
        print(f"[bench] built {n_local} series x {args.npts} pts: "
              f"{enc_bytes/1e9:.3f} GB encoded "
              f"({enc_bytes/(n_local*args.npts):.2f} B/pt) "
              f"in {time.time()-t0:.1f}s", flush=True)

    npts_total_local = n_local * args.npts
    if args.mode == "encode":
        bench_encode(args, torch, engine, workload, device, n_local, lo)
        return
    if args.mode == "rollup":
        bench_rollup(args, torch, engine, d_blob, d_offsets, d_lens,
                     enc_bytes, device, n_local)
        return
    if args.mode == "tiles":
        bench_tiles(args, torch, engine, d_blob, d_offsets, d_lens,
                    enc_bytes, device, n_local)
        return
    out_ts = torch.empty((n_local, args.npts), dtype=torch.int64, device=device)
    out_vals = torch.empty((n_local, args.npts), dtype=torch.float64, device=device)
    out_counts = torch.empty(n_local, dtype=torch.int32, device=device)
    out_errs = torch.empty(n_local, dtype=torch.int32, device=device)
    # length-sorted wave scheduling: every wavefront decodes 64 similar-cost
    # streams (stream length is known before decode)
    d_perm = None
    if not args.no_sort:
        d_perm = torch.argsort(d_lens).to(torch.int32)
    if args.layout is None:
        args.layout = "index" if args.data == "production" else "sorted"
    if args.layout == "sorted" and d_perm is not None:
        # physical layout pass: repack the blob in schedule order so each
        # wave's 64 streams are HBM neighbors; scheduling perm becomes
        # identity and is dropped
        new_lens = d_lens[d_perm.long()]
        aligned = ((new_lens.to(torch.int64) + 63) // 64) * 64
        # n+1 offsets: the end sentinel is needed by the oracle-side
        # cpu_baseline slice (the kernels only use offsets[series])
        new_off = torch.zeros(aligned.numel() + 1, dtype=torch.int64,
                              device=aligned.device)
        torch.cumsum(aligned, 0, out=new_off[1:])
        d_blob2 = torch.empty_like(d_blob)
        engine.regather_dev(d_blob, d_offsets, d_lens, d_perm,
                            new_off[:-1].contiguous(), d_blob2)
        torch.cuda.synchronize()
        d_blob, d_offsets, d_lens = d_blob2, new_off, new_lens.contiguous()
        # (d_offsets now has n+1 entries; kernels index only [0, n))
        # series identity = pack order: output row i now holds the stream
        # that was at perm[i] before the repack
        if expected_counts is not None:
            expected_counts = expected_counts[d_perm.long()]
        d_perm = None

    parse_only = args.mode == "parse-only"

    def step():
        if parse_only:
            # diagnostic: decode with outputs discarded (null out pointers)
            from m3_amd.engine import lib as _lib, _dev_ptr, _torch_stream, _check
            rc = _lib().m3gpu_decode_batch_dev_perm(
                _dev_ptr(d_blob), _dev_ptr(d_offsets), _dev_ptr(d_lens),
                _dev_ptr(d_perm) if d_perm is not None else None,
                d_lens.numel(), 1, 1, None, None,
                _dev_ptr(out_counts), _dev_ptr(out_errs), args.npts,
                _torch_stream())
            _check(rc, "decode parse-only")
        else:
            engine.decode_batch_dev(d_blob, d_offsets, d_lens, out_ts, out_vals,
                                    out_counts, out_errs, d_perm=d_perm)

    # correctness gate outside the timed region: every series decodes fully
    step()
    torch.cuda.synchronize()
    if expected_counts is None:
        counts_ok = int((out_counts != args.npts).sum().item()) == 0
    else:
        counts_ok = bool((out_counts == expected_counts).all().item())
    if int(out_errs.abs().sum().item()) != 0 or not counts_ok:
        raise SystemExit("[bench] decode errors in workload — aborting")
    npts_decoded_local = int(out_counts.to(torch.int64).sum().item())

    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()

    # timed region: K steps, barrier + sync on both sides, per-launch HIP
    # events on the decode kernel for the roofline
    ev = [(torch.cuda.Event(enable_timing=True), torch.cuda.Event(enable_timing=True))
          for _ in range(args.steps)]
    if dist:
        dist.barrier()
    torch.cuda.synchronize()
    wall0 = time.time()
    for k in range(args.steps):
        ev[k][0].record()
        step()
        ev[k][1].record()
    torch.cuda.synchronize()
    wall1 = time.time()
    if dist:
        dist.barrier()

    elapsed = wall1 - wall0
    if dist:
        t = torch.tensor([elapsed],
                         device=device if backend == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    kernel_ms = float(np.mean([a.elapsed_time(b) for a, b in ev]))
    npts_global = npts_decoded_local * world  # == nseries*npts for synthetic
    dps = npts_global * args.steps / elapsed  # datapoints/sec, whole job

    # roofline on the dominant kernel (decode): algorithmic bytes per launch
    algo_bytes = enc_bytes + 16 * npts_decoded_local  # stream in + (ts,val) out
    achieved_gbs = algo_bytes / (kernel_ms / 1e3) / 1e9
    read_only_gbs = enc_bytes / (kernel_ms / 1e3) / 1e9
    traffic = None
    tpath = os.path.join(REPO, "profiles", "traffic.json")
    if rank == 0 and os.path.exists(tpath):
        try:
            with open(tpath) as f:
                tj = json.load(f)
            if tj.get("kernel") == "k_decode_batch" and \
               tj.get("nseries") == n_local and tj.get("npts") == args.npts:
                traffic = tj.get("bytes_per_launch")
        except Exception:
            pass

    cpu_baseline = None
    if rank == 0 and world == 1 and not args.skip_cpu_baseline:
        cpu_baseline = run_cpu_baseline(args, d_blob, d_offsets, d_lens, n_local)

    if rank == 0:
        line = {
            "metric": "datapoints/sec M3TSZ decode (1M series x 1440 pts) + HBM GB/s vs peak",
            "value": dps,
            "unit": "datapoints/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1e3,
            "higher_is_better": True,
            "scaling": args.scaling,
            "vs_baseline": None,
            "dtype": "f64",
            "data": "synthetic" if args.data == "synthetic" else "replicated production streams",
            "config": {
                "workload": "1M series x 1440 pts batched M3TSZ decode, HBM-bandwidth microbench"
                            if args.data == "synthetic" else
                            "1M replicated production streams batched M3TSZ decode",
                "nseries": args.nseries,
                "npts": args.npts,
                "data_detail": args.data,
                "encoded_bytes_per_pt": enc_bytes / npts_decoded_local,
                "parallelism": f"series-sharded x{world} (no collectives)",
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved_gbs,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": achieved_gbs / HBM_PEAK_GBS,
                "read_only_gbs": read_only_gbs,
                "read_only_frac": read_only_gbs / HBM_PEAK_GBS,
                "traffic": traffic,
            },
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(line), flush=True)

    if dist:
        dist.destroy_process_group()


def run_cpu_baseline(args, d_blob, d_offsets, d_lens, n_local):
    """Oracle (C restatement of the reference decoder, kind='port') on the
    host cores, over a bounded sample of the same streams (~10-30 s)."""
    import oracle  # permitted: cpu_baseline leg only

    cores = os.cpu_count()
    # probe to size the sample (large enough that thread spin-up noise
    # does not dominate the per-series rate)
    probe_n = min(50000, n_local)
    off = d_offsets[:probe_n + 1].cpu().numpy().astype(np.uint64)
    lens = d_lens[:probe_n].cpu().numpy().astype(np.uint32)
    blob = d_blob[: int(off[-1])].cpu().numpy()
    t0 = time.time()
    oracle.decode_batch(blob, off, stride=args.npts, nthreads=cores)
    probe_t = time.time() - t0
    per_series = probe_t / probe_n
    if args.cpu_sample_series:
        target_series = min(args.cpu_sample_series, n_local)
    elif per_series * n_local <= 30.0:
        # the whole batch fits the 10-30s budget: unbiased full sample
        # (a PREFIX of the schedule-sorted blob would over-sample the
        # shortest streams)
        target_series = n_local
    else:
        target_series = int(min(n_local, max(probe_n, 15.0 / per_series)))
    off = d_offsets[:target_series + 1].cpu().numpy().astype(np.uint64)
    lens = d_lens[:target_series].cpu().numpy().astype(np.uint32)
    blob = d_blob[: int(off[-1])].cpu().numpy()
    t0 = time.time()
    _, _, counts = oracle.decode_batch(blob, off, stride=args.npts, nthreads=cores)
    dt = time.time() - t0
    pts = int(counts.astype(np.int64).sum())
    return {
        "value": pts / dt,
        "unit": "datapoints/sec",
        "cores": cores,
        "kind": "port",
        "sample": f"{target_series} of {n_local} series ({pts} pts) in {dt:.1f}s",
    }




def bench_encode(args, torch, engine, workload, device, n_local, lo):
    """BASELINE configs[2]: 1M-series batched M3TSZ encode, 1x MI355X.
    Input: decoded SoA resident in HBM; step = encode into strided rows."""
    import numpy as np
    npts = args.npts
    d_ts = torch.empty((n_local, npts), dtype=torch.int64, device=device)
    d_vals = torch.empty((n_local, npts), dtype=torch.float64, device=device)
    chunk = args.chunk
    for s0 in range(0, n_local, chunk):
        n = min(chunk, n_local - s0)
        ts, vals = workload.gen_chunk(lo + s0, n, npts)
        d_ts[s0:s0 + n] = torch.from_numpy(ts).to(device)
        d_vals[s0:s0 + n] = torch.from_numpy(vals).to(device)
    d_counts = torch.full((n_local,), npts, dtype=torch.int32, device=device)
    out_stride = (24 * npts + 32 + 7) & ~7
    d_out = torch.zeros((n_local, out_stride), dtype=torch.uint8, device=device)
    d_lens = torch.empty(n_local, dtype=torch.int32, device=device)
    d_errs = torch.empty(n_local, dtype=torch.int32, device=device)

    def step():
        engine.encode_batch_dev(d_ts, d_vals, d_counts, d_out, d_lens, d_errs)

    step()
    torch.cuda.synchronize()
    assert int(d_errs.abs().sum().item()) == 0
    enc_bytes = int(d_lens.to(torch.int64).sum().item())
    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dt = time.time() - t0
    npts_total = n_local * npts
    dps = npts_total * args.steps / dt
    algo = enc_bytes + 16 * npts_total
    print(json.dumps({
        "metric": "datapoints/sec M3TSZ encode (1M series x 1440 pts)",
        "value": dps, "unit": "datapoints/sec", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1e3, "higher_is_better": True,
        "scaling": args.scaling, "vs_baseline": None, "dtype": "f64",
        "data": "synthetic",
        "config": {"workload": "1M series batched M3TSZ encode, 1x MI355X",
                   "nseries": n_local, "npts": npts,
                   "encoded_bytes_per_pt": enc_bytes / npts_total},
        "roofline": {"bound": "hbm", "achieved": algo / (dt / args.steps) / 1e9,
                     "peak": HBM_PEAK_GBS, "unit": "GB/s",
                     "frac": algo / (dt / args.steps) / 1e9 / HBM_PEAK_GBS,
                     "traffic": None},
    }), flush=True)


def bench_rollup(args, torch, engine, d_blob, d_offsets, d_lens, enc_bytes,
                 device, n_local):
    """BASELINE configs[3]: fused decode->downsample, 10s -> 1m
    sum/min/max/p99 with m3aggregator semantics."""
    window = 60 * 10**9
    nbuckets = args.npts * 10 // 60
    aggs = ["sum", "min", "max", "p99"]
    out = torch.empty((n_local, nbuckets, len(aggs)), dtype=torch.float64,
                      device=device)
    wts = torch.empty((n_local, nbuckets), dtype=torch.int64, device=device)
    errs = torch.empty(n_local, dtype=torch.int32, device=device)

    def step():
        engine.rollup_batch_dev(d_blob, d_offsets, d_lens, engine.METRIC_TIMER,
                                window, nbuckets, aggs, out, wts, errs)

    step()
    torch.cuda.synchronize()
    assert int(errs.abs().sum().item()) == 0
    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dt = time.time() - t0
    npts_total = n_local * args.npts
    dps = npts_total * args.steps / dt
    algo = enc_bytes + n_local * nbuckets * (len(aggs) * 8 + 8)
    print(json.dumps({
        "metric": "datapoints/sec fused M3TSZ decode->1m rollup (sum/min/max/p99)",
        "value": dps, "unit": "datapoints/sec", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1e3, "higher_is_better": True,
        "scaling": args.scaling, "vs_baseline": None, "dtype": "f64",
        "data": "synthetic",
        "config": {"workload": "Fused decode->downsample 10s->1m rollup, 1x MI355X",
                   "nseries": n_local, "npts": args.npts, "nbuckets": nbuckets,
                   "aggs": aggs},
        "roofline": {"bound": "hbm", "achieved": algo / (dt / args.steps) / 1e9,
                     "peak": HBM_PEAK_GBS, "unit": "GB/s",
                     "frac": algo / (dt / args.steps) / 1e9 / HBM_PEAK_GBS,
                     "traffic": None},
    }), flush=True)


def build_production_batch(torch, nseries, device):
    """Replicate the 10 production-encoded streams from the reference's own
    benchmarks (tests/golden/production_streams.json, 719-720 pts each,
    0.41-2.97 B/pt) to nseries device-resident unique copies."""
    import base64
    import oracle  # stream prep + expected counts only (not the timed path)
    with open(os.path.join(REPO, "tests", "golden", "production_streams.json")) as f:
        ps = json.load(f)
    streams = [base64.b64decode(b) for b in ps["samples"]]
    counts1 = [len(oracle.decode_series(s)["ts"]) for s in streams]
    from m3_amd.engine import pack_streams
    blob1, offs1, lens1 = pack_streams(streams)
    period = int(offs1[-1])
    nrep = (nseries + len(streams) - 1) // len(streams)
    d_small = torch.from_numpy(blob1).to(device)
    d_blob = d_small.repeat(nrep)[: nrep * period]
    offsets = (np.tile(offs1[:-1], nrep)
               + np.repeat(np.arange(nrep, dtype=np.int64) * period, len(streams)))
    offsets = offsets[:nseries]
    offsets = np.concatenate([offsets, [nrep * period]]).astype(np.int64)
    lens = np.tile(lens1, nrep)[:nseries]
    counts = np.tile(np.asarray(counts1, np.int32), nrep)[:nseries]
    d_offsets = torch.from_numpy(offsets).to(device)
    d_lens = torch.from_numpy(lens.astype(np.int32)).to(device)
    expected_counts = torch.from_numpy(counts).to(device)
    return d_blob, d_offsets, d_lens, int(lens.astype(np.int64).sum()), expected_counts


def bench_tiles(args, torch, engine, d_blob, d_offsets, d_lens, enc_bytes,
                device, n_local):
    """AggregateTiles-shaped job: decode -> 10s->1m
    rollup -> re-encode the downsampled tile streams, all on-device."""
    window = 60 * 10**9
    nbuckets = args.npts * 10 // 60

    def step():
        return engine.aggregate_tiles_dev(torch, d_blob, d_offsets, d_lens,
                                          engine.METRIC_GAUGE, window,
                                          nbuckets, "last")
    tb, tl, tc = step()
    out_bytes = int(tl.to(torch.int64).sum().item())
    for _ in range(args.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(args.steps):
        step()
    torch.cuda.synchronize()
    dt = time.time() - t0
    npts_total = n_local * args.npts
    dps = npts_total * args.steps / dt
    print(json.dumps({
        "metric": "datapoints/sec AggregateTiles (decode->1m last->re-encode)",
        "value": dps, "unit": "datapoints/sec", "n_gpus": 1,
        "steps": args.steps, "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1e3, "higher_is_better": True,
        "scaling": args.scaling, "vs_baseline": None, "dtype": "f64",
        "data": "synthetic",
        "config": {"workload": "AggregateTiles 10s->1m, 1x MI355X",
                   "nseries": n_local, "npts": args.npts,
                   "tile_bytes_out": out_bytes},
    }), flush=True)


if __name__ == "__main__":
    main()
