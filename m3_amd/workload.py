"""Synthetic benchmark workloads (BASELINE.json configs).

Pure numpy + m3_amd product APIs — no oracle imports. Timestamps follow the
reference benchmark shape: start 1427162462e9 ns + i*10s (exact 10 s cadence
=> DoD == 0 => 1 timestamp bit/pt after the 2nd point). Values cycle through
four seeded distributions, grouped in 64-series runs
((i >> 6) % 4) so each wavefront's 64 lane parsers stay branch-coherent:
  0: counter-like random-walk 12-digit ints
  1: timer-like 7-digit.6-decimal floats
  2: gauge-like small one-decimal floats
  3: uniform random f64 bits (incompressible worst case)
"""
import numpy as np

START_NS = 1427162462 * 10**9
CADENCE_NS = 10 * 10**9


def gen_chunk(series0, nseries, npts, seed_base=42):
    """Generate (ts[int64 n x npts], vals[f64 n x npts]) for series indices
    [series0, series0+nseries)."""
    ts = START_NS + np.arange(npts, dtype=np.int64) * CADENCE_NS
    ts = np.broadcast_to(ts, (nseries, npts)).copy()
    vals = np.empty((nseries, npts), dtype=np.float64)
    # kinds alternate per 64-series group: every wavefront (one series per
    # lane) sees a single distribution — the host controls series placement,
    # exactly as the reference's shard assignment does, and grouping similar
    # series keeps the 64 lane parsers branch-coherent.
    import os
    kind_shift = int(os.environ.get("M3_KIND_SHIFT", "6"))
    kind_only = os.environ.get("M3_KIND_ONLY")
    idx = series0 + np.arange(nseries)
    for kind in range(4):
        if kind_only is not None:
            rows = np.arange(nseries) if kind == int(kind_only) else np.array([], int)
        else:
            rows = np.nonzero((idx >> kind_shift) % 4 == kind)[0]
        if not len(rows):
            continue
        rng = np.random.default_rng(seed_base + series0 * 7 + kind)
        n = len(rows)
        if kind == 0:
            start = rng.integers(10**11, 10**12, (n, 1)).astype(np.float64)
            steps = rng.integers(-10**5, 10**5, (n, npts)).astype(np.float64)
            steps[:, 0] = 0
            vals[rows] = start + np.cumsum(steps, axis=1)
        elif kind == 1:
            dig = rng.integers(0, 10**7, (n, npts)).astype(np.float64)
            dec = rng.integers(0, 10**6, (n, npts)).astype(np.float64)
            vals[rows] = dig + dec * 1e-6
        elif kind == 2:
            vals[rows] = np.round(rng.random((n, npts)) * 10, 1)
        else:
            bits = rng.integers(0, 2**52, (n, npts), dtype=np.uint64)
            bits |= np.uint64(0x3FF0000000000000)  # finite, exponent 0
            vals[rows] = bits.view(np.float64)
    return ts, vals


def encode_on_device(torch, nseries, npts, chunk=65536, device="cuda:0",
                     int_optimized=True, unit=1, seed_base=42, rank_offset=0,
                     verbose=False):
    """Build the device-resident encoded batch with the PRODUCT encoder:
    host-generate chunks -> H2D -> m3gpu encode kernel -> compact into the
    tight aligned blob. Returns (d_blob, d_offsets, d_lens, total_bytes)."""
    from . import engine

    out_stride = (24 * npts + 32 + 7) & ~7
    chunks = []
    lens_all = np.empty(nseries, dtype=np.uint32)
    d_scratch = torch.zeros(chunk * out_stride, dtype=torch.uint8, device=device)
    d_counts = torch.empty(chunk, dtype=torch.int32, device=device)
    d_errs = torch.empty(chunk, dtype=torch.int32, device=device)
    d_lens_t = torch.empty(chunk, dtype=torch.int32, device=device)
    for s0 in range(0, nseries, chunk):
        n = min(chunk, nseries - s0)
        ts, vals = gen_chunk(rank_offset + s0, n, npts, seed_base)
        d_ts = torch.from_numpy(ts).to(device, non_blocking=False)
        d_vals = torch.from_numpy(vals).to(device)
        d_counts[:n].fill_(npts)
        engine.encode_batch_dev(
            d_ts.view(n, npts), d_vals.view(n, npts),
            d_counts[:n], d_scratch.view(-1)[: n * out_stride].view(n, out_stride),
            d_lens_t[:n], d_errs[:n], int_optimized=int_optimized, unit=unit)
        if int(d_errs[:n].abs().sum().item()) != 0:
            raise engine.M3GpuError("encode errors while building workload")
        lens = d_lens_t[:n].cpu().numpy().astype(np.uint32)
        lens_all[s0:s0 + n] = lens
        padded = (lens.astype(np.int64) + 63) & ~63
        offs = np.zeros(n + 1, dtype=np.int64)
        np.cumsum(padded, out=offs[1:])
        d_blob_chunk = torch.zeros(int(offs[-1]), dtype=torch.uint8, device=device)
        d_offs = torch.from_numpy(offs[:n].astype(np.int64)).to(device)
        engine.compact_dev(d_scratch, out_stride, d_lens_t[:n], d_offs, d_blob_chunk)
        chunks.append((d_blob_chunk, offs[:-1]))
        if verbose:
            print(f"  encoded series {s0 + n}/{nseries}", flush=True)
    # stitch chunks
    sizes = [c[0].numel() for c in chunks]
    total = int(np.sum(sizes))
    d_blob = torch.empty(total, dtype=torch.uint8, device=device)
    offsets = np.zeros(nseries + 1, dtype=np.int64)
    pos = 0
    i0 = 0
    for (blob_c, offs_c), sz in zip(chunks, sizes):
        d_blob[pos:pos + sz] = blob_c
        offsets[i0:i0 + len(offs_c)] = offs_c + pos
        pos += sz
        i0 += len(offs_c)
    offsets[nseries] = pos
    d_offsets = torch.from_numpy(offsets.astype(np.int64)).to(device)
    d_lens = torch.from_numpy(lens_all.astype(np.int32)).to(device)
    return d_blob, d_offsets, d_lens, int(lens_all.astype(np.int64).sum())
