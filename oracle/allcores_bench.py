#!/usr/bin/env python3
"""All-cores CPU oracle throughput — TEST INFRASTRUCTURE / MEASUREMENT ONLY.

Runs the single-thread CPU oracle (oracle/liborcl.so) over the default
bench workload (BASELINE configs[1] shape) on every host core, by giving
each forked worker a contiguous chunk of blocks. Rows straddling a chunk
cut are attributed to whichever side scans them (at most nproc-1 rows in
total), so the aggregate is approximate by <= nproc-1 rows — fine for a
throughput figure, NEVER for parity. The number this prints feeds the
"vs full host" context line in DESIGN.md; bench.py's JSON `cpu_baseline`
stays the exact single-thread measurement.

Usage (on the GPU box, which has the big core count):
    python oracle/allcores_bench.py [--rows 20000000] [--procs 0=all]
"""
import argparse
import multiprocessing as mp
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import ybgpu as y  # noqa: E402

_G = {}


def _worker(rng):
    lo, hi = rng
    data, offsets, sc, spec = _G["data"], _G["offsets"], _G["sc"], _G["spec"]
    t0 = time.time()
    res, _ = y.orcl_scan(data, offsets, hi - lo, sc, spec, block_lo=lo)
    return res.rows_scanned, res.rows_matched, time.time() - t0


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=20_000_000)
    ap.add_argument("--procs", type=int, default=0)
    args = ap.parse_args()
    nproc = args.procs or os.cpu_count()

    schema = y.make_schema([y.KT_INT64],
                           [(10 + i, y.T_INT64, 1) for i in range(4)])
    preds = [y.Pred(0, 0, y.PRED_GT, 1 << 39, None, 0),
             y.Pred(0, 1, y.PRED_LT, 3 << 38, None, 0),
             y.Pred(0, 2, y.PRED_GE, 1 << 36, None, 0)]
    aggs = [y.Agg(y.AGG_SUM_INT64, 3), y.Agg(y.AGG_COUNT_STAR, 0)]

    t0 = time.time()
    data, offsets, nb, total, ne = y.generate(schema, rows=args.rows, seed=42)
    print(f"generated {args.rows} rows, {nb} blocks, "
          f"{total / 1e9:.2f} GB in {time.time() - t0:.1f}s", flush=True)

    sc = y.orcl_schema_from(schema)
    spec = y.OrclScanSpec()
    spec.read_time = y.orcl_read_time(1_700_000_000_000_000)
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = y.OrclPred(p.is_key_col, p.col, p.op, p.datum,
                                   p.bytes, p.bytes_len)
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = y.OrclAgg(a.op, a.col)

    _G.update(data=data, offsets=offsets, sc=sc, spec=spec)
    cuts = [nb * i // nproc for i in range(nproc + 1)]
    ranges = [(cuts[i], cuts[i + 1]) for i in range(nproc)
              if cuts[i + 1] > cuts[i]]

    ctx = mp.get_context("fork")  # workers inherit the tablet copy-on-write
    with ctx.Pool(len(ranges)) as pool:
        # warm the pool (fork + import cost) before timing
        pool.map(_worker, [(0, min(4, nb))] * len(ranges))
        t0 = time.time()
        out = pool.map(_worker, ranges)
        wall = time.time() - t0
    rows = sum(r[0] for r in out)
    matched = sum(r[1] for r in out)
    worker_max = max(r[2] for r in out)
    print(f"procs={len(ranges)} rows_scanned={rows} rows_matched={matched} "
          f"wall={wall:.2f}s worker_max={worker_max:.2f}s")
    print(f"throughput: {rows / wall / 1e6:.1f} Mrows/s wall, "
          f"{rows / worker_max / 1e6:.1f} Mrows/s by slowest worker")


if __name__ == "__main__":
    main()
