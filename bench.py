#!/usr/bin/env python3
"""bench.py — flagship benchmark: the DocDB SST-block scan-and-filter hot
path on MI355X (BASELINE.json metric: scanned rows/s, 100M-row filtered
aggregate).

One "step" = one pass of the hot path over the synthetic tablet set: GPU
block-decode + MVCC visibility + 3 int64 predicates + SUM(int64)/COUNT, plus
the cross-tablet aggregate merge (RCCL all-reduce when world_size > 1).

N=1 workload = BASELINE configs[1]: 1 tablet, 100M rows, 4KB SST blocks,
three_shared_parts encoding, packed-row V2, 3 int64 predicates + SUM(int64).
N>1 = configs[2]: 8 tablets x 12.5M rows sharded tablet->rank (strong
scaling), final RCCL all-reduce of the partial aggregates over xGMI.

Inputs are generated on host (seed 42) and resident in HBM before the timed
region. The CPU baseline leg times the oracle (oracle/ — test infrastructure)
on a bounded sample on this host; it is a reported baseline, not the target.
"""
import argparse
import ctypes
import json


def _wrap_i64(v):
    """Two's-complement wrap to int64 — the per-tablet partial SUMs are
    already wrapped C int64s; their PYTHON sum can exceed the int64 range
    the all-reduce tensor holds. Wrapping keeps rank-side math identical
    to the device/C semantics."""
    return ((v + (1 << 63)) & ((1 << 64) - 1)) - (1 << 63)
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

HBM_PEAK_GBS = 8000.0  # MI355X HBM3E spec peak (MI355X_MICROARCH.md)


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


# all-cores CPU-baseline worker state (module level: fork workers inherit
# the tablet copy-on-write; each worker scans a contiguous block chunk)
_CPU_G = {}


def _cpu_worker(rng):
    import ybgpu as y
    lo, hi = rng
    res, _ = y.orcl_scan(_CPU_G["data"], _CPU_G["offsets"], hi - lo,
                         _CPU_G["sc"], _CPU_G["spec"], block_lo=lo)
    return res.rows_scanned, res.rows_matched


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--rows", type=int, default=100_000_000,
                    help="total rows across all tablets")
    ap.add_argument("--workload", choices=["filtersum", "mvcc", "groupby"],
                    default="filtersum",
                    help="filtersum = BASELINE configs[1]/[2]; mvcc = "
                         "configs[3] (5 versions/row, COUNT); groupby = "
                         "configs[4]-shaped grouped aggregate")
    ap.add_argument("--cpu-sample-seconds", type=float, default=10.0)
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = world if world > 1 else args.gpus

    import ybgpu as y
    import gpu_scan

    dist = None
    torch = None
    backend = None
    if world > 1:
        import torch  # noqa
        import torch.distributed as dist_mod
        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        backend = os.environ.get(
            "YBG_DIST_BACKEND",
            "nccl" if torch.cuda.is_available() else "gloo")
        dist.init_process_group(backend=backend)
        ndev = torch.cuda.device_count() if torch.cuda.is_available() else 0
        dev = local_rank % ndev if ndev else 0
        if ndev:
            torch.cuda.set_device(dev)
        gpu_scan._lib().yb_gpu_set_device(dev)

    if not gpu_scan.gpu_available():
        log("FATAL: no HIP device — the product path has no CPU fallback")
        sys.exit(2)

    # ---- dataset ---------------------------------------------------------
    versions = 1
    group_col = 0
    read_micros = 1_700_000_000_000_000
    if args.workload == "filtersum":
        schema = y.make_schema([y.KT_INT64],
                               [(10 + i, y.T_INT64, 1) for i in range(4)])
        preds = [y.Pred(0, 0, y.PRED_GT, 1 << 39, None, 0),
                 y.Pred(0, 1, y.PRED_LT, 3 << 38, None, 0),
                 y.Pred(0, 2, y.PRED_GE, 1 << 36, None, 0)]
        aggs = [y.Agg(y.AGG_SUM_INT64, 3), y.Agg(y.AGG_COUNT_STAR, 0)]
    elif args.workload == "mvcc":
        # BASELINE configs[3]: 5 versions/row, visibility + COUNT
        schema = y.make_schema([y.KT_INT64],
                               [(10 + i, y.T_INT64, 1) for i in range(4)])
        preds = []
        aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0)]
        versions = 5
        # read BETWEEN versions 2 and 3 (versions at ht_base + k*1e6 us):
        # the two newer versions per row are in the future and must be
        # skipped (SkipFutureRecords semantics), the newest visible one is
        # deduped — the visibility work config 4 names, not just dedup
        read_micros = 1_600_000_002_500_000
    else:  # groupby — BASELINE configs[4]: mixed row incl. the 32-byte
        # string column, range predicates over the int64 and the string
        # (string EQUALITY parity is pinned in the test suite; over
        # uniform-random 32-byte strings an equality predicate selects
        # ~nothing, which would leave the grouped aggregates empty)
        schema = y.make_schema(
            [y.KT_INT64],
            [(10, y.T_INT64, 1), (11, y.T_INT64, 1), (12, y.T_DOUBLE, 1),
             (13, y.T_STRING, 1)])
        _slo = ctypes.create_string_buffer(b"m" * 32, 32)
        preds = [y.Pred(0, 1, y.PRED_GT, 1 << 38, None, 0),
                 y.Pred(0, 1, y.PRED_LT, 3 << 38, None, 0),
                 y.Pred(0, 3, y.PRED_GE, 0,
                        ctypes.cast(_slo, ctypes.POINTER(ctypes.c_uint8)),
                        32)]
        aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
        group_col = 1  # 1 + value col 0
    n_tablets = 1 if n_gpus == 1 and world <= 1 else 8
    rows_per_tablet = args.rows // n_tablets
    my_tablets = [t for t in range(n_tablets) if t % max(world, 1) == rank]

    def make_spec():
        spec = y.ScanSpec()
        spec.schema = schema
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.group_col = group_col
        spec.read_time = y.read_time(read_micros)
        spec.num_preds = len(preds)
        for i, p in enumerate(preds):
            spec.preds[i] = p
        # the mvcc workload scans a historical snapshot over 5-version
        # rows — the caller-known condition the expect_versions hint
        # models (it selects the version-chain kernel shape)
        spec.expect_versions = 1 if args.workload == "mvcc" else 0
        spec.num_aggs = len(aggs)
        for i, a in enumerate(aggs):
            spec.aggs[i] = a
        return spec

    scans = []
    total_bytes_local = 0
    gen_t0 = time.time()
    first_tablet_data = None
    for t in my_tablets:
        data, offsets, nb, total, ne = y.generate(
            schema, rows=rows_per_tablet, seed=42 + t, versions=versions,
            ht_base_micros=1_600_000_000_000_000,
            ht_step_micros=1_000_000 if versions > 1 else 1000,
            group_mod=65536 if group_col else 0)
        total_bytes_local += total
        s = gpu_scan.GpuScan(make_spec())
        s.feed_blocks_host(data, offsets, nb, total)
        scans.append(s)
        if first_tablet_data is None:
            first_tablet_data = (data, offsets, nb, total, ne)
        else:
            y.product().ybg_free(data)
            y.product().ybg_free(offsets)
    log(f"[rank {rank}] generated+uploaded {len(my_tablets)} tablet(s), "
        f"{total_bytes_local/1e9:.2f} GB in {time.time()-gen_t0:.1f}s "
        f"({total_bytes_local/max(rows_per_tablet*len(my_tablets),1):.1f} "
        f"bytes/row)")

    agg_buf = None
    if dist is not None:
        import torch
        agg_buf = torch.zeros(
            2, dtype=torch.int64,
            device="cuda" if backend == "nccl" else "cpu")

    def step():
        if group_col:
            # grouped partial aggregates; the cross-tablet merge of the
            # partial tables happens host-side (SURVEY §8e all-gather +
            # merge; single-node benchmark reduces the partials directly)
            import numpy as np
            total_sum = 0
            total_cnt = 0
            for s in scans:
                _keys, vals, cnts, _kb, n = s.group_aggregate_raw()
                if n:
                    va = np.frombuffer(vals, dtype=np.int64,
                                       count=n * y.MAX_AGGS)
                    va = va.reshape(n, y.MAX_AGGS)
                    total_cnt += int(va[:, 0].sum())
                    total_sum += int(va[:, 1].sum())
            if dist is not None:
                agg_buf[0] = _wrap_i64(total_sum)
                agg_buf[1] = total_cnt
                dist.all_reduce(agg_buf)
                total_sum = int(agg_buf[0].item())
                total_cnt = int(agg_buf[1].item())
            return _wrap_i64(total_sum), total_cnt
        for s in scans:
            s.execute()
        results = [s.aggregates() for s in scans]
        total_sum = sum(r.aggs[0].value_i64 for r in results)
        total_cnt = sum(r.aggs[1].value_i64 for r in results)
        if dist is not None:
            agg_buf[0] = _wrap_i64(total_sum)
            agg_buf[1] = total_cnt
            dist.all_reduce(agg_buf)  # RCCL over xGMI: the one collective
            total_sum = int(agg_buf[0].item())
            total_cnt = int(agg_buf[1].item())
        return _wrap_i64(total_sum), total_cnt

    def barrier_sync():
        if dist is not None:
            dist.barrier()
            import torch
            if torch.cuda.is_available():
                torch.cuda.synchronize()

    # ---- warmup ----------------------------------------------------------
    for _ in range(args.warmup):
        step()
    barrier_sync()

    # ---- timed region ----------------------------------------------------
    t0 = time.time()
    checksum = None
    kernel_decode_ms = 0.0
    kernel_total_ms = 0.0
    for _ in range(args.steps):
        checksum = step()
        for s in scans:
            tot, dec = s.kernel_ms()
            kernel_total_ms += tot
            kernel_decode_ms += dec
    barrier_sync()
    elapsed = time.time() - t0
    if dist is not None:
        import torch
        e = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if backend == "nccl" else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    total_rows = rows_per_tablet * n_tablets
    rows_per_s = total_rows * args.steps / elapsed
    ms_per_step = elapsed * 1000.0 / args.steps

    # roofline: dominant kernel = k_scan (block decode+filter+aggregate).
    # achieved = algorithmic bytes per launch / avg launch duration; one
    # launch per tablet per step, algorithmic bytes = that tablet's
    # serialized block bytes (every block byte crosses HBM once).
    n_launches = len(scans) * args.steps
    avg_decode_ms = kernel_decode_ms / max(n_launches, 1)
    bytes_per_launch = total_bytes_local / max(len(scans), 1)
    achieved_gbs = (bytes_per_launch / (avg_decode_ms / 1e3)) / 1e9 \
        if avg_decode_ms > 0 else 0.0
    traffic = os.environ.get("YBG_TRAFFIC_BYTES_PER_LAUNCH")
    if traffic:
        traffic = float(traffic)
    else:
        # committed PMC measurement for this exact workload, if present
        # (profiles/traffic_calibration.json; see profiles/README.md for the
        # collection commands and the gfx950 FETCH_SIZE caveat)
        try:
            cal = json.load(open(os.path.join(
                os.path.dirname(os.path.abspath(__file__)), "profiles",
                "traffic_calibration.json")))
            entries = cal if isinstance(cal, list) else [cal]
            traffic = None
            for c in entries:
                if (c.get("workload") == args.workload
                        and c.get("rows") == args.rows):
                    traffic = float(c["traffic_bytes_per_launch"])
                    break
        except Exception:
            traffic = None

    # ---- CPU baseline (oracle, rank 0, N=1 only) -------------------------
    cpu_baseline = None
    if rank == 0 and world <= 1 and not args.skip_cpu_baseline:
        data, offsets, nb, total, ne = first_tablet_data
        # calibrate on 256 blocks, then size the sample for the budget
        osc = y.orcl_schema_from(schema)
        ospec = y.OrclScanSpec()
        ospec.read_time = y.orcl_read_time(read_micros)
        ospec.num_preds = len(preds)
        for i, p in enumerate(preds):
            ospec.preds[i] = y.OrclPred(p.is_key_col, p.col, p.op, p.datum,
                                        p.bytes, p.bytes_len)
        ospec.num_aggs = len(aggs)
        for i, a in enumerate(aggs):
            ospec.aggs[i] = y.OrclAgg(a.op, a.col)

        def run_orcl(nblocks):
            t = time.time()
            res, _ = y.orcl_scan(data, offsets, nblocks, osc, ospec)
            return time.time() - t, res

        # single-thread sample (secondary figure + per-block calibration)
        cal_blocks = min(nb, 256)
        cal_t, cal_res = run_orcl(cal_blocks)
        per_block = max(cal_t / cal_blocks, 1e-9)
        sample_blocks = min(nb, max(cal_blocks,
                                    int(args.cpu_sample_seconds / per_block)))
        st, sres = run_orcl(sample_blocks)
        sample_rows = sres.rows_scanned
        cpu_1t_rows_per_s = sample_rows / st

        # all-cores oracle on the SAME tablet in the SAME run (north-star
        # measurement clause): fork pools, each worker a contiguous block
        # chunk, warm-up over own chunks before timing. The host's
        # parallel throughput PEAKS below the logical core count (2-node
        # NUMA + per-core saturation: measured 210 M rows/s at 32
        # workers vs 122 M at 256 — scripts/cpu_baseline_sweep.py), so
        # the worker count is PROBED and the best figure reported — the
        # baseline is "the best this host can do", not "all cores".
        import multiprocessing as _mp
        ncpu = os.cpu_count() or 1
        probe = sorted({min(c, ncpu, nb)
                        for c in (32, 64, 128, ncpu)
                        if min(c, ncpu, nb) >= 1})
        _CPU_G.update(data=data, offsets=offsets, sc=osc, spec=ospec)
        ctx = _mp.get_context("fork")

        def _passes(nproc, n_timed):
            cuts = [nb * i // nproc for i in range(nproc + 1)]
            ranges = [(cuts[i], cuts[i + 1]) for i in range(nproc)
                      if cuts[i + 1] > cuts[i]]
            best = None
            with ctx.Pool(len(ranges)) as pool:
                pool.map(_cpu_worker, ranges)  # warm-up over own chunks
                for _ in range(n_timed):
                    t0 = time.time()
                    out_w = pool.map(_cpu_worker, ranges)
                    wall = time.time() - t0
                    if best is None or wall < best[0]:
                        best = (wall, out_w)
            return len(ranges), best

        results = {}
        for c in probe:
            nw, (wall, out_w) = _passes(c, 1)
            results[nw] = (wall, out_w)
        best_n = min(results, key=lambda n: results[n][0])
        nw, (wall, out_w) = _passes(best_n, 2)
        if results[best_n][0] < wall:
            wall, out_w = results[best_n]
        par_rows = sum(r[0] for r in out_w)
        cpu_rows_per_s = par_rows / wall
        cpu_baseline = {
            "value": cpu_rows_per_s,
            "unit": "rows/s",
            "cores": nw,
            "kind": "port",
            "sample": f"full tablet ({par_rows} rows), worker count probed"
                      f" over {probe} on {ncpu} logical cores, best at "
                      f"{nw} workers, best-of-3 passes, {wall:.2f}s wall; "
                      f"single-thread {cpu_1t_rows_per_s/1e6:.1f} Mrows/s "
                      f"over {sample_blocks} blocks ({st:.1f}s)",
            "single_thread_value": cpu_1t_rows_per_s,
        }
        log(f"[cpu baseline] {cpu_rows_per_s/1e6:.1f} Mrows/s at {nw} "
            f"workers (probed {probe}); {cpu_1t_rows_per_s/1e6:.2f} "
            f"Mrows/s single-thread")

    if rank == 0:
        out = {
            "metric": "scanned_rows_per_s",
            "value": rows_per_s,
            "unit": "rows/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": (
                    {"filtersum":
                         ("1 tablet, 100M rows, 4KB SST blocks, "
                          "three_shared_parts, packed-row V2, 3 int64 "
                          "predicates + SUM(int64)" if n_tablets == 1 else
                          "8 tablets x 12.5M rows, same query, sharded "
                          "tablet->rank + RCCL all-reduce"),
                     "mvcc": "100M rows x 5 versions, read-at-HT visibility "
                             "+ COUNT",
                     "groupby": "mixed-type rows, predicate + GROUP-BY-key "
                                "partial aggregates"}[args.workload]),
                "rows": total_rows,
                "tablets": n_tablets,
                "block_size": 4096,
                "checksum": {"agg0": checksum[0], "agg1": checksum[1]},
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved_gbs,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": achieved_gbs / HBM_PEAK_GBS,
                "traffic": traffic,
            },
            "cpu_baseline": cpu_baseline,
        }
        print(json.dumps(out), flush=True)

    for s in scans:
        s.close()
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
