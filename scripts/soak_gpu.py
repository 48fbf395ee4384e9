#!/usr/bin/env python3
"""Randomized differential soak on a real MI355X — GPU kernels vs the CPU
oracle across every round-2 feature: random schemas/histories (the fuzz
generator), the fast-kernel dispatch, intents merges with random
transaction outcomes, backward delivery, option/bound pruning, grouped
aggregates. Run: python scripts/soak_gpu.py [cases] [seed]."""
import os
import random
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tests"))

import ybgpu as y  # noqa: E402
from gpu_scan import GpuScan  # noqa: E402
import test_fuzz_parity as fz  # noqa: E402
from parity_cases import make_orcl_spec, check_match  # noqa: E402

def run(cases=40, seed=20250915):
    rng = random.Random(seed)
    n_scan = n_intents = n_back = n_group = 0
    for it in range(cases):
        schema, kv_format, built, rows, dtypes = fz._random_case(rng)
        data, offsets, nb, total, ne = built
        for _ in range(2):
            read, preds, aggs = fz._random_spec(rng, len(dtypes), dtypes,
                                                rows)
            spec = y.ScanSpec()
            spec.schema = schema
            spec.kv_format = kv_format
            spec.read_time = y.read_time(read)
            spec.num_preds = len(preds)
            for i, p in enumerate(preds):
                spec.preds[i] = p
            spec.num_aggs = len(aggs)
            for i, a in enumerate(aggs):
                spec.aggs[i] = a
            s = GpuScan(spec)
            s.feed_blocks_host(data, offsets, nb, total)
            s.execute()
            gres = s.aggregates()
            s.close()
            osc = y.orcl_schema_from(schema)
            ospec = make_orcl_spec(read, preds, aggs)
            ores, _ = y.orcl_scan(data, offsets, nb, osc, ospec,
                                  kv_format=kv_format)
            check_match(gres, ores, aggs)
            n_scan += 1
    # intents soak
    import test_intents as ti
    rng2 = random.Random(seed + 1)
    for it in range(max(4, cases // 6)):
        schema, built = ti._base_tablet(rng2.randint(300, 2500))
        intents = y.Intents(schema)
        table = {}
        for t in range(rng2.randint(3, 25)):
            st = rng2.choice(["pending", "aborted", "c", "c"])
            table[t] = ("c", rng2.randint(1200, 4000)) if st == "c" else st
            for _ in range(rng2.randint(1, 5)):
                r = rng2.randint(0, 3000)
                k = rng2.random()
                if k < 0.6:
                    intents.add_packed_row(
                        rng2.randint(1100, 3900), t,
                        [(y.T_INT64, r), (y.T_INT64, r * 3)],
                        hash_=r // 512, key_datums=(r,))
                elif k < 0.8:
                    intents.add_column_update(
                        rng2.randint(1100, 3900), t, 1,
                        rng2.randint(0, 10000), hash_=r // 512,
                        key_datums=(r,))
                else:
                    intents.add_row_tombstone(rng2.randint(1100, 3900), t,
                                              hash_=r // 512,
                                              key_datums=(r,))
        blob, blen = intents.blob()
        txns, ntx = y.make_txns(table)
        for read in (rng2.randint(900, 4500), rng2.randint(900, 4500)):
            local = read + rng2.choice([0, rng2.randint(1, 2000)])
            aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
            spec = ti._spec(schema, read, local, local + 700, (), aggs)
            s = GpuScan(spec)
            s.feed_blocks_intents(data=built[0], offsets=built[1],
                                  n_blocks=built[2], intents_blob=blob,
                                  blob_len=blen, txns=txns, n_txns=ntx)
            s.execute()
            g = s.aggregates()
            restart = s.restart_data()
            s.close()
            osc = y.orcl_schema_from(schema)
            ospec = ti._orcl_spec(read, local, local + 700, (), aggs)
            o = y.orcl_scan_intents(built[0], built[1], built[2], osc,
                                    ospec, blob, blen, txns, ntx)
            assert (g.rows_scanned, g.rows_matched, g.aggs[1].value_i64) \
                == (o.rows_scanned, o.rows_matched, o.aggs[1].value_i64), it
            assert restart == bytes(o.restart_ht[:o.restart_ht_len])
            n_intents += 1
    # bloom point-scan soak: random sorted tablets; random point keys
    # scanned through feed_blocks_bloom must match the oracle over the
    # same bounds (present keys found, filter-rejected keys empty)
    import ctypes as C
    n_bloom = 0
    rng3 = random.Random(seed + 2)
    sc = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    for it in range(max(3, cases // 20)):
        rows = rng3.randint(800, 5000)
        hmod = rng3.choice([53, 211, 997])
        b = y.Builder(sc)
        seq = 1 << 50
        for r in sorted(range(rows), key=lambda r: (r % hmod, r)):
            seq += 1
            b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % hmod,
                             key_datums=(r,), seq=seq)
        data, offsets, nb, total = b.finish()[:4]
        filt = y.filter_from_sst(data, offsets, nb)
        for _ in range(20):
            if rng3.random() < 0.5:
                r = rng3.randrange(rows)
                hh, k0, expect = r % hmod, r, 1
            else:
                hh, k0 = rng3.randrange(8 * hmod), rng3.randrange(4 * rows)
                expect = 1 if (k0 < rows and k0 % hmod == hh) else 0
            lower = y.encode_dockey(sc, hash_=hh, key_datums=(k0,))
            upper = lower + b"\x00"
            spec = y.ScanSpec()
            spec.schema = sc
            spec.kv_format = y.ENC_THREE_SHARED_PARTS
            spec.read_time = y.read_time(9000)
            spec.num_aggs = 1
            spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
            lb = (C.c_uint8 * len(lower)).from_buffer_copy(lower)
            ub = (C.c_uint8 * len(upper)).from_buffer_copy(upper)
            spec.lower_bound, spec.lower_bound_len = lb, len(lower)
            spec.upper_bound, spec.upper_bound_len = ub, len(upper)
            s = GpuScan(spec)
            s.feed_blocks_bloom(data, offsets, nb, filt)
            s.execute()
            g = s.aggregates()
            s.close()
            assert g.rows_matched == expect, (it, hh, k0, g.rows_matched)
            n_bloom += 1
    # SST-file soak: finish_sst with random compression (none / snappy /
    # LZ4) + feed_sst (footer/index parse, device crc32c verify, on-GPU
    # decompression) vs the oracle over the same builder's raw blocks
    n_sst = 0
    rng4 = random.Random(seed + 3)
    sc2 = y.make_schema([y.KT_INT64],
                        [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    for it in range(max(3, cases // 25)):
        rows = rng4.randint(2000, 20000)
        comp = rng4.choice([0, 1, 4])
        b = y.Builder(sc2)
        seq = 1 << 50
        for r in range(rows):
            seq += 1
            b.add_packed_row(5000, [(y.T_INT64, r), (y.T_INT64, r * 7)],
                             hash_=r // 128, key_datums=(r,), seq=seq)
        sst_ptr, sst_total, _, _ = b.finish_sst(compression=comp)
        data2, offsets2, nb2, total2 = b.finish()[:4]
        for _ in range(3):
            thr = rng4.randint(0, rows)
            spec = y.ScanSpec()
            spec.schema = sc2
            spec.kv_format = y.ENC_THREE_SHARED_PARTS
            spec.read_time = y.read_time(9000)
            spec.num_preds = 1
            spec.preds[0] = y.Pred(0, 0, y.PRED_GE, thr, None, 0)
            spec.num_aggs = 2
            spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
            spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 1)
            s = GpuScan(spec)
            s.feed_sst(sst_ptr, sst_total, verify=True)
            s.execute()
            g = s.aggregates()
            s.close()
            osc2 = y.orcl_schema_from(sc2)
            ospec = y.OrclScanSpec()
            ospec.read_time = y.orcl_read_time(9000)
            ospec.num_preds = 1
            ospec.preds[0] = y.OrclPred(0, 0, y.PRED_GE, thr, None, 0)
            ospec.num_aggs = 2
            ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
            ospec.aggs[1] = y.OrclAgg(y.AGG_SUM_INT64, 1)
            o, _ = y.orcl_scan(data2, offsets2, nb2, osc2, ospec)
            assert (g.rows_scanned, g.rows_matched, g.aggs[1].value_i64) \
                == (o.rows_scanned, o.rows_matched,
                    o.aggs[1].value_i64), (it, comp, thr)
            n_sst += 1
    print(f"soak ok: {n_scan} scans, {n_intents} intent merges, "
          f"{n_bloom} bloom point scans, {n_sst} sst-file scans "
          f"(none/snappy/lz4), seed {seed}")

if __name__ == "__main__":
    run(int(sys.argv[1]) if len(sys.argv) > 1 else 40,
        int(sys.argv[2]) if len(sys.argv) > 2 else 20250915)
