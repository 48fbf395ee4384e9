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
    print(f"soak ok: {n_scan} scans, {n_intents} intent merges, "
          f"seed {seed}")

if __name__ == "__main__":
    run(int(sys.argv[1]) if len(sys.argv) > 1 else 40,
        int(sys.argv[2]) if len(sys.argv) > 2 else 20250915)
