"""Randomized differential parity: random schemas / MVCC histories /
scan specs, the host SIMULATOR of the device algorithm vs the CPU oracle.
Seeded (deterministic) and bounded; shapes cover what the reference's own
iterator tests sweep — packed V1/V2 mixes, multi-version rows, row
tombstones, column updates, NULLs, string columns, both KV encodings,
restart intervals 1/4/16, small blocks (rows straddling intervals and
blocks), typed/IN predicates and read-time sweeps."""
import ctypes as C

import pytest
import random
import struct

import ybgpu as y
from parity_cases import make_orcl_spec, check_match

KEEP = []


def _pack_in_list(vals):
    b = struct.pack("<%dQ" % len(vals), *[v & (2**64 - 1) for v in vals])
    arr = (C.c_uint8 * len(b)).from_buffer_copy(b)
    KEEP.append(arr)
    return arr, len(b)


def _random_case(rng):
    nval = rng.randint(1, 6)
    dtypes = [rng.choice([y.T_INT64, y.T_INT64, y.T_INT32, y.T_DOUBLE,
                          y.T_STRING, y.T_BOOL]) for _ in range(nval)]
    nullable = [rng.randint(0, 1) for _ in range(nval)]
    schema = y.make_schema(
        [y.KT_INT64],
        [(10 + i, dtypes[i], nullable[i]) for i in range(nval)])
    kv_format = rng.choice([y.ENC_THREE_SHARED_PARTS, y.ENC_SHARED_PREFIX])
    b = y.Builder(schema, kv_format=kv_format,
                  block_size=rng.choice([512, 1024, 4096]),
                  restart_interval=rng.choice([1, 4, 16]))
    rows = rng.randint(50, 1200)
    base_ht = 1_000_000
    seq = 1 << 50
    hash_div = rng.choice([16, 64, 256])
    for r in range(rows):
        hash_ = r // hash_div
        versions = rng.randint(1, 3)
        hts = sorted(rng.sample(range(base_ht, base_ht + 5000), versions),
                     reverse=True)
        if rng.random() < 0.05:
            seq += 1
            b.add_row_tombstone(base_ht + 6000, hash_=hash_, key_datums=(r,),
                                seq=seq + 1000)
        for ht in hts:
            vals = []
            for i in range(nval):
                if nullable[i] and rng.random() < 0.2:
                    vals.append((dtypes[i], None))
                elif dtypes[i] == y.T_STRING:
                    vals.append((dtypes[i],
                                 b"s%03d-%d" % (rng.randint(0, 200), r)))
                elif dtypes[i] == y.T_DOUBLE:
                    vals.append((dtypes[i], float(rng.randint(-500, 500))))
                elif dtypes[i] == y.T_BOOL:
                    vals.append((dtypes[i], rng.randint(0, 1)))
                else:
                    vals.append((dtypes[i], rng.randint(-10_000, 10_000)))
            seq += 1
            b.add_packed_row(ht, vals, hash_=hash_, key_datums=(r,),
                             packed_version=rng.choice([1, 2]), seq=seq)
        # column updates sort after the bare-row entries
        for _ in range(rng.randint(0, 2)):
            i = rng.randrange(nval)
            if dtypes[i] == y.T_STRING:
                continue
            seq += 1
            val = rng.randint(-10_000, 10_000)
            if dtypes[i] == y.T_DOUBLE:
                val = float(val)
            null = nullable[i] and rng.random() < 0.2
            b.add_column_update(rng.randint(base_ht, base_ht + 7000), i,
                                None if null else val, hash_=hash_,
                                key_datums=(r,), seq=seq, null=null)
    KEEP.append(b)
    return schema, kv_format, b.finish(), rows, dtypes


def _random_spec(rng, nval, dtypes, rows):
    preds = []
    for _ in range(rng.randint(0, 3)):
        i = rng.randrange(nval)
        if dtypes[i] == y.T_STRING:
            continue
        if dtypes[i] == y.T_DOUBLE:
            d = struct.unpack(
                "<Q", struct.pack("<d", float(rng.randint(-500, 500))))[0]
        else:
            d = rng.randint(-10_000, 10_000) & (2**64 - 1)
        op = rng.choice([y.PRED_GT, y.PRED_GE, y.PRED_LT, y.PRED_LE,
                         y.PRED_EQ, y.PRED_NE, y.PRED_IN])
        if op == y.PRED_IN:
            if dtypes[i] == y.T_DOUBLE:
                continue
            arr, ln = _pack_in_list(
                [rng.randint(-10_000, 10_000) for _ in range(5)])
            preds.append(y.Pred(0, i, op, 0, arr, ln))
        else:
            preds.append(y.Pred(0, i, op, d, None, 0))
    if rng.random() < 0.3:
        preds.append(y.Pred(1, 0, rng.choice([y.PRED_LT, y.PRED_GE]),
                            rng.randint(0, rows), None, 0))
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0)]
    for _ in range(rng.randint(0, 3)):
        i = rng.randrange(nval)
        if dtypes[i] in (y.T_INT64, y.T_INT32, y.T_BOOL):
            aggs.append(y.Agg(rng.choice(
                [y.AGG_SUM_INT64, y.AGG_MIN_INT64, y.AGG_MAX_INT64,
                 y.AGG_COUNT]), i))
        elif dtypes[i] == y.T_DOUBLE:
            aggs.append(y.Agg(rng.choice(
                [y.AGG_SUM_DOUBLE, y.AGG_MIN_DOUBLE, y.AGG_MAX_DOUBLE]), i))
        else:
            aggs.append(y.Agg(y.AGG_COUNT, i))
    read = rng.choice([999_000, 1_002_000, 1_004_000, 1_008_000, 2_000_000])
    return read, preds[:y.MAX_PREDS], aggs[:y.MAX_AGGS]


def test_fuzz_sim_vs_oracle():
    rng = random.Random(20260915)
    for it in range(40):
        schema, kv_format, built, rows, dtypes = _random_case(rng)
        data, offsets, nb, total, ne = built
        for run in range(3):
            read, preds, aggs = _random_spec(rng, len(dtypes), dtypes, rows)
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
            sres = y.sim_scan(spec, data, offsets, nb)

            osc = y.orcl_schema_from(schema)
            ospec = make_orcl_spec(read, preds, aggs)
            ores, _ = y.orcl_scan(data, offsets, nb, osc, ospec,
                                  kv_format=kv_format)
            try:
                check_match(sres, ores, aggs)
            except AssertionError as e:
                raise AssertionError(
                    f"fuzz iter {it} run {run} (rows={rows}, "
                    f"dtypes={dtypes}, read={read}): {e}") from e


@pytest.mark.gpu
def test_fuzz_gpu_vs_oracle():
    """The same randomized differential check against the REAL kernels:
    12 random cases x 2 specs, GPU scan vs oracle."""
    from gpu_scan import GpuScan

    rng = random.Random(77)
    for it in range(12):
        schema, kv_format, built, rows, dtypes = _random_case(rng)
        data, offsets, nb, total, ne = built
        for run in range(2):
            read, preds, aggs = _random_spec(rng, len(dtypes), dtypes, rows)
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
            try:
                check_match(gres, ores, aggs)
            except AssertionError as e:
                raise AssertionError(
                    f"gpu fuzz iter {it} run {run} (rows={rows}, "
                    f"dtypes={dtypes}, read={read}): {e}") from e


def test_fuzz_emit_vs_oracle_rows():
    """Row materialization (the PgFetchNext surface) on random cases: the
    simulator's emitted rows vs the oracle's collected rows."""
    rng = random.Random(5150)
    for it in range(12):
        schema, kv_format, built, rows, dtypes = _random_case(rng)
        data, offsets, nb, total, ne = built
        read, preds, _ = _random_spec(rng, len(dtypes), dtypes, rows)
        spec = y.ScanSpec()
        spec.schema = schema
        spec.kv_format = kv_format
        spec.read_time = y.read_time(read)
        spec.num_preds = len(preds)
        for i, p in enumerate(preds):
            spec.preds[i] = p
        got = y.sim_emit(spec, data, offsets, nb)

        osc = y.orcl_schema_from(schema)
        ospec = make_orcl_spec(read, preds, ())
        _, want = y.orcl_scan(data, offsets, nb, osc, ospec,
                              kv_format=kv_format, collect_rows=True)
        assert len(got) == len(want), (it, len(got), len(want))
        for g, w in zip(got, want):
            assert g == w, (it, g, w)


def test_fuzz_group_vs_oracle():
    """GROUP BY partial aggregates on random cases (int64 group keys,
    COUNT/SUM/MIN/MAX int64)."""
    rng = random.Random(8086)
    for it in range(12):
        schema, kv_format, built, rows, dtypes = _random_case(rng)
        int_cols = [i for i, d in enumerate(dtypes)
                    if d in (y.T_INT64, y.T_INT32)]
        if not int_cols:
            continue
        gcol = rng.choice(int_cols)
        data, offsets, nb, total, ne = built
        read, preds, _ = _random_spec(rng, len(dtypes), dtypes, rows)
        aggs = [y.Agg(y.AGG_COUNT_STAR, 0),
                y.Agg(rng.choice([y.AGG_SUM_INT64, y.AGG_MIN_INT64,
                                  y.AGG_MAX_INT64]), gcol)]
        spec = y.ScanSpec()
        spec.schema = schema
        spec.kv_format = kv_format
        spec.read_time = y.read_time(read)
        spec.group_col = 1 + gcol
        spec.num_preds = len(preds)
        for i, p in enumerate(preds):
            spec.preds[i] = p
        spec.num_aggs = len(aggs)
        for i, a in enumerate(aggs):
            spec.aggs[i] = a
        got = y.sim_group(spec, data, offsets, nb)

        osc = y.orcl_schema_from(schema)
        ospec = make_orcl_spec(read, preds, aggs)
        want = y.orcl_group(data, offsets, nb, osc, ospec, gcol,
                            kv_format=kv_format, num_aggs=2,
                            aggs=ospec.aggs)
        assert got == want, (it, gcol,
                             {k: got[k] for k in list(got)[:3]},
                             {k: want[k] for k in list(want)[:3]})
