"""Shared parity scenarios: each case builds a tablet (via the product
generator/builder) and a list of scan runs. Consumed by
 - test_sim_parity.py  (CPU: host simulator of the device algorithm vs oracle)
 - test_gpu_parity.py  (GPU: HIP kernels vs oracle)
"""
import ctypes as C

import ybgpu as y

SCHEMA_4I = y.make_schema([y.KT_INT64],
                          [(10 + i, y.T_INT64, 1) for i in range(4)])

_KEEP = []


def _case(name, schema, built, runs, kv_format=y.ENC_THREE_SHARED_PARTS):
    data, offsets, nb, total = built[0], built[1], built[2], built[3]
    return {
        "name": name,
        "schema": schema,
        "data": data,
        "offsets": offsets,
        "n_blocks": nb,
        "total": total,
        "runs": runs,
        "kv_format": kv_format,
    }


def build_cases():
    cases = []

    # config #2 shape: 3 int64 predicates + aggregates
    built = y.generate(SCHEMA_4I, rows=200_000, seed=42)
    preds = [y.Pred(0, 0, y.PRED_GT, 1 << 39, None, 0),
             y.Pred(0, 1, y.PRED_LT, 3 << 38, None, 0),
             y.Pred(0, 2, y.PRED_GE, 1 << 36, None, 0)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 3),
            y.Agg(y.AGG_MIN_INT64, 3), y.Agg(y.AGG_MAX_INT64, 2)]
    cases.append(_case("config2_filtered_sum", SCHEMA_4I, built,
                       [(1_700_000_000_000_000, preds, aggs)]))

    # config #4 shape: MVCC 5 versions, read times at version boundaries
    built = y.generate(SCHEMA_4I, rows=20_000, versions=5,
                       ht_base_micros=1_600_000_000_000_000,
                       ht_step_micros=1_000_000_000)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0)]
    cases.append(_case("mvcc_5_versions", SCHEMA_4I, built,
                       [(rm, (), aggs) for rm in
                        (1_599_999_999_000_000, 1_600_000_500_000_000,
                         1_602_000_000_500_000, 1_700_000_000_000_000)]))

    # shared_prefix encoding
    built = y.generate(SCHEMA_4I, rows=50_000, kv_format=y.ENC_SHARED_PREFIX)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    cases.append(_case("shared_prefix_fmt", SCHEMA_4I, built,
                       [(1_700_000_000_000_000, (), aggs)],
                       kv_format=y.ENC_SHARED_PREFIX))

    # packed V1
    built = y.generate(SCHEMA_4I, rows=50_000, packed_version=1)
    aggs = [y.Agg(y.AGG_COUNT, 2), y.Agg(y.AGG_SUM_INT64, 3)]
    cases.append(_case("packed_v1", SCHEMA_4I, built,
                       [(1_700_000_000_000_000, (), aggs)]))

    # config #5 shape: mixed types
    schema_mix = y.make_schema(
        [y.KT_INT64],
        [(10, y.T_INT64, 1), (11, y.T_DOUBLE, 1), (12, y.T_STRING, 1)])
    built = y.generate(schema_mix, rows=100_000)
    preds = [y.Pred(0, 0, y.PRED_GT, 1 << 38, None, 0),
             y.Pred(0, 0, y.PRED_LT, 1 << 39, None, 0)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_DOUBLE, 1),
            y.Agg(y.AGG_MIN_DOUBLE, 1), y.Agg(y.AGG_MAX_DOUBLE, 1)]
    cases.append(_case("mixed_types", schema_mix, built,
                       [(1_700_000_000_000_000, preds, aggs)]))

    # string equality predicate
    schema_s = y.make_schema([y.KT_INT64], [(10, y.T_STRING, 1),
                                            (11, y.T_INT64, 1)])
    b = y.Builder(schema_s)
    target = b"hello-world"
    for r in range(5000):
        s = target if r % 7 == 0 else b"other-%05d" % r
        b.add_packed_row(1000 + r, [(y.T_STRING, s), (y.T_INT64, r)],
                         hash_=r // 64, key_datums=(r,))
    buf = C.create_string_buffer(target, len(target))
    _KEEP.append((b, buf))
    pred = y.Pred(0, 0, y.PRED_EQ, 0, C.cast(buf, C.POINTER(C.c_uint8)),
                  len(target))
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    cases.append(_case("string_eq_pred", schema_s, b.finish(),
                       [(1_700_000_000_000_000, [pred], aggs)]))

    # column updates / tombstones (docrowwiseiterator-test.cc scenarios)
    schema_cu = y.make_schema([y.KT_INT64], [(30, y.T_INT64, 1),
                                             (40, y.T_INT64, 1)])
    b = y.Builder(schema_cu)
    b.add_packed_row(1000, [(y.T_INT64, 1), (y.T_INT64, 2)], hash_=0,
                     key_datums=(0,))
    b.add_column_update(2000, 1, 222, hash_=0, key_datums=(0,))
    b.add_row_tombstone(2500, hash_=1, key_datums=(1,), seq=(1 << 50) + 10)
    b.add_packed_row(1000, [(y.T_INT64, 3), (y.T_INT64, 4)], hash_=1,
                     key_datums=(1,), seq=(1 << 50) + 5)
    b.add_row_tombstone(2500, hash_=2, key_datums=(2,), seq=(1 << 50) + 20)
    b.add_packed_row(1000, [(y.T_INT64, 5), (y.T_INT64, 6)], hash_=2,
                     key_datums=(2,), seq=(1 << 50) + 15)
    b.add_column_update(3000, 0, 555, hash_=2, key_datums=(2,),
                        seq=(1 << 50) + 25)
    b.add_column_update(1000, 0, 7, hash_=3, key_datums=(3,))
    b.add_column_update(1500, 1, 8, hash_=3, key_datums=(3,))
    b.add_packed_row(1000, [(y.T_INT64, 9), (y.T_INT64, 10)], hash_=4,
                     key_datums=(4,))
    b.add_column_update(3000, 1, 333, hash_=4, key_datums=(4,),
                        seq=(1 << 50) + 40)
    b.add_column_update(2500, 1, None, hash_=4, key_datums=(4,),
                        seq=(1 << 50) + 38, null=True)
    _KEEP.append(b)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0),
            y.Agg(y.AGG_SUM_INT64, 1), y.Agg(y.AGG_COUNT, 1)]
    cases.append(_case("column_updates_tombstones", schema_cu, b.finish(),
                       [(rm, (), aggs) for rm in (1200, 2200, 2600, 3200,
                                                  5000)]))

    # NULLs in packed rows, both versions interleaved
    schema_n = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1),
                                            (11, y.T_INT64, 1)])
    b = y.Builder(schema_n)
    for r in range(4000):
        v0 = None if r % 3 == 0 else r
        v1 = None if r % 5 == 0 else r * 2
        pv = 1 if r % 2 == 0 else 2
        b.add_packed_row(1000 + r, [(y.T_INT64, v0), (y.T_INT64, v1)],
                         hash_=r // 64, key_datums=(r,), packed_version=pv)
    _KEEP.append(b)
    aggs = [y.Agg(y.AGG_COUNT, 0), y.Agg(y.AGG_COUNT, 1),
            y.Agg(y.AGG_SUM_INT64, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    cases.append(_case("nulls_packed", schema_n, b.finish(),
                       [(1_000_000, (), aggs)]))

    # wide all-fixed schema (32 int64 columns): the fixed-offset packed-V2
    # fast path at its column-count and offset bounds (v2_off up to 251)
    schema_w = y.make_schema([y.KT_INT64],
                             [(100 + i, y.T_INT64, 0) for i in range(32)])
    b = y.Builder(schema_w)
    for r in range(1500):
        b.add_packed_row(1000 + r,
                         [(y.T_INT64, r * 31 + c) for c in range(32)],
                         hash_=r // 64, key_datums=(r,), packed_version=2)
    _KEEP.append(b)
    aggs = [y.Agg(y.AGG_SUM_INT64, 0), y.Agg(y.AGG_SUM_INT64, 31),
            y.Agg(y.AGG_COUNT_STAR, 0)]
    preds = [y.Pred(0, 16, y.PRED_GT, 20_000, None, 0)]
    cases.append(_case("wide_fixed_32cols", schema_w, b.finish(),
                       [(1_000_000, (), aggs), (1_000_000, preds, aggs)]))

    # wide nullable rows: per-row null masks force the window/pointer
    # decoders off the fixed-offset path for exactly the rows with nulls
    schema_wn = y.make_schema([y.KT_INT64],
                              [(100 + i, y.T_INT64, 1) for i in range(12)])
    b = y.Builder(schema_wn)
    for r in range(3000):
        vals = [(y.T_INT64, None if (r + c) % 7 == 0 else r + c)
                for c in range(12)]
        b.add_packed_row(1000 + r, vals, hash_=r // 64, key_datums=(r,),
                         packed_version=2)
    _KEEP.append(b)
    aggs = [y.Agg(y.AGG_COUNT, 3), y.Agg(y.AGG_SUM_INT64, 11)]
    cases.append(_case("wide_nullable_mixed", schema_wn, b.finish(),
                       [(1_000_000, (), aggs)]))

    # key-column predicate
    built = y.generate(SCHEMA_4I, rows=30_000)
    preds = [y.Pred(1, 0, y.PRED_LT, 10_000, None, 0)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0)]
    cases.append(_case("key_pred", SCHEMA_4I, built,
                       [(1_700_000_000_000_000, preds, aggs)]))

    # IN-list predicates (hybrid-scan option filters): value column and key
    # column membership over packed 8-byte LE datum lists
    import ctypes as _C
    import struct as _struct
    built = y.generate(SCHEMA_4I, rows=25_000)
    # real col-0 datums (oracle row collection) + junk -> a list that both
    # matches and misses
    _osc = y.orcl_schema_from(SCHEMA_4I)
    _ospec = make_orcl_spec(1_700_000_000_000_000, (), [])
    _, _rows = y.orcl_scan(built[0], built[1], built[2], _osc, _ospec,
                           collect_rows=True)
    vals = sorted({int(r[1][0]) for r in _rows[:4000:97]
                   if r[1][0] is not None} | {0, 12345678901234})
    inbuf1 = _struct.pack("<%dQ" % len(vals), *[v & (2**64 - 1) for v in vals])
    a1 = (_C.c_uint8 * len(inbuf1)).from_buffer_copy(inbuf1)
    keys = list(range(100, 24_000, 311))
    inbuf2 = _struct.pack("<%dq" % len(keys), *keys)
    a2 = (_C.c_uint8 * len(inbuf2)).from_buffer_copy(inbuf2)
    _KEEP.append(a1)
    _KEEP.append(a2)
    preds_v = [y.Pred(0, 0, y.PRED_IN, 0, a1, len(inbuf1))]
    preds_k = [y.Pred(1, 0, y.PRED_IN, 0, a2, len(inbuf2))]
    preds_both = [y.Pred(0, 0, y.PRED_IN, 0, a1, len(inbuf1)),
                  y.Pred(1, 0, y.PRED_IN, 0, a2, len(inbuf2))]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 3)]
    cases.append(_case("in_list_preds", SCHEMA_4I, built,
                       [(1_700_000_000_000_000, preds_v, aggs),
                        (1_700_000_000_000_000, preds_k, aggs),
                        (1_700_000_000_000_000, preds_both, aggs)]))

    # string IN options: value column (plain bytes) and string KEY column
    # (zero-escaped in the rowkey — the escape-aware compare path)
    schema_si = y.make_schema([y.KT_INT64, y.KT_STRING],
                              [(10, y.T_STRING, 1), (11, y.T_INT64, 1)],
                              num_hash_cols=1)
    b = y.Builder(schema_si)
    for r in range(4000):
        sval = b"val-%04d" % (r % 50)
        skey = (b"k\x00z-%03d" % (r % 97)) if r % 3 == 0 else \
            (b"key-%04d" % r)
        b.add_packed_row(1000 + r, [(y.T_STRING, sval), (y.T_INT64, r)],
                         hash_=r // 64, key_datums=(r, 0),
                         key_strs=(None, skey))
    _KEEP.append(b)

    def _strlist(opts):
        out = b"".join(_struct.pack("<I", len(o)) + o for o in opts)
        arr = (_C.c_uint8 * len(out)).from_buffer_copy(out)
        _KEEP.append(arr)
        return arr, len(out)

    a3, l3 = _strlist([b"val-0003", b"val-0017", b"no-such"])
    a4, l4 = _strlist([b"k\x00z-005", b"key-0100", b"absent"])
    preds_sv = [y.Pred(0, 0, y.PRED_IN, 0, a3, l3)]
    preds_sk = [y.Pred(1, 1, y.PRED_IN, 0, a4, l4)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    cases.append(_case("in_list_string", schema_si, b.finish(),
                       [(1_700_000_000_000_000, preds_sv, aggs),
                        (1_700_000_000_000_000, preds_sk, aggs)]))

    # multi-column tuple options ((k0,k1) IN ((3,103),(7,257),(3,9999)) —
    # the reference's multi-column option groups)
    schema_t2 = y.make_schema([y.KT_INT64, y.KT_INT64],
                              [(10, y.T_INT64, 1)], num_hash_cols=1)
    b = y.Builder(schema_t2)
    for r in range(4000):
        b.add_packed_row(1000 + r, [(y.T_INT64, r * 3)],
                         hash_=r // 64, key_datums=(r % 50, r))
    _KEEP.append(b)
    tuples = [(3, 103), (7, 257), (3, 9999)]
    tb = _struct.pack("<I", 2) + _struct.pack("<II", 0, 1)
    for t2 in tuples:
        tb += _struct.pack("<QQ", t2[0] & (2**64 - 1), t2[1] & (2**64 - 1))
    at = (_C.c_uint8 * len(tb)).from_buffer_copy(tb)
    _KEEP.append(at)
    preds_t = [y.Pred(1, 0, y.PRED_IN_TUPLE, 0, at, len(tb))]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0)]
    cases.append(_case("in_tuple_keycols", schema_t2, b.finish(),
                       [(1_700_000_000_000_000, preds_t, aggs)]))

    # single row + empty result
    b = y.Builder(SCHEMA_4I)
    b.add_packed_row(1000, [(y.T_INT64, i) for i in range(4)], hash_=7,
                     key_datums=(123,))
    _KEEP.append(b)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 2)]
    cases.append(_case("single_row", SCHEMA_4I, b.finish(),
                       [(5000, (), aggs), (500, (), aggs)]))

    # rows straddling interval and block boundaries (23 versions, 1KB blocks)
    built = y.generate(SCHEMA_4I, rows=3_000, versions=23, block_size=1024,
                       ht_base_micros=1_600_000_000_000_000,
                       ht_step_micros=1_000)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0)]
    cases.append(_case("spanning_rows", SCHEMA_4I, built,
                       [(rm, (), aggs) for rm in
                        (1_600_000_000_000_005, 1_600_000_000_010_000,
                         1_700_000_000_000_000)]))

    # V1 packed rows with NON-NULLABLE fixed columns (truly fixed layout,
    # schema_packing.cc:45-49: only nullable/string columns are varlen)
    schema_nn = y.make_schema(
        [y.KT_INT64],
        [(10, y.T_INT64, 0), (11, y.T_INT32, 0), (12, y.T_STRING, 1),
         (13, y.T_INT64, 1)])
    b = y.Builder(schema_nn)
    for r in range(3000):
        b.add_packed_row(1000 + r,
                         [(y.T_INT64, r * 3), (y.T_INT32, r % 1000),
                          (y.T_STRING, b"s%04d" % (r % 50)),
                          (y.T_INT64, None if r % 4 == 0 else r)],
                         hash_=r // 64, key_datums=(r,),
                         packed_version=1)
    _KEEP.append(b)
    aggs = [y.Agg(y.AGG_SUM_INT64, 0), y.Agg(y.AGG_SUM_INT64, 1),
            y.Agg(y.AGG_COUNT, 3)]
    cases.append(_case("v1_nonnullable_fixed", schema_nn, b.finish(),
                       [(1_000_000, (), aggs)]))

    # non-default restart interval and the reference's default 32KB blocks
    built = y.generate(SCHEMA_4I, rows=40_000, restart_interval=4,
                       block_size=32768)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 2)]
    cases.append(_case("restart4_block32k", SCHEMA_4I, built,
                       [(1_700_000_000_000_000, (), aggs)]))

    # shared_prefix encoding + MVCC versions together
    built = y.generate(SCHEMA_4I, rows=5_000, versions=3,
                       kv_format=y.ENC_SHARED_PREFIX,
                       ht_base_micros=1_600_000_000_000_000,
                       ht_step_micros=1_000_000_000)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0)]
    cases.append(_case("shared_prefix_mvcc", SCHEMA_4I, built,
                       [(rm, (), aggs) for rm in
                        (1_600_000_500_000_000, 1_700_000_000_000_000)],
                       kv_format=y.ENC_SHARED_PREFIX))

    # bounds: lower inclusive / upper exclusive on encoded rowkey
    # (qlexpr/ql_scanspec.h:200-267 bounds; generator rows r have
    # hash = r*65536//rows, key col = r)
    built = y.generate(SCHEMA_4I, rows=30_000)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0)]

    def dockey(rows, r):
        h = (r * 65536) // rows
        return (b"G" + h.to_bytes(2, "big") + b"I" +
                ((r ^ (1 << 63)).to_bytes(8, "big")) + b"!!")

    lo = dockey(30_000, 5_000)
    hi = dockey(30_000, 25_000)
    lo_buf = C.create_string_buffer(lo, len(lo))
    hi_buf = C.create_string_buffer(hi, len(hi))
    _KEEP.append((lo_buf, hi_buf))
    cases.append(_case("bounded", SCHEMA_4I, built,
                       [(1_700_000_000_000_000, (), aggs,
                         (lo_buf, len(lo)), (hi_buf, len(hi)))]))
    return cases


def make_spec(case, read_micros, preds, aggs, lower=None, upper=None):
    spec = y.ScanSpec()
    spec.schema = case["schema"]
    spec.kv_format = case["kv_format"]
    spec.read_time = y.read_time(read_micros)
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = p
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = a
    if lower:
        spec.lower_bound = C.cast(lower[0], C.POINTER(C.c_uint8))
        spec.lower_bound_len = lower[1]
    if upper:
        spec.upper_bound = C.cast(upper[0], C.POINTER(C.c_uint8))
        spec.upper_bound_len = upper[1]
    return spec


def make_orcl_spec(read_micros, preds, aggs, lower=None, upper=None):
    spec = y.OrclScanSpec()
    spec.read_time = y.orcl_read_time(read_micros)
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = y.OrclPred(p.is_key_col, p.col, p.op, p.datum,
                                   p.bytes, p.bytes_len)
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = y.OrclAgg(a.op, a.col)
    if lower:
        spec.lower_bound = C.cast(lower[0], C.POINTER(C.c_uint8))
        spec.lower_bound_len = lower[1]
    if upper:
        spec.upper_bound = C.cast(upper[0], C.POINTER(C.c_uint8))
        spec.upper_bound_len = upper[1]
    return spec


def run_oracle(case, read_micros, preds, aggs, lower=None, upper=None):
    osc = y.orcl_schema_from(case["schema"])
    ospec = make_orcl_spec(read_micros, preds, aggs, lower, upper)
    res, _ = y.orcl_scan(case["data"], case["offsets"], case["n_blocks"],
                         osc, ospec, kv_format=case["kv_format"])
    return res


def check_match(gres, ores, aggs, f64_rel=1e-12, check_entries=True):
    if check_entries:
        # the oracle stops at the upper bound like the reference iterator;
        # the GPU filters instead, so entries_seen is compared only for
        # unbounded scans.
        assert gres.entries_seen == ores.entries_seen, \
            (gres.entries_seen, ores.entries_seen)
    assert gres.rows_scanned == ores.rows_scanned, \
        (gres.rows_scanned, ores.rows_scanned)
    assert gres.rows_matched == ores.rows_matched, \
        (gres.rows_matched, ores.rows_matched)
    for i, a in enumerate(aggs):
        assert gres.aggs[i].is_null == ores.aggs[i].is_null, i
        if ores.aggs[i].is_null:
            continue
        if a.op in (y.AGG_SUM_DOUBLE, y.AGG_MIN_DOUBLE, y.AGG_MAX_DOUBLE):
            gv, ov = gres.aggs[i].value_f64, ores.aggs[i].value_f64
            assert abs(gv - ov) <= f64_rel * max(1.0, abs(ov)), (i, gv, ov)
        else:
            assert gres.aggs[i].value_i64 == ores.aggs[i].value_i64, \
                (i, gres.aggs[i].value_i64, ores.aggs[i].value_i64)
