"""Index-guided block pruning + option RANGES (SURVEY §8f-1/2,
docdb/hybrid_scan_choices.cc seek plans, rocksdb index-based block
selection): bounded or option-constrained scans only feed the blocks
whose key range intersects the allowed key set; IN_RANGE adds the
reference's mixed bound options as a filter. Pruning is a superset
selection — results must be identical to unpruned scans and the oracle."""
import ctypes as C
import struct

import pytest

import ybgpu as y


def _range_tablet(rows=20000):
    # range-sharded (no hash): leading int64 range key drives the seek plan
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_INT64, 1)],
                           has_hash=False)
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(rows):
        seq += 1
        b.add_packed_row(1000, [(y.T_INT64, r), (y.T_INT64, r * 3)],
                         key_datums=(r,), seq=seq)
    return schema, b.finish()


def _in_range_bytes(ranges):
    """ranges: list of (lo, hi, lo_incl, hi_incl) int64."""
    blob = b"".join(
        struct.pack("<qqII", lo, hi, (1 if li else 0) | (2 if hi_i else 0), 0)
        for lo, hi, li, hi_i in ranges)
    buf = C.create_string_buffer(blob, len(blob))
    return C.cast(buf, C.POINTER(C.c_uint8)), len(blob), buf


def _spec(schema, preds, aggs, read=5000):
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(read)
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = p
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = a
    return spec


def _sel(spec, built):
    lib = y.product()
    f = y._sig(lib, "ybg_test_block_selection", C.c_int,
               [C.POINTER(y.ScanSpec), C.POINTER(C.c_uint8),
                C.POINTER(C.c_uint64), C.c_uint64, C.POINTER(C.c_uint8)])
    keep = (C.c_uint8 * built[2])()
    rc = f(C.byref(spec), built[0], built[1], built[2], keep)
    return rc, bytes(keep[:built[2]])


def _compact(built, keep):
    data = bytes(built[0][0:built[3]])
    offs = [built[1][i] for i in range(built[2] + 1)]
    blob = b""
    no = [0]
    for b in range(built[2]):
        if keep[b]:
            blob += data[offs[b]:offs[b + 1]]
            no.append(len(blob))
    d = (C.c_uint8 * len(blob)).from_buffer_copy(blob)
    o = (C.c_uint64 * len(no))(*no)
    return d, o, len(no) - 1


def _oracle_run(schema, built, preds, aggs, read=5000):
    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(read)
    ospec.num_preds = len(preds)
    for i, p in enumerate(preds):
        ospec.preds[i] = y.OrclPred(p.is_key_col, p.col, p.op, p.datum,
                                    p.bytes, p.bytes_len)
    ospec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        ospec.aggs[i] = y.OrclAgg(a.op, a.col)
    return y.orcl_scan(built[0], built[1], built[2], osc, ospec)[0]


def test_in_range_filter_parity():
    """IN_RANGE as a pure filter (value column): sim vs oracle."""
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    data, offsets, nb, total, ne = y.generate(schema, rows=30000, seed=5)
    built = (data, offsets, nb, total)
    ptr, ln, _keep = _in_range_bytes([(1 << 38, 1 << 39, True, False),
                                      (3 << 39, 2**40, False, True)])
    preds = [y.Pred(0, 0, y.PRED_IN_RANGE, 0, ptr, ln)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0)]
    spec = _spec(schema, preds, aggs, read=1_700_000_000_000_000)
    sres = y.sim_scan(spec, data, offsets, nb)
    ores = _oracle_run(schema, built, preds, aggs,
                       read=1_700_000_000_000_000)
    assert sres.rows_matched == ores.rows_matched > 0
    assert sres.aggs[1].value_i64 == ores.aggs[1].value_i64
    assert sres.rows_matched < sres.rows_scanned


def test_in_range_double_column():
    """IN_RANGE over a DOUBLE column (FP range semantics)."""
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_DOUBLE, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(4000):
        seq += 1
        b.add_packed_row(1000, [(y.T_DOUBLE, r * 0.25), (y.T_INT64, r)],
                         hash_=r // 512, key_datums=(r,), seq=seq)
    built = b.finish()
    lo = struct.unpack("<Q", struct.pack("<d", 100.0))[0]
    hi = struct.unpack("<Q", struct.pack("<d", 150.0))[0]
    blob = struct.pack("<QQII", lo, hi, 1, 0)  # [100.0, 150.0)
    buf = C.create_string_buffer(blob, len(blob))
    preds = [y.Pred(0, 0, y.PRED_IN_RANGE, 0,
                    C.cast(buf, C.POINTER(C.c_uint8)), len(blob))]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    spec = _spec(schema, preds, aggs)
    sres = y.sim_scan(spec, built[0], built[1], built[2])
    ores = _oracle_run(schema, (built[0], built[1], built[2]), preds, aggs)
    # rows with 100.0 <= r*0.25 < 150.0: r in [400, 600)
    assert sres.rows_matched == ores.rows_matched == 200
    assert sres.aggs[1].value_i64 == ores.aggs[1].value_i64 \
        == sum(range(400, 600))


def test_option_pruning_selective_scan():
    """<1%-selectivity IN options on the leading range key: most blocks
    are pruned; scanning only the kept subset is bit-exact with the
    oracle's full scan."""
    schema, built = _range_tablet(20000)
    opts = [5, 9177, 19998]
    blob = b"".join(struct.pack("<q", v) for v in opts)
    buf = C.create_string_buffer(blob, len(blob))
    preds = [y.Pred(1, 0, y.PRED_IN,
                    0, C.cast(buf, C.POINTER(C.c_uint8)), len(blob))]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    spec = _spec(schema, preds, aggs)
    rc, keep = _sel(spec, built)
    assert rc == 1
    kept = sum(keep)
    assert kept <= 6, f"kept {kept} of {built[2]} blocks"
    d, o, n = _compact(built, keep)
    sres = y.sim_scan(spec, d, o, n)
    ores = _oracle_run(schema, built, preds, aggs)
    assert sres.rows_matched == ores.rows_matched == 3
    assert sres.aggs[1].value_i64 == ores.aggs[1].value_i64 \
        == sum(v * 3 for v in opts)


def test_range_option_pruning():
    """IN_RANGE on the leading range key prunes to the covering blocks
    and filters exactly."""
    schema, built = _range_tablet(20000)
    ptr, ln, _k = _in_range_bytes([(100, 200, True, True),
                                   (15000, 15050, True, False)])
    preds = [y.Pred(1, 0, y.PRED_IN_RANGE, 0, ptr, ln)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    spec = _spec(schema, preds, aggs)
    rc, keep = _sel(spec, built)
    assert rc == 1 and 0 < sum(keep) < built[2] // 4
    d, o, n = _compact(built, keep)
    sres = y.sim_scan(spec, d, o, n)
    ores = _oracle_run(schema, built, preds, aggs)
    assert sres.rows_matched == ores.rows_matched == 101 + 50
    assert sres.aggs[1].value_i64 == ores.aggs[1].value_i64


def test_bounds_pruning_matches_unpruned():
    """DocKey bounds prune; pruned-subset scan == full-scan results."""
    schema, built = _range_tablet(8000)
    lo = y.encode_dockey(schema, key_datums=(2000,))
    hi = y.encode_dockey(schema, key_datums=(2500,))
    lob = C.create_string_buffer(lo, len(lo))
    hib = C.create_string_buffer(hi, len(hi))
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    spec = _spec(schema, (), aggs)
    spec.lower_bound = C.cast(lob, C.POINTER(C.c_uint8))
    spec.lower_bound_len = len(lo)
    spec.upper_bound = C.cast(hib, C.POINTER(C.c_uint8))
    spec.upper_bound_len = len(hi)
    rc, keep = _sel(spec, built)
    assert rc == 1 and 0 < sum(keep) < built[2]
    d, o, n = _compact(built, keep)
    sres = y.sim_scan(spec, d, o, n)
    full = y.sim_scan(spec, built[0], built[1], built[2])
    assert sres.rows_matched == full.rows_matched == 500
    assert sres.aggs[1].value_i64 == full.aggs[1].value_i64
    assert sres.entries_seen < full.entries_seen


@pytest.mark.gpu
def test_gpu_pruned_option_scan():
    """The ABI feed path prunes; device results match the oracle full
    scan and the scan touches far fewer entries."""
    from gpu_scan import GpuScan
    schema, built = _range_tablet(20000)
    opts = [5, 9177, 19998]
    blob = b"".join(struct.pack("<q", v) for v in opts)
    buf = C.create_string_buffer(blob, len(blob))
    preds = [y.Pred(1, 0, y.PRED_IN,
                    0, C.cast(buf, C.POINTER(C.c_uint8)), len(blob))]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    spec = _spec(schema, preds, aggs)
    s = GpuScan(spec)
    s.feed_blocks_host(built[0], built[1], built[2], built[3])
    s.execute()
    g = s.aggregates()
    s.close()
    ores = _oracle_run(schema, built, preds, aggs)
    assert g.rows_matched == ores.rows_matched == 3
    assert g.aggs[1].value_i64 == ores.aggs[1].value_i64
    # pruned: the scan decoded a tiny fraction of the tablet (unless the
    # A/B knob disabled the selection)
    import os
    if not os.environ.get("YBG_NOPRUNE"):
        assert g.entries_seen < 2000


@pytest.mark.gpu
def test_gpu_in_range_parity():
    from gpu_scan import GpuScan
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    data, offsets, nb, total, ne = y.generate(schema, rows=50000, seed=6)
    ptr, ln, _k = _in_range_bytes([(1 << 38, 1 << 39, True, False),
                                   (3 << 39, 2**40, False, True)])
    preds = [y.Pred(0, 0, y.PRED_IN_RANGE, 0, ptr, ln)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 0)]
    spec = _spec(schema, preds, aggs, read=1_700_000_000_000_000)
    s = GpuScan(spec)
    s.feed_blocks_host(data, offsets, nb, total)
    s.execute()
    g = s.aggregates()
    s.close()
    sres = y.sim_scan(spec, data, offsets, nb)
    assert (g.rows_matched, g.aggs[1].value_i64) == \
        (sres.rows_matched, sres.aggs[1].value_i64)
