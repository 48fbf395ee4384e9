"""The EXACT bench.py workload shapes at small scale, GPU vs oracle —
guards the benchmark specs themselves against drift from the parity-tested
surface (BASELINE configs 1/3/4 shapes incl. the config-5 string column)."""
import ctypes as C

import pytest

import ybgpu as y


def _gpu():
    import gpu_scan
    if not gpu_scan.gpu_available():
        pytest.skip("no GPU")
    return gpu_scan


@pytest.mark.gpu
def test_bench_workload_shapes_parity():
    gpu_scan = _gpu()
    rows = 200_000

    # filtersum (configs[1])
    schema = y.make_schema([y.KT_INT64],
                           [(10 + i, y.T_INT64, 1) for i in range(4)])
    preds = [y.Pred(0, 0, y.PRED_GT, 1 << 39, None, 0),
             y.Pred(0, 1, y.PRED_LT, 3 << 38, None, 0),
             y.Pred(0, 2, y.PRED_GE, 1 << 36, None, 0)]
    aggs = [y.Agg(y.AGG_SUM_INT64, 3), y.Agg(y.AGG_COUNT_STAR, 0)]
    for versions, read in ((1, 1_700_000_000_000_000),
                           (5, 1_600_000_002_500_000)):  # mvcc shape
        data, offsets, nb, total, ne = y.generate(
            schema, rows=rows, seed=42, versions=versions,
            ht_base_micros=1_600_000_000_000_000,
            ht_step_micros=1_000_000 if versions > 1 else 1000)
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
        s = gpu_scan.GpuScan(spec)
        s.feed_blocks_host(data, offsets, nb, total)
        s.execute()
        g = s.aggregates()
        s.close()
        osc = y.orcl_schema_from(schema)
        osp = y.OrclScanSpec()
        osp.read_time = y.orcl_read_time(read)
        osp.num_preds = len(preds)
        for i, p in enumerate(preds):
            osp.preds[i] = y.OrclPred(p.is_key_col, p.col, p.op, p.datum,
                                      p.bytes, p.bytes_len)
        osp.num_aggs = len(aggs)
        for i, a in enumerate(aggs):
            osp.aggs[i] = y.OrclAgg(a.op, a.col)
        o, _ = y.orcl_scan(data, offsets, nb, osc, osp)
        assert (g.rows_scanned, g.rows_matched, g.aggs[0].value_i64,
                g.aggs[1].value_i64) == \
            (o.rows_scanned, o.rows_matched, o.aggs[0].value_i64,
             o.aggs[1].value_i64), ("versions", versions)

    # groupby (configs[4] mixed row incl. 32-byte string)
    schema = y.make_schema(
        [y.KT_INT64],
        [(10, y.T_INT64, 1), (11, y.T_INT64, 1), (12, y.T_DOUBLE, 1),
         (13, y.T_STRING, 1)])
    slo = C.create_string_buffer(b"m" * 32, 32)
    preds = [y.Pred(0, 1, y.PRED_GT, 1 << 38, None, 0),
             y.Pred(0, 1, y.PRED_LT, 3 << 38, None, 0),
             y.Pred(0, 3, y.PRED_GE, 0,
                    C.cast(slo, C.POINTER(C.c_uint8)), 32)]
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    data, offsets, nb, total, ne = y.generate(schema, rows=rows, seed=42,
                                              group_mod=65536)
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(1_700_000_000_000_000)
    spec.group_col = 1
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = p
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = a
    s = gpu_scan.GpuScan(spec)
    s.feed_blocks_host(data, offsets, nb, total)
    got = s.group_aggregate()
    s.close()
    osc = y.orcl_schema_from(schema)
    osp = y.OrclScanSpec()
    osp.read_time = y.orcl_read_time(1_700_000_000_000_000)
    osp.num_preds = len(preds)
    for i, p in enumerate(preds):
        osp.preds[i] = y.OrclPred(p.is_key_col, p.col, p.op, p.datum,
                                  p.bytes, p.bytes_len)
    osp.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        osp.aggs[i] = y.OrclAgg(a.op, a.col)
    want = y.orcl_group(data, offsets, nb, osc, osp, 0)
    assert got == want
