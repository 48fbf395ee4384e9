"""Interval batching (YBG_IVB): one thread scans C consecutive restart
intervals as a single stream, deferring only the batch's first row and
walking its last row into the next batch. Results must be identical to the
C=1 protocol and the oracle for every C — including rows spanning several
intervals and whole batches."""
import os

import pytest

import ybgpu as y


def _sweep_ivb(run, ref):
    for ivb in (1, 2, 3, 7, 16, 64):
        os.environ["YBG_IVB"] = str(ivb)
        try:
            got = run()
        finally:
            del os.environ["YBG_IVB"]
        assert got == ref, f"ivb={ivb}"


def _res_tuple(r):
    return (r.entries_seen, r.rows_scanned, r.rows_matched,
            tuple((r.aggs[i].is_null, r.aggs[i].value_i64)
                  for i in range(2)),
            bytes(r.restart_ht[:r.restart_ht_len]))


def test_ivb_filtered_sum_parity():
    schema = y.make_schema([y.KT_INT64],
                           [(10 + i, y.T_INT64, 1) for i in range(4)])
    data, offsets, nb, total, ne = y.generate(schema, rows=50000, seed=7)
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(1_700_000_000_000_000)
    spec.num_preds = 1
    spec.preds[0] = y.Pred(0, 0, y.PRED_GT, 1 << 39, None, 0)
    spec.num_aggs = 2
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 3)
    ref = _res_tuple(y.sim_scan(spec, data, offsets, nb))
    _sweep_ivb(lambda: _res_tuple(y.sim_scan(spec, data, offsets, nb)), ref)


def test_ivb_multi_interval_rows():
    """Rows with many versions span interval and batch boundaries: 60
    versions per row at restart interval 16 means one row covers ~4
    intervals — whole batches at small C."""
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(40):
        for v in range(60):
            seq += 1
            b.add_packed_row(5000 - v, [(y.T_INT64, r * 100 + v)],
                             hash_=r // 16, key_datums=(r,), seq=seq)
    built = b.finish()
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(4000)  # mid-sweep: some versions future
    spec.num_aggs = 2
    spec.aggs[0] = y.AGG_COUNT_STAR and y.Agg(y.AGG_COUNT_STAR, 0)
    spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 0)
    ref = _res_tuple(y.sim_scan(spec, built[0], built[1], built[2]))
    # also pin against the oracle
    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(4000)
    ospec.num_aggs = 2
    ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
    ospec.aggs[1] = y.OrclAgg(y.AGG_SUM_INT64, 0)
    ores, _ = y.orcl_scan(built[0], built[1], built[2], osc, ospec)
    assert ref[1] == ores.rows_scanned and ref[2] == ores.rows_matched
    assert ref[3][1][1] == ores.aggs[1].value_i64
    _sweep_ivb(
        lambda: _res_tuple(y.sim_scan(spec, built[0], built[1], built[2])),
        ref)


def test_ivb_emit_and_group():
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(3000):
        seq += 1
        b.add_packed_row(1000 + (r % 5), [(y.T_INT64, r % 11),
                                          (y.T_INT64, r)],
                         hash_=r // 512, key_datums=(r,), seq=seq)
    built = b.finish()
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(2_000_000)
    spec.num_preds = 1
    spec.preds[0] = y.Pred(0, 1, y.PRED_LT, 1500, None, 0)
    ref_rows = y.sim_emit(spec, built[0], built[1], built[2])
    _sweep_ivb(lambda: y.sim_emit(spec, built[0], built[1], built[2]),
               ref_rows)
    gspec = y.ScanSpec()
    gspec.schema = schema
    gspec.kv_format = y.ENC_THREE_SHARED_PARTS
    gspec.read_time = y.read_time(2_000_000)
    gspec.group_col = 1
    gspec.num_aggs = 2
    gspec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    gspec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 1)
    ref_g = y.sim_group(gspec, built[0], built[1], built[2])
    _sweep_ivb(lambda: y.sim_group(gspec, built[0], built[1], built[2]),
               ref_g)


@pytest.mark.gpu
def test_ivb_gpu_parity():
    """The real kernels at several batch sizes vs the oracle-pinned C=1
    result (batch relay + cont-flag indexing are kernel-only code)."""
    from gpu_scan import GpuScan
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(5000):
        for v in range(3):
            seq += 1
            b.add_packed_row(3000 - v * 1000, [(y.T_INT64, r + v)],
                             hash_=r // 512, key_datums=(r,), seq=seq)
    built = b.finish()
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(2500, 3500, 4500)  # mid-sweep + restart
    spec.num_preds = 1
    spec.preds[0] = y.Pred(0, 0, y.PRED_GT, 1000, None, 0)
    spec.num_aggs = 2
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 0)
    ref = _res_tuple(y.sim_scan(spec, built[0], built[1], built[2]))

    def gpu_run():
        s = GpuScan(spec)
        s.feed_blocks_host(built[0], built[1], built[2], built[3])
        s.execute()
        r = s.aggregates()
        out = _res_tuple(r)
        s.close()
        return out

    _sweep_ivb(gpu_run, ref)
