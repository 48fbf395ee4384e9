"""GROUP BY parity (config #5 'GROUP-BY-key partial aggregates'):
the device grouping algorithm (host simulator on CPU; real kernels in the
gpu suite) must produce the same per-group aggregates as the oracle —
integer aggregates exact, grouped double SUM within tolerance."""
import pytest

import ybgpu as y


def _dataset_int_groups():
    schema = y.make_schema(
        [y.KT_INT64],
        [(10, y.T_INT64, 1), (11, y.T_INT64, 1), (12, y.T_DOUBLE, 1)])
    b = y.Builder(schema)
    for r in range(30_000):
        g = (r * 2654435761) % 97  # 97 groups
        v = None if r % 13 == 0 else r
        b.add_packed_row(1000 + r, [(y.T_INT64, g), (y.T_INT64, v),
                                    (y.T_DOUBLE, r * 0.5)],
                         hash_=r // 64, key_datums=(r,))
    return schema, b.finish(), b


def _dataset_str_groups():
    schema = y.make_schema(
        [y.KT_INT64], [(10, y.T_STRING, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema)
    names = [b"alpha", b"beta", b"gamma-longer-name", b"d"]
    for r in range(8_000):
        s = None if r % 11 == 0 else names[r % len(names)]
        b.add_packed_row(1000 + r, [(y.T_STRING, s), (y.T_INT64, r)],
                         hash_=r // 64, key_datums=(r,))
    return schema, b.finish(), b


def _specs(schema, group_col_idx, aggs, read=1_700_000_000_000_000):
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(read)
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = a
    spec.group_col = group_col_idx + 1
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(read)
    ospec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        ospec.aggs[i] = y.OrclAgg(a.op, a.col)
    return spec, ospec


def check_groups(got, want, aggs, f64_rel=1e-9):
    assert set(got.keys()) == set(want.keys()), \
        (len(got), len(want), set(got) ^ set(want))
    for k, wv in want.items():
        gv = got[k]
        for a, ag in enumerate(aggs):
            if wv[a] is None or gv[a] is None:
                assert wv[a] is None and gv[a] is None, (k, a, gv[a], wv[a])
            elif ag.op == y.AGG_SUM_DOUBLE:
                assert abs(gv[a] - wv[a]) <= f64_rel * max(1.0, abs(wv[a])), \
                    (k, a, gv[a], wv[a])
            else:
                assert gv[a] == wv[a], (k, a, gv[a], wv[a])


def test_sim_group_int_keys():
    schema, built, _b = _dataset_int_groups()
    data, offsets, nb, total, ne = built
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1),
            y.Agg(y.AGG_MIN_INT64, 1), y.Agg(y.AGG_MAX_INT64, 1),
            y.Agg(y.AGG_SUM_DOUBLE, 2), y.Agg(y.AGG_COUNT, 1)]
    spec, ospec = _specs(schema, 0, aggs)
    got = y.sim_group(spec, data, offsets, nb)
    osc = y.orcl_schema_from(schema)
    want = y.orcl_group(data, offsets, nb, osc, ospec, 0)
    assert len(want) == 97
    check_groups(got, want, aggs)


def test_sim_group_string_keys_and_nulls():
    schema, built, _b = _dataset_str_groups()
    data, offsets, nb, total, ne = built
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    spec, ospec = _specs(schema, 0, aggs)
    got = y.sim_group(spec, data, offsets, nb)
    osc = y.orcl_schema_from(schema)
    want = y.orcl_group(data, offsets, nb, osc, ospec, 0)
    assert None in want and len(want) == 5
    check_groups(got, want, aggs)


def _dataset_doubles():
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_DOUBLE, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    import random
    rng = random.Random(777)
    for r in range(20000):
        seq += 1
        b.add_packed_row(1000, [(y.T_INT64, r % 97),
                                (y.T_DOUBLE, rng.uniform(-1e6, 1e6))],
                         hash_=r // 512, key_datums=(r,), seq=seq)
    return schema, b.finish()


def test_grouped_double_min_max_sum():
    """Grouped double MIN/MAX (exact: order-isomorphic u64 atomics) and
    the DETERMINISTIC grouped double SUM (128-bit fixed-point at 2^-60:
    order-independent integer atomics) vs the oracle's sequential doubles.
    MIN/MAX must be bit-exact; SUM within 1e-9 relative (the fixed-point
    accumulator is more precise than a double-order-dependent sum)."""
    schema, built = _dataset_doubles()
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_MIN_DOUBLE, 1),
            y.Agg(y.AGG_MAX_DOUBLE, 1), y.Agg(y.AGG_SUM_DOUBLE, 1)]
    spec, ospec = _specs(schema, 0, aggs)
    got = y.sim_group(spec, built[0], built[1], built[2])
    osc = y.orcl_schema_from(schema)
    want = y.orcl_group(built[0], built[1], built[2], osc, ospec, 0)
    assert set(got) == set(want) and len(got) == 97
    for k in got:
        gv, wv = got[k], want[k]
        assert gv[0] == wv[0]                      # COUNT
        assert gv[1] == wv[1], (k, gv[1], wv[1])   # MIN exact
        assert gv[2] == wv[2], (k, gv[2], wv[2])   # MAX exact
        assert abs(gv[3] - wv[3]) <= 1e-9 * max(1.0, abs(wv[3])), k


def test_grouped_double_sum_poison():
    """Values outside the fixed-point range poison the slot -> NaN."""
    import math
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_DOUBLE, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    b.add_packed_row(1000, [(y.T_INT64, 0), (y.T_DOUBLE, 1e30)],
                     hash_=0, key_datums=(0,), seq=seq)
    seq += 1
    b.add_packed_row(1000, [(y.T_INT64, 1), (y.T_DOUBLE, 2.5)],
                     hash_=0, key_datums=(1,), seq=seq)
    built = b.finish()
    aggs = [y.Agg(y.AGG_SUM_DOUBLE, 1)]
    spec, _ = _specs(schema, 0, aggs)
    got = y.sim_group(spec, built[0], built[1], built[2])
    assert math.isnan(got[0][0])
    assert got[1][0] == 2.5


@pytest.mark.gpu
def test_grouped_double_gpu_deterministic():
    """Device grouped doubles: MIN/MAX bit-exact vs oracle; SUM
    bit-identical across repeated runs (the determinism the fixed-point
    accumulator buys over float atomics)."""
    import gpu_scan
    schema, built = _dataset_doubles()
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_MIN_DOUBLE, 1),
            y.Agg(y.AGG_MAX_DOUBLE, 1), y.Agg(y.AGG_SUM_DOUBLE, 1)]
    spec, ospec = _specs(schema, 0, aggs)
    runs = []
    for _ in range(3):
        s = gpu_scan.GpuScan(spec)
        s.feed_blocks_host(built[0], built[1], built[2], built[3])
        runs.append(s.group_aggregate())
        s.close()
    assert runs[0] == runs[1] == runs[2]  # bit-identical SUMs
    osc = y.orcl_schema_from(schema)
    want = y.orcl_group(built[0], built[1], built[2], osc, ospec, 0)
    for k in runs[0]:
        assert runs[0][k][1] == want[k][1]
        assert runs[0][k][2] == want[k][2]
        assert abs(runs[0][k][3] - want[k][3]) <= \
            1e-9 * max(1.0, abs(want[k][3]))
