"""GROUP BY parity (config #5 'GROUP-BY-key partial aggregates'):
the device grouping algorithm (host simulator on CPU; real kernels in the
gpu suite) must produce the same per-group aggregates as the oracle —
integer aggregates exact, grouped double SUM within tolerance."""
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
