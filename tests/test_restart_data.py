"""Read-restart data (GetReadRestartData analog): a visible record with
commit time in (read, local_limit] must be reported as the max-seen commit
time (its encoded DocHybridTime), and scans with local_limit == read never
report one (intent_aware_iterator.cc:815-827 UpdateMaxSeenHt,
:1400-1410 GetReadRestartData). Simulator vs oracle, byte-exact."""
import ctypes as C

import ybgpu as y


def _tablet():
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(200):
        for ht in (3000, 2000, 1000):  # newest first (micros)
            seq += 1
            b.add_packed_row(ht, [(y.T_INT64, r * 10 + ht)], hash_=r // 64,
                             key_datums=(r,), seq=seq)
    return schema, b.finish()


def _run(schema, built, read, local, glob):
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(read, local, glob)
    spec.num_aggs = 1
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    sres = y.sim_scan(spec, built[0], built[1], built[2])

    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(read, local, glob)
    ospec.num_aggs = 1
    ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
    ores, _ = y.orcl_scan(built[0], built[1], built[2], osc, ospec)
    assert sres.rows_scanned == ores.rows_scanned
    assert sres.restart_ht_len == ores.restart_ht_len, \
        (sres.restart_ht_len, ores.restart_ht_len)
    got = bytes(sres.restart_ht[:sres.restart_ht_len])
    want = bytes(ores.restart_ht[:ores.restart_ht_len])
    assert got == want, (got.hex(), want.hex())
    return sres


def test_restart_window_reported():
    schema, built = _tablet()
    # read at 1500, local_limit 2500: the ht=2000 versions are visible AND
    # committed after the read time -> restart data present
    r = _run(schema, built, 1500, 2500, 3500)
    assert r.restart_ht_len > 0
    # every row's newest visible version is ht=2000 -> all rows scanned
    assert r.rows_scanned == 200


def test_no_window_no_restart():
    schema, built = _tablet()
    r = _run(schema, built, 1500, 1500, 3500)
    assert r.restart_ht_len == 0
    assert r.rows_scanned == 200  # ht=1000 versions are visible


def test_window_without_candidates():
    schema, built = _tablet()
    # read after everything: window (5000, 6000] holds no records
    r = _run(schema, built, 5000, 6000, 7000)
    assert r.restart_ht_len == 0
    assert r.rows_scanned == 200


import pytest


@pytest.mark.gpu
def test_restart_window_gpu():
    """The same three windows through the real kernels (the restart-min
    wave fold + pre-reduction + final reduce are kernel-only code)."""
    from gpu_scan import GpuScan

    schema, built = _tablet()
    for (read, local, glob), expect_restart in (
            ((1500, 2500, 3500), True),
            ((1500, 1500, 3500), False),
            ((5000, 6000, 7000), False)):
        spec = y.ScanSpec()
        spec.schema = schema
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.read_time = y.read_time(read, local, glob)
        spec.num_aggs = 1
        spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        s = GpuScan(spec)
        s.feed_blocks_host(built[0], built[1], built[2], built[3])
        s.execute()
        gres = s.aggregates()
        s.close()
        sres = y.sim_scan(spec, built[0], built[1], built[2])
        assert gres.rows_scanned == sres.rows_scanned == 200
        assert gres.restart_ht_len == sres.restart_ht_len
        assert bytes(gres.restart_ht[:gres.restart_ht_len]) == \
            bytes(sres.restart_ht[:sres.restart_ht_len])
        assert bool(gres.restart_ht_len) == expect_restart, (read, local)


def test_restart_window_fuzz():
    """Random read/local/global windows over random multi-version tablets:
    sim and oracle must agree byte-exactly on the restart data."""
    import random
    rng = random.Random(31415)
    for it in range(15):
        schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
        b = y.Builder(schema)
        seq = 1 << 50
        rows = rng.randint(40, 400)
        for r in range(rows):
            hts = sorted(rng.sample(range(1000, 9000), rng.randint(1, 4)),
                         reverse=True)
            for ht in hts:
                seq += 1
                b.add_packed_row(ht, [(y.T_INT64, r)], hash_=r // 64,
                                 key_datums=(r,), seq=seq)
        built = b.finish()
        for _ in range(4):
            read = rng.randint(500, 9500)
            local = read + rng.choice([0, 0, rng.randint(1, 4000)])
            glob = local + rng.randint(0, 2000)
            _run(schema, built, read, local, glob)


def _grouped_tablet():
    # same 3-version rows, but with a second (group) column
    schema = y.make_schema([y.KT_INT64],
                          [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(200):
        for ht in (3000, 2000, 1000):
            seq += 1
            b.add_packed_row(ht, [(y.T_INT64, r % 7), (y.T_INT64, r)],
                             hash_=r // 64, key_datums=(r,), seq=seq)
    return schema, b.finish()


def test_group_scan_reports_restart():
    """GROUP BY scans must report read-restart data like plain scans
    (restart tracking is query-independent: it depends only on record
    visibility vs the read window). The sim group path shares the device
    code; its restart bytes must equal the plain scan's."""
    schema, built = _grouped_tablet()
    for (read, local, glob), expect in (
            ((1500, 2500, 3500), True),
            ((1500, 1500, 3500), False),
            ((5000, 6000, 7000), False)):
        spec = y.ScanSpec()
        spec.schema = schema
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.read_time = y.read_time(read, local, glob)
        spec.group_col = 1  # group by column id 10 (1-based slot)
        spec.num_aggs = 1
        spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        groups, restart = y.sim_group(spec, built[0], built[1], built[2],
                                      return_restart=True)
        assert sum(v[0] for v in groups.values()) == 200
        spec2 = y.ScanSpec()
        spec2.schema = schema
        spec2.kv_format = y.ENC_THREE_SHARED_PARTS
        spec2.read_time = y.read_time(read, local, glob)
        spec2.num_aggs = 1
        spec2.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        sres = y.sim_scan(spec2, built[0], built[1], built[2])
        want = bytes(sres.restart_ht[:sres.restart_ht_len])
        assert restart == want, (restart.hex(), want.hex())
        assert bool(restart) == expect


@pytest.mark.gpu
def test_group_scan_restart_gpu():
    """k_group's restart-min wave fold + the k_reduce fold into DevResult,
    surfaced through yb_gpu_scan_restart_data (the bht sizing fix: group
    kernels share the 6-slot LDS layout with k_scan)."""
    from gpu_scan import GpuScan
    schema, built = _grouped_tablet()
    for (read, local, glob), expect in (
            ((1500, 2500, 3500), True),
            ((1500, 1500, 3500), False),
            ((5000, 6000, 7000), False)):
        spec = y.ScanSpec()
        spec.schema = schema
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.read_time = y.read_time(read, local, glob)
        spec.group_col = 1
        spec.num_aggs = 1
        spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        s = GpuScan(spec)
        s.feed_blocks_host(built[0], built[1], built[2], built[3])
        groups = s.group_aggregate()
        restart = s.restart_data()
        s.close()
        sgroups, want = y.sim_group(spec, built[0], built[1], built[2],
                                    return_restart=True)
        assert groups == sgroups
        assert restart == want
        assert bool(restart) == expect
