"""Intents-DB merge: provisional records of in-flight
transactions resolved against the status table and merged into the scan —
docdb/intent_aware_iterator.cc:983-1011 (ProcessIntent),
transaction_status_cache.cc. Two independent implementations are
cross-checked: the product merges committed intents into the data blocks
at feed time (ybg_merge_intents); the oracle merges the two streams at
scan time (orcl_scan_intents)."""
import pytest

import ybgpu as y


def _res(r):
    return (r.rows_scanned, r.rows_matched,
            tuple((r.aggs[i].is_null, r.aggs[i].value_i64) for i in range(2)),
            bytes(r.restart_ht[:r.restart_ht_len]))


def _spec(schema, read, local=None, glob=None, preds=(), aggs=()):
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(read, local, glob)
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = p
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = a
    return spec


def _orcl_spec(read, local=None, glob=None, preds=(), aggs=()):
    sp = y.OrclScanSpec()
    sp.read_time = y.orcl_read_time(read, local, glob)
    sp.num_preds = len(preds)
    for i, p in enumerate(preds):
        sp.preds[i] = y.OrclPred(p.is_key_col, p.col, p.op, p.datum,
                                 p.bytes, p.bytes_len)
    sp.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        sp.aggs[i] = y.OrclAgg(a.op, a.col)
    return sp


def _base_tablet(rows=3000):
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(rows):
        seq += 1
        b.add_packed_row(1000, [(y.T_INT64, r), (y.T_INT64, r * 2)],
                         hash_=r // 512, key_datums=(r,), seq=seq)
    return schema, b.finish()


def _cross_check(schema, built, intents, txn_table, read, local=None,
                 glob=None, preds=(), aggs=None):
    """product feed-time merge + sim scan  vs  oracle runtime merge."""
    aggs = aggs or [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    blob, blen = intents.blob()
    txns, ntx = y.make_txns(txn_table)
    mb, mo, mn, mt = y.merge_intents(built[0], built[1], built[2], blob,
                                     blen, txns, ntx)
    spec = _spec(schema, read, local, glob, preds, aggs)
    sres = y.sim_scan(spec, mb, mo, mn)
    osc = y.orcl_schema_from(schema)
    ospec = _orcl_spec(read, local, glob, preds, aggs)
    ores = y.orcl_scan_intents(built[0], built[1], built[2], osc, ospec,
                               blob, blen, txns, ntx)
    assert _res(sres) == _res(ores), (_res(sres), _res(ores))
    return sres


def test_committed_intent_visible():
    schema, built = _base_tablet(200)
    it = y.Intents(schema)
    # txn 7 inserts a NEW row 500 (beyond existing) at write time 1500
    it.add_packed_row(1500, 7, [(y.T_INT64, 500), (y.T_INT64, 1000)],
                      hash_=0, key_datums=(500,))
    for read, commit, expect_extra in (
            (3000, 2000, 1),   # committed before read: visible
            (1800, 2000, 0),   # committed after read: not visible
            (2000, 2000, 1)):  # committed exactly at read: visible
        r = _cross_check(schema, built, it, {7: ("c", commit)}, read)
        assert r.rows_scanned == 200 + expect_extra, (read, commit)


def test_aborted_and_pending_invisible():
    schema, built = _base_tablet(200)
    it = y.Intents(schema)
    it.add_packed_row(1500, 7, [(y.T_INT64, 500), (y.T_INT64, 1)],
                      hash_=0, key_datums=(500,))
    it.add_packed_row(1500, 8, [(y.T_INT64, 501), (y.T_INT64, 2)],
                      hash_=0, key_datums=(501,))
    r = _cross_check(schema, built, it, {7: "aborted", 8: "pending"}, 5000)
    assert r.rows_scanned == 200


def test_intent_overwrites_existing_row():
    """A committed intent UPDATE of an existing row: newest-wins ordering
    between the resolved intent (at commit time) and the base row."""
    schema, built = _base_tablet(200)
    it = y.Intents(schema)
    it.add_packed_row(1500, 1, [(y.T_INT64, 50), (y.T_INT64, 999999)],
                      hash_=0, key_datums=(50,))
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    base_sum = sum(r * 2 for r in range(200))
    # committed at 2000 < read: intent value wins over base (ht 1000)
    r = _cross_check(schema, built, it, {1: ("c", 2000)}, 3000, aggs=aggs)
    assert r.aggs[1].value_i64 == base_sum - 100 + 999999
    # read below commit: base value stays
    r = _cross_check(schema, built, it, {1: ("c", 2000)}, 1500, aggs=aggs)
    assert r.aggs[1].value_i64 == base_sum


def test_intent_column_update_and_tombstone():
    schema, built = _base_tablet(100)
    it = y.Intents(schema)
    it.add_column_update(1600, 3, 1, 777, hash_=0, key_datums=(10,))
    it.add_row_tombstone(1700, 3, hash_=0, key_datums=(20,))
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    base_sum = sum(r * 2 for r in range(100))
    r = _cross_check(schema, built, it, {3: ("c", 2500)}, 4000, aggs=aggs)
    assert r.rows_scanned == 99          # row 20 deleted
    assert r.aggs[1].value_i64 == base_sum - 20 + 777 - 40


def test_intent_commit_in_restart_window():
    """An intent committed in (read, local_limit] is VISIBLE under the
    committed-intent rule only via global_limit; and a visible record with
    commit > read must surface as read-restart data
    (intent_aware_iterator.cc:1249-1267, :1400-1410)."""
    schema, built = _base_tablet(50)
    it = y.Intents(schema)
    it.add_packed_row(1200, 9, [(y.T_INT64, 300), (y.T_INT64, 5)],
                      hash_=0, key_datums=(300,))
    # commit at 2000, read 1500, local 2500, global 3500. Intent write
    # time 1200 <= local_limit -> the GLOBAL limit governs (A.5 rule /
    # intent_aware_iterator.cc:1249-1267): commit 2000 <= 3500 -> VISIBLE
    # even though commit > read; and a visible record with commit > read
    # is exactly a read-restart candidate.
    r = _cross_check(schema, built, it, {9: ("c", 2000)}, 1500, 2500, 3500)
    assert r.rows_scanned == 51
    assert r.restart_ht_len > 0
    # read at 2200 >= commit: visible through the plain rule, no restart
    r = _cross_check(schema, built, it, {9: ("c", 2000)}, 2200, 2200)
    assert r.rows_scanned == 51
    assert r.restart_ht_len == 0
    # tight window (local == read 1500): intent time 1200 <= local ->
    # global == local == read: commit 2000 > 1500 -> invisible
    r = _cross_check(schema, built, it, {9: ("c", 2000)}, 1500, 1500)
    assert r.rows_scanned == 50


def test_many_intents_fuzz():
    import random
    rng = random.Random(4242)
    schema, built = _base_tablet(1000)
    it = y.Intents(schema)
    table = {}
    for t in range(40):
        st = rng.choice(["pending", "aborted", "c"])
        if st == "c":
            table[t] = ("c", rng.randint(1200, 4000))
        else:
            table[t] = st
        for _ in range(rng.randint(1, 6)):
            r = rng.randint(0, 1400)  # some rows beyond the base tablet
            kind = rng.random()
            if kind < 0.6:
                it.add_packed_row(rng.randint(1100, 3900), t,
                                  [(y.T_INT64, r), (y.T_INT64, r * 3)],
                                  hash_=r // 512, key_datums=(r,),
                                  write_id=rng.randint(0, 5))
            elif kind < 0.8:
                it.add_column_update(rng.randint(1100, 3900), t, 1,
                                     rng.randint(0, 10000), hash_=r // 512,
                                     key_datums=(r,))
            else:
                it.add_row_tombstone(rng.randint(1100, 3900), t,
                                     hash_=r // 512, key_datums=(r,))
    for read in (1000, 1500, 2500, 3500, 5000):
        _cross_check(schema, built, it, table, read, read + 300, read + 800)


@pytest.mark.gpu
def test_intents_gpu():
    """The device path over the feed-time merged tablet vs the oracle
    runtime merge (yb_gpu_scan_feed_blocks_intents)."""
    from gpu_scan import GpuScan
    schema, built = _base_tablet(2000)
    it = y.Intents(schema)
    it.add_packed_row(1500, 1, [(y.T_INT64, 50), (y.T_INT64, 999999)],
                      hash_=0, key_datums=(50,))
    it.add_packed_row(1500, 2, [(y.T_INT64, 5000), (y.T_INT64, 7)],
                      hash_=3, key_datums=(5000,))
    it.add_row_tombstone(1600, 1, hash_=0, key_datums=(60,))
    it.add_packed_row(1500, 3, [(y.T_INT64, 70), (y.T_INT64, 1)],
                      hash_=0, key_datums=(70,))
    table = {1: ("c", 2000), 2: ("c", 1800), 3: "aborted"}
    blob, blen = it.blob()
    txns, ntx = y.make_txns(table)
    aggs = [y.Agg(y.AGG_COUNT_STAR, 0), y.Agg(y.AGG_SUM_INT64, 1)]
    for read, local in ((3000, 3000), (1700, 2600), (900, 900)):
        spec = _spec(schema, read, local, local + 1000, aggs=aggs)
        s = GpuScan(spec)
        s.feed_blocks_intents(built[0], built[1], built[2], blob, blen,
                              txns, ntx)
        s.execute()
        g = s.aggregates()
        s.close()
        osc = y.orcl_schema_from(schema)
        ospec = _orcl_spec(read, local, local + 1000, aggs=aggs)
        o = y.orcl_scan_intents(built[0], built[1], built[2], osc, ospec,
                                blob, blen, txns, ntx)
        assert _res(g) == _res(o), (read, local)


def test_restart_via_global_limit_only():
    """Regression (round-2 GPU soak): a committed intent visible through
    the GLOBAL limit needs restart tracking even when local_limit == read
    — the device gate used to consider only the local window."""
    schema, built = _base_tablet(60)
    it = y.Intents(schema)
    it.add_packed_row(1200, 9, [(y.T_INT64, 300), (y.T_INT64, 5)],
                      hash_=0, key_datums=(300,))
    # read == local == 1500, global 3500; commit 2000 in (read, global]:
    # intent time 1200 <= local -> global rule -> VISIBLE with commit >
    # read -> restart data required
    r = _cross_check(schema, built, it, {9: ("c", 2000)}, 1500, 1500, 3500)
    assert r.rows_scanned == 61
    assert r.restart_ht_len > 0


def test_intents_shared_prefix_format():
    """The feed-time merge re-encodes affected blocks in the tablet's KV
    format — cover ENC_SHARED_PREFIX end to end."""
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema, kv_format=y.ENC_SHARED_PREFIX)
    seq = 1 << 50
    for r in range(500):
        seq += 1
        b.add_packed_row(1000, [(y.T_INT64, r), (y.T_INT64, r * 2)],
                         hash_=r // 256, key_datums=(r,), seq=seq)
    built = b.finish()
    it = y.Intents(schema)
    it.add_packed_row(1500, 1, [(y.T_INT64, 100), (y.T_INT64, 424242)],
                      hash_=0, key_datums=(100,))
    blob, blen = it.blob()
    txns, ntx = y.make_txns({1: ("c", 2000)})
    mb, mo, mn, mt = y.merge_intents(built[0], built[1], built[2], blob,
                                     blen, txns, ntx,
                                     kv_format=y.ENC_SHARED_PREFIX)
    spec = _spec(schema, 3000,
                 aggs=[y.Agg(y.AGG_COUNT_STAR, 0),
                       y.Agg(y.AGG_SUM_INT64, 1)])
    spec.kv_format = y.ENC_SHARED_PREFIX
    sres = y.sim_scan(spec, mb, mo, mn)
    osc = y.orcl_schema_from(schema)
    ospec = _orcl_spec(3000, aggs=[y.Agg(y.AGG_COUNT_STAR, 0),
                                   y.Agg(y.AGG_SUM_INT64, 1)])
    ores = y.orcl_scan_intents(built[0], built[1], built[2], osc, ospec,
                               blob, blen, txns, ntx,
                               kv_format=y.ENC_SHARED_PREFIX)
    assert _res(sres) == _res(ores)
    assert sres.aggs[1].value_i64 == sum(r * 2 for r in range(500)) \
        - 200 + 424242
