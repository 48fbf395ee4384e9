"""Scenario tests for the CPU oracle, transcribed from the reference's own
iterator tests (docdb/docrowwiseiterator-test.cc) — same logical writes at the
same hybrid times, same expected visible rows. These pin the oracle's
semantics (visibility, overrides, tombstones, aggregates) on CPU; the GPU
parity suite (test_gpu_parity.py) then pins the GPU against the oracle."""
import ybgpu as y


def scan(schema, built, read_micros, preds=(), aggs=(), collect=False):
    data, offsets, nb, total, ne = built
    osc = y.orcl_schema_from(schema)
    spec = y.OrclScanSpec()
    spec.read_time = y.orcl_read_time(read_micros)
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = p
    spec.num_aggs = len(aggs)
    for i, a in enumerate(aggs):
        spec.aggs[i] = a
    return y.orcl_scan(data, offsets, nb, osc, spec, collect_rows=collect)


def test_docrowwiseiterator_scenario():
    """Mirrors SetupDocRowwiseIteratorData + TestDocRowwiseIterator
    (docrowwiseiterator-test.cc:864-937): YCQL-style per-column writes,
    column delete + overwrite, read at HT 2000 and 5000."""
    # schema: range key int64; columns c30 (int64 stand-in for "c" string),
    # c40 int64, c50 int64 — we use int64 for all three (the scenario's
    # visibility logic is type-independent; string columns are covered by
    # test_gpu_parity.test_mixed_types / test_string_equality_pred).
    schema = y.make_schema([y.KT_INT64],
                           [(30, y.T_INT64, 1), (40, y.T_INT64, 1),
                            (50, y.T_INT64, 1)],
                           has_hash=False, num_hash_cols=0)
    b = y.Builder(schema)
    # Row 1 (key 11111): c30@1000, c40@1000=10000, c50@1000
    b.add_column_update(1000, 0, 111, key_datums=(11111,))
    b.add_column_update(1000, 1, 10000, key_datums=(11111,))
    b.add_column_update(1000, 2, 115, key_datums=(11111,))
    # Row 2 (key 22222): c40: 30000@3000, DEL@2500, 20000@2000;
    #                    c50: row2_e_prime@4000, row2_e@2000
    b.add_column_update(3000, 1, 30000, key_datums=(22222,))
    b.add_column_update(2500, 1, None, key_datums=(22222,), null=True)
    b.add_column_update(2000, 1, 20000, key_datums=(22222,))
    b.add_column_update(4000, 2, 225, key_datums=(22222,))
    b.add_column_update(2000, 2, 224, key_datums=(22222,))
    built = b.finish()

    # Read at HT 2000 (docrowwiseiterator-test.cc:919-926):
    #   row1: (111, 10000, 115); row2: (null, 20000, 224)
    res, rows = scan(schema, built, 2000, collect=True)
    assert res.rows_scanned == 2
    assert rows == [((11111,), (111, 10000, 115)),
                    ((22222,), (None, 20000, 224))]

    # Read at HT 5000 (:928-936): row2: (null, 30000, 225)
    res, rows = scan(schema, built, 5000, collect=True)
    assert rows == [((11111,), (111, 10000, 115)),
                    ((22222,), (None, 30000, 225))]

    # Read at HT 2600: c40 deleted at 2500 -> null
    res, rows = scan(schema, built, 2600, collect=True)
    assert rows == [((11111,), (111, 10000, 115)),
                    ((22222,), (None, None, 224))]


def test_deleted_document_scenario():
    """Mirrors TestDocRowwiseIteratorDeletedDocument
    (docrowwiseiterator-test.cc:940-983): row tombstone hides older column
    writes; other rows unaffected."""
    schema = y.make_schema([y.KT_INT64],
                           [(30, y.T_INT64, 1), (40, y.T_INT64, 1)],
                           has_hash=False, num_hash_cols=0)
    b = y.Builder(schema)
    # row1: tombstone@2500 sorts before its column updates
    b.add_row_tombstone(2500, key_datums=(11111,))
    b.add_column_update(1000, 0, 111, key_datums=(11111,))
    b.add_column_update(1000, 1, 10000, key_datums=(11111,))
    # row2: c40@2000
    b.add_column_update(2000, 1, 20000, key_datums=(22222,))
    built = b.finish()

    res, rows = scan(schema, built, 5000, collect=True)
    assert rows == [((22222,), (None, 20000))]

    # before the delete both rows visible
    res, rows = scan(schema, built, 2000, collect=True)
    assert rows == [((11111,), (111, 10000)), ((22222,), (None, 20000))]


def test_packed_row_update_scenario():
    """Mirrors TestUpdatePackedRow / TestDeleteMarkerWithPackedRow
    (docrowwiseiterator-test.cc packed-row paths): packed row + newer
    column update wins by write time; newer packed row replaces all."""
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_INT64, 1)],
                           has_hash=False, num_hash_cols=0)
    b = y.Builder(schema)
    # newer packed row first in key order (enc HT desc)
    b.add_packed_row(3000, [(y.T_INT64, 100), (y.T_INT64, 200)],
                     key_datums=(1,), seq=(1 << 50) + 2)
    b.add_packed_row(1000, [(y.T_INT64, 1), (y.T_INT64, 2)],
                     key_datums=(1,), seq=(1 << 50) + 1)
    b.add_column_update(2000, 1, 999, key_datums=(1,), seq=(1 << 50) + 3)
    built = b.finish()

    # read@1500: packed(1000) = (1,2)
    _, rows = scan(schema, built, 1500, collect=True)
    assert rows == [((1,), (1, 2))]
    # read@2500: packed(1000) + col11 update@2000 -> (1, 999)
    _, rows = scan(schema, built, 2500, collect=True)
    assert rows == [((1,), (1, 999))]
    # read@5000: packed(3000) wins; update@2000 older than base -> ignored
    _, rows = scan(schema, built, 5000, collect=True)
    assert rows == [((1,), (100, 200))]


def test_aggregate_null_semantics():
    """doc_expr.cc:250-263 (COUNT skips NULL), :341-349 (SUM starts NULL)."""
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)],
                           has_hash=False, num_hash_cols=0)
    b = y.Builder(schema)
    b.add_packed_row(1000, [(y.T_INT64, None)], key_datums=(1,))
    b.add_packed_row(1000, [(y.T_INT64, 7)], key_datums=(2,))
    b.add_packed_row(1000, [(y.T_INT64, None)], key_datums=(3,))
    built = b.finish()
    aggs = [y.OrclAgg(y.AGG_COUNT, 0), y.OrclAgg(y.AGG_COUNT_STAR, 0),
            y.OrclAgg(y.AGG_SUM_INT64, 0), y.OrclAgg(y.AGG_MIN_INT64, 0)]
    res, _ = scan(schema, built, 5000, aggs=aggs)
    assert res.aggs[0].value_i64 == 1      # COUNT(col) skips NULLs
    assert res.aggs[1].value_i64 == 3      # COUNT(*)
    assert res.aggs[2].value_i64 == 7
    assert res.aggs[3].value_i64 == 7
    # empty scan: all aggregates NULL
    res, _ = scan(schema, built, 500, aggs=aggs)
    assert all(res.aggs[i].is_null for i in range(4))


def test_predicate_null_filtered():
    """NULL operand fails the predicate (PG comparison semantics via
    pgsql_operation.cc CheckFilter)."""
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)],
                           has_hash=False, num_hash_cols=0)
    b = y.Builder(schema)
    b.add_packed_row(1000, [(y.T_INT64, None)], key_datums=(1,))
    b.add_packed_row(1000, [(y.T_INT64, -5)], key_datums=(2,))
    built = b.finish()
    preds = [y.OrclPred(0, 0, y.PRED_LT, 0, None, 0)]
    res, _ = scan(schema, built, 5000, preds=preds,
                  aggs=[y.OrclAgg(y.AGG_COUNT_STAR, 0)])
    assert res.rows_scanned == 2
    assert res.rows_matched == 1
    assert res.aggs[0].value_i64 == 1


def test_intent_hybrid_time_value_prefix():
    """Values carrying a '#'+DocHybridTime intent-time prefix (committed txn
    records): visibility rule intent_aware_iterator.cc:1249-1267 and the
    prefix strip. Built via add_raw with a hand-encoded value."""
    import ctypes as C
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)],
                           has_hash=False, num_hash_cols=0)

    # encode user key: 'I' + BE64(1 ^ 1<<63) + '!' + '#' + dht(1000us)
    lib = y.oracle()
    enc_dht = lib.orcl_dht_encode
    enc_dht.restype = C.c_size_t
    enc_dht.argtypes = [C.c_uint64, C.c_uint32, C.POINTER(C.c_uint8)]
    buf = (C.c_uint8 * 16)()

    def dht(micros, wid=0):
        n = enc_dht(micros << 12, wid, buf)
        return bytes(buf[:n])

    def ukey(k, micros):
        kb = b"I" + ((k ^ (1 << 63)).to_bytes(8, "big")) + b"!"
        return kb + b"#" + dht(micros)

    # packed V2 body: '|' + uvarint(0 schema version) = 0x00 + flags 0x00
    def packed_v2(val):
        return bytes([0x7C, 0x00, 0x00]) + val.to_bytes(8, "little")

    prod = y.product()
    create = prod.ybg_builder_create
    create.restype = C.c_void_p
    create.argtypes = [C.POINTER(y.Schema), C.c_int, C.c_size_t, C.c_int]
    add_raw = prod.ybg_builder_add_raw
    add_raw.restype = C.c_int
    add_raw.argtypes = [C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t,
                        C.c_uint64, C.POINTER(C.c_uint8), C.c_size_t]
    fin = prod.ybg_builder_finish
    fin.restype = C.c_int
    fin.argtypes = [C.c_void_p, C.POINTER(C.POINTER(C.c_uint8)),
                    C.POINTER(C.POINTER(C.c_uint64)), C.POINTER(C.c_uint64),
                    C.POINTER(C.c_uint64), C.POINTER(C.c_uint64)]
    h = create(C.byref(schema), y.ENC_THREE_SHARED_PARTS, 4096, 16)

    def add(k, micros, value):
        kb = ukey(k, micros)
        ka = (C.c_uint8 * len(kb)).from_buffer_copy(kb)
        va = (C.c_uint8 * len(value)).from_buffer_copy(value)
        assert add_raw(h, ka, len(kb), 1 << 50, va, len(value)) == 0

    # row 1: committed txn record, commit ht 1000, intent time 900
    add(1, 1000, b"#" + dht(900) + packed_v2(41))
    # row 2: plain record at 1000
    add(2, 1000, packed_v2(42))
    data = C.POINTER(C.c_uint8)()
    offsets = C.POINTER(C.c_uint64)()
    nb = C.c_uint64()
    tot = C.c_uint64()
    ne = C.c_uint64()
    assert fin(h, C.byref(data), C.byref(offsets), C.byref(nb), C.byref(tot),
               C.byref(ne)) == 0
    built = (data, offsets, nb.value, tot.value, ne.value)

    aggs = [y.OrclAgg(y.AGG_COUNT_STAR, 0), y.OrclAgg(y.AGG_SUM_INT64, 0)]
    # read at 2000: both rows visible, intent prefix stripped before decode
    res, rows = scan(schema, built, 2000, aggs=aggs, collect=True)
    assert res.rows_scanned == 2
    assert res.aggs[1].value_i64 == 83
    # read at 950 (>= intent time 900, < commit 1000): neither visible
    res, _ = scan(schema, built, 950, aggs=aggs)
    assert res.rows_scanned == 0
