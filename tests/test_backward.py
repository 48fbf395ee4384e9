"""Backward (descending) scans — doc_rowwise_iterator.cc:690-818
FetchNextImpl<Direction::kBackward>, SkipFutureRecords<kBackward>
(intent_aware_iterator.cc:1319ff). On this engine the bandwidth-bound
parallel scan is direction-neutral; backward is a delivery-order contract
of the boundary: descending DocKey order, row_limit pages deliver the
highest keys first, and the paging state is the EXCLUSIVE upper bound of
the resumed page."""
import ctypes as C

import pytest

import ybgpu as y


def _gpu():
    import gpu_scan
    if not gpu_scan.gpu_available():
        pytest.skip("no GPU")
    return gpu_scan


def _tablet(rows=4000):
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(rows):
        seq += 1
        b.add_packed_row(1000 + (r % 7), [(y.T_INT64, r), (y.T_INT64, r * 3)],
                         hash_=r // 512, key_datums=(r,), seq=seq)
    return schema, b.finish()


def _spec(schema, read=5000, backward=0, row_limit=0, lower=None,
          upper=None, preds=()):
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(read)
    spec.num_preds = len(preds)
    for i, p in enumerate(preds):
        spec.preds[i] = p
    spec.emit_rows = 1
    spec.backward = backward
    spec.row_limit = row_limit
    if lower is not None:
        spec.lower_bound = C.cast(lower[0], C.POINTER(C.c_uint8))
        spec.lower_bound_len = lower[1]
    if upper is not None:
        spec.upper_bound = C.cast(upper[0], C.POINTER(C.c_uint8))
        spec.upper_bound_len = upper[1]
    return spec


def _drain(lib, spec, built, schema):
    h = lib.yb_host_iter_open(C.byref(spec), built[0], built[1], built[2])
    assert h
    kd = (C.c_uint64 * y.MAX_KEYCOLS)()
    vd = (C.c_uint64 * y.MAX_COLS)()
    nm = C.c_uint32()
    vl = C.POINTER(C.c_uint8)()
    nk = schema.num_hash_cols + schema.num_range_cols
    rows = []
    while lib.yb_host_iter_next(h, kd, vd, C.byref(nm), C.byref(vl)) == 1:
        vals = tuple(None if (nm.value >> c) & 1 else vd[c]
                     for c in range(schema.num_value_cols))
        rows.append((tuple(kd[i] for i in range(nk)), vals))
    pk = (C.c_uint8 * 64)()
    pl = C.c_size_t()
    assert lib.yb_host_iter_paging_state(h, pk, 64, C.byref(pl)) == 0
    lib.yb_host_iter_close(h)
    return rows, bytes(pk[:pl.value])


@pytest.mark.gpu
def test_backward_full_and_bounded():
    gpu_scan = _gpu()
    lib = gpu_scan._lib()
    lib.yb_host_iter_paging_state.restype = C.c_int
    lib.yb_host_iter_paging_state.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_size_t)]
    schema, built = _tablet()
    fwd, _ = _drain(lib, _spec(schema), built, schema)
    bwd, _ = _drain(lib, _spec(schema, backward=1), built, schema)
    assert bwd == list(reversed(fwd))
    # oracle pin: forward rows come from the oracle; backward = reversed
    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(5000)
    _, want = y.orcl_scan(built[0], built[1], built[2], osc, ospec,
                          collect_rows=True)
    want = [(k, v) for (k, v) in want]
    assert len(bwd) == len(want)
    assert bwd == list(reversed(want))
    # bounds apply identically in both directions
    with_pred = [y.Pred(0, 0, y.PRED_GE, 1000, None, 0),
                 y.Pred(0, 0, y.PRED_LT, 3000, None, 0)]
    f2, _ = _drain(lib, _spec(schema, preds=with_pred), built, schema)
    b2, _ = _drain(lib, _spec(schema, backward=1, preds=with_pred), built,
                   schema)
    assert b2 == list(reversed(f2)) and len(b2) == 2000


@pytest.mark.gpu
def test_backward_paging_resume():
    """Backward pages: highest keys first; resume with the paging state as
    the EXCLUSIVE upper bound + backward."""
    gpu_scan = _gpu()
    lib = gpu_scan._lib()
    lib.yb_host_iter_paging_state.restype = C.c_int
    lib.yb_host_iter_paging_state.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_size_t)]
    schema, built = _tablet()
    pages = []
    upper = None
    for _ in range(10):
        spec = _spec(schema, backward=1, row_limit=900, upper=upper)
        page, pkey = _drain(lib, spec, built, schema)
        pages.append(page)
        if not pkey:
            break
        buf = C.create_string_buffer(pkey, len(pkey))
        upper = (buf, len(pkey))
    all_rows = [r for p in pages for r in p]
    fwd, _ = _drain(lib, _spec(schema), built, schema)
    assert all_rows == list(reversed(fwd))
    assert len(pages) == 5 and [len(p) for p in pages[:4]] == [900] * 4


@pytest.mark.gpu
def test_backward_tuple_id_seek():
    gpu_scan = _gpu()
    lib = gpu_scan._lib()
    lib.yb_host_iter_tuple_id.restype = C.c_int
    lib.yb_host_iter_tuple_id.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_size_t)]
    lib.yb_host_iter_seek_tuple.restype = C.c_int
    lib.yb_host_iter_seek_tuple.argtypes = [
        C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t]
    schema, built = _tablet(500)
    spec = _spec(schema, backward=1)
    h = lib.yb_host_iter_open(C.byref(spec), built[0], built[1], built[2])
    kd = (C.c_uint64 * y.MAX_KEYCOLS)()
    vd = (C.c_uint64 * y.MAX_COLS)()
    nm = C.c_uint32()
    vl = C.POINTER(C.c_uint8)()
    # fetch 10 rows, remember the 10th ybctid
    tid = (C.c_uint8 * 64)()
    tl = C.c_size_t()
    last_key = None
    for i in range(10):
        assert lib.yb_host_iter_next(h, kd, vd, C.byref(nm), C.byref(vl)) == 1
        last_key = kd[0]
    assert lib.yb_host_iter_tuple_id(h, tid, 64, C.byref(tl)) == 0
    # seek back to it and re-fetch: same row again
    assert lib.yb_host_iter_seek_tuple(h, tid, tl.value) == 0
    assert lib.yb_host_iter_next(h, kd, vd, C.byref(nm), C.byref(vl)) == 1
    assert kd[0] == last_key
    lib.yb_host_iter_close(h)


@pytest.mark.gpu
def test_ql_row_form():
    """FetchNext(QLTableRow*) analog: the same rows keyed by COLUMN ID
    (ql_rowwise_iterator_interface.h:48-52)."""
    gpu_scan = _gpu()
    lib = gpu_scan._lib()
    lib.yb_host_iter_next_ql.restype = C.c_int
    lib.yb_host_iter_next_ql.argtypes = [
        C.c_void_p, C.POINTER(C.c_int32), C.POINTER(C.c_uint64),
        C.POINTER(C.c_int32), C.POINTER(C.c_uint64), C.POINTER(C.c_uint32),
        C.POINTER(C.POINTER(C.c_uint8))]
    schema, built = _tablet(300)
    spec = _spec(schema)
    h = lib.yb_host_iter_open(C.byref(spec), built[0], built[1], built[2])
    kci = (C.c_int32 * y.MAX_KEYCOLS)()
    kd = (C.c_uint64 * y.MAX_KEYCOLS)()
    ci = (C.c_int32 * y.MAX_COLS)()
    vd = (C.c_uint64 * y.MAX_COLS)()
    nm = C.c_uint32()
    vl = C.POINTER(C.c_uint8)()
    n = 0
    while lib.yb_host_iter_next_ql(h, kci, kd, ci, vd, C.byref(nm),
                                   C.byref(vl)) == 1:
        if n == 0:
            assert list(ci[:2]) == [10, 11]  # schema column ids
            assert kd[0] == 0 and vd[0] == 0 and vd[1] == 0
        n += 1
    lib.yb_host_iter_close(h)
    assert n == 300


@pytest.mark.gpu
def test_backward_over_merged_intents():
    """Backward delivery over a feed-time-merged tablet (intents +
    backward compose): descending rows equal the reversed oracle
    runtime-merge row set."""
    gpu_scan = _gpu()
    lib = gpu_scan._lib()
    schema = y.make_schema([y.KT_INT64],
                           [(10, y.T_INT64, 1), (11, y.T_INT64, 1)])
    b = y.Builder(schema)
    seq = 1 << 50
    for r in range(800):
        seq += 1
        b.add_packed_row(1000, [(y.T_INT64, r), (y.T_INT64, r * 2)],
                         hash_=r // 256, key_datums=(r,), seq=seq)
    built = b.finish()
    it = y.Intents(schema)
    it.add_packed_row(1500, 1, [(y.T_INT64, 5000), (y.T_INT64, 1)],
                      hash_=3, key_datums=(5000,))
    it.add_row_tombstone(1500, 1, hash_=0, key_datums=(10,))
    blob, blen = it.blob()
    txns, ntx = y.make_txns({1: ("c", 2000)})
    mb, mo, mn, mt = y.merge_intents(built[0], built[1], built[2], blob,
                                     blen, txns, ntx)
    spec = _spec(schema, read=3000, backward=1)
    rows_b, _ = _drain(lib, spec, (mb, mo, mn), schema)
    spec_f = _spec(schema, read=3000)
    rows_f, _ = _drain(lib, spec_f, (mb, mo, mn), schema)
    assert rows_b == list(reversed(rows_f))
    assert len(rows_b) == 800  # +1 inserted, -1 tombstoned
    assert rows_b[0][0][0] == 5000  # highest key first
