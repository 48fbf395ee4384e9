"""ybgpu — ctypes bindings for the MI355X-native DocDB scan path.

Product library:  yugabyte-db_amd/libybgpu.so  (C ABI: include/yb_gpu_scan.h)
Oracle library:   oracle/liborcl.so            (TEST INFRASTRUCTURE ONLY:
    only tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
    call the oracle — see oracle/orcl.h.)
"""
import ctypes as C
import os

_ROOT = os.path.dirname(os.path.abspath(__file__))
_PRODUCT_SO = os.environ.get(
    "YBG_SO", os.path.join(_ROOT, "yugabyte-db_amd", "libybgpu.so"))
_ORACLE_SO = os.path.join(_ROOT, "oracle", "liborcl.so")

MAX_COLS = 32
MAX_KEYCOLS = 8
MAX_PREDS = 8
MAX_AGGS = 8
MAX_HT = 16

# dtypes (ybg_dtype_t / orcl_dtype_t share values)
T_BOOL, T_INT8, T_INT16, T_INT32, T_INT64 = 0, 1, 2, 3, 4
T_UINT32, T_UINT64, T_FLOAT, T_DOUBLE, T_STRING = 5, 6, 7, 8, 9
KT_INT64, KT_INT32, KT_STRING = 0, 1, 2
ENC_SHARED_PREFIX, ENC_THREE_SHARED_PARTS = 0, 1
PRED_GT, PRED_GE, PRED_LT, PRED_LE, PRED_EQ, PRED_NE, PRED_IN, \
    PRED_IN_TUPLE, PRED_IN_RANGE = range(9)
(AGG_COUNT, AGG_COUNT_STAR, AGG_SUM_INT64, AGG_SUM_DOUBLE,
 AGG_MIN_INT64, AGG_MAX_INT64, AGG_MIN_DOUBLE, AGG_MAX_DOUBLE) = range(8)


class ValueCol(C.Structure):
    _fields_ = [("column_id", C.c_int32), ("dtype", C.c_int32),
                ("nullable", C.c_int32)]


class Schema(C.Structure):
    _fields_ = [
        ("has_hash", C.c_int32),
        ("num_hash_cols", C.c_int32),
        ("num_range_cols", C.c_int32),
        ("key_types", C.c_int32 * MAX_KEYCOLS),
        ("num_value_cols", C.c_int32),
        ("value_cols", ValueCol * MAX_COLS),
    ]


class ReadTime(C.Structure):
    _fields_ = [
        ("read", C.c_uint8 * MAX_HT), ("read_len", C.c_int32),
        ("local_limit", C.c_uint8 * MAX_HT), ("local_limit_len", C.c_int32),
        ("global_limit", C.c_uint8 * MAX_HT), ("global_limit_len", C.c_int32),
    ]


class Pred(C.Structure):
    _fields_ = [
        ("is_key_col", C.c_int32), ("col", C.c_int32), ("op", C.c_int32),
        ("datum", C.c_uint64), ("bytes", C.POINTER(C.c_uint8)),
        ("bytes_len", C.c_uint64),
    ]


class Agg(C.Structure):
    _fields_ = [("op", C.c_int32), ("col", C.c_int32)]


class ScanSpec(C.Structure):
    _fields_ = [
        ("schema", Schema),
        ("kv_format", C.c_int32),
        ("read_time", ReadTime),
        ("num_preds", C.c_int32),
        ("preds", Pred * MAX_PREDS),
        ("num_aggs", C.c_int32),
        ("aggs", Agg * MAX_AGGS),
        ("lower_bound", C.POINTER(C.c_uint8)), ("lower_bound_len", C.c_uint64),
        ("upper_bound", C.POINTER(C.c_uint8)), ("upper_bound_len", C.c_uint64),
        ("emit_rows", C.c_int32),
        ("row_limit", C.c_uint64),
        ("group_col", C.c_int32),  # 0 = none, else 1 + value column index
        ("backward", C.c_int32),   # descending delivery order
        ("expect_versions", C.c_int32),  # MVCC-heavy hint (kernel shape)
    ]


class AggResult(C.Structure):
    _fields_ = [("value_i64", C.c_int64), ("value_f64", C.c_double),
                ("is_null", C.c_int32), ("pad_", C.c_int32)]


class ScanResult(C.Structure):
    _fields_ = [("rows_scanned", C.c_uint64), ("rows_matched", C.c_uint64),
                ("entries_seen", C.c_uint64), ("aggs", AggResult * MAX_AGGS),
                ("restart_ht", C.c_uint8 * MAX_HT),
                ("restart_ht_len", C.c_uint32), ("pad2_", C.c_uint32)]


class RowBatch(C.Structure):
    _fields_ = [
        ("n_rows", C.c_uint64), ("n_key_cols", C.c_uint64),
        ("n_value_cols", C.c_uint64),
        ("sort_key", C.POINTER(C.c_uint64)),
        ("key_datums", C.POINTER(C.c_uint64)),
        ("datums", C.POINTER(C.c_uint64)),
        ("null_masks", C.POINTER(C.c_uint32)),
        ("hashes", C.POINTER(C.c_uint16)),
        ("varlen", C.POINTER(C.c_uint8)),
        ("varlen_size", C.c_uint64),
    ]


class Key(C.Structure):
    _fields_ = [
        ("hash", C.c_uint16),
        ("datums", C.c_uint64 * MAX_KEYCOLS),
        ("strs", C.POINTER(C.c_uint8) * MAX_KEYCOLS),
        ("str_lens", C.c_uint64 * MAX_KEYCOLS),
    ]


class RowVals(C.Structure):
    _fields_ = [
        ("datums", C.c_uint64 * MAX_COLS),
        ("strs", C.POINTER(C.c_uint8) * MAX_COLS),
        ("str_lens", C.c_uint64 * MAX_COLS),
        ("null", C.c_uint8 * MAX_COLS),
    ]


class GenParams(C.Structure):
    _fields_ = [
        ("rows", C.c_uint64), ("seed", C.c_uint64),
        ("packed_version", C.c_int32), ("kv_format", C.c_int32),
        ("block_size", C.c_uint32), ("restart_interval", C.c_int32),
        ("versions", C.c_int32),
        ("ht_base_micros", C.c_uint64), ("ht_step_micros", C.c_uint64),
        ("nthreads", C.c_int32),
        ("group_mod", C.c_uint64),
    ]


def _sig(lib, name, res, args):
    f = getattr(lib, name)
    f.restype = res
    f.argtypes = args
    return f


_product = None
_oracle = None


def product():
    """The product library (GPU path + generator)."""
    global _product
    if _product is None:
        _product = C.CDLL(_PRODUCT_SO)
    return _product


def oracle():
    """The CPU oracle — TEST INFRASTRUCTURE ONLY."""
    global _oracle
    if _oracle is None:
        _oracle = C.CDLL(_ORACLE_SO)
    return _oracle


def make_schema(key_types, value_cols, has_hash=True, num_hash_cols=1):
    """value_cols: list of (column_id, dtype, nullable)."""
    s = Schema()
    s.has_hash = 1 if has_hash else 0
    s.num_hash_cols = num_hash_cols if has_hash else 0
    s.num_range_cols = len(key_types) - s.num_hash_cols
    for i, kt in enumerate(key_types):
        s.key_types[i] = kt
    s.num_value_cols = len(value_cols)
    for i, (cid, dt, nul) in enumerate(value_cols):
        s.value_cols[i] = ValueCol(cid, dt, nul)
    return s


def read_time(read_micros, local_micros=None, global_micros=None, lib=None):
    rt = ReadTime()
    lib = lib or product()
    f = _sig(lib, "ybg_read_time_init", None,
             [C.POINTER(ReadTime), C.c_uint64, C.c_uint64, C.c_uint64])
    local_micros = read_micros if local_micros is None else local_micros
    global_micros = local_micros if global_micros is None else global_micros
    f(C.byref(rt), read_micros << 12, local_micros << 12, global_micros << 12)
    return rt


def generate(schema, rows, seed=42, packed_version=2,
             kv_format=ENC_THREE_SHARED_PARTS, block_size=4096,
             restart_interval=16, versions=1, ht_base_micros=1_600_000_000_000_000,
             ht_step_micros=1000, nthreads=0, group_mod=0):
    """Run the multithreaded dataset generator. Returns (data, offsets,
    n_blocks, total_bytes, n_entries); data/offsets are ctypes pointers owned
    by the caller (freed via ybg_free at process exit — we keep them)."""
    lib = product()
    f = _sig(lib, "ybg_generate", C.c_int,
             [C.POINTER(Schema), C.POINTER(GenParams),
              C.POINTER(C.POINTER(C.c_uint8)), C.POINTER(C.POINTER(C.c_uint64)),
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
              C.POINTER(C.c_uint64)])
    p = GenParams(rows, seed, packed_version, kv_format, block_size,
                  restart_interval, versions, ht_base_micros, ht_step_micros,
                  nthreads, group_mod)
    data = C.POINTER(C.c_uint8)()
    offsets = C.POINTER(C.c_uint64)()
    n_blocks = C.c_uint64()
    total = C.c_uint64()
    n_entries = C.c_uint64()
    rc = f(C.byref(schema), C.byref(p), C.byref(data), C.byref(offsets),
           C.byref(n_blocks), C.byref(total), C.byref(n_entries))
    assert rc == 0
    return data, offsets, n_blocks.value, total.value, n_entries.value


class Builder:
    """Low-level tablet builder (entries must be added in key order)."""

    def __init__(self, schema, kv_format=ENC_THREE_SHARED_PARTS,
                 block_size=4096, restart_interval=16):
        lib = product()
        self._lib = lib
        self._schema = schema
        self._create = _sig(lib, "ybg_builder_create", C.c_void_p,
                            [C.POINTER(Schema), C.c_int, C.c_size_t, C.c_int])
        self._h = self._create(C.byref(schema), kv_format, block_size,
                               restart_interval)
        self._add_packed = _sig(lib, "ybg_builder_add_packed_row", C.c_int,
                                [C.c_void_p, C.POINTER(Key), C.c_uint64,
                                 C.c_uint32, C.c_uint64, C.c_int,
                                 C.POINTER(RowVals)])
        self._add_col = _sig(lib, "ybg_builder_add_column_update", C.c_int,
                             [C.c_void_p, C.POINTER(Key), C.c_int, C.c_uint64,
                              C.c_uint32, C.c_uint64, C.c_uint64,
                              C.POINTER(C.c_uint8), C.c_uint64, C.c_int])
        self._add_tomb = _sig(lib, "ybg_builder_add_row_tombstone", C.c_int,
                              [C.c_void_p, C.POINTER(Key), C.c_uint64,
                               C.c_uint32, C.c_uint64])
        self._finish = _sig(lib, "ybg_builder_finish", C.c_int,
                            [C.c_void_p, C.POINTER(C.POINTER(C.c_uint8)),
                             C.POINTER(C.POINTER(C.c_uint64)),
                             C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
                             C.POINTER(C.c_uint64)])
        self._seq = 1 << 50
        self._keepalive = []

    def _key(self, hash_=0, datums=(), strs=()):
        k = Key()
        k.hash = hash_
        for i, d in enumerate(datums):
            k.datums[i] = d & 0xFFFFFFFFFFFFFFFF
        for i, s in enumerate(strs):
            if s is not None:
                buf = C.create_string_buffer(s, len(s))
                self._keepalive.append(buf)
                k.strs[i] = C.cast(buf, C.POINTER(C.c_uint8))
                k.str_lens[i] = len(s)
        return k

    def add_packed_row(self, ht_micros, values, hash_=0, key_datums=(),
                       key_strs=(), write_id=0, packed_version=2, seq=None,
                       logical=0):
        """values: list of (dtype, value-or-None[null]) per schema order;
        ints as int, doubles as float, strings as bytes."""
        k = self._key(hash_, key_datums, key_strs)
        v = RowVals()
        import struct
        for i, (dt, val) in enumerate(values):
            if val is None:
                v.null[i] = 1
            elif dt == T_DOUBLE:
                v.datums[i] = struct.unpack("<Q", struct.pack("<d", val))[0]
            elif dt == T_FLOAT:
                v.datums[i] = struct.unpack("<I", struct.pack("<f", val))[0]
            elif dt == T_STRING:
                buf = C.create_string_buffer(val, len(val))
                self._keepalive.append(buf)
                v.strs[i] = C.cast(buf, C.POINTER(C.c_uint8))
                v.str_lens[i] = len(val)
            else:
                v.datums[i] = val & 0xFFFFFFFFFFFFFFFF
        seq = self._next_seq(seq)
        rc = self._add_packed(self._h, C.byref(k), (ht_micros << 12) | logical,
                              write_id, seq, packed_version, C.byref(v))
        assert rc == 0

    def add_column_update(self, ht_micros, col_idx, value, hash_=0,
                          key_datums=(), key_strs=(), write_id=0, seq=None,
                          null=False, logical=0):
        import struct
        k = self._key(hash_, key_datums, key_strs)
        datum = 0
        sp = None
        slen = 0
        dt = self._schema.value_cols[col_idx].dtype
        if value is None:
            null = True
        elif dt == T_DOUBLE:
            datum = struct.unpack("<Q", struct.pack("<d", value))[0]
        elif dt == T_STRING:
            buf = C.create_string_buffer(value, len(value))
            self._keepalive.append(buf)
            sp = C.cast(buf, C.POINTER(C.c_uint8))
            slen = len(value)
        else:
            datum = value & 0xFFFFFFFFFFFFFFFF
        seq = self._next_seq(seq)
        rc = self._add_col(self._h, C.byref(k), col_idx,
                           (ht_micros << 12) | logical, write_id, seq, datum,
                           sp, slen, 1 if null else 0)
        assert rc == 0

    def add_row_tombstone(self, ht_micros, hash_=0, key_datums=(), key_strs=(),
                          write_id=0, seq=None, logical=0):
        k = self._key(hash_, key_datums, key_strs)
        seq = self._next_seq(seq)
        rc = self._add_tomb(self._h, C.byref(k), (ht_micros << 12) | logical,
                            write_id, seq)
        assert rc == 0

    def _next_seq(self, seq):
        if seq is None:
            seq = self._seq
            self._seq += 1
        return seq

    def finish(self):
        data = C.POINTER(C.c_uint8)()
        offsets = C.POINTER(C.c_uint64)()
        n_blocks = C.c_uint64()
        total = C.c_uint64()
        n_entries = C.c_uint64()
        rc = self._finish(self._h, C.byref(data), C.byref(offsets),
                          C.byref(n_blocks), C.byref(total), C.byref(n_entries))
        assert rc == 0
        return data, offsets, n_blocks.value, total.value, n_entries.value

    def finish_sst(self, compression=0):
        """Finish as a complete BlockBasedTable SST file; returns
        (data_ptr, total_bytes, n_blocks, n_entries). compression: 0 none,
        1 snappy (blocks that do not shrink stay uncompressed)."""
        lib = product()
        f = _sig(lib, "ybg_builder_finish_sst2", C.c_int,
                 [C.c_void_p, C.c_int, C.POINTER(C.POINTER(C.c_uint8)),
                  C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
                  C.POINTER(C.c_uint64)])
        data = C.POINTER(C.c_uint8)()
        total = C.c_uint64()
        n_blocks = C.c_uint64()
        n_entries = C.c_uint64()
        rc = f(self._h, compression, C.byref(data), C.byref(total),
               C.byref(n_blocks), C.byref(n_entries))
        assert rc == 0
        return data, total.value, n_blocks.value, n_entries.value


def sst_index(file_ptr, size=None, verify=True, cap=1 << 20):
    """Parse an SST file's footer + index: data-block (offset, size) lists.
    Raises RuntimeError on corrupt input. file_ptr: ctypes pointer or
    bytes."""
    lib = product()
    f = _sig(lib, "ybg_sst_index", C.c_int,
             [C.POINTER(C.c_uint8), C.c_uint64, C.c_int,
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64), C.c_uint64,
              C.POINTER(C.c_uint64)])
    if isinstance(file_ptr, (bytes, bytearray)):
        buf = (C.c_uint8 * len(file_ptr)).from_buffer_copy(file_ptr)
        file_ptr = C.cast(buf, C.POINTER(C.c_uint8))
        size = len(buf)
    offs = (C.c_uint64 * cap)()
    szs = (C.c_uint64 * cap)()
    n = C.c_uint64()
    rc = f(file_ptr, size, 1 if verify else 0, offs, szs, cap, C.byref(n))
    if rc:
        err = _sig(lib, "yb_gpu_last_error", C.c_char_p, [])
        raise RuntimeError("sst_index: rc=%d %s" % (rc, err().decode()))
    return list(offs[:n.value]), list(szs[:n.value])


# ---------------------------------------------------------------------------
# Oracle wrappers (TEST INFRASTRUCTURE ONLY)
# ---------------------------------------------------------------------------

class OrclRow(C.Structure):
    _fields_ = [
        ("key_datums", C.c_uint64 * MAX_KEYCOLS),
        ("key_str", C.POINTER(C.c_uint8) * MAX_KEYCOLS),
        ("key_str_len", C.c_uint32 * MAX_KEYCOLS),
        ("datums", C.c_uint64 * MAX_COLS),
        ("strp", C.POINTER(C.c_uint8) * MAX_COLS),
        ("strlen_", C.c_uint32 * MAX_COLS),
        ("null_mask", C.c_uint32),
        ("seq_in_scan", C.c_uint64),
    ]


ORCL_ROW_CB = C.CFUNCTYPE(C.c_int, C.POINTER(OrclRow), C.c_void_p)


class OrclValueCol(C.Structure):
    _fields_ = [("column_id", C.c_int32), ("dtype", C.c_int32),
                ("nullable", C.c_int32)]


class OrclSchema(C.Structure):
    _fields_ = [
        ("has_hash", C.c_int), ("num_hash_cols", C.c_int),
        ("num_range_cols", C.c_int), ("key_types", C.c_int * MAX_KEYCOLS),
        ("num_value_cols", C.c_int), ("value_cols", OrclValueCol * MAX_COLS),
    ]


class OrclReadTime(C.Structure):
    _fields_ = [
        ("read", C.c_uint8 * MAX_HT), ("read_len", C.c_size_t),
        ("local_limit", C.c_uint8 * MAX_HT), ("local_limit_len", C.c_size_t),
        ("global_limit", C.c_uint8 * MAX_HT), ("global_limit_len", C.c_size_t),
    ]


class OrclPred(C.Structure):
    _fields_ = [("is_key_col", C.c_int), ("col", C.c_int), ("op", C.c_int),
                ("datum", C.c_uint64), ("bytes", C.POINTER(C.c_uint8)),
                ("bytes_len", C.c_size_t)]


class OrclAgg(C.Structure):
    _fields_ = [("op", C.c_int), ("col", C.c_int)]


class OrclScanSpec(C.Structure):
    _fields_ = [
        ("read_time", OrclReadTime),
        ("num_preds", C.c_int), ("preds", OrclPred * MAX_PREDS),
        ("num_aggs", C.c_int), ("aggs", OrclAgg * MAX_AGGS),
        ("lower_bound", C.POINTER(C.c_uint8)), ("lower_bound_len", C.c_size_t),
        ("upper_bound", C.POINTER(C.c_uint8)), ("upper_bound_len", C.c_size_t),
    ]


class OrclAggResult(C.Structure):
    _fields_ = [("value_i64", C.c_int64), ("value_f64", C.c_double),
                ("is_null", C.c_int)]


class OrclScanResult(C.Structure):
    _fields_ = [("rows_scanned", C.c_uint64), ("rows_matched", C.c_uint64),
                ("entries_seen", C.c_uint64),
                ("aggs", OrclAggResult * MAX_AGGS),
                ("restart_ht", C.c_uint8 * MAX_HT),
                ("restart_ht_len", C.c_uint32), ("pad2_", C.c_uint32)]


def sim_scan(spec, data, offsets, n_blocks):
    """Run the host SIMULATOR of the GPU per-interval algorithm
    (scan_device.h via ybg_sim_scan) — TEST INFRASTRUCTURE; sequentially
    executes the exact device code path on CPU."""
    lib = product()
    f = _sig(lib, "ybg_sim_scan", C.c_int,
             [C.POINTER(ScanSpec), C.POINTER(C.c_uint8),
              C.POINTER(C.c_uint64), C.c_uint64, C.POINTER(ScanResult)])
    res = ScanResult()
    rc = f(C.byref(spec), data, offsets, n_blocks, C.byref(res))
    if rc != 0:
        raise RuntimeError(f"ybg_sim_scan failed rc={rc}")
    return res


def sim_scan_fast(spec, data, offsets, n_blocks):
    """Host simulator of the SPECIALIZED fast batch scanner
    (scan_batch_fast + general fallback per aborted batch) — TEST
    INFRASTRUCTURE. Returns (ScanResult, n_fallback_batches); raises on
    ineligible specs (rc 9)."""
    lib = product()
    f = _sig(lib, "ybg_sim_scan_fast", C.c_int,
             [C.POINTER(ScanSpec), C.POINTER(C.c_uint8),
              C.POINTER(C.c_uint64), C.c_uint64, C.POINTER(ScanResult),
              C.POINTER(C.c_uint64)])
    res = ScanResult()
    nf = C.c_uint64()
    rc = f(C.byref(spec), data, offsets, n_blocks, C.byref(res),
           C.byref(nf))
    if rc != 0:
        raise RuntimeError(f"ybg_sim_scan_fast failed rc={rc}")
    return res, nf.value


def sim_emit(spec, data, offsets, n_blocks, row_cap=1 << 20,
             varlen_cap=1 << 24):
    """Host-simulator row emission (exact device emit path, serial) —
    TEST INFRASTRUCTURE. Returns list of rows sorted by sort_key:
    (key_datums tuple, value tuple with None for NULL / bytes for strings)."""
    lib = product()
    f = _sig(lib, "ybg_sim_emit", C.c_int,
             [C.POINTER(ScanSpec), C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
              C.c_uint64, C.c_uint64, C.POINTER(C.c_uint64),
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64),
              C.POINTER(C.c_uint32), C.POINTER(C.c_uint8), C.c_uint64,
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint64)])
    nk = spec.schema.num_hash_cols + spec.schema.num_range_cols
    nc = spec.schema.num_value_cols
    sort_key = (C.c_uint64 * row_cap)()
    key_datums = (C.c_uint64 * (row_cap * max(nk, 1)))()
    datums = (C.c_uint64 * (row_cap * max(nc, 1)))()
    null_masks = (C.c_uint32 * row_cap)()
    varlen = (C.c_uint8 * varlen_cap)()
    n_rows = C.c_uint64()
    vl = C.c_uint64()
    rc = f(C.byref(spec), data, offsets, n_blocks, row_cap, sort_key,
           key_datums, datums, null_masks, varlen, varlen_cap,
           C.byref(n_rows), C.byref(vl))
    if rc != 0:
        raise RuntimeError(f"ybg_sim_emit rc={rc}")
    return decode_batch_rows(spec.schema, n_rows.value, sort_key, key_datums,
                             datums, null_masks, varlen)


def decode_batch_rows(schema, n_rows, sort_key, key_datums, datums,
                      null_masks, varlen):
    """Decode emitted rows into python tuples, sorted by sort_key."""
    nk = schema.num_hash_cols + schema.num_range_cols
    nc = schema.num_value_cols
    rows = []
    for r in range(n_rows):
        kd = []
        for c in range(nk):
            d = key_datums[r * max(nk, 1) + c]
            if schema.key_types[c] == KT_STRING:
                off = d & ((1 << 40) - 1)
                ln = d >> 40
                kd.append(bytes(varlen[off:off + ln]))
            else:
                kd.append(d)
        vals = []
        nm = null_masks[r]
        for c in range(nc):
            if (nm >> c) & 1:
                vals.append(None)
            elif schema.value_cols[c].dtype == T_STRING:
                d = datums[r * max(nc, 1) + c]
                off = d & ((1 << 40) - 1)
                ln = d >> 40
                vals.append(bytes(varlen[off:off + ln]))
            else:
                vals.append(datums[r * max(nc, 1) + c])
        rows.append((sort_key[r], tuple(kd), tuple(vals)))
    rows.sort(key=lambda t: t[0])
    return [(kd, vals) for (_, kd, vals) in rows]


def _decode_groups(schema, group_col, n, keys, vals, cnts, key_bytes,
                   num_aggs, aggs):
    """Decode group arrays into {key: (vals...)}; key None for the NULL
    group; doubles decoded from bit patterns."""
    import struct
    out = {}
    is_str = schema.value_cols[group_col].dtype == T_STRING
    for g in range(n):
        kv = keys[g]
        if kv == (1 << 64) - 1:
            k = None
        elif is_str:
            off, ln = kv & ((1 << 40) - 1), kv >> 40
            k = bytes(key_bytes[off:off + ln])
        else:
            k = kv
        row = []
        for a in range(num_aggs):
            c = cnts[g * MAX_AGGS + a]
            v = vals[g * MAX_AGGS + a]
            if c == 0:
                row.append(None)
            elif aggs[a].op in (AGG_SUM_DOUBLE, AGG_MIN_DOUBLE,
                                AGG_MAX_DOUBLE):
                row.append(struct.unpack("<d", struct.pack("<q", v))[0])
            else:
                row.append(v)
        out[k] = tuple(row)
    return out


def sim_group(spec, data, offsets, n_blocks, cap=1 << 20,
              key_bytes_cap=1 << 24, return_restart=False):
    """Host-simulator GROUP BY — TEST INFRASTRUCTURE."""
    lib = product()
    f = _sig(lib, "ybg_sim_group", C.c_int,
             [C.POINTER(ScanSpec), C.POINTER(C.c_uint8), C.POINTER(C.c_uint64),
              C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(C.c_int64),
              C.POINTER(C.c_uint64), C.POINTER(C.c_uint8), C.c_uint64,
              C.c_uint64, C.POINTER(C.c_uint64), C.POINTER(C.c_uint8),
              C.POINTER(C.c_uint32)])
    keys = (C.c_uint64 * cap)()
    vals = (C.c_int64 * (cap * MAX_AGGS))()
    cnts = (C.c_uint64 * (cap * MAX_AGGS))()
    kb = (C.c_uint8 * key_bytes_cap)()
    n = C.c_uint64()
    rht = (C.c_uint8 * MAX_HT)()
    rlen = C.c_uint32()
    rc = f(C.byref(spec), data, offsets, n_blocks, keys, vals, cnts, kb,
           key_bytes_cap, cap, C.byref(n), rht, C.byref(rlen))
    if rc != 0:
        raise RuntimeError(f"ybg_sim_group rc={rc}")
    groups = _decode_groups(spec.schema, spec.group_col - 1, n.value, keys,
                            vals, cnts, kb, spec.num_aggs, spec.aggs)
    if return_restart:
        return groups, bytes(rht[:rlen.value])
    return groups


def orcl_group(data, offsets, n_blocks, schema, spec, group_col,
               kv_format=ENC_THREE_SHARED_PARTS, cap=1 << 20,
               key_bytes_cap=1 << 24, num_aggs=None, aggs=None):
    """Oracle GROUP BY — TEST INFRASTRUCTURE."""
    lib = oracle()
    f = _sig(lib, "orcl_group_scan", C.c_int,
             [C.POINTER(C.POINTER(C.c_uint8)), C.POINTER(C.c_size_t),
              C.c_size_t, C.c_int, C.POINTER(OrclSchema),
              C.POINTER(OrclScanSpec), C.c_int, C.POINTER(C.c_uint64),
              C.POINTER(C.c_int64), C.POINTER(C.c_uint64),
              C.POINTER(C.c_uint8), C.c_size_t, C.c_size_t,
              C.POINTER(C.c_size_t)])
    base = C.cast(data, C.c_void_p).value
    blocks = (C.POINTER(C.c_uint8) * n_blocks)()
    sizes = (C.c_size_t * n_blocks)()
    for i in range(n_blocks):
        blocks[i] = C.cast(base + offsets[i], C.POINTER(C.c_uint8))
        sizes[i] = offsets[i + 1] - offsets[i]
    keys = (C.c_uint64 * cap)()
    vals = (C.c_int64 * (cap * MAX_AGGS))()
    cnts = (C.c_uint64 * (cap * MAX_AGGS))()
    kb = (C.c_uint8 * key_bytes_cap)()
    n = C.c_size_t()
    osc = schema
    rc = f(blocks, sizes, n_blocks, kv_format, C.byref(osc), C.byref(spec),
           group_col, keys, vals, cnts, kb, key_bytes_cap, cap, C.byref(n))
    if rc != 0:
        raise RuntimeError(f"orcl_group_scan rc={rc}")

    class _S:
        pass

    # adapt orcl schema for the decoder
    sch = _S()
    sch.value_cols = osc.value_cols
    return _decode_groups(sch, group_col, n.value, keys, vals, cnts, kb,
                          spec.num_aggs, spec.aggs)


def orcl_schema_from(schema):
    o = OrclSchema()
    o.has_hash = schema.has_hash
    o.num_hash_cols = schema.num_hash_cols
    o.num_range_cols = schema.num_range_cols
    for i in range(MAX_KEYCOLS):
        o.key_types[i] = schema.key_types[i]
    o.num_value_cols = schema.num_value_cols
    for i in range(schema.num_value_cols):
        o.value_cols[i] = OrclValueCol(schema.value_cols[i].column_id,
                                       schema.value_cols[i].dtype,
                                       schema.value_cols[i].nullable)
    return o


def orcl_read_time(read_micros, local_micros=None, global_micros=None,
                   logical=0):
    lib = oracle()
    rt = OrclReadTime()
    f = _sig(lib, "orcl_read_time_init", None,
             [C.POINTER(OrclReadTime), C.c_uint64, C.c_uint64, C.c_uint64])
    local_micros = read_micros if local_micros is None else local_micros
    global_micros = local_micros if global_micros is None else global_micros
    f(C.byref(rt), (read_micros << 12) | logical, local_micros << 12,
      global_micros << 12)
    return rt


def orcl_scan(data, offsets, n_blocks, schema, spec, kv_format=ENC_THREE_SHARED_PARTS,
              collect_rows=False, block_lo=0):
    """Run the oracle over a block array (as returned by generate()/Builder).
    `block_lo` starts the scan at a later block (contiguous sub-range; rows
    straddling the cut are attributed per-range — measurement use only).
    Returns (OrclScanResult, rows or None). TEST INFRASTRUCTURE ONLY."""
    lib = oracle()
    f = _sig(lib, "orcl_scan", C.c_int,
             [C.POINTER(C.POINTER(C.c_uint8)), C.POINTER(C.c_size_t),
              C.c_size_t, C.c_int, C.POINTER(OrclSchema),
              C.POINTER(OrclScanSpec), C.POINTER(OrclScanResult),
              ORCL_ROW_CB, C.c_void_p])
    base = C.cast(data, C.c_void_p).value
    blocks = (C.POINTER(C.c_uint8) * n_blocks)()
    sizes = (C.c_size_t * n_blocks)()
    for i in range(n_blocks):
        blocks[i] = C.cast(base + offsets[block_lo + i], C.POINTER(C.c_uint8))
        sizes[i] = offsets[block_lo + i + 1] - offsets[block_lo + i]
    res = OrclScanResult()
    rows = [] if collect_rows else None

    if collect_rows:
        sc = schema

        def cb(rowp, _arg):
            r = rowp.contents
            vals = []
            for i in range(sc.num_value_cols):
                if (r.null_mask >> i) & 1:
                    vals.append(None)
                elif sc.value_cols[i].dtype == T_STRING:
                    vals.append(C.string_at(r.strp[i], r.strlen_[i]))
                else:
                    vals.append(r.datums[i])
            nk = sc.num_hash_cols + sc.num_range_cols
            kd = []
            for i in range(nk):
                if sc.key_types[i] == KT_STRING:
                    kd.append(C.string_at(r.key_str[i], r.key_str_len[i]))
                else:
                    kd.append(r.key_datums[i])
            rows.append((tuple(kd), tuple(vals)))
            return 0

        cbf = ORCL_ROW_CB(cb)
    else:
        cbf = C.cast(None, ORCL_ROW_CB)
    rc = f(blocks, sizes, n_blocks, kv_format, C.byref(schema), C.byref(spec),
           C.byref(res), cbf, None)
    if rc != 0:
        raise RuntimeError(f"orcl_scan failed rc={rc}")
    return res, rows


def snappy_compress(data):
    """Host snappy compressor (generator codec; tests)."""
    lib = product()
    f = _sig(lib, "ybg_snappy_compress", C.c_int64,
             [C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint8),
              C.c_uint64])
    src = (C.c_uint8 * len(data)).from_buffer_copy(data)
    cap = len(data) * 2 + 64
    dst = (C.c_uint8 * cap)()
    n = f(src, len(data), dst, cap)
    assert n > 0
    return bytes(dst[:n])


def snappy_uncompress(data, cap):
    """Host snappy decompressor (same code the GPU kernel runs)."""
    lib = product()
    f = _sig(lib, "ybg_snappy_uncompress", C.c_int64,
             [C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint8),
              C.c_uint64])
    src = (C.c_uint8 * len(data)).from_buffer_copy(data)
    dst = (C.c_uint8 * cap)()
    n = f(src, len(data), dst, cap)
    if n < 0:
        raise RuntimeError("snappy_uncompress failed")
    return bytes(dst[:n])


class TxnStatus(C.Structure):
    """ybg_txn_status_t / orcl_txn_status_t (same layout)."""
    _fields_ = [("txn_id", C.c_uint32), ("status", C.c_int32),
                ("commit_ht", C.c_uint64)]


TXN_PENDING, TXN_COMMITTED, TXN_ABORTED = 0, 1, 2


def make_txns(table):
    """table: dict txn_id -> ("committed", commit_ht_micros, logical?) |
    "pending" | "aborted". Returns a ctypes TxnStatus array."""
    arr = (TxnStatus * max(len(table), 1))()
    for i, (tid, st) in enumerate(sorted(table.items())):
        arr[i].txn_id = tid
        if st == "pending":
            arr[i].status = TXN_PENDING
        elif st == "aborted":
            arr[i].status = TXN_ABORTED
        else:
            arr[i].status = TXN_COMMITTED
            ht = st[1] if isinstance(st, tuple) else st
            arr[i].commit_ht = ht << 12
    return arr, len(table)


class Intents:
    """Intent-stream builder (provisional records of in-flight
    transactions; post-DecodeStrongWriteIntent form — see
    include/yb_gpu_scan.h)."""

    def __init__(self, schema):
        lib = product()
        self._lib = lib
        self._schema = schema
        create = _sig(lib, "ybg_intents_create", C.c_void_p,
                      [C.POINTER(Schema)])
        self._h = create(C.byref(schema))
        self._add_packed = _sig(lib, "ybg_intents_add_packed_row", C.c_int,
                                [C.c_void_p, C.POINTER(Key), C.c_uint64,
                                 C.c_uint32, C.c_uint32, C.c_int,
                                 C.POINTER(RowVals)])
        self._add_col = _sig(lib, "ybg_intents_add_column_update", C.c_int,
                             [C.c_void_p, C.POINTER(Key), C.c_int,
                              C.c_uint64, C.c_uint32, C.c_uint32,
                              C.c_uint64, C.POINTER(C.c_uint8), C.c_uint64,
                              C.c_int])
        self._add_tomb = _sig(lib, "ybg_intents_add_row_tombstone", C.c_int,
                              [C.c_void_p, C.POINTER(Key), C.c_uint64,
                               C.c_uint32, C.c_uint32])
        self._data = _sig(lib, "ybg_intents_data", C.c_int,
                          [C.c_void_p, C.POINTER(C.POINTER(C.c_uint8)),
                           C.POINTER(C.c_uint64)])
        self._keepalive = []

    def _key(self, hash_=0, datums=(), strs=()):
        k = Key()
        k.hash = hash_
        for i, d in enumerate(datums):
            k.datums[i] = d & 0xFFFFFFFFFFFFFFFF
        for i, s in enumerate(strs):
            if s is not None:
                buf = C.create_string_buffer(s, len(s))
                self._keepalive.append(buf)
                k.strs[i] = C.cast(buf, C.POINTER(C.c_uint8))
                k.str_lens[i] = len(s)
        return k

    def add_packed_row(self, write_ht_micros, txn_id, values, hash_=0,
                       key_datums=(), key_strs=(), write_id=0,
                       packed_version=2, logical=0):
        import struct
        k = self._key(hash_, key_datums, key_strs)
        v = RowVals()
        for i, (dt, val) in enumerate(values):
            if val is None:
                v.null[i] = 1
            elif dt == T_DOUBLE:
                v.datums[i] = struct.unpack("<Q", struct.pack("<d", val))[0]
            elif dt == T_FLOAT:
                v.datums[i] = struct.unpack("<I", struct.pack("<f", val))[0]
            elif dt == T_STRING:
                buf = C.create_string_buffer(val, len(val))
                self._keepalive.append(buf)
                v.strs[i] = C.cast(buf, C.POINTER(C.c_uint8))
                v.str_lens[i] = len(val)
            else:
                v.datums[i] = val & 0xFFFFFFFFFFFFFFFF
        rc = self._add_packed(self._h, C.byref(k),
                              (write_ht_micros << 12) | logical, write_id,
                              txn_id, packed_version, C.byref(v))
        assert rc == 0

    def add_column_update(self, write_ht_micros, txn_id, col_idx, value,
                          hash_=0, key_datums=(), key_strs=(), write_id=0,
                          null=False, logical=0):
        import struct
        k = self._key(hash_, key_datums, key_strs)
        datum = 0
        sp = None
        slen = 0
        dt = self._schema.value_cols[col_idx].dtype
        if value is None:
            null = True
        elif dt == T_DOUBLE:
            datum = struct.unpack("<Q", struct.pack("<d", value))[0]
        elif dt == T_STRING:
            buf = C.create_string_buffer(value, len(value))
            self._keepalive.append(buf)
            sp = C.cast(buf, C.POINTER(C.c_uint8))
            slen = len(value)
        else:
            datum = value & 0xFFFFFFFFFFFFFFFF
        rc = self._add_col(self._h, C.byref(k), col_idx,
                           (write_ht_micros << 12) | logical, write_id,
                           txn_id, datum, sp, slen, 1 if null else 0)
        assert rc == 0

    def add_row_tombstone(self, write_ht_micros, txn_id, hash_=0,
                          key_datums=(), key_strs=(), write_id=0,
                          logical=0):
        k = self._key(hash_, key_datums, key_strs)
        rc = self._add_tomb(self._h, C.byref(k),
                            (write_ht_micros << 12) | logical, write_id,
                            txn_id)
        assert rc == 0

    def blob(self):
        p = C.POINTER(C.c_uint8)()
        n = C.c_uint64()
        rc = self._data(self._h, C.byref(p), C.byref(n))
        assert rc == 0
        return p, n.value


def merge_intents(data, offsets, n_blocks, intents_blob, blob_len, txns,
                  n_txns, kv_format=ENC_THREE_SHARED_PARTS):
    """Product-side feed-time resolve+merge (ybg_merge_intents). Returns
    (merged_data, merged_offsets, n_blocks, total) as ctypes buffers the
    caller can pass to feed/sim APIs (freed at process exit)."""
    lib = product()
    f = _sig(lib, "ybg_merge_intents", C.c_int,
             [C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
              C.c_int, C.POINTER(C.c_uint8), C.c_uint64,
              C.POINTER(TxnStatus), C.c_uint32,
              C.POINTER(C.POINTER(C.c_uint8)),
              C.POINTER(C.POINTER(C.c_uint64)), C.POINTER(C.c_uint64),
              C.POINTER(C.c_uint64)])
    ob = C.POINTER(C.c_uint8)()
    oo = C.POINTER(C.c_uint64)()
    on = C.c_uint64()
    ot = C.c_uint64()
    rc = f(data, offsets, n_blocks, kv_format, intents_blob, blob_len, txns,
           n_txns, C.byref(ob), C.byref(oo), C.byref(on), C.byref(ot))
    if rc != 0:
        raise RuntimeError(f"ybg_merge_intents rc={rc}")
    return ob, oo, on.value, ot.value


def orcl_scan_intents(data, offsets, n_blocks, schema, spec, intents_blob,
                      blob_len, txns, n_txns,
                      kv_format=ENC_THREE_SHARED_PARTS):
    """Oracle runtime two-stream merge scan — TEST INFRASTRUCTURE."""
    lib = oracle()
    f = _sig(lib, "orcl_scan_intents", C.c_int,
             [C.POINTER(C.POINTER(C.c_uint8)), C.POINTER(C.c_size_t),
              C.c_size_t, C.c_int, C.POINTER(OrclSchema),
              C.POINTER(OrclScanSpec), C.POINTER(C.c_uint8), C.c_size_t,
              C.POINTER(TxnStatus), C.c_uint32,
              C.POINTER(OrclScanResult), ORCL_ROW_CB, C.c_void_p])
    base = C.cast(data, C.c_void_p).value
    blocks = (C.POINTER(C.c_uint8) * n_blocks)()
    sizes = (C.c_size_t * n_blocks)()
    for i in range(n_blocks):
        blocks[i] = C.cast(base + offsets[i], C.POINTER(C.c_uint8))
        sizes[i] = offsets[i + 1] - offsets[i]
    res = OrclScanResult()
    rc = f(blocks, sizes, n_blocks, kv_format, C.byref(schema),
           C.byref(spec), intents_blob, blob_len, txns, n_txns,
           C.byref(res), C.cast(None, ORCL_ROW_CB), None)
    if rc != 0:
        raise RuntimeError(f"orcl_scan_intents rc={rc}")
    return res


def encode_dockey(schema, hash_=0, key_datums=(), key_strs=()):
    """Encode a DocKey from key-column datums (ybg_encode_dockey) —
    bounds / paging-state construction helper."""
    lib = product()
    f = _sig(lib, "ybg_encode_dockey", C.c_size_t,
             [C.POINTER(Schema), C.POINTER(Key), C.POINTER(C.c_uint8),
              C.c_size_t])
    k = Key()
    k.hash = hash_
    keep = []
    for i, d in enumerate(key_datums):
        k.datums[i] = d & 0xFFFFFFFFFFFFFFFF
    for i, s in enumerate(key_strs):
        if s is not None:
            buf = C.create_string_buffer(s, len(s))
            keep.append(buf)
            k.strs[i] = C.cast(buf, C.POINTER(C.c_uint8))
            k.str_lens[i] = len(s)
    out = (C.c_uint8 * 512)()
    n = f(C.byref(schema), C.byref(k), out, 512)
    assert n > 0
    return bytes(out[:n])


def lz4_compress(data):
    """Host lz4 compressor (generator codec; tests)."""
    lib = product()
    f = _sig(lib, "ybg_lz4_compress", C.c_int64,
             [C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint8),
              C.c_uint64])
    src = (C.c_uint8 * len(data)).from_buffer_copy(data)
    cap = len(data) * 2 + 64
    dst = (C.c_uint8 * cap)()
    n = f(src, len(data), dst, cap)
    assert n > 0
    return bytes(dst[:n])



def lz4_uncompress(data, cap):
    """Host lz4 decompressor (same code the GPU kernel runs)."""
    lib = product()
    f = _sig(lib, "ybg_lz4_uncompress", C.c_int64,
             [C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint8),
              C.c_uint64])
    src = (C.c_uint8 * len(data)).from_buffer_copy(data)
    dst = (C.c_uint8 * cap)()
    n = f(src, len(data), dst, cap)
    if n < 0:
        raise RuntimeError("lz4_uncompress failed")
    return bytes(dst[:n])




# ---- SST bloom filter (docdb_filter_policy / rocksdb FixedSizeFilter) ----

def filter_slice_size():
    lib = product()
    return _sig(lib, "ybg_filter_slice_size", C.c_uint64, [])()


def filter_from_sst(data, offsets, n_blocks, kv_format=1):
    """Build the tablet's bloom filter (concatenated fixed-size slices)."""
    lib = product()
    f = _sig(lib, "ybg_filter_from_sst", C.c_int,
             [C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
              C.c_int, C.POINTER(C.c_uint8), C.c_uint64,
              C.POINTER(C.c_uint64)])
    cap = filter_slice_size() * (2 + n_blocks)
    out = (C.c_uint8 * cap)()
    out_len = C.c_uint64(0)
    rc = f(data, offsets, n_blocks, kv_format, out, cap, C.byref(out_len))
    if rc:
        raise RuntimeError(f"ybg_filter_from_sst rc={rc}")
    return bytes(out[:out_len.value])


def filter_may_match(filt, key):
    lib = product()
    f = _sig(lib, "ybg_filter_may_match", C.c_int,
             [C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint8),
              C.c_uint64])
    fb = (C.c_uint8 * len(filt)).from_buffer_copy(filt)
    kb = (C.c_uint8 * len(key)).from_buffer_copy(key)
    return f(fb, len(filt), kb, len(key))


def filter_key_prefix_len(key):
    lib = product()
    f = _sig(lib, "ybg_filter_key_prefix_len", C.c_uint64,
             [C.POINTER(C.c_uint8), C.c_uint64])
    kb = (C.c_uint8 * len(key)).from_buffer_copy(key)
    return f(kb, len(key))
