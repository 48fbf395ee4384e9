"""Pruning/bloom correctness properties, CPU-only (no GPU needed):

1. Block selection (`ybg_test_block_selection`, the index-separator
   pruning `yb_gpu_scan_feed_blocks` applies) must be SOUND: scanning
   only the selected blocks yields bit-identical results to scanning
   every block, for random sorted tablets x random DocKey bounds and
   leading-key option predicates.
2. The bloom filter's reject is a PROOF of absence: whenever
   `ybg_filter_may_match` returns 0 for a point scan's pinned prefix,
   the full scan of the tablet finds nothing in those bounds.

Both mirror the reference's contract that filters/pruning are pure
optimizations (rocksdb/table/index_reader.cc seek pruning,
docdb/docdb_filter_policy.cc bloom: false positives allowed, false
negatives never)."""
import ctypes as C
import random

import ybgpu as y


def _selection(spec, data, offsets, n_blocks):
    lib = y.product()
    f = lib.ybg_test_block_selection
    f.restype = C.c_int
    f.argtypes = [C.POINTER(y.ScanSpec), C.POINTER(C.c_uint8),
                  C.POINTER(C.c_uint64), C.c_uint64, C.POINTER(C.c_uint8)]
    keep = (C.c_uint8 * n_blocks)()
    rc = f(C.byref(spec), data, offsets, n_blocks, keep)
    return rc, list(keep)


def _subset(data, offsets, n_blocks, keep):
    """Concatenate the kept blocks into a fresh (data, offsets) pair."""
    blob = bytearray()
    offs = [0]
    for b in range(n_blocks):
        if not keep[b]:
            continue
        blob += bytes(
            C.cast(C.addressof(data.contents) + offsets[b],
                   C.POINTER(C.c_uint8 * (offsets[b + 1] - offsets[b])))
            .contents)
        offs.append(len(blob))
    if len(offs) == 1:  # nothing selected: scan one block anyway (the
        blob += bytes(  # feed path's well-formed-table fallback)
            C.cast(C.addressof(data.contents),
                   C.POINTER(C.c_uint8 * (offsets[1] - offsets[0])))
            .contents)
        offs.append(len(blob))
    d = (C.c_uint8 * (len(blob) + 256)).from_buffer_copy(
        bytes(blob) + b"\x00" * 256)
    o = (C.c_uint64 * len(offs))(*offs)
    return C.cast(d, C.POINTER(C.c_uint8)), o, len(offs) - 1, (d, o)


def _res(r):
    return (r.rows_scanned, r.rows_matched,
            tuple((r.aggs[i].value_i64, r.aggs[i].is_null)
                  for i in range(2)))


def test_prune_soundness_fuzz():
    rng = random.Random(777)
    sc = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    for it in range(8):
        rows = rng.randint(500, 4000)
        hmod = rng.choice([17, 100, 997])
        b = y.Builder(sc)
        seq = 1 << 50
        for r in sorted(range(rows), key=lambda r: (r % hmod, r)):
            seq += 1
            b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % hmod,
                             key_datums=(r,), seq=seq)
        data, offsets, n_blocks, total = b.finish()[:4]
        for _ in range(6):
            lo_r = rng.randrange(rows)
            hi_r = rng.randrange(rows)
            lower = y.encode_dockey(sc, hash_=lo_r % hmod,
                                    key_datums=(lo_r,))
            upper = y.encode_dockey(sc, hash_=hi_r % hmod,
                                    key_datums=(hi_r,))
            if rng.random() < 0.3:
                upper = lower + b"\x00"  # point scan
            spec = y.ScanSpec()
            spec.schema = sc
            spec.kv_format = y.ENC_THREE_SHARED_PARTS
            spec.read_time = y.read_time(9000)
            spec.num_aggs = 2
            spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
            spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 0)
            lb = (C.c_uint8 * len(lower)).from_buffer_copy(lower)
            ub = (C.c_uint8 * len(upper)).from_buffer_copy(upper)
            spec.lower_bound, spec.lower_bound_len = lb, len(lower)
            spec.upper_bound, spec.upper_bound_len = ub, len(upper)
            full = _res(y.sim_scan(spec, data, offsets, n_blocks))
            rc, keep = _selection(spec, data, offsets, n_blocks)
            if not rc:
                continue  # selection unavailable for this spec: feed all
            sd, so, sn, _keepalive = _subset(data, offsets, n_blocks, keep)
            pruned = _res(y.sim_scan(spec, sd, so, sn))
            # entry counts differ (fewer blocks walked) — rows/aggs not
            assert pruned == full, (it, lo_r, hi_r, sum(keep), n_blocks)


def test_bloom_reject_is_proof_of_absence():
    rng = random.Random(4242)
    sc = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    rows, hmod = 3000, 211
    b = y.Builder(sc)
    seq = 1 << 50
    for r in sorted(range(rows), key=lambda r: (r % hmod, r)):
        seq += 1
        b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % hmod,
                         key_datums=(r,), seq=seq)
    data, offsets, n_blocks, total = b.finish()[:4]
    filt = y.filter_from_sst(data, offsets, n_blocks)
    rejected = checked = 0
    for _ in range(800):
        hh = rng.randrange(4 * hmod)
        k0 = rng.randrange(2 * rows)
        key = y.encode_dockey(sc, hash_=hh, key_datums=(k0,))
        if y.filter_may_match(filt, key):
            continue
        rejected += 1
        # the filter says "definitely absent": the full scan over the
        # point bounds must find nothing
        upper = key + b"\x00"
        spec = y.ScanSpec()
        spec.schema = sc
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.read_time = y.read_time(9000)
        spec.num_aggs = 1
        spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        lb = (C.c_uint8 * len(key)).from_buffer_copy(key)
        ub = (C.c_uint8 * len(upper)).from_buffer_copy(upper)
        spec.lower_bound, spec.lower_bound_len = lb, len(key)
        spec.upper_bound, spec.upper_bound_len = ub, len(upper)
        res = y.sim_scan(spec, data, offsets, n_blocks)
        assert res.rows_matched == 0, (hh, k0)
        checked += 1
    assert rejected > 200, rejected  # most random absent prefixes reject
    assert checked == rejected


def test_prune_point_scan_on_block_boundaries():
    """Regression: a point scan on a DocKey that opens a block used to
    lose that block — the upper bound key+\\x00 sorts below the block's
    first INTERNAL key (DocKey || '#' DHT || seq) even though its DocKey
    is in range. Selection must compare DocKeys. Exercise the exact
    first-DocKey of EVERY block."""
    sc = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    rows, hmod = 4993, 211
    b = y.Builder(sc)
    seq = 1 << 50
    by_key = sorted(range(rows), key=lambda r: (r % hmod, r))
    for r in by_key:
        seq += 1
        b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % hmod,
                         key_datums=(r,), seq=seq)
    data, offsets, n_blocks, total = b.finish()[:4]
    lib = y.product()
    fk = lib.ybg_block_first_key
    fk.restype = C.c_int
    fk.argtypes = [C.POINTER(C.c_uint8), C.c_uint64, C.c_int,
                   C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint64)]
    for blk in range(n_blocks):
        out = (C.c_uint8 * 256)()
        ln = C.c_uint64()
        rc = fk(C.cast(C.addressof(data.contents) + offsets[blk],
                       C.POINTER(C.c_uint8)),
                offsets[blk + 1] - offsets[blk], 1, out, 256, C.byref(ln))
        assert rc == 0
        ik = bytes(out[:ln.value])
        # DocKey = up to and including the second kGroupEnd
        first = ik.index(b"\x21")
        dk = ik[:ik.index(b"\x21", first + 1) + 1]
        lower, upper = dk, dk + b"\x00"
        spec = y.ScanSpec()
        spec.schema = sc
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.read_time = y.read_time(9000)
        spec.num_aggs = 1
        spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        lb = (C.c_uint8 * len(lower)).from_buffer_copy(lower)
        ub = (C.c_uint8 * len(upper)).from_buffer_copy(upper)
        spec.lower_bound, spec.lower_bound_len = lb, len(lower)
        spec.upper_bound, spec.upper_bound_len = ub, len(upper)
        full = _res(y.sim_scan(spec, data, offsets, n_blocks))
        assert full[1] == 1  # the row exists
        rc2, keep = _selection(spec, data, offsets, n_blocks)
        if not rc2:
            continue
        sd, so, sn, _ka = _subset(data, offsets, n_blocks, keep)
        assert _res(y.sim_scan(spec, sd, so, sn)) == full, blk


def test_prune_options_soundness_fuzz():
    """Same soundness property for the OPTION-driven seek plan: IN /
    IN_RANGE predicates on the leading range key of a range-sharded
    table derive key-prefix ranges that prune blocks — scanning only the
    kept blocks must be bit-identical to scanning all of them."""
    rng = random.Random(31337)
    sc = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)],
                       has_hash=False, num_hash_cols=0)
    for it in range(5):
        rows = rng.randint(1000, 5000)
        b = y.Builder(sc)
        seq = 1 << 50
        for r in range(rows):  # range table: key order == r order
            seq += 1
            b.add_packed_row(5000, [(y.T_INT64, r * 3)], key_datums=(r,),
                             seq=seq)
        data, offsets, n_blocks, total = b.finish()[:4]
        for _ in range(6):
            spec = y.ScanSpec()
            spec.schema = sc
            spec.kv_format = y.ENC_THREE_SHARED_PARTS
            spec.read_time = y.read_time(9000)
            spec.num_aggs = 2
            spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
            spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 0)
            spec.num_preds = 1
            keep_alive = []
            if rng.random() < 0.5:  # IN options on the range key
                opts = sorted(rng.sample(range(-100, rows + 100),
                                         rng.randint(1, 12)))
                blob = b"".join(
                    (v & (2**64 - 1)).to_bytes(8, "little") for v in opts)
                buf = (C.c_uint8 * len(blob)).from_buffer_copy(blob)
                keep_alive.append(buf)
                spec.preds[0] = y.Pred(1, 0, y.PRED_IN, 0, buf, len(blob))
            else:  # IN_RANGE option ranges
                recs = b""
                for _ in range(rng.randint(1, 4)):
                    lo = rng.randint(-100, rows)
                    hi = lo + rng.randint(0, rows // 3)
                    fl = rng.randint(0, 3)
                    recs += ((lo & (2**64 - 1)).to_bytes(8, "little") +
                             (hi & (2**64 - 1)).to_bytes(8, "little") +
                             fl.to_bytes(4, "little") + b"\x00" * 4)
                buf = (C.c_uint8 * len(recs)).from_buffer_copy(recs)
                keep_alive.append(buf)
                spec.preds[0] = y.Pred(1, 0, y.PRED_IN_RANGE, 0, buf,
                                       len(recs))
            # rows_scanned legitimately SHRINKS under option pruning
            # (out-of-option rows in skipped blocks are never visited;
            # an option predicate failing still counts a visited row as
            # scanned) — matched rows and aggregates are the invariant
            def _mres(r):
                return (r.rows_matched,
                        tuple((r.aggs[i].value_i64, r.aggs[i].is_null)
                              for i in range(2)))
            full = _mres(y.sim_scan(spec, data, offsets, n_blocks))
            rc, keep = _selection(spec, data, offsets, n_blocks)
            if not rc:
                continue
            sd, so, sn, _ka = _subset(data, offsets, n_blocks, keep)
            assert _mres(y.sim_scan(spec, sd, so, sn)) == full, (it,)
