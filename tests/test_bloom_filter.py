"""SST bloom filter: exact-bit parity of the rocksdb
FixedSizeFilter restatement (bloom_filter.h) against an INDEPENDENT
Python reimplementation of the cited reference algorithm
(rocksdb/util/hash.cc:32-77, bloom.cc:43-62,384-452), plus the
DocDbAwareV3FilterPolicy key transform and the feed-time pruning
semantics (results identical with and without the filter)."""
import struct

import pytest

import ybgpu as y


# --- independent restatement (test-side; deliberately separate code) ---

def ref_hash(data: bytes, seed=0xBC9F1D34) -> int:
    m, r = 0xC6A4A793, 24
    h = (seed ^ (len(data) * m)) & 0xFFFFFFFF
    i = 0
    while i + 4 <= len(data):
        (w,) = struct.unpack_from("<I", data, i)
        i += 4
        h = (h + w) & 0xFFFFFFFF
        h = (h * m) & 0xFFFFFFFF
        h ^= h >> 16
    tail = data[i:]
    # rocksdb adds tail bytes as SIGNED chars (disk-format quirk)
    def sgn(b):
        return b - 256 if b >= 128 else b
    if len(tail) >= 3:
        h = (h + ((sgn(tail[2]) << 16) & 0xFFFFFFFF)) & 0xFFFFFFFF
    if len(tail) >= 2:
        h = (h + ((sgn(tail[1]) << 8) & 0xFFFFFFFF)) & 0xFFFFFFFF
    if len(tail) >= 1:
        h = (h + sgn(tail[0])) & 0xFFFFFFFF
        h = (h * m) & 0xFFFFFFFF
        h ^= h >> r
    return h


CACHE_LINE = 64
META = 5


def ref_dims(total_bits_req=65536, err=0.01):
    import math
    num_lines = -(-total_bits_req // (CACHE_LINE * 8))
    if num_lines % 2 == 0:
        num_lines = num_lines + 1 if num_lines * CACHE_LINE < 4096 \
            else num_lines - 1
    total_bits = num_lines * CACHE_LINE * 8
    mler = -math.log(err)
    probes = max(1, min(255, int(mler / math.log(2))))
    max_keys = int(total_bits * math.log(2) ** 2 / mler)
    return num_lines, total_bits, probes, max_keys


def ref_build_slice(prefixes, num_lines, total_bits, probes):
    data = bytearray(total_bits // 8 + META)
    for k in prefixes:
        h = ref_hash(k)
        delta = ((h >> 17) | (h << 15)) & 0xFFFFFFFF
        b = (h % num_lines) * (CACHE_LINE * 8)
        for _ in range(probes):
            bit = b + (h % (CACHE_LINE * 8))
            data[bit // 8] |= 1 << (bit % 8)
            h = (h + delta) & 0xFFFFFFFF
    data[total_bits // 8] = probes
    struct.pack_into("<I", data, total_bits // 8 + 1, num_lines)
    return bytes(data)


def _schema():
    return y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])


def _enc_key(schema, hash_, k0):
    return y.encode_dockey(schema, hash_=hash_, key_datums=(k0,))


def test_prefix_extraction_boundaries():
    sc = _schema()
    k = _enc_key(sc, 0x1234, 77)
    plen = y.filter_key_prefix_len(k)
    # 'G' + 2B hash + ('I' + 8B int64) + '!' == 13, includes the group end
    assert plen == 13
    assert k[plen - 1:plen] == b"\x21"
    # unparseable (unknown entry type byte) => 0 => always-match
    assert y.filter_key_prefix_len(b"\x99abc") == 0
    assert y.filter_may_match(b"", b"\x99abc") == 1
    # string hashed component: zero-escaped, terminator included
    sc2 = y.make_schema([y.KT_STRING], [(10, y.T_INT64, 1)])
    k2 = y.encode_dockey(sc2, hash_=7, key_datums=(0,),
                         key_strs=(b"ab\x00cd",))
    plen2 = y.filter_key_prefix_len(k2)
    # G+hash(2) + 'S' + "ab\0\1cd" + 00 00 + '!' = 3+1+6+2+1 = 13
    assert plen2 == 13


def test_golden_bits_vs_independent_restatement():
    """Build a filter from a small tablet and reproduce it BIT-EXACTLY
    with the test-side Python implementation of the cited algorithm."""
    sc = _schema()
    b = y.Builder(sc)
    seq = 1 << 50
    keys = []
    for r in sorted(range(500), key=lambda r: (r % 37, r)):
        seq += 1
        b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % 37,
                         key_datums=(r,), seq=seq)
        keys.append(_enc_key(sc, r % 37, r))
    data, offsets, n_blocks, total = b.finish()[:4]
    filt = y.filter_from_sst(data, offsets, n_blocks)
    num_lines, total_bits, probes, max_keys = ref_dims()
    assert len(filt) == total_bits // 8 + META  # one slice at 500 keys
    # distinct prefixes in tablet key order, consecutive-deduped
    prefixes, last = [], None
    for k in sorted(keys):
        p = bytes(k[: y.filter_key_prefix_len(k)])
        if p != last:
            prefixes.append(p)
            last = p
    ref = ref_build_slice(prefixes, num_lines, total_bits, probes)
    assert filt == ref


def test_may_match_and_fp_rate():
    sc = _schema()
    b = y.Builder(sc)
    seq = 1 << 50
    for r in sorted(range(3000), key=lambda r: (r % 997, r)):
        seq += 1
        b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % 997,
                         key_datums=(r,), seq=seq)
    data, offsets, n_blocks, total = b.finish()[:4]
    filt = y.filter_from_sst(data, offsets, n_blocks)
    # every present prefix (hash code + hashed column value) must match
    for r in range(0, 3000, 7):
        assert y.filter_may_match(filt, _enc_key(sc, r % 997, r)) == 1
    # absent prefixes: false-positive rate must stay near the 1% target
    fp = sum(
        y.filter_may_match(filt, _enc_key(sc, (10_000 + i) % 997,
                                          10_000 + i))
        for i in range(4000))
    assert fp < 4000 * 0.05, fp
    # empty/absent filter never rejects
    assert y.filter_may_match(b"", _enc_key(sc, 5, 1)) == 1


def test_slice_rollover():
    """More distinct prefixes than max_keys => multiple slices; every
    added prefix still matches."""
    sc = _schema()
    b = y.Builder(sc)
    seq = 1 << 50
    n = 9000  # > max_keys (~6.8k) distinct hash prefixes
    for r in range(n):
        seq += 1
        b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r, key_datums=(r,),
                         seq=seq)
    data, offsets, n_blocks, total = b.finish()[:4]
    filt = y.filter_from_sst(data, offsets, n_blocks)
    ssz = y.filter_slice_size()
    assert len(filt) == 2 * ssz
    misses = sum(
        1 for hh in range(0, n, 97)
        if not y.filter_may_match(filt, _enc_key(sc, hh, hh)))
    assert misses == 0


@pytest.mark.gpu
def test_gpu_bloom_pruned_point_scan():
    """End-to-end: a point scan (EQ DocKey bounds) on an absent hash is
    answered empty via the filter with NO blocks fed; a present hash
    scans normally and matches the oracle."""
    import gpu_scan
    if not gpu_scan.gpu_available():
        pytest.fail("no HIP device visible")
    sc = _schema()
    b = y.Builder(sc)
    seq = 1 << 50
    for r in sorted(range(2000), key=lambda r: (r % 100, r)):
        seq += 1
        b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % 100,
                         key_datums=(r,), seq=seq)
    data, offsets, n_blocks, total = b.finish()[:4]
    filt = y.filter_from_sst(data, offsets, n_blocks)

    def point_scan(hh, k0):
        lower = _enc_key(sc, hh, k0)
        upper = lower + b"\x00"  # exclusive successor, same prefix
        spec = y.ScanSpec()
        spec.schema = sc
        spec.kv_format = y.ENC_THREE_SHARED_PARTS
        spec.read_time = y.read_time(9000)
        spec.num_aggs = 1
        spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
        lb = (__import__("ctypes").c_uint8 * len(lower)) \
            .from_buffer_copy(lower)
        ub = (__import__("ctypes").c_uint8 * len(upper)) \
            .from_buffer_copy(upper)
        spec.lower_bound = lb
        spec.lower_bound_len = len(lower)
        spec.upper_bound = ub
        spec.upper_bound_len = len(upper)
        s = gpu_scan.GpuScan(spec)
        s.feed_blocks_bloom(data, offsets, n_blocks, filt)
        s.execute()
        res = s.aggregates()
        s.close()
        return res

    hit = point_scan(42, 42)   # row 42 has hash 42, key 42
    assert hit.rows_matched == 1
    assert hit.aggs[0].value_i64 == 1
    miss = point_scan(7777, 1)  # hash 7777 never written
    assert miss.rows_matched == 0
    assert miss.entries_seen == 0  # nothing was fed, let alone scanned
    assert miss.aggs[0].is_null == 1


def test_filter_abi_edge_cases():
    """ABI error paths: undersized output cap (rc=8), corrupt block
    (rc=3), and broken/odd filter blobs never reject (fail-open, like
    the reference reader's broken-filter handling, bloom.cc:183-188)."""
    import ctypes as C
    lib = y.product()
    f = lib.ybg_filter_from_sst
    f.restype = C.c_int
    f.argtypes = [C.POINTER(C.c_uint8), C.POINTER(C.c_uint64), C.c_uint64,
                  C.c_int, C.POINTER(C.c_uint8), C.c_uint64,
                  C.POINTER(C.c_uint64)]
    sc = _schema()
    b = y.Builder(sc)
    seq = 1 << 50
    for r in range(200):
        seq += 1
        b.add_packed_row(5000, [(y.T_INT64, r)], hash_=r % 7,
                         key_datums=(r,), seq=seq)
    data, offsets, n_blocks, total = b.finish()[:4]
    out = (C.c_uint8 * 16)()
    ln = C.c_uint64(0)
    assert f(data, offsets, n_blocks, 1, out, 16, C.byref(ln)) == 8
    garbage = (C.c_uint8 * 64).from_buffer_copy(b"\x00" * 64)
    goff = (C.c_uint64 * 2)(0, 64)
    big = (C.c_uint8 * 20000)()
    assert f(C.cast(garbage, C.POINTER(C.c_uint8)), goff, 1, 1, big,
             20000, C.byref(ln)) == 3
    key = _enc_key(sc, 3, 3)
    # truncated / non-multiple-of-slice filters fail OPEN (may match)
    assert y.filter_may_match(b"\x01\x02\x03", key) == 1
    assert y.filter_may_match(b"\x00" * 100, key) == 1
