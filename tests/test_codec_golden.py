"""Pin the oracle and the product generator's codecs to the reference's own
golden byte vectors (tests/golden/reference_vectors.json, each citing the
reference test file:line), plus randomized round-trips mirroring
src/yb/util/fast_varint-test.cc and rocksdb/table/block_test.cc."""
import ctypes as C
import json
import os
import random
import struct

import ybgpu as y

HERE = os.path.dirname(os.path.abspath(__file__))
VEC = json.load(open(os.path.join(HERE, "golden", "reference_vectors.json")))


def _orcl():
    return y.oracle()


def test_signed_varint_golden():
    lib = _orcl()
    enc = lib.orcl_svarint_encode
    enc.restype = C.c_size_t
    enc.argtypes = [C.c_int64, C.POINTER(C.c_uint8)]
    buf = (C.c_uint8 * 16)()
    for v in VEC["signed_varint"]["vectors"]:
        n = enc(v["v"], buf)
        assert bytes(buf[:n]).hex() == v["hex"], v


def test_signed_varint_roundtrip():
    # mirrors fast_varint-test.cc:110-136 (powers of two, random, dense range)
    lib = _orcl()
    enc = lib.orcl_svarint_encode
    enc.restype = C.c_size_t
    enc.argtypes = [C.c_int64, C.POINTER(C.c_uint8)]
    dec = lib.orcl_svarint_decode
    dec.restype = C.c_size_t
    dec.argtypes = [C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_int64)]
    buf = (C.c_uint8 * 16)()
    out = C.c_int64()
    vals = [0, -1, 2**63 - 1, 2**63 - 2, -2**63, -2**63 + 1]
    for i in range(63):
        vals += [1 << i, (1 << i) + 1, (1 << i) - 1, -(1 << i)]
    rng = random.Random(1234)
    vals += [rng.getrandbits(64) - 2**63 for _ in range(2000)]
    vals += list(range(-1000, 1001))
    for v in vals:
        n = enc(v, buf)
        m = dec(buf, n, C.byref(out))
        assert m == n and out.value == v, v
        # ordering property: descending encode of a,b compares reversed
    # encoded DESCENDING varints (enc(-v)) compare opposite to values
    pairs = [(rng.getrandbits(40), rng.getrandbits(40)) for _ in range(500)]
    for a, b in pairs:
        na = enc(-a, buf)
        ea = bytes(buf[:na])
        nb = enc(-b, buf)
        eb = bytes(buf[:nb])
        if a < b:
            assert ea > eb
        elif a > b:
            assert ea < eb


def test_unsigned_varint_roundtrip():
    lib = _orcl()
    enc = lib.orcl_uvarint_encode
    enc.restype = C.c_size_t
    enc.argtypes = [C.c_uint64, C.POINTER(C.c_uint8)]
    dec = lib.orcl_uvarint_decode
    dec.restype = C.c_size_t
    dec.argtypes = [C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_uint64)]
    buf = (C.c_uint8 * 16)()
    out = C.c_uint64()
    rng = random.Random(99)
    vals = [0, 1, 127, 128, 2**64 - 1] + [rng.getrandbits(rng.randint(1, 64))
                                          for _ in range(3000)]
    for v in vals:
        n = enc(v, buf)
        m = dec(buf, n, C.byref(out))
        assert m == n and out.value == v, v


def test_doc_hybrid_time_golden():
    # doc_key-test.cc:373-390: HT physical=1000us logical=0 write_id=0
    lib = _orcl()
    enc = lib.orcl_dht_encode
    enc.restype = C.c_size_t
    enc.argtypes = [C.c_uint64, C.c_uint32, C.POINTER(C.c_uint8)]
    g = VEC["doc_hybrid_time_micros_1000_w0"]
    buf = (C.c_uint8 * 16)()
    n = enc((g["micros"] << 12) | g["logical"], g["write_id"], buf)
    assert bytes(buf[:n]).hex() == g["hex_after_hash_byte"]


def test_doc_hybrid_time_roundtrip_and_order():
    lib = _orcl()
    enc = lib.orcl_dht_encode
    enc.restype = C.c_size_t
    enc.argtypes = [C.c_uint64, C.c_uint32, C.POINTER(C.c_uint8)]
    dec = lib.orcl_dht_decode
    dec.restype = C.c_size_t
    dec.argtypes = [C.POINTER(C.c_uint8), C.c_size_t, C.POINTER(C.c_uint64),
                    C.POINTER(C.c_uint32)]
    buf = (C.c_uint8 * 16)()
    ht_out = C.c_uint64()
    wid_out = C.c_uint32()
    rng = random.Random(7)
    cases = []
    for _ in range(2000):
        micros = 1_500_000_000_000_000 + rng.getrandbits(40)
        logical = rng.getrandbits(12)
        wid = rng.choice([0, 1, rng.getrandbits(20), 0xFFFFFFFF])
        cases.append(((micros << 12) | logical, wid))
    encs = []
    for ht, wid in cases:
        n = enc(ht, wid, buf)
        b = bytes(buf[:n])
        # size in low 5 bits of last byte (doc_hybrid_time.cc:66-74)
        assert (b[-1] & 0x1F) == n
        m = dec(buf, n, C.byref(ht_out), C.byref(wid_out))
        assert m == n and ht_out.value == ht and wid_out.value == wid
        encs.append(((ht, wid), b))
    # reversed ordering: ht1 < ht2 (same wid) => enc1 > enc2
    # (doc_hybrid_time.h:89-95)
    for _ in range(1000):
        (a, wa), ea = rng.choice(encs)
        (b_, wb), eb = rng.choice(encs)
        if (a, wa) < (b_, wb):
            assert ea > eb
        elif (a, wa) > (b_, wb):
            assert ea < eb


def test_dockey_encoding_golden():
    """Product generator's DocKey bytes == the reference golden vectors
    (doc_key-test.cc TestDocKeyEncoding), extracted from a built block."""
    g = VEC["dockey_range_only"]
    schema = y.make_schema(
        [y.KT_STRING, y.KT_INT64, y.KT_STRING, y.KT_INT64],
        [(10, y.T_INT64, 1)], has_hash=False, num_hash_cols=0)
    b = y.Builder(schema)
    b.add_packed_row(
        1000, [(y.T_INT64, 5)],
        key_datums=(0, g["key_ints"][0], 0, g["key_ints"][1]),
        key_strs=(g["key_strs"][0].encode(), None, g["key_strs"][1].encode(),
                  None))
    data, offsets, n_blocks, total, n_entries = b.finish()
    assert n_blocks == 1 and n_entries == 1
    blk = C.string_at(data, total)
    expected_dockey = bytes.fromhex(g["hex"])
    # entry: header byte(s) + full key (restart) + value; locate the user key
    # by searching for the dockey bytes
    assert expected_dockey in blk
    # and the '#' + golden HT must follow it (same HT as the golden vector)
    ht = VEC["doc_hybrid_time_micros_1000_w0"]
    assert (expected_dockey + b"#" + bytes.fromhex(ht["hex_after_hash_byte"])
            ) in blk


def test_dockey_hashed_golden():
    g = VEC["dockey_hashed"]
    schema = y.make_schema(
        [y.KT_STRING, y.KT_STRING, y.KT_STRING, y.KT_INT64, y.KT_STRING,
         y.KT_INT64],
        [(10, y.T_INT64, 1)], has_hash=True, num_hash_cols=2)
    b = y.Builder(schema)
    b.add_packed_row(
        1000, [(y.T_INT64, 5)], hash_=g["hash"],
        key_datums=(0, 0, 0, 1000, 0, 2000),
        key_strs=(b"hashed1", b"hashed2", b"range1", None, b"range2", None))
    data, offsets, n_blocks, total, n_entries = b.finish()
    blk = C.string_at(data, total)
    assert bytes.fromhex(g["hex"]) in blk


def _roundtrip_block(fmt, keys_values):
    """Feed raw internal keys through the product BlockBuilder and decode with
    the oracle block iterator; mirrors rocksdb/table/block_test.cc:92 and
    :667-720 (EncodeThreeSharedParts iterator round-trip)."""
    lib = y.product()
    create = lib.ybg_builder_create
    create.restype = C.c_void_p
    create.argtypes = [C.POINTER(y.Schema), C.c_int, C.c_size_t, C.c_int]
    add_raw = lib.ybg_builder_add_raw
    add_raw.restype = C.c_int
    add_raw.argtypes = [C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t,
                        C.c_uint64, C.POINTER(C.c_uint8), C.c_size_t]
    fin = lib.ybg_builder_finish
    fin.restype = C.c_int
    fin.argtypes = [C.c_void_p, C.POINTER(C.POINTER(C.c_uint8)),
                    C.POINTER(C.POINTER(C.c_uint64)), C.POINTER(C.c_uint64),
                    C.POINTER(C.c_uint64), C.POINTER(C.c_uint64)]
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    h = create(C.byref(schema), fmt, 1 << 30, 16)  # one huge block
    for (ukey, seq, val) in keys_values:
        kb = (C.c_uint8 * len(ukey)).from_buffer_copy(ukey)
        vb = (C.c_uint8 * max(len(val), 1)).from_buffer_copy(val or b"\0")
        assert add_raw(h, kb, len(ukey), seq, vb, len(val)) == 0
    data = C.POINTER(C.c_uint8)()
    offsets = C.POINTER(C.c_uint64)()
    nb = C.c_uint64()
    tot = C.c_uint64()
    ne = C.c_uint64()
    assert fin(h, C.byref(data), C.byref(offsets), C.byref(nb), C.byref(tot),
               C.byref(ne)) == 0
    assert nb.value == 1

    # oracle block iterator
    olib = _orcl()

    class It(C.Structure):
        _fields_ = [("data", C.POINTER(C.c_uint8)), ("size", C.c_size_t),
                    ("restarts_offset", C.c_size_t),
                    ("num_restarts", C.c_uint32), ("fmt", C.c_int),
                    ("next_offset", C.c_size_t), ("key", C.c_uint8 * 256),
                    ("key_len", C.c_size_t), ("value", C.POINTER(C.c_uint8)),
                    ("value_len", C.c_size_t)]

    init = olib.orcl_block_iter_init
    init.restype = C.c_int
    init.argtypes = [C.POINTER(It), C.POINTER(C.c_uint8), C.c_size_t, C.c_int]
    nxt = olib.orcl_block_iter_next
    nxt.restype = C.c_int
    nxt.argtypes = [C.POINTER(It)]
    it = It()
    assert init(C.byref(it), data, tot.value, fmt) == 0
    out = []
    while True:
        r = nxt(C.byref(it))
        if r == 0:
            break
        assert r == 1, f"block decode error {r}"
        k = bytes(it.key[:it.key_len])
        v = C.string_at(it.value, it.value_len)
        out.append((k, v))
    return out


def _internal_key(ukey, seq, typ=1):
    return ukey + struct.pack("<Q", (seq << 8) | typ)


def test_block_roundtrip_both_formats_random():
    rng = random.Random(4242)
    # DocDB-shaped keys: shared prefix + varying mid + varying tail + 8B suffix
    entries = []
    seq = 1 << 50
    prefix = b"G\x12\x34I\x80\x00\x00\x00"
    for i in range(1000):
        mid = struct.pack(">I", i // 7)
        tail = bytes([rng.randint(0, 255) for _ in range(rng.randint(4, 12))])
        ukey = prefix + mid + b"!!#" + tail
        val = bytes([rng.randint(0, 255) for _ in range(rng.randint(0, 40))])
        entries.append((ukey, seq, val))
        seq += rng.choice([1, 1, 1, 7])
    # keys must be sorted for the builder; sort by ukey then seq desc
    entries.sort(key=lambda e: (e[0], -e[1]))
    expect = [(_internal_key(u, s), v) for (u, s, v) in entries]
    for fmt in (y.ENC_SHARED_PREFIX, y.ENC_THREE_SHARED_PARTS):
        got = _roundtrip_block(fmt, entries)
        assert got == expect, f"fmt={fmt}"


def test_block_roundtrip_pathological_keys():
    # empty-ish keys, max-shared keys, identical user keys with different seq,
    # long runs of equal bytes (middle-at-end quirk, block_builder.cc:118-141)
    rng = random.Random(7)
    entries = []
    seq = 1 << 50
    base = b"AAAAAAAABBBBBBBB"
    for i in range(300):
        ukey = base + bytes([i & 0xFF]) + base
        entries.append((ukey, seq, b"v" * (i % 5)))
        seq += 1
    entries.sort(key=lambda e: (e[0], -e[1]))
    expect = [(_internal_key(u, s), v) for (u, s, v) in entries]
    for fmt in (y.ENC_SHARED_PREFIX, y.ENC_THREE_SHARED_PARTS):
        got = _roundtrip_block(fmt, entries)
        assert got == expect, f"fmt={fmt}"


# ---------------------------------------------------------------------------
# Hand-derived three_shared_parts golden bytes. Every expected byte below is
# derived BY HAND from the reference's encoding definition
# (rocksdb/table/block_builder_internal.h:100-239 EncodeThreeSharedPartsSizes,
# block_builder.cc:118-262 CalculateComponents/MaxSharedRun), independently
# of both the generator and the oracle — pinning the encoder's component
# CHOICES (maximal shared run, left/right preference, last8 reuse/+0x100)
# with literal bytes instead of round-trips.
# ---------------------------------------------------------------------------

def _raw_builder_block(entries):
    import ctypes as C
    lib = y.product()
    create = y._sig(lib, "ybg_builder_create", C.c_void_p,
                    [C.POINTER(y.Schema), C.c_int, C.c_size_t, C.c_int])
    add_raw = y._sig(lib, "ybg_builder_add_raw", C.c_int,
                     [C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t,
                      C.c_uint64, C.POINTER(C.c_uint8), C.c_uint64])
    fin = y._sig(lib, "ybg_builder_finish", C.c_int,
                 [C.c_void_p, C.POINTER(C.POINTER(C.c_uint8)),
                  C.POINTER(C.POINTER(C.c_uint64)), C.POINTER(C.c_uint64),
                  C.POINTER(C.c_uint64), C.POINTER(C.c_uint64)])
    schema = y.make_schema([y.KT_INT64], [(10, y.T_INT64, 1)])
    h = create(C.byref(schema), y.ENC_THREE_SHARED_PARTS, 1 << 20, 16)
    for ukey, seq, val in entries:
        kb = (C.c_uint8 * len(ukey)).from_buffer_copy(ukey)
        vb = (C.c_uint8 * len(val)).from_buffer_copy(val)
        assert add_raw(h, kb, len(ukey), seq, vb, len(val)) == 0
    data = C.POINTER(C.c_uint8)()
    offs = C.POINTER(C.c_uint64)()
    nb = C.c_uint64()
    tot = C.c_uint64()
    ne = C.c_uint64()
    assert fin(h, C.byref(data), C.byref(offs), C.byref(nb), C.byref(tot),
               C.byref(ne)) == 0
    assert nb.value == 1
    return bytes(data[0:tot.value])


def _le64(v):
    return v.to_bytes(8, "little")


def test_three_shared_parts_hand_derived_bytes():
    """Restart (case 2.0), 2.1.1 with ns2=0, FREQUENT, and the general
    2.1.2 form with an ns1 delta — literal expected bytes."""
    blk = _raw_builder_block([
        (b"ABCDEFGHI", 0x50, b"v0"),
        (b"ABCDEFGHJ", 0x51, b"v1"),       # last8 = prev + 0x100 (inc)
        (b"ABCDXFGHK", 0x52, b"v2"),       # frequent: ns1=1 mid="FGH" ns2=1
        (b"ABQRSTUVWXYZ", 0x52, b"v3"),    # general: ns1=10 prev_ns1=7 d1=3
    ])
    want = b"".join([
        # e0 restart: e1=len(v0)<<2=0x08; no-reuse e2=key_size<<1 with
        # key_size=17 (9 user + 8 suffix) -> 0x22; full key; value
        bytes([0x08, 0x22]) + b"ABCDEFGHI" + _le64((0x50 << 8) | 1) + b"v0",
        # e1 2.1.1: sp=8 ("ABCDEFGH"), lhs rest "I" vs rhs rest "J":
        # no shared run -> ns1=1, ns2=0, deltas 0; last8 incremented.
        # e1=(2<<2)|inc(2)=0x0A; e2=0b01|ns1<<3=0x09; leb(sp)=0x08; "J"
        bytes([0x0A, 0x09, 0x08]) + b"J" + b"v1",
        # e2 FREQUENT: sp=4, prev rest "EFGHJ" vs "XFGHK": max shared run
        # "FGH" from the left -> ns1=1 ("X"), middle=3, ns2=1 ("K"),
        # deltas 0, last8 incremented again.
        # e1=(2<<2)|2|1=0x0B; leb(sp)=0x04; "X"; "K"
        bytes([0x0B, 0x04]) + b"XK" + b"v2",
        # e3 GENERAL 2.1.2: sp=2, lhs rest (minus reused last8)
        # "CDXFGHK" (7) vs rhs "QRSTUVWXYZ" (10): no shared run ->
        # ns1=10, prev_ns1=7 -> ns1_delta=3; ns2=0; last8 reused equal
        # (same seq). e1=2<<2=0x08; e2=0b11|reuse<<2|d1<<3=0x0F;
        # leb(ns1)=0x0A; svarint(+3)=0x83; leb(sp)=0x02; 10 key bytes
        bytes([0x08, 0x0F, 0x0A, 0x83, 0x02]) + b"QRSTUVWXYZ" + b"v3",
    ])
    assert blk[:len(want)] == want, (blk[:len(want)].hex(), want.hex())
    # trailer: one restart at 0
    assert blk[-8:] == (0).to_bytes(4, "little") + (1).to_bytes(4, "little")
    # and the oracle's decoder reconstructs the exact keys (independent
    # restatement of block.cc:287-346)
    _assert_oracle_roundtrip(blk, [
        b"ABCDEFGHI" + _le64((0x50 << 8) | 1),
        b"ABCDEFGHJ" + _le64((0x51 << 8) | 1),
        b"ABCDXFGHK" + _le64((0x52 << 8) | 1),
        b"ABQRSTUVWXYZ" + _le64((0x52 << 8) | 1),
    ], [b"v0", b"v1", b"v2", b"v3"])


def test_three_shared_parts_d2_case_bytes():
    """Case 2.1.1 with ns2_delta == 1 (the d2 bit): the new key's second
    non-shared part is one byte longer than the previous key's."""
    blk = _raw_builder_block([
        (b"AAEFGHJ", 0x60, b"w0"),
        (b"AAWFGHKL", 0x61, b"w1"),
    ])
    want = b"".join([
        # restart: key_size = 15 -> e2 = 0x1E
        bytes([0x08, 0x1E]) + b"AAEFGHJ" + _le64((0x60 << 8) | 1) + b"w0",
        # sp=2; lhs "EFGHJ" vs rhs "WFGHKL": left run "FGH" (len 3) at
        # offset 1 -> ns1=1 ("W"), middle=3, prev_ns2=1 ("J"),
        # ns2=2 ("KL") -> d2=1; inc. e1=(2<<2)|2=0x0A;
        # e2=0b01|d2<<2|ns1<<3|ns2<<6=0x8D; leb(sp)=0x02
        bytes([0x0A, 0x8D, 0x02]) + b"W" + b"KL" + b"w1",
    ])
    assert blk[:len(want)] == want, (blk[:len(want)].hex(), want.hex())
    _assert_oracle_roundtrip(blk, [
        b"AAEFGHJ" + _le64((0x60 << 8) | 1),
        b"AAWFGHKL" + _le64((0x61 << 8) | 1),
    ], [b"w0", b"w1"])


def _assert_oracle_roundtrip(blk, keys, values):
    import ctypes as C
    lib = y.product()
    f = y._sig(lib, "ybg_decode_block", C.c_int,
               [C.POINTER(C.c_uint8), C.c_uint64, C.c_int,
                C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint32),
                C.POINTER(C.c_uint8), C.c_uint64, C.POINTER(C.c_uint32),
                C.c_uint64, C.POINTER(C.c_uint64)])
    data = (C.c_uint8 * len(blk)).from_buffer_copy(blk)
    kb = (C.c_uint8 * 4096)()
    vb = (C.c_uint8 * 4096)()
    kl = (C.c_uint32 * 64)()
    vl = (C.c_uint32 * 64)()
    n = C.c_uint64()
    assert f(data, len(blk), y.ENC_THREE_SHARED_PARTS, kb, 4096, kl, vb,
             4096, vl, 64, C.byref(n)) == 0
    assert n.value == len(keys)
    ko = vo = 0
    for i, (k, v) in enumerate(zip(keys, values)):
        assert bytes(kb[ko:ko + kl[i]]) == k, i
        assert bytes(vb[vo:vo + vl[i]]) == v, i
        ko += kl[i]
        vo += vl[i]
