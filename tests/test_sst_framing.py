"""SST (BlockBasedTable) file framing: the generator's finish_sst output
and the product parser (ybg_sst_index / yb_gpu_scan_feed_sst path) checked
against each other, against the raw-block ground truth, and against an
INDEPENDENT pure-Python restatement of the trailer/footer format
(crc32c + mask per rocksdb/util/crc32c.h:51-61, footer layout per
rocksdb/table/format.cc:129-155, magic per
block_based_table_builder.cc:187-198)."""
import ctypes as C
import struct

import pytest

import ybgpu as y

MAGIC = 0x88E241B785F4CFF7
FOOTER_LEN = 53
TRAILER_LEN = 5


def _crc32c(data, crc=0):
    # Castagnoli, reflected poly 0x82F63B78 — bytewise, independent of the
    # C++ implementation under test
    tab = _crc32c.tab
    c = crc ^ 0xFFFFFFFF
    for b in data:
        c = tab[(c ^ b) & 0xFF] ^ (c >> 8)
    return c ^ 0xFFFFFFFF


def _mk_tab():
    tab = []
    for i in range(256):
        c = i
        for _ in range(8):
            c = (0x82F63B78 ^ (c >> 1)) if (c & 1) else (c >> 1)
        tab.append(c)
    return tab


_crc32c.tab = _mk_tab()


def _mask(crc):
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


def _build(rows=4000):
    schema = y.make_schema([y.KT_INT64],
                           [(10 + i, y.T_INT64, 1) for i in range(4)])
    b = y.Builder(schema)
    for r in range(rows):
        b.add_packed_row(1000 + r, [(y.T_INT64, r * 7 + c) for c in range(4)],
                         hash_=r // 64, key_datums=(r,), packed_version=2)
    return schema, b


def test_sst_roundtrip_matches_raw_blocks():
    schema, b = _build()
    data, offsets, n_blocks, total, n_entries = b.finish()
    raw = bytes(C.cast(data, C.POINTER(C.c_uint8 * total)).contents)
    raw_offs = [offsets[i] for i in range(n_blocks + 1)]

    sst_ptr, sst_total, sst_blocks, sst_entries = b.finish_sst()
    assert sst_blocks == n_blocks
    assert sst_entries == n_entries
    sst = bytes(C.cast(sst_ptr, C.POINTER(C.c_uint8 * sst_total)).contents)

    offs, szs = y.sst_index(sst, verify=True)
    assert len(offs) == n_blocks
    # every parsed data block equals the corresponding raw block
    for i in range(n_blocks):
        blk = sst[offs[i]:offs[i] + szs[i]]
        assert blk == raw[raw_offs[i]:raw_offs[i + 1]], f"block {i} differs"


def test_sst_footer_and_trailers_independent_check():
    schema, b = _build(rows=600)
    sst_ptr, sst_total, sst_blocks, _ = b.finish_sst()
    sst = bytes(C.cast(sst_ptr, C.POINTER(C.c_uint8 * sst_total)).contents)

    # footer: last 8 bytes = magic, version 2, checksum byte kCRC32c
    lo, hi = struct.unpack("<II", sst[-8:])
    assert ((hi << 32) | lo) == MAGIC
    version = struct.unpack("<I", sst[-12:-8])[0]
    assert version == 2
    footer = sst[-FOOTER_LEN:]
    assert footer[0] == 1  # kCRC32c

    # independently verify every data block's trailer crc
    offs, szs = y.sst_index(sst, verify=False)
    assert len(offs) == sst_blocks
    for off, sz in zip(offs, szs):
        blk = sst[off:off + sz]
        trailer = sst[off + sz:off + sz + TRAILER_LEN]
        assert trailer[0] == 0  # kNoCompression
        expect = _mask(_crc32c(bytes([trailer[0]]), _crc32c(blk)))
        assert struct.unpack("<I", trailer[1:])[0] == expect


def test_sst_corruption_detected():
    schema, b = _build(rows=600)
    sst_ptr, sst_total, _, _ = b.finish_sst()
    sst = bytearray(
        C.cast(sst_ptr, C.POINTER(C.c_uint8 * sst_total)).contents)

    offs, szs = y.sst_index(bytes(sst), verify=True)  # clean parse first
    # flip one data byte -> checksum verification must fail
    sst2 = bytearray(sst)
    sst2[offs[0] + szs[0] // 2] ^= 0x40
    with pytest.raises(RuntimeError, match="checksum"):
        y.sst_index(bytes(sst2), verify=True)
    # bad magic -> rejected
    sst3 = bytearray(sst)
    sst3[-1] ^= 0xFF
    with pytest.raises(RuntimeError, match="magic"):
        y.sst_index(bytes(sst3), verify=False)


def test_sst_scan_parity_oracle():
    """Scan over the blocks extracted from the SST file equals the oracle
    scan over the directly-built blocks (CPU: host simulator)."""
    schema, b = _build(rows=5000)
    data, offsets, n_blocks, total, _ = b.finish()
    sst_ptr, sst_total, _, _ = b.finish_sst()
    sst = bytes(C.cast(sst_ptr, C.POINTER(C.c_uint8 * sst_total)).contents)
    offs, szs = y.sst_index(sst, verify=True)
    # reassemble the concatenated-block layout the scan consumes
    blocks = b"".join(sst[o:o + s] for o, s in zip(offs, szs))
    boffs = [0]
    for s_ in szs:
        boffs.append(boffs[-1] + s_)
    buf = (C.c_uint8 * (len(blocks) + 48)).from_buffer_copy(
        blocks + b"\0" * 48)
    offarr = (C.c_uint64 * len(boffs))(*boffs)

    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(1_700_000_000_000_000)
    spec.num_preds = 1
    spec.preds[0] = y.Pred(0, 0, y.PRED_GT, 70, None, 0)
    spec.num_aggs = 2
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 3)
    sres = y.sim_scan(spec, buf, offarr, len(offs))

    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(1_700_000_000_000_000)
    ospec.num_preds = 1
    ospec.preds[0] = y.OrclPred(0, 0, y.PRED_GT, 70, None, 0)
    ospec.num_aggs = 2
    ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
    ospec.aggs[1] = y.OrclAgg(y.AGG_SUM_INT64, 3)
    data2, offsets2, nb2, total2, _ = b.finish()
    ores, _ = y.orcl_scan(data2, offsets2, nb2, osc, ospec)
    assert sres.rows_scanned == ores.rows_scanned
    assert sres.rows_matched == ores.rows_matched
    assert sres.aggs[0].value_i64 == ores.aggs[0].value_i64
    assert sres.aggs[1].value_i64 == ores.aggs[1].value_i64


@pytest.mark.gpu
def test_sst_feed_gpu_parity():
    """yb_gpu_scan_feed_sst end-to-end on the GPU vs the CPU oracle."""
    from gpu_scan import GpuScan

    schema, b = _build(rows=30000)
    sst_ptr, sst_total, _, _ = b.finish_sst()

    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(1_700_000_000_000_000)
    spec.num_preds = 1
    spec.preds[0] = y.Pred(0, 1, y.PRED_GE, 1000, None, 0)
    spec.num_aggs = 2
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 2)
    s = GpuScan(spec)
    s.feed_sst(sst_ptr, sst_total, verify=True)
    s.execute()
    res = s.aggregates()

    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(1_700_000_000_000_000)
    ospec.num_preds = 1
    ospec.preds[0] = y.OrclPred(0, 1, y.PRED_GE, 1000, None, 0)
    ospec.num_aggs = 2
    ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
    ospec.aggs[1] = y.OrclAgg(y.AGG_SUM_INT64, 2)
    data, offsets, nb, total, _ = b.finish()
    ores, _ = y.orcl_scan(data, offsets, nb, osc, ospec)
    assert res.rows_scanned == ores.rows_scanned
    assert res.rows_matched == ores.rows_matched
    assert res.aggs[0].value_i64 == ores.aggs[0].value_i64
    assert res.aggs[1].value_i64 == ores.aggs[1].value_i64
    s.close()


def test_snappy_codec_roundtrip():
    """The generator's snappy codec round-trips byte-exactly (the
    decompressor here is the SAME function the GPU kernel runs; the format
    is the public snappy raw-block spec — snappy_dev.h header comment)."""
    import random

    rng = random.Random(7)
    cases = [
        b"",
        b"a",
        b"abcabcabcabcabcabcabcabc" * 40,       # long matches
        bytes(rng.randrange(256) for _ in range(5000)),  # incompressible
        bytes(rng.choice(b"abcd") for _ in range(8000)),  # short matches
        (b"\x00" * 300) + b"xyz" + (b"\x00" * 300),       # RLE overlap
    ]
    for i, data in enumerate(cases):
        comp = y.snappy_compress(data)
        out = y.snappy_uncompress(comp, len(data) + 16)
        assert out == data, f"case {i} round-trip mismatch"


def test_sst_compressed_blocks():
    """Snappy SST: blocks shrink, parse/verify passes, and host
    decompression of every block reproduces the raw blocks byte-exactly."""
    schema, b = _build(rows=4000)
    data, offsets, n_blocks, total, _ = b.finish()
    raw = bytes(C.cast(data, C.POINTER(C.c_uint8 * total)).contents)
    raw_offs = [offsets[i] for i in range(n_blocks + 1)]

    sst_ptr, sst_total, sst_blocks, _ = b.finish_sst(compression=1)
    assert sst_blocks == n_blocks
    assert sst_total < total  # compression shrank the file
    sst = bytes(C.cast(sst_ptr, C.POINTER(C.c_uint8 * sst_total)).contents)
    offs, szs = y.sst_index(sst, verify=True)
    n_comp = 0
    for i in range(n_blocks):
        rb = raw[raw_offs[i]:raw_offs[i + 1]]
        blk = sst[offs[i]:offs[i] + szs[i]]
        t = sst[offs[i] + szs[i]]
        if t == 1:
            n_comp += 1
            assert y.snappy_uncompress(blk, len(rb) + 16) == rb, i
        else:
            assert blk == rb, i
    assert n_comp > 0  # the dataset must actually compress


@pytest.mark.gpu
def test_sst_compressed_feed_gpu_parity():
    """feed_sst on a snappy-compressed SST: the GPU decompression kernel +
    scan equals the CPU oracle over the uncompressed blocks."""
    from gpu_scan import GpuScan

    schema, b = _build(rows=30000)
    sst_ptr, sst_total, _, _ = b.finish_sst(compression=1)

    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(1_700_000_000_000_000)
    spec.num_preds = 1
    spec.preds[0] = y.Pred(0, 1, y.PRED_GE, 1000, None, 0)
    spec.num_aggs = 2
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 2)
    s = GpuScan(spec)
    s.feed_sst(sst_ptr, sst_total, verify=True)
    s.execute()
    res = s.aggregates()

    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(1_700_000_000_000_000)
    ospec.num_preds = 1
    ospec.preds[0] = y.OrclPred(0, 1, y.PRED_GE, 1000, None, 0)
    ospec.num_aggs = 2
    ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
    ospec.aggs[1] = y.OrclAgg(y.AGG_SUM_INT64, 2)
    data, offsets, nb, total, _ = b.finish()
    ores, _ = y.orcl_scan(data, offsets, nb, osc, ospec)
    assert res.rows_scanned == ores.rows_scanned
    assert res.rows_matched == ores.rows_matched
    assert res.aggs[0].value_i64 == ores.aggs[0].value_i64
    assert res.aggs[1].value_i64 == ores.aggs[1].value_i64
    s.close()


def test_lz4_codec_roundtrip():
    """LZ4 codec round-trips byte-exactly (the decompressor is the same
    function the GPU kernel runs; format = the public LZ4 block spec —
    lz4_dev.h header comment)."""
    import random
    rng = random.Random(11)
    cases = [
        b"",
        b"a" * 7,
        b"abcabcabcabcabcabcabcabc" * 60,
        bytes(rng.randrange(256) for _ in range(6000)),
        bytes(rng.choice(b"wxyz") for _ in range(9000)),
        (b"\x11" * 400) + b"tail" + (b"\x11" * 400),
    ]
    for i, data in enumerate(cases):
        comp = y.lz4_compress(data)
        out = y.lz4_uncompress(comp, len(data) + 16)
        assert out == data, f"case {i} round-trip mismatch"


def test_sst_lz4_blocks():
    """LZ4 SST (kLZ4Compression, trailer type 0x4, rocksdb varint32
    raw-length framing): blocks shrink, parse/verify passes, host
    decompression reproduces the raw blocks byte-exactly."""
    schema, b = _build(rows=4000)
    data, offsets, n_blocks, total, _ = b.finish()
    raw = bytes(C.cast(data, C.POINTER(C.c_uint8 * total)).contents)
    raw_offs = [offsets[i] for i in range(n_blocks + 1)]
    sst_ptr, sst_total, sst_blocks, _ = b.finish_sst(compression=4)
    assert sst_blocks == n_blocks and sst_total < total
    sst = bytes(C.cast(sst_ptr, C.POINTER(C.c_uint8 * sst_total)).contents)
    offs, szs = y.sst_index(sst, verify=True)
    n_comp = 0
    for i in range(n_blocks):
        rb = raw[raw_offs[i]:raw_offs[i + 1]]
        blk = sst[offs[i]:offs[i] + szs[i]]
        t = sst[offs[i] + szs[i]]
        if t == 4:
            n_comp += 1
            # strip the varint32 raw-length prefix
            o = 0
            ulen = 0
            sh = 0
            while blk[o] & 0x80:
                ulen |= (blk[o] & 0x7F) << sh
                o += 1
                sh += 7
            ulen |= blk[o] << sh
            o += 1
            assert ulen == len(rb)
            assert y.lz4_uncompress(blk[o:], len(rb) + 16) == rb, i
        else:
            assert blk == rb, i
    assert n_comp > 0


@pytest.mark.gpu
def test_sst_lz4_feed_gpu_parity():
    """feed_sst on an LZ4 SST: device crc32c verify + LZ4 decompression +
    scan equals the CPU oracle over the uncompressed blocks."""
    from gpu_scan import GpuScan
    schema, b = _build(rows=30000)
    sst_ptr, sst_total, _, _ = b.finish_sst(compression=4)
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(1_700_000_000_000_000)
    spec.num_preds = 1
    spec.preds[0] = y.Pred(0, 1, y.PRED_GE, 1000, None, 0)
    spec.num_aggs = 2
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    spec.aggs[1] = y.Agg(y.AGG_SUM_INT64, 2)
    s = GpuScan(spec)
    s.feed_sst(sst_ptr, sst_total, verify=True)
    s.execute()
    res = s.aggregates()
    s.close()
    osc = y.orcl_schema_from(schema)
    ospec = y.OrclScanSpec()
    ospec.read_time = y.orcl_read_time(1_700_000_000_000_000)
    ospec.num_preds = 1
    ospec.preds[0] = y.OrclPred(0, 1, y.PRED_GE, 1000, None, 0)
    ospec.num_aggs = 2
    ospec.aggs[0] = y.OrclAgg(y.AGG_COUNT_STAR, 0)
    ospec.aggs[1] = y.OrclAgg(y.AGG_SUM_INT64, 2)
    data, offsets, nb, total, _ = b.finish()
    ores, _ = y.orcl_scan(data, offsets, nb, osc, ospec)
    assert (res.rows_scanned, res.rows_matched, res.aggs[1].value_i64) == \
        (ores.rows_scanned, ores.rows_matched, ores.aggs[1].value_i64)


@pytest.mark.gpu
def test_sst_device_crc_rejects_corruption():
    """A flipped byte inside a data block must fail the DEVICE crc32c
    verify (k_crc32c) with the checksum error."""
    import gpu_scan
    schema, b = _build(rows=2000)
    sst_ptr, sst_total, _, _ = b.finish_sst(compression=1)
    sst = bytearray(
        C.cast(sst_ptr, C.POINTER(C.c_uint8 * sst_total)).contents)
    offs, szs = y.sst_index(bytes(sst), verify=False)
    sst[offs[0] + 5] ^= 0x40  # corrupt block 0
    buf = (C.c_uint8 * len(sst)).from_buffer(sst)
    spec = y.ScanSpec()
    spec.schema = schema
    spec.kv_format = y.ENC_THREE_SHARED_PARTS
    spec.read_time = y.read_time(1_700_000_000_000_000)
    spec.num_aggs = 1
    spec.aggs[0] = y.Agg(y.AGG_COUNT_STAR, 0)
    s = gpu_scan.GpuScan(spec)
    import pytest as _pytest
    with _pytest.raises(RuntimeError, match="checksum"):
        s.feed_sst(C.cast(buf, C.POINTER(C.c_uint8)), len(sst), verify=True)
    s.close()
