// yugabyte-db_amd/csrc/sstgen.cc — synthetic tablet (SST data block) writer:
// the write-path subset needed to build benchmark and parity datasets.
// Produces blocks in the reference's on-disk format:
//   BlockBuilder           src/yb/rocksdb/table/block_builder.cc
//   ThreeSharedParts sizes src/yb/rocksdb/table/block_builder_internal.h:100-239
//   DocKey                 src/yb/dockv/doc_key.h:40-63
//   DocHybridTime          src/yb/common/doc_hybrid_time.cc:39-76
//   Packed rows V1/V2      src/yb/dockv/packed_row.cc:453-461, 523-544
//   Single V1 values       src/yb/dockv/primitive_value.cc:1066-1125
#include "../../include/yb_gpu_scan.h"
#include "codec.h"
#include "sst_internal.h"
#include "snappy_dev.h"
#include "lz4_dev.h"
#include "bloom_filter.h"

#include <algorithm>
#include <atomic>
#include <cstdlib>
#include <memory>
#include <thread>

namespace ybg {
namespace {

// ---------------------------------------------------------------------------
// Block builder (restates rocksdb::BlockBuilder, block_builder.cc:63-412)
// ---------------------------------------------------------------------------

struct ComponentSizes {
  size_t prev_ns1 = 0, ns1 = 0, middle = 0, prev_ns2 = 0, ns2 = 0;
};

// block_builder.cc:118-141 (FindMaxSharedSubstringAtTheSamePos). Note the
// reference only records a run when it is TERMINATED by a mismatch — a run
// reaching the end of the slice is not counted; we replicate that.
static std::pair<size_t, size_t> MaxSharedRun(const uint8_t* a,
                                              const uint8_t* b, size_t n) {
  size_t max_size = 0, max_off = 0, run = 0;
  for (size_t i = 0; i < n; ++i) {
    if (a[i] == b[i]) {
      ++run;
    } else {
      if (run > max_size) {
        max_size = run;
        max_off = i - run;
      }
      run = 0;
    }
  }
  return {max_off, max_size};
}

// block_builder.cc:162-222 (FindMaxSharedMiddle)
static ComponentSizes FindMaxSharedMiddle(const uint8_t* lhs, size_t lhs_size,
                                          const uint8_t* rhs,
                                          size_t rhs_size) {
  size_t min_length;
  std::pair<size_t, size_t> max_shared;
  bool from_left = true;
  if (lhs_size == rhs_size) {
    min_length = rhs_size;
    max_shared = MaxSharedRun(lhs, rhs, min_length);
  } else {
    const uint8_t* lhs_suffix;
    const uint8_t* rhs_suffix;
    if (lhs_size > rhs_size) {
      min_length = rhs_size;
      lhs_suffix = lhs + lhs_size - min_length;
      rhs_suffix = rhs;
    } else {
      min_length = lhs_size;
      lhs_suffix = lhs;
      rhs_suffix = rhs + rhs_size - min_length;
    }
    max_shared = MaxSharedRun(lhs, rhs, min_length);
    auto from_right = MaxSharedRun(lhs_suffix, rhs_suffix, min_length);
    if (from_right.second > max_shared.second) {
      from_left = false;
      max_shared = from_right;
    }
  }
  ComponentSizes r;
  if (max_shared.second == 0) {
    r.prev_ns1 = lhs_size;
    r.ns1 = rhs_size;
    return r;
  }
  if (from_left) {
    size_t head = max_shared.first + max_shared.second;
    r.prev_ns1 = max_shared.first;
    r.ns1 = max_shared.first;
    r.middle = max_shared.second;
    r.prev_ns2 = lhs_size - head;
    r.ns2 = rhs_size - head;
  } else {
    size_t mid_plus_ns2 = min_length - max_shared.first;
    size_t ns2 = mid_plus_ns2 - max_shared.second;
    r.prev_ns1 = lhs_size - mid_plus_ns2;
    r.ns1 = rhs_size - mid_plus_ns2;
    r.middle = max_shared.second;
    r.prev_ns2 = ns2;
    r.ns2 = ns2;
  }
  return r;
}

class BlockBuilder {
 public:
  BlockBuilder(int restart_interval, ybg_kv_format_t fmt)
      : restart_interval_(restart_interval), fmt_(fmt) {
    restarts_.push_back(0);
  }

  void Reset() {
    buf_.clear();
    restarts_.clear();
    restarts_.push_back(0);
    counter_ = 0;
    last_key_.clear();
  }

  size_t SizeEstimate() const {
    return buf_.size() + restarts_.size() * 4 + 4;
  }
  bool Empty() const { return buf_.empty(); }

  // block_builder.cc:348-412 (Add)
  void Add(const uint8_t* key, size_t key_size, const uint8_t* value,
           size_t value_size) {
    size_t shared_prefix = 0;
    bool delta = false;
    if (counter_ >= restart_interval_) {
      restarts_.push_back((uint32_t)buf_.size());
      counter_ = 0;
    } else {
      delta = true;
      size_t min_len = std::min(last_key_.size(), key_size);
      while (shared_prefix < min_len && last_key_[shared_prefix] == key[shared_prefix])
        ++shared_prefix;
    }

    if (fmt_ == YBG_ENC_SHARED_PREFIX) {
      // block_builder.cc:389-399
      size_t non_shared = key_size - shared_prefix;
      Leb128Append(shared_prefix, &buf_);
      Leb128Append(non_shared, &buf_);
      Leb128Append(value_size, &buf_);
      buf_.insert(buf_.end(), key + shared_prefix, key + key_size);
      buf_.insert(buf_.end(), value, value + value_size);
    } else {
      AddThreeSharedParts(key, key_size, value, value_size, shared_prefix,
                          delta);
    }

    last_key_.assign(key, key + key_size);
    ++counter_;
  }

  // block_builder.cc:337-346 (Finish)
  void Finish() {
    for (uint32_t r : restarts_) Fixed32LEAppend(r, &buf_);
    Fixed32LEAppend((uint32_t)restarts_.size(), &buf_);
  }

  const Buf& data() const { return buf_; }
  size_t entries() const { return n_entries_; }
  const Buf& last_key() const { return last_key_; }

 private:
  void AddThreeSharedParts(const uint8_t* key, size_t key_size,
                           const uint8_t* value, size_t value_size,
                           size_t shared_prefix, bool delta) {
    ++n_entries_;
    size_t reuse = 0;
    bool inc = false;
    ComponentSizes cs;
    cs.prev_ns1 = last_key_.size();
    cs.ns1 = key_size;
    if (delta) {
      size_t min_len = std::min(last_key_.size(), key_size);
      // block_builder.cc:224-249 (CalculateLastInternalComponentReuse)
      if (min_len >= shared_prefix + 8) {
        uint64_t prev8, cur8;
        memcpy(&prev8, last_key_.data() + last_key_.size() - 8, 8);
        memcpy(&cur8, key + key_size - 8, 8);
        if (cur8 == prev8 + 0x100) {
          inc = true;
          reuse = 8;
        } else if (cur8 == prev8) {
          reuse = 8;
        }
      }
      cs = FindMaxSharedMiddle(last_key_.data() + shared_prefix,
                               last_key_.size() - shared_prefix - reuse,
                               key + shared_prefix,
                               key_size - shared_prefix - reuse);
    } else {
      shared_prefix = 0;
    }

    // block_builder_internal.h:100-239 (EncodeThreeSharedPartsSizes)
    int64_t ns1_delta = (int64_t)cs.ns1 - (int64_t)cs.prev_ns1;
    int64_t ns2_delta = (int64_t)cs.ns2 - (int64_t)cs.prev_ns2;
    bool frequent = reuse > 0 && cs.ns1 == 1 && cs.ns2 == 1 &&
                    ns1_delta == 0 && ns2_delta == 0;
    uint64_t encoded_1 =
        ((uint64_t)value_size << 2) | ((uint64_t)inc << 1) | (uint64_t)frequent;
    Leb128Append(encoded_1, &buf_);
    if (frequent) {
      Leb128Append(shared_prefix, &buf_);
    } else {
      bool reused = cs.ns1 < key_size;  // == shared_prefix+middle+reuse > 0
      if (reused) {
        if (reuse > 0 && ns1_delta == 0 && (ns2_delta == 0 || ns2_delta == 1) &&
            cs.ns1 < 8 && cs.ns2 < 4) {
          uint8_t encoded_2 = (uint8_t)(0b01 | ((ns2_delta == 1) << 2) |
                                        (cs.ns1 << 3) | (cs.ns2 << 6));
          buf_.push_back(encoded_2);
        } else {
          uint8_t encoded_2 = (uint8_t)(0b11 | ((reuse > 0) << 2) |
                                        ((ns1_delta != 0) << 3) |
                                        ((cs.ns2 != 0) << 4) |
                                        ((ns2_delta != 0) << 5));
          buf_.push_back(encoded_2);
          Leb128Append(cs.ns1, &buf_);
          if (ns1_delta != 0) SVarintAppend(ns1_delta, &buf_);
          if (cs.ns2 != 0) Leb128Append(cs.ns2, &buf_);
          if (ns2_delta != 0) SVarintAppend(ns2_delta, &buf_);
        }
        Leb128Append(shared_prefix, &buf_);
      } else {
        if (key_size > 0 && key_size < 128) {
          buf_.push_back((uint8_t)(key_size << 1));
        } else {
          buf_.push_back(0);
          Leb128Append(key_size, &buf_);
        }
      }
    }
    buf_.insert(buf_.end(), key + shared_prefix, key + shared_prefix + cs.ns1);
    buf_.insert(buf_.end(), key + key_size - reuse - cs.ns2,
                key + key_size - reuse);
    buf_.insert(buf_.end(), value, value + value_size);
  }

  int restart_interval_;
  ybg_kv_format_t fmt_;
  Buf buf_;
  std::vector<uint32_t> restarts_;
  Buf last_key_;
  int counter_ = 0;
  size_t n_entries_ = 0;
};

// ---------------------------------------------------------------------------
// DocKey / value encoding for a row
// ---------------------------------------------------------------------------

static void EncodeDocKey(const ybg_schema_t* sc, const ybg_key_t* k, Buf* out) {
  int col = 0;
  auto append_col = [&](int c) {
    switch ((ybg_keytype_t)sc->key_types[c]) {
      case YBG_KT_INT64:
        out->push_back(kInt64Byte);
        KeyInt64Append((int64_t)k->datums[c], out);
        break;
      case YBG_KT_INT32:
        out->push_back(kInt32Byte);
        KeyInt32Append((int32_t)k->datums[c], out);
        break;
      case YBG_KT_STRING:
        out->push_back(kStringByte);
        KeyStringAppend(k->strs[c], k->str_lens[c], out);
        break;
    }
  };
  if (sc->has_hash) {
    out->push_back(kUInt16Hash);
    out->push_back((uint8_t)(k->hash >> 8));
    out->push_back((uint8_t)k->hash);
    for (int i = 0; i < sc->num_hash_cols; ++i, ++col) append_col(col);
    out->push_back(kGroupEnd);
  }
  for (int i = 0; i < sc->num_range_cols; ++i, ++col) append_col(col);
  out->push_back(kGroupEnd);
}

// Single V1 value (primitive_value.cc:1066-1125). null => tombstone 'X'.
static void EncodeV1Value(ybg_dtype_t dt, uint64_t datum, const uint8_t* str,
                          uint64_t str_len, int null, Buf* out) {
  if (null) {
    out->push_back(kTombstoneByte);
    return;
  }
  switch (dt) {
    case YBG_T_BOOL:
      out->push_back(datum ? kTrueByte : kFalseByte);
      break;
    case YBG_T_INT8:
    case YBG_T_INT16:
    case YBG_T_INT32:
      out->push_back(kInt32Byte);
      BE32Append((uint32_t)(int32_t)(int64_t)datum, out);
      break;
    case YBG_T_INT64:
      out->push_back(kInt64Byte);
      BE64Append(datum, out);
      break;
    case YBG_T_UINT32:
      out->push_back(0x4F);  // 'O' kUInt32
      BE32Append((uint32_t)datum, out);
      break;
    case YBG_T_UINT64:
      out->push_back(0x55);  // 'U' kUInt64
      BE64Append(datum, out);
      break;
    case YBG_T_FLOAT:
      out->push_back(kFloatByte);
      BE32Append((uint32_t)datum, out);
      break;
    case YBG_T_DOUBLE:
      out->push_back(kDoubleByte);
      BE64Append(datum, out);
      break;
    case YBG_T_STRING:
      out->push_back(kStringByte);
      out->insert(out->end(), str, str + str_len);
      break;
  }
}

static size_t V2FixedSize(ybg_dtype_t dt) {
  switch (dt) {
    case YBG_T_BOOL: case YBG_T_INT8: return 1;
    case YBG_T_INT16: return 2;
    case YBG_T_INT32: case YBG_T_UINT32: case YBG_T_FLOAT: return 4;
    case YBG_T_INT64: case YBG_T_UINT64: case YBG_T_DOUBLE: return 8;
    default: return 0;
  }
}

// Packed row value body. V1: packed_row.cc:453-461 (Init) + ColumnPackerV1;
// V2: packed_row.cc:523-544 (Init) + ColumnPackerV2 + Complete :577-597.
static void EncodePackedRow(const ybg_schema_t* sc, int version,
                            const ybg_rowvals_t* vals, Buf* out) {
  if (version == 1) {
    out->push_back(kPackedV1Byte);
    UVarintAppend(0 /*schema version*/, out);
    // varlen end-offset array: nullable or string columns are varlen
    // (schema_packing.cc:45-49).
    size_t hdr_start = out->size();
    int nvarlen = 0;
    for (int i = 0; i < sc->num_value_cols; ++i) {
      if (sc->value_cols[i].nullable ||
          sc->value_cols[i].dtype == YBG_T_STRING)
        ++nvarlen;
    }
    out->resize(hdr_start + (size_t)nvarlen * 4);
    size_t body_start = out->size();
    int vi = 0;
    for (int i = 0; i < sc->num_value_cols; ++i) {
      ybg_dtype_t dt = (ybg_dtype_t)sc->value_cols[i].dtype;
      bool varlen = sc->value_cols[i].nullable || dt == YBG_T_STRING;
      if (!vals->null[i]) {
        // V1 NULL columns append nothing (ColumnPackerV1::DoPackValue).
        EncodeV1Value(dt, vals->datums[i], vals->strs[i], vals->str_lens[i],
                      0, out);
      }
      if (varlen) {
        uint32_t end = (uint32_t)(out->size() - body_start);
        memcpy(out->data() + hdr_start + (size_t)vi * 4, &end, 4);
        ++vi;
      }
    }
  } else {
    out->push_back(kPackedV2Byte);
    UVarintAppend(0, out);
    bool has_null = false;
    for (int i = 0; i < sc->num_value_cols; ++i)
      if (vals->null[i]) has_null = true;
    uint8_t flags = has_null ? kV2HasNullsFlag : 0;
    out->push_back(flags);
    if (has_null) {
      size_t mask_start = out->size();
      out->resize(mask_start + (size_t)((sc->num_value_cols + 7) / 8), 0);
      for (int i = 0; i < sc->num_value_cols; ++i)
        if (vals->null[i])
          (*out)[mask_start + (size_t)i / 8] |= (uint8_t)(1 << (i & 7));
    }
    for (int i = 0; i < sc->num_value_cols; ++i) {
      if (vals->null[i]) continue;
      ybg_dtype_t dt = (ybg_dtype_t)sc->value_cols[i].dtype;
      size_t fs = V2FixedSize(dt);
      if (fs) {
        uint64_t u = vals->datums[i];
        const uint8_t* p = reinterpret_cast<const uint8_t*>(&u);
        out->insert(out->end(), p, p + fs);  // raw little-endian
      } else {
        FieldLengthAppend((uint32_t)vals->str_lens[i], out);
        out->insert(out->end(), vals->strs[i], vals->strs[i] + vals->str_lens[i]);
      }
    }
  }
}

}  // namespace
}  // namespace ybg

// ---------------------------------------------------------------------------
// C ABI: builder
// ---------------------------------------------------------------------------

struct ybg_builder {
  ybg_schema_t schema;
  ybg_kv_format_t fmt;
  size_t block_target;
  int restart_interval;
  ybg::BlockBuilder bb;
  ybg::Buf all_blocks;
  std::vector<uint64_t> offsets;
  std::vector<ybg::Buf> block_last_keys;
  ybg::Buf sst_file;
  uint64_t n_entries = 0;
  ybg::Buf key_scratch, value_scratch;

  ybg_builder(const ybg_schema_t* sc, int f, size_t bt, int ri)
      : schema(*sc), fmt((ybg_kv_format_t)f), block_target(bt),
        restart_interval(ri), bb(ri, (ybg_kv_format_t)f) {
    offsets.push_back(0);
  }

  void FlushBlock() {
    if (bb.Empty()) return;
    block_last_keys.push_back(bb.last_key());
    bb.Finish();
    all_blocks.insert(all_blocks.end(), bb.data().begin(), bb.data().end());
    offsets.push_back(all_blocks.size());
    bb.Reset();
  }

  void AddInternal(const uint8_t* ukey, size_t ukey_len, uint64_t seq,
                   const uint8_t* value, size_t value_len) {
    // BlockBasedTableBuilder flushes when the block reaches the target size
    // BEFORE adding the next key (table_builder Add/ShouldFlush semantics).
    if (!bb.Empty() && bb.SizeEstimate() >= block_target) FlushBlock();
    key_scratch.assign(ukey, ukey + ukey_len);
    ybg::Fixed64LEAppend((seq << 8) | ybg::kTypeValue, &key_scratch);
    bb.Add(key_scratch.data(), key_scratch.size(), value, value_len);
    ++n_entries;
  }
};

extern "C" {

ybg_builder_t* ybg_builder_create(const ybg_schema_t* schema, int kv_format,
                                  size_t block_size_target,
                                  int restart_interval) {
  return new ybg_builder(schema, kv_format, block_size_target,
                         restart_interval);
}

int ybg_builder_add_packed_row(ybg_builder_t* b, const ybg_key_t* key,
                               uint64_t ht, uint32_t write_id, uint64_t seq,
                               int packed_version, const ybg_rowvals_t* vals) {
  ybg::Buf k;
  ybg::EncodeDocKey(&b->schema, key, &k);
  k.push_back(ybg::kHybridTimeByte);
  ybg::DocHtAppend(ht, write_id, &k);
  b->value_scratch.clear();
  ybg::EncodePackedRow(&b->schema, packed_version, vals, &b->value_scratch);
  b->AddInternal(k.data(), k.size(), seq, b->value_scratch.data(),
                 b->value_scratch.size());
  return 0;
}

int ybg_builder_add_column_update(ybg_builder_t* b, const ybg_key_t* key,
                                  int value_col_idx, uint64_t ht,
                                  uint32_t write_id, uint64_t seq,
                                  uint64_t datum, const uint8_t* str,
                                  uint64_t str_len, int null) {
  ybg::Buf k;
  ybg::EncodeDocKey(&b->schema, key, &k);
  k.push_back(ybg::kColByte);
  ybg::SVarintAppend(b->schema.value_cols[value_col_idx].column_id, &k);
  k.push_back(ybg::kHybridTimeByte);
  ybg::DocHtAppend(ht, write_id, &k);
  b->value_scratch.clear();
  ybg::EncodeV1Value((ybg_dtype_t)b->schema.value_cols[value_col_idx].dtype,
                     datum, str, str_len, null, &b->value_scratch);
  b->AddInternal(k.data(), k.size(), seq, b->value_scratch.data(),
                 b->value_scratch.size());
  return 0;
}

int ybg_builder_add_row_tombstone(ybg_builder_t* b, const ybg_key_t* key,
                                  uint64_t ht, uint32_t write_id,
                                  uint64_t seq) {
  ybg::Buf k;
  ybg::EncodeDocKey(&b->schema, key, &k);
  k.push_back(ybg::kHybridTimeByte);
  ybg::DocHtAppend(ht, write_id, &k);
  uint8_t tomb = ybg::kTombstoneByte;
  b->AddInternal(k.data(), k.size(), seq, &tomb, 1);
  return 0;
}

int ybg_builder_add_raw(ybg_builder_t* b, const uint8_t* user_key,
                        size_t user_key_len, uint64_t seq,
                        const uint8_t* value, size_t value_len) {
  b->AddInternal(user_key, user_key_len, seq, value, value_len);
  return 0;
}

int ybg_builder_finish(ybg_builder_t* b, const uint8_t** data,
                       const uint64_t** offsets, uint64_t* n_blocks,
                       uint64_t* total_bytes, uint64_t* n_entries) {
  b->FlushBlock();
  // 48 bytes of tail slack: the scan path's window init can read up to
  // 32 bytes past the position (scan_device.h Rdr contract).
  uint64_t sz = b->all_blocks.size();
  b->all_blocks.resize(sz + 48, 0);
  b->all_blocks.resize(sz);
  b->all_blocks.reserve(sz + 48);
  *data = b->all_blocks.data();
  *offsets = b->offsets.data();
  *n_blocks = b->offsets.size() - 1;
  *total_bytes = b->all_blocks.size();
  *n_entries = b->n_entries;
  return 0;
}

// Assemble a complete BlockBasedTable SST file: data blocks with
// [type|crc32c] trailers, an empty metaindex block, a shared-prefix index
// block (restart interval 1) whose values are BlockHandles, and the
// version-2 footer. Format citations: sst_internal.h.
int ybg_builder_finish_sst2(ybg_builder_t* b, int compression,
                            const uint8_t** data, uint64_t* total_bytes,
                            uint64_t* n_blocks, uint64_t* n_entries) {
  using namespace ybsst;
  b->FlushBlock();
  ybg::Buf& f = b->sst_file;
  f.clear();
  uint64_t nb = b->offsets.size() - 1;
  auto append_block_t = [&](const uint8_t* p, uint64_t n, uint8_t type,
                            uint64_t* h_off, uint64_t* h_sz) {
    *h_off = f.size();
    *h_sz = n;
    f.insert(f.end(), p, p + n);
    uint32_t crc = crc32c_extend(crc32c_value(p, n), &type, 1);
    f.push_back(type);
    ybg::Fixed32LEAppend(crc32c_mask(crc), &f);
  };
  auto append_block = [&](const uint8_t* p, uint64_t n, uint64_t* h_off,
                          uint64_t* h_sz) {
    append_block_t(p, n, 0 /* kNoCompression */, h_off, h_sz);
  };
  std::vector<uint64_t> h_off(nb), h_sz(nb);
  ybg::Buf cbuf;
  for (uint64_t i = 0; i < nb; ++i) {
    const uint8_t* bp = b->all_blocks.data() + b->offsets[i];
    uint64_t bn = b->offsets[i + 1] - b->offsets[i];
    if (compression == 1 /* kSnappyCompression */) {
      cbuf.resize(bn + bn / 2 + 32);
      int64_t cn = ybsnappy::snappy_compress(bp, bn, cbuf.data(),
                                             cbuf.size());
      // like the reference, keep the block uncompressed when compression
      // does not shrink it (CompressBlock GoodCompressionRatio)
      if (cn > 0 && (uint64_t)cn < bn) {
        append_block_t(cbuf.data(), (uint64_t)cn, 1, &h_off[i], &h_sz[i]);
        continue;
      }
    } else if (compression == 4 /* kLZ4Compression */) {
      // rocksdb LZ4 framing (compress_format_version 2):
      // varint32(raw length) then the LZ4 block (util/compression.h)
      ybg::Buf framed;
      ybg::Leb128Append(bn, &framed);
      cbuf.resize(bn + bn / 2 + 64);
      int64_t cn =
          yblz4::lz4_compress(bp, bn, cbuf.data(), cbuf.size());
      if (cn > 0 && framed.size() + (uint64_t)cn < bn) {
        framed.insert(framed.end(), cbuf.data(), cbuf.data() + cn);
        append_block_t(framed.data(), framed.size(), 4, &h_off[i],
                       &h_sz[i]);
        continue;
      }
    }
    append_block(bp, bn, &h_off[i], &h_sz[i]);
  }
  // empty metaindex block: restart array only (block_builder.cc Finish)
  uint64_t mi_off, mi_sz;
  {
    ybg::Buf mi;
    ybg::Fixed32LEAppend(0, &mi);  // restart[0]
    ybg::Fixed32LEAppend(1, &mi);  // num_restarts
    append_block(mi.data(), mi.size(), &mi_off, &mi_sz);
  }
  // index block: key = the block's last internal key, value = BlockHandle
  uint64_t ix_off, ix_sz;
  {
    ybg::BlockBuilder ib(1, YBG_ENC_SHARED_PREFIX);
    ybg::Buf hv;
    for (uint64_t i = 0; i < nb; ++i) {
      hv.clear();
      ybg::Leb128Append(h_off[i], &hv);
      ybg::Leb128Append(h_sz[i], &hv);
      const ybg::Buf& lk = b->block_last_keys[i];
      ib.Add(lk.data(), lk.size(), hv.data(), hv.size());
    }
    ib.Finish();
    append_block(ib.data().data(), ib.data().size(), &ix_off, &ix_sz);
  }
  // version-2 footer (format.cc:129-155): checksum byte, two handles,
  // zero padding to 41 bytes, version, magic
  {
    size_t base = f.size();
    f.push_back(1);  // kCRC32c
    ybg::Leb128Append(mi_off, &f);
    ybg::Leb128Append(mi_sz, &f);
    ybg::Leb128Append(ix_off, &f);
    ybg::Leb128Append(ix_sz, &f);
    f.resize(base + 41, 0);
    ybg::Fixed32LEAppend(2, &f);  // version
    ybg::Fixed32LEAppend((uint32_t)(kBlockBasedTableMagic & 0xffffffffu), &f);
    ybg::Fixed32LEAppend((uint32_t)(kBlockBasedTableMagic >> 32), &f);
  }
  // tail slack for the scan path's windowed loads (Rdr contract)
  uint64_t sz = f.size();
  f.resize(sz + 48, 0);
  f.resize(sz);
  f.reserve(sz + 48);
  *data = f.data();
  *total_bytes = sz;
  *n_blocks = nb;
  *n_entries = b->n_entries;
  return 0;
}

int ybg_builder_finish_sst(ybg_builder_t* b, const uint8_t** data,
                           uint64_t* total_bytes, uint64_t* n_blocks,
                           uint64_t* n_entries) {
  return ybg_builder_finish_sst2(b, 0, data, total_bytes, n_blocks,
                                 n_entries);
}

// Host snappy codec exports (tests pin the codec by round-trip)
int64_t ybg_snappy_compress(const uint8_t* src, uint64_t n, uint8_t* dst,
                            uint64_t cap) {
  return ybsnappy::snappy_compress(src, n, dst, cap);
}
int64_t ybg_lz4_compress(const uint8_t* src, uint64_t n, uint8_t* dst,
                         uint64_t cap) {
  return yblz4::lz4_compress(src, n, dst, cap);
}
int64_t ybg_lz4_uncompress(const uint8_t* src, uint64_t n, uint8_t* dst,
                           uint64_t cap) {
  return yblz4::lz4_uncompress(src, n, dst, cap);
}
int64_t ybg_snappy_uncompress(const uint8_t* src, uint64_t n, uint8_t* dst,
                              uint64_t cap) {
  return ybsnappy::snappy_uncompress(src, n, dst, cap);
}

void ybg_builder_destroy(ybg_builder_t* b) { delete b; }

void ybg_read_time_init(ybg_read_time_t* rt, uint64_t read_ht,
                        uint64_t local_limit_ht, uint64_t global_limit_ht) {
  rt->read_len =
      (int32_t)ybg::DocHtEncode(read_ht, 0xffffffffu, rt->read);
  rt->local_limit_len =
      (int32_t)ybg::DocHtEncode(local_limit_ht, 0xffffffffu, rt->local_limit);
  rt->global_limit_len =
      (int32_t)ybg::DocHtEncode(global_limit_ht, 0xffffffffu, rt->global_limit);
}

// ---------------------------------------------------------------------------
// Multi-threaded benchmark generator
// ---------------------------------------------------------------------------

static inline uint64_t splitmix64(uint64_t& x) {
  x += 0x9E3779B97f4A7C15ull;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  return z ^ (z >> 31);
}

int ybg_generate(const ybg_schema_t* schema, const ybg_gen_params_t* p,
                 uint8_t** data, uint64_t** offsets, uint64_t* n_blocks,
                 uint64_t* total_bytes, uint64_t* n_entries) {
  int nthreads = p->nthreads > 0 ? p->nthreads
                                 : (int)std::thread::hardware_concurrency();
  if (nthreads < 1) nthreads = 1;
  uint64_t rows = p->rows;
  if (rows < (uint64_t)nthreads * 64) nthreads = 1;

  struct Shard {
    std::unique_ptr<ybg_builder> b;
    uint64_t rows = 0;
  };
  std::vector<Shard> shards((size_t)nthreads);
  int versions = p->versions < 1 ? 1 : p->versions;

  auto worker = [&](int t) {
    uint64_t r0 = rows * (uint64_t)t / (uint64_t)nthreads;
    uint64_t r1 = rows * (uint64_t)(t + 1) / (uint64_t)nthreads;
    auto b = std::make_unique<ybg_builder>(schema, p->kv_format,
                                           p->block_size ? p->block_size : 4096,
                                           p->restart_interval
                                               ? p->restart_interval
                                               : 16);
    ybg_key_t key;
    memset(&key, 0, sizeof(key));
    ybg_rowvals_t vals;
    memset(&vals, 0, sizeof(vals));
    uint8_t strbufs[YBG_MAX_COLS][64];
    uint64_t seq = ybg::kInitialSeqno + r0 * (uint64_t)versions;
    for (uint64_t r = r0; r < r1; ++r) {
      // Hash walks monotonically so emission order == DocDB key order.
      key.hash = (uint16_t)((r * 65536ull) / (rows ? rows : 1));
      int kc = 0;
      for (int c = 0; c < schema->num_hash_cols + schema->num_range_cols;
           ++c, ++kc) {
        key.datums[kc] = r;  // row ordinal as the key column value
      }
      uint64_t vstate = p->seed * 0x9E3779B97f4A7C15ull + r * 2654435761ull;
      for (int i = 0; i < schema->num_value_cols; ++i) {
        ybg_dtype_t dt = (ybg_dtype_t)schema->value_cols[i].dtype;
        uint64_t rv = splitmix64(vstate);
        vals.null[i] = 0;
        if (dt == YBG_T_STRING) {
          // 32-byte lowercase alnum
          uint64_t s0 = rv, s1 = splitmix64(vstate);
          for (int j = 0; j < 32; ++j) {
            uint64_t v = (j < 16 ? s0 : s1) >> ((j & 15) * 4);
            strbufs[i][j] = (uint8_t)('a' + (v & 15));
          }
          vals.strs[i] = strbufs[i];
          vals.str_lens[i] = 32;
        } else if (dt == YBG_T_DOUBLE) {
          double d = (double)(rv >> 24) / (double)(1ull << 40);
          memcpy(&vals.datums[i], &d, 8);
        } else if (dt == YBG_T_FLOAT) {
          float f = (float)(rv >> 40) / (float)(1ull << 24);
          uint32_t u;
          memcpy(&u, &f, 4);
          vals.datums[i] = u;
        } else {
          vals.datums[i] = rv & ((1ull << 40) - 1);  // uniform in [0, 2^40)
          if (i == 0 && p->group_mod) vals.datums[i] %= p->group_mod;
        }
      }
      // MVCC versions, newest first (encoded DocHybridTime sorts newest
      // first — doc_hybrid_time.h:89-95). Only the newest version's values
      // are generated above; older versions get distinct values derived from
      // the version index so visibility mistakes change results.
      for (int v = versions - 1; v >= 0; --v) {
        // Distinct, monotonically increasing HT per row: physical micros walk
        // up every 4096 rows, the logical component distinguishes rows within
        // that window (hybrid_time.h:70-99). Versions are ht_step apart.
        uint64_t ht_micros =
            p->ht_base_micros + (uint64_t)v * p->ht_step_micros + (r >> 12);
        uint64_t ht = (ht_micros << 12) | (r & 0xfff);
        // Older versions carry value = newest + age*1000003 so that a
        // visibility mistake changes aggregate results deterministically.
        uint64_t age = (uint64_t)(versions - 1 - v);
        ybg_rowvals_t vv = vals;
        if (age) {
          for (int i = 0; i < schema->num_value_cols; ++i) {
            if ((ybg_dtype_t)schema->value_cols[i].dtype != YBG_T_STRING)
              vv.datums[i] = vals.datums[i] + age * 1000003ull;
          }
        }
        ybg_builder_add_packed_row(b.get(), &key, ht, 0, seq++,
                                   p->packed_version, &vv);
      }
    }
    shards[(size_t)t].b = std::move(b);
    shards[(size_t)t].rows = r1 - r0;
  };

  std::vector<std::thread> threads;
  for (int t = 0; t < nthreads; ++t) threads.emplace_back(worker, t);
  for (auto& th : threads) th.join();

  // stitch shards
  uint64_t total = 0, nb = 0, ne = 0;
  std::vector<std::pair<const uint8_t*, const uint64_t*>> parts;
  std::vector<std::pair<uint64_t, uint64_t>> sizes;  // (bytes, nblocks)
  for (auto& s : shards) {
    const uint8_t* d;
    const uint64_t* off;
    uint64_t nblk, bytes, ent;
    ybg_builder_finish(s.b.get(), &d, &off, &nblk, &bytes, &ent);
    parts.push_back({d, off});
    sizes.push_back({bytes, nblk});
    total += bytes;
    nb += nblk;
    ne += ent;
  }
  // +48 bytes tail slack for the scan path's windowed loads
  uint8_t* out = (uint8_t*)calloc(1, (total ? total : 1) + 48);
  uint64_t* out_off = (uint64_t*)malloc((nb + 1) * sizeof(uint64_t));
  uint64_t pos = 0, bi = 0;
  out_off[0] = 0;
  for (size_t s = 0; s < parts.size(); ++s) {
    memcpy(out + pos, parts[s].first, sizes[s].first);
    for (uint64_t k = 1; k <= sizes[s].second; ++k) {
      out_off[bi + k] = pos + parts[s].second[k];
    }
    bi += sizes[s].second;
    pos += sizes[s].first;
  }
  *data = out;
  *offsets = out_off;
  *n_blocks = nb;
  *total_bytes = total;
  *n_entries = ne;
  return 0;
}

void ybg_free(void* p) { free(p); }

// Encode a DocKey (doc_key.h:40-63) from key-column datums — used by the
// host adapter to serialize the resumable paging position
// (pgsql_operation.cc:2796-2806: the next row's key becomes the paging
// state). Returns encoded length.
size_t ybg_encode_dockey(const ybg_schema_t* schema, const ybg_key_t* key,
                         uint8_t* out, size_t cap) {
  ybg::Buf b;
  ybg::EncodeDocKey(schema, key, &b);
  if (b.size() > cap) return 0;
  memcpy(out, b.data(), b.size());
  return b.size();
}

}  // extern "C"

// ---------------------------------------------------------------------------
// Intents-DB merge (intent_aware_iterator.cc:983-1011 ProcessIntent +
// DecodeStrongWriteIntent; transaction_status_cache.cc). Committed intents
// become regular-format records at their COMMIT DocHybridTime whose value
// carries the intent WRITE time as the kHybridTime control prefix
// (:1249-1267); they are merge-rebuilt into the affected data blocks in
// internal-key order. See include/yb_gpu_scan.h for the boundary notes.
// ---------------------------------------------------------------------------

namespace ybg {
namespace {

// General data-block entry decoder (host): both KV encodings with every
// delta form the BlockBuilder above can emit (rocksdb/table/block.cc:
// 287-454, block_internal.h:54-160). last_key carries the reconstruction
// state across entries.
const uint8_t* HostLeb128(const uint8_t* p, const uint8_t* lim, uint64_t* v) {
  uint64_t r = 0;
  int sh = 0;
  while (p < lim && sh <= 63) {
    uint8_t b = *p++;
    if (b & 0x80) {
      r |= (uint64_t)(b & 0x7f) << sh;
    } else {
      *v = r | ((uint64_t)b << sh);
      return p;
    }
    sh += 7;
  }
  return nullptr;
}

const uint8_t* HostSvarint(const uint8_t* p, const uint8_t* lim, int64_t* v) {
  if (p >= lim) return nullptr;
  uint32_t b0 = p[0];
  uint32_t b1 = p + 1 < lim ? p[1] : 0;
  uint32_t header = (b0 << 8) | b1;
  bool neg = !(header & 0x8000);
  if (neg) header = (uint16_t)~header;
  int n = 0;
  for (uint32_t m = 0x4000; m && (header & m); m >>= 1) ++n;
  ++n;
  if (p + n > lim) return nullptr;
  uint64_t temp = 0;
  for (int i = 0; i < n; ++i)
    temp = (temp << 8) | (uint8_t)(neg ? ~p[i] : p[i]);
  uint64_t mask = n >= 10 ? ~0ull
                          : ((1ull << (7 * n - 1)) - 1);
  temp &= mask;
  *v = neg ? -(int64_t)temp : (int64_t)temp;
  return p + n;
}

const uint8_t* DecodeEntryHost(int fmt, const uint8_t* p, const uint8_t* lim,
                               Buf* last_key, Buf* value) {
  if (fmt == YBG_ENC_SHARED_PREFIX) {
    uint64_t shared, non_shared, vlen;
    if (!(p = HostLeb128(p, lim, &shared))) return nullptr;
    if (!(p = HostLeb128(p, lim, &non_shared))) return nullptr;
    if (!(p = HostLeb128(p, lim, &vlen))) return nullptr;
    if ((uint64_t)(lim - p) < non_shared + vlen ||
        shared > last_key->size())
      return nullptr;
    last_key->resize(shared);
    last_key->insert(last_key->end(), p, p + non_shared);
    p += non_shared;
    value->assign(p, p + vlen);
    return p + vlen;
  }
  uint64_t e1;
  if (!(p = HostLeb128(p, lim, &e1))) return nullptr;
  uint64_t vlen = e1 >> 2;
  bool inc = e1 & 2;
  uint64_t sp = 0, ns1 = 0, ns2 = 0, reuse = 0;
  int64_t d1 = 0, d2 = 0;
  bool shared_something;
  if (e1 & 1) {  // frequent
    if (!(p = HostLeb128(p, lim, &sp))) return nullptr;
    ns1 = 1;
    ns2 = 1;
    reuse = 8;
    shared_something = true;
  } else {
    if (p >= lim) return nullptr;
    uint8_t e2 = *p++;
    if (!(e2 & 1)) {
      shared_something = false;
      if (e2 == 0) {
        if (!(p = HostLeb128(p, lim, &ns1))) return nullptr;
      } else {
        ns1 = e2 >> 1;
      }
    } else {
      shared_something = true;
      if (!(e2 & 2)) {
        reuse = 8;
        d2 = (e2 >> 2) & 1;
        ns1 = (e2 >> 3) & 7;
        ns2 = (e2 >> 6) & 3;
      } else {
        reuse = (e2 & 4) ? 8 : 0;
        if (!(p = HostLeb128(p, lim, &ns1))) return nullptr;
        if (e2 & 8) {
          if (!(p = HostSvarint(p, lim, &d1))) return nullptr;
        }
        if (e2 & 16) {
          if (!(p = HostLeb128(p, lim, &ns2))) return nullptr;
        }
        if (e2 & 32) {
          if (!(p = HostSvarint(p, lim, &d2))) return nullptr;
        }
      }
      if (!(p = HostLeb128(p, lim, &sp))) return nullptr;
    }
  }
  if ((uint64_t)(lim - p) < ns1 + ns2 + vlen) return nullptr;
  if (!shared_something) {
    last_key->assign(p, p + ns1);
    p += ns1;
  } else {
    uint64_t prev_mid_start = sp + ns1 - (uint64_t)d1;
    uint64_t prev_ns2 = ns2 - (uint64_t)d2;
    uint64_t prev_except = prev_mid_start + prev_ns2 + reuse;
    if (last_key->size() < prev_except) return nullptr;
    uint64_t mid = last_key->size() - prev_except;
    Buf nk;
    nk.reserve(sp + ns1 + mid + ns2 + reuse);
    nk.insert(nk.end(), last_key->begin(), last_key->begin() + sp);
    nk.insert(nk.end(), p, p + ns1);
    nk.insert(nk.end(), last_key->begin() + prev_mid_start,
              last_key->begin() + prev_mid_start + mid);
    nk.insert(nk.end(), p + ns1, p + ns1 + ns2);
    if (reuse) {
      uint64_t l8;
      memcpy(&l8, last_key->data() + last_key->size() - 8, 8);
      if (inc) l8 += 0x100;
      size_t at = nk.size();
      nk.resize(at + 8);
      memcpy(nk.data() + at, &l8, 8);
    }
    p += ns1 + ns2;
    *last_key = std::move(nk);
  }
  value->assign(p, p + vlen);
  return p + vlen;
}

int DecodeBlockHost(const uint8_t* blk, size_t size, int fmt,
                    std::vector<std::pair<Buf, Buf>>* out) {
  if (size < 8) return 3;
  uint32_t nr;
  memcpy(&nr, blk + size - 4, 4);
  if (nr == 0 || (uint64_t)nr * 4 + 4 > size) return 3;
  const uint8_t* lim = blk + size - 4 - (uint64_t)nr * 4;
  const uint8_t* p = blk;
  Buf key, val;
  while (p < lim) {
    p = DecodeEntryHost(fmt, p, lim, &key, &val);
    if (!p) return 3;
    out->emplace_back(key, val);
  }
  return 0;
}

// rocksdb internal-key order (dbformat.h:84-110): user key asc, seq DESC
int IKeyCmp(const Buf& a, const Buf& b) {
  size_t ua = a.size() - 8, ub = b.size() - 8;
  size_t n = ua < ub ? ua : ub;
  int c = memcmp(a.data(), b.data(), n);
  if (c) return c;
  if (ua != ub) return ua < ub ? -1 : 1;
  uint64_t sa, sb;
  memcpy(&sa, a.data() + ua, 8);
  memcpy(&sb, b.data() + ub, 8);
  return sa > sb ? -1 : (sa < sb ? 1 : 0);
}

struct IntentRec {
  uint32_t txn_id, write_id;
  uint64_t write_ht;
  Buf key_prefix;  // user key without the '#' + DHT suffix
  Buf value_body;
};

// synthetic seqnos above the generator's 1<<50 range: equal-user-key ties
// order the resolved intent first (newest-wins sees it as newer)
constexpr uint64_t kIntentSeqBase = 1ull << 55;

}  // namespace
}  // namespace ybg

extern "C" {

struct ybg_intents {
  ybg_schema_t schema;
  std::vector<ybg::IntentRec> recs;
  ybg::Buf blob;
};

ybg_intents_t* ybg_intents_create(const ybg_schema_t* schema) {
  auto* it = new ybg_intents();
  it->schema = *schema;
  return it;
}

int ybg_intents_add_packed_row(ybg_intents_t* it, const ybg_key_t* key,
                               uint64_t write_ht, uint32_t write_id,
                               uint32_t txn_id, int packed_version,
                               const ybg_rowvals_t* vals) {
  ybg::IntentRec r;
  r.txn_id = txn_id;
  r.write_id = write_id;
  r.write_ht = write_ht;
  ybg::EncodeDocKey(&it->schema, key, &r.key_prefix);
  ybg::EncodePackedRow(&it->schema, packed_version, vals, &r.value_body);
  it->recs.push_back(std::move(r));
  return 0;
}

int ybg_intents_add_column_update(ybg_intents_t* it, const ybg_key_t* key,
                                  int value_col_idx, uint64_t write_ht,
                                  uint32_t write_id, uint32_t txn_id,
                                  uint64_t datum, const uint8_t* str,
                                  uint64_t str_len, int null) {
  ybg::IntentRec r;
  r.txn_id = txn_id;
  r.write_id = write_id;
  r.write_ht = write_ht;
  ybg::EncodeDocKey(&it->schema, key, &r.key_prefix);
  r.key_prefix.push_back(ybg::kColByte);
  ybg::SVarintAppend(it->schema.value_cols[value_col_idx].column_id,
                     &r.key_prefix);
  ybg::EncodeV1Value((ybg_dtype_t)it->schema.value_cols[value_col_idx].dtype,
                     datum, str, str_len, null, &r.value_body);
  it->recs.push_back(std::move(r));
  return 0;
}

int ybg_intents_add_row_tombstone(ybg_intents_t* it, const ybg_key_t* key,
                                  uint64_t write_ht, uint32_t write_id,
                                  uint32_t txn_id) {
  ybg::IntentRec r;
  r.txn_id = txn_id;
  r.write_id = write_id;
  r.write_ht = write_ht;
  ybg::EncodeDocKey(&it->schema, key, &r.key_prefix);
  r.value_body.push_back(ybg::kTombstoneByte);
  it->recs.push_back(std::move(r));
  return 0;
}

int ybg_intents_data(ybg_intents_t* it, const uint8_t** blob,
                     uint64_t* len) {
  it->blob.clear();
  for (const auto& r : it->recs) {
    ybg::Fixed32LEAppend(r.txn_id, &it->blob);
    ybg::Fixed32LEAppend(r.write_id, &it->blob);
    ybg::Fixed64LEAppend(r.write_ht, &it->blob);
    ybg::Fixed32LEAppend((uint32_t)r.key_prefix.size(), &it->blob);
    ybg::Fixed32LEAppend((uint32_t)r.value_body.size(), &it->blob);
    it->blob.insert(it->blob.end(), r.key_prefix.begin(),
                    r.key_prefix.end());
    it->blob.insert(it->blob.end(), r.value_body.begin(),
                    r.value_body.end());
  }
  *blob = it->blob.data();
  *len = it->blob.size();
  return 0;
}

void ybg_intents_destroy(ybg_intents_t* it) { delete it; }

int ybg_merge_intents(const uint8_t* blocks, const uint64_t* offsets,
                      uint64_t n_blocks, int kv_format,
                      const uint8_t* intents, uint64_t intents_len,
                      const ybg_txn_status_t* txns, uint32_t n_txns,
                      uint8_t** out_blocks, uint64_t** out_offsets,
                      uint64_t* out_n_blocks, uint64_t* out_total) {
  using ybg::Buf;
  // --- parse + resolve (ProcessIntent + the status cache lookup) ---
  std::vector<std::pair<Buf, Buf>> resolved;  // (internal key, value)
  {
    uint64_t o = 0, idx = 0;
    while (o < intents_len) {
      if (o + 24 > intents_len) return 9;
      uint32_t txn_id, write_id, klen, vlen;
      uint64_t wht;
      memcpy(&txn_id, intents + o, 4);
      memcpy(&write_id, intents + o + 4, 4);
      memcpy(&wht, intents + o + 8, 8);
      memcpy(&klen, intents + o + 16, 4);
      memcpy(&vlen, intents + o + 20, 4);
      o += 24;
      if (o + klen + vlen > intents_len) return 9;
      const uint8_t* kp = intents + o;
      const uint8_t* vp = intents + o + klen;
      o += klen + vlen;
      const ybg_txn_status_t* ts = nullptr;
      for (uint32_t t = 0; t < n_txns; ++t)
        if (txns[t].txn_id == txn_id) { ts = &txns[t]; break; }
      if (!ts) return 9;  // unknown transaction: caller bug
      if (ts->status != 1) continue;  // pending/aborted: invisible
      Buf ik(kp, kp + klen);
      ik.push_back(ybg::kHybridTimeByte);
      ybg::DocHtAppend(ts->commit_ht, write_id, &ik);
      ybg::Fixed64LEAppend(((ybg::kIntentSeqBase + idx) << 8) |
                               ybg::kTypeValue,
                           &ik);
      Buf v;
      v.push_back(ybg::kHybridTimeByte);
      ybg::DocHtAppend(wht, write_id, &v);
      v.insert(v.end(), vp, vp + vlen);
      resolved.emplace_back(std::move(ik), std::move(v));
      ++idx;
    }
  }
  std::stable_sort(resolved.begin(), resolved.end(),
                   [](const std::pair<Buf, Buf>& a,
                      const std::pair<Buf, Buf>& b) {
                     return ybg::IKeyCmp(a.first, b.first) < 0;
                   });

  // --- block assignment by first keys ---
  std::vector<Buf> first_keys(n_blocks);
  for (uint64_t b = 0; b < n_blocks; ++b) {
    Buf val;
    const uint8_t* blk = blocks + offsets[b];
    size_t sz = offsets[b + 1] - offsets[b];
    if (sz < 8) return 3;
    uint32_t nr;
    memcpy(&nr, blk + sz - 4, 4);
    if (nr == 0 || (uint64_t)nr * 4 + 4 > sz) return 3;
    const uint8_t* lim = blk + sz - 4 - (uint64_t)nr * 4;
    if (!ybg::DecodeEntryHost(kv_format, blk, lim, &first_keys[b], &val))
      return 3;
  }
  std::vector<std::vector<size_t>> per_block(n_blocks);
  for (size_t i = 0; i < resolved.size(); ++i) {
    // last block whose first key <= intent key (upper_bound - 1)
    uint64_t lo = 0, hi = n_blocks;
    while (lo < hi) {
      uint64_t mid = (lo + hi) / 2;
      if (ybg::IKeyCmp(first_keys[mid], resolved[i].first) <= 0)
        lo = mid + 1;
      else
        hi = mid;
    }
    uint64_t b = lo > 0 ? lo - 1 : 0;
    per_block[b].push_back(i);
  }

  // --- rebuild affected blocks, copy the rest ---
  Buf out;
  std::vector<uint64_t> ooff;
  ooff.push_back(0);
  for (uint64_t b = 0; b < n_blocks; ++b) {
    const uint8_t* blk = blocks + offsets[b];
    size_t sz = offsets[b + 1] - offsets[b];
    if (per_block[b].empty()) {
      out.insert(out.end(), blk, blk + sz);
      ooff.push_back(out.size());
      continue;
    }
    std::vector<std::pair<Buf, Buf>> entries;
    int rc = ybg::DecodeBlockHost(blk, sz, kv_format, &entries);
    if (rc) return rc;
    ybg::BlockBuilder bb(16, (ybg_kv_format_t)kv_format);
    size_t ei = 0;
    size_t ii = 0;
    const auto& ids = per_block[b];
    while (ei < entries.size() || ii < ids.size()) {
      bool take_intent;
      if (ei >= entries.size()) take_intent = true;
      else if (ii >= ids.size()) take_intent = false;
      else
        take_intent =
            ybg::IKeyCmp(resolved[ids[ii]].first, entries[ei].first) < 0;
      const auto& e = take_intent ? resolved[ids[ii]] : entries[ei];
      bb.Add(e.first.data(), e.first.size(), e.second.data(),
             e.second.size());
      if (take_intent) ++ii; else ++ei;
    }
    bb.Finish();
    out.insert(out.end(), bb.data().begin(), bb.data().end());
    ooff.push_back(out.size());
  }
  uint8_t* ob = (uint8_t*)malloc(out.size() ? out.size() : 1);
  memcpy(ob, out.data(), out.size());
  uint64_t* oo = (uint64_t*)malloc(ooff.size() * 8);
  memcpy(oo, ooff.data(), ooff.size() * 8);
  *out_blocks = ob;
  *out_offsets = oo;
  *out_n_blocks = n_blocks;
  *out_total = out.size();
  return 0;
}


/* Test hook: decode a data block's entries with the host decoder the
 * intent merge uses (DecodeBlockHost). Flattened output:
 * keys/values concatenated + per-entry lengths. */
int ybg_decode_block(const uint8_t* blk, uint64_t size, int kv_format,
                     uint8_t* keys_out, uint64_t keys_cap,
                     uint32_t* key_lens, uint8_t* vals_out,
                     uint64_t vals_cap, uint32_t* val_lens,
                     uint64_t cap_entries, uint64_t* n_entries) {
  std::vector<std::pair<ybg::Buf, ybg::Buf>> es;
  int rc = ybg::DecodeBlockHost(blk, size, kv_format, &es);
  if (rc) return rc;
  uint64_t ko = 0, vo = 0;
  *n_entries = es.size();
  for (size_t i = 0; i < es.size() && i < cap_entries; ++i) {
    if (ko + es[i].first.size() > keys_cap ||
        vo + es[i].second.size() > vals_cap)
      return 8;
    memcpy(keys_out + ko, es[i].first.data(), es[i].first.size());
    memcpy(vals_out + vo, es[i].second.data(), es[i].second.size());
    key_lens[i] = (uint32_t)es[i].first.size();
    val_lens[i] = (uint32_t)es[i].second.size();
    ko += es[i].first.size();
    vo += es[i].second.size();
  }
  return 0;
}


/* First internal key of a data block (the restart entry) — feed-time
 * block pruning uses these as the block separators, the role the SST
 * index's keys play in the reference (rocksdb/table/index_reader.cc). */
int ybg_block_first_key(const uint8_t* blk, uint64_t size, int kv_format,
                        uint8_t* out, uint64_t cap, uint64_t* len) {
  if (size < 8) return 3;
  uint32_t nr;
  memcpy(&nr, blk + size - 4, 4);
  if (nr == 0 || (uint64_t)nr * 4 + 4 > size) return 3;
  const uint8_t* lim = blk + size - 4 - (uint64_t)nr * 4;
  ybg::Buf key, val;
  if (!ybg::DecodeEntryHost(kv_format, blk, lim, &key, &val)) return 3;
  if (key.size() > cap) return 8;
  memcpy(out, key.data(), key.size());
  *len = key.size();
  return 0;
}

/* ---- SST bloom filter (docdb_filter_policy / rocksdb FixedSizeFilter,
 * SURVEY §8f-2; see bloom_filter.h for the format citations). The filter
 * is built from a finished tablet's blocks (the flush-time role of
 * FixedSizeFilterBlockBuilder) and queried at feed time. ------------- */

uint64_t ybg_filter_slice_size(void) {
  return ybg::bloom_dims().slice_size;
}

/* Build the filter over every distinct key prefix in the tablet.
 * Returns 0 ok (out_len bytes written, a multiple of
 * ybg_filter_slice_size), 8 if cap is too small, 3 on block corruption. */
int ybg_filter_from_sst(const uint8_t* data, const uint64_t* offsets,
                        uint64_t n_blocks, int kv_format, uint8_t* out,
                        uint64_t cap, uint64_t* out_len) {
  ybg::BloomBuilder fb;
  for (uint64_t b = 0; b < n_blocks; ++b) {
    std::vector<std::pair<ybg::Buf, ybg::Buf>> es;
    int rc = ybg::DecodeBlockHost(data + offsets[b],
                                  offsets[b + 1] - offsets[b], kv_format,
                                  &es);
    if (rc) return rc;
    for (auto& e : es) {
      if (e.first.size() < 8) return 3;
      /* user key = internal key minus the 8-byte seqno/type suffix */
      fb.AddDocKey(e.first.data(), e.first.size() - 8);
    }
  }
  ybg::Buf f = fb.Finish();
  if (f.size() > cap) return 8;
  memcpy(out, f.data(), f.size());
  *out_len = f.size();
  return 0;
}

/* KeyMayMatch (DocDbAwareV3FilterPolicy semantics): key = encoded
 * DocKey (no internal suffix). 1 = may exist, 0 = definitely absent. */
int ybg_filter_may_match(const uint8_t* filt, uint64_t filt_len,
                         const uint8_t* key, uint64_t key_len) {
  return ybg::bloom_may_match(filt, filt_len, key, key_len,
                              ybg::bloom_dims().slice_size)
             ? 1
             : 0;
}

/* Test hook: the kUpToHashOrFirstRange prefix length the transform
 * extracts (0 = unparseable = always-match). */
uint64_t ybg_filter_key_prefix_len(const uint8_t* key, uint64_t key_len) {
  return ybg::filter_key_prefix_len(key, key_len);
}

}  // extern "C"
