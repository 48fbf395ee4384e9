// yugabyte-db_amd/csrc/bloom_filter.h — host-side restatement of the
// reference's SST bloom filter (docdb_filter_policy):
//
//  * bit format + probing: rocksdb FixedSizeFilter
//    (src/yb/rocksdb/util/bloom.cc:43-62 AddHash, :384-452
//    FixedSizeFilterBitsBuilder, :174-259 FullFilterBitsReader) — one
//    CACHE_LINE (64 B) block per key, num_probes bits within it, 5-byte
//    trailer [num_probes u8][num_lines fixed32 LE]; num_lines forced ODD
//    (:394-404); defaults filter_block_size_bits = 65536, error = 1%
//    (src/yb/rocksdb/filter_policy.h:180-181);
//  * hash: rocksdb::Hash(seed 0xbc9f1d34) with the SIGNED-char tail
//    (src/yb/rocksdb/util/hash.cc:32-77, util/hash.h:40-42);
//  * key transform: DocDbAwareV3FilterPolicy extracts the encoded-DocKey
//    prefix DocKeyPart::kUpToHashOrFirstRange — through the hashed
//    group's kGroupEnd when a hash is present, else exactly one range
//    component (src/yb/docdb/docdb_filter_policy.cc:22-44,110-117;
//    src/yb/dockv/doc_key.cc:524-590). A key the transform cannot parse
//    maps to the EMPTY prefix, which always matches (:33-40) — pruning
//    must never reject what it cannot prove absent.
//
// The filter is a pure OPTIMIZATION: results are identical with or
// without it, so parity tests assert equal scan results both ways plus
// exact bit-layout goldens against an independent restatement.
#pragma once

#include <cmath>
#include <cstdint>
#include <cstring>
#include <vector>

#include "codec.h"

namespace ybg {

constexpr uint32_t kBloomCacheLine = 64;  // rocksdb port CACHE_LINE_SIZE
constexpr uint32_t kBloomMetaSize = 5;
constexpr uint32_t kBloomDefaultBits = 65536;  // filter_policy.h:180
constexpr double kBloomDefaultError = 0.01;    // filter_policy.h:181

// rocksdb::Hash (util/hash.cc:32-77). The tail bytes are added as
// SIGNED chars — a documented disk-format quirk kept bit-compatible.
inline uint32_t BloomHash(const uint8_t* data, size_t n) {
  const uint32_t m = 0xc6a4a793;
  const uint32_t r = 24;
  const uint8_t* limit = data + n;
  uint32_t h = (uint32_t)(0xbc9f1d34 ^ (n * m));
  while (data + 4 <= limit) {
    uint32_t w;
    memcpy(&w, data, 4);  // little-endian host
    data += 4;
    h += w;
    h *= m;
    h ^= (h >> 16);
  }
  switch (limit - data) {
    case 3:
      h += (uint32_t)((int8_t)data[2] << 16);
      [[fallthrough]];
    case 2:
      h += (uint32_t)((int8_t)data[1] << 8);
      [[fallthrough]];
    case 1:
      h += (uint32_t)((int8_t)data[0]);
      h *= m;
      h ^= (h >> r);
      break;
  }
  return h;
}

struct BloomDims {
  uint32_t num_lines;
  uint32_t total_bits;
  uint32_t num_probes;
  uint32_t max_keys;
  uint32_t slice_size;  // total_bits/8 + kBloomMetaSize
};

// FixedSizeFilterBitsBuilder sizing (bloom.cc:389-421).
inline BloomDims bloom_dims(uint32_t total_bits_req = kBloomDefaultBits,
                            double error_rate = kBloomDefaultError) {
  BloomDims d;
  d.num_lines =
      (total_bits_req + kBloomCacheLine * 8 - 1) / (kBloomCacheLine * 8);
  if (d.num_lines % 2 == 0) {
    if (d.num_lines * kBloomCacheLine < 4096) d.num_lines++;
    else d.num_lines--;
  }
  d.total_bits = d.num_lines * kBloomCacheLine * 8;
  const double mler = -std::log(error_rate);
  uint32_t probes = (uint32_t)(mler / std::log(2.0));
  if (probes < 1) probes = 1;
  if (probes > 255) probes = 255;
  d.num_probes = probes;
  d.max_keys = (uint32_t)(d.total_bits * std::log(2.0) * std::log(2.0) /
                          mler);
  d.slice_size = d.total_bits / 8 + kBloomMetaSize;
  return d;
}

// AddHash (bloom.cc:43-62).
inline void bloom_add_hash(uint32_t h, uint8_t* data, const BloomDims& d) {
  const uint32_t delta = (h >> 17) | (h << 15);
  size_t b = (size_t)(h % d.num_lines) * (kBloomCacheLine * 8);
  for (uint32_t i = 0; i < d.num_probes; ++i) {
    const size_t bitpos = b + (h % (kBloomCacheLine * 8));
    data[bitpos / 8] |= (uint8_t)(1u << (bitpos % 8));
    h += delta;
  }
}

// FullFilterBitsReader::HashMayMatch over ONE slice (bloom.cc:235-259).
inline bool bloom_slice_may_match(const uint8_t* slice, uint32_t slice_len,
                                  const uint8_t* key, size_t key_len) {
  if (slice_len <= kBloomMetaSize) return false;  // empty filter: no keys
  const uint32_t num_probes = slice[slice_len - kBloomMetaSize];
  uint32_t num_lines;
  memcpy(&num_lines, slice + slice_len - 4, 4);
  if (num_lines == 0 || num_probes == 0) return true;  // broken: pass
  if (slice_len != num_lines * kBloomCacheLine + kBloomMetaSize)
    return true;  // broken: pass (reader logs + disables, :183-188)
  uint32_t h = BloomHash(key, key_len);
  const uint32_t delta = (h >> 17) | (h << 15);
  const uint8_t* base = slice + (size_t)(h % num_lines) * kBloomCacheLine;
  for (uint32_t i = 0; i < num_probes; ++i) {
    const uint32_t bitpos = h % (kBloomCacheLine * 8);
    if ((base[bitpos / 8] & (1u << (bitpos % 8))) == 0) return false;
    h += delta;
  }
  return true;
}

// Skip one encoded key-entry value; returns bytes consumed, 0 on error.
// Covers the entry encodings this engine writes (codec.h EncodeDocKey):
// kInt64/kInt32 (kv_util.h:102-158) and zero-escaped strings
// (doc_kv_util.h:101-167).
inline size_t skip_key_entry(const uint8_t* p, size_t len) {
  if (len == 0) return 0;
  switch (p[0]) {
    case kInt64Byte:
      return len >= 9 ? 9 : 0;
    case kInt32Byte:
      return len >= 5 ? 5 : 0;
    case kStringByte: {
      size_t i = 1;
      while (i + 1 < len) {
        if (p[i] == 0) {
          if (p[i + 1] == 0) return i + 2;
          if (p[i + 1] != 1) return 0;
          i += 2;
        } else {
          ++i;
        }
      }
      return 0;
    }
    default:
      return 0;  // unknown entry type: caller treats key as unparseable
  }
}

// DocKeyPart::kUpToHashOrFirstRange prefix length of an encoded DocKey
// (doc_key.cc:524-590 DoDecode + MaxRangeComponentsToDecode): hash
// present -> through the hashed group's kGroupEnd; no hash -> exactly
// one range component. 0 = unparseable (empty prefix, always matches).
inline size_t filter_key_prefix_len(const uint8_t* key, size_t len) {
  if (len == 0) return 0;
  if (key[0] == kUInt16Hash) {
    if (len < 3) return 0;
    size_t p = 3;
    while (p < len && key[p] != kGroupEnd) {
      size_t n = skip_key_entry(key + p, len - p);
      if (!n) return 0;
      p += n;
    }
    if (p >= len) return 0;
    return p + 1;  // include the hashed group's kGroupEnd
  }
  return skip_key_entry(key, len);  // first range component only
}

// Multi-slice builder: a new slice starts when the current one reaches
// max_keys (FixedSizeFilterBlockBuilder rollover). Slices concatenate;
// the query probes every slice (a superset of the reference's
// slice-index dispatch — correct, slightly higher false-positive rate).
class BloomBuilder {
 public:
  explicit BloomBuilder(uint32_t total_bits_req = kBloomDefaultBits,
                        double error_rate = kBloomDefaultError)
      : dims_(bloom_dims(total_bits_req, error_rate)) {}

  // key = full encoded DocKey (user key); the V3 transform is applied
  // here. Consecutive duplicate prefixes are added once (rows arrive in
  // key order, so equal prefixes are adjacent).
  void AddDocKey(const uint8_t* key, size_t len) {
    size_t plen = filter_key_prefix_len(key, len);
    if (plen == 0) return;  // unparseable: always-match, nothing to add
    if (plen == last_.size() && memcmp(last_.data(), key, plen) == 0)
      return;
    last_.assign(key, key + plen);
    if (keys_in_slice_ == 0) OpenSlice();
    bloom_add_hash(BloomHash(key, plen), CurSlice(), dims_);
    if (++keys_in_slice_ >= dims_.max_keys) CloseSlice();
  }

  // Concatenated slice sequence (each dims_.slice_size bytes).
  Buf Finish() {
    if (keys_in_slice_ > 0) CloseSlice();
    return std::move(out_);
  }

  const BloomDims& dims() const { return dims_; }

 private:
  uint8_t* CurSlice() { return out_.data() + slice_off_; }
  void OpenSlice() {
    slice_off_ = out_.size();
    out_.resize(out_.size() + dims_.slice_size, 0);
  }
  void CloseSlice() {  // trailer (bloom.cc:432-437)
    uint8_t* s = CurSlice();
    s[dims_.total_bits / 8] = (uint8_t)dims_.num_probes;
    memcpy(s + dims_.total_bits / 8 + 1, &dims_.num_lines, 4);
    keys_in_slice_ = 0;
  }

  BloomDims dims_;
  Buf out_;
  Buf last_;
  size_t slice_off_ = 0;
  uint32_t keys_in_slice_ = 0;
};

// KeyMayMatch over a slice sequence: empty prefix always matches;
// otherwise match if ANY slice may contain the prefix.
inline bool bloom_may_match(const uint8_t* filt, size_t filt_len,
                            const uint8_t* key, size_t key_len,
                            uint32_t slice_size) {
  size_t plen = filter_key_prefix_len(key, key_len);
  if (plen == 0) return true;
  if (filt_len == 0 || slice_size == 0 || filt_len % slice_size != 0)
    return true;  // absent/broken filter never rejects
  for (size_t off = 0; off + slice_size <= filt_len; off += slice_size)
    if (bloom_slice_may_match(filt + off, slice_size, key, plen))
      return true;
  return false;
}

}  // namespace ybg
