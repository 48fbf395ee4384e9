// Snappy raw-block codec, shared by the host compressor (generator), the
// host decompressor (tests) and the gfx950 decompression kernel — the same
// function compiles for both sides like scan_device.h.
//
// The reference compresses data blocks with Snappy when configured
// (rocksdb CompressBlock / kSnappyCompression; the block trailer type byte
// is 0x1 — table/block_based_table_builder.cc:658-700). Snappy itself
// lives in the absent thirdparty tree, so this is a restatement of the
// PUBLIC raw-format specification (google/snappy format_description.txt,
// stable since release 1.0):
//   preamble: uvarint32 uncompressed length
//   elements: tag byte, low 2 bits = type
//     00 literal: len-1 in tag>>2 if < 60; 60..63 => that many extra bytes
//                 hold len-1 little-endian
//     01 copy, 1-byte offset: len = ((tag>>2)&7)+4, off = (tag>>5)<<8|next
//     10 copy, 2-byte offset: len = (tag>>2)+1, off = 2-byte LE
//     11 copy, 4-byte offset: len = (tag>>2)+1, off = 4-byte LE
// Copies may overlap their output (RLE-style), so the copy loop is
// byte-ordered. Parity for this codec is pinned by byte-exact round-trip
// tests (compressor -> both decompressors) — the reference's own snappy
// binaries are not available in this container (DESIGN.md "Oracle").
#ifndef YBG_SNAPPY_DEV_H
#define YBG_SNAPPY_DEV_H

#ifndef YBG_DEV_QUAL
#define YBG_SNAPPY_HOST_ONLY 1
#define SNPDEV static inline
#else
#define SNPDEV YBG_DEV_QUAL
#endif

#include <stdint.h>

namespace ybsnappy {

// Decompress a raw snappy stream into dst (capacity dst_cap). Returns the
// uncompressed length, or -1 on malformed input / overflow.
SNPDEV int64_t snappy_uncompress(const uint8_t* src, uint64_t n,
                                 uint8_t* dst, uint64_t dst_cap) {
  uint64_t ip = 0, op = 0;
  // uvarint32 uncompressed length
  uint64_t ulen = 0;
  int shift = 0;
  for (;;) {
    if (ip >= n || shift > 28) return -1;
    uint8_t b = src[ip++];
    ulen |= (uint64_t)(b & 0x7f) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  if (ulen > dst_cap) return -1;
  while (ip < n) {
    uint8_t tag = src[ip++];
    uint32_t kind = tag & 3;
    if (kind == 0) {  // literal
      uint64_t len = (tag >> 2) + 1;
      if (len > 60) {
        uint32_t nb = (uint32_t)len - 60;  // 1..4 extra length bytes
        if (ip + nb > n) return -1;
        uint64_t l = 0;
        for (uint32_t i = 0; i < nb; ++i) l |= (uint64_t)src[ip + i] << (8 * i);
        ip += nb;
        len = l + 1;
      }
      if (ip + len > n || op + len > dst_cap) return -1;
      for (uint64_t i = 0; i < len; ++i) dst[op + i] = src[ip + i];
      ip += len;
      op += len;
    } else {
      uint64_t len, off;
      if (kind == 1) {
        len = ((tag >> 2) & 7) + 4;
        if (ip >= n) return -1;
        off = ((uint64_t)(tag >> 5) << 8) | src[ip++];
      } else if (kind == 2) {
        len = (tag >> 2) + 1;
        if (ip + 2 > n) return -1;
        off = (uint64_t)src[ip] | ((uint64_t)src[ip + 1] << 8);
        ip += 2;
      } else {
        len = (tag >> 2) + 1;
        if (ip + 4 > n) return -1;
        off = (uint64_t)src[ip] | ((uint64_t)src[ip + 1] << 8) |
              ((uint64_t)src[ip + 2] << 16) | ((uint64_t)src[ip + 3] << 24);
        ip += 4;
      }
      if (off == 0 || off > op || op + len > dst_cap) return -1;
      // overlapping copies are byte-ordered by the format
      for (uint64_t i = 0; i < len; ++i) dst[op + i] = dst[op - off + i];
      op += len;
    }
  }
  return op == ulen ? (int64_t)op : -1;
}

#ifdef YBG_SNAPPY_HOST_ONLY
// Greedy hash-match compressor (host only — the generator's write path).
// Emits literals and 2-byte-offset copies: a strict subset of the format
// every conforming decompressor accepts. Returns compressed length or -1
// if dst_cap is too small.
static inline int64_t snappy_compress(const uint8_t* src, uint64_t n,
                                      uint8_t* dst, uint64_t dst_cap) {
  uint64_t op = 0;
  // preamble
  {
    uint64_t v = n;
    do {
      if (op >= dst_cap) return -1;
      uint8_t b = v & 0x7f;
      v >>= 7;
      dst[op++] = v ? (b | 0x80) : b;
    } while (v);
  }
  auto emit_literal = [&](uint64_t from, uint64_t len) -> bool {
    while (len > 0) {
      uint64_t l = len;
      if (l <= 60) {
        if (op + 1 + l > dst_cap) return false;
        dst[op++] = (uint8_t)((l - 1) << 2);
      } else {
        uint64_t lm = l - 1;
        uint32_t nb = lm < (1u << 8) ? 1 : (lm < (1u << 16) ? 2 : 3);
        if (op + 1 + nb + l > dst_cap) return false;
        dst[op++] = (uint8_t)((59 + nb) << 2);
        for (uint32_t i = 0; i < nb; ++i) dst[op++] = (uint8_t)(lm >> (8 * i));
      }
      for (uint64_t i = 0; i < l; ++i) dst[op++] = src[from + i];
      from += l;
      len -= l;
    }
    return true;
  };
  constexpr uint32_t kHashBits = 12;
  uint32_t table[1 << kHashBits];
  for (uint32_t i = 0; i < (1u << kHashBits); ++i) table[i] = 0xffffffffu;
  uint64_t ip = 0, lit_start = 0;
  while (ip + 4 <= n) {
    uint32_t w;
    __builtin_memcpy(&w, src + ip, 4);
    uint32_t h = (w * 0x1e35a7bdu) >> (32 - kHashBits);
    uint32_t cand = table[h];
    table[h] = (uint32_t)ip;
    if (cand != 0xffffffffu && ip - cand <= 0xffff && ip - cand > 0) {
      uint32_t cw;
      __builtin_memcpy(&cw, src + cand, 4);
      if (cw == w) {
        // extend the match
        uint64_t len = 4;
        while (ip + len < n && src[cand + len] == src[ip + len] && len < 64)
          ++len;
        if (!emit_literal(lit_start, ip - lit_start)) return -1;
        uint64_t off = ip - cand;
        if (op + 3 > dst_cap) return -1;
        dst[op++] = (uint8_t)(((len - 1) << 2) | 2);
        dst[op++] = (uint8_t)off;
        dst[op++] = (uint8_t)(off >> 8);
        ip += len;
        lit_start = ip;
        continue;
      }
    }
    ++ip;
  }
  if (!emit_literal(lit_start, n - lit_start)) return -1;
  return (int64_t)op;
}
#endif  // YBG_SNAPPY_HOST_ONLY

}  // namespace ybsnappy

#undef SNPDEV
#endif  // YBG_SNAPPY_DEV_H
