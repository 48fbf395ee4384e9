// LZ4 block-format codec, shared by the host compressor (generator), the
// host decompressor (tests) and the gfx950 decompression kernel — same
// both-sides pattern as snappy_dev.h.
//
// The reference compresses data blocks with LZ4 when configured (rocksdb
// CompressBlock / kLZ4Compression, trailer type byte 0x4;
// rocksdb/util/compression.h LZ4_Compress/LZ4_Uncompress). With
// compress_format_version 2 the stored bytes are
//   varint32(uncompressed length) ‖ LZ4 BLOCK data
// (compression.h: "output_header_len = PutVarint32(&output, length)").
// LZ4 itself lives in the absent thirdparty tree, so this restates the
// PUBLIC LZ4 block format (lz4.org block format description, stable):
//   sequence: token byte — high nibble literal length, low nibble match
//   length - 4; nibble 15 extends with 255-bytes + terminator byte;
//   literals; 2-byte LE match offset (>= 1); match copies may overlap
//   (byte-ordered copy). The final sequence holds literals only.
// Parity is pinned by byte-exact round-trip tests across the three
// compilations (host compressor feeding host + device decompressors).
#ifndef YBG_LZ4_DEV_H
#define YBG_LZ4_DEV_H

#ifndef YBG_DEV_QUAL
#define YBG_LZ4_HOST_ONLY 1
#define LZ4DEV static inline
#else
#define LZ4DEV YBG_DEV_QUAL
#endif

#include <stdint.h>

namespace yblz4 {

// Decompress an LZ4 block into dst. Returns bytes written or -1.
LZ4DEV int64_t lz4_uncompress(const uint8_t* src, uint64_t n, uint8_t* dst,
                              uint64_t dst_cap) {
  uint64_t ip = 0, op = 0;
  while (ip < n) {
    uint8_t token = src[ip++];
    uint64_t lit = token >> 4;
    if (lit == 15) {
      uint8_t b;
      do {
        if (ip >= n) return -1;
        b = src[ip++];
        lit += b;
      } while (b == 255);
    }
    if (ip + lit > n || op + lit > dst_cap) return -1;
    for (uint64_t i = 0; i < lit; ++i) dst[op + i] = src[ip + i];
    ip += lit;
    op += lit;
    if (ip >= n) break;  // final sequence: literals only
    if (ip + 2 > n) return -1;
    uint64_t off = (uint64_t)src[ip] | ((uint64_t)src[ip + 1] << 8);
    ip += 2;
    if (off == 0 || off > op) return -1;
    uint64_t mlen = (token & 0xf);
    if (mlen == 15) {
      uint8_t b;
      do {
        if (ip >= n) return -1;
        b = src[ip++];
        mlen += b;
      } while (b == 255);
    }
    mlen += 4;
    if (op + mlen > dst_cap) return -1;
    const uint8_t* m = dst + op - off;
    for (uint64_t i = 0; i < mlen; ++i) dst[op + i] = m[i];  // may overlap
    op += mlen;
  }
  return (int64_t)op;
}

#ifdef YBG_LZ4_HOST_ONLY
// Greedy hash-chain-free compressor (any valid LZ4 stream is acceptable;
// byte-identity with the reference's liblz4 output is NOT required — the
// FORMAT is what parity pins). Emits format-conforming end conditions:
// last 5 bytes are literals and the last match starts 12+ bytes before
// the end (lz4 block format restrictions).
static inline int64_t lz4_compress(const uint8_t* src, uint64_t n,
                                   uint8_t* dst, uint64_t cap) {
  const uint64_t kTable = 1 << 14;
  static thread_local uint32_t table[1 << 14];
  for (uint64_t i = 0; i < kTable; ++i) table[i] = 0xffffffffu;
  uint64_t ip = 0, op = 0, anchor = 0;
  auto hash4 = [&](uint64_t p) {
    uint32_t v;
    __builtin_memcpy(&v, src + p, 4);
    return (v * 2654435761u) >> (32 - 14);
  };
  auto emit = [&](uint64_t lit_from, uint64_t lit_n, uint64_t off,
                  uint64_t mlen) -> bool {
    uint64_t need = 1 + lit_n + lit_n / 255 + 3 + mlen / 255 + 1;
    if (op + need > cap) return false;
    uint64_t ml = mlen ? mlen - 4 : 0;
    dst[op++] = (uint8_t)(((lit_n < 15 ? lit_n : 15) << 4) |
                          (mlen ? (ml < 15 ? ml : 15) : 0));
    if (lit_n >= 15) {
      uint64_t r = lit_n - 15;
      while (r >= 255) { dst[op++] = 255; r -= 255; }
      dst[op++] = (uint8_t)r;
    }
    for (uint64_t i = 0; i < lit_n; ++i) dst[op + i] = src[lit_from + i];
    op += lit_n;
    if (!mlen) return true;
    dst[op++] = (uint8_t)(off & 0xff);
    dst[op++] = (uint8_t)(off >> 8);
    if (ml >= 15) {
      uint64_t r = ml - 15;
      while (r >= 255) { dst[op++] = 255; r -= 255; }
      dst[op++] = (uint8_t)r;
    }
    return true;
  };
  if (n >= 13) {
    const uint64_t mflimit = n - 12;  // matches must start before here
    while (ip < mflimit) {
      uint32_t h = hash4(ip);
      uint64_t cand = table[h];
      table[h] = (uint32_t)ip;
      uint32_t a, b;
      __builtin_memcpy(&a, src + ip, 4);
      if (cand != 0xffffffffu && ip - cand <= 65535) {
        __builtin_memcpy(&b, src + cand, 4);
        if (a == b) {
          uint64_t mlen = 4;
          const uint64_t matchlimit = n - 5;  // last 5 bytes literal-only
          while (ip + mlen < matchlimit &&
                 src[cand + mlen] == src[ip + mlen])
            ++mlen;
          if (!emit(anchor, ip - anchor, ip - cand, mlen)) return -1;
          ip += mlen;
          anchor = ip;
          continue;
        }
      }
      ++ip;
    }
  }
  if (!emit(anchor, n - anchor, 0, 0)) return -1;
  return (int64_t)op;
}
#endif  // YBG_LZ4_HOST_ONLY

}  // namespace yblz4
#endif  // YBG_LZ4_DEV_H
