// SST (BlockBasedTable) file framing shared between the generator
// (ybg_builder_finish_sst) and the product feed path (yb_gpu_scan_feed_sst).
//
// Format, from the reference (paths relative to yugabyte/yugabyte-db,
// src/yb/rocksdb/):
//   BlockHandle encoding (varint64 offset + varint64 size)
//       table/format.cc:59-75, table/format.h:55-103
//   Footer (new format): [checksum byte][metaindex handle][index handle]
//       [pad to 41][version fixed32][magic lo fixed32][magic hi fixed32]
//       = 53 bytes — table/format.cc:129-155, format.h:164-176
//   Magic numbers — table/block_based_table_builder.cc:187-198
//   Block trailer: [type 1B][Mask(crc32c(contents+type)) fixed32]
//       table/block_based_table_builder.cc:658-700, table/format.h:41-43
//   crc32c mask — util/crc32c.h:51-61
//   Index block: kKeyDeltaEncodingSharedPrefix entries whose values are
//       BlockHandles — table/format.h:46-51 (restart interval 1 by
//       default; the parser accepts any)
#ifndef YBG_SST_INTERNAL_H
#define YBG_SST_INTERNAL_H

#include <stdint.h>

#include <cstddef>
#include <string>
#include <vector>

namespace ybsst {

constexpr uint64_t kBlockBasedTableMagic = 0x88e241b785f4cff7ull;
constexpr uint64_t kLegacyBlockBasedTableMagic = 0xdb4775248b80fb57ull;
constexpr size_t kBlockTrailerSize = 5;
constexpr size_t kNewFooterLen = 53;  // 1 + 2*20 + 4 + 8
constexpr uint32_t kCrcMaskDelta = 0xa282ead8ul;

// crc32c (Castagnoli, reflected poly 0x82F63B78), slicing-by-8 tables —
// the bytewise version measured ~0.4 GB/s on the host verify path, which
// dominated compressed-SST feed time.
inline const uint32_t (*crc32c_tables())[256] {
  static uint32_t t[8][256];
  static bool init = false;
  if (!init) {
    for (uint32_t i = 0; i < 256; ++i) {
      uint32_t c = i;
      for (int k = 0; k < 8; ++k)
        c = (c & 1) ? (0x82F63B78u ^ (c >> 1)) : (c >> 1);
      t[0][i] = c;
    }
    for (int s = 1; s < 8; ++s)
      for (uint32_t i = 0; i < 256; ++i)
        t[s][i] = t[0][t[s - 1][i] & 0xff] ^ (t[s - 1][i] >> 8);
    init = true;
  }
  return t;
}

inline uint32_t crc32c_extend(uint32_t crc, const uint8_t* p, size_t n) {
  const uint32_t(*t)[256] = crc32c_tables();
  uint32_t c = crc ^ 0xffffffffu;
  while (n >= 8) {
    uint32_t lo, hi;
    __builtin_memcpy(&lo, p, 4);
    __builtin_memcpy(&hi, p + 4, 4);
    lo ^= c;
    c = t[7][lo & 0xff] ^ t[6][(lo >> 8) & 0xff] ^ t[5][(lo >> 16) & 0xff] ^
        t[4][lo >> 24] ^ t[3][hi & 0xff] ^ t[2][(hi >> 8) & 0xff] ^
        t[1][(hi >> 16) & 0xff] ^ t[0][hi >> 24];
    p += 8;
    n -= 8;
  }
  for (size_t i = 0; i < n; ++i) c = t[0][(c ^ p[i]) & 0xff] ^ (c >> 8);
  return c ^ 0xffffffffu;
}

inline uint32_t crc32c_value(const uint8_t* p, size_t n) {
  return crc32c_extend(0, p, n);
}

// util/crc32c.h:58-61
inline uint32_t crc32c_mask(uint32_t crc) {
  return ((crc >> 15) | (crc << 17)) + kCrcMaskDelta;
}

// Parse an SST file: footer -> index block -> data-block handles.
// On success fills (offset, size, trailer type) per data block, in file
// order. verify=1 checks every data block's crc32c trailer; types other
// than kNoCompression(0) / kSnappyCompression(1) are rejected.
int parse_sst(const uint8_t* file, uint64_t size, int verify,
              std::vector<uint64_t>* offsets, std::vector<uint64_t>* sizes,
              std::vector<uint8_t>* types, std::string* err);

}  // namespace ybsst

#endif  // YBG_SST_INTERNAL_H
