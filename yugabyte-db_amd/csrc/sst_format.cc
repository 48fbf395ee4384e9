// SST (BlockBasedTable) file parsing: footer -> index block -> data-block
// handles. Host-side product code (the feed path calls parse_sst, then the
// existing block pipeline takes over). Reference citations in
// sst_internal.h.
#include "sst_internal.h"

#include <cstring>

namespace ybsst {

namespace {

inline uint32_t fixed32(const uint8_t* p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;  // little-endian hosts only (x86-64)
}

// LEB128 varint64 (rocksdb util/coding.cc GetVarint64Ptr)
const uint8_t* get_varint64(const uint8_t* p, const uint8_t* limit,
                            uint64_t* v) {
  uint64_t result = 0;
  for (uint32_t shift = 0; shift <= 63 && p < limit; shift += 7) {
    uint64_t byte = *p++;
    if (byte & 0x80) {
      result |= (byte & 0x7f) << shift;
    } else {
      result |= byte << shift;
      *v = result;
      return p;
    }
  }
  return nullptr;
}

}  // namespace

int parse_sst(const uint8_t* file, uint64_t size, int verify,
              std::vector<uint64_t>* offsets, std::vector<uint64_t>* sizes,
              std::vector<uint8_t>* types, std::string* err) {
  offsets->clear();
  sizes->clear();
  types->clear();
  if (size < kNewFooterLen) {
    *err = "file too small for an SST footer";
    return 3;
  }
  uint64_t magic = ((uint64_t)fixed32(file + size - 4) << 32) |
                   fixed32(file + size - 8);
  if (magic == kLegacyBlockBasedTableMagic) {
    *err = "legacy (version-0) SST footer not supported";
    return 3;
  }
  if (magic != kBlockBasedTableMagic) {
    *err = "bad SST magic number";
    return 3;
  }
  uint32_t version = fixed32(file + size - 12);
  if (version > 2) {
    *err = "unsupported block-based table version";
    return 3;
  }
  const uint8_t* f = file + size - kNewFooterLen;
  uint8_t checksum = f[0];
  if (checksum != 1 /* kCRC32c */) {
    *err = "unsupported checksum type (only kCRC32c)";
    return 3;
  }
  uint64_t mi_off, mi_sz, ix_off, ix_sz;
  const uint8_t* p = f + 1;
  const uint8_t* lim = file + size - 12;
  if (!(p = get_varint64(p, lim, &mi_off)) ||
      !(p = get_varint64(p, lim, &mi_sz)) ||
      !(p = get_varint64(p, lim, &ix_off)) ||
      !(p = get_varint64(p, lim, &ix_sz))) {
    *err = "corrupt footer block handles";
    return 3;
  }
  // overflow-safe range check: crafted varint handles can wrap uint64
  if (ix_off > size || ix_sz > size - ix_off ||
      kBlockTrailerSize > size - ix_off - ix_sz) {
    *err = "index handle out of range";
    return 3;
  }
  const uint8_t* ix = file + ix_off;
  if (verify) {
    uint8_t type = ix[ix_sz];
    uint32_t crc = crc32c_extend(crc32c_value(ix, ix_sz), &type, 1);
    if (crc32c_mask(crc) != fixed32(ix + ix_sz + 1)) {
      *err = "index block checksum mismatch";
      return 3;
    }
    if (type != 0) {
      *err = "compressed index block not supported";
      return 3;
    }
  }
  // index block: restart array trailer, then sequential shared-prefix
  // entries (rocksdb/table/block_builder.cc layout; values = BlockHandles)
  if (ix_sz < 8) {
    *err = "index block too small";
    return 3;
  }
  uint32_t nrestarts = fixed32(ix + ix_sz - 4);
  if ((uint64_t)nrestarts * 4 + 4 > ix_sz) {
    *err = "corrupt index restart array";
    return 3;
  }
  const uint8_t* e = ix;
  const uint8_t* elim = ix + ix_sz - 4 - (uint64_t)nrestarts * 4;
  std::vector<uint8_t> key;
  while (e < elim) {
    uint64_t shared, non_shared, vlen;
    if (!(e = get_varint64(e, elim, &shared)) ||
        !(e = get_varint64(e, elim, &non_shared)) ||
        !(e = get_varint64(e, elim, &vlen)) ||
        (uint64_t)(elim - e) < non_shared + vlen || shared > key.size()) {
      *err = "corrupt index entry";
      return 3;
    }
    key.resize(shared);
    key.insert(key.end(), e, e + non_shared);
    e += non_shared;
    const uint8_t* v = e;
    uint64_t b_off, b_sz;
    const uint8_t* q = get_varint64(v, v + vlen, &b_off);
    if (!q || !(q = get_varint64(q, v + vlen, &b_sz))) {
      *err = "corrupt index block handle";
      return 3;
    }
    e += vlen;
    if (b_off > size || b_sz > size - b_off ||
        kBlockTrailerSize > size - b_off - b_sz) {
      *err = "data block handle out of range";
      return 3;
    }
    offsets->push_back(b_off);
    sizes->push_back(b_sz);
    types->push_back(file[b_off + b_sz]);
  }
  for (size_t i = 0; i < offsets->size(); ++i) {
    const uint8_t* b = file + (*offsets)[i];
    uint64_t n = (*sizes)[i];
    uint8_t type = (*types)[i];
    if (type != 0 && type != 1 && type != 4) {
      *err = "unsupported block compression type (none/snappy/lz4 only)";
      return 3;
    }
    if (verify == 1) {  // verify 2 = footer/index only (data-block
                        // trailers are verified on DEVICE by k_crc32c)
      uint32_t crc = crc32c_extend(crc32c_value(b, n), &type, 1);
      if (crc32c_mask(crc) != fixed32(b + n + 1)) {
        *err = "data block checksum mismatch";
        return 3;
      }
    }
  }
  return 0;
}

}  // namespace ybsst
