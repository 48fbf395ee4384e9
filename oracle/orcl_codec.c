/*
 * oracle/orcl_codec.c — leaf codecs, restated from the reference.
 * TEST INFRASTRUCTURE ONLY (see orcl.h header comment).
 */
#include "orcl.h"
#include <string.h>

/* ---- yb fast signed varint ---------------------------------------------- */

/* src/yb/util/fast_varint.cc:49-58 (SignedPositiveVarIntLength) */
static int signed_positive_varint_length(uint64_t v) {
  v >>= 6;
  int n = 1;
  while (v != 0) {
    v >>= 7;
    n += 1;
  }
  return n;
}

/* src/yb/util/fast_varint.cc:80-137 (FastEncodeSignedVarInt) */
size_t orcl_svarint_encode(int64_t v, uint8_t *dest) {
  int negative = v < 0;
  uint64_t uv = (uint64_t)v;
  if (negative) uv = 1 + ~uv;
  const int n = signed_positive_varint_length(uv);
  int i;
  if (n == 10) {
    dest[0] = 0xff;
    dest[1] = 0xc0;
    i = 2;
  } else if (n == 9) {
    dest[0] = 0xff;
    dest[1] = (uint8_t)(0x80 | (uv >> 56));
    i = 2;
  } else {
    dest[0] = (uint8_t)(~((1 << (8 - n)) - 1) | (uv >> (8 * (n - 1))));
    i = 1;
  }
  for (; i < n; ++i) dest[i] = (uint8_t)(uv >> (8 * (n - 1 - i)));
  if (negative)
    for (i = 0; i < n; ++i) dest[i] = (uint8_t)~dest[i];
  return (size_t)n;
}

/* src/yb/util/fast_varint.cc:146-159 (kVarIntMasks) */
static const uint64_t kVarIntMasks[] = {
    0,
    0x3fULL,
    0x1fffULL,
    0xfffffULL,
    0x7ffffffULL,
    0x3ffffffffULL,
    0x1ffffffffffULL,
    0xffffffffffffULL,
    0x7fffffffffffffULL,
    0x3fffffffffffffffULL,
    0xffffffffffffffffULL,
};

/* src/yb/util/fast_varint.cc:171-227 (FastDecodeSignedVarInt) */
size_t orcl_svarint_decode(const uint8_t *src, size_t src_size, int64_t *v) {
  if (src_size == 0) return 0;
  uint16_t header = (uint16_t)((src[0] << 8) | (src_size > 1 ? src[1] : 0));
  uint64_t negative = -(uint64_t)((header & 0x8000) == 0);
  header ^= (uint16_t)negative;
  const size_t n_bytes = (size_t)(__builtin_clz((unsigned)((~header & 0x7fff) | 0x20)) - 16);
  if (src_size < n_bytes) return 0;
  uint64_t mask = kVarIntMasks[n_bytes];
  uint64_t temp = 0;
  for (size_t i = 0; i < n_bytes; ++i) temp = (temp << 8) | src[i];
  *v = (int64_t)(((temp & mask) | (~mask & negative)) - negative);
  return n_bytes;
}

/* src/yb/util/fast_varint.cc:161-169 (FastDecodeDescendingSignedVarIntSize) */
size_t orcl_desc_svarint_size(const uint8_t *src, size_t src_size) {
  if (src_size == 0) return 0;
  uint16_t header = (uint16_t)((src[0] << 8) | (src_size > 1 ? src[1] : 0));
  uint64_t negative = -(uint64_t)((header & 0x8000) == 0);
  header ^= (uint16_t)negative;
  return (size_t)(__builtin_clz((unsigned)((~header & 0x7fff) | 0x20)) - 16);
}

/* ---- yb fast unsigned varint -------------------------------------------- */

/* src/yb/util/fast_varint.cc:259-265 (UnsignedVarIntLength) */
static size_t unsigned_varint_length(uint64_t v) {
  size_t result = 1;
  v >>= 7;
  while (v != 0) {
    v >>= 7;
    ++result;
  }
  return result;
}

/* src/yb/util/fast_varint.cc:267-289 (FastEncodeUnsignedVarInt) */
size_t orcl_uvarint_encode(uint64_t v, uint8_t *dest) {
  const size_t n = unsigned_varint_length(v);
  size_t i;
  if (n == 10) {
    dest[0] = 0xff;
    dest[1] = 0x80;
    i = 2;
  } else if (n == 9) {
    dest[0] = 0xff;
    dest[1] = (uint8_t)(v >> 56);
    i = 2;
  } else {
    dest[0] = (uint8_t)(~((1 << (9 - n)) - 1) | (v >> (8 * (n - 1))));
    i = 1;
  }
  for (; i < n; ++i) dest[i] = (uint8_t)(v >> (8 * (n - 1 - i)));
  return n;
}

/* src/yb/util/fast_varint.cc:28-37 (MakeUnsignedVarIntSize):
 * size = clz((i<<1)^0x1ff)-23+1 for first byte i. */
static size_t uvarint_size_from_first_byte(uint8_t b) {
  return (size_t)(__builtin_clz((unsigned)(((unsigned)b << 1) ^ 0x1ff)) - 23 + 1);
}

/* src/yb/util/fast_varint.cc:291-334 (FastDecodeUnsignedVarInt) */
size_t orcl_uvarint_decode(const uint8_t *src, size_t src_size, uint64_t *v) {
  if (src_size == 0) return 0;
  uint8_t first_byte = src[0];
  size_t n_bytes = uvarint_size_from_first_byte(first_byte);
  if (src_size < n_bytes) return 0;
  if (n_bytes == 1) {
    *v = first_byte & 0x7f;
    return 1;
  }
  uint64_t result = 0;
  size_t i = 0;
  if (n_bytes == 9) {
    if (src[1] & 0x80) {
      n_bytes = 10;
      result = src[1] & 0x3f;
      i = 2;
    }
    if (src_size < n_bytes) return 0;
    if (i == 0) { /* n_bytes stayed 9 */
      result = 0;
      i = 1;
      /* reference: result = src[0] & ((1 << (8 - n_bytes)) - 1) is skipped
       * for n_bytes==9; first byte is 0xff and contributes nothing.
       * fast_varint.cc:311-323. */
    }
  } else {
    result = src[0] & (uint8_t)((1 << (8 - n_bytes)) - 1);
    i = 1;
  }
  for (; i < n_bytes; ++i) result = (result << 8) | src[i];
  *v = result;
  return n_bytes;
}

/* ---- LEB128 (rocksdb coding) -------------------------------------------- */

/* src/yb/rocksdb/util/coding.h:224-233 (EncodeVarint64) */
size_t orcl_leb128_encode(uint64_t v, uint8_t *dest) {
  size_t n = 0;
  while (v >= 128) {
    dest[n++] = (uint8_t)((v & 127) | 128);
    v >>= 7;
  }
  dest[n++] = (uint8_t)v;
  return n;
}

/* src/yb/rocksdb/util/coding.cc (GetVarint64Ptr) */
size_t orcl_leb128_decode(const uint8_t *src, size_t src_size, uint64_t *v) {
  uint64_t result = 0;
  for (size_t i = 0, shift = 0; i < src_size && shift <= 63; ++i, shift += 7) {
    uint64_t byte = src[i];
    if (byte & 128) {
      result |= (byte & 127) << shift;
    } else {
      result |= byte << shift;
      *v = result;
      return i + 1;
    }
  }
  return 0;
}

/* ---- packed-row V2 field length ----------------------------------------- */

/* src/yb/util/fast_varint.cc:358-371 (EncodeFieldLength) */
size_t orcl_field_length_encode(uint32_t len, uint8_t *out) {
  if (len < 0x80) {
    *out = (uint8_t)(len << 1);
    return 1;
  }
  uint32_t enc = (len << 1) | 1;
  memcpy(out, &enc, 4); /* little-endian host */
  return 4;
}

/* src/yb/util/fast_varint.cc:373-384 (DecodeFieldLength) */
size_t orcl_field_length_decode(const uint8_t *inp, uint32_t *len) {
  uint8_t b = *inp;
  if ((b & 1) == 0) {
    *len = b >> 1;
    return 1;
  }
  uint32_t v;
  memcpy(&v, inp, 4);
  *len = v >> 1;
  return 4;
}

/* ---- key int codecs ------------------------------------------------------ */

/* src/yb/util/kv_util.h:148-158 (AppendInt64ToKey): BE64(v ^ 1<<63) */
void orcl_key_int64_encode(int64_t v, uint8_t *dest8) {
  uint64_t u = (uint64_t)v ^ 0x8000000000000000ull;
  for (int i = 0; i < 8; ++i) dest8[i] = (uint8_t)(u >> (56 - 8 * i));
}
int64_t orcl_key_int64_decode(const uint8_t *src8) {
  uint64_t u = 0;
  for (int i = 0; i < 8; ++i) u = (u << 8) | src8[i];
  return (int64_t)(u ^ 0x8000000000000000ull);
}
/* src/yb/util/kv_util.h:84-100 (kInt32SignBitFlipMask variant) */
void orcl_key_int32_encode(int32_t v, uint8_t *dest4) {
  uint32_t u = (uint32_t)v ^ 0x80000000u;
  for (int i = 0; i < 4; ++i) dest4[i] = (uint8_t)(u >> (24 - 8 * i));
}
int32_t orcl_key_int32_decode(const uint8_t *src4) {
  uint32_t u = 0;
  for (int i = 0; i < 4; ++i) u = (u << 8) | src4[i];
  return (int32_t)(u ^ 0x80000000u);
}

/* ---- key string codec ---------------------------------------------------- */

/* src/yb/dockv/doc_kv_util.h:101-167: '\0' -> "\0\1", terminate "\0\0". */
size_t orcl_key_string_encode(const uint8_t *s, size_t len, uint8_t *dest) {
  size_t n = 0;
  for (size_t i = 0; i < len; ++i) {
    dest[n++] = s[i];
    if (s[i] == 0) dest[n++] = 1;
  }
  dest[n++] = 0;
  dest[n++] = 0;
  return n;
}

size_t orcl_key_string_decode(const uint8_t *src, size_t src_size,
                              uint8_t *out, size_t *out_len) {
  size_t n = 0, o = 0;
  while (n + 1 < src_size) {
    if (src[n] == 0) {
      if (src[n + 1] == 0) {
        *out_len = o;
        return n + 2;
      }
      if (src[n + 1] == 1) {
        out[o++] = 0;
        n += 2;
        continue;
      }
      return 0; /* corruption */
    }
    out[o++] = src[n++];
  }
  return 0;
}

/* ---- DocHybridTime ------------------------------------------------------- */

/* src/yb/common/doc_hybrid_time.cc:39-76 (EncodedInDocDbFormat):
 * FastEncodeDescendingSignedVarInt(v) == FastEncodeSignedVarInt(-v)
 * (fast_varint.h:63-66). */
size_t orcl_dht_encode(uint64_t ht, uint32_t write_id, uint8_t *dest) {
  uint8_t *out = dest;
  out += orcl_svarint_encode(-(int64_t)0, out); /* generation */
  int64_t micros = (int64_t)(ht >> ORCL_HT_LOGICAL_BITS);
  int64_t logical = (int64_t)(ht & ((1 << ORCL_HT_LOGICAL_BITS) - 1));
  out += orcl_svarint_encode(-(micros - (int64_t)ORCL_YB_EPOCH_MICROS), out);
  out += orcl_svarint_encode(-logical, out);
  out += orcl_svarint_encode(-(((int64_t)write_id + 1) << ORCL_HT_SIZE_BITS), out);
  uint8_t last = out[-1];
  uint8_t encoded_size = (uint8_t)(out - dest);
  out[-1] = (uint8_t)((last & ~ORCL_HT_SIZE_MASK) | encoded_size);
  return encoded_size;
}

/* src/yb/common/doc_hybrid_time.h (GetEncodedSize): low 5 bits of last byte. */
size_t orcl_dht_encoded_size_from_end(const uint8_t *key, size_t key_len) {
  if (key_len == 0) return 0;
  size_t sz = key[key_len - 1] & ORCL_HT_SIZE_MASK;
  if (sz == 0 || sz > key_len) return 0;
  return sz;
}

/* src/yb/common/doc_hybrid_time.cc:86-101 (EncodedFromStart) */
size_t orcl_dht_size_from_start(const uint8_t *src, size_t src_size) {
  size_t off = 0;
  for (int i = 0; i != 4; ++i) {
    size_t sz = orcl_desc_svarint_size(src + off, src_size - off);
    if (sz == 0 || off + sz > src_size) return 0;
    off += sz;
  }
  return off;
}

/* src/yb/common/doc_hybrid_time.cc:104-146 (DecodeFrom). Note: the write_id
 * varint's low 5 bits were overwritten with the size, so we decode it and
 * shift right by 5 after negation (the reference does the same:
 * (decoded >> kNumBitsForHybridTimeSize) - 1). */
size_t orcl_dht_decode(const uint8_t *src, size_t src_size,
                       uint64_t *ht, uint32_t *write_id) {
  size_t off = 0;
  int64_t v;
  size_t sz = orcl_svarint_decode(src + off, src_size - off, &v);
  if (!sz) return 0;
  off += sz; /* generation, ignored */
  sz = orcl_svarint_decode(src + off, src_size - off, &v);
  if (!sz) return 0;
  off += sz;
  int64_t micros = (int64_t)ORCL_YB_EPOCH_MICROS + (-v);
  sz = orcl_svarint_decode(src + off, src_size - off, &v);
  if (!sz) return 0;
  off += sz;
  int64_t logical = -v;
  sz = orcl_svarint_decode(src + off, src_size - off, &v);
  if (!sz) return 0;
  off += sz;
  int64_t shifted = -v;
  if (shifted < 0) return 0;
  *write_id = (uint32_t)((shifted >> ORCL_HT_SIZE_BITS) - 1);
  *ht = ((uint64_t)micros << ORCL_HT_LOGICAL_BITS) | (uint64_t)logical;
  /* size check: low 5 bits of last byte == bytes decoded */
  if ((size_t)(src[off - 1] & ORCL_HT_SIZE_MASK) != off) return 0;
  return off;
}

/* src/yb/docdb/intent_aware_iterator.cc:1446-1455: read limits use
 * write_id = kMaxWriteId = numeric_limits<uint32_t>::max()
 * (common/doc_hybrid_time.h:33). */
#define ORCL_MAX_WRITE_ID 0xffffffffu

void orcl_read_time_init(orcl_read_time_t *rt, uint64_t read_ht,
                         uint64_t local_limit_ht, uint64_t global_limit_ht) {
  rt->read_len = orcl_dht_encode(read_ht, ORCL_MAX_WRITE_ID, rt->read);
  rt->local_limit_len =
      orcl_dht_encode(local_limit_ht, ORCL_MAX_WRITE_ID, rt->local_limit);
  rt->global_limit_len =
      orcl_dht_encode(global_limit_ht, ORCL_MAX_WRITE_ID, rt->global_limit);
}
