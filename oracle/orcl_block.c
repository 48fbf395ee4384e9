/*
 * oracle/orcl_block.c — RocksDB data-block entry iteration (both key-value
 * encoding formats), restated from the reference.
 * TEST INFRASTRUCTURE ONLY (see orcl.h header comment).
 *
 * Block layout (src/yb/rocksdb/table/block_builder.cc:24-46, Finish :337-346):
 *   entries ‖ uint32_LE restarts[num_restarts] ‖ uint32_LE num_restarts
 * First restart is always offset 0 (block_builder.cc:75, :81).
 */
#include "orcl.h"
#include <string.h>

static uint32_t load_le32(const uint8_t *p) {
  uint32_t v;
  memcpy(&v, p, 4);
  return v;
}

static uint64_t load_le64(const uint8_t *p) {
  uint64_t v;
  memcpy(&v, p, 8);
  return v;
}

static void store_le64(uint8_t *p, uint64_t v) { memcpy(p, &v, 8); }

int orcl_block_iter_init(orcl_block_iter_t *it, const uint8_t *data,
                         size_t size, orcl_kv_format_t fmt) {
  if (size < 8) return -1;
  it->data = data;
  it->size = size;
  it->fmt = fmt;
  it->num_restarts = load_le32(data + size - 4);
  if (it->num_restarts == 0 ||
      (size_t)it->num_restarts * 4 + 4 > size)
    return -1;
  it->restarts_offset = size - 4 - (size_t)it->num_restarts * 4;
  it->next_offset = 0;
  it->key_len = 0;
  it->value = NULL;
  it->value_len = 0;
  return 0;
}

/* DecodeEntry for kKeyDeltaEncodingSharedPrefix:
 * src/yb/rocksdb/table/block.cc:411-436 decode path; layout
 * varint32 shared ‖ varint32 non_shared ‖ varint32 value_len ‖ delta ‖ value
 * (block_builder.cc:389-399). */
static int next_shared_prefix(orcl_block_iter_t *it) {
  const uint8_t *p = it->data + it->next_offset;
  const uint8_t *limit = it->data + it->restarts_offset;
  uint64_t shared, non_shared, value_len;
  size_t sz;
  if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &shared))) return -2;
  p += sz;
  if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &non_shared))) return -2;
  p += sz;
  if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &value_len))) return -2;
  p += sz;
  if ((size_t)(limit - p) < non_shared + value_len) return -2;
  if (shared > it->key_len || shared + non_shared > ORCL_MAX_KEY) return -2;
  memcpy(it->key + shared, p, non_shared);
  it->key_len = (size_t)(shared + non_shared);
  it->value = p + non_shared;
  it->value_len = (size_t)value_len;
  it->next_offset = (size_t)(it->value + value_len - it->data);
  return 1;
}

/* DecodeEntryThreeSharedParts (block_internal.h:54-160) +
 * ParseNextKeyThreeSharedParts (block.cc:287-346) +
 * IterKey::Update (db/dbformat.h:405-476). */
static int next_three_shared(orcl_block_iter_t *it) {
  const uint8_t *p = it->data + it->next_offset;
  const uint8_t *limit = it->data + it->restarts_offset;
  if (limit - p < 2) return -2;

  uint64_t encoded_1;
  size_t sz = orcl_leb128_decode(p, (size_t)(limit - p), &encoded_1);
  if (!sz) return -2;
  p += sz;

  uint32_t value_size = (uint32_t)(encoded_1 >> 2);
  uint64_t last8_increase = (encoded_1 & 2) << 7; /* 0 or 0x100 */
  const int is_frequent_case_1 = (int)(encoded_1 & 1);

  uint32_t shared_prefix_size = 0, non_shared_1_size = 0, non_shared_2_size = 0;
  uint32_t shared_last_component_size = 0;
  int64_t ns1_delta = 0, ns2_delta = 0;
  int is_something_shared;
  uint64_t tmp;

  if (is_frequent_case_1) {
    if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &tmp))) return -2;
    p += sz;
    shared_prefix_size = (uint32_t)tmp;
    shared_last_component_size = 8;
    is_something_shared = 1;
    non_shared_1_size = 1;
    non_shared_2_size = 1;
  } else {
    uint8_t encoded_2 = *p++;
    if ((encoded_2 & 1) == 0) {
      is_something_shared = 0;
      if (encoded_2 == 0) {
        if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &tmp))) return -2;
        p += sz;
        non_shared_1_size = (uint32_t)tmp;
      } else {
        non_shared_1_size = encoded_2 >> 1;
      }
    } else {
      is_something_shared = 1;
      if ((encoded_2 & 2) == 0) {
        shared_last_component_size = 8;
        ns2_delta = (encoded_2 >> 2) & 1;
        non_shared_1_size = (encoded_2 >> 3) & 7;
        non_shared_2_size = (encoded_2 >> 6) & 3;
      } else {
        shared_last_component_size = (encoded_2 & 4) ? 8 : 0;
        if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &tmp))) return -2;
        p += sz;
        non_shared_1_size = (uint32_t)tmp;
        if (encoded_2 & 8) {
          if (!(sz = orcl_svarint_decode(p, (size_t)(limit - p), &ns1_delta)))
            return -2;
          p += sz;
        }
        if (encoded_2 & 16) {
          if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &tmp))) return -2;
          p += sz;
          non_shared_2_size = (uint32_t)tmp;
        }
        if (encoded_2 & 32) {
          if (!(sz = orcl_svarint_decode(p, (size_t)(limit - p), &ns2_delta)))
            return -2;
          p += sz;
        }
      }
      if (!(sz = orcl_leb128_decode(p, (size_t)(limit - p), &tmp))) return -2;
      p += sz;
      shared_prefix_size = (uint32_t)tmp;
    }
  }

  if ((size_t)(limit - p) <
      (size_t)non_shared_1_size + non_shared_2_size + value_size)
    return -2;

  if (!is_something_shared) {
    /* block.cc:311-317: full key stored inline. */
    if (non_shared_1_size > ORCL_MAX_KEY) return -2;
    memcpy(it->key, p, non_shared_1_size);
    it->key_len = non_shared_1_size;
    it->value = p + non_shared_1_size;
    it->value_len = value_size;
    it->next_offset = (size_t)(it->value + value_size - it->data);
    return 1;
  }

  /* block.cc:319-344: reconstruct from previous key. */
  const uint64_t prev_shared_middle_start =
      (uint64_t)shared_prefix_size + non_shared_1_size - (uint64_t)ns1_delta;
  const uint64_t prev_non_shared_2_size =
      (uint64_t)non_shared_2_size - (uint64_t)ns2_delta;
  const uint64_t prev_size_except_middle_shared =
      prev_shared_middle_start + prev_non_shared_2_size +
      shared_last_component_size;
  if (it->key_len < prev_size_except_middle_shared) return -2;
  const uint64_t shared_middle_size =
      it->key_len - prev_size_except_middle_shared;
  if (shared_prefix_size + shared_middle_size + shared_last_component_size == 0)
    return -2;

  /* IterKey::Update (dbformat.h:413-476): new key =
   * prev[0:prefix) ‖ ns1 ‖ prev[pms:pms+mid) ‖ ns2 ‖ (prev_last8 + inc). */
  const uint64_t new_middle_start = shared_prefix_size + non_shared_1_size;
  const uint64_t new_ns2_start = new_middle_start + shared_middle_size;
  const uint64_t new_last8_start = new_ns2_start + non_shared_2_size;
  const uint64_t new_key_size = new_last8_start + shared_last_component_size;
  if (new_key_size > ORCL_MAX_KEY) return -2;

  uint64_t last_component = 0;
  if (shared_last_component_size > 0) {
    last_component = load_le64(it->key + it->key_len - 8) + last8_increase;
  }
  if (new_middle_start != prev_shared_middle_start && shared_middle_size > 0) {
    memmove(it->key + new_middle_start, it->key + prev_shared_middle_start,
            shared_middle_size);
  }
  memcpy(it->key + shared_prefix_size, p, non_shared_1_size);
  memcpy(it->key + new_ns2_start, p + non_shared_1_size, non_shared_2_size);
  if (shared_last_component_size > 0) {
    store_le64(it->key + new_last8_start, last_component);
  }
  it->key_len = (size_t)new_key_size;
  it->value = p + non_shared_1_size + non_shared_2_size;
  it->value_len = value_size;
  it->next_offset = (size_t)(it->value + value_size - it->data);
  return 1;
}

int orcl_block_iter_next(orcl_block_iter_t *it) {
  if (it->next_offset >= it->restarts_offset) return 0;
  if (it->fmt == ORCL_ENC_SHARED_PREFIX) return next_shared_prefix(it);
  return next_three_shared(it);
}
