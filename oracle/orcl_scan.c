/*
 * oracle/orcl_scan.c — the scan-and-filter hot path, restated from:
 *   DocRowwiseIterator::FetchNextImpl  src/yb/docdb/doc_rowwise_iterator.cc:690-818
 *   SkipFutureRecords<kForward>        src/yb/docdb/intent_aware_iterator.cc:1223-1317
 *   FlatGetHelper (row assembly)       src/yb/docdb/doc_reader.cc:1826-1925
 *   Packed row decode V1/V2            src/yb/dockv/schema_packing.cc:489-497,1043-1121
 *   Value control fields               src/yb/dockv/value.cc:75-130
 *   Aggregates                         src/yb/docdb/doc_expr.cc:248-395
 * TEST INFRASTRUCTURE ONLY (see orcl.h header comment).
 */
#include "orcl.h"
#include <stdlib.h>
#include <string.h>

/* value/key entry type bytes — src/yb/dockv/value_type.h */
#define VT_GROUP_END 0x21    /* '!' kGroupEnd */
#define VT_HYBRID_TIME 0x23  /* '#' kHybridTime */
#define VT_MERGE_FLAGS 0x6B  /* 'k' */
#define VT_TTL 0x74          /* 't' */
#define VT_USER_TS 0x75      /* 'u' */
#define VT_UINT16_HASH 0x47  /* 'G' */
#define VT_INT32 0x48        /* 'H' */
#define VT_INT64 0x49        /* 'I' */
#define VT_SYS_COL 0x4A      /* 'J' kSystemColumnId */
#define VT_COL 0x4B          /* 'K' kColumnId */
#define VT_STRING 0x53       /* 'S' */
#define VT_FLOAT 0x43        /* 'C' */
#define VT_DOUBLE 0x44       /* 'D' */
#define VT_FALSE 0x46        /* 'F' */
#define VT_TRUE 0x54         /* 'T' */
#define VT_NULL_LOW 0x24     /* '$' kNullLow */
#define VT_TOMBSTONE 0x58    /* 'X' */
#define VT_PACKED_V1 0x7A    /* 'z' */
#define VT_PACKED_V2 0x7C    /* '|' */
#define V2_HAS_NULLS_FLAG 1  /* src/yb/dockv/packed_row.h:197 */

/* lexicographic slice compare, rocksdb Slice::compare semantics */
/* escape-aware compare: lhs = zero-escaped key bytes (00 -> 00 01,
 * doc_kv_util.h:101-167), rhs = plain bytes */
static int key_slice_cmp(const uint8_t *e, size_t elen, const uint8_t *b,
                         size_t blen) {
  size_t si = 0, ri = 0;
  while (si < elen && ri < blen) {
    uint8_t cb = e[si];
    si += (cb == 0) ? 2 : 1;
    if (cb != b[ri]) return cb < b[ri] ? -1 : 1;
    ++ri;
  }
  {
    int le = si >= elen, re = ri >= blen;
    return (le && re) ? 0 : (le ? -1 : 1);
  }
}

static int slice_cmp(const uint8_t *a, size_t alen, const uint8_t *b,
                     size_t blen) {
  size_t n = alen < blen ? alen : blen;
  int r = memcmp(a, b, n);
  if (r) return r;
  if (alen < blen) return -1;
  if (alen > blen) return 1;
  return 0;
}

/* ---- DocKey parse --------------------------------------------------------
 * Layout (src/yb/dockv/doc_key.h:40-63):
 *   ['G' hash_be16] hashed_cols '!' range_cols '!'
 * Each key col: type byte + payload (kv_util.h int encodings, doc_kv_util.h
 * zero-escaped strings). Returns dockey length or 0 on corruption. Also fills
 * per-key-column datum/slice info when row != NULL. */
static size_t parse_dockey(const orcl_schema_t *sc, const uint8_t *p,
                           size_t len, orcl_row_t *row) {
  size_t off = 0;
  int col = 0;
  if (sc->has_hash) {
    if (len < 3 || p[0] != VT_UINT16_HASH) return 0;
    off = 3;
  }
  for (int group = 0; group < 2; ++group) {
    int ncols = group == 0 ? sc->num_hash_cols : sc->num_range_cols;
    if (!sc->has_hash && group == 0) ncols = 0;
    for (int i = 0; i < ncols; ++i, ++col) {
      if (off >= len) return 0;
      uint8_t t = p[off];
      switch (sc->key_types[col]) {
        case ORCL_KT_INT64:
          if (t != VT_INT64 || off + 9 > len) return 0;
          if (row) row->key_datums[col] = (uint64_t)orcl_key_int64_decode(p + off + 1);
          off += 9;
          break;
        case ORCL_KT_INT32:
          if (t != VT_INT32 || off + 5 > len) return 0;
          if (row) {
            int64_t v = orcl_key_int32_decode(p + off + 1);
            row->key_datums[col] = (uint64_t)v;
          }
          off += 5;
          break;
        case ORCL_KT_STRING: {
          if (t != VT_STRING) return 0;
          /* find unescaped 00 00 terminator */
          size_t s = off + 1;
          if (row) { row->key_str[col] = p + s; }
          while (s + 1 < len) {
            if (p[s] == 0) {
              if (p[s + 1] == 0) break;
              if (p[s + 1] != 1) return 0;
              s += 2;
            } else {
              ++s;
            }
          }
          if (s + 1 >= len) return 0;
          if (row) row->key_str_len[col] = (uint32_t)(s - (off + 1));
          off = s + 2;
          break;
        }
        default:
          return 0;
      }
    }
    if (group == 0 && !sc->has_hash) {
      /* No hashed group at all: DocKey = range items ‖ '!' only
       * (doc_key.h:352-369 NoHash path — no leading group end). */
      continue;
    }
    if (off >= len || p[off] != VT_GROUP_END) return 0;
    ++off; /* consume '!' */
  }
  return off;
}

/* ---- control fields ------------------------------------------------------
 * src/yb/dockv/value.cc:75-130 (DecodeControlFields): order
 * 'k' merge flags, '#' intent doc ht, 't' ttl, 'u' user timestamp. The '#'
 * intent time is stripped by the visibility check (intent_aware_iterator
 * .cc:1260-1263) before row assembly, but we handle it here too for
 * robustness. Returns bytes consumed, (size_t)-1 on corruption. */
static size_t skip_control_fields(const uint8_t *v, size_t len) {
  size_t off = 0;
  uint64_t tmp;
  int64_t stmp;
  size_t sz;
  if (off < len && v[off] == VT_MERGE_FLAGS) {
    ++off;
    if (!(sz = orcl_uvarint_decode(v + off, len - off, &tmp))) return (size_t)-1;
    off += sz;
  }
  if (off < len && v[off] == VT_HYBRID_TIME) {
    ++off;
    if (!(sz = orcl_dht_size_from_start(v + off, len - off))) return (size_t)-1;
    off += sz;
  }
  if (off < len && v[off] == VT_TTL) {
    ++off;
    if (!(sz = orcl_svarint_decode(v + off, len - off, &stmp))) return (size_t)-1;
    off += sz;
  }
  if (off < len && v[off] == VT_USER_TS) {
    ++off;
    if (off + 8 > len) return (size_t)-1;
    off += 8;
  }
  return off;
}

/* ---- row assembly state -------------------------------------------------- */

typedef struct {
  uint8_t rowkey[ORCL_MAX_KEY];
  size_t rowkey_len;
  int active;

  int base_seen;            /* newest visible bare entry recorded */
  const uint8_t *base_value; /* value slice after intent-ht strip */
  size_t base_value_len;
  uint8_t base_ht[ORCL_MAX_HT_SIZE];
  size_t base_ht_len;

  /* newest visible column update per value column (by packing index) */
  uint32_t col_seen_mask;   /* bit i: update recorded for value col i */
  const uint8_t *col_value[ORCL_MAX_COLS];
  size_t col_value_len[ORCL_MAX_COLS];
  uint8_t col_ht[ORCL_MAX_COLS][ORCL_MAX_HT_SIZE];
  size_t col_ht_len[ORCL_MAX_COLS];
  int liveness_seen;        /* 'J' system column (liveness) update recorded */
  uint8_t liveness_ht[ORCL_MAX_HT_SIZE];
  size_t liveness_ht_len;
  const uint8_t *liveness_value;
  size_t liveness_value_len;
} row_state_t;

/* V1 single-value decode into datum/slice. Returns 1 non-null applied,
 * 0 tombstone/null, -1 corruption.
 * src/yb/dockv/primitive_value.cc:1066-1125 (DoAppendEncodedValue). */
static int decode_v1_value(const uint8_t *v, size_t len, orcl_dtype_t dt,
                           uint64_t *datum, const uint8_t **strp,
                           uint32_t *strl) {
  if (len == 0) return 0; /* empty = null (packed V1 null column) */
  uint8_t t = v[0];
  if (t == VT_TOMBSTONE) return 0;
  if (t == VT_NULL_LOW) return 0;
  switch (dt) {
    case ORCL_T_BOOL:
      if (t == VT_TRUE) { *datum = 1; return 1; }
      if (t == VT_FALSE) { *datum = 0; return 1; }
      return -1;
    case ORCL_T_INT8:
    case ORCL_T_INT16:
    case ORCL_T_INT32: {
      if (t != VT_INT32 || len < 5) return -1;
      uint32_t u = 0;
      for (int i = 0; i < 4; ++i) u = (u << 8) | v[1 + i];
      *datum = (uint64_t)(int64_t)(int32_t)u;
      return 1;
    }
    case ORCL_T_INT64: {
      if (t != VT_INT64 || len < 9) return -1;
      uint64_t u = 0;
      for (int i = 0; i < 8; ++i) u = (u << 8) | v[1 + i];
      *datum = u; /* BE64 of the raw int64 (no sign flip in values) */
      return 1;
    }
    case ORCL_T_FLOAT: {
      if (t != VT_FLOAT || len < 5) return -1;
      uint32_t u = 0;
      for (int i = 0; i < 4; ++i) u = (u << 8) | v[1 + i];
      *datum = u;
      return 1;
    }
    case ORCL_T_DOUBLE: {
      if (t != VT_DOUBLE || len < 9) return -1;
      uint64_t u = 0;
      for (int i = 0; i < 8; ++i) u = (u << 8) | v[1 + i];
      *datum = u;
      return 1;
    }
    case ORCL_T_STRING:
      if (t != VT_STRING) return -1;
      *strp = v + 1;
      *strl = (uint32_t)(len - 1);
      return 1;
    default:
      return -1;
  }
}

/* V2 fixed sizes (raw little-endian) — value_packing_v2.cc:56-86. */
static size_t v2_fixed_size(orcl_dtype_t dt) {
  switch (dt) {
    case ORCL_T_BOOL: case ORCL_T_INT8: return 1;
    case ORCL_T_INT16: return 2;
    case ORCL_T_INT32: case ORCL_T_UINT32: case ORCL_T_FLOAT: return 4;
    case ORCL_T_INT64: case ORCL_T_UINT64: case ORCL_T_DOUBLE: return 8;
    default: return 0; /* varlen */
  }
}

/* V1 fixed encoded sizes — schema_packing.cc:51-65 (EncodedColumnSize). */
static size_t v1_fixed_size(orcl_dtype_t dt) {
  switch (dt) {
    case ORCL_T_BOOL: return 1;
    case ORCL_T_INT8: case ORCL_T_INT16: case ORCL_T_INT32: return 5;
    case ORCL_T_UINT32: case ORCL_T_FLOAT: return 5;
    case ORCL_T_INT64: case ORCL_T_UINT64: case ORCL_T_DOUBLE: return 9;
    default: return 0;
  }
}

typedef struct {
  const orcl_schema_t *sc;
  /* per-column V1 layout (nullable columns are varlen in V1 —
   * schema_packing.cc:45-49 IsVarlenColumn) */
  int v1_varlen[ORCL_MAX_COLS];
  int v1_num_varlen_before[ORCL_MAX_COLS];
  size_t v1_offset_after_prev_varlen[ORCL_MAX_COLS];
  int v1_varlen_count;
  int nullable[ORCL_MAX_COLS];
} packing_t;

static void packing_init(packing_t *pk, const orcl_schema_t *sc,
                         const int *nullable) {
  pk->sc = sc;
  int nvar = 0;
  size_t off_after = 0;
  for (int i = 0; i < sc->num_value_cols; ++i) {
    pk->nullable[i] = nullable ? nullable[i] : 1;
    int varlen = pk->nullable[i] || sc->value_cols[i].dtype == ORCL_T_STRING;
    pk->v1_varlen[i] = varlen;
    pk->v1_num_varlen_before[i] = nvar;
    pk->v1_offset_after_prev_varlen[i] = off_after;
    if (varlen) {
      ++nvar;
      off_after = 0;
    } else {
      off_after += v1_fixed_size(sc->value_cols[i].dtype);
    }
  }
  pk->v1_varlen_count = nvar;
}

/* Decode a packed row (V1 or V2) into row datums.
 * body points at the 'z'/'|' byte. Returns 1 ok, -1 corruption.
 * V1: schema_packing.cc:489-497 (FetchV1), 1043-1062; body = 'z' ‖
 *     uvarint(version) ‖ u32le end-offsets (varlen cols) ‖ bodies.
 * V2: packed_row.cc:523-544 (RowPackerV2::Init), schema_packing.cc:1076-1121
 *     (PackedRowDecoderV2); body = '|' ‖ uvarint(version) ‖ flags ‖
 *     [null mask] ‖ values (fixed raw LE / varlen field-length-prefixed). */
static int decode_packed_row(const packing_t *pk, const uint8_t *body,
                             size_t len, orcl_row_t *row) {
  const orcl_schema_t *sc = pk->sc;
  uint8_t kind = body[0];
  size_t off = 1;
  uint64_t version;
  size_t sz = orcl_uvarint_decode(body + off, len - off, &version);
  if (!sz) return -1;
  off += sz;

  if (kind == VT_PACKED_V1) {
    const uint8_t *header = body + off;
    size_t prefix_len = (size_t)pk->v1_varlen_count * 4;
    if (off + prefix_len > len) return -1;
    const uint8_t *data = header + prefix_len;
    size_t data_len = len - off - prefix_len;
    for (int i = 0; i < sc->num_value_cols; ++i) {
      size_t start = pk->v1_offset_after_prev_varlen[i];
      if (pk->v1_num_varlen_before[i]) {
        uint32_t e;
        memcpy(&e, header + (pk->v1_num_varlen_before[i] - 1) * 4, 4);
        start += e;
      }
      size_t end;
      if (pk->v1_varlen[i]) {
        uint32_t e;
        memcpy(&e, header + pk->v1_num_varlen_before[i] * 4, 4);
        end = e;
      } else {
        end = start + v1_fixed_size(sc->value_cols[i].dtype);
      }
      if (end < start || end > data_len) return -1;
      int r = decode_v1_value(data + start, end - start,
                              sc->value_cols[i].dtype, &row->datums[i],
                              &row->strp[i], &row->strlen_[i]);
      if (r < 0) return -1;
      if (r == 0) row->null_mask |= 1u << i;
    }
    return 1;
  }

  if (kind == VT_PACKED_V2) {
    if (off >= len) return -1;
    uint8_t flags = body[off++];
    const uint8_t *null_mask = NULL;
    if (flags & V2_HAS_NULLS_FLAG) {
      null_mask = body + off;
      off += (size_t)((sc->num_value_cols + 7) / 8);
      if (off > len) return -1;
    }
    const uint8_t *data = body + off;
    const uint8_t *end = body + len;
    for (int i = 0; i < sc->num_value_cols; ++i) {
      if (null_mask && (null_mask[i / 8] & (1 << (i & 7)))) {
        row->null_mask |= 1u << i;
        continue;
      }
      orcl_dtype_t dt = sc->value_cols[i].dtype;
      size_t fs = v2_fixed_size(dt);
      if (fs) {
        if (data + fs > end) return -1;
        uint64_t u = 0;
        memcpy(&u, data, fs); /* raw little-endian */
        if (dt == ORCL_T_INT8) u = (uint64_t)(int64_t)(int8_t)u;
        else if (dt == ORCL_T_INT16) u = (uint64_t)(int64_t)(int16_t)u;
        else if (dt == ORCL_T_INT32) u = (uint64_t)(int64_t)(int32_t)u;
        row->datums[i] = u;
        data += fs;
      } else {
        uint32_t flen;
        if (data >= end) return -1;
        size_t consumed = orcl_field_length_decode(data, &flen);
        data += consumed;
        if (data + flen > end) return -1;
        row->strp[i] = data;
        row->strlen_[i] = flen;
        data += flen;
      }
    }
    return 1;
  }
  return -1;
}

/* Finalize current row: assemble, bounds/predicates/aggregates.
 * Returns 0 ok, 1 stop scan (upper bound), -1 error. */
typedef struct {
  const packing_t *pk;
  const orcl_scan_spec_t *spec;
  orcl_scan_result_t *res;
  orcl_row_cb cb;
  void *cb_arg;
  uint64_t row_seq;
} scan_ctx_t;

static int pred_eval(const orcl_pred_t *pr, const orcl_schema_t *sc,
                     const orcl_row_t *row) {
  long double lhs_f = 0, rhs_f = 0;
  int64_t lhs_i = 0, rhs_i = 0;
  int numeric = 1;
  int cmp;
  if (pr->op == ORCL_PRED_IN_TUPLE) {
    /* multi-column option group (hybrid_scan_choices.h:43-77) */
    uint32_t nc, t, c;
    const uint8_t *cols, *tup;
    size_t tup_sz, ntup;
    if (pr->bytes_len < 4) return 0;
    memcpy(&nc, pr->bytes, 4);
    cols = pr->bytes + 4;
    tup = cols + 4ull * nc;
    tup_sz = 8ull * nc;
    if (!tup_sz || pr->bytes_len < 4 + 4ull * nc) return 0;
    ntup = (pr->bytes_len - 4 - 4ull * nc) / tup_sz;
    for (t = 0; t < ntup; ++t) {
      int all = 1;
      for (c = 0; c < nc && all; ++c) {
        uint32_t ci;
        uint64_t want;
        memcpy(&ci, cols + 4ull * c, 4);
        memcpy(&want, tup + t * tup_sz + 8ull * c, 8);
        if (row->key_datums[ci] != want) all = 0;
      }
      if (all) return 1;
    }
    return 0;
  }
  if (pr->is_key_col) {
    orcl_keytype_t kt = sc->key_types[pr->col];
    if (kt == ORCL_KT_STRING) {
      /* key_str points at the zero-ESCAPED bytes inside the encoded key
       * (doc_kv_util.h:101-167); compare unescaping on the fly */
      cmp = key_slice_cmp(row->key_str[pr->col], row->key_str_len[pr->col],
                          pr->bytes, pr->bytes_len);
      numeric = 0;
    } else {
      lhs_i = (int64_t)row->key_datums[pr->col];
      rhs_i = (int64_t)pr->datum;
    }
  } else {
    if (row->null_mask & (1u << pr->col)) return 0; /* NULL -> filtered out */
    orcl_dtype_t dt = sc->value_cols[pr->col].dtype;
    if (dt == ORCL_T_STRING) {
      cmp = slice_cmp(row->strp[pr->col], row->strlen_[pr->col], pr->bytes,
                      pr->bytes_len);
      numeric = 0;
    } else if (dt == ORCL_T_DOUBLE) {
      double a, b;
      memcpy(&a, &row->datums[pr->col], 8);
      memcpy(&b, &pr->datum, 8);
      lhs_f = a; rhs_f = b;
      numeric = 2;
    } else if (dt == ORCL_T_FLOAT) {
      uint32_t au = (uint32_t)row->datums[pr->col];
      float a, b;
      memcpy(&a, &au, 4);
      uint32_t bu = (uint32_t)pr->datum;
      memcpy(&b, &bu, 4);
      lhs_f = a; rhs_f = b;
      numeric = 2;
    } else {
      lhs_i = (int64_t)row->datums[pr->col];
      rhs_i = (int64_t)pr->datum;
    }
  }
  if (pr->op == ORCL_PRED_IN) {
    /* membership over the option list */
    size_t n = pr->bytes_len / 8, i;
    if (numeric == 0) {
      /* string options: [u32 LE length][bytes] records */
      const uint8_t *lp;
      size_t ll;
      if (pr->is_key_col) {
        lp = row->key_str[pr->col];
        ll = row->key_str_len[pr->col];
      } else {
        lp = row->strp[pr->col];
        ll = row->strlen_[pr->col];
      }
      const uint8_t *q = pr->bytes, *qe = pr->bytes + pr->bytes_len;
      while (q + 4 <= qe) {
        uint32_t ol;
        memcpy(&ol, q, 4);
        q += 4;
        if (q + ol > qe) return 0;
        if (pr->is_key_col) {
          if (key_slice_cmp(lp, ll, q, ol) == 0) return 1;
        } else if (ol == ll && memcmp(q, lp, ol) == 0) {
          return 1;
        }
        q += ol;
      }
      return 0;
    }
    for (i = 0; i < n; ++i) {
      uint64_t raw;
      memcpy(&raw, pr->bytes + 8 * i, 8);
      if (numeric == 1) {
        if (lhs_i == (int64_t)raw) return 1;
      } else {
        double b;
        if (sc->value_cols[pr->col].dtype == ORCL_T_FLOAT) {
          uint32_t bu = (uint32_t)raw;
          float bf;
          memcpy(&bf, &bu, 4);
          b = bf;
        } else {
          memcpy(&b, &raw, 8);
        }
        if (lhs_f == b) return 1;
      }
    }
    return 0;
  }
  if (pr->op == ORCL_PRED_IN_RANGE) {
    /* option ranges (hybrid_scan_choices.h:43-77 OptionRange) */
    size_t n = pr->bytes_len / 24, i;
    if (numeric == 0) return 0; /* numeric columns only (open rejects) */
    for (i = 0; i < n; ++i) {
      uint64_t lo, hi;
      uint32_t fl;
      memcpy(&lo, pr->bytes + 24 * i, 8);
      memcpy(&hi, pr->bytes + 24 * i + 8, 8);
      memcpy(&fl, pr->bytes + 24 * i + 16, 4);
      if (numeric == 1) {
        int okl = (fl & 1) ? lhs_i >= (int64_t)lo : lhs_i > (int64_t)lo;
        int okh = (fl & 2) ? lhs_i <= (int64_t)hi : lhs_i < (int64_t)hi;
        if (okl && okh) return 1;
      } else {
        double l, hgh;
        memcpy(&l, &lo, 8);
        memcpy(&hgh, &hi, 8);
        int okl = (fl & 1) ? lhs_f >= l : lhs_f > l;
        int okh = (fl & 2) ? lhs_f <= hgh : lhs_f < hgh;
        if (okl && okh) return 1;
      }
    }
    return 0;
  }
  if (numeric == 1) cmp = lhs_i < rhs_i ? -1 : (lhs_i > rhs_i ? 1 : 0);
  else if (numeric == 2) cmp = lhs_f < rhs_f ? -1 : (lhs_f > rhs_f ? 1 : 0);
  switch (pr->op) {
    case ORCL_PRED_GT: return cmp > 0;
    case ORCL_PRED_GE: return cmp >= 0;
    case ORCL_PRED_LT: return cmp < 0;
    case ORCL_PRED_LE: return cmp <= 0;
    case ORCL_PRED_EQ: return cmp == 0;
    case ORCL_PRED_NE: return cmp != 0;
    case ORCL_PRED_IN: return 0;       /* handled above */
    case ORCL_PRED_IN_TUPLE: return 0; /* handled above */
    case ORCL_PRED_IN_RANGE: return 0; /* handled above */
  }
  return 0;
}

/* doc_expr.cc:248-395 aggregate semantics. */
static void agg_update(orcl_agg_result_t *a, const orcl_agg_t *spec,
                       const orcl_schema_t *sc, const orcl_row_t *row) {
  (void)sc;
  int col = spec->col;
  int isnull = spec->op == ORCL_AGG_COUNT_STAR
                   ? 0
                   : (row->null_mask >> col) & 1;
  switch (spec->op) {
    case ORCL_AGG_COUNT_STAR:
      a->value_i64 = a->is_null ? 1 : a->value_i64 + 1;
      a->is_null = 0;
      break;
    case ORCL_AGG_COUNT: /* skips NULL column — doc_expr.cc:250-263 */
      if (isnull) break;
      a->value_i64 = a->is_null ? 1 : a->value_i64 + 1;
      a->is_null = 0;
      break;
    case ORCL_AGG_SUM_INT64: /* starts NULL, adopts first — :341-349 */
      if (isnull) break;
      if (a->is_null) {
        a->value_i64 = (int64_t)row->datums[col];
        a->is_null = 0;
      } else {
        a->value_i64 += (int64_t)row->datums[col];
      }
      break;
    case ORCL_AGG_SUM_DOUBLE: {
      if (isnull) break;
      double v;
      memcpy(&v, &row->datums[col], 8);
      if (a->is_null) {
        a->value_f64 = v;
        a->is_null = 0;
      } else {
        a->value_f64 += v;
      }
      break;
    }
    case ORCL_AGG_MIN_INT64:
    case ORCL_AGG_MAX_INT64: {
      if (isnull) break;
      int64_t v = (int64_t)row->datums[col];
      if (a->is_null) { a->value_i64 = v; a->is_null = 0; }
      else if (spec->op == ORCL_AGG_MIN_INT64 ? v < a->value_i64
                                              : v > a->value_i64)
        a->value_i64 = v;
      break;
    }
    case ORCL_AGG_MIN_DOUBLE:
    case ORCL_AGG_MAX_DOUBLE: {
      if (isnull) break;
      double v;
      memcpy(&v, &row->datums[col], 8);
      if (a->is_null) { a->value_f64 = v; a->is_null = 0; }
      else if (spec->op == ORCL_AGG_MIN_DOUBLE ? v < a->value_f64
                                               : v > a->value_f64)
        a->value_f64 = v;
      break;
    }
  }
}

static int finalize_row(scan_ctx_t *cx, row_state_t *st) {
  if (!st->active) return 0;
  st->active = 0;
  const orcl_schema_t *sc = cx->pk->sc;
  const orcl_scan_spec_t *spec = cx->spec;

  /* bounds on encoded rowkey (doc_rowwise_iterator.cc:766-771) */
  if (spec->lower_bound &&
      slice_cmp(st->rowkey, st->rowkey_len, spec->lower_bound,
                spec->lower_bound_len) < 0)
    return 0;
  if (spec->upper_bound &&
      slice_cmp(st->rowkey, st->rowkey_len, spec->upper_bound,
                spec->upper_bound_len) >= 0)
    return 1; /* keys are ordered: stop */

  orcl_row_t row;
  memset(&row, 0, sizeof(row));
  row.null_mask = (uint32_t)((1ull << sc->num_value_cols) - 1); /* start all null */
  if (!parse_dockey(sc, st->rowkey, st->rowkey_len, &row)) return -1;

  int found = 0;
  const uint8_t *base_ht = NULL;
  size_t base_ht_len = 0;

  if (st->base_seen) {
    size_t cf = skip_control_fields(st->base_value, st->base_value_len);
    if (cf == (size_t)-1) return -1;
    const uint8_t *body = st->base_value + cf;
    size_t body_len = st->base_value_len - cf;
    base_ht = st->base_ht;
    base_ht_len = st->base_ht_len;
    if (body_len > 0 &&
        (body[0] == VT_PACKED_V1 || body[0] == VT_PACKED_V2)) {
      row.null_mask = 0;
      /* decode_packed_row sets null bits per column */
      uint32_t nm = 0;
      orcl_row_t tmp = row;
      tmp.null_mask = 0;
      if (decode_packed_row(cx->pk, body, body_len, &tmp) < 0) return -1;
      nm = tmp.null_mask;
      row = tmp;
      row.null_mask = nm;
      found = 1; /* doc_reader.cc:1894-1900 InitRowValue found_ = true */
    } else {
      /* tombstone or non-packed base: null row, not found via base
       * (doc_reader.cc:1893-1897 SetNullOrMissingResult) */
    }
  }

  /* column updates override when newer than base
   * (doc_reader.cc:1847-1869 ProcessEntry: skip if row_write_time >= wt;
   * encoded compare is reversed: newer ⇔ memcmp-smaller). */
  if (st->liveness_seen) {
    if (!base_ht ||
        slice_cmp(st->liveness_ht, st->liveness_ht_len, base_ht, base_ht_len) <
            0) {
      /* liveness column sets found only (kLivenessColumnIndex path) */
      size_t cf = skip_control_fields(st->liveness_value, st->liveness_value_len);
      if (cf == (size_t)-1) return -1;
      if (st->liveness_value_len - cf > 0 &&
          st->liveness_value[cf] != VT_TOMBSTONE)
        found = 1;
    }
  }
  for (int i = 0; i < sc->num_value_cols; ++i) {
    if (!(st->col_seen_mask & (1u << i))) continue;
    if (base_ht &&
        slice_cmp(st->col_ht[i], st->col_ht_len[i], base_ht, base_ht_len) >= 0)
      continue; /* older than (or same as) base */
    size_t cf = skip_control_fields(st->col_value[i], st->col_value_len[i]);
    if (cf == (size_t)-1) return -1;
    const uint8_t *body = st->col_value[i] + cf;
    size_t body_len = st->col_value_len[i] - cf;
    int r = decode_v1_value(body, body_len, sc->value_cols[i].dtype,
                            &row.datums[i], &row.strp[i], &row.strlen_[i]);
    if (r < 0) return -1;
    if (r == 0) {
      row.null_mask |= 1u << i; /* column tombstone -> NULL */
    } else {
      row.null_mask &= ~(1u << i);
      found = 1; /* doc_reader.cc:1863-1866 */
    }
  }

  if (!found) return 0;
  cx->res->rows_scanned++;

  for (int p = 0; p < spec->num_preds; ++p) {
    if (!pred_eval(&spec->preds[p], sc, &row)) return 0;
  }
  cx->res->rows_matched++;
  for (int g = 0; g < spec->num_aggs; ++g) {
    agg_update(&cx->res->aggs[g], &spec->aggs[g], sc, &row);
  }
  if (cx->cb) {
    /* emitted rows expose DECODED key strings: unescape the zero-encoded
     * rowkey bytes (doc_kv_util.h:101-167) into per-column scratch. The
     * predicate/bound paths above keep the escaped in-key form. */
    static uint8_t key_unesc[ORCL_MAX_KEYCOLS][256];
    int nk = sc->num_hash_cols + sc->num_range_cols;
    for (int kc = 0; kc < nk; ++kc) {
      if (sc->key_types[kc] != ORCL_KT_STRING || !row.key_str[kc]) continue;
      const uint8_t *e = row.key_str[kc];
      size_t elen = row.key_str_len[kc], o = 0;
      for (size_t si = 0; si < elen && o < sizeof(key_unesc[0]); ) {
        uint8_t cb = e[si];
        key_unesc[kc][o++] = cb;
        si += (cb == 0) ? 2 : 1;
      }
      row.key_str[kc] = key_unesc[kc];
      row.key_str_len[kc] = (uint32_t)o;
    }
    row.seq_in_scan = cx->row_seq++;
    if (cx->cb(&row, cx->cb_arg)) return 1;
  }
  return 0;
}

/* One merged-stream entry through the full visibility / row-assembly /
 * restart-tracking pipeline (the body of the orcl_scan loop, shared with
 * the intents-merging driver below). Returns 0 = continue, 1 = stop early
 * (row callback), -1 = error. */
static int scan_feed_entry(const orcl_schema_t *schema,
                           const orcl_scan_spec_t *spec,
                           orcl_scan_result_t *res, scan_ctx_t *cx,
                           row_state_t *st, const uint8_t *reg_limit,
                           size_t reg_limit_len, const uint8_t *key,
                           size_t key_len, const uint8_t *value,
                           size_t value_len) {
  const orcl_read_time_t *rt = &spec->read_time;
  res->entries_seen++;
  /* internal key = user_key ‖ fixed64(seq<<8|type) — dbformat.h:84-110 */
  if (key_len < 9) return -1;
  const uint8_t *ukey = key;
  size_t ukey_len = key_len - 8;
  size_t ht_size = orcl_dht_encoded_size_from_end(ukey, ukey_len);
  if (!ht_size || ukey_len < ht_size + 1) return -1;
  const uint8_t *ht_enc = ukey + ukey_len - ht_size;
  if (ukey[ukey_len - ht_size - 1] != VT_HYBRID_TIME) return -1;
  size_t prefix_len = ukey_len - ht_size - 1; /* DocKey ‖ [subkey] */

  /* --- visibility (SkipFutureRecords, intent_aware_iterator.cc:1223-1317) */

  int visible;
  if (value_len > 0 && value[0] == VT_HYBRID_TIME) {
    /* committed-txn record with intent time (:1249-1267) */
    const uint8_t *v1 = value + 1;
    size_t v1_len = value_len - 1;
    const uint8_t *max_allowed;
    size_t max_allowed_len;
    if (slice_cmp(v1, v1_len, rt->local_limit, rt->local_limit_len) > 0) {
      max_allowed = rt->global_limit;
      max_allowed_len = rt->global_limit_len;
    } else {
      max_allowed = rt->read;
      max_allowed_len = rt->read_len;
    }
    visible = slice_cmp(ht_enc, ht_size, max_allowed, max_allowed_len) >= 0;
    if (visible) {
      size_t iht = orcl_dht_size_from_start(v1, v1_len);
      if (!iht) return -1;
      value = v1 + iht;
      value_len = v1_len - iht;
    }
  } else {
    visible = slice_cmp(ht_enc, ht_size, reg_limit, reg_limit_len) >= 0;
  }
  if (!visible) return 0;

  /* --- read-restart tracking (UpdateMaxSeenHt,
   * intent_aware_iterator.cc:815-827; GetReadRestartData :1400-1410):
   * a visible record with commit > read (encoded bytes BELOW
   * encoded(read)) is a restart candidate; keep the MIN encoded form
   * (= max commit time). */
  if (slice_cmp(ht_enc, ht_size, rt->read, rt->read_len) < 0) {
    if (res->restart_ht_len == 0 ||
        slice_cmp(ht_enc, ht_size, res->restart_ht,
                  res->restart_ht_len) < 0) {
      size_t n = ht_size < ORCL_MAX_HT_SIZE ? ht_size : ORCL_MAX_HT_SIZE;
      memcpy(res->restart_ht, ht_enc, n);
      res->restart_ht_len = (uint32_t)n;
    }
  }

  /* --- row grouping: split DocKey from subkeys (InitIterKey,
   * doc_rowwise_iterator.cc:363-404) */
  size_t dockey_len = parse_dockey(schema, ukey, prefix_len, NULL);
  if (!dockey_len || dockey_len > prefix_len) return -1;

  if (!st->active || st->rowkey_len != dockey_len ||
      memcmp(st->rowkey, ukey, dockey_len) != 0) {
    int fr = finalize_row(cx, st);
    if (fr) return fr > 0 ? 1 : -1;
    memcpy(st->rowkey, ukey, dockey_len);
    st->rowkey_len = dockey_len;
    st->active = 1;
    st->base_seen = 0;
    st->col_seen_mask = 0;
    st->liveness_seen = 0;
  }

  if (dockey_len == prefix_len) {
    /* bare row entry (packed row / tombstone) */
    if (!st->base_seen) {
      st->base_seen = 1;
      st->base_value = value;
      st->base_value_len = value_len;
      memcpy(st->base_ht, ht_enc, ht_size);
      st->base_ht_len = ht_size;
    }
  } else {
    /* column subkey: 'K' svarint(column_id) or 'J' svarint(id) —
     * key_bytes.cc:54-56 (AppendColumnId) */
    const uint8_t *sk = ukey + dockey_len;
    size_t sk_len = prefix_len - dockey_len;
    if (sk_len < 2) return -1;
    int64_t col_id;
    size_t sz = orcl_svarint_decode(sk + 1, sk_len - 1, &col_id);
    if (!sz || 1 + sz != sk_len) return -1;
    if (sk[0] == VT_SYS_COL) {
      if (!st->liveness_seen) {
        st->liveness_seen = 1;
        st->liveness_value = value;
        st->liveness_value_len = value_len;
        memcpy(st->liveness_ht, ht_enc, ht_size);
        st->liveness_ht_len = ht_size;
      }
    } else if (sk[0] == VT_COL) {
      int idx = -1;
      for (int i = 0; i < schema->num_value_cols; ++i) {
        if (schema->value_cols[i].column_id == (int32_t)col_id) {
          idx = i;
          break;
        }
      }
      if (idx >= 0 && !(st->col_seen_mask & (1u << idx))) {
        st->col_seen_mask |= 1u << idx;
        st->col_value[idx] = value;
        st->col_value_len[idx] = value_len;
        memcpy(st->col_ht[idx], ht_enc, ht_size);
        st->col_ht_len[idx] = ht_size;
      }
    } else {
      return -1;
    }
  }
  return 0;
}

int orcl_scan(const uint8_t *const *blocks, const size_t *sizes, size_t nblocks,
              orcl_kv_format_t fmt, const orcl_schema_t *schema,
              const orcl_scan_spec_t *spec, orcl_scan_result_t *result,
              orcl_row_cb row_cb, void *cb_arg) {
  memset(result, 0, sizeof(*result));
  for (int g = 0; g < spec->num_aggs; ++g) result->aggs[g].is_null = 1;

  packing_t pk;
  int nullable[ORCL_MAX_COLS];
  for (int i = 0; i < schema->num_value_cols; ++i)
    nullable[i] = schema->value_cols[i].nullable;
  packing_init(&pk, schema, nullable);

  scan_ctx_t cx = {.pk = &pk, .spec = spec, .res = result,
                   .cb = row_cb, .cb_arg = cb_arg, .row_seq = 0};

  /* regular_limit = memcmp-min of encoded(read), encoded(local_limit)
   * == encoded(max(read, local_limit)) — intent_aware_iterator.h:74-77. */
  const orcl_read_time_t *rt = &spec->read_time;
  const uint8_t *reg_limit = rt->read;
  size_t reg_limit_len = rt->read_len;
  if (slice_cmp(rt->local_limit, rt->local_limit_len, rt->read, rt->read_len) <
      0) {
    reg_limit = rt->local_limit;
    reg_limit_len = rt->local_limit_len;
  }

  row_state_t st;
  memset(&st, 0, sizeof(st));

  orcl_block_iter_t it;
  for (size_t b = 0; b < nblocks; ++b) {
    if (orcl_block_iter_init(&it, blocks[b], sizes[b], fmt)) return -1;
    int r;
    while ((r = orcl_block_iter_next(&it)) == 1) {
      int fe = scan_feed_entry(schema, spec, result, &cx, &st, reg_limit,
                               reg_limit_len, it.key, it.key_len, it.value,
                               it.value_len);
      if (fe) return fe > 0 ? 0 : -1;
    }
    if (r < 0) return -1;
  }
  int fr = finalize_row(&cx, &st);
  if (fr < 0) return -1;
  return 0;
}


/* ---------------------------------------------------------------------------
 * Intents-DB merge oracle: see orcl.h. The regular stream and the resolved
 * intent records are merged in rocksdb internal-key order (user key asc,
 * seqno desc — db/dbformat.h:84-110) and fed one at a time through
 * scan_feed_entry — the same pipeline orcl_scan uses.
 * ------------------------------------------------------------------------- */

typedef struct {
  uint8_t *key;
  size_t key_len;
  uint8_t *value;
  size_t value_len;
} resolved_intent_t;

static int ikey_cmp(const uint8_t *a, size_t alen, const uint8_t *b,
                    size_t blen) {
  size_t ua = alen - 8, ub = blen - 8;
  size_t n = ua < ub ? ua : ub;
  int c = memcmp(a, b, n);
  if (c) return c;
  if (ua != ub) return ua < ub ? -1 : 1;
  uint64_t sa, sb;
  memcpy(&sa, a + ua, 8);
  memcpy(&sb, b + ub, 8);
  return sa > sb ? -1 : (sa < sb ? 1 : 0); /* seq DESC */
}

static int resolved_cmp_qsort(const void *pa, const void *pb) {
  const resolved_intent_t *a = (const resolved_intent_t *)pa;
  const resolved_intent_t *b = (const resolved_intent_t *)pb;
  return ikey_cmp(a->key, a->key_len, b->key, b->key_len);
}

int orcl_scan_intents(const uint8_t *const *blocks, const size_t *sizes,
                      size_t nblocks, orcl_kv_format_t fmt,
                      const orcl_schema_t *schema,
                      const orcl_scan_spec_t *spec, const uint8_t *intents,
                      size_t intents_len, const orcl_txn_status_t *txns,
                      uint32_t n_txns, orcl_scan_result_t *result,
                      orcl_row_cb row_cb, void *cb_arg) {
  memset(result, 0, sizeof(*result));
  for (int g = 0; g < spec->num_aggs; ++g) result->aggs[g].is_null = 1;

  /* resolve (ProcessIntent + status-cache lookup) */
  resolved_intent_t *res = NULL;
  size_t n_res = 0, cap_res = 0;
  int rc = -1;
  {
    size_t o = 0, idx = 0;
    while (o < intents_len) {
      uint32_t txn_id, write_id, klen, vlen;
      uint64_t wht;
      if (o + 24 > intents_len) goto done;
      memcpy(&txn_id, intents + o, 4);
      memcpy(&write_id, intents + o + 4, 4);
      memcpy(&wht, intents + o + 8, 8);
      memcpy(&klen, intents + o + 16, 4);
      memcpy(&vlen, intents + o + 20, 4);
      o += 24;
      if (o + klen + vlen > intents_len) goto done;
      const uint8_t *kp = intents + o;
      const uint8_t *vp = intents + o + klen;
      o += (size_t)klen + vlen;
      const orcl_txn_status_t *ts = NULL;
      for (uint32_t t = 0; t < n_txns; ++t)
        if (txns[t].txn_id == txn_id) {
          ts = &txns[t];
          break;
        }
      if (!ts) goto done;
      if (ts->status != 1) continue; /* pending / aborted: invisible */
      if (n_res == cap_res) {
        cap_res = cap_res ? cap_res * 2 : 16;
        res = (resolved_intent_t *)realloc(res,
                                           cap_res * sizeof(*res));
      }
      resolved_intent_t *r = &res[n_res++];
      /* internal key: user key + '#' + DHT(commit, write_id) + seq desc */
      r->key = (uint8_t *)malloc((size_t)klen + 1 + 16 + 8);
      memcpy(r->key, kp, klen);
      size_t kl = klen;
      r->key[kl++] = VT_HYBRID_TIME;
      kl += orcl_dht_encode(ts->commit_ht, write_id, r->key + kl);
      uint64_t seq = (((uint64_t)1 << 55) + idx) << 8 | 0x1 /*kTypeValue*/;
      memcpy(r->key + kl, &seq, 8);
      kl += 8;
      r->key_len = kl;
      /* value: kHybridTime + DHT(write time) + provisional body */
      r->value = (uint8_t *)malloc((size_t)vlen + 1 + 16);
      r->value[0] = VT_HYBRID_TIME;
      size_t vl = 1 + orcl_dht_encode(wht, write_id, r->value + 1);
      memcpy(r->value + vl, vp, vlen);
      r->value_len = vl + vlen;
      ++idx;
    }
  }
  if (n_res > 1) qsort(res, n_res, sizeof(*res), resolved_cmp_qsort);

  {
    packing_t pk;
    int nullable[ORCL_MAX_COLS];
    for (int i = 0; i < schema->num_value_cols; ++i)
      nullable[i] = schema->value_cols[i].nullable;
    packing_init(&pk, schema, nullable);
    scan_ctx_t cx = {.pk = &pk, .spec = spec, .res = result,
                     .cb = row_cb, .cb_arg = cb_arg, .row_seq = 0};
    const orcl_read_time_t *rt = &spec->read_time;
    const uint8_t *reg_limit = rt->read;
    size_t reg_limit_len = rt->read_len;
    if (slice_cmp(rt->local_limit, rt->local_limit_len, rt->read,
                  rt->read_len) < 0) {
      reg_limit = rt->local_limit;
      reg_limit_len = rt->local_limit_len;
    }
    row_state_t st;
    memset(&st, 0, sizeof(st));
    size_t ii = 0;
    orcl_block_iter_t it;
    for (size_t b = 0; b < nblocks; ++b) {
      if (orcl_block_iter_init(&it, blocks[b], sizes[b], fmt)) goto done;
      int r2;
      while ((r2 = orcl_block_iter_next(&it)) == 1) {
        /* flush intents ordered before this regular entry */
        while (ii < n_res &&
               ikey_cmp(res[ii].key, res[ii].key_len, it.key, it.key_len) <
                   0) {
          int fe = scan_feed_entry(schema, spec, result, &cx, &st,
                                   reg_limit, reg_limit_len, res[ii].key,
                                   res[ii].key_len, res[ii].value,
                                   res[ii].value_len);
          if (fe) { rc = fe > 0 ? 0 : -1; goto done; }
          ++ii;
        }
        int fe = scan_feed_entry(schema, spec, result, &cx, &st, reg_limit,
                                 reg_limit_len, it.key, it.key_len,
                                 it.value, it.value_len);
        if (fe) { rc = fe > 0 ? 0 : -1; goto done; }
      }
      if (r2 < 0) goto done;
    }
    while (ii < n_res) {
      int fe = scan_feed_entry(schema, spec, result, &cx, &st, reg_limit,
                               reg_limit_len, res[ii].key, res[ii].key_len,
                               res[ii].value, res[ii].value_len);
      if (fe) { rc = fe > 0 ? 0 : -1; goto done; }
      ++ii;
    }
    int fr = finalize_row(&cx, &st);
    rc = fr < 0 ? -1 : 0;
  }
done:
  for (size_t i = 0; i < n_res; ++i) {
    free(res[i].key);
    free(res[i].value);
  }
  free(res);
  return rc;
}
