/*
 * oracle/orcl.h — CPU oracle for the DocDB SST-block scan-and-filter hot path.
 *
 * TEST INFRASTRUCTURE ONLY. This library is a plain-C restatement of the
 * reference algorithm (yugabyte/yugabyte-db @ /root/reference), used solely as
 * the parity checker and as the reported CPU baseline (bench.py cpu_baseline
 * leg). It is never part of the product path: only tests/, __graft_entry__
 * .smoke() and bench.py's cpu_baseline leg may link or call it. The product
 * GPU path must fail loudly if its HIP extension is missing — it never falls
 * back to this code.
 *
 * Every function cites the reference file:line it restates.
 */
#ifndef ORCL_H
#define ORCL_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---------------------------------------------------------------------------
 * Leaf codecs (yb/util/fast_varint.cc, yb/util/kv_util.h, yb/dockv/doc_kv_util.h)
 * ------------------------------------------------------------------------- */

/* yb fast SIGNED varint: first bit sign (1 positive), n-1 leading ones size
 * prefix, one's complement for negatives.
 * Encode: src/yb/util/fast_varint.cc:80-137 (FastEncodeSignedVarInt).
 * Returns encoded size (1..10). */
size_t orcl_svarint_encode(int64_t v, uint8_t *dest);

/* Decode: src/yb/util/fast_varint.cc:171-227 (FastDecodeSignedVarInt).
 * Returns decoded size, or 0 on error. */
size_t orcl_svarint_decode(const uint8_t *src, size_t src_size, int64_t *v);

/* Size of a DESCENDING signed varint from its first byte(s):
 * src/yb/util/fast_varint.cc:161-169 (FastDecodeDescendingSignedVarIntSize). */
size_t orcl_desc_svarint_size(const uint8_t *src, size_t src_size);

/* yb fast UNSIGNED varint (n-1 leading ones, no sign bit).
 * Encode: src/yb/util/fast_varint.cc:267-289 (FastEncodeUnsignedVarInt). */
size_t orcl_uvarint_encode(uint64_t v, uint8_t *dest);
/* Decode: src/yb/util/fast_varint.cc:291-334 (FastDecodeUnsignedVarInt). */
size_t orcl_uvarint_decode(const uint8_t *src, size_t src_size, uint64_t *v);

/* Classic LEB128 varint32/64 used by the RocksDB block format.
 * src/yb/rocksdb/util/coding.{h,cc} (EncodeVarint32/64, GetVarint32/64Ptr). */
size_t orcl_leb128_encode(uint64_t v, uint8_t *dest);
size_t orcl_leb128_decode(const uint8_t *src, size_t src_size, uint64_t *v);

/* Packed-row-V2 field length: 1 byte (len<<1) if len<128 else 4-byte LE
 * ((len<<1)|1). src/yb/util/fast_varint.cc:358-384 (EncodeFieldLength /
 * DecodeFieldLength). */
size_t orcl_field_length_encode(uint32_t len, uint8_t *out);
size_t orcl_field_length_decode(const uint8_t *inp, uint32_t *len);

/* Key-encoded int64/int32: sign-bit flip + big endian.
 * src/yb/util/kv_util.h:148-158 (AppendInt64ToKey) / :102-113. */
void    orcl_key_int64_encode(int64_t v, uint8_t *dest8);
int64_t orcl_key_int64_decode(const uint8_t *src8);
void    orcl_key_int32_encode(int32_t v, uint8_t *dest4);
int32_t orcl_key_int32_decode(const uint8_t *src4);

/* Key-encoded string: '\0' -> "\0\1", terminated by "\0\0".
 * src/yb/dockv/doc_kv_util.h:101-167 (AppendEncodedStrToKey asc,
 * ZeroEncodeAndAppendStrToKey). Returns bytes written incl. terminator. */
size_t orcl_key_string_encode(const uint8_t *s, size_t len, uint8_t *dest);
/* Decode in place: returns encoded size consumed (incl. "\0\0"), fills
 * out/out_len (out buffer must hold >= encoded size). */
size_t orcl_key_string_decode(const uint8_t *src, size_t src_size,
                              uint8_t *out, size_t *out_len);

/* ---------------------------------------------------------------------------
 * DocHybridTime (yb/common/doc_hybrid_time.cc)
 * ------------------------------------------------------------------------- */

/* YugaByte epoch: src/yb/common/doc_hybrid_time.h:108. */
#define ORCL_YB_EPOCH_MICROS (1500000000ull * 1000000ull)
/* HybridTime repr: micros << 12 | logical. src/yb/common/hybrid_time.h:70-99. */
#define ORCL_HT_LOGICAL_BITS 12
#define ORCL_HT_SIZE_BITS 5
#define ORCL_HT_SIZE_MASK ((1 << ORCL_HT_SIZE_BITS) - 1)
#define ORCL_MAX_HT_SIZE 16 /* 4 descending varints, typically 6-9 bytes */

/* Encode DocHybridTime: 4 descending signed varints (generation=0,
 * micros-epoch, logical, (write_id+1)<<5), then overwrite low 5 bits of last
 * byte with total encoded size. src/yb/common/doc_hybrid_time.cc:39-76
 * (EncodedInDocDbFormat). Returns encoded size. */
size_t orcl_dht_encode(uint64_t ht /* micros<<12|logical */, uint32_t write_id,
                       uint8_t *dest);

/* Encoded size from the last byte (low 5 bits):
 * src/yb/common/doc_hybrid_time.h GetEncodedSize. Returns 0 if invalid. */
size_t orcl_dht_encoded_size_from_end(const uint8_t *key, size_t key_len);

/* Size of an encoded DocHybridTime read from the FRONT (4 descending
 * varints): src/yb/common/doc_hybrid_time.cc:86-101 (EncodedFromStart).
 * Returns 0 on error. */
size_t orcl_dht_size_from_start(const uint8_t *src, size_t src_size);

/* Full decode: src/yb/common/doc_hybrid_time.cc:104-146 (DecodeFrom).
 * Returns consumed size or 0 on error. */
size_t orcl_dht_decode(const uint8_t *src, size_t src_size,
                       uint64_t *ht, uint32_t *write_id);

/* ---------------------------------------------------------------------------
 * RocksDB data-block iteration (yb/rocksdb/table/block.cc, block_internal.h)
 * ------------------------------------------------------------------------- */

#define ORCL_MAX_KEY 256

/* Key-value encoding formats: src/yb/rocksdb/types.h:50-56. */
typedef enum {
  ORCL_ENC_SHARED_PREFIX = 0,
  ORCL_ENC_THREE_SHARED_PARTS = 1,
} orcl_kv_format_t;

typedef struct {
  const uint8_t *data;    /* block bytes (without 5-byte file trailer) */
  size_t size;            /* total block size incl. restart array */
  size_t restarts_offset; /* offset of restart array */
  uint32_t num_restarts;
  orcl_kv_format_t fmt;
  /* iteration state */
  size_t next_offset;          /* offset of next entry */
  uint8_t key[ORCL_MAX_KEY];   /* current (reconstructed) internal key */
  size_t key_len;
  const uint8_t *value;        /* current value slice */
  size_t value_len;
} orcl_block_iter_t;

/* Init from raw block contents. Reads the restart trailer
 * (src/yb/rocksdb/table/block_builder.cc:337-346 Finish). Returns 0 ok. */
int orcl_block_iter_init(orcl_block_iter_t *it, const uint8_t *data,
                         size_t size, orcl_kv_format_t fmt);

/* Decode next entry; returns 1 when an entry was produced, 0 at end, <0 on
 * corruption. shared_prefix: src/yb/rocksdb/table/block.cc:411-436;
 * three_shared_parts: block.cc:287-346 (ParseNextKeyThreeSharedParts) +
 * block_internal.h:54-160 (DecodeEntryThreeSharedParts) +
 * db/dbformat.h:405-476 (IterKey::Update). */
int orcl_block_iter_next(orcl_block_iter_t *it);

/* ---------------------------------------------------------------------------
 * Scan: schema, read time, predicates, aggregates
 * ------------------------------------------------------------------------- */

/* Data types for value columns (subset used by the hot path). */
typedef enum {
  ORCL_T_BOOL = 0,
  ORCL_T_INT8 = 1,
  ORCL_T_INT16 = 2,
  ORCL_T_INT32 = 3,
  ORCL_T_INT64 = 4,
  ORCL_T_UINT32 = 5,
  ORCL_T_UINT64 = 6,
  ORCL_T_FLOAT = 7,
  ORCL_T_DOUBLE = 8,
  ORCL_T_STRING = 9, /* varlen (string/binary) */
} orcl_dtype_t;

/* Key column types (order matters for DocKey parse). */
typedef enum {
  ORCL_KT_INT64 = 0,  /* 'I' + BE64(v ^ sign) — value_type.h kInt64 */
  ORCL_KT_INT32 = 1,  /* 'H' + BE32 — kInt32 */
  ORCL_KT_STRING = 2, /* 'S' + zero-escaped + 00 00 — kString */
} orcl_keytype_t;

typedef struct {
  int32_t column_id;     /* value column id (subkey encoding uses this) */
  orcl_dtype_t dtype;
  int32_t nullable;      /* V1 packing: nullable columns are varlen
                            (schema_packing.cc:45-49) */
} orcl_value_col_t;

#define ORCL_MAX_COLS 32
#define ORCL_MAX_KEYCOLS 8

typedef struct {
  /* DocKey shape: [hash prefix 'G'+BE16 when num_hash_cols>0] + hashed cols +
   * '!' + range cols + '!'. src/yb/dockv/doc_key.h:40-63. */
  int has_hash;             /* 1 if keys carry the kUInt16Hash prefix */
  int num_hash_cols;
  int num_range_cols;
  orcl_keytype_t key_types[ORCL_MAX_KEYCOLS]; /* hashed then range */
  /* Value columns in packing order (schema order after key columns):
   * src/yb/dockv/schema_packing.cc:499-530 (SchemaPacking ctor). */
  int num_value_cols;
  orcl_value_col_t value_cols[ORCL_MAX_COLS];
} orcl_schema_t;

/* Encoded read time limits (EncodedReadHybridTime):
 * src/yb/docdb/intent_aware_iterator.h:61-77, .cc:1446-1455.
 * Each is an encoded DocHybridTime (with write_id = kMaxWriteId by default). */
typedef struct {
  uint8_t read[ORCL_MAX_HT_SIZE];          size_t read_len;
  uint8_t local_limit[ORCL_MAX_HT_SIZE];   size_t local_limit_len;
  uint8_t global_limit[ORCL_MAX_HT_SIZE];  size_t global_limit_len;
} orcl_read_time_t;

/* Build limits from plain hybrid times (micros<<12|logical), write_id=max.
 * kMaxWriteId = 0x7fffffff (src/yb/common/doc_hybrid_time.h). */
void orcl_read_time_init(orcl_read_time_t *rt, uint64_t read_ht,
                         uint64_t local_limit_ht, uint64_t global_limit_ht);

typedef enum {
  ORCL_PRED_GT = 0, ORCL_PRED_GE, ORCL_PRED_LT, ORCL_PRED_LE,
  ORCL_PRED_EQ, ORCL_PRED_NE,
  /* IN list over a numeric column: bytes = n x 8-byte LE datum patterns
   * (hybrid_scan_choices.h:43-60; ql_scanspec.cc:323-346) */
  ORCL_PRED_IN,
  /* tuple membership over multiple numeric key columns
   * (hybrid_scan_choices.h:43-77); bytes = [u32 ncols][u32 colidx x n]
   * [tuples of n x 8-byte LE datums] */
  ORCL_PRED_IN_TUPLE,
  /* option RANGES over a numeric column (hybrid_scan_choices.h
   * OptionRange / mixed bound options): bytes = n x 24-byte records
   * [u64 lo][u64 hi][u32 flags: bit0 lo inclusive, bit1 hi inclusive]
   * [u32 pad]; true when the datum falls in ANY range. */
  ORCL_PRED_IN_RANGE,
} orcl_pred_op_t;

typedef struct {
  int is_key_col;      /* 1: key column index, 0: value column index */
  int col;             /* index into key cols (hashed+range) or value cols */
  orcl_pred_op_t op;
  /* rhs: for numeric, datum holds the value bit-pattern of the column dtype;
   * for string, bytes/len. */
  uint64_t datum;
  const uint8_t *bytes;
  size_t bytes_len;
} orcl_pred_t;

typedef enum {
  ORCL_AGG_COUNT = 0,       /* COUNT(col): skips NULL — doc_expr.cc:250-263 */
  ORCL_AGG_COUNT_STAR = 1,
  ORCL_AGG_SUM_INT64 = 2,   /* starts NULL, adopts first — doc_expr.cc:341-349 */
  ORCL_AGG_SUM_DOUBLE = 3,
  ORCL_AGG_MIN_INT64 = 4,
  ORCL_AGG_MAX_INT64 = 5,
  ORCL_AGG_MIN_DOUBLE = 6,
  ORCL_AGG_MAX_DOUBLE = 7,
} orcl_agg_op_t;

typedef struct {
  orcl_agg_op_t op;
  int col; /* value column index (ignored for COUNT_STAR) */
} orcl_agg_t;

#define ORCL_MAX_PREDS 8
#define ORCL_MAX_AGGS 8

typedef struct {
  orcl_read_time_t read_time;
  int num_preds;
  orcl_pred_t preds[ORCL_MAX_PREDS];
  int num_aggs;
  orcl_agg_t aggs[ORCL_MAX_AGGS];
  /* Optional encoded rowkey bounds (DocKey bytes, no HT): lower inclusive,
   * upper exclusive. NULL => unbounded. qlexpr/ql_scanspec.h:200-267. */
  const uint8_t *lower_bound; size_t lower_bound_len;
  const uint8_t *upper_bound; size_t upper_bound_len;
} orcl_scan_spec_t;

/* Assembled row (PgTableRow analog — src/yb/dockv/pg_row.h:91-179). */
typedef struct {
  /* key column datums: int64 value, or (ptr,len) for strings */
  uint64_t key_datums[ORCL_MAX_KEYCOLS];
  const uint8_t *key_str[ORCL_MAX_KEYCOLS];
  uint32_t key_str_len[ORCL_MAX_KEYCOLS];
  /* value columns */
  uint64_t datums[ORCL_MAX_COLS];     /* numeric bit patterns (host endian) */
  const uint8_t *strp[ORCL_MAX_COLS]; /* string columns: pointer into block/value */
  uint32_t strlen_[ORCL_MAX_COLS];
  uint32_t null_mask;                 /* bit i = value col i NULL */
  /* provenance for deterministic ordering in tests */
  uint64_t seq_in_scan;
} orcl_row_t;

typedef struct {
  int64_t value_i64;
  double value_f64;
  int is_null;
} orcl_agg_result_t;

typedef struct {
  uint64_t rows_scanned;    /* logical rows visited (visible base rows) */
  uint64_t rows_matched;    /* rows passing predicates */
  uint64_t entries_seen;    /* KV entries decoded */
  orcl_agg_result_t aggs[ORCL_MAX_AGGS];
  /* read-restart data (intent_aware_iterator.cc:815-827, 1400-1410):
   * encoded DocHybridTime of the newest visible record with commit time
   * in (read, local_limit]; len 0 = no restart needed */
  uint8_t restart_ht[ORCL_MAX_HT_SIZE];
  uint32_t restart_ht_len;
  uint32_t pad2_;
} orcl_scan_result_t;

/* Row callback for non-aggregate scans (parity tests). Return 0 to continue. */
typedef int (*orcl_row_cb)(const orcl_row_t *row, void *arg);

/* Scan a sequence of blocks (one tablet, in key order).
 * blocks[i]/sizes[i]: raw block contents (no file trailer, no compression —
 * kNoCompression per docdb_rocksdb_util.cc:200-221).
 * Restates: DocRowwiseIterator::FetchNextImpl (doc_rowwise_iterator.cc:690-818),
 * SkipFutureRecords (intent_aware_iterator.cc:1223-1317), FlatGetHelper
 * (doc_reader.cc:1826-1925), packed row decode (schema_packing.cc:489-497,
 * 1043-1121), aggregates (doc_expr.cc:248-395).
 * Returns 0 on success. */
int orcl_scan(const uint8_t *const *blocks, const size_t *sizes, size_t nblocks,
              orcl_kv_format_t fmt, const orcl_schema_t *schema,
              const orcl_scan_spec_t *spec, orcl_scan_result_t *result,
              orcl_row_cb row_cb, void *cb_arg);

/* ---- intents-DB merge oracle (intent_aware_iterator.cc:983-1011
 * ProcessIntent; transaction_status_cache.cc) -------------------------
 * Runtime TWO-STREAM merge: the regular block stream and the resolved
 * intent stream are consumed in internal-key order through the same
 * visibility/row pipeline. Committed intents enter at their COMMIT
 * DocHybridTime with the intent WRITE time as the value's kHybridTime
 * prefix; pending and aborted intents are invisible. (The GPU product
 * path merges at feed time instead — the two independent algorithms are
 * cross-checked by the parity tests.) Intent blob layout per record:
 * [u32 txn_id][u32 write_id][u64 write_ht][u32 klen][u32 vlen]
 * [user key without HT suffix][value body]. */
typedef struct {
  uint32_t txn_id;
  int32_t status; /* 0 pending, 1 committed, 2 aborted */
  uint64_t commit_ht;
} orcl_txn_status_t;

int orcl_scan_intents(const uint8_t *const *blocks, const size_t *sizes,
                      size_t nblocks, orcl_kv_format_t fmt,
                      const orcl_schema_t *schema,
                      const orcl_scan_spec_t *spec, const uint8_t *intents,
                      size_t intents_len, const orcl_txn_status_t *txns,
                      uint32_t n_txns, orcl_scan_result_t *result,
                      orcl_row_cb row_cb, void *cb_arg);

#ifdef __cplusplus
}
#endif
#endif /* ORCL_H */
