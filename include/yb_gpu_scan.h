/*
 * include/yb_gpu_scan.h — the drop-in C ABI for the MI355X-native DocDB
 * SST-block scan-and-filter path.
 *
 * This ABI is the boundary a YugabyteDB tablet server would call through in
 * place of the CPU DocRowwiseIterator row loop. Each entry point names the
 * reference interface it replaces (paths relative to yugabyte/yugabyte-db):
 *
 *   yb_gpu_scan_open        ~ YQLStorageIf::GetIterator + DocRowwiseIterator
 *                             ctor/Init  (src/yb/docdb/ql_storage_interface.h:
 *                             37-120, doc_rowwise_iterator.h:52-81,
 *                             doc_rowwise_iterator.cc:165-231)
 *   yb_gpu_scan_feed_blocks ~ the BoundedRocksDbIterator data source — the
 *                             already-flushed, decompressed data blocks a
 *                             tablet scan walks (rocksdb/table/block.cc)
 *   yb_gpu_scan_execute     ~ PgsqlReadOperation::ExecuteScalar row loop
 *                             (src/yb/docdb/pgsql_operation.cc:2808-2931)
 *   yb_gpu_scan_aggregate   ~ EvalAggregate/PopulateAggregate
 *                             (src/yb/docdb/pgsql_operation.cc:3171-3186)
 *   yb_gpu_scan_next_batch  ~ YQLRowwiseIteratorIf::PgFetchNext(PgTableRow*)
 *                             batched (src/yb/docdb/
 *                             ql_rowwise_iterator_interface.h:32-97)
 *   yb_gpu_scan_paging_state~ SetPagingState (pgsql_operation.cc:2908-2922)
 *
 * Error convention mirrors yb::Status (src/yb/util/status_fwd.h): int code
 * (0 = OK) + message via yb_gpu_last_error. No exceptions cross the ABI.
 * Threading: one handle == one HIP stream, single caller per handle
 * (CheckInitOnce, doc_rowwise_iterator.cc:128-135); different handles are
 * independent (one per tablet/GPU).
 *
 * The same library also exports the synthetic tablet generator (the write
 * path needed to build benchmark/parity datasets — BlockBuilder
 * rocksdb/table/block_builder.cc, RowPacker dockv/packed_row.cc).
 */
#ifndef YB_GPU_SCAN_H
#define YB_GPU_SCAN_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- shared descriptors -------------------------------------------------- */

/* Value-column data types (DataType subset — src/yb/common/value.messages.h).
 * Values match oracle/orcl.h orcl_dtype_t for test convenience. */
typedef enum {
  YBG_T_BOOL = 0,
  YBG_T_INT8 = 1,
  YBG_T_INT16 = 2,
  YBG_T_INT32 = 3,
  YBG_T_INT64 = 4,
  YBG_T_UINT32 = 5,
  YBG_T_UINT64 = 6,
  YBG_T_FLOAT = 7,
  YBG_T_DOUBLE = 8,
  YBG_T_STRING = 9,
} ybg_dtype_t;

typedef enum {
  YBG_KT_INT64 = 0,
  YBG_KT_INT32 = 1,
  YBG_KT_STRING = 2,
} ybg_keytype_t;

/* src/yb/rocksdb/types.h:50-56 KeyValueEncodingFormat */
typedef enum {
  YBG_ENC_SHARED_PREFIX = 0,
  YBG_ENC_THREE_SHARED_PARTS = 1,
} ybg_kv_format_t;

#define YBG_MAX_COLS 32
#define YBG_MAX_KEYCOLS 8
#define YBG_MAX_PREDS 8
#define YBG_MAX_AGGS 8
#define YBG_MAX_HT 16

typedef struct {
  int32_t column_id;
  int32_t dtype;    /* ybg_dtype_t */
  int32_t nullable; /* affects V1 packing varlen-ness (schema_packing.cc:45-49) */
} ybg_value_col_t;

typedef struct {
  int32_t has_hash;
  int32_t num_hash_cols;
  int32_t num_range_cols;
  int32_t key_types[YBG_MAX_KEYCOLS]; /* ybg_keytype_t */
  int32_t num_value_cols;
  ybg_value_col_t value_cols[YBG_MAX_COLS];
} ybg_schema_t;

/* Encoded read-time limits (EncodedReadHybridTime —
 * src/yb/docdb/intent_aware_iterator.h:61-77). */
typedef struct {
  uint8_t read[YBG_MAX_HT];         int32_t read_len;
  uint8_t local_limit[YBG_MAX_HT];  int32_t local_limit_len;
  uint8_t global_limit[YBG_MAX_HT]; int32_t global_limit_len;
} ybg_read_time_t;

/* Helper: build the three encoded limits from plain hybrid times
 * (micros<<12|logical) with write_id = kMaxWriteId
 * (intent_aware_iterator.cc:1446-1455). */
void ybg_read_time_init(ybg_read_time_t *rt, uint64_t read_ht,
                        uint64_t local_limit_ht, uint64_t global_limit_ht);

typedef enum {
  YBG_PRED_GT = 0, YBG_PRED_GE, YBG_PRED_LT, YBG_PRED_LE,
  YBG_PRED_EQ, YBG_PRED_NE,
  /* IN-list membership. NUMERIC column: bytes = n x 8-byte little-endian
   * datum bit patterns (the column dtype's representation), bytes_len =
   * 8n. STRING column (value or key): bytes = [u32 LE length][raw bytes]
   * records back to back, bytes_len covering them exactly (key-column
   * options are given UNESCAPED; the kernels compare against the
   * zero-escaped rowkey form on the fly, doc_kv_util.h:101-167).
   * This is the scan-side equivalent of the reference's hybrid-scan
   * option filters (docdb/hybrid_scan_choices.h:43-60, IN extraction
   * qlexpr/ql_scanspec.cc:323-346): on a brute-force bandwidth-bound GPU
   * scan the discrete options become a filter, not a seek plan. */
  YBG_PRED_IN,
  /* Tuple membership over MULTIPLE numeric KEY columns — the reference's
   * multi-column option groups ((r1,r3) IN ((1,3),(5,6)) ...),
   * docdb/hybrid_scan_choices.h:43-77. bytes layout:
   *   [u32 LE ncols][u32 LE key-col index x ncols]
   *   [tuples: ncols x 8-byte LE datum patterns each]
   * is_key_col must be 1; col is ignored. */
  YBG_PRED_IN_TUPLE,
  /* Option RANGES over a NUMERIC column (key or value) — the reference's
   * mixed bound options (docdb/hybrid_scan_choices.h:43-77 OptionRange):
   * bytes = n x 24-byte records
   *   [u64 LE lo][u64 LE hi][u32 LE flags: bit0 lo incl, bit1 hi incl]
   *   [u32 pad]
   * true when the column datum falls in ANY range. Numeric bounds use
   * the column dtype's bit pattern (int compares signed, double/float
   * as FP). Range options on the leading range-key column of a
   * range-sharded table also PRUNE the scanned block set (see
   * yb_gpu_scan_feed_blocks). */
  YBG_PRED_IN_RANGE,
} ybg_pred_op_t;

typedef struct {
  int32_t is_key_col;
  int32_t col;
  int32_t op;       /* ybg_pred_op_t */
  uint64_t datum;   /* numeric rhs bit pattern (dtype of the column) */
  const uint8_t *bytes; /* string rhs */
  uint64_t bytes_len;
} ybg_pred_t;

typedef enum {
  YBG_AGG_COUNT = 0,
  YBG_AGG_COUNT_STAR = 1,
  YBG_AGG_SUM_INT64 = 2,
  YBG_AGG_SUM_DOUBLE = 3,
  YBG_AGG_MIN_INT64 = 4,
  YBG_AGG_MAX_INT64 = 5,
  YBG_AGG_MIN_DOUBLE = 6,
  YBG_AGG_MAX_DOUBLE = 7,
} ybg_agg_op_t;

typedef struct {
  int32_t op;  /* ybg_agg_op_t */
  int32_t col; /* value column index */
} ybg_agg_t;

/* Scan spec — the YQLScanSpec/DocPgsqlScanSpec surface reduced to what the
 * hot path consumes (qlexpr/ql_scanspec.h:200-267 bounds;
 * pgsql_operation.cc:602-668 predicates; :3171-3186 aggregates). */
typedef struct {
  ybg_schema_t schema;
  int32_t kv_format; /* ybg_kv_format_t */
  ybg_read_time_t read_time;
  int32_t num_preds;
  ybg_pred_t preds[YBG_MAX_PREDS];
  int32_t num_aggs;
  ybg_agg_t aggs[YBG_MAX_AGGS];
  const uint8_t *lower_bound; uint64_t lower_bound_len; /* incl., encoded DocKey */
  const uint8_t *upper_bound; uint64_t upper_bound_len; /* excl. */
  int32_t emit_rows;  /* 1: materialize matching rows (next_batch) */
  uint64_t row_limit; /* 0 = unlimited; else paging after this many rows */
  /* GROUP BY (config #5 "GROUP-BY-key partial aggregates"):
   * 0 = plain aggregates, else 1 + value-column index to group on. */
  int32_t group_col;
  /* Scan direction (doc_rowwise_iterator.cc:690-818 FetchNextImpl is
   * direction-templated; SkipFutureRecords<Direction::kBackward>,
   * intent_aware_iterator.cc:1319ff). On this engine the bandwidth-bound
   * full scan is direction-NEUTRAL (every block is scanned in parallel
   * and visibility picks the same newest version either way); backward is
   * a DELIVERY-ORDER property of the boundary: rows come back in
   * descending DocKey order, row_limit pages deliver the highest keys
   * first, and the paging state becomes an EXCLUSIVE upper bound for the
   * resumed scan. Aggregate results are unaffected by direction. */
  int32_t backward;
  /* Caller hint: nonzero when the scan expects multi-version rows — a
   * historical-snapshot read (ReadHybridTime in the past of recent
   * writes) or a not-yet-compacted history window. The reference's
   * callers know this when they build the read operation
   * (docdb/doc_read_context.h read time + retention policy); here it
   * selects the version-chain-optimized decode shape of the fast scan
   * kernel (a measured +16% on MVCC-heavy data, -4% on single-version
   * data — results are identical either way, 0 is always safe). */
  int32_t expect_versions;
} ybg_scan_spec_t;

/* ---- scan handle --------------------------------------------------------- */

typedef struct ybg_scan ybg_scan_t;

/* Last error message for the calling thread. */
const char *yb_gpu_last_error(void);

/* Open a scan. Returns 0 and sets *out on success; nonzero error code
 * otherwise. Requires a visible MI355X (gfx950) device: this is the GPU
 * product path — there is NO CPU fallback. */
int yb_gpu_scan_open(const ybg_scan_spec_t *spec, ybg_scan_t **out);

/* Feed the tablet's data blocks. blocks: concatenated raw block contents
 * (uncompressed, no 5-byte file trailer); offsets[i]..offsets[i+1] delimit
 * block i (offsets has n+1 entries). The memory is borrowed for the handle's
 * lifetime. `device` nonzero means `blocks` is already a device pointer
 * (resident in HBM); otherwise it is copied host->device once here. */
int yb_gpu_scan_feed_blocks(ybg_scan_t *s, const uint8_t *blocks,
                            const uint64_t *offsets, uint64_t n_blocks,
                            int device);

/* Feed a complete BlockBasedTable SST file (host memory): parses the
 * version-2 footer + index block (rocksdb/table/format.cc:59-155,
 * block_based_table_builder.cc:658-700), optionally verifies every data
 * block's masked-crc32c trailer, strips trailers and feeds the data
 * blocks. kNoCompression blocks only (the reference-supported uncompressed
 * configuration, docdb_rocksdb_util.cc:200-221). */
int yb_gpu_scan_feed_sst(ybg_scan_t *s, const uint8_t *file, uint64_t size,
                         int verify_checksums);

/* Run the scan asynchronously on the handle's stream. */
int yb_gpu_scan_execute(ybg_scan_t *s);

/* Block until the scan completes; returns status. */
int yb_gpu_scan_wait(ybg_scan_t *s);

typedef struct {
  int64_t value_i64;
  double value_f64;
  int32_t is_null;
  int32_t pad_;
} ybg_agg_result_t;

typedef struct {
  uint64_t rows_scanned;  /* visible rows visited */
  uint64_t rows_matched;  /* rows passing predicates */
  uint64_t entries_seen;  /* KV entries decoded */
  ybg_agg_result_t aggs[YBG_MAX_AGGS];
  /* Read-restart data (GetReadRestartData analog,
   * intent_aware_iterator.cc:1400-1410): when local_limit > read and a
   * visible record committed in (read, local_limit], the ENCODED
   * DocHybridTime of the newest such record (the max seen commit time,
   * smallest encoded bytes). restart_ht_len == 0 means no restart. */
  uint8_t restart_ht[YBG_MAX_HT];
  uint32_t restart_ht_len;
  uint32_t pad2_;
} ybg_scan_result_t;

/* Fetch aggregate results (implies wait). */
int yb_gpu_scan_aggregate(ybg_scan_t *s, ybg_scan_result_t *out);

/* GROUP BY (spec.group_col >= 0): run the grouped scan and return the
 * per-group partials. keys[g]: numeric group key datum, or for string group
 * columns (len<<40)|offset into key_bytes. vals/cnts: [n_groups * num_aggs]
 * (value bit pattern + non-null contribution count; cnt==0 <=> NULL).
 * Returns 0 and *n_groups on success; error if cap exceeded. Integer
 * aggregates are exact; grouped double SUM uses device atomics and is only
 * reproducible up to summation order. */
int yb_gpu_scan_group_aggregate(ybg_scan_t *s, uint64_t *keys,
                                int64_t *vals, uint64_t *cnts,
                                uint8_t *key_bytes, uint64_t key_bytes_cap,
                                uint64_t cap, uint64_t *n_groups);

/* Materialized row batch (PgTableRow analog — dockv/pg_row.h:91-179).
 * Row order within the batch is by (sort_key) = the row's position in the
 * tablet (block index << 32 | entry ordinal); rows are emitted unsorted —
 * callers needing key order sort by sort_key. */
typedef struct {
  uint64_t n_rows;
  uint64_t n_key_cols;
  uint64_t n_value_cols;
  const uint64_t *sort_key;   /* [n_rows] */
  const uint64_t *key_datums; /* [n_rows * n_key_cols] */
  const uint64_t *datums;     /* [n_rows * n_value_cols]; string cols: offset
                                 into varlen heap (hi32 = len) */
  const uint32_t *null_masks; /* [n_rows] bit i = value col i NULL */
  const uint16_t *hashes;     /* [n_rows] kUInt16Hash prefix (hash schemas) */
  const uint8_t *varlen;      /* varlen heap */
  uint64_t varlen_size;
} ybg_row_batch_t;

/* Fetch the materialized matching rows (host-visible; implies wait).
 * Valid until the next execute/close on this handle. */
int yb_gpu_scan_next_batch(ybg_scan_t *s, ybg_row_batch_t *out);

/* Read-restart data of the last execute/group_aggregate
 * (GetReadRestartData analog — intent_aware_iterator.cc:1400-1410): the
 * encoded DocHybridTime (smallest encoded = max commit time) of a visible
 * record committed in (read, local_limit], i.e. the time the statement
 * must restart at. *len_out == 0 means no restart. The plain aggregate
 * path also reports this in ybg_scan_result_t.restart_ht; this entry
 * point exists for the GROUP BY path, whose result shape has no restart
 * field. ht_out must hold YBG_MAX_HT bytes. */
int yb_gpu_scan_restart_data(ybg_scan_t *s, uint8_t *ht_out,
                             uint32_t *len_out);

/* ---- SST bloom filter (docdb_filter_policy / rocksdb FixedSizeFilter,
 * SURVEY 8f-2). Exact reference bit format: 64-byte cache-line blocks,
 * 5-byte trailer, rocksdb::Hash(0xbc9f1d34) with signed-char tail
 * (rocksdb/util/bloom.cc:43-62,384-452; util/hash.cc:32-77). Keys are
 * transformed to the DocKeyPart::kUpToHashOrFirstRange prefix
 * (DocDbAwareV3FilterPolicy, docdb/docdb_filter_policy.cc:110-117).
 * The filter is an optimization only: scan results are identical with
 * or without it. */
uint64_t ybg_filter_slice_size(void);
/* Build a filter over every distinct key prefix of a finished tablet.
 * out receives concatenated fixed-size slices (out_len a multiple of
 * ybg_filter_slice_size()). 0 ok, 8 cap too small, 3 corruption. */
int ybg_filter_from_sst(const uint8_t *data, const uint64_t *offsets,
                        uint64_t n_blocks, int kv_format, uint8_t *out,
                        uint64_t cap, uint64_t *out_len);
/* KeyMayMatch: 1 = prefix may exist, 0 = definitely absent. */
int ybg_filter_may_match(const uint8_t *filt, uint64_t filt_len,
                         const uint8_t *key, uint64_t key_len);
/* Test hook: extracted kUpToHashOrFirstRange prefix length (0 =
 * unparseable => always-match). */
uint64_t ybg_filter_key_prefix_len(const uint8_t *key, uint64_t key_len);
/* feed_blocks + bloom consultation: when the spec's DocKey bounds pin a
 * single filter prefix and the filter proves it absent, the scan is
 * answered empty without uploading any block. filter may be NULL. */
int yb_gpu_scan_feed_blocks_bloom(ybg_scan_t *s, const uint8_t *blocks,
                                  const uint64_t *offsets,
                                  uint64_t n_blocks, int device_resident,
                                  const uint8_t *filter,
                                  uint64_t filter_len);

/* Resumable position (pgsql_operation.cc:2796-2806, 2908-2922): for an
 * unlimited scan reports length 0 (complete). row_limit paging requires
 * delivered-row ORDER, which the batch ABI leaves to the row-at-a-time
 * adapter — use yb_host_iter_paging_state (host_iterator.h); calling this
 * on a limited scan returns an error directing there. */
int yb_gpu_scan_paging_state(ybg_scan_t *s, uint8_t *key_out, size_t cap,
                             size_t *len_out);

/* Kernel-time accounting for the last execute (HIP events on the handle's
 * stream): total kernel ms and the dominant (decode) kernel's ms. */
int yb_gpu_scan_kernel_ms(ybg_scan_t *s, double *total_ms, double *decode_ms);

int yb_gpu_scan_close(ybg_scan_t *s);

/* Returns 1 if a gfx950-class HIP device is visible. */
int yb_gpu_available(void);

/* Select the HIP device subsequent handles bind to (one rank per GPU). */
int yb_gpu_set_device(int device);

/* ---- synthetic tablet generator (write path) ----------------------------- */

typedef struct ybg_builder ybg_builder_t;

/* Create a tablet builder. block_size_target: flush threshold in bytes
 * (--db_block_size_bytes, dockv/packed_row.cc:39; configs use 4096).
 * restart_interval: 16 (rocksdb/table.h:150). */
ybg_builder_t *ybg_builder_create(const ybg_schema_t *schema, int kv_format,
                                  size_t block_size_target,
                                  int restart_interval);

/* Row key for the builder: datums for int key cols, bytes for string cols. */
typedef struct {
  uint16_t hash;       /* kUInt16Hash prefix value (when schema.has_hash) */
  uint64_t datums[YBG_MAX_KEYCOLS];
  const uint8_t *strs[YBG_MAX_KEYCOLS];
  uint64_t str_lens[YBG_MAX_KEYCOLS];
} ybg_key_t;

/* Column values for a packed row: numeric bit patterns / string slices;
 * null[i] nonzero => NULL. */
typedef struct {
  uint64_t datums[YBG_MAX_COLS];
  const uint8_t *strs[YBG_MAX_COLS];
  uint64_t str_lens[YBG_MAX_COLS];
  uint8_t null[YBG_MAX_COLS];
} ybg_rowvals_t;

/* Append entries IN KEY ORDER (caller guarantees DocDB ordering:
 * user_key asc; for equal user keys impossible here since HT differs).
 * ht = micros<<12|logical. seq: rocksdb sequence number for the internal
 * key suffix (dbformat.h:84-110; initial seqno 1<<50 per
 * docdb_rocksdb_util.cc:170). packed_version: 1 or 2. */
int ybg_builder_add_packed_row(ybg_builder_t *b, const ybg_key_t *key,
                               uint64_t ht, uint32_t write_id, uint64_t seq,
                               int packed_version, const ybg_rowvals_t *vals);

/* Column update entry ('K' + svarint(column_id) subkey; value = single V1
 * value). null nonzero writes a column tombstone. */
int ybg_builder_add_column_update(ybg_builder_t *b, const ybg_key_t *key,
                                  int value_col_idx, uint64_t ht,
                                  uint32_t write_id, uint64_t seq,
                                  uint64_t datum, const uint8_t *str,
                                  uint64_t str_len, int null);

/* Row tombstone (bare key, value = kTombstone 'X'). */
int ybg_builder_add_row_tombstone(ybg_builder_t *b, const ybg_key_t *key,
                                  uint64_t ht, uint32_t write_id, uint64_t seq);

/* Append a fully custom entry (tests): raw user key + raw value. The
 * internal key suffix fixed64(seq<<8|kTypeValue) is appended here. */
int ybg_builder_add_raw(ybg_builder_t *b, const uint8_t *user_key,
                        size_t user_key_len, uint64_t seq,
                        const uint8_t *value, size_t value_len);

/* Finish: flush last block. Returns pointers valid until destroy. */
int ybg_builder_finish(ybg_builder_t *b, const uint8_t **data,
                       const uint64_t **offsets, uint64_t *n_blocks,
                       uint64_t *total_bytes, uint64_t *n_entries);

/* Finish as a complete SST FILE (data blocks + [type|masked-crc32c]
 * trailers + empty metaindex + shared-prefix index block + version-2
 * footer). Pointer valid until destroy. */
int ybg_builder_finish_sst(ybg_builder_t *b, const uint8_t **data,
                           uint64_t *total_bytes, uint64_t *n_blocks,
                           uint64_t *n_entries);

/* Host block-entry decode + first-key helpers (the intent merge's and
 * block pruning's decoder; exposed for CPU tests). */
int ybg_decode_block(const uint8_t *blk, uint64_t size, int kv_format,
                     uint8_t *keys_out, uint64_t keys_cap,
                     uint32_t *key_lens, uint8_t *vals_out,
                     uint64_t vals_cap, uint32_t *val_lens,
                     uint64_t cap_entries, uint64_t *n_entries);
int ybg_block_first_key(const uint8_t *blk, uint64_t size, int kv_format,
                        uint8_t *out, uint64_t cap, uint64_t *len);

/* Parse an SST file and report its data-block (offset, size) handles —
 * the exact parser yb_gpu_scan_feed_sst uses (exposed for CPU tests). */
int ybg_sst_index(const uint8_t *file, uint64_t size, int verify,
                  uint64_t *offsets, uint64_t *sizes, uint64_t cap,
                  uint64_t *n_blocks);

void ybg_builder_destroy(ybg_builder_t *b);

/* ---- Intents-DB merge (docdb/intent_aware_iterator.cc:983-1011
 * ProcessIntent + DecodeStrongWriteIntent; transaction status resolution
 * docdb/transaction_status_cache.cc) ----------------------------------
 *
 * The reference scans TWO sorted streams: the regular DB and the intents
 * DB holding provisional (in-flight transaction) records; each intent is
 * resolved against the transaction status cache, and a committed intent
 * behaves as a regular record positioned at its COMMIT hybrid time whose
 * value carries the intent WRITE time as a kHybridTime control prefix
 * (the committed-intent visibility rule, :1249-1267). Aborted and
 * still-pending foreign intents are invisible.
 *
 * This implementation resolves and merges at FEED time: committed intents
 * are synthesized into regular-format records and merge-rebuilt into the
 * affected data blocks (intents are sparse; untouched blocks are reused
 * byte-identical), so every scan then exercises the same committed-intent
 * visibility rule on device. The intent stream is accepted in its
 * post-DecodeStrongWriteIntent form: user key (DocKey [+ subkey], no HT
 * suffix) + write DocHybridTime fields + provisional value body. The
 * intents-DB KEY encoding (intent type sets, reverse txn index) is not
 * transcribed — it is consumed before this boundary. */
typedef struct {
  uint32_t txn_id;
  int32_t status;      /* 0 pending, 1 committed, 2 aborted */
  uint64_t commit_ht;  /* micros<<12|logical, valid when committed */
} ybg_txn_status_t;

/* Intent-stream builder (test/ingest helper). Records may be added in any
 * order; resolution sorts. Value forms mirror the regular builder. */
typedef struct ybg_intents ybg_intents_t;
ybg_intents_t *ybg_intents_create(const ybg_schema_t *schema);
int ybg_intents_add_packed_row(ybg_intents_t *it, const ybg_key_t *key,
                               uint64_t write_ht, uint32_t write_id,
                               uint32_t txn_id, int packed_version,
                               const ybg_rowvals_t *vals);
int ybg_intents_add_column_update(ybg_intents_t *it, const ybg_key_t *key,
                                  int value_col_idx, uint64_t write_ht,
                                  uint32_t write_id, uint32_t txn_id,
                                  uint64_t datum, const uint8_t *str,
                                  uint64_t str_len, int null);
int ybg_intents_add_row_tombstone(ybg_intents_t *it, const ybg_key_t *key,
                                  uint64_t write_ht, uint32_t write_id,
                                  uint32_t txn_id);
int ybg_intents_data(ybg_intents_t *it, const uint8_t **blob,
                     uint64_t *len);
void ybg_intents_destroy(ybg_intents_t *it);

/* Resolve the intent stream against the status table and merge committed
 * intents into the data blocks (host-side; affected blocks are decoded,
 * merge-inserted in internal-key order and re-encoded with the standard
 * BlockBuilder; untouched blocks copy through). Outputs are malloc'd —
 * free with ybg_free. */
int ybg_merge_intents(const uint8_t *blocks, const uint64_t *offsets,
                      uint64_t n_blocks, int kv_format,
                      const uint8_t *intents, uint64_t intents_len,
                      const ybg_txn_status_t *txns, uint32_t n_txns,
                      uint8_t **out_blocks, uint64_t **out_offsets,
                      uint64_t *out_n_blocks, uint64_t *out_total);

/* Feed regular blocks + an intent stream: resolve + merge + feed. */
int yb_gpu_scan_feed_blocks_intents(ybg_scan_t *s, const uint8_t *blocks,
                                    const uint64_t *offsets,
                                    uint64_t n_blocks,
                                    const uint8_t *intents,
                                    uint64_t intents_len,
                                    const ybg_txn_status_t *txns,
                                    uint32_t n_txns);

/* One-call multi-threaded benchmark dataset generator.
 * Produces `rows` rows of the given schema in one tablet: hash prefix walks
 * 0..65535 monotonically, one int64 range key column ascending, packed-row
 * (version as given) values seeded deterministically (seed). versions>1
 * writes that many MVCC versions per row at increasing HTs (newest at
 * ht_base + (versions-1)*ht_step). Caller frees *data with ybg_free. */
typedef struct {
  uint64_t rows;
  uint64_t seed;
  int packed_version;      /* 1 or 2 */
  int kv_format;           /* ybg_kv_format_t */
  uint32_t block_size;     /* e.g. 4096 */
  int restart_interval;    /* 16 */
  int versions;            /* MVCC versions per row (config 4: 5) */
  uint64_t ht_base_micros; /* first version's HT physical micros */
  uint64_t ht_step_micros; /* HT increment between versions */
  int nthreads;            /* 0 = hw concurrency */
  uint64_t group_mod;      /* >0: value column 0 becomes (value %% group_mod)
                              — bounded group-key cardinality for GROUP BY
                              benchmarks */
} ybg_gen_params_t;

int ybg_generate(const ybg_schema_t *schema, const ybg_gen_params_t *p,
                 uint8_t **data, uint64_t **offsets, uint64_t *n_blocks,
                 uint64_t *total_bytes, uint64_t *n_entries);

void ybg_free(void *p);

/* Encode a DocKey (doc_key.h:40-63) from key-column values (paging-state
 * serialization). Returns encoded length, 0 if cap too small. */
size_t ybg_encode_dockey(const ybg_schema_t *schema, const ybg_key_t *key,
                         uint8_t *out, size_t cap);

#ifdef __cplusplus
}
#endif
#endif /* YB_GPU_SCAN_H */
