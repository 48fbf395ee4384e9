// yugabyte-db_amd/csrc/host_iterator.cc — see host_iterator.h.
#include "host_iterator.h"

#include <algorithm>
#include <cstring>

namespace ybg {

GpuDocRowwiseIterator::GpuDocRowwiseIterator(const ybg_scan_spec_t& spec)
    : spec_(spec) {
  spec_.emit_rows = 1;
  limit_ = spec.row_limit;
  open_rc_ = yb_gpu_scan_open(&spec_, &handle_);
}

GpuDocRowwiseIterator::~GpuDocRowwiseIterator() {
  if (handle_) yb_gpu_scan_close(handle_);
}

int GpuDocRowwiseIterator::FeedBlocks(const uint8_t* blocks,
                                      const uint64_t* offsets,
                                      uint64_t n_blocks,
                                      bool device_resident) {
  if (open_rc_) return open_rc_;
  return yb_gpu_scan_feed_blocks(handle_, blocks, offsets, n_blocks,
                                 device_resident ? 1 : 0);
}

int GpuDocRowwiseIterator::FeedBlocksBloom(const uint8_t* blocks,
                                           const uint64_t* offsets,
                                           uint64_t n_blocks,
                                           const uint8_t* filter,
                                           uint64_t filter_len) {
  if (open_rc_) return open_rc_;
  return yb_gpu_scan_feed_blocks_bloom(handle_, blocks, offsets, n_blocks,
                                       0, filter, filter_len);
}

int GpuDocRowwiseIterator::MaterializeBatch() {
  int rc = yb_gpu_scan_next_batch(handle_, &batch_);
  if (rc) return rc;
  // Restore tablet key order: the kernels emit rows in completion order
  // with a scan-position sort key (interval, entry offset). Backward
  // scans deliver in DESCENDING key order (FetchNextImpl<kBackward>,
  // doc_rowwise_iterator.cc:690-818) — the parallel scan itself is
  // direction-neutral, direction is a delivery property here.
  order_.resize(batch_.n_rows);
  for (uint64_t i = 0; i < batch_.n_rows; ++i) order_[i] = i;
  const uint64_t* sk = batch_.sort_key;
  if (spec_.backward)
    std::sort(order_.begin(), order_.end(),
              [sk](uint64_t a, uint64_t b) { return sk[a] > sk[b]; });
  else
    std::sort(order_.begin(), order_.end(),
              [sk](uint64_t a, uint64_t b) { return sk[a] < sk[b]; });
  pos_ = 0;
  batch_ready_ = true;
  return 0;
}

int GpuDocRowwiseIterator::PgFetchNext(PgRow* row) {
  if (open_rc_) return -open_rc_;
  if (!batch_ready_) {
    int rc = MaterializeBatch();
    if (rc) return -rc;
  }
  if (pos_ >= batch_.n_rows) return 0;
  if (limit_ && pos_ >= limit_) return 0;  // page exhausted (row_limit)
  uint64_t r = order_[pos_++];
  uint64_t nk = batch_.n_key_cols, nc = batch_.n_value_cols;
  for (uint64_t c = 0; c < nk; ++c)
    row->key_datums[c] = batch_.key_datums[r * nk + c];
  for (uint64_t c = 0; c < nc; ++c)
    row->datums[c] = batch_.datums[r * nc + c];
  row->null_mask = batch_.null_masks[r];
  row->varlen = batch_.varlen;
  return 1;
}

int GpuDocRowwiseIterator::Aggregate(ybg_scan_result_t* out) {
  if (open_rc_) return open_rc_;
  int rc = yb_gpu_scan_execute(handle_);
  if (rc) return rc;
  return yb_gpu_scan_aggregate(handle_, out);
}

// Resumable position (pgsql_operation.cc:2796-2806, 2908-2922): the encoded
// DocKey of the first undelivered row; a follow-up scan resumes with it as
// the inclusive lower bound. len = 0 when the scan is complete.
size_t GpuDocRowwiseIterator::EncodeRowKey(uint64_t row, uint8_t* out,
                                           size_t cap) {
  ybg_key_t k = {};
  const ybg_schema_t& sc = spec_.schema;
  int nk = sc.num_hash_cols + sc.num_range_cols;
  std::vector<std::vector<uint8_t>> strs((size_t)nk);
  for (int c = 0; c < nk; ++c) {
    uint64_t d = batch_.key_datums[row * (uint64_t)nk + c];
    if (sc.key_types[c] == YBG_KT_STRING) {
      uint64_t off = d & ((1ull << 40) - 1);
      uint32_t ln = (uint32_t)(d >> 40);
      strs[c].assign(batch_.varlen + off, batch_.varlen + off + ln);
      k.strs[c] = strs[c].data();
      k.str_lens[c] = ln;
    } else {
      k.datums[c] = d;
    }
  }
  if (sc.has_hash) k.hash = batch_.hashes[row];
  return ybg_encode_dockey(&sc, &k, out, cap);
}

size_t GpuDocRowwiseIterator::EncodeRowKeyDyn(uint64_t row) {
  if (keybuf_.size() < 256) keybuf_.resize(256);
  for (;;) {
    size_t n = EncodeRowKey(row, keybuf_.data(), keybuf_.size());
    if (n) return n;  // 0 = capacity exceeded (ybg_encode_dockey contract)
    if (keybuf_.size() >= (1u << 20)) return 0;  // corrupt, not just long
    keybuf_.resize(keybuf_.size() * 2);
  }
}

int GpuDocRowwiseIterator::PagingState(uint8_t* key_out, size_t cap,
                                       size_t* len) {
  *len = 0;
  if (!batch_ready_ || !limit_ || limit_ >= batch_.n_rows) return 0;
  // forward: the first undelivered key resumes as an INCLUSIVE lower
  // bound; backward: the LAST DELIVERED key resumes as the EXCLUSIVE
  // upper bound of the next (still backward) page
  size_t n = EncodeRowKeyDyn(
      spec_.backward ? order_[limit_ - 1] : order_[limit_]);
  if (!n) return -1;        // row key failed to encode (corrupt state)
  if (n > cap) return -1;   // explicit error, never a silent empty state
  memcpy(key_out, keybuf_.data(), n);
  *len = n;
  return 0;
}

// ~ YQLRowwiseIteratorIf::GetTupleId (ql_rowwise_iterator_interface.h:
// 62-66): the ybctid of the row the last PgFetchNext returned — its
// encoded DocKey.
int GpuDocRowwiseIterator::GetTupleId(uint8_t* key_out, size_t cap,
                                      size_t* len) {
  *len = 0;
  if (!batch_ready_ || pos_ == 0 || pos_ > batch_.n_rows) return -1;
  size_t n = EncodeRowKeyDyn(order_[pos_ - 1]);
  if (!n || n > cap) return -1;
  memcpy(key_out, keybuf_.data(), n);
  *len = n;
  return 0;
}

// ~ YQLRowwiseIteratorIf::SeekTuple (ql_rowwise_iterator_interface.h:
// 68-71): position so the next PgFetchNext returns the row with this
// exact encoded DocKey (ybctid). Returns 0 positioned, 1 not found.
int GpuDocRowwiseIterator::SeekTuple(const uint8_t* dockey, size_t len) {
  if (open_rc_) return -open_rc_;
  if (!batch_ready_) {
    int rc = MaterializeBatch();
    if (rc) return -rc;
  }
  // rows are sorted by tablet key order (descending for backward scans):
  // binary search on encoded keys (keys encode into the growable keybuf_
  // — long string key columns must not truncate, or the order breaks)
  const int dir = spec_.backward ? -1 : 1;
  uint64_t lo = 0, hi = batch_.n_rows;
  while (lo < hi) {
    uint64_t mid = (lo + hi) / 2;
    size_t n = EncodeRowKeyDyn(order_[mid]);
    if (!n) return -1;
    int cmp = memcmp(keybuf_.data(), dockey, n < len ? n : len);
    if (cmp == 0) cmp = (n < len) ? -1 : (n > len ? 1 : 0);
    if (dir * cmp < 0) lo = mid + 1;
    else hi = mid;
  }
  if (lo >= batch_.n_rows) return 1;
  size_t n = EncodeRowKeyDyn(order_[lo]);
  if (n != len || memcmp(keybuf_.data(), dockey, len) != 0) return 1;
  pos_ = lo;
  return 0;
}

const char* GpuDocRowwiseIterator::LastError() const {
  return yb_gpu_last_error();
}

}  // namespace ybg

// ---------------------------------------------------------------------------
// C wrapper so the test suite can drive the C++ adapter row-at-a-time.
// ---------------------------------------------------------------------------
extern "C" {

void* yb_host_iter_open(const ybg_scan_spec_t* spec, const uint8_t* blocks,
                        const uint64_t* offsets, uint64_t n_blocks) {
  auto* it = new ybg::GpuDocRowwiseIterator(*spec);
  if (it->FeedBlocks(blocks, offsets, n_blocks, false) != 0) {
    delete it;
    return nullptr;
  }
  return it;
}

// Returns 1 and fills the row arrays, 0 at end, <0 on error.
int yb_host_iter_next(void* h, uint64_t* key_datums, uint64_t* datums,
                      uint32_t* null_mask, const uint8_t** varlen) {
  auto* it = static_cast<ybg::GpuDocRowwiseIterator*>(h);
  ybg::PgRow row;
  int rc = it->PgFetchNext(&row);
  if (rc != 1) return rc;
  memcpy(key_datums, row.key_datums, sizeof(row.key_datums));
  memcpy(datums, row.datums, sizeof(row.datums));
  *null_mask = row.null_mask;
  *varlen = row.varlen;
  return 1;
}

// YCQL row form — FetchNext(QLTableRow*) analog
// (ql_rowwise_iterator_interface.h:48-52): the same row keyed by COLUMN
// ID instead of projection position (QLTableRow::AllocColumn semantics).
// Fills one (column_id, datum) pair per schema value column plus the key
// columns' ids/datums; NULL columns report their id with the null bit.
int yb_host_iter_next_ql(void* h, int32_t* key_col_ids,
                         uint64_t* key_datums, int32_t* col_ids,
                         uint64_t* datums, uint32_t* null_mask,
                         const uint8_t** varlen) {
  auto* it = static_cast<ybg::GpuDocRowwiseIterator*>(h);
  ybg::PgRow row;
  int rc = it->PgFetchNext(&row);
  if (rc != 1) return rc;
  const ybg_schema_t& sc = it->schema();
  int nk = sc.num_hash_cols + sc.num_range_cols;
  for (int c = 0; c < nk; ++c) {
    key_col_ids[c] = c;  // key columns are identified by position
    key_datums[c] = row.key_datums[c];
  }
  for (int c = 0; c < sc.num_value_cols; ++c) {
    col_ids[c] = sc.value_cols[c].column_id;
    datums[c] = row.datums[c];
  }
  *null_mask = row.null_mask;
  *varlen = row.varlen;
  return 1;
}

int yb_host_iter_paging_state(void* h, uint8_t* key_out, size_t cap,
                              size_t* len) {
  return static_cast<ybg::GpuDocRowwiseIterator*>(h)->PagingState(key_out,
                                                                  cap, len);
}

int yb_host_iter_tuple_id(void* h, uint8_t* key_out, size_t cap,
                          size_t* len) {
  return static_cast<ybg::GpuDocRowwiseIterator*>(h)->GetTupleId(key_out,
                                                                 cap, len);
}

int yb_host_iter_seek_tuple(void* h, const uint8_t* dockey, size_t len) {
  return static_cast<ybg::GpuDocRowwiseIterator*>(h)->SeekTuple(dockey, len);
}

void yb_host_iter_close(void* h) {
  delete static_cast<ybg::GpuDocRowwiseIterator*>(h);
}

}  // extern "C"
