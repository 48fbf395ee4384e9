// yugabyte-db_amd/csrc/host_iterator.h — host-side C++ adapter that mirrors
// the reference's YQLRowwiseIteratorIf surface
// (src/yb/docdb/ql_rowwise_iterator_interface.h:32-97) on top of the batch
// C ABI. A tserver integration would register this class in place of
// DocRowwiseIterator.
#pragma once

#include <cstdint>
#include <memory>
#include <vector>

#include "../../include/yb_gpu_scan.h"

namespace ybg {

// PgTableRow analog (dockv/pg_row.h:91-179): null flags + 64-bit datums;
// varlen datums reference bytes in the batch's varlen heap.
struct PgRow {
  uint64_t key_datums[YBG_MAX_KEYCOLS];
  uint64_t datums[YBG_MAX_COLS];
  uint32_t null_mask;
  const uint8_t* varlen;  // heap base; string datum = (len<<40)|offset
};

class GpuDocRowwiseIterator {
 public:
  // ~ DocRowwiseIterator ctor + Init(spec) (doc_rowwise_iterator.cc:165-231)
  explicit GpuDocRowwiseIterator(const ybg_scan_spec_t& spec);
  ~GpuDocRowwiseIterator();

  // ~ the BoundedRocksDbIterator data source: the tablet's data blocks
  int FeedBlocks(const uint8_t* blocks, const uint64_t* offsets,
                 uint64_t n_blocks, bool device_resident);

  // ~ the BloomFilterAwareIterator role (docdb_filter_policy.cc): feed
  // with the tablet's bloom filter consulted — a point scan the filter
  // proves empty uploads nothing and every fetch reports end-of-scan.
  int FeedBlocksBloom(const uint8_t* blocks, const uint64_t* offsets,
                      uint64_t n_blocks, const uint8_t* filter,
                      uint64_t filter_len);

  // ~ YQLRowwiseIteratorIf::PgFetchNext (row-at-a-time over the GPU batch;
  // rows are delivered in tablet key order). Returns 1 row fetched, 0 end
  // of scan, <0 error.
  int PgFetchNext(PgRow* row);

  // ~ PgsqlReadOperation::ExecuteScalar aggregate branch
  // (pgsql_operation.cc:2858-2893)
  int Aggregate(ybg_scan_result_t* out);

  // ~ GetSubDocKey paging position (pgsql_operation.cc:2908-2922)
  int PagingState(uint8_t* key_out, size_t cap, size_t* len);

  // ~ GetTupleId / SeekTuple — the ybctid surface
  // (ql_rowwise_iterator_interface.h:62-71)
  int GetTupleId(uint8_t* key_out, size_t cap, size_t* len);
  int SeekTuple(const uint8_t* dockey, size_t len);

  const char* LastError() const;
  const ybg_schema_t& schema() const { return spec_.schema; }

 private:
  int MaterializeBatch();
  size_t EncodeRowKey(uint64_t row, uint8_t* out, size_t cap);
  // Encode into the growable keybuf_ (no fixed cap: string key columns
  // make encoded DocKeys unbounded). Returns the length, 0 on error.
  size_t EncodeRowKeyDyn(uint64_t row);

  ybg_scan_t* handle_ = nullptr;
  ybg_scan_spec_t spec_;
  ybg_row_batch_t batch_ = {};
  std::vector<uint64_t> order_;  // row indices sorted by sort_key
  std::vector<uint8_t> keybuf_;  // EncodeRowKeyDyn scratch
  uint64_t pos_ = 0;
  uint64_t limit_ = 0;  // rows to deliver this page (0 = all)
  bool batch_ready_ = false;
  int open_rc_ = 0;
};

}  // namespace ybg
