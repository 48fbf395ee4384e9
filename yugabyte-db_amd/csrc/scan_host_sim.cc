// yugabyte-db_amd/csrc/scan_host_sim.cc — host-side simulator of the GPU
// per-interval scan algorithm (scan_device.h), sequentially executing the
// exact device code path including head-row deferral, tail walks and
// continuation-flag resolution. TEST INFRASTRUCTURE: lets the CPU test suite
// pin the device algorithm against the oracle without a GPU. Exported as
// ybg_sim_scan from the product library (never used on the product path).
#define YBG_HOST_SIM 1
#define YBG_DEV_QUAL inline
#include "scan_device.h"

#include <vector>

using namespace ybgdev;

extern "C" {

// Mirrors yb_gpu_scan_open's spec translation + k_scan + k_reduce, serially.
int ybg_sim_scan(const ybg_scan_spec_t* spec, const uint8_t* data,
                 const uint64_t* offsets, uint64_t n_blocks,
                 ybg_scan_result_t* out) {
  // ---- DevSpec from ABI spec (same as yb_gpu_scan_open) ----
  DevSpec d;
  memset(&d, 0, sizeof(d));
  const ybg_schema_t& sc = spec->schema;
  d.has_hash = sc.has_hash;
  d.num_hash_cols = sc.num_hash_cols;
  d.num_range_cols = sc.num_range_cols;
  for (int i = 0; i < YBG_MAX_KEYCOLS; ++i) d.key_types[i] = sc.key_types[i];
  d.num_value_cols = sc.num_value_cols;
  int nvar = 0, off_after = 0;
  for (int i = 0; i < sc.num_value_cols; ++i) {
    DevCol& c = d.cols[i];
    c.id = sc.value_cols[i].column_id;
    c.dtype = sc.value_cols[i].dtype;
    bool varlen =
        sc.value_cols[i].nullable || sc.value_cols[i].dtype == YBG_T_STRING;
    c.v1_varlen = varlen;
    c.v1_nvb = nvar;
    c.v1_off = off_after;
    int v1sz;
    switch (sc.value_cols[i].dtype) {
      case YBG_T_BOOL: v1sz = 1; break;
      case YBG_T_INT8: case YBG_T_INT16: case YBG_T_INT32:
      case YBG_T_UINT32: case YBG_T_FLOAT: v1sz = 5; break;
      default: v1sz = 9; break;
    }
    if (varlen) { ++nvar; off_after = 0; }
    else off_after += v1sz;
    switch (sc.value_cols[i].dtype) {
      case YBG_T_BOOL: case YBG_T_INT8: c.v2_fixed = 1; break;
      case YBG_T_INT16: c.v2_fixed = 2; break;
      case YBG_T_INT32: case YBG_T_UINT32: case YBG_T_FLOAT:
        c.v2_fixed = 4; break;
      case YBG_T_STRING: c.v2_fixed = 0; break;
      default: c.v2_fixed = 8; break;
    }
  }
  d.v1_varlen_count = nvar;
  d.fmt = spec->kv_format;
  auto htlim = [](const uint8_t* b, int32_t len, HtLim* o) {
    uint64_t hi = 0, lo = 0;
    for (int i = 0; i < len && i < 16; ++i) {
      uint64_t v = b[i];
      if (i < 8) hi |= v << (56 - 8 * i);
      else lo |= v << (56 - 8 * (i - 8));
    }
    o->hi = hi;
    o->lo = lo;
    o->len = (uint32_t)len;
  };
  htlim(spec->read_time.read, spec->read_time.read_len, &d.read);
  htlim(spec->read_time.local_limit, spec->read_time.local_limit_len,
        &d.local_lim);
  htlim(spec->read_time.global_limit, spec->read_time.global_limit_len,
        &d.global_lim);
  {
    int rl = spec->read_time.read_len, ll = spec->read_time.local_limit_len;
    int n = rl < ll ? rl : ll;
    int cmp = memcmp(spec->read_time.local_limit, spec->read_time.read, n);
    bool local_smaller = cmp < 0 || (cmp == 0 && ll < rl);
    d.reg_lim = local_smaller ? d.local_lim : d.read;
  }
  std::vector<uint8_t> aux;
  d.num_preds = spec->num_preds;
  for (int i = 0; i < spec->num_preds; ++i) {
    const ybg_pred_t& p = spec->preds[i];
    DevPred& dp = d.preds[i];
    dp.is_key_col = p.is_key_col;
    dp.col = p.col;
    dp.op = p.op;
    dp.datum = p.datum;
    dp.str_len = (uint32_t)p.bytes_len;
    dp.rhs_off = (uint32_t)aux.size();
    if (p.bytes && p.bytes_len)
      aux.insert(aux.end(), p.bytes, p.bytes + p.bytes_len);
  }
  d.num_aggs = spec->num_aggs;
  for (int i = 0; i < spec->num_aggs; ++i) {
    d.aggs[i].op = spec->aggs[i].op;
    d.aggs[i].col = spec->aggs[i].col;
  }
  d.lower_off = (uint32_t)aux.size();
  d.lower_len = (uint32_t)spec->lower_bound_len;
  if (spec->lower_bound && spec->lower_bound_len)
    aux.insert(aux.end(), spec->lower_bound,
               spec->lower_bound + spec->lower_bound_len);
  d.upper_off = (uint32_t)aux.size();
  d.upper_len = (uint32_t)spec->upper_bound_len;
  if (spec->upper_bound && spec->upper_bound_len)
    aux.insert(aux.end(), spec->upper_bound,
               spec->upper_bound + spec->upper_bound_len);
  if (aux.empty()) aux.push_back(0);

  // ---- interval table (k_count_restarts + k_emit_intervals, serial) ----
  std::vector<Interval> ivs;
  for (uint64_t b = 0; b < n_blocks; ++b) {
    uint64_t sz = offsets[b + 1] - offsets[b];
    if (sz < 8) return 3;
    const uint8_t* blk = data + offsets[b];
    uint32_t nr = load_le32_u(blk + sz - 4);
    if (nr == 0 || (uint64_t)nr * 4 + 4 > sz) return 3;
    uint32_t restarts_off = (uint32_t)(sz - 4 - (uint64_t)nr * 4);
    for (uint32_t r = 0; r < nr; ++r) {
      uint32_t start = load_le32_u(blk + restarts_off + 4ull * r);
      uint32_t end = (r + 1 < nr)
                         ? load_le32_u(blk + restarts_off + 4ull * (r + 1))
                         : restarts_off;
      ivs.push_back(Interval{(uint32_t)b, start, end});
    }
  }
  uint64_t n_ivs = ivs.size();

  // ---- per-interval scan (k_scan body, serial) ----
  uint64_t entries = 0, scanned = 0, matched = 0;
  uint64_t agg_val[YBG_MAX_AGGS] = {0}, agg_cnt[YBG_MAX_AGGS] = {0};
  std::vector<HeadOut> heads(n_ivs);
  std::vector<uint8_t> walked(n_ivs, 0);
  uint8_t key[kKeyCap], rk_save[kKeyCap];
  for (uint64_t j = 0; j < n_ivs; ++j) {
    bool wn = false;
    if (!scan_one_interval(d, data, offsets, ivs.data(), n_ivs, j, aux.data(),
                           key, rk_save, &entries, &scanned, &matched,
                           agg_val, agg_cnt, &heads[j], &wn))
      return 6;
    walked[j] = wn ? 1 : 0;
  }
  // head ownership resolution (shfl relay / cont flags, serial equivalent)
  for (uint64_t j = 0; j < n_ivs; ++j) {
    bool consumed = j > 0 && walked[j - 1];
    if (consumed) continue;
    scanned += heads[j].scanned;
    matched += heads[j].matched;
    agg_combine(d, agg_val, agg_cnt, heads[j].val, heads[j].cnt);
  }

  memset(out, 0, sizeof(*out));
  out->entries_seen = entries;
  out->rows_scanned = scanned;
  out->rows_matched = matched;
  for (int g = 0; g < d.num_aggs; ++g) {
    out->aggs[g].is_null = (agg_cnt[g] == 0);
    switch (d.aggs[g].op) {
      case YBG_AGG_SUM_DOUBLE:
      case YBG_AGG_MIN_DOUBLE:
      case YBG_AGG_MAX_DOUBLE: {
        double dd;
        memcpy(&dd, &agg_val[g], 8);
        out->aggs[g].value_f64 = dd;
        break;
      }
      default:
        out->aggs[g].value_i64 = (int64_t)agg_val[g];
        break;
    }
  }
  return 0;
}

}  // extern "C"
