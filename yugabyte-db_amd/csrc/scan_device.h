// yugabyte-db_amd/csrc/scan_device.h — the per-interval scan algorithm,
// compiled BOTH into the gfx950 kernels (scan_gpu.hip) and into a host
// simulator (scan_host_sim.cc) so the exact device logic is parity-testable
// on CPU. Reference citations are in scan_gpu.hip's header comment.
#pragma once
#include <cstdint>
#include <cstring>

#include "../../include/yb_gpu_scan.h"

#ifndef YBG_DEV_QUAL
#define YBG_DEV_QUAL __device__ __forceinline__
#endif
#define DEV YBG_DEV_QUAL

// tail-update strategy for the fast scanner (see scan_batch_fast)
#ifndef YBG_TAILMODE
#define YBG_TAILMODE 0
#endif

#ifdef YBG_HOST_SIM
// host shims for the HIP bit-cast intrinsics
static inline double __longlong_as_double(long long v) {
  double d;
  memcpy(&d, &v, 8);
  return d;
}
static inline long long __double_as_longlong(double d) {
  long long v;
  memcpy(&v, &d, 8);
  return v;
}
static inline float __uint_as_float(unsigned v) {
  float f;
  memcpy(&f, &v, 4);
  return f;
}
static inline unsigned long long atomicAdd(unsigned long long* p,
                                           unsigned long long v) {
  unsigned long long old = *p;
  *p += v;
  return old;
}
static inline unsigned atomicCAS(unsigned* p, unsigned cmp, unsigned val) {
  unsigned old = *p;
  if (old == cmp) *p = val;
  return old;
}
static inline double ybg_atomic_add_f64(double* p, double v) {
  double old = *p;
  *p += v;
  return old;
}
static inline long long ybg_atomic_min_i64(long long* p, long long v) {
  long long old = *p;
  if (v < old) *p = v;
  return old;
}
static inline long long ybg_atomic_max_i64(long long* p, long long v) {
  long long old = *p;
  if (v > old) *p = v;
  return old;
}
static inline unsigned long long ybg_atomic_exch_u64(unsigned long long* p,
                                                     unsigned long long v) {
  unsigned long long old = *p;
  *p = v;
  return old;
}
static inline unsigned long long ybg_atomic_min_u64(unsigned long long* p,
                                                    unsigned long long v) {
  unsigned long long old = *p;
  if (v < old) *p = v;
  return old;
}
static inline unsigned long long ybg_atomic_max_u64(unsigned long long* p,
                                                    unsigned long long v) {
  unsigned long long old = *p;
  if (v > old) *p = v;
  return old;
}
static inline unsigned ybg_atomic_or_u32(unsigned* p, unsigned v) {
  unsigned old = *p;
  *p |= v;
  return old;
}
static inline unsigned long long ybg_atomic_load_u64(unsigned long long* p) {
  return *p;
}
static inline unsigned ybg_atomic_load_u32(unsigned* p) { return *p; }
static inline void ybg_atomic_store_rel_u32(unsigned* p, unsigned v) {
  *p = v;
}
static inline unsigned ybg_atomic_load_acq_u32(unsigned* p) { return *p; }
#else
// device wrappers (agent-scope L2 atomics; no L1 staleness)
__device__ __forceinline__ double ybg_atomic_add_f64(double* p, double v) {
  return atomicAdd(p, v);
}
__device__ __forceinline__ long long ybg_atomic_min_i64(long long* p,
                                                        long long v) {
  return atomicMin(p, v);
}
__device__ __forceinline__ long long ybg_atomic_max_i64(long long* p,
                                                        long long v) {
  return atomicMax(p, v);
}
__device__ __forceinline__ unsigned long long ybg_atomic_exch_u64(
    unsigned long long* p, unsigned long long v) {
  return atomicExch(p, v);
}
__device__ __forceinline__ unsigned long long ybg_atomic_min_u64(
    unsigned long long* p, unsigned long long v) {
  return atomicMin(p, v);
}
__device__ __forceinline__ unsigned long long ybg_atomic_max_u64(
    unsigned long long* p, unsigned long long v) {
  return atomicMax(p, v);
}
__device__ __forceinline__ unsigned ybg_atomic_or_u32(unsigned* p,
                                                      unsigned v) {
  return atomicOr(p, v);
}
__device__ __forceinline__ unsigned long long ybg_atomic_load_u64(
    unsigned long long* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ unsigned ybg_atomic_load_u32(unsigned* p) {
  return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ void ybg_atomic_store_rel_u32(unsigned* p,
                                                         unsigned v) {
  __hip_atomic_store(p, v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
}
__device__ __forceinline__ unsigned ybg_atomic_load_acq_u32(unsigned* p) {
  return __hip_atomic_load(p, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_AGENT);
}
#endif

namespace ybgdev {

// ---------------------------------------------------------------------------
// Device-side spec (POD kernel arg)
// ---------------------------------------------------------------------------

struct HtLim {      // zero-padded big-endian <=16-byte encoded DocHybridTime
  uint64_t hi, lo;  // hi = bytes [0,8), lo = [8,16)
  uint32_t len;
};

struct DevCol {
  int32_t id;
  int32_t dtype;
  int32_t v1_varlen;  // nullable || string (schema_packing.cc:45-49)
  int32_t v1_nvb;     // varlen columns before
  int32_t v1_off;     // offset after prev varlen column
  int32_t v2_fixed;   // 0 varlen, else 1/2/4/8
};

struct DevPred {
  int32_t is_key_col, col, op;
  uint64_t datum;
  uint32_t str_len;
  uint32_t rhs_off;  // rhs bytes offset in aux buffer
};

struct DevAgg {
  int32_t op, col;
};

// Compact predicate record for the hot eval path: one 16-byte load fetches
// everything a typed compare needs. For string predicates, datum packs
// (str_len << 32) | rhs_off (aux-buffer offset of the RHS bytes).
struct PredC {
  uint64_t datum;
  uint32_t opdt;  // op | (dtype << 8)
  uint32_t pad_;
};

struct DevSpec {
  int32_t has_hash, num_hash_cols, num_range_cols;
  int32_t key_types[YBG_MAX_KEYCOLS];
  int32_t num_value_cols;
  DevCol cols[YBG_MAX_COLS];
  int32_t fmt;
  HtLim read, local_lim, global_lim, reg_lim;
  int32_t num_preds;
  DevPred preds[YBG_MAX_PREDS];
  int32_t num_aggs;
  DevAgg aggs[YBG_MAX_AGGS];
  uint32_t lower_len, lower_off, upper_len, upper_off;
  int32_t v1_varlen_count;
  uint32_t value_pred_mask;  // bit i: predicate i is on a value column
  uint32_t fixed_rk_len;     // encoded DocKey length when schema has no
                             // string key columns (0 = variable)
  int32_t need_rowkey;       // bounds / key predicates / row emission need
                             // the finalized row's key bytes (rk_save)
  int32_t group_col;         // value column to GROUP BY, -1 = none

  // --- Hot-path compaction (host-precomputed). The generic arrays above
  // stay for the cold paths (key predicates, V1 decode); the hot loop reads
  // only these, so the compiler keeps a handful of uniform values live
  // instead of every slot of every array (measured: the full-unroll
  // eval_col held ~600 spilled SGPRs of hoisted spec fields).
  //
  // col_act[i] — one action word per value column:
  //   bits 0-7   mask of predicates evaluated on this column
  //   bits 8-15  mask of aggregate slots fed by this column
  //   bit  16    this is the GROUP BY column
  //   bits 17-20 column dtype (YBG_T_*)
  //   bits 21-24 packed-V2 fixed width (0 = varlen, else 1/2/4/8)
  uint32_t col_act[YBG_MAX_COLS];
  PredC predc[YBG_MAX_PREDS];
  int32_t agg_op[YBG_MAX_AGGS];
  uint32_t key_pred_mask;  // mask of predicates on key columns
  int32_t col_ids[YBG_MAX_COLS];  // column ids (kColB update lookup)
  // When every value column is fixed-width: total packed-V2 body length
  // ('|' + version + flags + fixed bodies; single-byte version assumed) and
  // per-column body offsets from the value start. 0 = fast path off.
  uint32_t v2_fixed_len;
  // host-side dispatch hint mirrored from spec->expect_versions: selects
  // the FUSE template variant of the fast kernel (not read on device)
  int32_t fuse_hint;
  // leading columns with STATIC offsets (everything before the first
  // varlen column, offset-capped at 255): v2_off[0..v2_nfp) are valid and
  // column v2_nfp starts at v2_tail_off. v2_nfp == num_value_cols when the
  // whole schema is fixed-width.
  int32_t v2_nfp;
  uint32_t v2_tail_off;
  uint8_t v2_off[YBG_MAX_COLS];
  // Read-restart tracking (intent_aware_iterator.cc:815-827 UpdateMaxSeenHt
  // + GetReadRestartData :1400-1410): only possible when local_limit >
  // read — a visible record committed in (read, local_limit] forces a
  // restart. 0 disables all per-entry work.
  int32_t track_restart;
};

constexpr uint32_t kActPredM = 0xffu;
constexpr uint32_t kActAggShift = 8, kActAggM = 0xffu;
constexpr uint32_t kActGroup = 1u << 16;
// any scan-visible role: predicates, aggregates or the group key. A
// column with none of these is INERT — decode loops skip its eval_col
// entirely unless rows are being materialized (emit_datums needs every
// projected column).
constexpr uint32_t kActLiveM = 0x1ffffu;
constexpr uint32_t kActDtShift = 17, kActDtM = 0xfu;
constexpr uint32_t kActV2Shift = 21, kActV2M = 0xfu;

#define YBG_UNLIKELY(x) __builtin_expect(!!(x), 0)
#define YBG_LIKELY(x) __builtin_expect(!!(x), 1)

struct Interval {
  uint32_t block;
  uint32_t start;
  uint32_t end;
};

constexpr int kThreads = 256;
constexpr int kKeyCap = 128;
// the fast kernel's per-thread LDS key slot (fixed rowkeys only need the
// rowkey prefix + chunk slack; 80 covers 5 + 8 key cols x 9)
constexpr int kFastKeyCap = 80;
// wave partial: entries, scanned, matched, err, {val,cnt} x MAX_AGGS,
// restart-min {hi, lo, len} (encoded-HT MIN among visible restart
// candidates; len 0 = none)
constexpr int kPartialStride = 4 + 2 * YBG_MAX_AGGS + 3;
// head record: {val,cnt} x MAX_AGGS, scanned, matched
constexpr int kHeadStride = 2 * YBG_MAX_AGGS + 2;

constexpr uint8_t kGroupEnd = 0x21, kHybridTimeByte = 0x23, kNullLow = 0x24,
                  kFloatB = 0x43, kDoubleB = 0x44, kFalseB = 0x46,
                  kUInt16Hash = 0x47, kInt32B = 0x48, kInt64B = 0x49,
                  kSysColB = 0x4A, kColB = 0x4B, kStringB = 0x53,
                  kTrueB = 0x54, kTombB = 0x58, kMergeFlagsB = 0x6B,
                  kTtlB = 0x74, kUserTsB = 0x75, kPackedV1B = 0x7A,
                  kPackedV2B = 0x7C;

// ---------------------------------------------------------------------------
// Byte-level device helpers
// ---------------------------------------------------------------------------

// Unaligned little-endian u64 via three aligned dword loads + funnel shift.
// The compiler emits per-byte loads for unaligned memcpy on amdgcn; this is
// 3 loads instead of 8. May read up to 11 bytes past p — every data buffer
// (generator output, device copy) carries >= 16 bytes of tail slack.
DEV uint64_t load_u64_una(const uint8_t* p) {
  uintptr_t a = (uintptr_t)p;
  const uint32_t* q = (const uint32_t*)(a & ~(uintptr_t)3);
  uint32_t sh = (uint32_t)(a & 3) * 8;
  uint64_t lo = ((uint64_t)q[1] << 32) | q[0];
  if (sh == 0) return lo;
  return (lo >> sh) | ((uint64_t)q[2] << (64 - sh));
}
DEV uint64_t load_be64(const uint8_t* p) {
  return __builtin_bswap64(load_u64_una(p));
}
DEV uint32_t load_be32(const uint8_t* p) {
  return __builtin_bswap32((uint32_t)load_u64_una(p));
}
DEV uint64_t load_le64_u(const uint8_t* p) { return load_u64_una(p); }
DEV uint32_t load_le32_u(const uint8_t* p) { return (uint32_t)load_u64_una(p); }

// LEB128 varint (rocksdb util/coding.h) — one windowed load covers the
// 1-4 byte encodings that dominate (block headers); longer forms fall back.
DEV const uint8_t* leb128(const uint8_t* p, const uint8_t* limit,
                          uint64_t* v) {
  if (p >= limit) return nullptr;
  uint64_t w = load_u64_una(p);
  uint64_t b0 = w & 0xff;
  if (!(b0 & 0x80)) {
    *v = b0;
    return p + 1;
  }
  if (p + 2 <= limit) {
    uint64_t b1 = (w >> 8) & 0xff;
    if (!(b1 & 0x80)) {
      *v = (b0 & 0x7f) | (b1 << 7);
      return p + 2;
    }
    if (p + 3 <= limit) {
      uint64_t b2 = (w >> 16) & 0xff;
      if (!(b2 & 0x80)) {
        *v = (b0 & 0x7f) | ((b1 & 0x7f) << 7) | (b2 << 14);
        return p + 3;
      }
      if (p + 4 <= limit) {
        uint64_t b3 = (w >> 24) & 0xff;
        if (!(b3 & 0x80)) {
          *v = (b0 & 0x7f) | ((b1 & 0x7f) << 7) | ((b2 & 0x7f) << 14) |
               (b3 << 21);
          return p + 4;
        }
      }
    }
  }
  // slow path (>= 5 bytes or close to limit)
  uint64_t result = 0;
  int shift = 0;
  while (p < limit && shift <= 63) {
    uint64_t b = *p++;
    if (b & 128) {
      result |= (b & 127) << shift;
    } else {
      result |= b << shift;
      *v = result;
      return p;
    }
    shift += 7;
  }
  return nullptr;
}

// yb fast signed varint (util/fast_varint.cc:171-227)
DEV const uint8_t* svarint(const uint8_t* p, const uint8_t* limit,
                           int64_t* v) {
  if (p >= limit) return nullptr;
  uint64_t wbe = __builtin_bswap64(load_u64_una(p));  // bytes big-endian
  uint32_t header = (uint32_t)(wbe >> 48);
  uint64_t neg = -(uint64_t)((header & 0x8000u) == 0);
  header ^= (uint32_t)neg;
  int n = __builtin_clz((~header & 0x7fffu) | 0x20u) - 16;
  if (p + n > limit) return nullptr;
  if (n <= 8) {
    uint64_t temp = wbe >> (8 * (8 - n));
    uint64_t mask = (1ull << (7 * n - 1)) - 1;
    *v = (int64_t)(((temp & mask) | (~mask & neg)) - neg);
    return p + n;
  }
  uint64_t temp = 0;
  for (int i = 0; i < n; ++i) temp = (temp << 8) | p[i];
  uint64_t mask = (n >= 10) ? ~0ull : 0x3fffffffffffffffull;
  *v = (int64_t)(((temp & mask) | (~mask & neg)) - neg);
  return p + n;
}

DEV int desc_svarint_size(const uint8_t* p, const uint8_t* limit) {
  if (p >= limit) return 0;
  uint32_t b0 = p[0];
  uint32_t b1 = (p + 1 < limit) ? p[1] : 0;
  uint32_t header = (b0 << 8) | b1;
  uint64_t neg = -(uint64_t)((header & 0x8000u) == 0);
  header ^= (uint32_t)neg;
  return __builtin_clz((~header & 0x7fffu) | 0x20u) - 16;
}

// yb fast unsigned varint (util/fast_varint.cc:291-334)
DEV const uint8_t* uvarint(const uint8_t* p, const uint8_t* limit,
                           uint64_t* v) {
  if (p >= limit) return nullptr;
  uint32_t first = p[0];
  // kUnsignedVarIntSize (fast_varint.cc:28-37): clz32((i<<1)^0x1ff)-22;
  // shifting the 9-bit value left by 22 folds the -22 into the clz.
  int n = __builtin_clz((((first << 1) ^ 0x1ffu) << 22) | 1u);
  if (p + n > limit) return nullptr;
  if (n == 1) {
    *v = first & 0x7f;
    return p + 1;
  }
  if (n <= 8) {
    uint64_t wbe = __builtin_bswap64(load_u64_una(p));
    uint64_t masked =
        (wbe >> (8 * (8 - n))) & ((1ull << (8 * n - n)) - 1);
    // top n-1 bits of the first byte are the size prefix; keep 8-n bits
    uint64_t first_bits = (uint64_t)(first & ((1u << (8 - n)) - 1))
                          << (8 * (n - 1));
    uint64_t tail = (wbe >> (8 * (8 - n))) & ((n > 1) ? ((1ull << (8 * (n - 1))) - 1) : 0);
    *v = first_bits | tail;
    (void)masked;
    return p + n;
  }
  uint64_t result = 0;
  int i = 0;
  if (n == 9) {
    if (p[1] & 0x80) {
      n = 10;
      if (p + n > limit) return nullptr;
      result = p[1] & 0x3f;
      i = 2;
    } else {
      result = 0;
      i = 1;
    }
  } else {
    result = first & ((1u << (8 - n)) - 1);
    i = 1;
  }
  for (; i < n; ++i) result = (result << 8) | p[i];
  *v = result;
  return p + n;
}

DEV int dht_size_from_start(const uint8_t* p, const uint8_t* limit) {
  int off = 0;
  for (int i = 0; i < 4; ++i) {
    int sz = desc_svarint_size(p + off, limit);
    if (sz == 0 || p + off + sz > limit) return 0;
    off += sz;
  }
  return off;
}

template <class PtrT>
DEV void slice_u128(PtrT p, uint32_t len, uint64_t* hi, uint64_t* lo) {
  uint32_t n = len < 16 ? len : 16;
  uint64_t h = __builtin_bswap64(load_u64_una(&p[0]));
  if (n < 8) {
    uint32_t sh = 8 * (8 - n);
    h = (h >> sh) << sh;
    *hi = h;
    *lo = 0;
    return;
  }
  *hi = h;
  uint64_t l = 0;
  if (n > 8) {
    l = __builtin_bswap64(load_u64_una(&p[8]));
    uint32_t sh = 8 * (16 - n);
    l = (l >> sh) << sh;
  }
  *lo = l;
}

// Register-resident user-key tail: big-endian bytes
// [ukey_len-16, ukey_len) as (thi, tlo) — thi = bytes [-16,-8), tlo =
// [-8,0). The DocHybridTime suffix, its size byte and the kHybridTime
// marker all live in this window for every key the benchmark configs
// produce, so per-entry visibility runs on registers; the fast decode path
// patches the window in place and every other path invalidates it (rebuilt
// from LDS once per invalidation). Requires ukey_len >= 16.
DEV void tail_from_lds(const uint8_t* key, uint32_t ukey_len, uint64_t* thi,
                       uint64_t* tlo) {
  *thi = __builtin_bswap64(load_u64_una(key + ukey_len - 16));
  *tlo = __builtin_bswap64(load_u64_una(key + ukey_len - 8));
}

// patch byte at absolute key position pos (window-relative, branchless)
DEV void tail_patch(uint64_t* thi, uint64_t* tlo, uint32_t ukey_len,
                    uint32_t pos, uint8_t b) {
  int32_t off = (int32_t)pos - (int32_t)(ukey_len - 16);
  uint32_t sh = 8 * (7 - ((uint32_t)off & 7));
  uint64_t m = off >= 0 ? (0xffull << sh) : 0;
  uint64_t v = (uint64_t)b << sh;
  bool hi = off < 8;
  *thi = (*thi & ~(hi ? m : 0)) | (hi ? (v & m) : 0);
  *tlo = (*tlo & ~(hi ? 0 : m)) | (hi ? 0 : (v & m));
}

// memcmp + length tiebreak over zero-padded 16-byte slices (Slice::compare)
DEV int u128_slice_cmp(uint64_t ahi, uint64_t alo, uint32_t alen,
                       uint64_t bhi, uint64_t blo, uint32_t blen) {
  if (ahi != bhi) return ahi < bhi ? -1 : 1;
  if (alo != blo) return alo < blo ? -1 : 1;
  if (alen != blen) return alen < blen ? -1 : 1;
  return 0;
}

// ---------------------------------------------------------------------------
// Register-window stream reader. The scan's decode is a strictly forward
// byte stream; parsing through per-byte global loads makes every entry a
// chain of dependent ~200-900-cycle memory ops. Rdr keeps 32 bytes in four
// registers; peeks are funnel shifts (VALU), consumes shift the window and
// refill with ALIGNED 8-byte loads whose results are not needed until >=16
// bytes later — the latency hides behind parsing. Buffers must carry >= 48
// bytes of tail slack (generator/device allocations do).
// ---------------------------------------------------------------------------
struct Rdr {
  uint64_t q0, q1, q2, q3;  // aligned u64s: [base, base+32)
  const uint8_t* base;      // 8-byte aligned
  uint32_t k;               // logical position = base + k, k in [0,16)

  DEV void init(const uint8_t* p) {
    base = (const uint8_t*)((uintptr_t)p & ~(uintptr_t)7);
    k = (uint32_t)((uintptr_t)p & 7);
    const uint64_t* q = (const uint64_t*)base;
    q0 = q[0];
    q1 = q[1];
    q2 = q[2];
    q3 = q[3];
  }
  DEV const uint8_t* pos() const { return base + k; }
  // All window selection below is ARITHMETIC on values (mask/ternary on
  // u64s), never an if/else over which member to read: the optimizer was
  // turning the branchy form into a dynamically-indexed q[] array in
  // scratch memory — every peek/refill became an HBM-backed scratch
  // round-trip (~5 GB/dispatch of WRITE_SIZE on the 100M-row bench).
  // 8 bytes at the current position, little-endian
  DEV uint64_t peek8() const {
    uint32_t kk = k;
    uint64_t m1 = (uint64_t)0 - (uint64_t)(kk >= 8);
    uint64_t lo = (q0 & ~m1) | (q1 & m1);
    uint64_t hi = (q1 & ~m1) | (q2 & m1);
    uint32_t sh = 8 * (kk & 7);
    return sh ? ((lo >> sh) | (hi << (64 - sh))) : lo;
  }
  // Slide the window so k < 8 (one aligned load): callers that peek deep
  // (peek8_at(16), body bytes up to 24 ahead) must run this first — the
  // window holds [base, base+32) and deep peeks at large k silently wrap.
  DEV void align8() {
    if (k >= 8) {
      q0 = q1;
      q1 = q2;
      q2 = q3;
      base += 8;
      q3 = ((const uint64_t*)base)[3];
      k -= 8;
    }
  }
  // 8 bytes at position + off (k + off + 7 must be < 32, i.e. call
  // align8() first when off can reach 16)
  DEV uint64_t peek8_at(uint32_t off) const {
    uint32_t kk = k + off;
    uint64_t m1 = (uint64_t)0 - (uint64_t)(kk >= 8);
    uint64_t m2 = (uint64_t)0 - (uint64_t)(kk >= 16);
    uint64_t lo = (q0 & ~m1) | (q1 & (m1 & ~m2)) | (q2 & m2);
    uint64_t hi = (q1 & ~m1) | (q2 & (m1 & ~m2)) | (q3 & m2);
    uint32_t sh = 8 * (kk & 7);
    return sh ? ((lo >> sh) | (hi << (64 - sh))) : lo;
  }
  DEV void consume(uint32_t n) {  // n <= 16
    // refill lazily in 16-byte steps: the two adjacent u64 loads hit the
    // same (or neighbouring) cache line back-to-back, halving the number
    // of temporally-separated line touches vs an 8-byte slide (the L2 sees
    // ~10^6 concurrent per-lane streams, so a line rarely survives between
    // two separate visits)
    k += n;
    while (k >= 16) {
      q0 = q2;
      q1 = q3;
      base += 16;
      const uint64_t* q = (const uint64_t*)base;
      q2 = q[2];
      q3 = q[3];
#if !defined(YBG_NO_PF) && defined(__HIP_DEVICE_COMPILE__)
      // sliding L2 prefetch: one extra independent load per 64 consumed
      // bytes, ~192 B ahead of the stream (device buffers carry 256 B of
      // tail slack for this). The asm keeps the otherwise-dead load.
      if (((uintptr_t)base & 63) == 0) {
        uint64_t pf = q[24];
        asm volatile("" ::"v"(pf));
      }
#endif
      k -= 16;
    }
  }
  DEV void skip(uint32_t n) {
    if (n <= 16) consume(n);
    else init(pos() + n);
  }
  DEV void seek(const uint8_t* p) {
    if (p == pos()) return;
    uintptr_t d = (uintptr_t)p - (uintptr_t)base;
    if (d < 16) {  // still inside the register window
      k = (uint32_t)d;
      return;
    }
    init(p);
  }
};

// ---------------------------------------------------------------------------
// Interval table construction
// ---------------------------------------------------------------------------

#ifndef YBG_HOST_SIM
__global__ void k_count_restarts(const uint8_t* __restrict__ data,
                                 const uint64_t* __restrict__ offsets,
                                 uint64_t n_blocks,
                                 uint32_t* __restrict__ counts,
                                 int* __restrict__ error) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (; i < n_blocks; i += stride) {
    uint64_t sz = offsets[i + 1] - offsets[i];
    uint32_t nr = sz >= 8 ? load_le32_u(data + offsets[i + 1] - 4) : 0;
    if (sz < 8 || nr == 0 || (uint64_t)nr * 4 + 4 > sz) {
      atomicExch(error, 1);
      counts[i] = 0;
      continue;
    }
    counts[i] = nr;
  }
}

__global__ void k_emit_intervals(const uint8_t* __restrict__ data,
                                 const uint64_t* __restrict__ offsets,
                                 uint64_t n_blocks,
                                 const uint64_t* __restrict__ iv_base,
                                 Interval* __restrict__ ivs) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x;
  for (; i < n_blocks; i += stride) {
    uint64_t sz = offsets[i + 1] - offsets[i];
    const uint8_t* blk = data + offsets[i];
    uint32_t nr = load_le32_u(blk + sz - 4);
    uint32_t restarts_off = (uint32_t)(sz - 4 - (uint64_t)nr * 4);
    uint64_t base = iv_base[i];
    for (uint32_t r = 0; r < nr; ++r) {
      uint32_t start = load_le32_u(blk + restarts_off + 4ull * r);
      uint32_t end = (r + 1 < nr)
                         ? load_le32_u(blk + restarts_off + 4ull * (r + 1))
                         : restarts_off;
      ivs[base + r] = Interval{(uint32_t)i, start, end};
    }
  }
}
#endif  // !YBG_HOST_SIM

// ---------------------------------------------------------------------------
// Entry decode (both formats) into the LDS key scratch
// ---------------------------------------------------------------------------

struct EntryRef {
  const uint8_t* value;
  uint32_t value_len;
  uint32_t shared;  // shared-prefix bytes reused from the previous key
};

// Mirrors DecodeEntryThreeSharedParts (block_internal.h:54-160) +
// IterKey::Update (db/dbformat.h:405-476) and the shared_prefix DecodeEntry
// (block.cc:411-436). Returns pointer past the entry or nullptr.
//
// Key representation: bytes [0, key_len-8) live in the LDS array `key`; the
// trailing 8-byte internal component (seqno<<8|type, dbformat.h:84-110) is
// cached in *last8 (little-endian). Nothing on the read path parses the
// internal component, and the dominant "last8 reused (+0x100)" case then
// performs no LDS traffic for it. For the rare paths that materialize tail
// bytes into LDS (restarts, no-reuse deltas), the register is reloaded from
// LDS afterwards; stale tail bytes in LDS are never read (the middle-copy
// consults *last8 for source positions inside the previous key's tail).
// rkb/changed: when rkb > 0, any write that modifies a key byte at a
// position < rkb sets *changed — used for row-change detection without a
// separate saved-rowkey compare (valid for fixed-length rowkeys).
DEV const uint8_t* decode_entry_ptr(int fmt, const uint8_t* p,
                                    const uint8_t* limit, uint8_t* key,
                                    uint32_t* key_len, uint64_t* last8,
                                    uint32_t rkb, bool* changed,
                                    EntryRef* out) {
  auto wr = [&](uint64_t pos, uint8_t b) {
    if (pos < rkb && key[pos] != b) *changed = true;
    key[pos] = b;
  };
  if (fmt == YBG_ENC_SHARED_PREFIX) {
    uint64_t shared, non_shared, value_len;
    if (!(p = leb128(p, limit, &shared))) return nullptr;
    if (!(p = leb128(p, limit, &non_shared))) return nullptr;
    if (!(p = leb128(p, limit, &value_len))) return nullptr;
    if ((uint64_t)(limit - p) < non_shared + value_len) return nullptr;
    uint64_t new_len = shared + non_shared;
    if (shared > *key_len || new_len > kKeyCap || new_len < 9)
      return nullptr;
    uint64_t prev_len = *key_len;
    // ensure LDS holds prev tail bytes the shared prefix reaches into
    if (prev_len >= 8 && shared > prev_len - 8) {
      uint64_t old8 = *last8;
      for (int i = 0; i < 8; ++i)
        wr(prev_len - 8 + i, (uint8_t)(old8 >> (8 * i)));
    }
    for (uint32_t i = 0; i < (uint32_t)non_shared; ++i)
      wr(shared + i, p[i]);
    *key_len = (uint32_t)new_len;
    *last8 = load_u64_una(key + new_len - 8);
    out->value = p + non_shared;
    out->value_len = (uint32_t)value_len;
    out->shared = (uint32_t)shared;
    return out->value + value_len;
  }

  if (limit - p < 2) return nullptr;
  uint64_t encoded_1;
  if (!(p = leb128(p, limit, &encoded_1))) return nullptr;
  uint32_t value_size = (uint32_t)(encoded_1 >> 2);
  uint64_t last8_inc = (encoded_1 & 2) << 7;
  bool frequent = encoded_1 & 1;

  uint32_t shared_prefix = 0, ns1 = 0, ns2 = 0, reuse8 = 0;
  int64_t ns1_delta = 0, ns2_delta = 0;
  bool shared_something;
  uint64_t tmp;

  if (frequent) {
    if (!(p = leb128(p, limit, &tmp))) return nullptr;
    shared_prefix = (uint32_t)tmp;
    reuse8 = 8;
    shared_something = true;
    ns1 = 1;
    ns2 = 1;
  } else {
    uint32_t e2 = *p++;
    if ((e2 & 1) == 0) {
      shared_something = false;
      if (e2 == 0) {
        if (!(p = leb128(p, limit, &tmp))) return nullptr;
        ns1 = (uint32_t)tmp;
      } else {
        ns1 = e2 >> 1;
      }
    } else {
      shared_something = true;
      if ((e2 & 2) == 0) {
        reuse8 = 8;
        ns2_delta = (e2 >> 2) & 1;
        ns1 = (e2 >> 3) & 7;
        ns2 = (e2 >> 6) & 3;
      } else {
        reuse8 = (e2 & 4) ? 8 : 0;
        if (!(p = leb128(p, limit, &tmp))) return nullptr;
        ns1 = (uint32_t)tmp;
        if (e2 & 8) {
          if (!(p = svarint(p, limit, &ns1_delta))) return nullptr;
        }
        if (e2 & 16) {
          if (!(p = leb128(p, limit, &tmp))) return nullptr;
          ns2 = (uint32_t)tmp;
        }
        if (e2 & 32) {
          if (!(p = svarint(p, limit, &ns2_delta))) return nullptr;
        }
      }
      if (!(p = leb128(p, limit, &tmp))) return nullptr;
      shared_prefix = (uint32_t)tmp;
    }
  }
  if ((uint64_t)(limit - p) < (uint64_t)ns1 + ns2 + value_size) return nullptr;

  if (!shared_something) {
    // restart / full key inline (block.cc:311-317)
    if (ns1 > kKeyCap || ns1 < 9) return nullptr;
    if (ns1 <= kKeyCap - 8) {
      // u64-chunk copy (key slots are 8-aligned, cap leaves write slack);
      // row-change compare runs over the first rkb bytes up front
      if (rkb) {
        bool ch = false;
        uint32_t full = rkb & ~7u;
        for (uint32_t i = 0; i < full; i += 8)
          if (load_u64_una(&key[i]) != load_u64_una(p + i)) ch = true;
        if (rkb & 7) {
          uint64_t m = (1ull << (8 * (rkb & 7))) - 1;
          if (((load_u64_una(&key[full]) ^ load_u64_una(p + full)) & m) != 0)
            ch = true;
        }
        if (ch) *changed = true;
      }
      for (uint32_t i = 0; i < ns1; i += 8) {
        uint64_t w8 = load_u64_una(p + i);
        __builtin_memcpy(&key[i], &w8, 8);
      }
    } else {
      for (uint32_t i = 0; i < ns1; ++i) wr(i, p[i]);
    }
    *key_len = ns1;
    *last8 = load_u64_una(key + ns1 - 8);
    out->value = p + ns1;
    out->value_len = value_size;
    out->shared = 0;
    return out->value + value_size;
  }

  uint64_t prev_mid_start = (uint64_t)shared_prefix + ns1 - (uint64_t)ns1_delta;
  uint64_t prev_ns2 = (uint64_t)ns2 - (uint64_t)ns2_delta;
  uint64_t prev_except_mid = prev_mid_start + prev_ns2 + reuse8;
  if (*key_len < prev_except_mid) return nullptr;
  uint64_t mid = *key_len - prev_except_mid;
  if (shared_prefix + mid + reuse8 == 0) return nullptr;

  uint64_t new_mid_start = shared_prefix + ns1;
  uint64_t new_ns2_start = new_mid_start + mid;
  uint64_t new_last8_start = new_ns2_start + ns2;
  uint64_t new_key_size = new_last8_start + reuse8;
  if (new_key_size > kKeyCap || new_key_size < 9) return nullptr;
  uint64_t prev_len = *key_len;

  if (reuse8) {
    // FAST PATH: the internal component is carried in the register; only
    // body bytes move. The middle source region always lies before the
    // previous tail (prev_except_mid accounting), so plain LDS moves apply.
    if (new_mid_start != prev_mid_start && mid > 0) {
      if (new_mid_start < prev_mid_start) {
        for (uint64_t i = 0; i < mid; ++i)
          wr(new_mid_start + i, key[prev_mid_start + i]);
      } else {
        for (uint64_t i = mid; i-- > 0;)
          wr(new_mid_start + i, key[prev_mid_start + i]);
      }
    }
    for (uint32_t i = 0; i < ns1; ++i) wr(shared_prefix + i, p[i]);
    for (uint32_t i = 0; i < ns2; ++i) wr(new_ns2_start + i, p[ns1 + i]);
    *last8 += last8_inc;
  } else {
    // No tail reuse (rare): the retained prefix or the shared middle may
    // reach into the previous key's register-resident tail. Materialize the
    // old tail into LDS first, then proceed with plain LDS moves and reload
    // the register from the new tail bytes.
    if (prev_len >= 8) {
      uint64_t old8 = *last8;
      for (int i = 0; i < 8; ++i)
        wr(prev_len - 8 + i, (uint8_t)(old8 >> (8 * i)));
    }
    if (new_mid_start != prev_mid_start && mid > 0) {
      if (new_mid_start < prev_mid_start) {
        for (uint64_t i = 0; i < mid; ++i)
          wr(new_mid_start + i, key[prev_mid_start + i]);
      } else {
        for (uint64_t i = mid; i-- > 0;)
          wr(new_mid_start + i, key[prev_mid_start + i]);
      }
    }
    for (uint32_t i = 0; i < ns1; ++i) wr(shared_prefix + i, p[i]);
    for (uint32_t i = 0; i < ns2; ++i) wr(new_ns2_start + i, p[ns1 + i]);
    *last8 = load_u64_una(key + new_key_size - 8);
  }
  *key_len = (uint32_t)new_key_size;
  out->value = p + ns1 + ns2;
  out->value_len = value_size;
  out->shared = shared_prefix;
  return out->value + value_size;
}

// Fast-entry decode core for the dominant three_shared_parts encodings
// (frequent case and case 2.1.1, block_builder_internal.h:139-183) — the
// whole header plus the <=10 non-shared key bytes are parsed from the
// register window with no per-byte memory chain, and in these cases the
// shared middle never moves (ns1_delta == 0), so only ns1+ns2 LDS bytes
// are written. Returns the pointer past the entry (reader at the value
// start) or nullptr when the entry is NOT one of the fast forms (state
// untouched; caller falls back — or, in the specialized kernel, aborts
// the batch). Shared verbatim by decode_entry and k_scan_fast.
DEV const uint8_t* decode_entry_fast(Rdr* rdr, const uint8_t* limit,
                                     uint8_t* key, uint32_t* key_len,
                                     uint64_t* last8, uint32_t rkb,
                                     bool* changed, EntryRef* out,
                                     uint64_t* thi, uint64_t* tlo) {
  const uint8_t* p = rdr->pos();
  if (limit - p < 8) return nullptr;
  {
    uint64_t w = rdr->peek8();
    uint32_t b0 = (uint32_t)(w & 0xff);
    uint64_t e1;
    uint32_t e1len;
    bool ok = true;
    if (!(b0 & 0x80)) {
      e1 = b0;
      e1len = 1;
    } else {
      uint32_t b1 = (uint32_t)((w >> 8) & 0xff);
      if (!(b1 & 0x80)) {
        e1 = (b0 & 0x7f) | ((uint64_t)b1 << 7);
        e1len = 2;
      } else {
        e1 = 0;
        e1len = 0;
        ok = false;
      }
    }
    if (ok) {
      uint32_t value_size = (uint32_t)(e1 >> 2);
      uint64_t inc = (e1 & 2) << 7;
      uint32_t ns1, ns2, d2 = 0, hl;
      bool fast = false;
      if (e1 & 1) {  // frequent: <sp>
        uint32_t spb = (uint32_t)((w >> (8 * e1len)) & 0xff);
        if (!(spb & 0x80)) {
          ns1 = 1;
          ns2 = 1;
          hl = e1len + 1;
          fast = true;
          out->shared = spb;
        }
      } else {
        uint32_t e2 = (uint32_t)((w >> (8 * e1len)) & 0xff);
        if ((e2 & 3) == 1) {  // case 2.1.1: reuse, ns1<8, ns2<4, d1=0
          uint32_t spb = (uint32_t)((w >> (8 * (e1len + 1))) & 0xff);
          if (!(spb & 0x80)) {
            d2 = (e2 >> 2) & 1;
            ns1 = (e2 >> 3) & 7;
            ns2 = (e2 >> 6) & 3;
            hl = e1len + 2;
            fast = true;
            out->shared = spb;
          }
        }
      }
      if (fast) {
        uint32_t sp = out->shared;
        uint64_t prev_len = *key_len;
        uint64_t prev_ns2 = (uint64_t)ns2 - d2;
        uint64_t prev_except = (uint64_t)sp + ns1 + prev_ns2 + 8;
        uint64_t total = (uint64_t)hl + ns1 + ns2 + value_size;
        if (prev_len >= prev_except && (uint64_t)(limit - p) >= total) {
          uint64_t mid = prev_len - prev_except;
          uint64_t new_ns2_start = sp + ns1 + mid;
          uint64_t new_len = new_ns2_start + ns2 + 8;
          if (new_len <= kKeyCap) {
            // Row-change detection without LDS readback: the reference's
            // BlockBuilder emits the MAXIMAL shared prefix
            // (rocksdb/table/block_builder.cc delta encode), so a shared
            // prefix below the fixed rowkey length means the byte at
            // `shared_prefix` differs from the previous key — the row
            // changed; a shared prefix >= rkb means bytes [0, rkb) are
            // untouched. (The general decode path below keeps the
            // compare-on-write detection and covers restart entries.)
            if (sp < rkb) *changed = true;
            // register tail maintenance: d2 lengthens the key by one byte
            // (window slides by one); the written bytes are patched below
            const uint32_t new_ukey = (uint32_t)new_len - 8;
            if (d2) {
              *thi = (*thi << 8) | (*tlo >> 56);
              *tlo <<= 8;
            }
            // key bytes [hl, hl+ns1+ns2) — within the first 16 window bytes
            uint64_t w2 = rdr->peek8_at(8);
            for (uint32_t i = 0; i < ns1; ++i) {
              uint32_t j = hl + i;
              uint64_t src = j < 8 ? w : w2;
              uint8_t b = (uint8_t)(src >> (8 * (j & 7)));
              key[sp + i] = b;
              tail_patch(thi, tlo, new_ukey, sp + i, b);
            }
            for (uint32_t i = 0; i < ns2; ++i) {
              uint32_t j = hl + ns1 + i;
              uint64_t src = j < 8 ? w : w2;
              uint8_t b = (uint8_t)(src >> (8 * (j & 7)));
              key[new_ns2_start + i] = b;
              tail_patch(thi, tlo, new_ukey, new_ns2_start + i, b);
            }
            *last8 += inc;
            *key_len = (uint32_t)new_len;
            rdr->consume(hl + ns1 + ns2);
            out->value = rdr->pos();
            out->value_len = value_size;
            return out->value + value_size;
          }
        }
      }
    }
  }
  return nullptr;
}

// Full entry decode: fast core first, general fallback. On success the
// reader is positioned AT THE VALUE START (value not consumed).
DEV const uint8_t* decode_entry(int fmt, Rdr* rdr, const uint8_t* limit,
                                uint8_t* key, uint32_t* key_len,
                                uint64_t* last8, uint32_t rkb, bool* changed,
                                EntryRef* out, uint64_t* thi, uint64_t* tlo,
                                bool* tail_valid) {
  if (fmt == YBG_ENC_THREE_SHARED_PARTS) {
    const uint8_t* q = decode_entry_fast(rdr, limit, key, key_len, last8,
                                         rkb, changed, out, thi, tlo);
    if (q) return q;
  }
  // general path: the register tail no longer mirrors LDS
  const uint8_t* p = rdr->pos();
  const uint8_t* q =
      decode_entry_ptr(fmt, p, limit, key, key_len, last8, rkb, changed, out);
  if (!q) return nullptr;
  *tail_valid = false;
  rdr->seek(out->value);
  return q;
}

// ---------------------------------------------------------------------------
// DocKey length (doc_key.h:40-63); key lives in LDS
// ---------------------------------------------------------------------------

DEV uint32_t dockey_len(const DevSpec& sp, const uint8_t* p, uint32_t len) {
  uint32_t off = 0;
  int col = 0;
  if (sp.has_hash) {
    if (len < 3 || p[0] != kUInt16Hash) return 0;
    off = 3;
  }
  for (int group = 0; group < 2; ++group) {
    if (group == 0 && !sp.has_hash) continue;
    int ncols = group == 0 ? sp.num_hash_cols : sp.num_range_cols;
    for (int i = 0; i < ncols; ++i, ++col) {
      if (off >= len) return 0;
      uint8_t t = p[off];
      int kt = sp.key_types[col];
      if (kt == YBG_KT_INT64) {
        if (t != kInt64B || off + 9 > len) return 0;
        off += 9;
      } else if (kt == YBG_KT_INT32) {
        if (t != kInt32B || off + 5 > len) return 0;
        off += 5;
      } else {
        if (t != kStringB) return 0;
        uint32_t s = off + 1;
        for (;;) {
          if (s + 1 >= len) return 0;
          if (p[s] == 0) {
            if (p[s + 1] == 0) break;
            if (p[s + 1] != 1) return 0;
            s += 2;
          } else {
            ++s;
          }
        }
        off = s + 2;
      }
    }
    if (off >= len || p[off] != kGroupEnd) return 0;
    ++off;
  }
  return off;
}

DEV int skip_control(const uint8_t* v, uint32_t len) {
  uint32_t off = 0;
  if (off < len && v[off] == kMergeFlagsB) {
    ++off;
    uint64_t tmp;
    const uint8_t* q = uvarint(v + off, v + len, &tmp);
    if (!q) return -1;
    off = (uint32_t)(q - v);
  }
  if (off < len && v[off] == kHybridTimeByte) {
    ++off;
    int sz = dht_size_from_start(v + off, v + len);
    if (!sz) return -1;
    off += sz;
  }
  if (off < len && v[off] == kTtlB) {
    ++off;
    int64_t tmp;
    const uint8_t* q = svarint(v + off, v + len, &tmp);
    if (!q) return -1;
    off = (uint32_t)(q - v);
  }
  if (off < len && v[off] == kUserTsB) {
    off += 9;
    if (off > len) return -1;
  }
  return (int)off;
}


// ---------------------------------------------------------------------------
// Streaming row context (templated on the aggregate-slot capacity NA so the
// per-thread register state shrinks to what the query needs). Value-column
// predicates are evaluated EAGERLY at column decode into a pass bitmask —
// no operand values are retained; aggregate operands keep only a datum +
// null bit per slot.
// ---------------------------------------------------------------------------

// Row materialization target (next_batch / PgFetchNext path). All pointers
// device-global (host in the simulator).
struct EmitCtx {
  uint64_t* sort_key;    // [row_cap]
  uint64_t* key_datums;  // [row_cap * nk]
  uint64_t* datums;      // [row_cap * nc]; strings: (len<<40)|varlen offset
  uint32_t* null_masks;  // [row_cap]
  uint16_t* hashes;      // [row_cap] kUInt16Hash prefix (doc_key.h:54)
  uint8_t* varlen;
  uint64_t varlen_cap;
  unsigned long long* row_counter;
  unsigned long long* varlen_counter;
  uint64_t row_cap;
  int nk, nc;
  const uint32_t* head_consumed;  // [n_ivs] from the flags pre-pass
  unsigned long long* overflow;   // set nonzero when a cap was exceeded
};

template <int NA>
struct RowCtxT {
  bool base_seen;
  bool found;
  // The packed row's write time, needed ONLY when a later column-update
  // entry must be ordered against it (doc_reader.cc:1847-1869). Written
  // once per row but read on the cold update path only — kept in LDS (the
  // kernel passes a per-thread slot) instead of registers, where the
  // allocator was spilling it to HBM-backed scratch (~2.4 GB of write
  // traffic per 100M-row dispatch).
  uint64_t* bht;  // [3]: hi, lo, len
  int32_t cur_col;  // current column-update group (-2 liveness, -1 none)
  bool cur_col_done;
  uint32_t pred_pass;  // bit i: value-col predicate i passes (key preds at
                       // finalize)
  uint32_t agg_null;   // bit g: aggregate operand g is NULL
  uint64_t agg_datum[NA];
  // emit mode only (nullptr otherwise): per-thread row buffer in LDS
  uint64_t* emit_datums;  // [nc]: numeric datum or (global str address)
  uint32_t* emit_lens;    // [nc]: string length (numeric: 0)
  uint32_t emit_null;     // bit i: column i NULL
  uint32_t emit_str;      // bit i: column i holds a string address
  // GROUP BY operand (captured when sp.group_col matches)
  uint64_t grp_datum;     // numeric datum / global string address
  uint32_t grp_len;       // string length (0 numeric)
  bool grp_null;
};

template <int NA>
DEV void row_reset(RowCtxT<NA>* rc, const DevSpec& sp) {
  rc->base_seen = false;
  rc->found = false;
  rc->cur_col = -1;
  rc->cur_col_done = false;
  rc->pred_pass = 0;
  rc->agg_null = 0xffffffffu;
  rc->emit_null = 0xffffffffu;
  rc->emit_str = 0;
  rc->grp_null = true;
  rc->grp_len = 0;
}

// Value-column predicate compare (pgsql_operation.cc:602-668 typed-compare
// subset). sptr/slen only for string columns.
DEV bool pred_compare(const PredC& pr, uint64_t datum, const uint8_t* sptr,
                      uint32_t slen, const uint8_t* aux) {
  int dtype = (int)(pr.opdt >> 8);
  int cmp;
  if (YBG_UNLIKELY((pr.opdt & 0xff) == YBG_PRED_IN_RANGE)) {
    // option ranges (hybrid_scan_choices.h:43-77 OptionRange): n 24-byte
    // records; datum packs (count << 32) | aux offset
    uint32_t n = (uint32_t)(pr.datum >> 32);
    const uint8_t* rec = aux + (uint32_t)pr.datum;
    for (uint32_t i = 0; i < n; ++i, rec += 24) {
      uint64_t lo = load_u64_una(rec);
      uint64_t hi = load_u64_una(rec + 8);
      uint32_t fl = (uint32_t)load_u64_una(rec + 16);
      bool okl, okh;
      if (dtype == YBG_T_DOUBLE) {
        double a = __longlong_as_double((long long)datum);
        double l = __longlong_as_double((long long)lo);
        double h = __longlong_as_double((long long)hi);
        okl = (fl & 1) ? a >= l : a > l;
        okh = (fl & 2) ? a <= h : a < h;
      } else if (dtype == YBG_T_FLOAT) {
        float a = __uint_as_float((uint32_t)datum);
        float l = __uint_as_float((uint32_t)lo);
        float h = __uint_as_float((uint32_t)hi);
        okl = (fl & 1) ? a >= l : a > l;
        okh = (fl & 2) ? a <= h : a < h;
      } else {
        int64_t a = (int64_t)datum;
        okl = (fl & 1) ? a >= (int64_t)lo : a > (int64_t)lo;
        okh = (fl & 2) ? a <= (int64_t)hi : a < (int64_t)hi;
      }
      if (okl && okh) return true;
    }
    return false;
  }
  if (YBG_UNLIKELY((pr.opdt & 0xff) == YBG_PRED_IN)) {
    // membership over the option list (hybrid_scan_choices.h:43-60).
    // datum packs (count << 32) | aux offset. Numeric options are 8-byte
    // LE patterns; string options are [u32 LE length][bytes] records.
    uint32_t n = (uint32_t)(pr.datum >> 32);
    const uint8_t* lst = aux + (uint32_t)pr.datum;
    if (dtype == YBG_T_STRING) {
      for (uint32_t i = 0; i < n; ++i) {
        uint32_t ol = (uint32_t)load_u64_una(lst) & 0xffffffffu;
        const uint8_t* ob = lst + 4;
        lst += 4 + ol;
        if (ol != slen) continue;
        bool eq = true;
        uint32_t k = 0;
        for (; k + 8 <= ol; k += 8)
          if (load_u64_una(sptr + k) != load_u64_una(ob + k)) {
            eq = false;
            break;
          }
        if (eq)
          for (; k < ol; ++k)
            if (sptr[k] != ob[k]) { eq = false; break; }
        if (eq) return true;
      }
      return false;
    }
    for (uint32_t i = 0; i < n; ++i) {
      uint64_t rv = load_u64_una(lst + 8ull * i);
      if (dtype == YBG_T_DOUBLE) {
        if (__longlong_as_double((long long)datum) ==
            __longlong_as_double((long long)rv))
          return true;
      } else if (dtype == YBG_T_FLOAT) {
        if (__uint_as_float((uint32_t)datum) == __uint_as_float((uint32_t)rv))
          return true;
      } else {
        if (datum == rv) return true;
      }
    }
    return false;
  }
  if (YBG_LIKELY(dtype != YBG_T_STRING && dtype != YBG_T_DOUBLE &&
                 dtype != YBG_T_FLOAT)) {
    int64_t a = (int64_t)datum, b = (int64_t)pr.datum;
    cmp = a < b ? -1 : (a > b ? 1 : 0);
  } else if (dtype == YBG_T_STRING) {
    const uint8_t* rhs = aux + (uint32_t)pr.datum;
    uint32_t rlen = (uint32_t)(pr.datum >> 32);
    uint32_t n = slen < rlen ? slen : rlen;
    cmp = 0;
    // memcmp order == big-endian word order: compare 8 bytes per step
    // (byteswapped unaligned loads) instead of a serial byte loop; the
    // sub-8 tail stays a byte loop (no tail-slack assumption on aux)
    uint32_t k = 0;
    for (; k + 8 <= n; k += 8) {
      uint64_t a = __builtin_bswap64(load_u64_una(sptr + k));
      uint64_t b = __builtin_bswap64(load_u64_una(rhs + k));
      if (a != b) {
        cmp = a < b ? -1 : 1;
        break;
      }
    }
    if (cmp == 0)
      for (; k < n; ++k)
        if (sptr[k] != rhs[k]) {
          cmp = sptr[k] < rhs[k] ? -1 : 1;
          break;
        }
    if (cmp == 0 && slen != rlen) cmp = slen < rlen ? -1 : 1;
  } else if (dtype == YBG_T_DOUBLE) {
    double a = __longlong_as_double((long long)datum);
    double b = __longlong_as_double((long long)pr.datum);
    cmp = a < b ? -1 : (a > b ? 1 : 0);
  } else {
    float a = __uint_as_float((uint32_t)datum);
    float b = __uint_as_float((uint32_t)pr.datum);
    cmp = a < b ? -1 : (a > b ? 1 : 0);
  }
  switch (pr.opdt & 0xff) {
    case YBG_PRED_GT: return cmp > 0;
    case YBG_PRED_GE: return cmp >= 0;
    case YBG_PRED_LT: return cmp < 0;
    case YBG_PRED_LE: return cmp <= 0;
    case YBG_PRED_EQ: return cmp == 0;
    default: return cmp != 0;
  }
}

// Column value arrives (from packed row or column update): evaluate the
// predicates that reference it and stash aggregate operands.
template <int NA>
DEV void eval_col(const DevSpec& sp, const uint8_t* aux, RowCtxT<NA>* rc,
                  int col, bool is_null, uint64_t datum, const uint8_t* sptr,
                  uint32_t slen) {
  const uint32_t act = sp.col_act[col];
  if (YBG_UNLIKELY(act & kActGroup)) {
    rc->grp_null = is_null;
    rc->grp_datum = sptr ? (uint64_t)(uintptr_t)sptr : datum;
    rc->grp_len = sptr ? slen : 0;
  }
  if (rc->emit_datums) {
    rc->emit_null = (rc->emit_null & ~(1u << col)) | ((uint32_t)is_null << col);
    if (!is_null) {
      if (sptr) {
        rc->emit_datums[col] = (uint64_t)(uintptr_t)sptr;
        rc->emit_lens[col] = slen;
        rc->emit_str |= 1u << col;
      } else {
        rc->emit_datums[col] = datum;
        rc->emit_lens[col] = 0;
        rc->emit_str &= ~(1u << col);
      }
    }
  }
  // predicates on this column: iterate set bits only (the full
  // MAX_PREDS unroll kept every slot's fields live as uniform registers)
  uint32_t pm = act & kActPredM;
  while (pm) {
    int i = __builtin_ctz(pm);
    pm &= pm - 1;
    bool pass = !is_null && pred_compare(sp.predc[i], datum, sptr, slen, aux);
    rc->pred_pass = (rc->pred_pass & ~(1u << i)) | ((uint32_t)pass << i);
  }
  uint32_t am = (act >> kActAggShift) & kActAggM;
  if (am) {
#pragma unroll
    for (int g = 0; g < NA; ++g) {
      if (am & (1u << g)) {
        rc->agg_datum[g] = datum;
        rc->agg_null = (rc->agg_null & ~(1u << g)) | ((uint32_t)is_null << g);
      }
    }
  }
}

// Decode one V1-encoded single value (primitive_value.cc:1066-1125).
// Returns 0 null/tombstone, 1 applied, -1 error.
template <int NA>
DEV int decode_single_v1(const DevSpec& sp, const uint8_t* base,
                         const uint8_t* aux, int col, const uint8_t* vp,
                         uint32_t vlen, RowCtxT<NA>* rc) {
  if (vlen == 0) {
    eval_col(sp, aux, rc, col, true, 0, nullptr, 0);
    return 0;
  }
  uint8_t t = vp[0];
  if (t == kTombB || t == kNullLow) {
    eval_col(sp, aux, rc, col, true, 0, nullptr, 0);
    return 0;
  }
  const int dtype = (int)((sp.col_act[col] >> kActDtShift) & kActDtM);
  uint64_t datum = 0;
  const uint8_t* sptr = nullptr;
  uint32_t slen = 0;
  switch (dtype) {
    case YBG_T_BOOL:
      if (t != kTrueB && t != kFalseB) return -1;
      datum = (t == kTrueB);
      break;
    case YBG_T_INT8:
    case YBG_T_INT16:
    case YBG_T_INT32:
      if (t != kInt32B || vlen < 5) return -1;
      datum = (uint64_t)(int64_t)(int32_t)load_be32(vp + 1);
      break;
    case YBG_T_INT64:
      if (t != kInt64B || vlen < 9) return -1;
      datum = load_be64(vp + 1);
      break;
    case YBG_T_UINT64:
      if (vlen < 9) return -1;
      datum = load_be64(vp + 1);
      break;
    case YBG_T_FLOAT:
      if (t != kFloatB || vlen < 5) return -1;
      datum = load_be32(vp + 1);
      break;
    case YBG_T_UINT32:
      if (vlen < 5) return -1;
      datum = load_be32(vp + 1);
      break;
    case YBG_T_DOUBLE:
      if (t != kDoubleB || vlen < 9) return -1;
      datum = load_be64(vp + 1);
      break;
    case YBG_T_STRING:
      if (t != kStringB) return -1;
      sptr = vp + 1;
      slen = vlen - 1;
      break;
    default:
      return -1;
  }
  eval_col(sp, aux, rc, col, false, datum, sptr, slen);
  (void)base;
  return 1;
}

// Packed row decode streaming into predicate/aggregate slots.
// `body` points at 'z'/'|'.
template <int NA>
DEV bool decode_packed(const DevSpec& sp, const uint8_t* base,
                       const uint8_t* aux, const uint8_t* body, uint32_t len,
                       RowCtxT<NA>* rc) {
  uint8_t kind = body[0];
  uint32_t off = 1;
  uint64_t version;
  const uint8_t* q = uvarint(body + off, body + len, &version);
  if (!q) return false;
  off = (uint32_t)(q - body);

  if (kind == kPackedV1B) {
    const uint8_t* header = body + off;
    uint32_t prefix_len = (uint32_t)sp.v1_varlen_count * 4;
    if (off + prefix_len > len) return false;
    const uint8_t* data = header + prefix_len;
    uint32_t data_len = len - off - prefix_len;
    for (int i = 0; i < sp.num_value_cols; ++i) {
      const DevCol& c = sp.cols[i];
      uint32_t start = (uint32_t)c.v1_off;
      if (c.v1_nvb) start += load_le32_u(header + (c.v1_nvb - 1) * 4);
      uint32_t end;
      if (c.v1_varlen) {
        end = load_le32_u(header + c.v1_nvb * 4);
      } else {
        uint32_t fs;
        switch (c.dtype) {
          case YBG_T_BOOL: fs = 1; break;
          case YBG_T_INT8: case YBG_T_INT16: case YBG_T_INT32:
          case YBG_T_UINT32: case YBG_T_FLOAT: fs = 5; break;
          default: fs = 9; break;
        }
        end = start + fs;
      }
      if (end < start || end > data_len) return false;
      int r = decode_single_v1(sp, base, aux, i, data + start, end - start,
                               rc);
      if (r < 0) return false;
    }
    return true;
  }

  if (kind == kPackedV2B) {
    if (off >= len) return false;
    uint8_t flags = body[off++];
    const uint8_t* null_mask = nullptr;
    if (flags & 1) {  // kHasNullsFlag (packed_row.h:197)
      null_mask = body + off;
      off += (uint32_t)((sp.num_value_cols + 7) / 8);
      if (off > len) return false;
    }
    const uint8_t* data = body + off;
    const uint8_t* end = body + len;
    for (int i = 0; i < sp.num_value_cols; ++i) {
      if (null_mask && (null_mask[i >> 3] & (1 << (i & 7)))) {
        eval_col(sp, aux, rc, i, true, 0, nullptr, 0);
        continue;
      }
      const DevCol& c = sp.cols[i];
      if (c.v2_fixed) {
        if (data + c.v2_fixed > end) return false;
        uint64_t u = 0;
        switch (c.v2_fixed) {  // raw little-endian (value_packing_v2.cc:73-77)
          case 1: u = data[0]; break;
          case 2: { uint16_t x; __builtin_memcpy(&x, data, 2); u = x; break; }
          case 4: u = load_le32_u(data); break;
          default: u = load_le64_u(data); break;
        }
        if (c.dtype == YBG_T_INT8) u = (uint64_t)(int64_t)(int8_t)u;
        else if (c.dtype == YBG_T_INT16) u = (uint64_t)(int64_t)(int16_t)u;
        else if (c.dtype == YBG_T_INT32) u = (uint64_t)(int64_t)(int32_t)u;
        eval_col(sp, aux, rc, i, false, u, nullptr, 0);
        data += c.v2_fixed;
      } else {
        if (data >= end) return false;
        uint32_t flen, consumed;
        uint8_t b0 = *data;
        if ((b0 & 1) == 0) {  // field length (fast_varint.cc:373-384)
          flen = b0 >> 1;
          consumed = 1;
        } else {
          flen = load_le32_u(data) >> 1;
          consumed = 4;
        }
        data += consumed;
        if (data + flen > end) return false;
        eval_col(sp, aux, rc, i, false, 0, data, flen);
        data += flen;
      }
    }
    return true;
  }
  return false;
}

// key-column datum from saved rowkey bytes (global scratch)
DEV bool key_col_value(const DevSpec& sp, const uint8_t* rk, uint32_t rk_len,
                       int target, uint64_t* out_datum, uint32_t* out_soff,
                       uint32_t* out_slen) {
  uint32_t off = sp.has_hash ? 3u : 0u;
  int col = 0;
  for (int group = 0; group < 2; ++group) {
    if (group == 0 && !sp.has_hash) continue;
    int ncols = group == 0 ? sp.num_hash_cols : sp.num_range_cols;
    for (int k = 0; k < ncols; ++k, ++col) {
      int kt = sp.key_types[col];
      if (kt == YBG_KT_INT64) {
        if (col == target) {
          *out_datum = (uint64_t)(int64_t)(load_be64(rk + off + 1) ^
                                           0x8000000000000000ull);
          *out_slen = 0;
          return true;
        }
        off += 9;
      } else if (kt == YBG_KT_INT32) {
        if (col == target) {
          *out_datum = (uint64_t)(int64_t)(int32_t)(load_be32(rk + off + 1) ^
                                                    0x80000000u);
          *out_slen = 0;
          return true;
        }
        off += 5;
      } else {
        uint32_t s = off + 1;
        while (s + 1 < rk_len && !(rk[s] == 0 && rk[s + 1] == 0))
          s += (rk[s] == 0) ? 2 : 1;
        if (col == target) {
          *out_soff = off + 1;
          *out_slen = s - (off + 1);
          return true;
        }
        off = s + 2;
      }
    }
    ++off;  // group end
  }
  return false;
}

// Key-column predicates (evaluated at row finalize; the rowkey bytes come
// from the saved row key). Key strings are zero-escaped in the key
// (doc_kv_util.h:101-167): unescape on the fly.
DEV bool eval_key_preds(const DevSpec& sp, const uint8_t* rk, uint32_t rk_len,
                        const uint8_t* aux) {
  uint32_t kp = sp.key_pred_mask;
  while (kp) {
    int i = __builtin_ctz(kp);
    kp &= kp - 1;
    const DevPred& pr = sp.preds[i];
    if (YBG_UNLIKELY(pr.op == YBG_PRED_IN_TUPLE)) {
      // multi-column option group (hybrid_scan_choices.h:43-77):
      // [u32 ncols][u32 colidx x n][tuples of n x u64]
      const uint8_t* q = aux + pr.rhs_off;
      uint32_t nc = (uint32_t)load_u64_una(q) & 0xffffffffu;
      const uint8_t* cols = q + 4;
      const uint8_t* tup = cols + 4ull * nc;
      uint32_t tup_sz = 8 * nc;
      uint32_t ntup =
          tup_sz ? (uint32_t)((pr.str_len - 4 - 4ull * nc) / tup_sz) : 0;
      bool hit = false;
      for (uint32_t t = 0; t < ntup && !hit; ++t) {
        bool all = true;
        for (uint32_t c = 0; c < nc && all; ++c) {
          uint32_t ci = (uint32_t)load_u64_una(cols + 4ull * c) & 0xffffffffu;
          uint64_t dv = 0;
          uint32_t so = 0, sn = 0;
          if (!key_col_value(sp, rk, rk_len, (int)ci, &dv, &so, &sn))
            return false;
          if (dv != load_u64_una(tup + (uint64_t)t * tup_sz + 8ull * c))
            all = false;
        }
        hit = all;
      }
      if (!hit) return false;
      continue;
    }
    uint64_t d = 0;
    uint32_t soff = 0, sl = 0;
    if (!key_col_value(sp, rk, rk_len, pr.col, &d, &soff, &sl)) return false;
    int cmp;
    if (sp.key_types[pr.col] == YBG_KT_STRING && pr.op == YBG_PRED_IN) {
      // option membership over [u32 len][bytes] records; the key bytes
      // are zero-escaped in the rowkey (doc_kv_util.h:101-167) so each
      // candidate compare unescapes on the fly like the EQ path
      const uint8_t* lst = aux + pr.rhs_off;
      const uint8_t* lend = lst + pr.str_len;
      bool hit = false;
      while (lst + 4 <= lend && !hit) {
        uint32_t ol = (uint32_t)load_u64_una(lst) & 0xffffffffu;
        const uint8_t* ob = lst + 4;
        lst += 4 + ol;
        const uint8_t* lstr = rk + soff;
        uint32_t si = 0, ri = 0;
        bool eq = true;
        while (si < sl && ri < ol) {
          uint8_t cb = lstr[si];
          si += (cb == 0) ? 2 : 1;
          if (cb != ob[ri]) { eq = false; break; }
          ++ri;
        }
        if (eq && si >= sl && ri >= ol) hit = true;
      }
      if (!hit) return false;
      continue;
    }
    if (sp.key_types[pr.col] == YBG_KT_STRING) {
      const uint8_t* lstr = rk + soff;
      const uint8_t* rhs = aux + pr.rhs_off;
      uint32_t si = 0, ri = 0;
      cmp = 0;
      while (si < sl && ri < pr.str_len) {
        uint8_t cb = lstr[si];
        si += (cb == 0) ? 2 : 1;
        if (cb != rhs[ri]) {
          cmp = cb < rhs[ri] ? -1 : 1;
          break;
        }
        ++ri;
      }
      if (cmp == 0) {
        bool le = si >= sl, re = ri >= pr.str_len;
        cmp = (le && re) ? 0 : (le ? -1 : 1);
      }
    } else if (pr.op == YBG_PRED_IN) {
      uint32_t n = pr.str_len / 8;
      bool hit = false;
      for (uint32_t k2 = 0; k2 < n; ++k2)
        if (d == load_u64_una(aux + pr.rhs_off + 8ull * k2)) hit = true;
      if (!hit) return false;
      continue;
    } else if (pr.op == YBG_PRED_IN_RANGE) {
      // option ranges on a numeric key column
      uint32_t n = pr.str_len / 24;
      bool hit = false;
      int64_t a = (int64_t)d;
      for (uint32_t k2 = 0; k2 < n && !hit; ++k2) {
        const uint8_t* rec = aux + pr.rhs_off + 24ull * k2;
        int64_t lo = (int64_t)load_u64_una(rec);
        int64_t hi2 = (int64_t)load_u64_una(rec + 8);
        uint32_t fl = (uint32_t)load_u64_una(rec + 16);
        bool okl = (fl & 1) ? a >= lo : a > lo;
        bool okh = (fl & 2) ? a <= hi2 : a < hi2;
        hit = okl && okh;
      }
      if (!hit) return false;
      continue;
    } else {
      int64_t a = (int64_t)d, b = (int64_t)pr.datum;
      cmp = a < b ? -1 : (a > b ? 1 : 0);
    }
    bool pass;
    switch (pr.op) {
      case YBG_PRED_GT: pass = cmp > 0; break;
      case YBG_PRED_GE: pass = cmp >= 0; break;
      case YBG_PRED_LT: pass = cmp < 0; break;
      case YBG_PRED_LE: pass = cmp <= 0; break;
      case YBG_PRED_EQ: pass = cmp == 0; break;
      default: pass = cmp != 0; break;
    }
    if (!pass) return false;
  }
  return true;
}

DEV bool in_bounds(const DevSpec& sp, const uint8_t* rk, uint32_t rk_len,
                   const uint8_t* aux) {
  if (sp.lower_len) {
    const uint8_t* b = aux + sp.lower_off;
    uint32_t n = rk_len < sp.lower_len ? rk_len : sp.lower_len;
    int cmp = 0;
    for (uint32_t i = 0; i < n; ++i)
      if (rk[i] != b[i]) { cmp = rk[i] < b[i] ? -1 : 1; break; }
    if (cmp == 0 && rk_len != sp.lower_len)
      cmp = rk_len < sp.lower_len ? -1 : 1;
    if (cmp < 0) return false;
  }
  if (sp.upper_len) {
    const uint8_t* b = aux + sp.upper_off;
    uint32_t n = rk_len < sp.upper_len ? rk_len : sp.upper_len;
    int cmp = 0;
    for (uint32_t i = 0; i < n; ++i)
      if (rk[i] != b[i]) { cmp = rk[i] < b[i] ? -1 : 1; break; }
    if (cmp == 0 && rk_len != sp.upper_len)
      cmp = rk_len < sp.upper_len ? -1 : 1;
    if (cmp >= 0) return false;
  }
  return true;
}

// Aggregate accumulate (doc_expr.cc:248-395); cnt counts non-null
// contributions (SUM/MIN/MAX null-ness = cnt == 0).
template <int NA>
DEV void acc_row(const DevSpec& sp, const RowCtxT<NA>& rc, uint64_t* agg_val,
                 uint64_t* agg_cnt) {
#pragma unroll
  for (int g = 0; g < NA; ++g) {
    if (g >= sp.num_aggs) continue;
    const int op = sp.agg_op[g];
    bool isnull =
        (op == YBG_AGG_COUNT_STAR) ? false : ((rc.agg_null >> g) & 1);
    if (isnull) continue;
    uint64_t v = rc.agg_datum[g];
    switch (op) {
      case YBG_AGG_COUNT_STAR:
      case YBG_AGG_COUNT:
        agg_val[g] += 1;
        break;
      case YBG_AGG_SUM_INT64:
        agg_val[g] = (uint64_t)((int64_t)agg_val[g] + (int64_t)v);
        break;
      case YBG_AGG_SUM_DOUBLE: {
        double cur = __longlong_as_double((long long)agg_val[g]) +
                     __longlong_as_double((long long)v);
        agg_val[g] = (uint64_t)__double_as_longlong(cur);
        break;
      }
      case YBG_AGG_MIN_INT64:
        if (agg_cnt[g] == 0 || (int64_t)v < (int64_t)agg_val[g]) agg_val[g] = v;
        break;
      case YBG_AGG_MAX_INT64:
        if (agg_cnt[g] == 0 || (int64_t)v > (int64_t)agg_val[g]) agg_val[g] = v;
        break;
      case YBG_AGG_MIN_DOUBLE: {
        double dd = __longlong_as_double((long long)v);
        if (agg_cnt[g] == 0 || dd < __longlong_as_double((long long)agg_val[g]))
          agg_val[g] = (uint64_t)__double_as_longlong(dd);
        break;
      }
      case YBG_AGG_MAX_DOUBLE: {
        double dd = __longlong_as_double((long long)v);
        if (agg_cnt[g] == 0 || dd > __longlong_as_double((long long)agg_val[g]))
          agg_val[g] = (uint64_t)__double_as_longlong(dd);
        break;
      }
    }
    agg_cnt[g] += 1;
  }
}

// Combine one partial (val,cnt) pair into an accumulator. Fold order must be
// kept FIXED by callers so double SUM is deterministic.
DEV void combine1(int op, uint64_t* av, uint64_t* ac, uint64_t bv,
                  uint64_t bc) {
  if (bc == 0) return;
  switch (op) {
    case YBG_AGG_COUNT_STAR:
    case YBG_AGG_COUNT:
    case YBG_AGG_SUM_INT64:
      *av = (uint64_t)((int64_t)*av + (int64_t)bv);
      break;
    case YBG_AGG_SUM_DOUBLE: {
      double cur = (*ac ? __longlong_as_double((long long)*av) : 0.0) +
                   __longlong_as_double((long long)bv);
      *av = (uint64_t)__double_as_longlong(cur);
      break;
    }
    case YBG_AGG_MIN_INT64:
      if (*ac == 0 || (int64_t)bv < (int64_t)*av) *av = bv;
      break;
    case YBG_AGG_MAX_INT64:
      if (*ac == 0 || (int64_t)bv > (int64_t)*av) *av = bv;
      break;
    case YBG_AGG_MIN_DOUBLE:
      if (*ac == 0 || __longlong_as_double((long long)bv) <
                          __longlong_as_double((long long)*av))
        *av = bv;
      break;
    case YBG_AGG_MAX_DOUBLE:
      if (*ac == 0 || __longlong_as_double((long long)bv) >
                          __longlong_as_double((long long)*av))
        *av = bv;
      break;
  }
  *ac += bc;
}

// Combine full partial arrays in fixed order.
DEV void agg_combine(const DevSpec& sp, uint64_t* a_val, uint64_t* a_cnt,
                     const uint64_t* b_val, const uint64_t* b_cnt) {
#pragma unroll
  for (int g = 0; g < YBG_MAX_AGGS; ++g) {
    if (g >= sp.num_aggs) continue;
    combine1(sp.agg_op[g], &a_val[g], &a_cnt[g], b_val[g], b_cnt[g]);
  }
}

// Rdr-based packed-row V2 decode fast path: schema version < 128, no null
// mask (the dominant shape); anything else falls back to the pointer
// decoder. Assumes rdr is positioned at the 'kPackedRowV2' byte; consumes
// the whole body on success.
template <int NA>
DEV bool decode_packed_v2_rdr(const DevSpec& sp, const uint8_t* base,
                              const uint8_t* aux, Rdr* rdr, uint32_t len,
                              RowCtxT<NA>* rc, bool* done) {
  uint64_t w = rdr->peek8();
  uint32_t ver = (uint32_t)((w >> 8) & 0xff);
  uint32_t flags = (uint32_t)((w >> 16) & 0xff);
  if ((ver & 0x80) || flags != 0) {
    *done = false;  // caller falls back
    return true;
  }
  *done = true;
  const uint8_t* end = rdr->pos() + len;
  rdr->consume(3);
  for (int i = 0; i < sp.num_value_cols; ++i) {
    const uint32_t act = sp.col_act[i];
    const uint32_t fw = (act >> kActV2Shift) & kActV2M;
    if (fw) {
      if (YBG_UNLIKELY(rdr->pos() + fw > end)) return false;
      uint64_t u = rdr->peek8();
      switch (fw) {
        case 1: u &= 0xff; break;
        case 2: u &= 0xffff; break;
        case 4: u &= 0xffffffffull; break;
        default: break;
      }
      const uint32_t dt = (act >> kActDtShift) & kActDtM;
      if (dt == YBG_T_INT8) u = (uint64_t)(int64_t)(int8_t)u;
      else if (dt == YBG_T_INT16) u = (uint64_t)(int64_t)(int16_t)u;
      else if (dt == YBG_T_INT32) u = (uint64_t)(int64_t)(int32_t)u;
      if ((act & kActLiveM) || rc->emit_datums)
        eval_col(sp, aux, rc, i, false, u, nullptr, 0);
      rdr->consume(fw);
    } else {
      if (rdr->pos() >= end) return false;
      uint64_t w0 = rdr->peek8();
      uint32_t flen;
      if ((w0 & 1) == 0) {
        flen = (uint32_t)((w0 & 0xff) >> 1);
        rdr->consume(1);
      } else {
        flen = (uint32_t)((w0 & 0xffffffffull) >> 1);
        rdr->consume(4);
      }
      if (rdr->pos() + flen > end) return false;
      if ((act & kActLiveM) || rc->emit_datums)
        eval_col(sp, aux, rc, i, false, 0, rdr->pos(), flen);
      rdr->skip(flen);
    }
  }
  (void)base;
  return true;
}

// Fixed-offset packed-V2 decode: when the schema is all-fixed-width and the
// row carries no null mask (flags == 0, single-byte schema version), every
// column body sits at a host-precomputed offset from the value start
// (schema_packing.cc:1076-1121 layout with no varlen entries). The loads
// are INDEPENDENT global dword loads on lines the reader window just
// touched (L1-resident) — no serial window-consume chain.
template <int NA>
DEV void decode_packed_v2_fixed(const DevSpec& sp, const uint8_t* aux,
                                const uint8_t* value, RowCtxT<NA>* rc) {
  // one shared alignment for the whole body: every column extract is a
  // funnel shift over two aligned words (the words are shared between
  // adjacent columns instead of re-deriving per-column unaligned loads)
  // software-pipelined one deep: column i+1's word pair issues BEFORE
  // column i's eval, so the (uniform-trip) loop hides each L1 latency
  // behind the previous column's eval work instead of serializing on it
  const uintptr_t a = (uintptr_t)value;
  const uint64_t* qw = (const uint64_t*)(a & ~(uintptr_t)7);
  const uint32_t abase = (uint32_t)(a & 7);
  const int n = sp.num_value_cols;
  uint32_t ob = abase + sp.v2_off[0];
  uint64_t w0 = qw[ob >> 3], w1 = qw[(ob >> 3) + 1];
  for (int i = 0; i < n; ++i) {
    const uint32_t act = sp.col_act[i];
    const uint32_t sh = (ob & 7) * 8;
    uint64_t u = sh ? (w0 >> sh) | (w1 << (64 - sh)) : w0;
    if (i + 1 < n) {
      ob = abase + sp.v2_off[i + 1];
      w0 = qw[ob >> 3];
      w1 = qw[(ob >> 3) + 1];
    }
    const uint32_t dt = (act >> kActDtShift) & kActDtM;
    switch ((act >> kActV2Shift) & kActV2M) {
      case 1:
        u = (dt == YBG_T_INT8) ? (uint64_t)(int64_t)(int8_t)u : (u & 0xff);
        break;
      case 2:
        u = (dt == YBG_T_INT16) ? (uint64_t)(int64_t)(int16_t)u
                                : (u & 0xffff);
        break;
      case 4:
        u = (dt == YBG_T_INT32) ? (uint64_t)(int64_t)(int32_t)u
                                : (u & 0xffffffffull);
        break;
      default:
        break;
    }
    if ((act & kActLiveM) || rc->emit_datums)
      eval_col(sp, aux, rc, i, false, u, nullptr, 0);
  }
}

// Direct-load packed-V2 decode for schemas whose leading v2_nfp columns
// are fixed-width: the prefix decodes from static offsets with the same
// software-pipelined word-pair loads as decode_packed_v2_fixed, and the
// varlen tail walks with direct unaligned loads — no window-consume
// chain at all (the serial chain is what bounds wide mixed schemas).
// Caller guarantees ver < 128, flags == 0, v2_nfp > 0; value points at
// the kPackedV2B byte. Returns false on corruption.
template <int NA>
DEV bool decode_packed_v2_mixed(const DevSpec& sp, const uint8_t* aux,
                                const uint8_t* value, uint32_t value_len,
                                RowCtxT<NA>* rc) {
  const uint8_t* end = value + value_len;
  const int nfp = sp.v2_nfp, n = sp.num_value_cols;
  if (YBG_UNLIKELY(value + sp.v2_tail_off > end)) return false;
  const uintptr_t a = (uintptr_t)value;
  const uint64_t* qw = (const uint64_t*)(a & ~(uintptr_t)7);
  const uint32_t abase = (uint32_t)(a & 7);
  uint32_t ob = abase + sp.v2_off[0];
  uint64_t w0 = qw[ob >> 3], w1 = qw[(ob >> 3) + 1];
  for (int i = 0; i < nfp; ++i) {
    const uint32_t act = sp.col_act[i];
    const uint32_t sh = (ob & 7) * 8;
    uint64_t u = sh ? (w0 >> sh) | (w1 << (64 - sh)) : w0;
    if (i + 1 < nfp) {
      ob = abase + sp.v2_off[i + 1];
      w0 = qw[ob >> 3];
      w1 = qw[(ob >> 3) + 1];
    }
    const uint32_t dt = (act >> kActDtShift) & kActDtM;
    switch ((act >> kActV2Shift) & kActV2M) {
      case 1:
        u = (dt == YBG_T_INT8) ? (uint64_t)(int64_t)(int8_t)u : (u & 0xff);
        break;
      case 2:
        u = (dt == YBG_T_INT16) ? (uint64_t)(int64_t)(int16_t)u
                                : (u & 0xffff);
        break;
      case 4:
        u = (dt == YBG_T_INT32) ? (uint64_t)(int64_t)(int32_t)u
                                : (u & 0xffffffffull);
        break;
      default:
        break;
    }
    if ((act & kActLiveM) || rc->emit_datums)
      eval_col(sp, aux, rc, i, false, u, nullptr, 0);
  }
  const uint8_t* p = value + sp.v2_tail_off;
  for (int i = nfp; i < n; ++i) {
    const uint32_t act = sp.col_act[i];
    const uint32_t fw = (act >> kActV2Shift) & kActV2M;
    uint64_t u;
    memcpy(&u, p, 8);  // unaligned; block tail slack covers the overread
    if (fw) {
      if (YBG_UNLIKELY(p + fw > end)) return false;
      switch (fw) {
        case 1: u &= 0xff; break;
        case 2: u &= 0xffff; break;
        case 4: u &= 0xffffffffull; break;
        default: break;
      }
      const uint32_t dt = (act >> kActDtShift) & kActDtM;
      if (dt == YBG_T_INT8) u = (uint64_t)(int64_t)(int8_t)u;
      else if (dt == YBG_T_INT16) u = (uint64_t)(int64_t)(int16_t)u;
      else if (dt == YBG_T_INT32) u = (uint64_t)(int64_t)(int32_t)u;
      if ((act & kActLiveM) || rc->emit_datums)
        eval_col(sp, aux, rc, i, false, u, nullptr, 0);
      p += fw;
    } else {
      if (YBG_UNLIKELY(p >= end)) return false;
      uint32_t flen;
      if ((u & 1) == 0) {
        flen = (uint32_t)((u & 0xff) >> 1);
        p += 1;
      } else {
        flen = (uint32_t)((u & 0xffffffffull) >> 1);
        p += 4;
      }
      if (YBG_UNLIKELY(p + flen > end)) return false;
      if ((act & kActLiveM) || rc->emit_datums)
        eval_col(sp, aux, rc, i, false, 0, p, flen);
      p += flen;
    }
  }
  return true;
}

// Visibility + row-state update for one entry. key/rowkey live in LDS.
// rdr is positioned at the value start. Returns false on corruption.
template <int NA>
DEV bool process_entry(const DevSpec& sp, const uint8_t* base,
                       const uint8_t* aux, const uint8_t* key,
                       uint32_t key_len, const uint8_t* value,
                       uint32_t value_len, uint32_t rowkey_len,
                       RowCtxT<NA>* rc, Rdr* rdr, uint32_t ht_size,
                       uint64_t ht_hi, uint64_t ht_lo) {
  uint32_t ukey_len = key_len - 8;
  uint32_t prefix_len = ukey_len - ht_size - 1;
  uint32_t vb0 = value_len > 0 ? (uint32_t)(rdr->peek8() & 0xff) : 0u;
  bool visible;
  if (value_len > 0 && vb0 == kHybridTimeByte) {
    // committed-txn record with intent time
    // (intent_aware_iterator.cc:1249-1267)
    uint64_t v_hi, v_lo;
    slice_u128(value + 1, value_len - 1, &v_hi, &v_lo);
    bool use_global =
        u128_slice_cmp(v_hi, v_lo, value_len - 1, sp.local_lim.hi,
                       sp.local_lim.lo, sp.local_lim.len) > 0;
    const HtLim& lim = use_global ? sp.global_lim : sp.read;
    visible =
        u128_slice_cmp(ht_hi, ht_lo, ht_size, lim.hi, lim.lo, lim.len) >= 0;
    if (visible) {
      int iht = dht_size_from_start(value + 1, value + value_len);
      if (!iht) return false;
      value += 1 + iht;
      value_len -= 1 + iht;
    }
  } else {
    visible = u128_slice_cmp(ht_hi, ht_lo, ht_size, sp.reg_lim.hi,
                             sp.reg_lim.lo, sp.reg_lim.len) >= 0;
  }
  if (!visible) return true;
  if (YBG_UNLIKELY(sp.track_restart)) {
    // visible record with commit > read (encoded bytes BELOW enc(read)):
    // a restart candidate — keep the MIN encoded (max commit time). The
    // slot lives beside bht in LDS (rc->bht[3..5]).
    if (u128_slice_cmp(ht_hi, ht_lo, ht_size, sp.read.hi, sp.read.lo,
                       sp.read.len) < 0) {
      uint64_t* rr = rc->bht + 3;
      if (rr[2] == 0 || u128_slice_cmp(ht_hi, ht_lo, ht_size, rr[0], rr[1],
                                       (uint32_t)rr[2]) < 0) {
        rr[0] = ht_hi;
        rr[1] = ht_lo;
        rr[2] = ht_size;
      }
    }
  }

  if (rowkey_len == prefix_len) {
    if (!rc->base_seen) {
      rc->base_seen = true;
      rc->bht[0] = ht_hi;
      rc->bht[1] = ht_lo;
      rc->bht[2] = ht_size;
      if (value_len > 0 && vb0 == kPackedV2B && value == rdr->pos()) {
        if (sp.v2_fixed_len && value_len == sp.v2_fixed_len &&
            (rdr->peek8() & 0xff8000u) == 0) {
          // all-fixed schema, no null mask, 1-byte version: direct
          // fixed-offset column loads, no window-consume chain. Reposition
          // the reader FIRST: the next entry's window loads issue here and
          // land while the columns below decode (they are otherwise the
          // first thing the next iteration stalls on).
          rdr->seek(value + value_len);
          decode_packed_v2_fixed(sp, aux, value, rc);
          rc->found = true;
          return true;
        }
        if (sp.v2_nfp > 0 && (rdr->peek8() & 0xff8000u) == 0) {
          // mixed fixed/varlen schema, no null mask: direct loads for the
          // static-offset prefix + unaligned walk for the varlen tail
          rdr->seek(value + value_len);
          if (!decode_packed_v2_mixed(sp, aux, value, value_len, rc))
            return false;
          rc->found = true;
          return true;
        }
        // remaining shapes (leading varlen column) — register-window
        // decode; falls back when a null mask / big schema version appears
        bool done;
        if (!decode_packed_v2_rdr(sp, base, aux, rdr, value_len, rc, &done))
          return false;
        if (done) {
          rc->found = true;
          return true;
        }
      }
      int cf = skip_control(value, value_len);
      if (cf < 0) return false;
      const uint8_t* body = value + cf;
      uint32_t body_len = value_len - cf;
      if (body_len > 0 && (body[0] == kPackedV1B || body[0] == kPackedV2B)) {
        if (!decode_packed(sp, base, aux, body, body_len, rc)) return false;
        rc->found = true;  // doc_reader.cc:1894-1900
      }
    }
  } else {
    const uint8_t* sk = key + rowkey_len;
    uint32_t sk_len = prefix_len - rowkey_len;
    if (sk_len < 2) return false;
    if (sk[0] == kSysColB) {
      if (rc->cur_col != -2) {
        rc->cur_col = -2;
        rc->cur_col_done = false;
      }
      if (!rc->cur_col_done) {
        rc->cur_col_done = true;
        bool newer = !rc->base_seen ||
                     u128_slice_cmp(ht_hi, ht_lo, ht_size, rc->bht[0],
                                    rc->bht[1], (uint32_t)rc->bht[2]) < 0;
        if (newer) {
          int cf = skip_control(value, value_len);
          if (cf < 0) return false;
          if (value_len - (uint32_t)cf > 0 && value[cf] != kTombB)
            rc->found = true;
        }
      }
    } else if (sk[0] == kColB) {
      int64_t col_id;
      const uint8_t* q = svarint(sk + 1, sk + sk_len, &col_id);
      if (!q || q != sk + sk_len) return false;
      int idx = -1;
#pragma clang loop unroll(disable)
      for (int i = 0; i < sp.num_value_cols; ++i)
        if (sp.col_ids[i] == (int32_t)col_id) { idx = i; break; }
      if (idx < 0) return true;
      if (rc->cur_col != idx) {
        rc->cur_col = idx;
        rc->cur_col_done = false;
      }
      if (!rc->cur_col_done) {
        rc->cur_col_done = true;
        bool newer = !rc->base_seen ||
                     u128_slice_cmp(ht_hi, ht_lo, ht_size, rc->bht[0],
                                    rc->bht[1], (uint32_t)rc->bht[2]) < 0;
        if (newer) {
          int cf = skip_control(value, value_len);
          if (cf < 0) return false;
          int r = decode_single_v1(sp, base, aux, idx, value + cf,
                                   value_len - cf, rc);
          if (r < 0) return false;
          if (r > 0) rc->found = true;
        }
      }
    } else {
      return false;
    }
  }
  return true;
}

// ---------------------------------------------------------------------------
// One interval, the whole algorithm (head deferral + tail walk). Called by
// the k_scan kernel (one thread per interval) and by the host simulator.
// ---------------------------------------------------------------------------
template <int NA>
struct HeadOut {
  uint64_t val[NA];
  uint64_t cnt[NA];
  uint32_t scanned, matched;
};

// ---------------------------------------------------------------------------
// GROUP BY partial aggregates (config #5). Open-addressing hash table in
// device memory: state[] 0=empty / 2=claiming / 1=ready, gkey[] the group
// key (numeric datum, or for string group columns (len<<40)|offset into the
// immutable block data — exemplar bytes are compared, never copied). Probe
// loops always complete an iteration before retrying a claiming slot so
// divergent lanes reconverge (no intra-wave spin deadlock). Integer
// aggregate updates are atomic adds/min/max (exact, order-independent);
// grouped double SUM uses atomic f64 adds (order nondeterministic,
// documented).
// ---------------------------------------------------------------------------
struct GroupCtx {
  unsigned long long* gkey;  // [cap + 1]; slot cap = the NULL-key group
  unsigned* state;           // [cap + 1]
  long long* vals;           // [(cap + 1) * YBG_MAX_AGGS]
  unsigned long long* cnts;  // [(cap + 1) * YBG_MAX_AGGS]
  unsigned long long* vals_hi;  // [(cap+1)*MAX_AGGS] SUM_DOUBLE high word
  unsigned* poison;          // [(cap+1)*MAX_AGGS] non-finite/overflow flag
  uint64_t cap;              // power of two
  unsigned long long* overflow;
  const uint8_t* data;       // block data base (string exemplars)
};

DEV uint64_t grp_hash_bytes(const uint8_t* p, uint32_t len) {
  uint64_t h = 1469598103934665603ull;
  for (uint32_t i = 0; i < len; ++i) {
    h ^= p[i];
    h *= 1099511628211ull;
  }
  return h;
}

DEV uint64_t grp_mix(uint64_t x) {
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdull;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ull;
  x ^= x >> 33;
  return x;
}

// Returns slot index or ~0ull on table overflow. kv: numeric datum, or for
// strings (len<<40)|(address - data).
DEV uint64_t grp_find_or_insert(const GroupCtx& gc, uint64_t kv,
                                bool is_str) {
  uint64_t h;
  const uint8_t* sp1 = nullptr;
  uint32_t sl = 0;
  if (is_str) {
    sl = (uint32_t)(kv >> 40);
    sp1 = gc.data + (kv & ((1ull << 40) - 1));
    h = grp_hash_bytes(sp1, sl);
  } else {
    h = grp_mix(kv);
  }
  uint64_t mask = gc.cap - 1;
  uint64_t probe = 0;
  for (uint64_t iters = 0; probe < gc.cap && iters < gc.cap * 64; ++iters) {
    uint64_t i = (h + probe) & mask;
    unsigned st = ybg_atomic_load_u32(&gc.state[i]);
    if (st == 0) {
      st = atomicCAS(&gc.state[i], 0u, 2u);
      if (st == 0) {
        ybg_atomic_exch_u64(&gc.gkey[i], kv);
        ybg_atomic_store_rel_u32(&gc.state[i], 1u);
        return i;
      }
    }
    if (st == 1) {
      // no acquire reload: gkey is read with a device-scope atomic load
      // (coherence-point direct) and the branch on st==1 orders it after
      // the state observation; the per-row L1 invalidate an acquire load
      // implies costs more than the whole probe
      uint64_t k2 = ybg_atomic_load_u64(&gc.gkey[i]);
      bool match;
      if (is_str) {
        uint32_t l2 = (uint32_t)(k2 >> 40);
        match = (l2 == sl);
        if (match) {
          const uint8_t* p2 = gc.data + (k2 & ((1ull << 40) - 1));
          for (uint32_t b = 0; b < sl; ++b)
            if (p2[b] != sp1[b]) { match = false; break; }
        }
      } else {
        match = (k2 == kv);
      }
      if (match) return i;
      ++probe;
    }
    // st == 2: another lane is publishing this slot; retry same probe
  }
  return ~0ull;
}

// Order-isomorphic u64 image of a double (sign-magnitude -> biased):
// unsigned compares/min/max on the image match double ordering, so
// grouped double MIN/MAX run as plain integer atomics — exact and
// order-independent (deterministic). NaNs map above +inf.
DEV uint64_t f64_ordered(uint64_t bits) {
  return bits ^ ((bits >> 63) ? ~0ull : (1ull << 63));
}
DEV uint64_t f64_unordered(uint64_t u) {
  return u ^ ((u >> 63) ? (1ull << 63) : ~0ull);
}

// Grouped double SUM accumulates in 128-bit two's-complement FIXED POINT
// (2^-60 scaling) with carry-propagating integer atomics: bit-exact and
// order-independent => deterministic run to run and across GPU counts,
// unlike a float atomic-add. Values outside +-2^66 or non-finite poison
// the slot (exported as NaN). doc_expr.cc:341-395 semantics otherwise.
DEV void group_sum_f64_fixed(const GroupCtx& gc, uint64_t slot, int g,
                             uint64_t dbits) {
  double v = __longlong_as_double((long long)dbits);
  if (!(v == v) || v > 7.3786976294838206e19 ||
      v < -7.3786976294838206e19) {
    ybg_atomic_or_u32(&gc.poison[slot * YBG_MAX_AGGS + g], 1u);
    return;
  }
  double sc = v * 1152921504606846976.0;  // 2^60
  __int128 f = (__int128)sc;
  unsigned long long lo = (unsigned long long)(unsigned __int128)f;
  unsigned long long hi =
      (unsigned long long)((unsigned __int128)f >> 64);
  unsigned long long* vlo =
      (unsigned long long*)&gc.vals[slot * YBG_MAX_AGGS + g];
  unsigned long long* vhi = &gc.vals_hi[slot * YBG_MAX_AGGS + g];
  unsigned long long old = atomicAdd(vlo, lo);
  if (old + lo < old) atomicAdd(vhi, 1ull);
  atomicAdd(vhi, hi);
}

// Final per-slot value decode (shared by the device export kernel and the
// host simulator's export).
DEV long long group_export_value(int op, long long raw,
                                 unsigned long long hi, unsigned poison) {
  if (op == YBG_AGG_SUM_DOUBLE) {
    double dv;
    if (poison) {
      dv = __longlong_as_double(0x7ff8000000000000ll);  // NaN
    } else {
      unsigned __int128 u = ((unsigned __int128)hi << 64) |
                            (unsigned long long)raw;
      __int128 sf = (__int128)u;
      dv = (double)sf * 8.673617379884035e-19;  // 2^-60
    }
    return (long long)__double_as_longlong(dv);
  }
  if (op == YBG_AGG_MIN_DOUBLE || op == YBG_AGG_MAX_DOUBLE)
    return (long long)f64_unordered((uint64_t)raw);
  return raw;
}

// Deferred head-row record for the single-pass GROUP kernel: the head
// row's raw operands (ownership unknown until the lane relay / cont-flag
// resolution). agg_datum is fixed-size for a stable memory layout; the
// unrolled NA loops keep register copies constant-indexed.
struct GroupHead {
  uint64_t grp_datum;
  uint64_t agg_datum[YBG_MAX_AGGS];
  uint32_t grp_len;
  uint32_t agg_null;
  uint32_t grp_null;
  uint32_t hit;
};

template <int NA>
DEV void group_accum(const DevSpec& sp, const GroupCtx& gc,
                     const RowCtxT<NA>& rc) {
#if YBG_GABL == 1
  return;
#endif
  uint64_t slot;
  if (rc.grp_null) {
    slot = gc.cap;  // the NULL-key group
    if (ybg_atomic_load_u32(&gc.state[slot]) != 1)
      ybg_atomic_store_rel_u32(&gc.state[slot], 1u);
  } else {
    bool is_str = rc.grp_len != 0;
    uint64_t kv = is_str ? (((uint64_t)rc.grp_len << 40) |
                            (rc.grp_datum - (uint64_t)(uintptr_t)gc.data))
                         : rc.grp_datum;
    slot = grp_find_or_insert(gc, kv, is_str);
    if (slot == ~0ull) {
      atomicAdd(gc.overflow, 1ull);
      return;
    }
  }
#if YBG_GABL == 2
  return;
#endif
  long long* v = gc.vals + slot * YBG_MAX_AGGS;
  unsigned long long* c = gc.cnts + slot * YBG_MAX_AGGS;
#pragma unroll
  for (int g = 0; g < NA; ++g) {
    if (g >= sp.num_aggs) continue;
    int op = sp.aggs[g].op;
    bool isnull =
        (op == YBG_AGG_COUNT_STAR) ? false : ((rc.agg_null >> g) & 1);
    if (isnull) continue;
    uint64_t d = rc.agg_datum[g];
    switch (op) {
      case YBG_AGG_COUNT_STAR:
      case YBG_AGG_COUNT:
        atomicAdd((unsigned long long*)&v[g], 1ull);
        break;
      case YBG_AGG_SUM_INT64:
        atomicAdd((unsigned long long*)&v[g], d);
        break;
      case YBG_AGG_SUM_DOUBLE:
        group_sum_f64_fixed(gc, slot, g, d);
        break;
      case YBG_AGG_MIN_INT64:
        ybg_atomic_min_i64(&v[g], (long long)d);
        break;
      case YBG_AGG_MAX_INT64:
        ybg_atomic_max_i64(&v[g], (long long)d);
        break;
      case YBG_AGG_MIN_DOUBLE:
        ybg_atomic_min_u64((unsigned long long*)&v[g], f64_ordered(d));
        break;
      case YBG_AGG_MAX_DOUBLE:
        ybg_atomic_max_u64((unsigned long long*)&v[g], f64_ordered(d));
        break;
      default:
        break;
    }
    // COUNT ops carry their contribution count in the value itself; the
    // export reconstructs cnt = val for them (one fewer atomic per row)
    if (op != YBG_AGG_COUNT_STAR && op != YBG_AGG_COUNT)
      atomicAdd(&c[g], 1ull);
  }
}

// Same accumulate from a deferred head record (the heads-resolution pass
// and the in-kernel not-consumed head path).
template <int NA>
DEV void group_accum_rec(const DevSpec& sp, const GroupCtx& gc,
                         const GroupHead& r) {
  uint64_t slot;
  if (r.grp_null) {
    slot = gc.cap;
    if (ybg_atomic_load_u32(&gc.state[slot]) != 1)
      ybg_atomic_store_rel_u32(&gc.state[slot], 1u);
  } else {
    bool is_str = r.grp_len != 0;
    uint64_t kv = is_str ? (((uint64_t)r.grp_len << 40) |
                            (r.grp_datum - (uint64_t)(uintptr_t)gc.data))
                         : r.grp_datum;
    slot = grp_find_or_insert(gc, kv, is_str);
    if (slot == ~0ull) {
      atomicAdd(gc.overflow, 1ull);
      return;
    }
  }
#if YBG_GABL == 2
  return;
#endif
  long long* v = gc.vals + slot * YBG_MAX_AGGS;
  unsigned long long* c = gc.cnts + slot * YBG_MAX_AGGS;
#pragma unroll
  for (int g = 0; g < NA; ++g) {
    if (g >= sp.num_aggs) continue;
    int op = sp.aggs[g].op;
    bool isnull =
        (op == YBG_AGG_COUNT_STAR) ? false : ((r.agg_null >> g) & 1);
    if (isnull) continue;
    uint64_t d = r.agg_datum[g];
    switch (op) {
      case YBG_AGG_COUNT_STAR:
      case YBG_AGG_COUNT:
        atomicAdd((unsigned long long*)&v[g], 1ull);
        break;
      case YBG_AGG_SUM_INT64:
        atomicAdd((unsigned long long*)&v[g], d);
        break;
      case YBG_AGG_SUM_DOUBLE:
        group_sum_f64_fixed(gc, slot, g, d);
        break;
      case YBG_AGG_MIN_INT64:
        ybg_atomic_min_i64(&v[g], (long long)d);
        break;
      case YBG_AGG_MAX_INT64:
        ybg_atomic_max_i64(&v[g], (long long)d);
        break;
      case YBG_AGG_MIN_DOUBLE:
        ybg_atomic_min_u64((unsigned long long*)&v[g], f64_ordered(d));
        break;
      case YBG_AGG_MAX_DOUBLE:
        ybg_atomic_max_u64((unsigned long long*)&v[g], f64_ordered(d));
        break;
      default:
        break;
    }
    if (op != YBG_AGG_COUNT_STAR && op != YBG_AGG_COUNT)
      atomicAdd(&c[g], 1ull);
  }
}

// Write one materialized row (PgTableRow analog — dockv/pg_row.h:91-179).
// rk bytes = the finalized row's key (rk_save), used for key-column datums.
template <int NA>
DEV void emit_row(const DevSpec& sp, EmitCtx* ec, const RowCtxT<NA>& rc,
                  const uint8_t* rk, uint32_t rk_len, uint64_t sort_key) {
  unsigned long long slot = atomicAdd(ec->row_counter, 1ull);
  if (slot >= ec->row_cap) {
    atomicAdd(ec->overflow, 1ull);
    return;
  }
  ec->sort_key[slot] = sort_key;
  int nk = ec->nk;
  for (int c = 0; c < nk; ++c) {
    uint64_t d = 0;
    uint32_t soff = 0, sl = 0;
    if (key_col_value(sp, rk, rk_len, c, &d, &soff, &sl)) {
      if (sp.key_types[c] == YBG_KT_STRING) {
        // unescape into the varlen heap ('\0\1' -> '\0',
        // doc_kv_util.h:101-167)
        unsigned long long off = atomicAdd(ec->varlen_counter, sl);
        if (off + sl > ec->varlen_cap) {
          atomicAdd(ec->overflow, 1ull);
          d = 0;
        } else {
          uint32_t o = 0;
          for (uint32_t i = 0; i < sl; ++i) {
            uint8_t b = rk[soff + i];
            ec->varlen[off + o++] = b;
            if (b == 0) ++i;  // skip escape byte
          }
          d = ((uint64_t)o << 40) | off;
        }
      }
      ec->key_datums[slot * nk + c] = d;
    }
  }
  int nc = ec->nc;
  for (int c = 0; c < nc; ++c) {
    uint64_t d = 0;
    if (!((rc.emit_null >> c) & 1)) {
      d = rc.emit_datums[c];
      if ((rc.emit_str >> c) & 1) {
        uint32_t sl = rc.emit_lens[c];
        unsigned long long off = atomicAdd(ec->varlen_counter, sl);
        if (off + sl > ec->varlen_cap) {
          atomicAdd(ec->overflow, 1ull);
          d = 0;
        } else {
          const uint8_t* src = (const uint8_t*)(uintptr_t)d;
          for (uint32_t i = 0; i < sl; ++i) ec->varlen[off + i] = src[i];
          d = ((uint64_t)sl << 40) | off;
        }
      }
    }
    ec->datums[slot * nc + c] = d;
  }
  ec->null_masks[slot] = rc.emit_null;
  ec->hashes[slot] =
      sp.has_hash ? (uint16_t)(((uint16_t)rk[1] << 8) | rk[2]) : 0;
}

// ---------------------------------------------------------------------------
// Specialized fast batch scanner (k_scan_fast): the dominant shape —
// fixed-length rowkeys, packed-V2 all-fixed-width values, frequent/2.1.1
// entry encodings, value-column typed compares, <= NA aggregates — as a
// SMALL straight-line loop. The general scan_one_interval below compiles
// to a ~380K-instruction mega-kernel whose issue slots drown in phi-copies,
// exec-mask bookkeeping and SGPR-spill lane shuffles; this loop carries a
// fraction of the live state so the compiler emits none of that. Anything
// outside the shape ABORTS the batch (return 0) with no side effects on
// the accumulators; the caller re-runs the batch through the general path.
// Semantics are pinned against scan_one_interval by sim + GPU parity tests.
// ---------------------------------------------------------------------------

DEV void u128_shr(uint64_t* hi, uint64_t* lo, uint32_t s) {
  if (s >= 64) {
    *lo = s >= 128 ? 0 : (*hi >> (s - 64));
    *hi = 0;
  } else if (s) {
    *lo = (*lo >> s) | (*hi << (64 - s));
    *hi >>= s;
  }
}

DEV void u128_shl(uint64_t* hi, uint64_t* lo, uint32_t s) {
  if (s >= 64) {
    *hi = s >= 128 ? 0 : (*lo << (s - 64));
    *lo = 0;
  } else if (s) {
    *hi = (*hi << s) | (*lo >> (64 - s));
    *lo <<= s;
  }
}

// Splice n (1..16) big-endian bytes (sv_hi:sv_lo, first byte at sv_hi's
// MSB) into the 16-byte register tail at SIGNED tail byte offset o
// (o = key_pos - (ukey_len - 16)); bytes falling outside [0,16) drop out.
// Replaces a per-byte tail_patch loop (~18 instructions per byte) with
// ~30 instructions per RANGE.
DEV void tail_splice(uint64_t* thi, uint64_t* tlo, int32_t o, uint32_t n,
                     uint64_t sv_hi, uint64_t sv_lo) {
  if (o <= -(int32_t)n || o >= 16) return;
  uint64_t mh = n >= 8 ? ~0ull : (~0ull << (64 - 8 * n));
  uint64_t ml = n <= 8 ? 0ull : (n >= 16 ? ~0ull : (~0ull << (128 - 8 * n)));
  if (o >= 0) {
    u128_shr(&sv_hi, &sv_lo, 8 * (uint32_t)o);
    u128_shr(&mh, &ml, 8 * (uint32_t)o);
  } else {
    u128_shl(&sv_hi, &sv_lo, 8 * (uint32_t)(-o));
    u128_shl(&mh, &ml, 8 * (uint32_t)(-o));
  }
  *thi = (*thi & ~mh) | (sv_hi & mh);
  *tlo = (*tlo & ~ml) | (sv_lo & ml);
}

// Accumulate ONE row's aggregate operand (acc_row's per-slot body).
DEV void combine_datum(int op, uint64_t* val, uint64_t* cnt, uint64_t v) {
  switch (op) {
    case YBG_AGG_COUNT_STAR:
    case YBG_AGG_COUNT:
      *val += 1;
      break;
    case YBG_AGG_SUM_INT64:
      *val = (uint64_t)((int64_t)*val + (int64_t)v);
      break;
    case YBG_AGG_SUM_DOUBLE: {
      double cur = __longlong_as_double((long long)*val) +
                   __longlong_as_double((long long)v);
      *val = (uint64_t)__double_as_longlong(cur);
      break;
    }
    case YBG_AGG_MIN_INT64:
      if (*cnt == 0 || (int64_t)v < (int64_t)*val) *val = v;
      break;
    case YBG_AGG_MAX_INT64:
      if (*cnt == 0 || (int64_t)v > (int64_t)*val) *val = v;
      break;
    case YBG_AGG_MIN_DOUBLE: {
      double dd = __longlong_as_double((long long)v);
      if (*cnt == 0 || dd < __longlong_as_double((long long)*val))
        *val = (uint64_t)__double_as_longlong(dd);
      break;
    }
    case YBG_AGG_MAX_DOUBLE: {
      double dd = __longlong_as_double((long long)v);
      if (*cnt == 0 || dd > __longlong_as_double((long long)*val))
        *val = (uint64_t)__double_as_longlong(dd);
      break;
    }
  }
  *cnt += 1;
}

// Lean typed compare (pred_compare's non-string, non-IN subset).
DEV bool pred_cmp_fast(const PredC& pr, uint64_t datum) {
  const int dtype = (int)(pr.opdt >> 8);
  int cmp;
  if (YBG_LIKELY(dtype != YBG_T_DOUBLE && dtype != YBG_T_FLOAT)) {
    int64_t a = (int64_t)datum, b = (int64_t)pr.datum;
    cmp = a < b ? -1 : (a > b ? 1 : 0);
  } else if (dtype == YBG_T_DOUBLE) {
    double a = __longlong_as_double((long long)datum);
    double b = __longlong_as_double((long long)pr.datum);
    cmp = a < b ? -1 : (a > b ? 1 : 0);
  } else {
    float a = __uint_as_float((uint32_t)datum);
    float b = __uint_as_float((uint32_t)pr.datum);
    cmp = a < b ? -1 : (a > b ? 1 : 0);
  }
  switch (pr.opdt & 0xff) {
    case YBG_PRED_GT: return cmp > 0;
    case YBG_PRED_GE: return cmp >= 0;
    case YBG_PRED_LT: return cmp < 0;
    case YBG_PRED_LE: return cmp <= 0;
    case YBG_PRED_EQ: return cmp == 0;
    default: return cmp != 0;
  }
}

// Host-side: may this spec take the fast kernel at all?
inline bool fast_eligible(const DevSpec& d) {
  if (d.fmt != YBG_ENC_THREE_SHARED_PARTS) return false;
  if (!d.fixed_rk_len || d.need_rowkey || d.group_col >= 0) return false;
  if (!d.v2_fixed_len || d.num_value_cols <= 0) return false;
  if (d.num_value_cols > 8) return false;  // NC template cap
  if (d.num_aggs > 2) return false;
  for (int i = 0; i < d.num_preds; ++i) {
    if (d.preds[i].is_key_col) return false;
    if (d.preds[i].op > YBG_PRED_NE) return false;  // no IN / IN_TUPLE
  }
  return true;
}

#ifndef YBG_FABORT
#define YBG_FABORT(n) return 0
#endif

// YBG_ABL: perf-forensics ablations of scan_batch_fast (results WRONG by
// design — never in a default build): 1 = skip register-tail patches and
// force visibility true; 2 = skip packed-value decode; 3 = skip the
// per-row finalize accumulation.
#ifndef YBG_ABL
#define YBG_ABL 0
#endif

// YBG_GABL: grouped-kernel ablations (results WRONG by design, never in
// a default build): 1 = skip the group hash table entirely (measures the
// scan/eval side alone); 2 = probe/insert but skip the per-agg atomics.
#ifndef YBG_GABL
#define YBG_GABL 0
#endif

// Returns 1 = batch done (accumulators updated), 0 = abort (accumulators
// and ho untouched except bht[3..5] restart candidates, which are benign
// duplicates under the MIN fold when the batch is retried).
// NC = compile-time value-column cap (4 or 8, >= sp.num_value_cols): the
// packed-row column loads unroll to NC branch-free word-pair loads that
// issue back-to-back, so a row costs ONE L1 latency instead of one per
// column (a runtime-bound loop cannot unroll and serializes the loads on
// the per-iteration vmcnt wait). Columns [num_value_cols, NC) are
// zero-padded in the spec (col_act = 0, v2_off = 0) and evaluate to
// no-ops.
// FUSE selects the changed-byte extraction shape (spec->expect_versions
// dispatch): false = separate LDS/tail-patch loops with below-rkb LDS
// bounds (best on single-version data), true = one fused pass per byte
// with unconditional LDS writes (best on MVCC version chains, where the
// below-rkb bound makes lane trip counts diverge and every byte would
// otherwise be extracted twice). Results are identical.
template <int NA, int NC, bool FUSE = false>
DEV int scan_batch_fast(const DevSpec& sp, const uint8_t* data,
                        const uint64_t* block_offsets, const Interval* ivs,
                        uint64_t n_ivs, uint64_t j_lo, uint64_t j_hi,
                        uint8_t* key, uint64_t* rmin,
                        uint32_t* entries, uint32_t* scanned,
                        uint32_t* matched, uint64_t* agg_val,
                        uint64_t* agg_cnt, HeadOut<NA>* ho,
                        bool* walked_next_out) {
  Interval iv = ivs[j_lo];
  const uint8_t* blk = data + block_offsets[iv.block];
  const uint8_t* p = blk + iv.start;
  const uint8_t* limit = blk + iv.end;
  uint64_t cur_iv = j_lo;
  Rdr rdr;
  rdr.init(p);
  bool at_restart = true;  // interval starts are restart points

#pragma unroll
  for (int g = 0; g < NA; ++g) { ho->val[g] = 0; ho->cnt[g] = 0; }
  ho->scanned = 0;
  ho->matched = 0;

  // batch-local accumulators: folded into the caller's only on success
  uint32_t e_b = 0, s_b = 0, m_b = 0;
  uint64_t av_b[NA], ac_b[NA];
#pragma unroll
  for (int g = 0; g < NA; ++g) { av_b[g] = 0; ac_b[g] = 0; }

  uint32_t key_len = 0;
  uint64_t thi = 0, tlo = 0;
  bool row_open = false, in_head = true, walked = false;
  bool base_seen = false, found = false;
  uint32_t pred_pass = 0, agg_null = 0xffffffffu;
  uint64_t agg_datum[NA];
#pragma unroll
  for (int g = 0; g < NA; ++g) agg_datum[g] = 0;
  const uint32_t rkb = sp.fixed_rk_len;
  EntryRef er;

  for (;;) {
    if (p >= limit) {
      if (cur_iv + 1 >= n_ivs) break;
      Interval nx = ivs[cur_iv + 1];
      cur_iv += 1;
      at_restart = true;
      const uint8_t* nblk = data + block_offsets[nx.block];
      const uint8_t* np = nblk + nx.start;
      limit = nblk + nx.end;
      blk = nblk;
      if (np != p) {
        p = np;
        rdr.init(p);
      }
      continue;
    }
    bool kchg;
    const uint8_t* q = nullptr;
    if (YBG_UNLIKELY(at_restart)) {
      // compact restart decode: three_shared_parts no-reuse form
      // (block_builder_internal.h case 2.0; full key inline). Keys longer
      // than 127 B (e2 == 0 -> varint ns1) or shorter than 24 B (register
      // tail needs ukey >= 16) abort to the general path.
      at_restart = false;
      if (limit - p < 3) YBG_FABORT(1);
      uint64_t w = load_u64_una(p);
      uint32_t b0 = (uint32_t)(w & 0xff);
      uint64_t e1;
      uint32_t e1len;
      if (!(b0 & 0x80)) {
        e1 = b0;
        e1len = 1;
      } else {
        uint32_t b1 = (uint32_t)((w >> 8) & 0xff);
        if (b1 & 0x80) YBG_FABORT(2);
        e1 = (b0 & 0x7f) | ((uint64_t)b1 << 7);
        e1len = 2;
      }
      uint32_t value_size = (uint32_t)(e1 >> 2);
      if (e1 & 1) YBG_FABORT(3);  // frequent case never starts a restart
      uint32_t e2 = (uint32_t)((w >> (8 * e1len)) & 0xff);
      if (e2 & 1) YBG_FABORT(4);  // reuse forms: not a self-contained restart
      uint32_t ns1 = e2 >> 1;
      // upper bound: the u64-chunk copy below writes up to ns1 rounded up
      // to 8 into the caller's kFastKeyCap LDS slot
      if (ns1 < 24 || ns1 > kFastKeyCap - 8) YBG_FABORT(5);
      uint32_t hl = e1len + 1;
      if ((uint64_t)(limit - p) < (uint64_t)hl + ns1 + value_size) YBG_FABORT(6);
      const uint8_t* kp2 = p + hl;
      // row-change detection: compare the first rkb bytes before copying
      kchg = !row_open;
      if (row_open) {
        uint32_t full = rkb & ~7u;
        for (uint32_t i = 0; i < full; i += 8)
          if (load_u64_una(&key[i]) != load_u64_una(kp2 + i)) kchg = true;
        if (rkb & 7) {
          uint64_t m = (1ull << (8 * (rkb & 7))) - 1;
          if (((load_u64_una(&key[full]) ^ load_u64_una(kp2 + full)) & m))
            kchg = true;
        }
      }
      for (uint32_t i = 0; i < ns1; i += 8) {
        uint64_t w8 = load_u64_una(kp2 + i);
        __builtin_memcpy(&key[i], &w8, 8);
      }
      key_len = ns1;
      tail_from_lds(kp2, ns1 - 8, &thi, &tlo);
      er.value = kp2 + ns1;
      er.value_len = value_size;
      er.shared = 0;
      q = er.value + value_size;
      rdr.seek(er.value);
    } else {
      // Unified STABLE-LENGTH delta decode: frequent, 2.1.1 (d2 == 0) and
      // the general form with last8 reuse and zero deltas
      // (block_builder_internal.h:100-239). Zero deltas mean prev_ns2 ==
      // ns2 and prev_mid_start == new_mid_start: the key LENGTH and every
      // byte position are unchanged, the shared middle never moves, and
      // the only changed bytes are [sp, sp+ns1) and the ns2 range. Bytes
      // below the fixed rowkey length go to LDS (the restart row-compare
      // reads them); bytes inside the register tail window are patched in
      // registers; bytes between rkb and the tail window need no store at
      // all (nothing reads them before the next restart overwrites).
      if (limit - p < 8) YBG_FABORT(7);
      rdr.align8();  // body bytes peek up to 24 ahead of the position
      uint64_t w = rdr.peek8();
      uint32_t b0 = (uint32_t)(w & 0xff);
      uint64_t e1;
      uint32_t e1len;
      if (!(b0 & 0x80)) {
        e1 = b0;
        e1len = 1;
      } else {
        uint32_t b1 = (uint32_t)((w >> 8) & 0xff);
        if (b1 & 0x80) YBG_FABORT(8);
        e1 = (b0 & 0x7f) | ((uint64_t)b1 << 7);
        e1len = 2;
      }
      const uint32_t value_size = (uint32_t)(e1 >> 2);
      const uint64_t inc = (e1 & 2) << 7;
      uint32_t ns1, ns2, hl, sp;
      uint64_t w2 = rdr.peek8_at(8);
      auto hdr_byte = [&](uint32_t j) -> uint32_t {
        return (uint32_t)((j < 8 ? (w >> (8 * j)) : (w2 >> (8 * (j - 8)))) &
                          0xff);
      };
      if (e1 & 1) {  // frequent
        sp = hdr_byte(e1len);
        if (sp & 0x80) YBG_FABORT(9);
        ns1 = 1;
        ns2 = 1;
        hl = e1len + 1;
      } else {
        uint32_t e2 = hdr_byte(e1len);
        if ((e2 & 3) == 1) {  // case 2.1.1; d2 != 0 changes the length
          if (e2 & 4) YBG_FABORT(10);
          ns1 = (e2 >> 3) & 7;
          ns2 = (e2 >> 6) & 3;
          sp = hdr_byte(e1len + 1);
          if (sp & 0x80) YBG_FABORT(11);
          hl = e1len + 2;
        } else if ((e2 & 7) == 7 && !(e2 & 40)) {
          // general form: reuse8 set, ns1_delta and ns2_delta absent
          uint32_t o = e1len + 1;
          ns1 = hdr_byte(o);
          if (ns1 & 0x80) YBG_FABORT(12);
          ++o;
          ns2 = 0;
          if (e2 & 16) {
            ns2 = hdr_byte(o);
            if (ns2 & 0x80) YBG_FABORT(13);
            ++o;
          }
          sp = hdr_byte(o);
          if (sp & 0x80) YBG_FABORT(14);
          hl = o + 1;
        } else {
          YBG_FABORT(15);  // restart mid-interval / delta forms: general path
        }
      }
      const uint32_t nb = hl + ns1 + ns2;
      // splice-source widths: ns1 from a 16-byte load pair, ns2 from one
      if (nb > 24 || ns1 > 16 || ns2 > 8) YBG_FABORT(16);
      const uint64_t prev_len = key_len;
      const uint64_t prev_except = (uint64_t)sp + ns1 + ns2 + 8;
      if (prev_len < prev_except ||
          (uint64_t)(limit - p) < (uint64_t)nb + value_size)
        YBG_FABORT(17);
      // stable length: new_len == prev_len; ns2 range stays in place
      const uint32_t ukey = (uint32_t)prev_len - 8;
      const uint32_t ns2s = (uint32_t)(prev_len - ns2 - 8);
      // a ns2 range dipping below rkb implies sp < rkb (row change), so
      // kchg is already correct; those bytes just need LDS writes too
      kchg = sp < rkb || !row_open;
      {
        // changed bytes: LDS only below rkb (the restart row-compare
        // source), register tail inside the window, nothing in between.
        // YBG_TAILMODE selects the tail-update strategy (A/B-measured):
        //   0 per-byte tail_patch from window bytes (needs align8 + w3)
        //   1 range splice sourced by direct loads from the entry bytes
        //   2 range splice sourced by window peeks
#if YBG_TAILMODE == 0
        rdr.align8();
        w = rdr.peek8();
        w2 = rdr.peek8_at(8);
        uint64_t w3 = rdr.peek8_at(16);
        auto body_byte = [&](uint32_t j) -> uint8_t {
          uint64_t src = j < 8 ? w : (j < 16 ? w2 : w3);
          return (uint8_t)(src >> (8 * (j & 7)));
        };
        // fused: each changed byte is extracted ONCE and feeds both the
        // LDS row-compare copy (positions below rkb) and the register
        // tail patch — body_byte's 3-way select is the per-byte cost
        uint32_t lds_n = sp < rkb ? (rkb - sp < ns1 ? rkb - sp : ns1) : 0;
        uint32_t lds2 = ns2s < rkb ? (rkb - ns2s < ns2 ? rkb - ns2s : ns2)
                                   : 0;
        if constexpr (FUSE) {
          // MVCC shape: one pass per changed byte, LDS write
          // unconditional (key bytes >= rkb are dead on this path — only
          // [0, rkb) feeds the restart row-compare, the slot is rewritten
          // at the next restart, and sp + ns1 <= key_len <= kFastKeyCap -
          // 8 keeps it in-slot). Measured +16% on version-chain data,
          // -4% on single-version data (reproduced twice each way).
          (void)lds_n;
          (void)lds2;
          for (uint32_t i = 0; i < ns1; ++i) {
            uint8_t b = body_byte(hl + i);
            key[sp + i] = b;
#if YBG_ABL != 1
            tail_patch(&thi, &tlo, ukey, sp + i, b);
#endif
          }
          for (uint32_t i = 0; i < ns2; ++i) {
            uint8_t b = body_byte(hl + ns1 + i);
            key[ns2s + i] = b;
#if YBG_ABL != 1
            tail_patch(&thi, &tlo, ukey, ns2s + i, b);
#endif
          }
        } else {
          // single-version shape: separate loops, LDS bounded below rkb
          for (uint32_t i = 0; i < lds_n; ++i)
            key[sp + i] = body_byte(hl + i);
          for (uint32_t i = 0; i < lds2; ++i)
            key[ns2s + i] = body_byte(hl + ns1 + i);
#if YBG_ABL != 1
          for (uint32_t i = 0; i < ns1; ++i)
            tail_patch(&thi, &tlo, ukey, sp + i, body_byte(hl + i));
          for (uint32_t i = 0; i < ns2; ++i)
            tail_patch(&thi, &tlo, ukey, ns2s + i, body_byte(hl + ns1 + i));
#endif
        }
#else
        uint32_t lds_n = sp < rkb ? (rkb - sp < ns1 ? rkb - sp : ns1) : 0;
        for (uint32_t i = 0; i < lds_n; ++i)
          key[sp + i] = p[hl + i];
        uint32_t lds2 = ns2s < rkb ? (rkb - ns2s < ns2 ? rkb - ns2s : ns2)
                                   : 0;
        for (uint32_t i = 0; i < lds2; ++i)
          key[ns2s + i] = p[hl + ns1 + i];
        const int32_t T = (int32_t)ukey - 16;
#if YBG_TAILMODE == 2
        rdr.align8();
        uint64_t s1h = __builtin_bswap64(rdr.peek8_at(hl));
        uint64_t s1l =
            ns1 > 8 ? __builtin_bswap64(rdr.peek8_at(hl + 8)) : 0;
        uint64_t s2h =
            ns2 ? __builtin_bswap64(hl + ns1 <= 17
                                        ? rdr.peek8_at(hl + ns1)
                                        : load_u64_una(p + hl + ns1))
                : 0;
#else
        uint64_t s1h = __builtin_bswap64(load_u64_una(p + hl));
        uint64_t s1l =
            ns1 > 8 ? __builtin_bswap64(load_u64_una(p + hl + 8)) : 0;
        uint64_t s2h =
            ns2 ? __builtin_bswap64(load_u64_una(p + hl + ns1)) : 0;
#endif
        tail_splice(&thi, &tlo, (int32_t)sp - T, ns1, s1h, s1l);
        if (ns2) tail_splice(&thi, &tlo, (int32_t)ns2s - T, ns2, s2h, 0);
#endif
      }
      (void)inc;
      rdr.consume(nb);
      er.value = rdr.pos();
      er.value_len = value_size;
      q = er.value + value_size;
    }
    const uint32_t ukey_len = key_len - 8;
    if (ukey_len < rkb + 2 || ukey_len < 16) YBG_FABORT(19);
    const uint32_t ht_sz = (uint32_t)tlo & 0x1f;
    // ht_sz >= 16: the marker byte lies outside the register tail window
    // (LDS would be stale on this path) — general path handles it
    if (ht_sz == 0 || ht_sz >= 16) YBG_FABORT(20);
    const uint32_t mb = ht_sz + 1;
    uint32_t marker;
    if (mb <= 8) marker = (uint32_t)(tlo >> (8 * (mb - 1))) & 0xff;
    else marker = (uint32_t)(thi >> (8 * (mb - 9))) & 0xff;
    if (marker != kHybridTimeByte) YBG_FABORT(21);
    // subkey entries (column updates / liveness) take the general path
    if (ukey_len != rkb + ht_sz + 1) YBG_FABORT(22);
    uint64_t ht_hi, ht_lo;
    {
      uint32_t s = 8 * (16 - ht_sz);
      if (s == 0) { ht_hi = thi; ht_lo = tlo; }
      else if (s < 64) {
        ht_hi = (thi << s) | (tlo >> (64 - s));
        ht_lo = tlo << s;
      } else {
        ht_hi = tlo << (s - 64);
        ht_lo = 0;
      }
    }
    // row boundary
    if (kchg) {
      if (row_open) {
#if YBG_ABL == 3
        if (false) {
#else
        if (found) {
#endif
          bool hit = (pred_pass & sp.value_pred_mask) == sp.value_pred_mask;
          if (in_head) {
            ho->scanned += 1;
            if (hit) {
              ho->matched += 1;
#pragma unroll
              for (int g = 0; g < NA; ++g) {
                if (g >= sp.num_aggs) continue;
                bool nul = (sp.agg_op[g] != YBG_AGG_COUNT_STAR) &&
                           ((agg_null >> g) & 1);
                if (!nul) {
                  combine_datum(sp.agg_op[g], &ho->val[g], &ho->cnt[g],
                                agg_datum[g]);
                }
              }
            }
          } else {
            s_b += 1;
            if (hit) {
              m_b += 1;
#pragma unroll
              for (int g = 0; g < NA; ++g) {
                if (g >= sp.num_aggs) continue;
                bool nul = (sp.agg_op[g] != YBG_AGG_COUNT_STAR) &&
                           ((agg_null >> g) & 1);
                if (!nul)
                  combine_datum(sp.agg_op[g], &av_b[g], &ac_b[g],
                                agg_datum[g]);
              }
            }
          }
        }
        in_head = false;
        row_open = false;
        if (cur_iv >= j_hi) break;
      }
      row_open = true;
      base_seen = false;
      found = false;
      pred_pass = 0;
      agg_null = 0xffffffffu;
    }
    e_b += (cur_iv < j_hi);
    walked |= (cur_iv == j_hi);
    // visibility (no intent-prefixed values in the fast shape)
    const uint32_t vb0 =
        er.value_len ? (uint32_t)(rdr.peek8() & 0xff) : 0u;
    if (YBG_UNLIKELY(vb0 == kHybridTimeByte)) YBG_FABORT(23);
#if YBG_ABL == 1
    const bool visible = true;
#else
    const bool visible =
        u128_slice_cmp(ht_hi, ht_lo, ht_sz, sp.reg_lim.hi, sp.reg_lim.lo,
                       sp.reg_lim.len) >= 0;
#endif
    if (visible) {
      if (YBG_UNLIKELY(sp.track_restart)) {
        if (u128_slice_cmp(ht_hi, ht_lo, ht_sz, sp.read.hi, sp.read.lo,
                           sp.read.len) < 0) {
          uint64_t* rr = rmin;
          if (rr[2] == 0 || u128_slice_cmp(ht_hi, ht_lo, ht_sz, rr[0],
                                           rr[1], (uint32_t)rr[2]) < 0) {
            rr[0] = ht_hi;
            rr[1] = ht_lo;
            rr[2] = ht_sz;
          }
        }
      }
      if (!base_seen) {
        base_seen = true;
#if YBG_ABL == 2
        if (true) {
          found = true;
          pred_pass = sp.value_pred_mask;
          agg_null = 0;
          rdr.seek(q);
          p = q;
          continue;
        } else
#endif
        if (YBG_LIKELY(er.value_len == sp.v2_fixed_len && vb0 == kPackedV2B &&
                       (rdr.peek8() & 0xff8000u) == 0)) {
          const uint8_t* value = rdr.pos();
          rdr.seek(value + er.value_len);  // next window loads issue now
          // fixed-offset column extraction (decode_packed_v2_fixed, lean
          // eval) in two fully-unrolled passes: a branch-free load pass
          // issuing all NC word pairs, then the eval pass over registers
          // (see the NC doc on scan_batch_fast)
          const uintptr_t a = (uintptr_t)value;
          const uint64_t* qw = (const uint64_t*)(a & ~(uintptr_t)7);
          const uint32_t abase = (uint32_t)(a & 7);
          uint64_t uv[NC];
#pragma unroll
          for (int i = 0; i < NC; ++i) {
            const uint32_t ob = abase + sp.v2_off[i];
            const uint32_t wi = ob >> 3, shb = (ob & 7) * 8;
            const uint64_t w0 = qw[wi], w1 = qw[wi + 1];
            uv[i] = shb ? (w0 >> shb) | (w1 << (64 - shb)) : w0;
          }
#pragma unroll
          for (int i = 0; i < NC; ++i) {
            const uint32_t act = sp.col_act[i];
            uint64_t u = uv[i];
            const uint32_t dt = (act >> kActDtShift) & kActDtM;
            switch ((act >> kActV2Shift) & kActV2M) {
              case 1:
                u = (dt == YBG_T_INT8) ? (uint64_t)(int64_t)(int8_t)u
                                       : (u & 0xff);
                break;
              case 2:
                u = (dt == YBG_T_INT16) ? (uint64_t)(int64_t)(int16_t)u
                                        : (u & 0xffff);
                break;
              case 4:
                u = (dt == YBG_T_INT32) ? (uint64_t)(int64_t)(int32_t)u
                                        : (u & 0xffffffffull);
                break;
              default:
                break;
            }
            uint32_t pm = act & kActPredM;
            while (pm) {
              int pi = __builtin_ctz(pm);
              pm &= pm - 1;
              bool pass = pred_cmp_fast(sp.predc[pi], u);
              pred_pass =
                  (pred_pass & ~(1u << pi)) | ((uint32_t)pass << pi);
            }
            uint32_t am = (act >> kActAggShift) & kActAggM;
            if (am) {
#pragma unroll
              for (int g = 0; g < NA; ++g) {
                if (am & (1u << g)) {
                  agg_datum[g] = u;
                  agg_null &= ~(1u << g);
                }
              }
            }
          }
          found = true;
          p = q;
          continue;  // reader already past the value
        } else if (er.value_len == 1 && vb0 == kTombB) {
          // row tombstone: row exists but is deleted (found stays false)
        } else {
          YBG_FABORT(24);  // control fields / V1 / null-mask V2: general path
        }
      }
    }
    rdr.seek(q);
    p = q;
  }
  // final open row
  if (row_open && found) {
    bool hit = (pred_pass & sp.value_pred_mask) == sp.value_pred_mask;
    if (in_head) {
      ho->scanned += 1;
      if (hit) {
        ho->matched += 1;
#pragma unroll
        for (int g = 0; g < NA; ++g) {
          if (g >= sp.num_aggs) continue;
          bool nul = (sp.agg_op[g] != YBG_AGG_COUNT_STAR) &&
                     ((agg_null >> g) & 1);
          if (!nul)
            combine_datum(sp.agg_op[g], &ho->val[g], &ho->cnt[g],
                          agg_datum[g]);
        }
      }
    } else {
      s_b += 1;
      if (hit) {
        m_b += 1;
#pragma unroll
        for (int g = 0; g < NA; ++g) {
          if (g >= sp.num_aggs) continue;
          bool nul = (sp.agg_op[g] != YBG_AGG_COUNT_STAR) &&
                     ((agg_null >> g) & 1);
          if (!nul)
            combine_datum(sp.agg_op[g], &av_b[g], &ac_b[g], agg_datum[g]);
        }
      }
    }
  }
  *entries += e_b;
  *scanned += s_b;
  *matched += m_b;
#pragma unroll
  for (int g = 0; g < NA; ++g) {
    if (g < sp.num_aggs)
      combine1(sp.agg_op[g], &agg_val[g], &agg_cnt[g], av_b[g], ac_b[g]);
  }
  *walked_next_out = walked;
  return 1;
}

// Scan the interval range [j, j_hi) as ONE continuous stream (j_hi = j+1
// is the classic one-interval form). The batch DEFERS its first (head)
// row and WALKS its last row into interval j_hi; interior interval
// boundaries are just stream positions (contiguous intervals of one block
// continue without re-initializing the reader). iv_flags, when non-null
// (the emit flags pre-pass), receives per-INTERVAL head-consumption:
// iv_flags[i] = 1 iff interval i's first row is a continuation of a row
// from interval i-1 — exactly what k_emit's per-interval workers need.
template <int NA, bool EMIT = false, bool GROUP = false>
DEV bool scan_one_interval(const DevSpec& sp, const uint8_t* data,
                           const uint64_t* block_offsets, const Interval* ivs,
                           uint64_t n_ivs, uint64_t j, const uint8_t* aux,
                           uint8_t* key, uint8_t* rk_save, uint64_t* bht,
                           uint32_t* entries,
                           uint32_t* scanned, uint32_t* matched,
                           uint64_t* agg_val, uint64_t* agg_cnt,
                           HeadOut<NA>* ho, bool* walked_next_out,
                           EmitCtx* ec = nullptr,
                           uint64_t* emit_datums = nullptr,
                           uint32_t* emit_lens = nullptr,
                           const GroupCtx* gc = nullptr,
                           const uint32_t* head_flags = nullptr,
                           GroupHead* gh_out = nullptr,
                           uint64_t j_hi_in = 0,
                           uint32_t* iv_flags = nullptr) {
  const uint64_t j_hi = j_hi_in ? j_hi_in : j + 1;
  Interval iv = ivs[j];
  const uint8_t* blk = data + block_offsets[iv.block];
  const uint8_t* p = blk + iv.start;
  const uint8_t* limit = blk + iv.end;
  uint64_t cur_iv = j;
  bool crossed = false;  // next decoded entry is the first of its interval
  Rdr rdr;
  rdr.init(p);

#pragma unroll
  for (int g = 0; g < NA; ++g) { ho->val[g] = 0; ho->cnt[g] = 0; }
  ho->scanned = 0;
  ho->matched = 0;
  bool walked_next = false;
  bool fail = false;

  uint32_t key_len = 0;
  uint64_t reg_last8 = 0;
  uint64_t thi = 0, tlo = 0;  // register user-key tail (see tail_from_lds)
  bool tail_ok = false;
  uint32_t rk_len = 0;
  bool row_open = false;
  bool in_head = true;
  RowCtxT<NA> rc;
  rc.bht = bht;
  rc.emit_datums = EMIT ? emit_datums : nullptr;
  rc.emit_lens = EMIT ? emit_lens : nullptr;
  row_reset(&rc, sp);
  EntryRef er;
  uint64_t row_sort_key = 0;
  // Fixed-rowkey fast mode: row changes are detected inside the decoder
  // (compare-on-write below rkb); rk_save is not maintained (finalize needs
  // no rowkey bytes when no bounds/key predicates are set).
  const uint32_t rkb =
      (sp.fixed_rk_len && !sp.need_rowkey) ? sp.fixed_rk_len : 0;

  for (;;) {
    if (p >= limit) {
      if (cur_iv + 1 >= n_ivs) break;
      Interval nx = ivs[cur_iv + 1];
      cur_iv += 1;
      crossed = true;
      const uint8_t* nblk = data + block_offsets[nx.block];
      const uint8_t* np = nblk + nx.start;
      limit = nblk + nx.end;
      if (np != p) {  // non-contiguous (block crossing): reposition
        blk = nblk;
        p = np;
        rdr.init(p);
      } else {
        blk = nblk;
      }
      // first entry of a restart interval is self-contained; the carried
      // key state is overwritten by its full-key decode.
      continue;
    }
    bool kchg = false;
    const uint8_t* q = decode_entry(sp.fmt, &rdr, limit, key, &key_len,
                                    &reg_last8, rkb, &kchg, &er, &thi, &tlo,
                                    &tail_ok);
    if (!q || key_len < 10) { fail = true; break; }
    uint32_t ukey_len = key_len - 8;
    uint32_t ht_sz;
    uint64_t ht_hi, ht_lo;
    if (YBG_LIKELY(ukey_len >= 16)) {
      if (!tail_ok) {
        tail_from_lds(key, ukey_len, &thi, &tlo);
        tail_ok = true;
      }
      ht_sz = (uint32_t)tlo & 0x1f;
      uint32_t mb = ht_sz + 1;  // marker distance from the key end
      uint32_t marker;
      if (mb <= 8) marker = (uint32_t)(tlo >> (8 * (mb - 1))) & 0xff;
      else if (mb <= 16) marker = (uint32_t)(thi >> (8 * (mb - 9))) & 0xff;
      else marker = key[ukey_len - mb];  // ht_sz == 16 only
      if (ht_sz == 0 || ukey_len < ht_sz + 2 ||
          marker != kHybridTimeByte) { fail = true; break; }
      // visibility slice = (thi:tlo) << 8*(16-ht_sz): the top ht_sz bytes
      // are the encoded DocHybridTime, the shifted-in bits are zero — the
      // exact zero-padded slice_u128 layout
      uint32_t s = 8 * (16 - ht_sz);
      if (s == 0) { ht_hi = thi; ht_lo = tlo; }
      else if (s < 64) {
        ht_hi = (thi << s) | (tlo >> (64 - s));
        ht_lo = tlo << s;
      } else {
        ht_hi = tlo << (s - 64);
        ht_lo = 0;
      }
    } else {
      ht_sz = key[ukey_len - 1] & 0x1f;
      if (ht_sz == 0 || ukey_len < ht_sz + 2 ||
          key[ukey_len - ht_sz - 1] != kHybridTimeByte) { fail = true; break; }
      slice_u128(key + ukey_len - ht_sz, ht_sz, &ht_hi, &ht_lo);
    }
    uint32_t prefix_len = ukey_len - ht_sz - 1;
    uint32_t rk;
    if (sp.fixed_rk_len) {
      rk = sp.fixed_rk_len;
      if (rk > prefix_len) { fail = true; break; }
    } else {
      rk = dockey_len(sp, key, prefix_len);
      if (!rk || rk > prefix_len) { fail = true; break; }
    }

    bool row_change;
    if (rkb) {
      row_change = !row_open || kchg;
    } else {
      // Bytes [0, er.shared) are identical to the previous entry's key by
      // construction (delta encoding), and while the row is open the
      // previous entry's rowkey == rk_save, so only [er.shared, rk)
      // needs comparing.
      row_change = !row_open || (rk != rk_len);
      if (!row_change && er.shared < rk) {
        for (uint32_t i = er.shared; i < rk; ++i) {
          if (key[i] != rk_save[i]) { row_change = true; break; }
        }
      }
    }
    if (row_change) {
      if (row_open) {
        // finalize previous row (doc_rowwise_iterator row boundary)
        const uint8_t* rkp = (rkb && !EMIT) ? key : rk_save;
        if (rc.found && in_bounds(sp, rkp, rk_len, aux)) {
          // explicit in_head/main branches (never a runtime-selected
          // pointer): both destinations must stay register-promotable
          bool hit = (rc.pred_pass & sp.value_pred_mask) ==
                         sp.value_pred_mask &&
                     eval_key_preds(sp, rkp, rk_len, aux);
          if (in_head) {
            ho->scanned += 1;
            if (hit) {
              ho->matched += 1;
              acc_row(sp, rc, ho->val, ho->cnt);
            }
          } else {
            *scanned += 1;
            if (hit) {
              *matched += 1;
              acc_row(sp, rc, agg_val, agg_cnt);
            }
          }
          if (hit) {
            if (EMIT && (!in_head || ec->head_consumed[j] == 0))
              emit_row(sp, ec, rc, rkp, rk_len, row_sort_key);
            if (GROUP) {
              if (!in_head) {
                group_accum(sp, *gc, rc);
              } else if (gh_out) {
                // defer the head row (ownership unknown): raw operands
                gh_out->grp_datum = rc.grp_datum;
                gh_out->grp_len = rc.grp_len;
                gh_out->grp_null = rc.grp_null ? 1u : 0u;
                gh_out->agg_null = rc.agg_null;
#pragma unroll
                for (int g = 0; g < NA; ++g)
                  gh_out->agg_datum[g] = rc.agg_datum[g];
                gh_out->hit = 1;
              } else if (head_flags && head_flags[j] == 0) {
                group_accum(sp, *gc, rc);
              }
            }
          }
        }
        in_head = false;
        row_open = false;
        if (cur_iv >= j_hi) break;  // tail walk ended at a new row
      }
      if (!rkb || EMIT) {
        for (uint32_t i = 0; i < rk; ++i) rk_save[i] = key[i];
      }
      rk_len = rk;
      row_open = true;
      if (EMIT)  // scan position: (interval, entry offset within block)
        row_sort_key = ((uint64_t)cur_iv << 16) | (uint64_t)(p - blk);
      row_reset(&rc, sp);
    }
    if (iv_flags && crossed && !row_change) iv_flags[cur_iv] = 1;
    crossed = false;
    *entries += (cur_iv < j_hi);
    walked_next |= (cur_iv == j_hi);
    if (!process_entry(sp, data, aux, key, key_len, er.value, er.value_len,
                       rk_len, &rc, &rdr, ht_sz, ht_hi, ht_lo)) {
      fail = true;
      break;
    }
    rdr.seek(q);
    p = q;
  }
  const uint8_t* rkp_end = (rkb && !EMIT) ? key : rk_save;
  if (!fail && row_open && rc.found && in_bounds(sp, rkp_end, rk_len, aux)) {
    bool hit = (rc.pred_pass & sp.value_pred_mask) == sp.value_pred_mask &&
               eval_key_preds(sp, rkp_end, rk_len, aux);
    if (in_head) {
      ho->scanned += 1;
      if (hit) {
        ho->matched += 1;
        acc_row(sp, rc, ho->val, ho->cnt);
      }
    } else {
      *scanned += 1;
      if (hit) {
        *matched += 1;
        acc_row(sp, rc, agg_val, agg_cnt);
      }
    }
    if (hit) {
      if (EMIT && (!in_head || ec->head_consumed[j] == 0))
        emit_row(sp, ec, rc, rkp_end, rk_len, row_sort_key);
      if (GROUP) {
        if (!in_head) {
          group_accum(sp, *gc, rc);
        } else if (gh_out) {
          gh_out->grp_datum = rc.grp_datum;
          gh_out->grp_len = rc.grp_len;
          gh_out->grp_null = rc.grp_null ? 1u : 0u;
          gh_out->agg_null = rc.agg_null;
#pragma unroll
          for (int g = 0; g < NA; ++g)
            gh_out->agg_datum[g] = rc.agg_datum[g];
          gh_out->hit = 1;
        } else if (head_flags && head_flags[j] == 0) {
          group_accum(sp, *gc, rc);
        }
      }
    }
  }
  *walked_next_out = walked_next;
  return !fail;
}

// ---------------------------------------------------------------------------
// Host-side spec translation (shared by the ABI and the host simulator).
// ---------------------------------------------------------------------------
inline void build_dev_spec(const ybg_scan_spec_t* spec, DevSpec* dp,
                           unsigned char* aux, uint32_t* aux_len,
                           uint32_t aux_cap) {
  DevSpec& d = *dp;
  memset(&d, 0, sizeof(d));
  const ybg_schema_t& sc = spec->schema;
  d.has_hash = sc.has_hash;
  d.num_hash_cols = sc.num_hash_cols;
  d.num_range_cols = sc.num_range_cols;
  for (int i = 0; i < YBG_MAX_KEYCOLS; ++i) d.key_types[i] = sc.key_types[i];
  d.num_value_cols = sc.num_value_cols;
  d.fuse_hint = spec->expect_versions != 0;
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
    // regular_limit = memcmp-min(enc(read), enc(local)) == enc(max(read,
    // local)) — intent_aware_iterator.h:74-77
    int rl = spec->read_time.read_len, ll = spec->read_time.local_limit_len;
    int n = rl < ll ? rl : ll;
    int cmp = memcmp(spec->read_time.local_limit, spec->read_time.read, n);
    bool local_smaller = cmp < 0 || (cmp == 0 && ll < rl);
    d.reg_lim = local_smaller ? d.local_lim : d.read;
    // Restart tracking (the reference's UpdateMaxSeenHt is unconditional;
    // GetReadRestartData reports when max-seen > read). A visible record
    // with commit > read can exist when EITHER limit extends beyond read:
    // regular records through local_limit, committed-intent records
    // through GLOBAL_limit (the :1249-1267 rule) — gating on the local
    // window alone missed intent-carrying records when local == read <
    // global (caught by the round-2 GPU/oracle soak). Encoded order is
    // reversed, so "limit beyond read" == enc(limit) < enc(read).
    int gl = spec->read_time.global_limit_len;
    int n2 = gl < rl ? gl : rl;
    int cmp2 = memcmp(spec->read_time.global_limit, spec->read_time.read,
                      n2);
    bool global_beyond = cmp2 < 0 || (cmp2 == 0 && gl < rl);
    d.track_restart = (local_smaller || global_beyond) ? 1 : 0;
  }
  uint32_t pos = 0;
  auto put = [&](const uint8_t* b, uint64_t n) {
    uint32_t off = pos;
    if (b && n && pos + n <= aux_cap) {
      memcpy(aux + pos, b, n);
      pos += (uint32_t)n;
    }
    return off;
  };
  d.num_preds = spec->num_preds;
  d.value_pred_mask = 0;
  for (int i = 0; i < spec->num_preds; ++i) {
    const ybg_pred_t& p = spec->preds[i];
    DevPred& pr = d.preds[i];
    pr.is_key_col = p.is_key_col;
    pr.col = p.col;
    pr.op = p.op;
    pr.datum = p.datum;
    pr.str_len = (uint32_t)p.bytes_len;
    pr.rhs_off = put(p.bytes, p.bytes_len);
    if (!p.is_key_col) d.value_pred_mask |= 1u << i;
  }
  d.num_aggs = spec->num_aggs;
  for (int i = 0; i < spec->num_aggs; ++i) {
    d.aggs[i].op = spec->aggs[i].op;
    d.aggs[i].col = spec->aggs[i].col;
    d.agg_op[i] = spec->aggs[i].op;
  }
  d.group_col = spec->group_col > 0 ? spec->group_col - 1 : -1;
  // hot-path compaction: per-column action words + compact predicate records
  d.key_pred_mask = 0;
  for (int i = 0; i < spec->num_preds; ++i) {
    const DevPred& pr = d.preds[i];
    PredC& pc = d.predc[i];
    if (pr.is_key_col) {
      d.key_pred_mask |= 1u << i;
      pc.datum = 0;
      pc.opdt = 0;
      continue;
    }
    int dt = d.cols[pr.col].dtype;
    pc.opdt = (uint32_t)pr.op | ((uint32_t)dt << 8);
    if (pr.op == YBG_PRED_IN_RANGE) {
      pc.datum = (((uint64_t)(pr.str_len / 24) << 32) | pr.rhs_off);
    } else if (pr.op == YBG_PRED_IN) {
      uint64_t cnt;
      if (dt == YBG_T_STRING) {
        // [u32 len][bytes] records
        cnt = 0;
        uint32_t o = 0;
        while (o + 4 <= pr.str_len) {
          uint32_t ol;
          memcpy(&ol, aux + pr.rhs_off + o, 4);
          o += 4 + ol;
          ++cnt;
        }
      } else {
        cnt = pr.str_len / 8;
      }
      pc.datum = ((cnt << 32) | pr.rhs_off);
    } else if (dt == YBG_T_STRING)
      pc.datum = (((uint64_t)pr.str_len << 32) | pr.rhs_off);
    else
      pc.datum = pr.datum;
    pc.pad_ = 0;
  }
  for (int c = 0; c < sc.num_value_cols; ++c) {
    uint32_t act = 0;
    for (int i = 0; i < spec->num_preds; ++i)
      if (!d.preds[i].is_key_col && d.preds[i].col == c) act |= 1u << i;
    for (int g = 0; g < spec->num_aggs; ++g)
      if (d.aggs[g].col == c && d.aggs[g].op != YBG_AGG_COUNT_STAR)
        act |= 1u << (kActAggShift + g);
    if (d.group_col == c) act |= kActGroup;
    act |= ((uint32_t)d.cols[c].dtype & kActDtM) << kActDtShift;
    act |= ((uint32_t)d.cols[c].v2_fixed & kActV2M) << kActV2Shift;
    d.col_act[c] = act;
    d.col_ids[c] = d.cols[c].id;
  }
  {
    // fixed-offset packed-V2 fast path: usable when every value column is
    // fixed-width ('|' + 1-byte version + flags + bodies)
    uint32_t off = 3;
    int nfp = 0;
    bool all_fixed = sc.num_value_cols > 0;
    for (int c = 0; c < sc.num_value_cols; ++c) {
      if (!d.cols[c].v2_fixed || off > 255) { all_fixed = false; break; }
      d.v2_off[c] = (uint8_t)off;
      off += (uint32_t)d.cols[c].v2_fixed;
      ++nfp;
    }
    d.v2_fixed_len = all_fixed ? off : 0;
    d.v2_nfp = nfp;
    d.v2_tail_off = off;
  }
  d.lower_len = (uint32_t)spec->lower_bound_len;
  d.lower_off = put(spec->lower_bound, spec->lower_bound_len);
  d.upper_len = (uint32_t)spec->upper_bound_len;
  d.upper_off = put(spec->upper_bound, spec->upper_bound_len);
  {
    bool kp = false;
    for (int i = 0; i < spec->num_preds; ++i)
      if (spec->preds[i].is_key_col) kp = true;
    d.need_rowkey = kp || spec->lower_bound_len || spec->upper_bound_len ||
                    spec->emit_rows;
  }
  // fixed rowkey length fast path (doc_key.h:40-63 layout) when no string
  // key columns
  {
    bool fixed = true;
    // 'G' + hash16 + hashed cols + '!' + range cols + '!'  (with hash)
    // range cols + '!'                                     (without)
    uint32_t L = sc.has_hash ? 5u : 1u;
    int nk = (sc.has_hash ? sc.num_hash_cols : 0) + sc.num_range_cols;
    for (int i = 0; i < nk; ++i) {
      if (sc.key_types[i] == YBG_KT_INT64) L += 9;
      else if (sc.key_types[i] == YBG_KT_INT32) L += 5;
      else fixed = false;
    }
    d.fixed_rk_len = fixed ? L : 0;
  }
  *aux_len = pos ? pos : 1;
}

}  // namespace ybgdev
