/*
 * oracle/orcl_group.c — GROUP BY partial aggregates over the scan, built on
 * the row callback (row assembly semantics identical to orcl_scan). Mirrors
 * the per-group accumulate semantics of doc_expr.cc:248-395 applied per
 * group key (config #5). TEST INFRASTRUCTURE ONLY (see orcl.h).
 */
#include "orcl.h"

#include <stdlib.h>
#include <string.h>

typedef struct {
  int used;
  int key_is_null;
  uint64_t key;            /* numeric datum */
  uint8_t *key_bytes;      /* string group key (owned copy) */
  uint32_t key_len;
  int64_t vals[ORCL_MAX_AGGS];
  uint64_t cnts[ORCL_MAX_AGGS];
} grp_slot_t;

typedef struct {
  const orcl_schema_t *sc;
  const orcl_scan_spec_t *spec;
  int group_col;
  grp_slot_t *slots;
  size_t cap; /* power of two; slot cap = NULL-key group */
  int overflow;
} grp_ctx_t;

static uint64_t mix64(uint64_t x) {
  x ^= x >> 33;
  x *= 0xff51afd7ed558ccdull;
  x ^= x >> 33;
  x *= 0xc4ceb9fe1a85ec53ull;
  x ^= x >> 33;
  return x;
}

static uint64_t fnv(const uint8_t *p, uint32_t n) {
  uint64_t h = 1469598103934665603ull;
  for (uint32_t i = 0; i < n; ++i) {
    h ^= p[i];
    h *= 1099511628211ull;
  }
  return h;
}

static grp_slot_t *find_slot(grp_ctx_t *g, int is_null, uint64_t key,
                             const uint8_t *bytes, uint32_t len) {
  if (is_null) {
    grp_slot_t *s = &g->slots[g->cap];
    s->used = 1;
    s->key_is_null = 1;
    return s;
  }
  uint64_t h = bytes ? fnv(bytes, len) : mix64(key);
  for (size_t probe = 0; probe < g->cap; ++probe) {
    grp_slot_t *s = &g->slots[(h + probe) & (g->cap - 1)];
    if (!s->used) {
      s->used = 1;
      if (bytes) {
        s->key_bytes = (uint8_t *)malloc(len ? len : 1);
        memcpy(s->key_bytes, bytes, len);
        s->key_len = len;
      } else {
        s->key = key;
      }
      /* init MIN/MAX */
      for (int a = 0; a < g->spec->num_aggs; ++a) {
        if (g->spec->aggs[a].op == ORCL_AGG_MIN_INT64)
          s->vals[a] = INT64_MAX;
        else if (g->spec->aggs[a].op == ORCL_AGG_MAX_INT64)
          s->vals[a] = INT64_MIN;
      }
      return s;
    }
    if (bytes) {
      if (s->key_bytes && s->key_len == len &&
          memcmp(s->key_bytes, bytes, len) == 0)
        return s;
    } else if (!s->key_bytes && s->key == key) {
      return s;
    }
  }
  g->overflow = 1;
  return NULL;
}

static int grp_cb(const orcl_row_t *row, void *arg) {
  grp_ctx_t *g = (grp_ctx_t *)arg;
  const orcl_schema_t *sc = g->sc;
  int gi = g->group_col;
  int gnull = (row->null_mask >> gi) & 1;
  grp_slot_t *s;
  if (sc->value_cols[gi].dtype == ORCL_T_STRING && !gnull)
    s = find_slot(g, 0, 0, row->strp[gi], row->strlen_[gi]);
  else
    s = find_slot(g, gnull, row->datums[gi], NULL, 0);
  if (!s) return 1; /* overflow: stop */
  for (int a = 0; a < g->spec->num_aggs; ++a) {
    const orcl_agg_t *ag = &g->spec->aggs[a];
    int isnull = ag->op == ORCL_AGG_COUNT_STAR
                     ? 0
                     : (row->null_mask >> ag->col) & 1;
    if (isnull) continue;
    uint64_t d = row->datums[ag->col];
    switch (ag->op) {
      case ORCL_AGG_COUNT_STAR:
      case ORCL_AGG_COUNT:
        s->vals[a] += 1;
        break;
      case ORCL_AGG_SUM_INT64:
        s->vals[a] = (int64_t)((uint64_t)s->vals[a] + d);
        break;
      case ORCL_AGG_SUM_DOUBLE: {
        double v, cur;
        memcpy(&v, &d, 8);
        memcpy(&cur, &s->vals[a], 8);
        cur = (s->cnts[a] == 0) ? v : cur + v;
        memcpy(&s->vals[a], &cur, 8);
        break;
      }
      case ORCL_AGG_MIN_INT64:
        if ((int64_t)d < s->vals[a]) s->vals[a] = (int64_t)d;
        break;
      case ORCL_AGG_MAX_INT64:
        if ((int64_t)d > s->vals[a]) s->vals[a] = (int64_t)d;
        break;
      case ORCL_AGG_MIN_DOUBLE: {
        double v, cur;
        memcpy(&v, &d, 8);
        memcpy(&cur, &s->vals[a], 8);
        if (s->cnts[a] == 0 || v < cur) memcpy(&s->vals[a], &v, 8);
        break;
      }
      case ORCL_AGG_MAX_DOUBLE: {
        double v, cur;
        memcpy(&v, &d, 8);
        memcpy(&cur, &s->vals[a], 8);
        if (s->cnts[a] == 0 || v > cur) memcpy(&s->vals[a], &v, 8);
        break;
      }
      default:
        break;
    }
    s->cnts[a] += 1;
  }
  return 0;
}

/* GROUP BY group_col (value column index). Outputs per group: key datum
 * (numeric; ~0 for the NULL group; strings: (len<<40)|offset into
 * key_bytes_out), vals/cnts [n * ORCL_MAX_AGGS]. Returns 0 ok. */
int orcl_group_scan(const uint8_t *const *blocks, const size_t *sizes,
                    size_t nblocks, orcl_kv_format_t fmt,
                    const orcl_schema_t *schema,
                    const orcl_scan_spec_t *spec, int group_col,
                    uint64_t *keys_out, int64_t *vals_out,
                    uint64_t *cnts_out, uint8_t *key_bytes_out,
                    size_t key_bytes_cap, size_t cap, size_t *n_out) {
  grp_ctx_t g;
  memset(&g, 0, sizeof(g));
  g.sc = schema;
  g.spec = spec;
  g.group_col = group_col;
  g.cap = 1u << 20;
  g.slots = (grp_slot_t *)calloc(g.cap + 1, sizeof(grp_slot_t));
  orcl_scan_result_t res;
  int rc = orcl_scan(blocks, sizes, nblocks, fmt, schema, spec, &res, grp_cb,
                     &g);
  if (rc == 0 && g.overflow) rc = -8;
  size_t n = 0;
  size_t boff = 0;
  if (rc == 0) {
    for (size_t i = 0; i <= g.cap; ++i) {
      grp_slot_t *s = &g.slots[i];
      if (!s->used) continue;
      if (n >= cap) {
        rc = -8;
        break;
      }
      uint64_t kv;
      if (s->key_is_null) {
        kv = ~0ull;
      } else if (s->key_bytes) {
        if (boff + s->key_len > key_bytes_cap) {
          rc = -8;
          break;
        }
        memcpy(key_bytes_out + boff, s->key_bytes, s->key_len);
        kv = ((uint64_t)s->key_len << 40) | boff;
        boff += s->key_len;
      } else {
        kv = s->key;
      }
      keys_out[n] = kv;
      for (int a = 0; a < ORCL_MAX_AGGS; ++a) {
        vals_out[n * ORCL_MAX_AGGS + a] = s->vals[a];
        cnts_out[n * ORCL_MAX_AGGS + a] = s->cnts[a];
      }
      ++n;
    }
  }
  for (size_t i = 0; i <= g.cap; ++i) free(g.slots[i].key_bytes);
  free(g.slots);
  *n_out = n;
  return rc;
}
