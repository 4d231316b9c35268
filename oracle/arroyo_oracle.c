/* CPU restatement of Arroyo's sliding/tumbling window-aggregate operator.
 *
 * TEST INFRASTRUCTURE + CPU BASELINE ONLY.  Only tests/, __graft_entry__'s
 * smoke() and bench.py's cpu_baseline leg may load this library; the product
 * path is the HIP library (arroyo_amd/csrc) and fails loudly if its extension
 * is missing -- it never falls back to this code.
 *
 * Restated, statement for statement, from the reference
 * (ArroyoSystems/arroyo, mounted read-only at /root/reference during the
 * build; citations are file:line into that tree):
 *   sliding: crates/arroyo-worker/src/arrow/sliding_aggregating_window.rs
 *     bin_start :90-99, should_advance :102-113, advance :115-210,
 *     process_batch :598-674 (late drop :631-633)
 *   tumbling: crates/arroyo-worker/src/arrow/tumbling_aggregating_window.rs
 *     process_batch :250-319, handle_watermark :321-392
 *   state table: crates/arroyo-state/src/tables/expiring_time_key_map.rs
 *     :826-929 (flush retention cutoff, exact-timestamp expire, get_min_time);
 *     retention = width (timestamp_table_config, sliding :739-752)
 *   output projection: crates/arroyo-planner/src/extension/aggregate.rs
 *     :292-390 (window=(bin_start, bin_start+width), _timestamp=end-1ns)
 *   partial/final aggregate split: crates/arroyo-planner/src/builder.rs
 *     :135-199 over DataFusion 48.0.1 (ArroyoSystems fork 48.0.1/arroyo,
 *     not vendored in the reference); aggregate semantics pinned by the
 *     reference's golden vectors (tests/golden/, see oracle/gen_golden.py).
 *
 * The reference cannot be compiled here (Rust workspace; no rustc/cargo in
 * the container), so parity is pinned by its golden test vectors instead:
 * tests/test_oracle_golden.py checks this oracle against four of them.
 *
 * Pipeline-level semantics (watermark generation, batching) live in the test
 * harness (arroyo_amd/pipeline.py), not here: this library is the operator.
 */
#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <math.h>
#include <stdio.h>

#include "../include/arroyo_amd_types.h"

#define ORACLE_API __attribute__((visibility("default")))

/* ------------------------------------------------------------------ */
/* aggregate state: per entry, n_aggs x 2 u64 words.
 * COUNT: w0=count.  SUM: w0=sum(i64).  MIN/MAX: w0=value.
 * AVG: w0=count, w1=bits(double sum).                                  */

static inline double bits_to_d(int64_t b) { double d; memcpy(&d, &b, 8); return d; }
static inline int64_t d_to_bits(double d) { int64_t b; memcpy(&b, &d, 8); return b; }

typedef struct {
    int cap;          /* power of two */
    int64_t n;
    int64_t *keys;
    uint8_t *used;
    int64_t *st;      /* [cap][n_aggs][2] */
} Table;

typedef struct {
    AmdWindowConfig cfg;
    /* open bins (execs) and closed panes (TieredRecordBatchHolder),
     * sorted by bin */
    int n_open, cap_open;
    uint64_t *open_bins; Table **open_tables;
    int n_closed, cap_closed;
    uint64_t *closed_bins; Table **closed_tables;
    /* state-table bins (ExpiringTimeKeyView keys), sorted */
    int n_table, cap_table;
    uint64_t *table_bins;
    /* sliding state machine (sliding_aggregating_window.rs:63-73) */
    int state;               /* 0 NoData, 1 OnlyBufferedData, 2 InMemoryData */
    uint64_t earliest, next;
    int has_wm; uint64_t wm;  /* last present watermark */
    /* output accumulator */
    int64_t out_rows, out_cap;
    int out_cols;
    int64_t **out;            /* [out_cols][out_cap] */
    char err[256];
} Op;

static uint64_t hash64(uint64_t x) {
    /* splitmix64 finalizer */
    x += 0x9e3779b97f4a7c15ULL;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
    return x ^ (x >> 31);
}

/* internal key width: tuples of up to 4 i64 key columns (ArroyoSchema
 * key_indices, crates/arroyo-rpc/src/df.rs:24-30); unkeyed ops store one
 * zero word */
#define NKI(c) ((c)->n_keys > 0 ? (c)->n_keys : 1)

static Table *table_new(int cap_log2, int n_aggs, int nk) {
    Table *t = calloc(1, sizeof(Table));
    t->cap = 1 << cap_log2;
    t->keys = malloc((size_t)t->cap * nk * 8);
    t->used = calloc((size_t)t->cap, 1);
    t->st = malloc((size_t)t->cap * n_aggs * 16);
    return t;
}

static void table_free(Table *t) {
    if (!t) return;
    free(t->keys); free(t->used); free(t->st); free(t);
}

/* value column of agg i is f64 (bit patterns in the i64 plane) */
static int agg_isf(const AmdWindowConfig *c, int i) {
    return c->agg_col[i] >= 0 && c->val_is_f64[c->agg_col[i]];
}

static void st_init(const AmdWindowConfig *c, int64_t *s) {
    for (int i = 0; i < c->n_aggs; i++) {
        int f = agg_isf(c, i);
        switch (c->agg_ops[i]) {
        case AMD_AGG_COUNT: s[2*i] = 0; break;
        case AMD_AGG_SUM:   s[2*i] = f ? d_to_bits(0.0) : 0; break;
        case AMD_AGG_MIN:
            s[2*i] = f ? d_to_bits(INFINITY) : INT64_MAX;
            break;
        case AMD_AGG_MAX:
            s[2*i] = f ? d_to_bits(-INFINITY) : INT64_MIN;
            break;
        case AMD_AGG_AVG:   s[2*i] = 0; s[2*i+1] = d_to_bits(0.0); break;
        }
    }
}

static void table_grow(const AmdWindowConfig *c, Table *t);

static uint64_t hash_tuple(const int64_t *key, int nk) {
    uint64_t h = hash64((uint64_t)key[0]);
    for (int k = 1; k < nk; k++)
        h = hash64(h ^ (uint64_t)key[k]);
    return h;
}

static int tuple_eq(const int64_t *a, const int64_t *b, int nk) {
    for (int k = 0; k < nk; k++)
        if (a[k] != b[k]) return 0;
    return 1;
}

static int64_t *table_slot(const AmdWindowConfig *c, Table *t,
                           const int64_t *key) {
    int nk = NKI(c);
    if (t->n * 10 >= (int64_t)t->cap * 7) table_grow(c, t);
    uint64_t m = (uint64_t)t->cap - 1;
    uint64_t i = hash_tuple(key, nk) & m;
    while (t->used[i] && !tuple_eq(&t->keys[i * nk], key, nk))
        i = (i + 1) & m;
    if (!t->used[i]) {
        t->used[i] = 1;
        memcpy(&t->keys[i * nk], key, (size_t)nk * 8);
        t->n++;
        st_init(c, &t->st[i * c->n_aggs * 2]);
    }
    return &t->st[i * c->n_aggs * 2];
}

static void table_grow(const AmdWindowConfig *c, Table *t) {
    int nk = NKI(c);
    int ocap = t->cap;
    int64_t *ok = t->keys; uint8_t *ou = t->used; int64_t *os = t->st;
    t->cap <<= 1; t->n = 0;
    t->keys = malloc((size_t)t->cap * nk * 8);
    t->used = calloc((size_t)t->cap, 1);
    t->st = malloc((size_t)t->cap * c->n_aggs * 16);
    for (int i = 0; i < ocap; i++) {
        if (!ou[i]) continue;
        int64_t *s = table_slot(c, t, &ok[(size_t)i * nk]);
        memcpy(s, &os[(size_t)i * c->n_aggs * 2], (size_t)c->n_aggs * 16);
        /* table_slot counted it as new and re-inited; restore by copy above */
    }
    free(ok); free(ou); free(os);
}

static void st_update(const AmdWindowConfig *c, int64_t *s,
                      const int64_t *const *vcols, int64_t row) {
    for (int i = 0; i < c->n_aggs; i++) {
        int col = c->agg_col[i];
        int f = agg_isf(c, i);
        switch (c->agg_ops[i]) {
        case AMD_AGG_COUNT: s[2*i]++; break;
        case AMD_AGG_SUM:
            if (f)
                s[2*i] = d_to_bits(bits_to_d(s[2*i]) +
                                   bits_to_d(vcols[col][row]));
            else
                s[2*i] += vcols[col][row];
            break;
        case AMD_AGG_MIN: {
            int64_t v = vcols[col][row];
            if (f ? bits_to_d(v) < bits_to_d(s[2*i]) : v < s[2*i])
                s[2*i] = v;
        } break;
        case AMD_AGG_MAX: {
            int64_t v = vcols[col][row];
            if (f ? bits_to_d(v) > bits_to_d(s[2*i]) : v > s[2*i])
                s[2*i] = v;
        } break;
        case AMD_AGG_AVG:
            s[2*i]++;
            s[2*i+1] = d_to_bits(bits_to_d(s[2*i+1]) +
                                 (f ? bits_to_d(vcols[col][row])
                                    : (double)vcols[col][row]));
            break;
        }
    }
}

static void st_merge(const AmdWindowConfig *c, int64_t *a, const int64_t *b) {
    for (int i = 0; i < c->n_aggs; i++) {
        int f = agg_isf(c, i);
        switch (c->agg_ops[i]) {
        case AMD_AGG_COUNT: a[2*i] += b[2*i]; break;
        case AMD_AGG_SUM:
            if (f)
                a[2*i] = d_to_bits(bits_to_d(a[2*i]) + bits_to_d(b[2*i]));
            else
                a[2*i] += b[2*i];
            break;
        case AMD_AGG_MIN:
            if (f ? bits_to_d(b[2*i]) < bits_to_d(a[2*i]) : b[2*i] < a[2*i])
                a[2*i] = b[2*i];
            break;
        case AMD_AGG_MAX:
            if (f ? bits_to_d(b[2*i]) > bits_to_d(a[2*i]) : b[2*i] > a[2*i])
                a[2*i] = b[2*i];
            break;
        case AMD_AGG_AVG:
            a[2*i] += b[2*i];
            a[2*i+1] = d_to_bits(bits_to_d(a[2*i+1]) + bits_to_d(b[2*i+1]));
            break;
        }
    }
}

/* sorted (bins,tables) vector helpers */
static int vec_find(const uint64_t *bins, int n, uint64_t b) {
    int lo = 0, hi = n;
    while (lo < hi) { int mid = (lo + hi) / 2; if (bins[mid] < b) lo = mid + 1; else hi = mid; }
    return lo; /* insertion point; bins[lo]==b if present */
}

static Table *panes_get(int *n, int *cap, uint64_t **bins, Table ***tables,
                        uint64_t b, const AmdWindowConfig *c, int create) {
    int i = vec_find(*bins, *n, b);
    if (i < *n && (*bins)[i] == b) return (*tables)[i];
    if (!create) return NULL;
    if (*n == *cap) {
        *cap = *cap ? *cap * 2 : 16;
        *bins = realloc(*bins, (size_t)*cap * 8);
        *tables = realloc(*tables, (size_t)*cap * sizeof(Table *));
    }
    memmove(*bins + i + 1, *bins + i, (size_t)(*n - i) * 8);
    memmove(*tables + i + 1, *tables + i, (size_t)(*n - i) * sizeof(Table *));
    (*bins)[i] = b;
    Table *t = table_new(8, c->n_aggs, NKI(c));
    (*tables)[i] = t;
    (*n)++;
    return t;
}

static Table *panes_remove(int *n, uint64_t *bins, Table **tables, uint64_t b) {
    int i = vec_find(bins, *n, b);
    if (i >= *n || bins[i] != b) return NULL;
    Table *t = tables[i];
    memmove(bins + i, bins + i, 0);
    memmove(bins + i, bins + i + 1, (size_t)(*n - i - 1) * 8);
    memmove(tables + i, tables + i + 1, (size_t)(*n - i - 1) * sizeof(Table *));
    (*n)--;
    return t;
}

static void tset_add(Op *o, uint64_t b) {
    int i = vec_find(o->table_bins, o->n_table, b);
    if (i < o->n_table && o->table_bins[i] == b) return;
    if (o->n_table == o->cap_table) {
        o->cap_table = o->cap_table ? o->cap_table * 2 : 16;
        o->table_bins = realloc(o->table_bins, (size_t)o->cap_table * 8);
    }
    memmove(o->table_bins + i + 1, o->table_bins + i, (size_t)(o->n_table - i) * 8);
    o->table_bins[i] = b;
    o->n_table++;
}

static void tset_remove_exact(Op *o, uint64_t b) {
    int i = vec_find(o->table_bins, o->n_table, b);
    if (i < o->n_table && o->table_bins[i] == b) {
        memmove(o->table_bins + i, o->table_bins + i + 1,
                (size_t)(o->n_table - i - 1) * 8);
        o->n_table--;
    }
}

static void tset_retain_ge(Op *o, uint64_t cutoff) {
    int i = vec_find(o->table_bins, o->n_table, cutoff);
    if (i > 0) {
        memmove(o->table_bins, o->table_bins + i, (size_t)(o->n_table - i) * 8);
        o->n_table -= i;
    }
}

static inline uint64_t bin_of(uint64_t ts, uint64_t w) {
    return w ? ts - ts % w : ts;
}

/* output accumulation: columns [key?, aggs..., win_start, win_end, _ts] */
static void out_reserve(Op *o, int64_t add) {
    if (o->out_rows + add <= o->out_cap) return;
    int64_t ncap = o->out_cap ? o->out_cap : 1024;
    while (ncap < o->out_rows + add) ncap *= 2;
    for (int i = 0; i < o->out_cols; i++)
        o->out[i] = realloc(o->out[i], (size_t)ncap * 8);
    o->out_cap = ncap;
}

static void emit_table(Op *o, Table *t, uint64_t ws, uint64_t we) {
    const AmdWindowConfig *c = &o->cfg;
    out_reserve(o, t->n);
    for (int i = 0; i < t->cap; i++) {
        if (!t->used[i]) continue;
        int64_t r = o->out_rows++;
        int col = 0;
        for (int kk = 0; kk < c->n_keys; kk++)
            o->out[col++][r] = t->keys[(size_t)i * NKI(c) + kk];
        const int64_t *s = &t->st[(size_t)i * c->n_aggs * 2];
        for (int a = 0; a < c->n_aggs; a++) {
            if (c->agg_ops[a] == AMD_AGG_AVG) {
                double v = s[2*a] ? bits_to_d(s[2*a+1]) / (double)s[2*a] : 0.0;
                o->out[col++][r] = d_to_bits(v);
            } else {
                o->out[col++][r] = s[2*a];
            }
        }
        o->out[col++][r] = (int64_t)ws;
        o->out[col++][r] = (int64_t)we;
        o->out[col++][r] = (int64_t)(we - 1);
    }
}

/* advance(): sliding_aggregating_window.rs:115-210 */
static void advance(Op *o) {
    const AmdWindowConfig *c = &o->cfg;
    uint64_t b = (o->state == 1) ? o->earliest : o->next;
    uint64_t E = b + c->slide_nanos;

    /* partial_table.flush(Some(bin_end)): retention(=width) cutoff :131 */
    if (E >= c->width_nanos) tset_retain_ge(o, E - c->width_nanos);

    Table *pane = panes_remove(&o->n_open, o->open_bins, o->open_tables, b);
    if (pane) {
        Table *tgt = panes_get(&o->n_closed, &o->cap_closed, &o->closed_bins,
                               &o->closed_tables, b, c, 0);
        if (!tgt) {
            /* move the pane wholesale */
            int i = vec_find(o->closed_bins, o->n_closed, b);
            if (o->n_closed == o->cap_closed) {
                o->cap_closed = o->cap_closed ? o->cap_closed * 2 : 16;
                o->closed_bins = realloc(o->closed_bins, (size_t)o->cap_closed * 8);
                o->closed_tables = realloc(o->closed_tables,
                                           (size_t)o->cap_closed * sizeof(Table *));
            }
            memmove(o->closed_bins + i + 1, o->closed_bins + i,
                    (size_t)(o->n_closed - i) * 8);
            memmove(o->closed_tables + i + 1, o->closed_tables + i,
                    (size_t)(o->n_closed - i) * sizeof(Table *));
            o->closed_bins[i] = b;
            o->closed_tables[i] = pane;
            o->n_closed++;
        } else {
            for (int i = 0; i < pane->cap; i++)
                if (pane->used[i])
                    st_merge(c, table_slot(c, tgt,
                                           &pane->keys[(size_t)i * NKI(c)]),
                             &pane->st[(size_t)i * c->n_aggs * 2]);
            table_free(pane);
        }
        tset_add(o, b); /* partial_table.insert :151 */
    }
    /* expire_timestamp(bin_end - width + slide), exact key :160 */
    if (E + c->slide_nanos >= c->width_nanos)
        tset_remove_exact(o, E + c->slide_nanos - c->width_nanos);

    /* merge closed panes in [E-width, E) :161-196 */
    uint64_t lo = (E >= c->width_nanos) ? E - c->width_nanos : 0;
    Table *merged = table_new(8, c->n_aggs, NKI(c));
    for (int p = 0; p < o->n_closed; p++) {
        if (o->closed_bins[p] < lo || o->closed_bins[p] >= E) continue;
        Table *t = o->closed_tables[p];
        for (int i = 0; i < t->cap; i++)
            if (t->used[i])
                st_merge(c, table_slot(c, merged,
                                       &t->keys[(size_t)i * NKI(c)]),
                         &t->st[(size_t)i * c->n_aggs * 2]);
    }
    /* delete_before(bin_end + slide - width) :173-174 */
    uint64_t del = (E + c->slide_nanos >= c->width_nanos)
                       ? E + c->slide_nanos - c->width_nanos : 0;
    int keep = 0;
    for (int p = 0; p < o->n_closed; p++) {
        if (o->closed_bins[p] >= del) {
            o->closed_bins[keep] = o->closed_bins[p];
            o->closed_tables[keep] = o->closed_tables[p];
            keep++;
        } else {
            table_free(o->closed_tables[p]);
        }
    }
    o->n_closed = keep;

    emit_table(o, merged, E - c->width_nanos, E);
    table_free(merged);

    /* state transition :176-187 */
    if (o->n_closed == 0) {
        if (o->n_table > 0) {
            o->state = 1;
            o->earliest = bin_of(o->table_bins[0], c->slide_nanos);
        } else {
            o->state = 0;
        }
    } else {
        o->state = 2;
        o->next = E;
    }
}

/* ------------------------------------------------------------------ */
ORACLE_API void *oracle_create(const AmdWindowConfig *cfg) {
    if (!cfg || cfg->n_aggs < 1 || cfg->n_aggs > AMD_MAX_AGGS ||
        cfg->n_keys < 0 || cfg->n_keys > 4 || cfg->width_nanos == 0)
        return NULL;
    /* the reference's planner rejects non-divisible hop parameters
     * ("hop() width currently must be a multiple of slide",
     * arroyo-planner/src/lib.rs:644); the operator's behavior there is
     * unreachable (and its earliest-bin state machine would not
     * terminate), so reject them here too */
    if (!cfg->is_tumbling && cfg->slide_nanos != 0 &&
        cfg->width_nanos % cfg->slide_nanos != 0)
        return NULL;
    if (!cfg->is_tumbling && cfg->slide_nanos == 0)
        return NULL;
    Op *o = calloc(1, sizeof(Op));
    o->cfg = *cfg;
    /* hop(x, x) is a tumble window (arroyo-planner/src/lib.rs:649-651) */
    if (!o->cfg.is_tumbling && o->cfg.slide_nanos == o->cfg.width_nanos)
        o->cfg.is_tumbling = 1;
    if (o->cfg.is_tumbling) o->cfg.slide_nanos = o->cfg.width_nanos;
    o->out_cols = cfg->n_keys + cfg->n_aggs + 3;
    o->out = calloc((size_t)o->out_cols, sizeof(int64_t *));
    return o;
}

ORACLE_API const char *oracle_last_error(void *h) {
    return h ? ((Op *)h)->err : "null handle";
}

ORACLE_API int oracle_process_batch(void *h, const int64_t *const *cols,
                                    int32_t n_cols, int64_t n_rows) {
    Op *o = h;
    const AmdWindowConfig *c = &o->cfg;
    if (n_cols != c->n_keys + c->n_value_cols + 1) {
        snprintf(o->err, sizeof o->err, "expected %d cols, got %d",
                 c->n_keys + c->n_value_cols + 1, n_cols);
        return 1;
    }
    const int64_t *const *vcols = cols + c->n_keys;
    const int64_t *ts = cols[n_cols - 1];
    uint64_t wmb = o->has_wm ? bin_of(o->wm, c->slide_nanos) : 0;
    uint64_t cached_bin = UINT64_MAX; Table *cached = NULL;
    for (int64_t r = 0; r < n_rows; r++) {
        uint64_t b = bin_of((uint64_t)ts[r], c->slide_nanos);
        if (o->has_wm && b < wmb) continue; /* late drop */
        if (!c->is_tumbling) {
            if (o->state == 0) { o->state = 1; o->earliest = b; }
            else if (o->state == 1 && b < o->earliest) o->earliest = b;
        }
        if (b != cached_bin) {
            cached = panes_get(&o->n_open, &o->cap_open, &o->open_bins,
                               &o->open_tables, b, c, 1);
            cached_bin = b;
        }
        int64_t key[4] = {0, 0, 0, 0};
        for (int kk = 0; kk < c->n_keys; kk++) key[kk] = cols[kk][r];
        st_update(c, table_slot(c, cached, key), vcols, r);
    }
    return 0;
}

ORACLE_API int oracle_handle_watermark(void *h, uint64_t wm, AmdOutBatch *out) {
    Op *o = h;
    const AmdWindowConfig *c = &o->cfg;
    o->has_wm = 1;
    o->wm = wm;
    if (c->is_tumbling) {
        /* tumbling_aggregating_window.rs:321-392 */
        uint64_t wb = bin_of(wm, c->width_nanos);
        while (o->n_open > 0 && o->open_bins[0] < wb) {
            uint64_t b = o->open_bins[0];
            Table *pane = panes_remove(&o->n_open, o->open_bins, o->open_tables, b);
            emit_table(o, pane, b, b + c->width_nanos);
            table_free(pane);
        }
    } else {
        while (o->state != 0) {
            uint64_t base = (o->state == 1) ? o->earliest : o->next;
            if (!(base + c->slide_nanos <= bin_of(wm, c->slide_nanos))) break;
            advance(o);
        }
    }
    if (out) {
        memset(out, 0, sizeof *out);
        out->n_rows = o->out_rows;
        out->n_cols = o->out_cols;
        out->cols = calloc((size_t)o->out_cols, sizeof(void *));
        out->is_f64 = calloc((size_t)o->out_cols, sizeof(int32_t));
        for (int i = 0; i < o->out_cols; i++) {
            out->cols[i] = malloc((size_t)(o->out_rows ? o->out_rows : 1) * 8);
            if (o->out_rows)
                memcpy(out->cols[i], o->out[i], (size_t)o->out_rows * 8);
        }
        for (int a = 0; a < c->n_aggs; a++)
            if (c->agg_ops[a] == AMD_AGG_AVG || agg_isf(c, a))
                out->is_f64[c->n_keys + a] = 1;
        o->out_rows = 0;
    }
    return 0;
}

/* handle_checkpoint (sliding :693-737): drain open bins' partial states into
 * the state table.  Output columns: [key?, partial state words
 * (1 per agg, 2 for AVG)..., bin _timestamp]. */
ORACLE_API int oracle_checkpoint_drain(void *h, AmdOutBatch *out) {
    Op *o = h;
    const AmdWindowConfig *c = &o->cfg;
    int swords = 0;
    for (int a = 0; a < c->n_aggs; a++)
        swords += (c->agg_ops[a] == AMD_AGG_AVG) ? 2 : 1;
    int ncols = c->n_keys + swords + 1;
    int64_t total = 0;
    for (int p = 0; p < o->n_open; p++) total += o->open_tables[p]->n;
    memset(out, 0, sizeof *out);
    out->n_rows = total;
    out->n_cols = ncols;
    out->cols = calloc((size_t)ncols, sizeof(void *));
    out->is_f64 = calloc((size_t)ncols, sizeof(int32_t));
    for (int i = 0; i < ncols; i++)
        out->cols[i] = malloc((size_t)(total ? total : 1) * 8);
    {
        int col = c->n_keys;
        for (int a = 0; a < c->n_aggs; a++) {
            if (c->agg_ops[a] == AMD_AGG_AVG) { out->is_f64[col + 1] = 1; col += 2; }
            else { if (agg_isf(c, a)) out->is_f64[col] = 1; col += 1; }
        }
    }
    int64_t r = 0;
    for (int p = 0; p < o->n_open; p++) {
        Table *t = o->open_tables[p];
        uint64_t b = o->open_bins[p];
        tset_add(o, b);
        for (int i = 0; i < t->cap; i++) {
            if (!t->used[i]) continue;
            int col = 0;
            for (int kk = 0; kk < c->n_keys; kk++)
                ((int64_t *)out->cols[col++])[r] =
                    t->keys[(size_t)i * NKI(c) + kk];
            const int64_t *s = &t->st[(size_t)i * c->n_aggs * 2];
            for (int a = 0; a < c->n_aggs; a++) {
                ((int64_t *)out->cols[col++])[r] = s[2*a];
                if (c->agg_ops[a] == AMD_AGG_AVG)
                    ((int64_t *)out->cols[col++])[r] = s[2*a+1];
            }
            ((int64_t *)out->cols[col])[r] = (int64_t)b;
            r++;
        }
    }
    return 0;
}

ORACLE_API void oracle_free_out(AmdOutBatch *out) {
    if (!out) return;
    for (int i = 0; i < out->n_cols; i++) free(out->cols[i]);
    free(out->cols);
    free(out->is_f64);
    memset(out, 0, sizeof *out);
}

ORACLE_API void oracle_destroy(void *h) {
    Op *o = h;
    if (!o) return;
    for (int p = 0; p < o->n_open; p++) table_free(o->open_tables[p]);
    for (int p = 0; p < o->n_closed; p++) table_free(o->closed_tables[p]);
    free(o->open_bins); free(o->open_tables);
    free(o->closed_bins); free(o->closed_tables);
    free(o->table_bins);
    for (int i = 0; i < o->out_cols; i++) free(o->out[i]);
    free(o->out);
    free(o);
}

/* ================================================================== */
/* Instant (windowed stream-stream) join oracle.
 *
 * Restated from crates/arroyo-worker/src/arrow/instant_join.rs:
 *   - rows are routed to a per-exact-timestamp exec (process_side :109-172:
 *     batches spanning several instants are sort+partitioned; we route
 *     row-by-row, same result);
 *   - a batch whose min timestamp is behind the watermark is a fatal error
 *     (:129-139 panics);
 *   - on watermark, every instant < watermark fires in timestamp order
 *     (BTreeMap pop_first loop, :265-281) and emits the join of its two
 *     sides.  The join itself is the reference's HashJoinExec built on the
 *     left side and probed with the right (LockedJoinPair,
 *     arroyo-planner/src/physical.rs:177-268); here: equi-join on the i64
 *     key, or the cross product when n_keys == 0 (join on the instant
 *     itself, e.g. windowed_inner_join.sql ON a.window = b.window);
 *   - checkpoint drains each side's buffered rows (handle_checkpoint
 *     :285-303 flushes the left/right ExpiringTimeKeyTables);
 *   - restore re-processes the drained batches (on_start :205-230).
 */

typedef struct {
    int64_t cap, n;
    int64_t **cols;          /* [n_cols][cap]; n_cols = n_keys + n_vals */
    int n_cols;
} RowBuf;

typedef struct {
    uint64_t instant;
    RowBuf l, r;
} JInstant;

typedef struct {
    AmdJoinConfig cfg;
    int n_inst, cap_inst;
    JInstant *inst;          /* sorted by instant */
    int has_wm; uint64_t wm;
    int out_cols;
    int64_t out_rows, out_cap;
    int64_t **out;
    char err[256];
} JOp;

static void rowbuf_init(RowBuf *b, int n_cols) {
    memset(b, 0, sizeof *b);
    b->n_cols = n_cols;
    b->cols = calloc((size_t)n_cols, sizeof(int64_t *));
}

static void rowbuf_free(RowBuf *b) {
    for (int c = 0; c < b->n_cols; c++) free(b->cols[c]);
    free(b->cols);
}

static void rowbuf_push(RowBuf *b, const int64_t *vals) {
    if (b->n == b->cap) {
        b->cap = b->cap ? b->cap * 2 : 64;
        for (int c = 0; c < b->n_cols; c++)
            b->cols[c] = realloc(b->cols[c], (size_t)b->cap * 8);
    }
    for (int c = 0; c < b->n_cols; c++) b->cols[c][b->n] = vals[c];
    b->n++;
}

static JInstant *jinstant_get(JOp *o, uint64_t t) {
    int lo = 0, hi = o->n_inst;
    while (lo < hi) { int m = (lo + hi) / 2; if (o->inst[m].instant < t) lo = m + 1; else hi = m; }
    if (lo < o->n_inst && o->inst[lo].instant == t) return &o->inst[lo];
    if (o->n_inst == o->cap_inst) {
        o->cap_inst = o->cap_inst ? o->cap_inst * 2 : 16;
        o->inst = realloc(o->inst, (size_t)o->cap_inst * sizeof(JInstant));
    }
    memmove(o->inst + lo + 1, o->inst + lo,
            (size_t)(o->n_inst - lo) * sizeof(JInstant));
    o->inst[lo].instant = t;
    rowbuf_init(&o->inst[lo].l, o->cfg.n_keys + o->cfg.n_left_vals);
    rowbuf_init(&o->inst[lo].r, o->cfg.n_keys + o->cfg.n_right_vals);
    o->n_inst++;
    return &o->inst[lo];
}

ORACLE_API void *oracle_join_create(const AmdJoinConfig *cfg) {
    if (!cfg || cfg->n_keys < 0 || cfg->n_keys > 1 ||
        cfg->n_left_vals < 0 || cfg->n_right_vals < 0 ||
        cfg->join_type < 0 || cfg->join_type > AMD_JOIN_FULL)
        return NULL;
    JOp *o = calloc(1, sizeof(JOp));
    o->cfg = *cfg;
    /* non-inner joins append [left_present, right_present] columns */
    o->out_cols = cfg->n_keys + cfg->n_left_vals + cfg->n_right_vals + 1 +
                  (cfg->join_type != AMD_JOIN_INNER ? 2 : 0);
    o->out = calloc((size_t)o->out_cols, sizeof(int64_t *));
    return o;
}

ORACLE_API const char *oracle_join_last_error(void *h) {
    return h ? ((JOp *)h)->err : "null handle";
}

ORACLE_API int oracle_join_process_batch(void *h, int32_t side,
                                         const int64_t *const *cols,
                                         int32_t n_cols, int64_t n_rows) {
    JOp *o = h;
    const AmdJoinConfig *c = &o->cfg;
    int nv = side == 0 ? c->n_left_vals : c->n_right_vals;
    int want = c->n_keys + nv + 1;
    if (n_cols != want) {
        snprintf(o->err, sizeof o->err, "side %d expects %d cols, got %d",
                 side, want, n_cols);
        return 1;
    }
    const int64_t *ts = cols[n_cols - 1];
    int64_t tmp[64];
    for (int64_t r = 0; r < n_rows; r++) {
        if (o->has_wm && (uint64_t)ts[r] < o->wm) {
            /* instant_join.rs:129-139 panics on pre-watermark data */
            snprintf(o->err, sizeof o->err,
                     "batch with timestamp %lld before the watermark %llu",
                     (long long)ts[r], (unsigned long long)o->wm);
            return 1;
        }
        JInstant *in = jinstant_get(o, (uint64_t)ts[r]);
        RowBuf *b = side == 0 ? &in->l : &in->r;
        for (int cidx = 0; cidx < n_cols - 1; cidx++) tmp[cidx] = cols[cidx][r];
        rowbuf_push(b, tmp);
    }
    return 0;
}

static void jout_reserve(JOp *o, int64_t add) {
    if (o->out_rows + add <= o->out_cap) return;
    int64_t ncap = o->out_cap ? o->out_cap : 1024;
    while (ncap < o->out_rows + add) ncap *= 2;
    for (int i = 0; i < o->out_cols; i++)
        o->out[i] = realloc(o->out[i], (size_t)ncap * 8);
    o->out_cap = ncap;
}

/* li/ri == -1: that side is absent (outer join); its value columns are
 * zero-filled and its presence flag 0 */
static void jemit(JOp *o, const JInstant *in, int64_t li, int64_t ri) {
    const AmdJoinConfig *c = &o->cfg;
    jout_reserve(o, 1);
    int64_t r = o->out_rows++;
    int col = 0;
    if (c->n_keys)
        o->out[col++][r] = li >= 0 ? in->l.cols[0][li] : in->r.cols[0][ri];
    for (int v = 0; v < c->n_left_vals; v++)
        o->out[col++][r] = li >= 0 ? in->l.cols[c->n_keys + v][li] : 0;
    for (int v = 0; v < c->n_right_vals; v++)
        o->out[col++][r] = ri >= 0 ? in->r.cols[c->n_keys + v][ri] : 0;
    o->out[col++][r] = (int64_t)in->instant;
    if (c->join_type != AMD_JOIN_INNER) {
        o->out[col++][r] = li >= 0;
        o->out[col][r] = ri >= 0;
    }
}

static void fire_instant(JOp *o, JInstant *in) {
    const AmdJoinConfig *c = &o->cfg;
    int jt = c->join_type;
    int emit_l = jt == AMD_JOIN_LEFT || jt == AMD_JOIN_FULL;
    int emit_r = jt == AMD_JOIN_RIGHT || jt == AMD_JOIN_FULL;
    if (c->n_keys == 0) {
        /* window-condition join: cross product; a row is unmatched only
         * when the other side of the instant is empty */
        if (in->l.n && in->r.n) {
            for (int64_t li = 0; li < in->l.n; li++)
                for (int64_t ri = 0; ri < in->r.n; ri++)
                    jemit(o, in, li, ri);
        } else if (in->l.n && emit_l) {
            for (int64_t li = 0; li < in->l.n; li++) jemit(o, in, li, -1);
        } else if (in->r.n && emit_r) {
            for (int64_t ri = 0; ri < in->r.n; ri++) jemit(o, in, -1, ri);
        }
        return;
    }
    /* hash multimap over the left (build) side, probe with the right:
     * HashJoinExec via LockedJoinPair, planner/physical.rs:177-268;
     * JoinType Left/Right/Full additionally emits null-padded rows for
     * the unmatched side (plan/join.rs join_type passthrough) */
    if (in->l.n == 0 || in->r.n == 0) {
        if (in->l.n && emit_l)
            for (int64_t li = 0; li < in->l.n; li++) jemit(o, in, li, -1);
        if (in->r.n && emit_r)
            for (int64_t ri = 0; ri < in->r.n; ri++) jemit(o, in, -1, ri);
        return;
    }
    int64_t H = 64;
    while (H < in->l.n * 2) H <<= 1;
    int64_t *head = malloc((size_t)H * 8);
    int64_t *next = malloc((size_t)in->l.n * 8);
    char *l_hit = calloc((size_t)in->l.n, 1);
    for (int64_t i = 0; i < H; i++) head[i] = -1;
    for (int64_t li = 0; li < in->l.n; li++) {
        uint64_t s = hash64((uint64_t)in->l.cols[0][li]) & (uint64_t)(H - 1);
        next[li] = head[s];
        head[s] = li;
    }
    for (int64_t ri = 0; ri < in->r.n; ri++) {
        int64_t key = in->r.cols[0][ri];
        uint64_t s = hash64((uint64_t)key) & (uint64_t)(H - 1);
        int hit = 0;
        for (int64_t li = head[s]; li >= 0; li = next[li])
            if (in->l.cols[0][li] == key) {
                jemit(o, in, li, ri);
                l_hit[li] = 1;
                hit = 1;
            }
        if (!hit && emit_r) jemit(o, in, -1, ri);
    }
    if (emit_l)
        for (int64_t li = 0; li < in->l.n; li++)
            if (!l_hit[li]) jemit(o, in, li, -1);
    free(head);
    free(next);
    free(l_hit);
}

static void jbuild_out(JOp *o, AmdOutBatch *out) {
    memset(out, 0, sizeof *out);
    out->n_rows = o->out_rows;
    out->n_cols = o->out_cols;
    out->cols = calloc((size_t)o->out_cols, sizeof(void *));
    out->is_f64 = calloc((size_t)o->out_cols, sizeof(int32_t));
    for (int i = 0; i < o->out_cols; i++) {
        out->cols[i] = malloc((size_t)(o->out_rows ? o->out_rows : 1) * 8);
        if (o->out_rows)
            memcpy(out->cols[i], o->out[i], (size_t)o->out_rows * 8);
    }
    o->out_rows = 0;
}

ORACLE_API int oracle_join_handle_watermark(void *h, uint64_t wm,
                                            AmdOutBatch *out) {
    JOp *o = h;
    o->has_wm = 1;
    o->wm = wm;
    int fired = 0;
    while (fired < o->n_inst && o->inst[fired].instant < wm) {
        fire_instant(o, &o->inst[fired]);
        rowbuf_free(&o->inst[fired].l);
        rowbuf_free(&o->inst[fired].r);
        fired++;
    }
    if (fired) {
        memmove(o->inst, o->inst + fired,
                (size_t)(o->n_inst - fired) * sizeof(JInstant));
        o->n_inst -= fired;
    }
    if (out) jbuild_out(o, out);
    return 0;
}

/* drain one side's buffered rows: [key?, vals..., _timestamp] */
ORACLE_API int oracle_join_checkpoint_drain(void *h, int32_t side,
                                            AmdOutBatch *out) {
    JOp *o = h;
    const AmdJoinConfig *c = &o->cfg;
    int nv = side == 0 ? c->n_left_vals : c->n_right_vals;
    int ncols = c->n_keys + nv + 1;
    int64_t total = 0;
    for (int i = 0; i < o->n_inst; i++)
        total += side == 0 ? o->inst[i].l.n : o->inst[i].r.n;
    memset(out, 0, sizeof *out);
    out->n_rows = total;
    out->n_cols = ncols;
    out->cols = calloc((size_t)ncols, sizeof(void *));
    out->is_f64 = calloc((size_t)ncols, sizeof(int32_t));
    for (int i = 0; i < ncols; i++)
        out->cols[i] = malloc((size_t)(total ? total : 1) * 8);
    int64_t r = 0;
    for (int i = 0; i < o->n_inst; i++) {
        RowBuf *b = side == 0 ? &o->inst[i].l : &o->inst[i].r;
        for (int64_t j = 0; j < b->n; j++) {
            for (int cidx = 0; cidx < ncols - 1; cidx++)
                ((int64_t *)out->cols[cidx])[r] = b->cols[cidx][j];
            ((int64_t *)out->cols[ncols - 1])[r] = (int64_t)o->inst[i].instant;
            r++;
        }
    }
    return 0;
}

ORACLE_API void oracle_join_destroy(void *h) {
    JOp *o = h;
    if (!o) return;
    for (int i = 0; i < o->n_inst; i++) {
        rowbuf_free(&o->inst[i].l);
        rowbuf_free(&o->inst[i].r);
    }
    free(o->inst);
    for (int i = 0; i < o->out_cols; i++) free(o->out[i]);
    free(o->out);
    free(o);
}

/* ================================================================== */
/* Session (gap) window aggregate oracle.
 *
 * Restated from crates/arroyo-worker/src/arrow/session_aggregating_window.rs:
 *   - process_batch :849-893: rows with ts < last watermark are filtered out
 *     (gt_eq keeps ts >= watermark); the rest is buffered per key;
 *   - session formation (ActiveSession::add_batch :424-495 +
 *     fill_active_session :610-645): over the key's time-sorted rows, a row
 *     joins the current session iff ts < data_end + gap (STRICT); data_end
 *     is the running max;
 *   - firing (KeyComputingHolder::watermark_update :559-608, driven from
 *     handle_watermark -> advance :76-98): sessions fire, oldest first,
 *     while data_end + gap < watermark (STRICT);
 *   - output (to_record_batch :316-380): [key?, final aggregates,
 *     window_start = data_start, window_end = data_end + gap,
 *     _timestamp = window_end - 1].
 */

typedef struct {
    int64_t cap, n;
    int64_t **vals;           /* [n_value_cols][cap] */
    int64_t *ts;              /* [cap] */
} SRows;

typedef struct {
    AmdSessionConfig cfg;
    /* open-addressing key -> SRows map (n_keys==0: single global entry) */
    int64_t map_cap, map_n;
    int64_t *map_keys;
    uint8_t *map_used;
    SRows *map_rows;
    int has_wm; uint64_t wm;
    int out_cols;
    int64_t out_rows, out_cap;
    int64_t **out;
    char err[256];
} SOp;

static void srows_push(SRows *b, int nv, const int64_t *vals, int64_t ts) {
    if (b->n == b->cap) {
        b->cap = b->cap ? b->cap * 2 : 16;
        for (int c = 0; c < nv; c++)
            b->vals[c] = realloc(b->vals[c], (size_t)b->cap * 8);
        b->ts = realloc(b->ts, (size_t)b->cap * 8);
    }
    for (int c = 0; c < nv; c++) b->vals[c][b->n] = vals[c];
    b->ts[b->n++] = ts;
}

static void sgrow(SOp *o);

static SRows *skey_slot(SOp *o, int64_t key) {
    if (o->map_n * 10 >= o->map_cap * 7) sgrow(o);
    uint64_t m = (uint64_t)o->map_cap - 1;
    uint64_t i = hash64((uint64_t)key) & m;
    while (o->map_used[i] && o->map_keys[i] != key) i = (i + 1) & m;
    if (!o->map_used[i]) {
        o->map_used[i] = 1;
        o->map_keys[i] = key;
        o->map_n++;
        SRows *b = &o->map_rows[i];
        memset(b, 0, sizeof *b);
        b->vals = calloc((size_t)o->cfg.n_value_cols, sizeof(int64_t *));
    }
    return &o->map_rows[i];
}

static void sgrow(SOp *o) {
    int64_t ocap = o->map_cap;
    int64_t *ok = o->map_keys; uint8_t *ou = o->map_used;
    SRows *orr = o->map_rows;
    o->map_cap <<= 1; o->map_n = 0;
    o->map_keys = malloc((size_t)o->map_cap * 8);
    o->map_used = calloc((size_t)o->map_cap, 1);
    o->map_rows = calloc((size_t)o->map_cap, sizeof(SRows));
    for (int64_t i = 0; i < ocap; i++) {
        if (!ou[i]) continue;
        SRows *b = skey_slot(o, ok[i]);
        free(b->vals);
        *b = orr[i];
    }
    free(ok); free(ou); free(orr);
}

ORACLE_API void *oracle_session_create(const AmdSessionConfig *cfg) {
    if (!cfg || cfg->n_keys < 0 || cfg->n_keys > 1 || cfg->n_value_cols < 0 ||
        cfg->n_aggs < 1 || cfg->n_aggs > AMD_MAX_AGGS || cfg->gap_nanos == 0)
        return NULL;
    SOp *o = calloc(1, sizeof(SOp));
    o->cfg = *cfg;
    o->map_cap = 64;
    o->map_keys = malloc((size_t)o->map_cap * 8);
    o->map_used = calloc((size_t)o->map_cap, 1);
    o->map_rows = calloc((size_t)o->map_cap, sizeof(SRows));
    o->out_cols = cfg->n_keys + cfg->n_aggs + 3;
    o->out = calloc((size_t)o->out_cols, sizeof(int64_t *));
    return o;
}

ORACLE_API const char *oracle_session_last_error(void *h) {
    return h ? ((SOp *)h)->err : "null handle / invalid config";
}

ORACLE_API int oracle_session_process_batch(void *h,
                                            const int64_t *const *cols,
                                            int32_t n_cols, int64_t n_rows) {
    SOp *o = h;
    const AmdSessionConfig *c = &o->cfg;
    int want = c->n_keys + c->n_value_cols + 1;
    if (n_cols != want) {
        snprintf(o->err, sizeof o->err, "expected %d cols, got %d", want,
                 n_cols);
        return 1;
    }
    const int64_t *ts = cols[n_cols - 1];
    int64_t tmp[AMD_MAX_AGGS * 2 + 8];
    for (int64_t r = 0; r < n_rows; r++) {
        /* late-data filter: keep ts >= watermark (:856-867, gt_eq) */
        if (o->has_wm && (uint64_t)ts[r] < o->wm) continue;
        int64_t key = c->n_keys ? cols[0][r] : 0;
        SRows *b = skey_slot(o, key);
        for (int v = 0; v < c->n_value_cols; v++)
            tmp[v] = cols[c->n_keys + v][r];
        srows_push(b, c->n_value_cols, tmp, ts[r]);
    }
    return 0;
}

static void sout_reserve(SOp *o, int64_t add) {
    if (o->out_rows + add <= o->out_cap) return;
    int64_t ncap = o->out_cap ? o->out_cap : 1024;
    while (ncap < o->out_rows + add) ncap *= 2;
    for (int i = 0; i < o->out_cols; i++)
        o->out[i] = realloc(o->out[i], (size_t)ncap * 8);
    o->out_cap = ncap;
}

static int cmp_i64(const void *a, const void *b) {
    int64_t x = *(const int64_t *)a, y = *(const int64_t *)b;
    return x < y ? -1 : x > y ? 1 : 0;
}

/* emit one session [lo, hi) of the key's order-sorted rows */
static void semit(SOp *o, int64_t key, SRows *b, const int64_t *order,
                  int64_t lo, int64_t hi, int64_t data_start,
                  int64_t data_end) {
    const AmdSessionConfig *c = &o->cfg;
    sout_reserve(o, 1);
    int64_t r = o->out_rows++;
    int col = 0;
    if (c->n_keys) o->out[col++][r] = key;
    for (int a = 0; a < c->n_aggs; a++, col++) {
        int op = c->agg_ops[a];
        int vc = c->agg_col[a];
        int64_t acc_i = 0;
        double acc_d = 0.0;
        int64_t mn = INT64_MAX, mx = INT64_MIN;
        for (int64_t i = lo; i < hi; i++) {
            int64_t v = vc >= 0 ? b->vals[vc][order[i]] : 0;
            acc_i += vc >= 0 ? v : 1;
            acc_d += (double)v;
            if (v < mn) mn = v;
            if (v > mx) mx = v;
        }
        switch (op) {
        case AMD_AGG_COUNT: o->out[col][r] = hi - lo; break;
        case AMD_AGG_SUM:   o->out[col][r] = acc_i; break;
        case AMD_AGG_MIN:   o->out[col][r] = mn; break;
        case AMD_AGG_MAX:   o->out[col][r] = mx; break;
        case AMD_AGG_AVG:
            o->out[col][r] = d_to_bits(acc_d / (double)(hi - lo));
            break;
        case AMD_AGG_COUNT_DISTINCT: {
            /* exact distinct over the session's rows (the reference runs
             * its final aggregate over the buffered rows at close) */
            int64_t n = hi - lo;
            int64_t *vals = malloc((size_t)n * 8);
            for (int64_t i = lo; i < hi; i++)
                vals[i - lo] = b->vals[vc][order[i]];
            qsort(vals, (size_t)n, 8, cmp_i64);
            int64_t d = 0;
            for (int64_t i = 0; i < n; i++)
                if (i == 0 || vals[i] != vals[i - 1]) d++;
            free(vals);
            o->out[col][r] = d;
            break;
        }
        }
    }
    int64_t end = data_end + (int64_t)c->gap_nanos;
    o->out[col++][r] = data_start;
    o->out[col++][r] = end;
    o->out[col][r] = end - 1;
}

ORACLE_API int oracle_session_handle_watermark(void *h, uint64_t wm,
                                               AmdOutBatch *out) {
    SOp *o = h;
    const AmdSessionConfig *c = &o->cfg;
    o->has_wm = 1;
    o->wm = wm;
    int64_t gap = (int64_t)c->gap_nanos;
    for (int64_t s = 0; s < o->map_cap; s++) {
        if (!o->map_used[s]) continue;
        SRows *b = &o->map_rows[s];
        if (b->n == 0) continue;
        /* sort row indices by ts (stable not needed: rows within a session
         * are aggregated, order irrelevant for the supported aggs; AVG uses
         * f64 accumulation with stated tolerance) */
        int64_t *order = malloc((size_t)b->n * 16);
        for (int64_t i = 0; i < b->n; i++) {
            order[2 * i] = b->ts[i];
            order[2 * i + 1] = i;
        }
        /* sort (ts, idx) pairs */
        qsort(order, (size_t)b->n, 16, cmp_i64);
        int64_t *idx = malloc((size_t)b->n * 8);
        for (int64_t i = 0; i < b->n; i++) idx[i] = order[2 * i + 1];
        int64_t lo = 0, kept_from = -1;
        while (lo < b->n) {
            int64_t data_start = b->ts[idx[lo]];
            int64_t data_end = data_start;
            int64_t hi = lo + 1;
            while (hi < b->n && b->ts[idx[hi]] < data_end + gap) {
                data_end = b->ts[idx[hi]];
                hi++;
            }
            if ((uint64_t)(data_end + gap) < wm) {
                semit(o, o->map_keys[s], b, idx, lo, hi, data_start,
                      data_end);
            } else {
                kept_from = lo;
                break;
            }
            lo = hi;
        }
        if (kept_from != 0) {
            /* compact: keep rows of unfired sessions */
            if (kept_from < 0) {
                b->n = 0;
            } else {
                int64_t keep = b->n - kept_from;
                int64_t *nts = malloc((size_t)keep * 8);
                for (int64_t i = 0; i < keep; i++)
                    nts[i] = b->ts[idx[kept_from + i]];
                for (int v = 0; v < c->n_value_cols; v++) {
                    int64_t *nv = malloc((size_t)keep * 8);
                    for (int64_t i = 0; i < keep; i++)
                        nv[i] = b->vals[v][idx[kept_from + i]];
                    free(b->vals[v]);
                    b->vals[v] = nv;
                }
                free(b->ts);
                b->ts = nts;
                b->n = b->cap = keep;
            }
        }
        free(order);
        free(idx);
    }
    if (out) {
        memset(out, 0, sizeof *out);
        out->n_rows = o->out_rows;
        out->n_cols = o->out_cols;
        out->cols = calloc((size_t)o->out_cols, sizeof(void *));
        out->is_f64 = calloc((size_t)o->out_cols, sizeof(int32_t));
        for (int a = 0; a < c->n_aggs; a++)
            if (c->agg_ops[a] == AMD_AGG_AVG)
                out->is_f64[c->n_keys + a] = 1;
        for (int i = 0; i < o->out_cols; i++) {
            out->cols[i] = malloc((size_t)(o->out_rows ? o->out_rows : 1) * 8);
            if (o->out_rows)
                memcpy(out->cols[i], o->out[i], (size_t)o->out_rows * 8);
        }
        o->out_rows = 0;
    }
    return 0;
}

/* drain buffered (unfired) rows: [key?, vals..., _timestamp] */
ORACLE_API int oracle_session_checkpoint_drain(void *h, AmdOutBatch *out) {
    SOp *o = h;
    const AmdSessionConfig *c = &o->cfg;
    int ncols = c->n_keys + c->n_value_cols + 1;
    int64_t total = 0;
    for (int64_t s = 0; s < o->map_cap; s++)
        if (o->map_used[s]) total += o->map_rows[s].n;
    memset(out, 0, sizeof *out);
    out->n_rows = total;
    out->n_cols = ncols;
    out->cols = calloc((size_t)ncols, sizeof(void *));
    out->is_f64 = calloc((size_t)ncols, sizeof(int32_t));
    for (int i = 0; i < ncols; i++)
        out->cols[i] = malloc((size_t)(total ? total : 1) * 8);
    int64_t r = 0;
    for (int64_t s = 0; s < o->map_cap; s++) {
        if (!o->map_used[s]) continue;
        SRows *b = &o->map_rows[s];
        for (int64_t j = 0; j < b->n; j++, r++) {
            int col = 0;
            if (c->n_keys) ((int64_t *)out->cols[col++])[r] = o->map_keys[s];
            for (int v = 0; v < c->n_value_cols; v++)
                ((int64_t *)out->cols[col++])[r] = b->vals[v][j];
            ((int64_t *)out->cols[col])[r] = b->ts[j];
        }
    }
    return 0;
}

ORACLE_API int oracle_session_restore(void *h, const int64_t *const *cols,
                                      int32_t n_cols, int64_t n_rows) {
    /* on_start re-adds drained rows (:803-846) */
    return oracle_session_process_batch(h, cols, n_cols, n_rows);
}

ORACLE_API void oracle_session_destroy(void *h) {
    SOp *o = h;
    if (!o) return;
    for (int64_t s = 0; s < o->map_cap; s++) {
        if (!o->map_used[s]) continue;
        SRows *b = &o->map_rows[s];
        for (int v = 0; v < o->cfg.n_value_cols; v++) free(b->vals[v]);
        free(b->vals);
        free(b->ts);
    }
    free(o->map_keys); free(o->map_used); free(o->map_rows);
    for (int i = 0; i < o->out_cols; i++) free(o->out[i]);
    free(o->out);
    free(o);
}

/* ================================================================== */
/* Non-windowed (TTL'd) stream-stream join oracle.
 *
 * Restated from crates/arroyo-worker/src/arrow/join_with_expiration.rs:
 *   - process_batch_index :162-180 routes to process_left/process_right
 *     :42-108: the incoming batch is inserted into its side's per-key state
 *     (KeyTimeView::insert, expiring_time_key_map.rs:997-1050) and joined
 *     against the OTHER side's stored rows for the batch's keys (get_batch
 *     :970-985) through the inner HashJoinExec (compute_pair :110-130) —
 *     each cross-side pair is emitted exactly once, when its later row
 *     arrives;
 *   - output _timestamp = max(left._timestamp, right._timestamp)
 *     (post_join_timestamp_projection, arroyo-planner/src/plan/join.rs
 *     :121-191);
 *   - the live in-memory view never evicts during a run; the TTL filters
 *     state on restore (table_manager.rs:533-570, get_view(watermark)).
 *     oracle_expjoin_expire applies the same cutoff (watermark - ttl)
 *     explicitly, for parity with the GPU path's bounded-memory mode.
 */

typedef struct {
    int64_t cap, n;
    int64_t **vals;
    int64_t *ts;
} ERows;

typedef struct {
    int64_t map_cap, map_n;
    int64_t *map_keys;
    uint8_t *map_used;
    ERows *rows;
    int nv;
} ESide;

typedef struct {
    AmdExpJoinConfig cfg;
    ESide side[2];
    int has_wm; uint64_t wm;
    int out_cols;
    int64_t out_rows, out_cap;
    int64_t **out;
    char err[256];
} EOp;

static void eside_grow(ESide *sd);

static ERows *eside_slot(ESide *sd, int64_t key) {
    if (sd->map_n * 10 >= sd->map_cap * 7) eside_grow(sd);
    uint64_t m = (uint64_t)sd->map_cap - 1;
    uint64_t i = hash64((uint64_t)key) & m;
    while (sd->map_used[i] && sd->map_keys[i] != key) i = (i + 1) & m;
    if (!sd->map_used[i]) {
        sd->map_used[i] = 1;
        sd->map_keys[i] = key;
        sd->map_n++;
        ERows *b = &sd->rows[i];
        memset(b, 0, sizeof *b);
        b->vals = calloc((size_t)sd->nv, sizeof(int64_t *));
    }
    return &sd->rows[i];
}

static void eside_grow(ESide *sd) {
    int64_t ocap = sd->map_cap;
    int64_t *ok = sd->map_keys; uint8_t *ou = sd->map_used;
    ERows *orr = sd->rows;
    sd->map_cap <<= 1; sd->map_n = 0;
    sd->map_keys = malloc((size_t)sd->map_cap * 8);
    sd->map_used = calloc((size_t)sd->map_cap, 1);
    sd->rows = calloc((size_t)sd->map_cap, sizeof(ERows));
    for (int64_t i = 0; i < ocap; i++) {
        if (!ou[i]) continue;
        ERows *b = eside_slot(sd, ok[i]);
        free(b->vals);
        *b = orr[i];
    }
    free(ok); free(ou); free(orr);
}

static void erows_push(ERows *b, int nv, const int64_t *const *cols,
                       int64_t r, int64_t ts) {
    if (b->n == b->cap) {
        b->cap = b->cap ? b->cap * 2 : 8;
        for (int c = 0; c < nv; c++)
            b->vals[c] = realloc(b->vals[c], (size_t)b->cap * 8);
        b->ts = realloc(b->ts, (size_t)b->cap * 8);
    }
    for (int c = 0; c < nv; c++) b->vals[c][b->n] = cols[1 + c][r];
    b->ts[b->n++] = ts;
}

ORACLE_API void *oracle_expjoin_create(const AmdExpJoinConfig *cfg) {
    if (!cfg || cfg->n_keys != 1 || cfg->n_left_vals < 0 ||
        cfg->n_right_vals < 0 || cfg->ttl_nanos == 0 ||
        cfg->join_type < 0 || cfg->join_type > AMD_JOIN_FULL)
        return NULL;
    EOp *o = calloc(1, sizeof(EOp));
    o->cfg = *cfg;
    for (int s = 0; s < 2; s++) {
        ESide *sd = &o->side[s];
        sd->nv = s == 0 ? cfg->n_left_vals : cfg->n_right_vals;
        sd->map_cap = 64;
        sd->map_keys = malloc((size_t)sd->map_cap * 8);
        sd->map_used = calloc((size_t)sd->map_cap, 1);
        sd->rows = calloc((size_t)sd->map_cap, sizeof(ERows));
    }
    /* non-inner: + [left_present, right_present]; non-inner or updating:
     * + trailing is_retract (a later match retracts the earlier
     * null-padded row) */
    o->out_cols = 1 + cfg->n_left_vals + cfg->n_right_vals + 1 +
                  (cfg->join_type != AMD_JOIN_INNER ? 2 : 0) +
                  (cfg->join_type != AMD_JOIN_INNER || cfg->updating ? 1 : 0);
    o->out = calloc((size_t)o->out_cols, sizeof(int64_t *));
    return o;
}

ORACLE_API const char *oracle_expjoin_last_error(void *h) {
    return h ? ((EOp *)h)->err : "null handle / invalid config";
}

static void eout_reserve(EOp *o, int64_t add) {
    if (o->out_rows + add <= o->out_cap) return;
    int64_t ncap = o->out_cap ? o->out_cap : 1024;
    while (ncap < o->out_rows + add) ncap *= 2;
    for (int i = 0; i < o->out_cols; i++)
        o->out[i] = realloc(o->out[i], (size_t)ncap * 8);
    o->out_cap = ncap;
}

static void ebuild_out(EOp *o, AmdOutBatch *out) {
    memset(out, 0, sizeof *out);
    out->n_rows = o->out_rows;
    out->n_cols = o->out_cols;
    out->cols = calloc((size_t)o->out_cols, sizeof(void *));
    out->is_f64 = calloc((size_t)o->out_cols, sizeof(int32_t));
    for (int i = 0; i < o->out_cols; i++) {
        out->cols[i] = malloc((size_t)(o->out_rows ? o->out_rows : 1) * 8);
        if (o->out_rows)
            memcpy(out->cols[i], o->out[i], (size_t)o->out_rows * 8);
    }
    o->out_rows = 0;
}

/* emit one output row; lv/rv NULL = that side absent (outer join) */
static void eemit2(EOp *o, int64_t key, const int64_t *lv, int64_t lts,
                   const int64_t *rv, int64_t rts, int retract) {
    const AmdExpJoinConfig *c = &o->cfg;
    eout_reserve(o, 1);
    int64_t rr = o->out_rows++;
    int col = 0;
    o->out[col++][rr] = key;
    for (int v = 0; v < c->n_left_vals; v++)
        o->out[col++][rr] = lv ? lv[v] : 0;
    for (int v = 0; v < c->n_right_vals; v++)
        o->out[col++][rr] = rv ? rv[v] : 0;
    o->out[col++][rr] = lv ? (rv ? (lts > rts ? lts : rts) : lts) : rts;
    if (c->join_type != AMD_JOIN_INNER) {
        o->out[col++][rr] = lv != NULL;
        o->out[col++][rr] = rv != NULL;
    }
    if (c->join_type != AMD_JOIN_INNER || c->updating)
        o->out[col][rr] = retract;
}

/* pair emission with arrival-side routing: `mine` is the incoming row's
 * values, `ob` a stored row of the other side */
static void eemit_pair(EOp *o, int32_t side, int64_t key,
                       const int64_t *mine, int64_t mts, const ERows *ob,
                       int64_t j, int retract) {
    int64_t ovals[64];
    ESide *other = &o->side[1 - side];
    for (int v = 0; v < other->nv; v++) ovals[v] = ob->vals[v][j];
    if (side == 0)
        eemit2(o, key, mine, mts, ovals, ob->ts[j], retract);
    else
        eemit2(o, key, ovals, ob->ts[j], mine, mts, retract);
}

/* null-padded emission for a stored row of side `s` */
static void eemit_null(EOp *o, int32_t s, int64_t key, const ERows *b,
                       int64_t j, int retract) {
    int64_t vals[64];
    for (int v = 0; v < o->side[s].nv; v++) vals[v] = b->vals[v][j];
    if (s == 0)
        eemit2(o, key, vals, b->ts[j], NULL, 0, retract);
    else
        eemit2(o, key, NULL, 0, vals, b->ts[j], retract);
}

static ERows *eside_find(ESide *sd, int64_t key) {
    uint64_t m = (uint64_t)sd->map_cap - 1;
    uint64_t i = hash64((uint64_t)key) & m;
    while (sd->map_used[i] && sd->map_keys[i] != key) i = (i + 1) & m;
    return sd->map_used[i] ? &sd->rows[i] : NULL;
}

static int expjoin_insert(EOp *o, int32_t side, const int64_t *const *cols,
                          int32_t n_cols, int64_t n_rows, int emit,
                          AmdOutBatch *out) {
    const AmdExpJoinConfig *c = &o->cfg;
    int nv = side == 0 ? c->n_left_vals : c->n_right_vals;
    int upd = c->updating ? 1 : 0;
    int jt = c->join_type;
    int want = 1 + nv + upd + 1;
    if (n_cols != want) {
        snprintf(o->err, sizeof o->err, "side %d expects %d cols, got %d",
                 side, want, n_cols);
        return 1;
    }
    /* which null-padded rows this join type emits (plan/join.rs join_type
     * into the DataFusion join: LEFT pads unmatched left rows, etc.) */
    int null_own = jt == AMD_JOIN_FULL ||
                   (side == 0 ? jt == AMD_JOIN_LEFT : jt == AMD_JOIN_RIGHT);
    int null_other = jt == AMD_JOIN_FULL ||
                     (side == 0 ? jt == AMD_JOIN_RIGHT : jt == AMD_JOIN_LEFT);
    const int64_t *ts = cols[n_cols - 1];
    const int64_t *retr = upd ? cols[1 + nv] : NULL;
    for (int64_t r = 0; r < n_rows; r++) {
        int64_t key = cols[0][r];
        int64_t mine[64];
        for (int v = 0; v < nv; v++) mine[v] = cols[1 + v][r];
        ERows *ob = eside_find(&o->side[1 - side], key);
        int64_t on = ob ? ob->n : 0;
        if (!upd || !retr[r]) {
            /* append */
            ERows *own = eside_slot(&o->side[side], key);
            if (emit) {
                if (on > 0) {
                    for (int64_t j = 0; j < on; j++)
                        eemit_pair(o, side, key, mine, ts[r], ob, j, 0);
                    if (null_other && own->n == 0)
                        /* the other side's rows were unmatched until now:
                         * retract their null-padded emissions */
                        for (int64_t j = 0; j < on; j++)
                            eemit_null(o, 1 - side, key, ob, j, 1);
                } else if (null_own) {
                    eemit2(o, key, side == 0 ? mine : NULL,
                           side == 0 ? ts[r] : 0,
                           side == 0 ? NULL : mine,
                           side == 0 ? 0 : ts[r], 0);
                }
            }
            erows_push(own, nv, cols, r, ts[r]);
        } else {
            /* retract: remove one stored row with equal values */
            ERows *own = eside_find(&o->side[side], key);
            int64_t idx = -1;
            if (own)
                for (int64_t j = 0; j < own->n && idx < 0; j++) {
                    int eq = 1;
                    for (int v = 0; v < nv && eq; v++)
                        eq = own->vals[v][j] == mine[v];
                    if (eq) idx = j;
                }
            if (idx < 0) {
                snprintf(o->err, sizeof o->err,
                         "retract of unknown row for key %lld",
                         (long long)key);
                return 1;
            }
            int64_t sts = own->ts[idx];
            if (emit) {
                if (on > 0) {
                    for (int64_t j = 0; j < on; j++)
                        eemit_pair(o, side, key, mine, sts, ob, j, 1);
                    if (null_other && own->n == 1)
                        /* the other side's rows lose their last match:
                         * their null-padded rows come back */
                        for (int64_t j = 0; j < on; j++)
                            eemit_null(o, 1 - side, key, ob, j, 0);
                } else if (null_own) {
                    eemit2(o, key, side == 0 ? mine : NULL,
                           side == 0 ? sts : 0,
                           side == 0 ? NULL : mine,
                           side == 0 ? 0 : sts, 1);
                }
            }
            /* remove idx (order within the multiset is not observable) */
            for (int v = 0; v < nv; v++)
                own->vals[v][idx] = own->vals[v][own->n - 1];
            own->ts[idx] = own->ts[own->n - 1];
            own->n--;
        }
    }
    if (out) ebuild_out(o, out);
    return 0;
}

ORACLE_API int oracle_expjoin_process_batch(void *h, int32_t side,
                                            const int64_t *const *cols,
                                            int32_t n_cols, int64_t n_rows,
                                            AmdOutBatch *out) {
    return expjoin_insert((EOp *)h, side, cols, n_cols, n_rows, 1, out);
}

ORACLE_API int oracle_expjoin_handle_watermark(void *h, uint64_t wm) {
    EOp *o = h;
    o->has_wm = 1;
    o->wm = wm;
    return 0;
}

/* drop stored rows with ts < watermark - ttl (the cutoff the reference
 * applies to this table across checkpoint/restore) */
ORACLE_API int oracle_expjoin_expire(void *h) {
    EOp *o = h;
    if (!o->has_wm || o->wm < o->cfg.ttl_nanos) return 0;
    uint64_t cutoff = o->wm - o->cfg.ttl_nanos;
    for (int s = 0; s < 2; s++) {
        ESide *sd = &o->side[s];
        for (int64_t i = 0; i < sd->map_cap; i++) {
            if (!sd->map_used[i]) continue;
            ERows *b = &sd->rows[i];
            int64_t w = 0;
            for (int64_t j = 0; j < b->n; j++) {
                if ((uint64_t)b->ts[j] >= cutoff) {
                    for (int v = 0; v < sd->nv; v++)
                        b->vals[v][w] = b->vals[v][j];
                    b->ts[w++] = b->ts[j];
                }
            }
            b->n = w;
        }
    }
    return 0;
}

ORACLE_API int oracle_expjoin_checkpoint_drain(void *h, int32_t side,
                                               AmdOutBatch *out) {
    EOp *o = h;
    ESide *sd = &o->side[side];
    int ncols = 1 + sd->nv + 1;
    int64_t total = 0;
    for (int64_t i = 0; i < sd->map_cap; i++)
        if (sd->map_used[i]) total += sd->rows[i].n;
    memset(out, 0, sizeof *out);
    out->n_rows = total;
    out->n_cols = ncols;
    out->cols = calloc((size_t)ncols, sizeof(void *));
    out->is_f64 = calloc((size_t)ncols, sizeof(int32_t));
    for (int i = 0; i < ncols; i++)
        out->cols[i] = malloc((size_t)(total ? total : 1) * 8);
    int64_t r = 0;
    for (int64_t i = 0; i < sd->map_cap; i++) {
        if (!sd->map_used[i]) continue;
        ERows *b = &sd->rows[i];
        for (int64_t j = 0; j < b->n; j++, r++) {
            int col = 0;
            ((int64_t *)out->cols[col++])[r] = sd->map_keys[i];
            for (int v = 0; v < sd->nv; v++)
                ((int64_t *)out->cols[col++])[r] = b->vals[v][j];
            ((int64_t *)out->cols[col])[r] = b->ts[j];
        }
    }
    return 0;
}

ORACLE_API int oracle_expjoin_restore(void *h, int32_t side,
                                      const int64_t *const *cols,
                                      int32_t n_cols, int64_t n_rows,
                                      int has_watermark,
                                      uint64_t watermark_nanos) {
    EOp *o = h;
    const AmdExpJoinConfig *c = &o->cfg;
    int nv = side == 0 ? c->n_left_vals : c->n_right_vals;
    int want = 1 + nv + 1;
    if (n_cols != want) {
        snprintf(o->err, sizeof o->err, "side %d expects %d cols, got %d",
                 side, want, n_cols);
        return 1;
    }
    uint64_t cutoff = 0;
    if (has_watermark && watermark_nanos > c->ttl_nanos)
        cutoff = watermark_nanos - c->ttl_nanos;
    const int64_t *ts = cols[n_cols - 1];
    for (int64_t r = 0; r < n_rows; r++) {
        if ((uint64_t)ts[r] < cutoff) continue;  /* restore-time TTL filter */
        erows_push(eside_slot(&o->side[side], cols[0][r]), nv, cols, r,
                   ts[r]);
    }
    return 0;
}

ORACLE_API void oracle_expjoin_destroy(void *h) {
    EOp *o = h;
    if (!o) return;
    for (int s = 0; s < 2; s++) {
        ESide *sd = &o->side[s];
        for (int64_t i = 0; i < sd->map_cap; i++) {
            if (!sd->map_used[i]) continue;
            for (int v = 0; v < sd->nv; v++) free(sd->rows[i].vals[v]);
            free(sd->rows[i].vals);
            free(sd->rows[i].ts);
        }
        free(sd->map_keys); free(sd->map_used); free(sd->rows);
    }
    for (int i = 0; i < o->out_cols; i++) free(o->out[i]);
    free(o->out);
    free(o);
}

/* ================================================================== */
/* Updating (non-windowed) aggregate oracle.
 *
 * Restated from crates/arroyo-worker/src/arrow/incremental_aggregator.rs:
 *   - process_batch :931-949 routes to keyed_aggregate :826-884 /
 *     global_aggregate :778-824: per row, the key's accumulators are
 *     updated (or retracted when the batch's _updating_meta.is_retract is
 *     set for the row, get_retracts :740-758);
 *   - COUNT/SUM/AVG use retractable "Sliding" accumulators
 *     (IncrementalState::Sliding :78-83 -- retract = subtract);
 *     COUNT DISTINCT uses the "Batch" value-multiset accumulator
 *     (IncrementalState::Batch :84-174: per-value refcount, evaluate counts
 *     values with count > 0); MIN/MAX would also be Batch-type -- here they
 *     are append-only and a retraction is a loud error (stated scope);
 *   - flush :637-737 (driven by the caller, the reference's tick timer):
 *     every key touched since the last flush emits
 *     retract(value as of last flush) + append(current value); the retract
 *     is omitted for keys first seen this interval (updated_keys value None,
 *     keyed_aggregate :839-856), the append is omitted when all the key's
 *     rows are retracted (null timestamp check :670-678 -- here: live row
 *     count 0), and the pair is skipped when the value is unchanged
 *     (:649-661).
 */

typedef struct {
    int64_t cap, n;
    int64_t *vals;            /* distinct values */
    int64_t *cnt;             /* net refcounts */
} UMultiset;

#define UAGG_MAX_SW 128   /* total state words cap (BIT ops take 32);
                             matches the GPU path */

typedef struct {
    int64_t rows;             /* live row count (presence) */
    int64_t st[UAGG_MAX_SW];
    UMultiset ms[AMD_MAX_AGGS];   /* per COUNT_DISTINCT agg */
    int emitted;
    int64_t last[AMD_MAX_AGGS];   /* last emitted values (AVG: f64 bits) */
    int changed;
    int64_t touched_epoch;        /* last flush epoch with activity */
} UEntry;

typedef struct {
    AmdUpdatingConfig cfg;
    /* variable-width states: agg a's words at st[soff[a]..+usw[a]) */
    int soff[AMD_MAX_AGGS];
    int usw[AMD_MAX_AGGS];
    int sw_total;
    int64_t map_cap, map_n;
    int64_t *map_keys;
    uint8_t *map_used;
    UEntry *ent;
    int out_cols;
    int64_t out_rows, out_cap;
    int64_t **out;
    int64_t epoch;
    char err[256];
} UOp;

static int uagg_width(int op) {
    if (op >= AMD_AGG_COVAR_POP && op <= AMD_AGG_REGR_SXY) return 5;
    if (op == AMD_AGG_BIT_AND || op == AMD_AGG_BIT_OR) return 32;
    return 2;
}

static void ugrow(UOp *o);

static int64_t ukey_slot(UOp *o, int64_t key) {
    if (o->map_n * 10 >= o->map_cap * 7) ugrow(o);
    uint64_t m = (uint64_t)o->map_cap - 1;
    uint64_t i = hash64((uint64_t)key) & m;
    while (o->map_used[i] && o->map_keys[i] != key) i = (i + 1) & m;
    if (!o->map_used[i]) {
        o->map_used[i] = 1;
        o->map_keys[i] = key;
        o->map_n++;
        UEntry *e = &o->ent[i];
        memset(e, 0, sizeof *e);
        for (int a = 0; a < o->cfg.n_aggs; a++) {
            switch (o->cfg.agg_ops[a]) {
            case AMD_AGG_MIN: e->st[o->soff[a]] = INT64_MAX; break;
            case AMD_AGG_MAX: e->st[o->soff[a]] = INT64_MIN; break;
            }
        }
    }
    return (int64_t)i;
}

static void ugrow(UOp *o) {
    int64_t ocap = o->map_cap;
    int64_t *ok = o->map_keys; uint8_t *ou = o->map_used;
    UEntry *oe = o->ent;
    o->map_cap <<= 1; o->map_n = 0;
    o->map_keys = malloc((size_t)o->map_cap * 8);
    o->map_used = calloc((size_t)o->map_cap, 1);
    o->ent = calloc((size_t)o->map_cap, sizeof(UEntry));
    for (int64_t i = 0; i < ocap; i++) {
        if (!ou[i]) continue;
        int64_t s = ukey_slot(o, ok[i]);
        o->ent[s] = oe[i];
    }
    free(ok); free(ou); free(oe);
}

ORACLE_API void *oracle_updagg_create(const AmdUpdatingConfig *cfg) {
    if (!cfg || cfg->n_keys < 0 || cfg->n_keys > 1 || cfg->n_value_cols < 0 ||
        cfg->n_aggs < 1 || cfg->n_aggs > AMD_MAX_AGGS)
        return NULL;
    UOp *o = calloc(1, sizeof(UOp));
    o->cfg = *cfg;
    for (int a = 0; a < cfg->n_aggs; a++) {
        o->soff[a] = o->sw_total;
        o->usw[a] = uagg_width(cfg->agg_ops[a]);
        o->sw_total += o->usw[a];
    }
    if (o->sw_total > UAGG_MAX_SW) { free(o); return NULL; }
    o->map_cap = 64;
    o->map_keys = malloc((size_t)o->map_cap * 8);
    o->map_used = calloc((size_t)o->map_cap, 1);
    o->ent = calloc((size_t)o->map_cap, sizeof(UEntry));
    o->out_cols = cfg->n_keys + cfg->n_aggs + 1;  /* + is_retract */
    o->out = calloc((size_t)o->out_cols, sizeof(int64_t *));
    return o;
}

ORACLE_API const char *oracle_updagg_last_error(void *h) {
    return h ? ((UOp *)h)->err : "null handle / invalid config";
}

static void ums_add(UMultiset *m, int64_t v, int64_t d) {
    for (int64_t i = 0; i < m->n; i++)
        if (m->vals[i] == v) { m->cnt[i] += d; return; }
    if (m->n == m->cap) {
        m->cap = m->cap ? m->cap * 2 : 16;
        m->vals = realloc(m->vals, (size_t)m->cap * 8);
        m->cnt = realloc(m->cnt, (size_t)m->cap * 8);
    }
    m->vals[m->n] = v;
    m->cnt[m->n++] = d;
}

ORACLE_API int oracle_updagg_process_batch(void *h,
                                           const int64_t *const *cols,
                                           int32_t n_cols, int64_t n_rows) {
    UOp *o = h;
    const AmdUpdatingConfig *c = &o->cfg;
    int want = c->n_keys + c->n_value_cols + 1;
    if (n_cols != want) {
        snprintf(o->err, sizeof o->err, "expected %d cols, got %d", want,
                 n_cols);
        return 1;
    }
    const int64_t *retr = cols[n_cols - 1];
    for (int64_t r = 0; r < n_rows; r++) {
        int64_t key = c->n_keys ? cols[0][r] : 0;
        int64_t s = ukey_slot(o, key);
        UEntry *e = &o->ent[s];
        int64_t d = retr[r] ? -1 : 1;
        e->rows += d;
        e->changed = 1;
        e->touched_epoch = o->epoch;
        for (int a = 0; a < c->n_aggs; a++) {
            int64_t v = c->agg_col[a] >= 0
                            ? cols[c->n_keys + c->agg_col[a]][r] : 0;
            int64_t *st = &e->st[o->soff[a]];
            switch (c->agg_ops[a]) {
            case AMD_AGG_COUNT: st[0] += d; break;
            case AMD_AGG_SUM:   st[0] += d * v; break;
            case AMD_AGG_AVG:
                st[0] += d;
                st[1] = d_to_bits(bits_to_d(st[1]) + (double)d * v);
                break;
            case AMD_AGG_MIN:
                if (d < 0) {
                    snprintf(o->err, sizeof o->err,
                             "MIN does not support retraction");
                    return 1;
                }
                if (v < st[0]) st[0] = v;
                break;
            case AMD_AGG_MAX:
                if (d < 0) {
                    snprintf(o->err, sizeof o->err,
                             "MAX does not support retraction");
                    return 1;
                }
                if (v > st[0]) st[0] = v;
                break;
            case AMD_AGG_COUNT_DISTINCT:
                ums_add(&e->ms[a], v, d);
                break;
            case AMD_AGG_STDDEV:
            case AMD_AGG_STDDEV_POP:
            case AMD_AGG_VAR:
            case AMD_AGG_VAR_POP:
                /* retractable co-moments (sum x, sum x^2); n = live rows */
                st[0] = d_to_bits(bits_to_d(st[0]) + (double)d * (double)v);
                st[1] = d_to_bits(bits_to_d(st[1]) +
                                  (double)d * (double)v * (double)v);
                break;
            case AMD_AGG_BIT_XOR:
                st[0] ^= v;   /* self-inverse: retract == append */
                break;
            case AMD_AGG_COVAR_POP: case AMD_AGG_COVAR_SAMP:
            case AMD_AGG_CORR: case AMD_AGG_REGR_SLOPE:
            case AMD_AGG_REGR_INTERCEPT: case AMD_AGG_REGR_R2:
            case AMD_AGG_REGR_AVGX: case AMD_AGG_REGR_AVGY:
            case AMD_AGG_REGR_COUNT: case AMD_AGG_REGR_SXX:
            case AMD_AGG_REGR_SYY: case AMD_AGG_REGR_SXY: {
                /* (sum y, sum x, sum xy, sum y^2, sum x^2); y = agg_col
                 * (SQL's first arg), x = agg_col2 */
                double y = (double)v;
                double x = (double)cols[c->n_keys + c->agg_col2[a]][r];
                st[0] = d_to_bits(bits_to_d(st[0]) + d * y);
                st[1] = d_to_bits(bits_to_d(st[1]) + d * x);
                st[2] = d_to_bits(bits_to_d(st[2]) + d * x * y);
                st[3] = d_to_bits(bits_to_d(st[3]) + d * y * y);
                st[4] = d_to_bits(bits_to_d(st[4]) + d * x * x);
                break;
            }
            case AMD_AGG_BIT_AND:
            case AMD_AGG_BIT_OR: {
                uint32_t *cnt = (uint32_t *)st;
                uint64_t uv = (uint64_t)v;
                while (uv) {
                    int b = __builtin_ctzll(uv);
                    cnt[b] += (uint32_t)d;   /* u32 wraps; net is exact */
                    uv &= uv - 1;
                }
                break;
            }
            }
        }
    }
    return 0;
}

static void ueval(const UOp *o, const UEntry *e, int64_t *out) {
    const AmdUpdatingConfig *c = &o->cfg;
    for (int a = 0; a < c->n_aggs; a++) {
        const int64_t *st = &e->st[o->soff[a]];
        switch (c->agg_ops[a]) {
        case AMD_AGG_COUNT:
        case AMD_AGG_SUM:
        case AMD_AGG_MIN:
        case AMD_AGG_MAX:
        case AMD_AGG_BIT_XOR:
            out[a] = st[0];
            break;
        case AMD_AGG_AVG:
            out[a] = d_to_bits(bits_to_d(st[1]) / (double)st[0]);
            break;
        case AMD_AGG_COUNT_DISTINCT: {
            int64_t n = 0;
            for (int64_t i = 0; i < e->ms[a].n; i++)
                if (e->ms[a].cnt[i] > 0) n++;
            out[a] = n;
            break;
        }
        case AMD_AGG_STDDEV:
        case AMD_AGG_STDDEV_POP:
        case AMD_AGG_VAR:
        case AMD_AGG_VAR_POP: {
            double n = (double)e->rows;
            double sx = bits_to_d(st[0]);
            double sxx = bits_to_d(st[1]);
            double mean = sx / n;
            double m2 = sxx - n * mean * mean;
            if (m2 < 0.0) m2 = 0.0;   /* fp noise */
            int samp = c->agg_ops[a] == AMD_AGG_STDDEV ||
                       c->agg_ops[a] == AMD_AGG_VAR;
            double var = samp ? (e->rows > 1 ? m2 / (n - 1.0) : NAN)
                              : m2 / n;
            if (c->agg_ops[a] == AMD_AGG_STDDEV ||
                c->agg_ops[a] == AMD_AGG_STDDEV_POP)
                var = sqrt(var);
            out[a] = d_to_bits(var);
            break;
        }
        case AMD_AGG_COVAR_POP: case AMD_AGG_COVAR_SAMP:
        case AMD_AGG_CORR: case AMD_AGG_REGR_SLOPE:
        case AMD_AGG_REGR_INTERCEPT: case AMD_AGG_REGR_R2:
        case AMD_AGG_REGR_AVGX: case AMD_AGG_REGR_AVGY:
        case AMD_AGG_REGR_COUNT: case AMD_AGG_REGR_SXX:
        case AMD_AGG_REGR_SYY: case AMD_AGG_REGR_SXY: {
            double n = (double)e->rows;
            double sy = bits_to_d(st[0]), sx = bits_to_d(st[1]);
            double sxy = bits_to_d(st[2]), syy = bits_to_d(st[3]);
            double sxx2 = bits_to_d(st[4]);
            double my = sy / n, mx = sx / n;
            double Sxy = sxy - n * mx * my;
            double Sxx = sxx2 - n * mx * mx;
            double Syy = syy - n * my * my;
            if (Sxx < 0.0) Sxx = 0.0;
            if (Syy < 0.0) Syy = 0.0;
            double v;
            switch (c->agg_ops[a]) {
            case AMD_AGG_COVAR_POP:  v = Sxy / n; break;
            case AMD_AGG_COVAR_SAMP:
                v = e->rows > 1 ? Sxy / (n - 1.0) : NAN;
                break;
            case AMD_AGG_CORR:
                v = (e->rows > 1 && Sxx > 0.0 && Syy > 0.0)
                        ? Sxy / sqrt(Sxx * Syy) : NAN;
                break;
            case AMD_AGG_REGR_SLOPE:
                v = Sxx > 0.0 ? Sxy / Sxx : NAN;
                break;
            case AMD_AGG_REGR_INTERCEPT:
                v = Sxx > 0.0 ? my - (Sxy / Sxx) * mx : NAN;
                break;
            case AMD_AGG_REGR_R2:
                v = Sxx <= 0.0 ? NAN
                    : (Syy <= 0.0 ? 1.0 : (Sxy * Sxy) / (Sxx * Syy));
                break;
            case AMD_AGG_REGR_AVGX: v = mx; break;
            case AMD_AGG_REGR_AVGY: v = my; break;
            case AMD_AGG_REGR_SXX: v = Sxx; break;
            case AMD_AGG_REGR_SYY: v = Syy; break;
            case AMD_AGG_REGR_SXY: v = Sxy; break;
            default: v = 0.0; break;   /* unreachable */
            }
            if (c->agg_ops[a] == AMD_AGG_REGR_COUNT)
                out[a] = e->rows;              /* i64 output */
            else
                out[a] = d_to_bits(v);
            break;
        }
        case AMD_AGG_BIT_AND:
        case AMD_AGG_BIT_OR: {
            const uint32_t *cnt = (const uint32_t *)st;
            uint64_t r2 = 0;
            for (int b = 0; b < 64; b++) {
                uint32_t nb = cnt[b];
                int set = c->agg_ops[a] == AMD_AGG_BIT_AND
                              ? (int64_t)nb == e->rows && e->rows > 0
                              : nb > 0;
                if (set) r2 |= 1ULL << b;
            }
            out[a] = (int64_t)r2;
            break;
        }
        }
    }
}

static void uout_reserve(UOp *o, int64_t add) {
    if (o->out_rows + add <= o->out_cap) return;
    int64_t ncap = o->out_cap ? o->out_cap : 1024;
    while (ncap < o->out_rows + add) ncap *= 2;
    for (int i = 0; i < o->out_cols; i++)
        o->out[i] = realloc(o->out[i], (size_t)ncap * 8);
    o->out_cap = ncap;
}

static void uemit(UOp *o, int64_t key, const int64_t *vals, int retract) {
    uout_reserve(o, 1);
    int64_t r = o->out_rows++;
    int col = 0;
    if (o->cfg.n_keys) o->out[col++][r] = key;
    for (int a = 0; a < o->cfg.n_aggs; a++) o->out[col++][r] = vals[a];
    o->out[col][r] = retract;
}

ORACLE_API int oracle_updagg_flush(void *h, AmdOutBatch *out) {
    UOp *o = h;
    const AmdUpdatingConfig *c = &o->cfg;
    for (int64_t s = 0; s < o->map_cap; s++) {
        if (!o->map_used[s] || !o->ent[s].changed) continue;
        UEntry *e = &o->ent[s];
        e->changed = 0;
        int64_t cur[AMD_MAX_AGGS];
        if (e->rows > 0) ueval(o, e, cur);
        if (e->emitted) {
            if (e->rows > 0 &&
                memcmp(cur, e->last, (size_t)c->n_aggs * 8) == 0)
                continue;  /* unchanged: skip the retract/append pair */
            uemit(o, o->map_keys[s], e->last, 1);
            e->emitted = 0;
        }
        if (e->rows > 0) {
            uemit(o, o->map_keys[s], cur, 0);
            memcpy(e->last, cur, (size_t)c->n_aggs * 8);
            e->emitted = 1;
        }
    }
    o->epoch++;
    if (out) {
        memset(out, 0, sizeof *out);
        out->n_rows = o->out_rows;
        out->n_cols = o->out_cols;
        out->cols = calloc((size_t)o->out_cols, sizeof(void *));
        out->is_f64 = calloc((size_t)o->out_cols, sizeof(int32_t));
        for (int a = 0; a < c->n_aggs; a++)
            if (c->agg_ops[a] == AMD_AGG_AVG ||
                (c->agg_ops[a] >= AMD_AGG_STDDEV &&
                 c->agg_ops[a] <= AMD_AGG_VAR_POP) ||
                (c->agg_ops[a] >= AMD_AGG_COVAR_POP &&
                 c->agg_ops[a] <= AMD_AGG_REGR_SXY &&
                 c->agg_ops[a] != AMD_AGG_REGR_COUNT))
                out->is_f64[c->n_keys + a] = 1;
        for (int i = 0; i < o->out_cols; i++) {
            out->cols[i] = malloc((size_t)(o->out_rows ? o->out_rows : 1) * 8);
            if (o->out_rows)
                memcpy(out->cols[i], o->out[i], (size_t)o->out_rows * 8);
        }
        o->out_rows = 0;
    }
    return 0;
}

/* TTL eviction: the reference's UpdatingCache::time_out (updating_cache.rs)
 * retracts and removes keys idle longer than the ttl during flush, measured
 * in wall-clock time; here the cutoff is flush epochs (idle_flushes) so
 * tests are deterministic.  Evicted keys emit retract(last emitted) and
 * their state resets (a re-arriving key starts fresh, as in the
 * reference's remove()). */
ORACLE_API int oracle_updagg_expire(void *h, int64_t idle_flushes,
                                    AmdOutBatch *out) {
    UOp *o = h;
    const AmdUpdatingConfig *c = &o->cfg;
    for (int64_t s = 0; s < o->map_cap; s++) {
        if (!o->map_used[s]) continue;
        UEntry *e = &o->ent[s];
        if (o->epoch - e->touched_epoch <= idle_flushes) continue;
        if (e->emitted) uemit(o, o->map_keys[s], e->last, 1);
        for (int a = 0; a < c->n_aggs; a++) {
            free(e->ms[a].vals);
            free(e->ms[a].cnt);
        }
        int64_t te = e->touched_epoch;
        memset(e, 0, sizeof *e);
        e->touched_epoch = te;  /* stays idle unless touched again */
        for (int a = 0; a < c->n_aggs; a++) {
            switch (c->agg_ops[a]) {
            case AMD_AGG_MIN: e->st[2 * a] = INT64_MAX; break;
            case AMD_AGG_MAX: e->st[2 * a] = INT64_MIN; break;
            }
        }
    }
    if (out) {
        memset(out, 0, sizeof *out);
        out->n_rows = o->out_rows;
        out->n_cols = o->out_cols;
        out->cols = calloc((size_t)o->out_cols, sizeof(void *));
        out->is_f64 = calloc((size_t)o->out_cols, sizeof(int32_t));
        for (int i = 0; i < o->out_cols; i++) {
            out->cols[i] = malloc((size_t)(o->out_rows ? o->out_rows : 1) * 8);
            if (o->out_rows)
                memcpy(out->cols[i], o->out[i], (size_t)o->out_rows * 8);
        }
        o->out_rows = 0;
    }
    return 0;
}

/* which=0: scalar accumulator states [key?, rows, st words..., emitted,
 * last words...]; which=1: distinct-value multiset rows
 * [key?, agg_index, value, net_count] */
ORACLE_API int oracle_updagg_checkpoint_drain(void *h, int32_t which,
                                              AmdOutBatch *out) {
    UOp *o = h;
    const AmdUpdatingConfig *c = &o->cfg;
    int ncols;
    int64_t total = 0;
    if (which == 0) {
        ncols = c->n_keys + 1 + o->sw_total + 1 + c->n_aggs;
        for (int64_t s = 0; s < o->map_cap; s++)
            if (o->map_used[s]) total++;
    } else {
        ncols = c->n_keys + 3;
        for (int64_t s = 0; s < o->map_cap; s++)
            if (o->map_used[s])
                for (int a = 0; a < c->n_aggs; a++)
                    for (int64_t i = 0; i < o->ent[s].ms[a].n; i++)
                        if (o->ent[s].ms[a].cnt[i] != 0) total++;
    }
    memset(out, 0, sizeof *out);
    out->n_rows = total;
    out->n_cols = ncols;
    out->cols = calloc((size_t)ncols, sizeof(void *));
    out->is_f64 = calloc((size_t)ncols, sizeof(int32_t));
    for (int i = 0; i < ncols; i++)
        out->cols[i] = malloc((size_t)(total ? total : 1) * 8);
    int64_t r = 0;
    for (int64_t s = 0; s < o->map_cap; s++) {
        if (!o->map_used[s]) continue;
        UEntry *e = &o->ent[s];
        if (which == 0) {
            int col = 0;
            if (c->n_keys) ((int64_t *)out->cols[col++])[r] = o->map_keys[s];
            ((int64_t *)out->cols[col++])[r] = e->rows;
            for (int w = 0; w < o->sw_total; w++)
                ((int64_t *)out->cols[col++])[r] = e->st[w];
            ((int64_t *)out->cols[col++])[r] = e->emitted;
            for (int a = 0; a < c->n_aggs; a++)
                ((int64_t *)out->cols[col++])[r] = e->last[a];
            r++;
        } else {
            for (int a = 0; a < c->n_aggs; a++)
                for (int64_t i = 0; i < e->ms[a].n; i++) {
                    if (e->ms[a].cnt[i] == 0) continue;
                    int col = 0;
                    if (c->n_keys)
                        ((int64_t *)out->cols[col++])[r] = o->map_keys[s];
                    ((int64_t *)out->cols[col++])[r] = a;
                    ((int64_t *)out->cols[col++])[r] = e->ms[a].vals[i];
                    ((int64_t *)out->cols[col])[r] = e->ms[a].cnt[i];
                    r++;
                }
        }
    }
    return 0;
}

ORACLE_API int oracle_updagg_restore(void *h, int32_t which,
                                     const int64_t *const *cols,
                                     int32_t n_cols, int64_t n_rows) {
    UOp *o = h;
    const AmdUpdatingConfig *c = &o->cfg;
    if (which == 0) {
        int want = c->n_keys + 1 + o->sw_total + 1 + c->n_aggs;
        if (n_cols != want) {
            snprintf(o->err, sizeof o->err, "restore(0) expects %d cols",
                     want);
            return 1;
        }
        for (int64_t r = 0; r < n_rows; r++) {
            int64_t s = ukey_slot(o, c->n_keys ? cols[0][r] : 0);
            UEntry *e = &o->ent[s];
            int col = c->n_keys;
            e->rows = cols[col++][r];
            for (int w = 0; w < o->sw_total; w++)
                e->st[w] = cols[col++][r];
            e->emitted = (int)cols[col++][r];
            for (int a = 0; a < c->n_aggs; a++)
                e->last[a] = cols[col++][r];
        }
    } else {
        int want = c->n_keys + 3;
        if (n_cols != want) {
            snprintf(o->err, sizeof o->err, "restore(1) expects %d cols",
                     want);
            return 1;
        }
        for (int64_t r = 0; r < n_rows; r++) {
            int64_t s = ukey_slot(o, c->n_keys ? cols[0][r] : 0);
            int a = (int)cols[c->n_keys][r];
            ums_add(&o->ent[s].ms[a], cols[c->n_keys + 1][r],
                    cols[c->n_keys + 2][r]);
        }
    }
    return 0;
}

ORACLE_API void oracle_updagg_destroy(void *h) {
    UOp *o = h;
    if (!o) return;
    for (int64_t s = 0; s < o->map_cap; s++) {
        if (!o->map_used[s]) continue;
        for (int a = 0; a < o->cfg.n_aggs; a++) {
            free(o->ent[s].ms[a].vals);
            free(o->ent[s].ms[a].cnt);
        }
    }
    free(o->map_keys); free(o->map_used); free(o->ent);
    for (int i = 0; i < o->out_cols; i++) free(o->out[i]);
    free(o->out);
    free(o);
}

/* ================================================================== */
/* SQL window-function (ROW_NUMBER per instant) oracle.
 *
 * Restated from crates/arroyo-worker/src/arrow/window_fn.rs:
 *   - process_batch :275-300 buffers rows per exact `_timestamp` instant,
 *     silently filtering rows with ts < watermark
 *     (filter_and_split_batches :52-93, filter_by_time);
 *   - handle_watermark :220-246 fires every instant < watermark in
 *     timestamp order through the BoundedWindowAggExec: here ROW_NUMBER()
 *     OVER (PARTITION BY part_col ORDER BY order cols), ties resolved by
 *     input order (DataFusion's stable sort);
 *   - rows with row_number > limit are dropped (the reference's downstream
 *     filter; limit 0 keeps all).
 */

typedef struct {
    uint64_t instant;
    int64_t cap, n;
    int64_t **cols;           /* [n_cols-1][cap] (ts implied) */
} WfInstant;

typedef struct {
    AmdWindowFnConfig cfg;
    int n_inst, cap_inst;
    WfInstant *inst;          /* sorted by instant */
    int has_wm; uint64_t wm;
    int out_cols;
    int64_t out_rows, out_cap;
    int64_t **out;
    char err[256];
} WfOp;

static WfInstant *wf_get(WfOp *o, uint64_t t) {
    int lo = 0, hi = o->n_inst;
    while (lo < hi) { int m = (lo + hi) / 2; if (o->inst[m].instant < t) lo = m + 1; else hi = m; }
    if (lo < o->n_inst && o->inst[lo].instant == t) return &o->inst[lo];
    if (o->n_inst == o->cap_inst) {
        o->cap_inst = o->cap_inst ? o->cap_inst * 2 : 16;
        o->inst = realloc(o->inst, (size_t)o->cap_inst * sizeof(WfInstant));
    }
    memmove(o->inst + lo + 1, o->inst + lo,
            (size_t)(o->n_inst - lo) * sizeof(WfInstant));
    WfInstant *in = &o->inst[lo];
    memset(in, 0, sizeof *in);
    in->instant = t;
    in->cols = calloc((size_t)o->cfg.n_cols - 1, sizeof(int64_t *));
    o->n_inst++;
    return in;
}

ORACLE_API void *oracle_windowfn_create(const AmdWindowFnConfig *cfg) {
    if (!cfg || cfg->n_cols < 2 || cfg->n_cols > 12 || cfg->n_order < 1 ||
        cfg->n_order > 2 || cfg->part_col >= cfg->n_cols - 1)
        return NULL;
    WfOp *o = calloc(1, sizeof(WfOp));
    o->cfg = *cfg;
    o->out_cols = cfg->n_cols + 1;
    o->out = calloc((size_t)o->out_cols, sizeof(int64_t *));
    return o;
}

ORACLE_API const char *oracle_windowfn_last_error(void *h) {
    return h ? ((WfOp *)h)->err : "null handle / invalid config";
}

ORACLE_API int oracle_windowfn_process_batch(void *h,
                                             const int64_t *const *cols,
                                             int32_t n_cols,
                                             int64_t n_rows) {
    WfOp *o = h;
    if (n_cols != o->cfg.n_cols) {
        snprintf(o->err, sizeof o->err, "expected %d cols, got %d",
                 o->cfg.n_cols, n_cols);
        return 1;
    }
    const int64_t *ts = cols[n_cols - 1];
    for (int64_t r = 0; r < n_rows; r++) {
        /* late rows silently filtered (filter_by_time) */
        if (o->has_wm && (uint64_t)ts[r] < o->wm) continue;
        WfInstant *in = wf_get(o, (uint64_t)ts[r]);
        if (in->n == in->cap) {
            in->cap = in->cap ? in->cap * 2 : 64;
            for (int c = 0; c < n_cols - 1; c++)
                in->cols[c] = realloc(in->cols[c], (size_t)in->cap * 8);
        }
        for (int c = 0; c < n_cols - 1; c++) in->cols[c][in->n] = cols[c][r];
        in->n++;
    }
    return 0;
}

static const WfOp *g_wf_sort_op;
static const WfInstant *g_wf_sort_in;

static int wf_cmp(const void *a, const void *b) {
    int64_t i = *(const int64_t *)a, j = *(const int64_t *)b;
    const AmdWindowFnConfig *c = &g_wf_sort_op->cfg;
    const WfInstant *in = g_wf_sort_in;
    if (c->part_col >= 0) {
        int64_t pa = in->cols[c->part_col][i], pb = in->cols[c->part_col][j];
        if (pa != pb) return pa < pb ? -1 : 1;
    }
    for (int k = 0; k < c->n_order; k++) {
        int64_t va = in->cols[c->order_col[k]][i];
        int64_t vb = in->cols[c->order_col[k]][j];
        if (va != vb) {
            int lt = va < vb ? -1 : 1;
            return c->order_desc[k] ? -lt : lt;
        }
    }
    return i < j ? -1 : i > j ? 1 : 0;  /* stable: input order */
}

static void wf_out_reserve(WfOp *o, int64_t add) {
    if (o->out_rows + add <= o->out_cap) return;
    int64_t ncap = o->out_cap ? o->out_cap : 1024;
    while (ncap < o->out_rows + add) ncap *= 2;
    for (int i = 0; i < o->out_cols; i++)
        o->out[i] = realloc(o->out[i], (size_t)ncap * 8);
    o->out_cap = ncap;
}

static void wf_fire(WfOp *o, WfInstant *in) {
    const AmdWindowFnConfig *c = &o->cfg;
    int64_t *idx = malloc((size_t)in->n * 8);
    for (int64_t i = 0; i < in->n; i++) idx[i] = i;
    g_wf_sort_op = o;
    g_wf_sort_in = in;
    qsort(idx, (size_t)in->n, 8, wf_cmp);
    int64_t rn = 0;
    for (int64_t i = 0; i < in->n; i++) {
        if (i == 0 ||
            (c->part_col >= 0 &&
             in->cols[c->part_col][idx[i]] !=
                 in->cols[c->part_col][idx[i - 1]]))
            rn = 0;
        rn++;
        if (c->limit && rn > c->limit) continue;
        wf_out_reserve(o, 1);
        int64_t r = o->out_rows++;
        for (int cc = 0; cc < c->n_cols - 1; cc++)
            o->out[cc][r] = in->cols[cc][idx[i]];
        o->out[c->n_cols - 1][r] = (int64_t)in->instant;
        o->out[c->n_cols][r] = rn;
    }
    free(idx);
}

ORACLE_API int oracle_windowfn_handle_watermark(void *h, uint64_t wm,
                                                AmdOutBatch *out) {
    WfOp *o = h;
    o->has_wm = 1;
    o->wm = wm;
    int fired = 0;
    while (fired < o->n_inst && o->inst[fired].instant < wm) {
        wf_fire(o, &o->inst[fired]);
        for (int c = 0; c < o->cfg.n_cols - 1; c++)
            free(o->inst[fired].cols[c]);
        free(o->inst[fired].cols);
        fired++;
    }
    if (fired) {
        memmove(o->inst, o->inst + fired,
                (size_t)(o->n_inst - fired) * sizeof(WfInstant));
        o->n_inst -= fired;
    }
    if (out) {
        memset(out, 0, sizeof *out);
        out->n_rows = o->out_rows;
        out->n_cols = o->out_cols;
        out->cols = calloc((size_t)o->out_cols, sizeof(void *));
        out->is_f64 = calloc((size_t)o->out_cols, sizeof(int32_t));
        for (int i = 0; i < o->out_cols; i++) {
            out->cols[i] = malloc((size_t)(o->out_rows ? o->out_rows : 1) * 8);
            if (o->out_rows)
                memcpy(out->cols[i], o->out[i], (size_t)o->out_rows * 8);
        }
        o->out_rows = 0;
    }
    return 0;
}

ORACLE_API int oracle_windowfn_checkpoint_drain(void *h, AmdOutBatch *out) {
    WfOp *o = h;
    int ncols = o->cfg.n_cols;
    int64_t total = 0;
    for (int i = 0; i < o->n_inst; i++) total += o->inst[i].n;
    memset(out, 0, sizeof *out);
    out->n_rows = total;
    out->n_cols = ncols;
    out->cols = calloc((size_t)ncols, sizeof(void *));
    out->is_f64 = calloc((size_t)ncols, sizeof(int32_t));
    for (int i = 0; i < ncols; i++)
        out->cols[i] = malloc((size_t)(total ? total : 1) * 8);
    int64_t r = 0;
    for (int i = 0; i < o->n_inst; i++) {
        WfInstant *in = &o->inst[i];
        for (int64_t j = 0; j < in->n; j++, r++) {
            for (int c = 0; c < ncols - 1; c++)
                ((int64_t *)out->cols[c])[r] = in->cols[c][j];
            ((int64_t *)out->cols[ncols - 1])[r] = (int64_t)in->instant;
        }
    }
    return 0;
}

/* restore = re-ingest the drained rows: windowfn state IS the raw
 * buffered rows (drain preserves per-instant arrival order, so the
 * ROW_NUMBER arrival-sequence tiebreak is reproduced), mirroring
 * WindowFunctionOperator's table restore of buffered instants
 * (crates/arroyo-worker/src/arrow/window_fn.rs ExpiringTimeKeyTable). */
ORACLE_API int oracle_windowfn_restore(void *h, const int64_t *const *cols,
                                       int32_t n_cols, int64_t n_rows) {
    return oracle_windowfn_process_batch(h, cols, n_cols, n_rows);
}

ORACLE_API void oracle_windowfn_destroy(void *h) {
    WfOp *o = h;
    if (!o) return;
    for (int i = 0; i < o->n_inst; i++) {
        for (int c = 0; c < o->cfg.n_cols - 1; c++) free(o->inst[i].cols[c]);
        free(o->inst[i].cols);
    }
    free(o->inst);
    for (int i = 0; i < o->out_cols; i++) free(o->out[i]);
    free(o->out);
    free(o);
}

/* ================================================================== */
/* Stateless map/filter/projection oracle.
 *
 * Restated from crates/arroyo-worker/src/arrow/mod.rs: the expression
 * operators evaluate a DataFusion plan per batch
 * (StatelessPhysicalExecutor::process_batch :245-290) and emit the result
 * immediately; filters preserve row order.  The register program carries
 * the plan's information content for the supported expression set. */

typedef struct {
    AmdMapConfig cfg;
    char err[256];
} MOp;

ORACLE_API void *oracle_map_create(const AmdMapConfig *cfg) {
    if (!cfg || cfg->n_in_cols < 1 || cfg->n_in_cols > AMD_MAP_MAX_REGS ||
        cfg->n_prog < 0 || cfg->n_prog > AMD_MAP_MAX_PROG ||
        cfg->n_out < 1 || cfg->n_out > AMD_MAP_MAX_OUT)
        return NULL;
    for (int i = 0; i < cfg->n_prog; i++)
        if (cfg->prog[i].dst < 0 || cfg->prog[i].dst >= AMD_MAP_MAX_REGS ||
            cfg->prog[i].a < 0 || cfg->prog[i].a >= AMD_MAP_MAX_REGS ||
            cfg->prog[i].b < 0 || cfg->prog[i].b >= AMD_MAP_MAX_REGS)
            return NULL;
    MOp *o = calloc(1, sizeof(MOp));
    o->cfg = *cfg;
    return o;
}

ORACLE_API const char *oracle_map_last_error(void *h) {
    return h ? ((MOp *)h)->err : "null handle / invalid config";
}

static int map_eval_row(const AmdMapConfig *c, const int64_t *const *cols,
                        int64_t r, int64_t *regs, char *err) {
    for (int i = 0; i < c->n_in_cols; i++) regs[i] = cols[i][r];
    for (int i = 0; i < c->n_prog; i++) {
        const AmdMapInstr *in = &c->prog[i];
        int64_t a = regs[in->a], b = regs[in->b], v = 0;
        double fa = bits_to_d(a), fb = bits_to_d(b);
        switch (in->op) {
        case AMD_MOP_CONST: v = in->imm; break;
        case AMD_MOP_ADD: v = a + b; break;
        case AMD_MOP_SUB: v = a - b; break;
        case AMD_MOP_MUL: v = a * b; break;
        case AMD_MOP_DIV:
            if (b == 0) {
                snprintf(err, 256, "division by zero");
                return 1;
            }
            v = a / b;
            break;
        case AMD_MOP_MOD:
            if (b == 0) {
                snprintf(err, 256, "division by zero");
                return 1;
            }
            v = a % b;
            break;
        case AMD_MOP_EQ: v = a == b; break;
        case AMD_MOP_NE: v = a != b; break;
        case AMD_MOP_LT: v = a < b; break;
        case AMD_MOP_LE: v = a <= b; break;
        case AMD_MOP_GT: v = a > b; break;
        case AMD_MOP_GE: v = a >= b; break;
        case AMD_MOP_AND: v = (a != 0) && (b != 0); break;
        case AMD_MOP_OR: v = (a != 0) || (b != 0); break;
        case AMD_MOP_NOT: v = a == 0; break;
        case AMD_MOP_I2F: v = d_to_bits((double)a); break;
        case AMD_MOP_F2I: v = (int64_t)fa; break;
        case AMD_MOP_FADD: v = d_to_bits(fa + fb); break;
        case AMD_MOP_FSUB: v = d_to_bits(fa - fb); break;
        case AMD_MOP_FMUL: v = d_to_bits(fa * fb); break;
        case AMD_MOP_FDIV: v = d_to_bits(fa / fb); break;
        }
        regs[in->dst] = v;
    }
    return 0;
}

ORACLE_API int oracle_map_process_batch(void *h, const int64_t *const *cols,
                                        int32_t n_cols, int64_t n_rows,
                                        AmdOutBatch *out) {
    MOp *o = h;
    const AmdMapConfig *c = &o->cfg;
    if (n_cols != c->n_in_cols) {
        snprintf(o->err, sizeof o->err, "expected %d cols, got %d",
                 c->n_in_cols, n_cols);
        return 1;
    }
    memset(out, 0, sizeof *out);
    out->n_cols = c->n_out;
    out->cols = calloc((size_t)c->n_out, sizeof(void *));
    out->is_f64 = calloc((size_t)c->n_out, sizeof(int32_t));
    for (int i = 0; i < c->n_out; i++) {
        out->cols[i] = malloc((size_t)(n_rows ? n_rows : 1) * 8);
        out->is_f64[i] = c->out_is_f64[i];
    }
    int64_t w = 0;
    int64_t regs[AMD_MAP_MAX_REGS] = {0};
    for (int64_t r = 0; r < n_rows; r++) {
        if (map_eval_row(c, cols, r, regs, o->err)) return 1;
        if (c->filter_reg >= 0 && regs[c->filter_reg] == 0) continue;
        for (int i = 0; i < c->n_out; i++)
            ((int64_t *)out->cols[i])[w] = regs[c->out_reg[i]];
        w++;
    }
    out->n_rows = w;
    return 0;
}

ORACLE_API void oracle_map_destroy(void *h) { free(h); }
