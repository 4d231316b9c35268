/* arroyo-amd session (gap) window aggregate: MI355X-native (gfx950)
 * equivalent of SessionAggregatingWindowFunc
 * (crates/arroyo-worker/src/arrow/session_aggregating_window.rs) behind the
 * arroyo_amd_session_* C ABI (include/arroyo_amd.h).
 *
 * MI355X-first design (NOT a translation of the reference's per-key
 * BTreeMap-of-batches + per-key DataFusion exec streams):
 *   - Per-batch pre-aggregation (k_sess_update): one fused kernel folds each
 *     row into a per-batch open-addressing table keyed by the session key,
 *     accumulating (min_ts, max_ts, aggregate states) on memset-zero-
 *     identity encoded words, with an LDS staging table absorbing hot-key
 *     atomics first.  Because a batch's
 *     event-time span is < gap (enforced; the host splits wider batches into
 *     gap/2 buckets), a key's rows within one batch always belong to ONE
 *     session, so (min_ts, max_ts, states) is a valid partial session —
 *     this replaces the reference's per-row BTreeMap insertion and
 *     ActiveSession::add_batch scan (:424-520).
 *   - Session store merge (k_sess_merge): one thread per DISTINCT key of the
 *     batch (the pre-agg table's occupied slots) merges its partial session
 *     into the device-resident session store: an open-addressing key table
 *     whose slots hold up to max_sessions live sessions inline.  Two
 *     sessions are one iff their [start, end+gap) closures touch
 *     (P.start < A.end + gap && A.start < P.end + gap, both strict) — the
 *     interval form of the reference's strictly-within-gap row rule.  One
 *     thread per distinct key means no locks and no CAS loops on session
 *     data; only the key-slot claim is a CAS.
 *   - Watermark firing (k_sess_fire): one thread per key slot scans its
 *     sessions and emits every session with data_end + gap < watermark
 *     (KeyComputingHolder::watermark_update :559-608) through a global
 *     output cursor; output columns [key?, finals..., window_start,
 *     window_end = data_end + gap, _timestamp = window_end - 1]
 *     (to_record_batch :316-380).  Key slots are never un-claimed (open
 *     addressing with deletions would corrupt probe chains); an idle key
 *     costs its slot only.
 *   - Late rows (ts < watermark) are dropped in the update kernel
 *     (process_batch :849-874, gt_eq filter) — silently, as the reference
 *     does (unlike the window operators, where pre-watermark data is a
 *     fatal error).
 *   - COUNT DISTINCT keeps an exact per-session value set in single-writer
 *     hash regions (see the cd_* primitives below): phase 1 appends raw
 *     values to a per-batch chain, the per-key phase-2 thread dedupes.
 *
 * Parity is pinned against oracle/arroyo_oracle.c (itself pinned against
 * the reference's session_window / global_session_window golden vectors) by
 * tests/test_session.py.
 */
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "../../include/arroyo_amd_types.h"

#define API extern "C" __attribute__((visibility("default")))

namespace sess {

#define EMPTY_KEY (-1LL)
#define SERR_TABLE_FULL 1
#define SERR_BATCH_FULL 2
#define SERR_SESSIONS   3
#define SERR_OUT_CAP    4

__host__ __device__ inline uint64_t enc_min(int64_t v) {
    return ~(((uint64_t)v) ^ 0x8000000000000000ULL);
}
__host__ __device__ inline int64_t dec_min(uint64_t e) {
    return (int64_t)((~e) ^ 0x8000000000000000ULL);
}
__host__ __device__ inline uint64_t enc_max(int64_t v) {
    return ((uint64_t)v) ^ 0x8000000000000000ULL;
}
__host__ __device__ inline int64_t dec_max(uint64_t e) {
    return (int64_t)(e ^ 0x8000000000000000ULL);
}

__device__ inline uint64_t hash64(uint64_t x) {
    x += 0x9e3779b97f4a7c15ULL;
    x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ULL;
    x = (x ^ (x >> 27)) * 0x94d049bb133111ebULL;
    return x ^ (x >> 31);
}

/* aggregate state: n_aggs x 2 u64 words, zero = identity (memset-clear):
 * COUNT/SUM w0 add; MIN w0 = enc_min max-domain; MAX w0 = enc_max;
 * AVG w0 = count, w1 = bits(f64 sum) folded with CAS-add. */
struct AggSpec {
    int32_t n_aggs;
    int32_t op[AMD_MAX_AGGS];
    int32_t col[AMD_MAX_AGGS];
};

__device__ inline void atomic_fold(uint64_t *st, const AggSpec &a,
                                   const int64_t *const *vcols, int64_t r) {
    for (int i = 0; i < a.n_aggs; i++) {
        uint64_t *w = st + 2 * i;
        int64_t v = a.col[i] >= 0 ? vcols[i][r] : 0;
        switch (a.op[i]) {
        case AMD_AGG_COUNT:
            atomicAdd((unsigned long long *)w, 1ULL);
            break;
        case AMD_AGG_SUM:
            atomicAdd((unsigned long long *)w, (unsigned long long)v);
            break;
        case AMD_AGG_MIN:
            atomicMax((unsigned long long *)w,
                      (unsigned long long)enc_min(v));
            break;
        case AMD_AGG_MAX:
            atomicMax((unsigned long long *)w,
                      (unsigned long long)enc_max(v));
            break;
        case AMD_AGG_AVG: {
            atomicAdd((unsigned long long *)w, 1ULL);
            double dv = (double)v;
            unsigned long long old = w[1], assumed;
            do {
                assumed = old;
                double cur;
                memcpy(&cur, &assumed, 8);
                cur += dv;
                unsigned long long nv2;
                memcpy(&nv2, &cur, 8);
                old = atomicCAS((unsigned long long *)&w[1], assumed, nv2);
            } while (old != assumed);
            break;
        }
        case AMD_AGG_COUNT_DISTINCT:
            /* handled by the batch value chain, not the scalar words */
            break;
        }
    }
}

/* -------- per-session exact distinct sets (COUNT DISTINCT) ----------
 * Single-writer hash regions: only the one phase-2/restore thread that
 * owns a key ever touches its sessions' regions, so inserts are plain
 * stores.  Region layout: [D/64 bitmap words][D value slots]; regions are
 * pool-allocated and zero-initialised once (never reused -- fired
 * sessions orphan theirs; bounded by log2_cd_regions, loud error). */
#define SERR_CD_FULL    5
#define SERR_CD_REGIONS 6
#define SERR_BV_POOL    7

struct CdPool {
    uint64_t *regions;        /* [n_regions][D/64 + D] */
    unsigned long long *rcur;
    int64_t n_regions;
    uint32_t D;
};

__device__ inline uint64_t cd_region_words(uint32_t D) {
    return D / 64 + D;
}

__device__ inline int64_t cd_alloc(const CdPool &P, int *err) {
    int64_t idx = (int64_t)atomicAdd(P.rcur, 1ULL);
    if (idx >= P.n_regions) { *err = SERR_CD_REGIONS; return -1; }
    return idx;
}

/* insert v into region r (single writer); returns 1 if newly inserted */
__device__ inline int cd_insert(const CdPool &P, int64_t r, int64_t v,
                                int *err) {
    uint64_t *bm = P.regions + (size_t)r * cd_region_words(P.D);
    int64_t *vals = (int64_t *)(bm + P.D / 64);
    uint32_t h = (uint32_t)hash64((uint64_t)v) & (P.D - 1);
    for (uint32_t pr = 0; pr < P.D; pr++) {
        uint32_t sl = (h + pr) & (P.D - 1);
        uint64_t bit = 1ULL << (sl & 63);
        if (!(bm[sl >> 6] & bit)) {
            bm[sl >> 6] |= bit;
            vals[sl] = v;
            return 1;
        }
        if (vals[sl] == v) return 0;
    }
    *err = SERR_CD_FULL;
    return 0;
}

/* merge region src into the session state acc (w0 count, w1 region+1) */
__device__ inline void cd_merge_into(const CdPool &P, uint64_t *acc,
                                     uint64_t src_count, uint64_t src_reg1,
                                     int *err) {
    if (src_reg1 == 0) return;
    if (acc[1] == 0) {
        acc[0] = src_count;
        acc[1] = src_reg1;
        return;
    }
    /* reinsert the smaller set into the larger */
    uint64_t big_reg1 = acc[1], big_cnt = acc[0];
    uint64_t small_reg1 = src_reg1, small_cnt = src_count;
    if (small_cnt > big_cnt) {
        big_reg1 = src_reg1; big_cnt = src_count;
        small_reg1 = acc[1]; small_cnt = acc[0];
    }
    int64_t sr = (int64_t)small_reg1 - 1, br = (int64_t)big_reg1 - 1;
    const uint64_t *bm = P.regions + (size_t)sr * cd_region_words(P.D);
    const int64_t *vals = (const int64_t *)(bm + P.D / 64);
    for (uint32_t sl = 0; sl < P.D; sl++)
        if (bm[sl >> 6] & (1ULL << (sl & 63)))
            big_cnt += cd_insert(P, br, vals[sl], err);
    acc[0] = big_cnt;
    acc[1] = big_reg1;
}

/* atomically merge encoded partial words (LDS flush -> global table) */
__device__ inline void state_merge_atomic(uint64_t *dst, const uint64_t *src,
                                          const AggSpec &a) {
    for (int i = 0; i < a.n_aggs; i++) {
        switch (a.op[i]) {
        case AMD_AGG_COUNT:
        case AMD_AGG_SUM:
            atomicAdd((unsigned long long *)&dst[2 * i],
                      (unsigned long long)src[2 * i]);
            break;
        case AMD_AGG_MIN:
        case AMD_AGG_MAX:
            atomicMax((unsigned long long *)&dst[2 * i],
                      (unsigned long long)src[2 * i]);
            break;
        case AMD_AGG_AVG: {
            atomicAdd((unsigned long long *)&dst[2 * i],
                      (unsigned long long)src[2 * i]);
            double v;
            memcpy(&v, &src[2 * i + 1], 8);
            atomicAdd((double *)&dst[2 * i + 1], v);
            break;
        }
        }
    }
}

/* merge partial state words into a session's state (single thread) */
__device__ inline void state_merge(uint64_t *dst, const uint64_t *src,
                                   const AggSpec &a) {
    for (int i = 0; i < a.n_aggs; i++) {
        switch (a.op[i]) {
        case AMD_AGG_COUNT:
        case AMD_AGG_SUM:
            dst[2 * i] += src[2 * i];
            break;
        case AMD_AGG_MIN:
        case AMD_AGG_MAX:
            if (src[2 * i] > dst[2 * i]) dst[2 * i] = src[2 * i];
            break;
        case AMD_AGG_AVG: {
            dst[2 * i] += src[2 * i];
            double x, y;
            memcpy(&x, &dst[2 * i + 1], 8);
            memcpy(&y, &src[2 * i + 1], 8);
            x += y;
            memcpy(&dst[2 * i + 1], &x, 8);
            break;
        }
        }
    }
}

/* claim-or-find the open-addressing slot for `key`; spec slot = index C for
 * key == EMPTY_KEY (the sentinel value is a legal key). */
__device__ inline int64_t key_slot(int64_t *keys, uint32_t C, int64_t key,
                                   int *err, int which_err) {
    if (key == EMPTY_KEY) return (int64_t)C;
    uint64_t m = C - 1;
    uint64_t j = hash64((uint64_t)key) & m;
    for (uint32_t probes = 0; probes < C; probes++) {
        int64_t cur = keys[j];
        if (cur == key) return (int64_t)j;
        if (cur == EMPTY_KEY) {
            int64_t old = (int64_t)atomicCAS(
                (unsigned long long *)&keys[j],
                (unsigned long long)EMPTY_KEY, (unsigned long long)key);
            if (old == EMPTY_KEY || old == key) return (int64_t)j;
        }
        j = (j + 1) & m;
    }
    *err = which_err;
    return -1;
}

struct UpdateArgs {
    const int64_t *cols[12];  /* [key?], vals..., ts */
    int32_t n_keys, n_vals;
    int64_t n_rows;
    uint64_t ts_offset;
    int has_wm; uint64_t wm;
    /* per-batch pre-agg table, B slots + 1 spec */
    int64_t *bkeys;
    uint64_t *bst;            /* [B+1][2 + n_aggs*2]: min_ts,max_ts,aggs */
    uint32_t B;
    AggSpec agg;
    /* COUNT DISTINCT: batch value chain (raw multiset per batch key; the
     * single-writer phase-2 thread dedupes into the session's region) */
    int32_t cd_agg;           /* agg index, -1 when absent */
    int64_t *bv_val;
    int32_t *bv_next;
    unsigned long long *bv_cur;
    int64_t bv_cap;
    int *err;
};

/* LDS-staged: a per-workgroup table absorbs hot keys (Zipf streams send a
 * large share of a batch's rows to one key, and serialized same-address
 * HBM atomics were ~70 us of a 64K-row batch) before flushing distinct
 * entries into the per-batch global table.  Same pattern as the window
 * path's k_update_lds.  LDS entry = key + (min,max,aggs) encoded words. */
#define SESS_LDS_SLOTS 512

__global__ void __launch_bounds__(256)
k_sess_update(UpdateArgs A) {
    __shared__ int64_t lkey[SESS_LDS_SLOTS];
    extern __shared__ uint64_t lst[];    /* [SLOTS][2 + n_aggs*2] */
    int sw = 2 + 2 * A.agg.n_aggs;
    for (int i = threadIdx.x; i < SESS_LDS_SLOTS; i += blockDim.x) {
        lkey[i] = EMPTY_KEY;
        for (int w = 0; w < sw; w++) lst[(size_t)i * sw + w] = 0;
    }
    __syncthreads();
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t *ts = A.cols[A.n_keys + A.n_vals];
    const int64_t *vcols[AMD_MAX_AGGS];
    for (int i = 0; i < A.agg.n_aggs; i++)
        vcols[i] = A.agg.col[i] >= 0 ? A.cols[A.n_keys + A.agg.col[i]]
                                     : nullptr;
    for (int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         r < A.n_rows; r += stride) {
        uint64_t t = (uint64_t)ts[r] + A.ts_offset;
        if (A.has_wm && t < A.wm) continue;  /* late: silently dropped */
        int64_t key = A.n_keys ? A.cols[0][r] : 0;
        /* LDS first (hot keys collapse here); EMPTY_KEY sentinel collides
         * with a real key of -1: that rare key goes straight to global.
         * COUNT DISTINCT rows need the global slot for the value chain, so
         * LDS staging is bypassed. */
        bool done = false;
        if (key != EMPTY_KEY && A.cd_agg < 0) {
            uint32_t h = (uint32_t)hash64((uint64_t)key * 0x9e37u) &
                         (SESS_LDS_SLOTS - 1);
            for (int pr = 0; pr < 4 && !done; pr++) {
                uint32_t sl = (h + pr) & (SESS_LDS_SLOTS - 1);
                int64_t k = lkey[sl];
                if (k == EMPTY_KEY) {
                    int64_t old = (int64_t)atomicCAS(
                        (unsigned long long *)&lkey[sl],
                        (unsigned long long)EMPTY_KEY,
                        (unsigned long long)key);
                    k = old == EMPTY_KEY ? key : old;
                }
                if (k == key) {
                    uint64_t *st = lst + (size_t)sl * sw;
                    atomicMax((unsigned long long *)&st[0],
                              (unsigned long long)enc_min((int64_t)t));
                    atomicMax((unsigned long long *)&st[1],
                              (unsigned long long)enc_max((int64_t)t));
                    atomic_fold(st + 2, A.agg, vcols, r);
                    done = true;
                }
            }
        }
        if (!done) {
            int64_t s = key_slot(A.bkeys, A.B, key, A.err, SERR_BATCH_FULL);
            if (s < 0) continue;
            uint64_t *st = A.bst + (size_t)s * sw;
            atomicMax((unsigned long long *)&st[0],
                      (unsigned long long)enc_min((int64_t)t));
            atomicMax((unsigned long long *)&st[1],
                      (unsigned long long)enc_max((int64_t)t));
            atomic_fold(st + 2, A.agg, vcols, r);
            if (A.cd_agg >= 0) {
                int64_t v = A.cols[A.n_keys + A.agg.col[A.cd_agg]][r];
                int64_t idx = (int64_t)atomicAdd(A.bv_cur, 1ULL);
                if (idx >= A.bv_cap) { *A.err = SERR_BV_POOL; continue; }
                A.bv_val[idx] = v;
                A.bv_next[idx] = (int32_t)(
                    (int64_t)atomicExch(
                        (unsigned long long *)&st[2 + 2 * A.cd_agg],
                        (unsigned long long)(idx + 1)) - 1);
            }
        }
    }
    __syncthreads();
    /* flush LDS entries into the per-batch global table */
    for (int i = threadIdx.x; i < SESS_LDS_SLOTS; i += blockDim.x) {
        int64_t key = lkey[i];
        if (key == EMPTY_KEY) continue;
        int64_t s = key_slot(A.bkeys, A.B, key, A.err, SERR_BATCH_FULL);
        if (s < 0) continue;
        uint64_t *dst = A.bst + (size_t)s * sw;
        const uint64_t *src = lst + (size_t)i * sw;
        atomicMax((unsigned long long *)&dst[0],
                  (unsigned long long)src[0]);
        atomicMax((unsigned long long *)&dst[1],
                  (unsigned long long)src[1]);
        state_merge_atomic(dst + 2, src + 2, A.agg);
    }
}

/* quad-batched phase-1 kernel (keyed, no COUNT DISTINCT): the same
 * latency treatment the window path's k_update_batch measured out — four
 * rows per thread, sequential same-key runs collapsed, a small LDS hot
 * cache for the Zipf head, and the remaining rows' first probe loads
 * issued together before any is resolved.  Every state update is a
 * no-return atomic (atomicMax / atomicAdd), so the probe is the only
 * latency chain. */
__device__ inline int64_t sess_probe_resolve(int64_t *keys, uint32_t B,
                                             int *err, int64_t key,
                                             uint64_t j, int64_t kk) {
    uint64_t m = B - 1;
    for (uint32_t probes = 0; probes < B; probes++) {
        if (kk == key) return (int64_t)j;
        if (kk == EMPTY_KEY) {
            int64_t old = (int64_t)atomicCAS((unsigned long long *)&keys[j],
                                             (unsigned long long)EMPTY_KEY,
                                             (unsigned long long)key);
            if (old == EMPTY_KEY || old == key) return (int64_t)j;
        }
        j = (j + 1) & m;
        kk = keys[j];
    }
    *err = SERR_BATCH_FULL;
    return -1;
}

__global__ void __launch_bounds__(256)
k_sess_update_batch(UpdateArgs A) {
    __shared__ int64_t lkey[SESS_LDS_SLOTS];
    extern __shared__ uint64_t lst[];    /* [SLOTS][2 + n_aggs*2] */
    int sw = 2 + 2 * A.agg.n_aggs;
    for (int i = threadIdx.x; i < SESS_LDS_SLOTS; i += blockDim.x) {
        lkey[i] = EMPTY_KEY;
        for (int w = 0; w < sw; w++) lst[(size_t)i * sw + w] = 0;
    }
    __syncthreads();
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const int64_t *ts = A.cols[A.n_keys + A.n_vals];
    const int64_t *vcols[AMD_MAX_AGGS];
    for (int i = 0; i < A.agg.n_aggs; i++)
        vcols[i] = A.agg.col[i] >= 0 ? A.cols[A.n_keys + A.agg.col[i]]
                                     : nullptr;
    const uint64_t m = A.B - 1;
    int64_t nq = A.n_rows / 4;
    for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nq;
         v += stride) {
        int64_t key[4];
        uint64_t traw[4];
        for (int j = 0; j < 2; j++) {
            ulonglong2 tp = ((const ulonglong2 *)ts)[2 * v + j];
            ulonglong2 kp = ((const ulonglong2 *)A.cols[0])[2 * v + j];
            traw[2 * j] = tp.x;
            traw[2 * j + 1] = tp.y;
            key[2 * j] = (int64_t)kp.x;
            key[2 * j + 1] = (int64_t)kp.y;
        }
        bool need[4];
        for (int j = 0; j < 4; j++) {
            traw[j] += A.ts_offset;
            need[j] = !(A.has_wm && traw[j] < A.wm) && key[j] != EMPTY_KEY;
            /* sentinel-valued key (-1): rare, straight to the global
             * table's spec slot */
            if (!(A.has_wm && traw[j] < A.wm) && key[j] == EMPTY_KEY) {
                uint64_t *st = A.bst + (size_t)A.B * sw;
                atomicMax((unsigned long long *)&st[0],
                          (unsigned long long)enc_min((int64_t)traw[j]));
                atomicMax((unsigned long long *)&st[1],
                          (unsigned long long)enc_max((int64_t)traw[j]));
                atomic_fold(st + 2, A.agg, vcols, 4 * v + j);
            }
        }
        /* LDS hot cache (2-probe) */
        for (int j = 0; j < 4; j++) {
            if (!need[j]) continue;
            uint32_t h = (uint32_t)hash64((uint64_t)key[j] * 0x9e37u) &
                         (SESS_LDS_SLOTS - 1);
            for (int pr = 0; pr < 2; pr++) {
                uint32_t sl = (h + pr) & (SESS_LDS_SLOTS - 1);
                int64_t k = lkey[sl];
                if (k == EMPTY_KEY) {
                    int64_t old = (int64_t)atomicCAS(
                        (unsigned long long *)&lkey[sl],
                        (unsigned long long)EMPTY_KEY,
                        (unsigned long long)key[j]);
                    k = old == EMPTY_KEY ? key[j] : old;
                }
                if (k == key[j]) {
                    uint64_t *st = lst + (size_t)sl * sw;
                    atomicMax((unsigned long long *)&st[0],
                              (unsigned long long)enc_min((int64_t)traw[j]));
                    atomicMax((unsigned long long *)&st[1],
                              (unsigned long long)enc_max((int64_t)traw[j]));
                    atomic_fold(st + 2, A.agg, vcols, 4 * v + j);
                    need[j] = false;
                    break;
                }
            }
        }
        /* issue the remaining rows' first probe loads together */
        uint64_t h[4];
        int64_t firstk[4];
        for (int j = 0; j < 4; j++)
            if (need[j]) {
                h[j] = hash64((uint64_t)key[j]) & m;
                firstk[j] = A.bkeys[h[j]];
            }
        for (int j = 0; j < 4; j++) {
            if (!need[j]) continue;
            int64_t s = sess_probe_resolve(A.bkeys, A.B, A.err, key[j],
                                           h[j], firstk[j]);
            if (s < 0) continue;
            uint64_t *st = A.bst + (size_t)s * sw;
            atomicMax((unsigned long long *)&st[0],
                      (unsigned long long)enc_min((int64_t)traw[j]));
            atomicMax((unsigned long long *)&st[1],
                      (unsigned long long)enc_max((int64_t)traw[j]));
            atomic_fold(st + 2, A.agg, vcols, 4 * v + j);
        }
    }
    __syncthreads();
    for (int i = threadIdx.x; i < SESS_LDS_SLOTS; i += blockDim.x) {
        int64_t key = lkey[i];
        if (key == EMPTY_KEY) continue;
        int64_t s = key_slot(A.bkeys, A.B, key, A.err, SERR_BATCH_FULL);
        if (s < 0) continue;
        uint64_t *dst = A.bst + (size_t)s * sw;
        const uint64_t *src = lst + (size_t)i * sw;
        atomicMax((unsigned long long *)&dst[0],
                  (unsigned long long)src[0]);
        atomicMax((unsigned long long *)&dst[1],
                  (unsigned long long)src[1]);
        state_merge_atomic(dst + 2, src + 2, A.agg);
    }
}


struct Store {
    int64_t *keys;     /* [C+1]; slot C = spec (key == EMPTY_KEY) */
    /* Occupancy stays a DENSE u32 plane (the watermark fire scans every
     * slot: strided reads through the records cost ~1 GB of line traffic
     * per fire, measured 765 us — the dense plane scans at full
     * bandwidth).  Live sessions sit in a per-slot interleaved record
     * [(start, end, st[sw]) x MS] so the phase-2 merge touches ~2 random
     * lines per key (key probe + record) instead of the 5 the split
     * planes cost (profiles/r01_session_note.md). */
    uint32_t *ns;      /* [C+1] live sessions per key (dense) */
    /* live-slot index: the watermark fire visits only slots with ns > 0
     * instead of scanning all C+1 (the dense scan wasted 7/8 of its
     * threads at the bench occupancy).  merge_partial appends a slot on
     * its 0 -> n transition; the fire compacts survivors into the flip
     * buffer (no duplicates: n reaches 0 only inside a fire). */
    uint32_t *live;               /* append target for merges */
    unsigned long long *live_n;
    int64_t *recs;     /* [(C+1)][rec_w] */
    uint32_t C, MS;
    uint32_t sess_w;   /* words per session = 2 + 2*n_aggs */
    uint32_t rec_w;    /* words per record  = MS*sess_w padded to 64B */
};

struct MergeArgs {
    /* batch pre-agg table (cleared inline by k_sess_merge) */
    int64_t *bkeys;
    uint64_t *bst;
    unsigned long long *bv_cur;   /* reset with the table (may be null) */
    uint32_t B;
    Store store;
    uint64_t gap;
    AggSpec agg;
    int32_t cd_agg;
    CdPool cd;
    const int64_t *bv_val;
    const int32_t *bv_next;
    int *err;
};

/* one thread per occupied pre-agg slot: merge the partial session into the
 * store.  No two threads share a key, so session-list surgery is plain
 * single-threaded code. */
__device__ inline void merge_partial(const MergeArgs &M, int64_t key,
                                     int64_t pmin, int64_t pmax,
                                     const uint64_t *pst, int *err) {
    const Store &S = M.store;
    int64_t slot = key_slot(S.keys, S.C, key, err, SERR_TABLE_FULL);
    if (slot < 0) return;
    int sw = 2 * M.agg.n_aggs;
    int64_t *rec = S.recs + (size_t)slot * S.rec_w;
    uint32_t n = S.ns[slot];
    /* absorb every stored session whose gap-closure touches the partial's */
    int64_t cs = pmin, ce = pmax;
    uint64_t acc[AMD_MAX_AGGS * 2];
    for (int i = 0; i < sw; i++) acc[i] = pst[i];
    if (M.cd_agg >= 0) {
        /* the partial's CD word is the batch chain head, folded in below;
         * acc starts with no region */
        acc[2 * M.cd_agg] = 0;
        acc[2 * M.cd_agg + 1] = 0;
    }
    uint32_t w = 0;
    for (uint32_t i = 0; i < n; i++) {
        int64_t *si = rec + (size_t)i * S.sess_w;
        const uint64_t *sist = (const uint64_t *)(si + 2);
        if (cs < si[1] + (int64_t)M.gap && si[0] < ce + (int64_t)M.gap) {
            if (si[0] < cs) cs = si[0];
            if (si[1] > ce) ce = si[1];
            if (M.cd_agg >= 0) {
                uint64_t *aw = acc + 2 * M.cd_agg;
                const uint64_t *sv = sist + 2 * M.cd_agg;
                cd_merge_into(M.cd, aw, sv[0], sv[1], err);
                /* scalar merge must skip the CD words: zero them on the
                 * source copy path by merging around */
                uint64_t saved0 = aw[0], saved1 = aw[1];
                state_merge(acc, sist, M.agg);
                aw[0] = saved0;
                aw[1] = saved1;
            } else {
                state_merge(acc, sist, M.agg);
            }
        } else {
            if (w != i) {
                int64_t *sd = rec + (size_t)w * S.sess_w;
                for (uint32_t k = 0; k < S.sess_w; k++) sd[k] = si[k];
            }
            w++;
        }
    }
    if (w >= S.MS) { *err = SERR_SESSIONS; return; }
    if (M.cd_agg >= 0 && pst[2 * M.cd_agg] != 0) {
        /* fold the batch's raw value chain into the merged session's set */
        uint64_t *aw = acc + 2 * M.cd_agg;
        if (aw[1] == 0) {
            int64_t r = cd_alloc(M.cd, err);
            if (r < 0) return;
            aw[1] = (uint64_t)(r + 1);
        }
        for (int32_t j = (int32_t)((int64_t)pst[2 * M.cd_agg] - 1); j >= 0;
             j = M.bv_next[j])
            aw[0] += cd_insert(M.cd, (int64_t)aw[1] - 1, M.bv_val[j], err);
    }
    int64_t *wr = rec + (size_t)w * S.sess_w;
    wr[0] = cs;
    wr[1] = ce;
    for (int k = 0; k < sw; k++) ((uint64_t *)(wr + 2))[k] = acc[k];
    S.ns[slot] = w + 1;
    if (n == 0 && S.live) {
        unsigned long long p = atomicAdd(S.live_n, 1ULL);
        S.live[p] = (uint32_t)slot;
    }
}

__global__ void __launch_bounds__(256)
k_sess_merge(MergeArgs M) {
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int sw = 2 + 2 * M.agg.n_aggs;
    for (int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         s <= (int64_t)M.B; s += stride) {
        uint64_t *st = M.bst + (size_t)s * sw;
        if (st[0] == 0 && st[1] == 0) continue;      /* untouched slot */
        int64_t key = s == (int64_t)M.B ? EMPTY_KEY : M.bkeys[s];
        if (s != (int64_t)M.B && key == EMPTY_KEY) continue;
        merge_partial(M, key, dec_min(st[0]), dec_max(st[1]), st + 2, M.err);
        /* clear the slot inline (the lines are already dirty): the next
         * batch starts from an empty table without the two 5 MB fill
         * launches per batch this scan used to require */
        for (int w = 0; w < sw; w++) st[w] = 0;
        if (s != (int64_t)M.B) M.bkeys[s] = EMPTY_KEY;
    }
    if (M.bv_cur && blockIdx.x == 0 && threadIdx.x == 0) *M.bv_cur = 0;
}

struct FireArgs {
    Store store;
    uint64_t gap, wm;
    AggSpec agg;
    int64_t *out[AMD_MAX_AGGS + 4];
    unsigned long long *n_out;
    int64_t out_cap;
    int32_t n_keys;
    int *err;
};

__global__ void __launch_bounds__(256)
k_sess_fire(FireArgs F) {
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const Store &S = F.store;
    int sw = 2 * F.agg.n_aggs;
    for (int64_t slot = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         slot <= (int64_t)S.C; slot += stride) {
        uint32_t n = S.ns[slot];
        if (n == 0) continue;
        if (slot < (int64_t)S.C && S.keys[slot] == EMPTY_KEY) continue;
        int64_t key = slot == (int64_t)S.C ? EMPTY_KEY : S.keys[slot];
        int64_t *rec = S.recs + (size_t)slot * S.rec_w;
        uint32_t w = 0;
        for (uint32_t i = 0; i < n; i++) {
            int64_t *si = rec + (size_t)i * S.sess_w;
            const uint64_t *sist = (const uint64_t *)(si + 2);
            uint64_t close = (uint64_t)(si[1] + (int64_t)F.gap);
            if (close < F.wm) {
                int64_t r = (int64_t)atomicAdd(F.n_out, 1ULL);
                if (r >= F.out_cap) { *F.err = SERR_OUT_CAP; continue; }
                int col = 0;
                if (F.n_keys) F.out[col++][r] = key;
                for (int a = 0; a < F.agg.n_aggs; a++, col++) {
                    uint64_t w0 = sist[2 * a];
                    uint64_t w1 = sist[2 * a + 1];
                    int64_t v;
                    switch (F.agg.op[a]) {
                    case AMD_AGG_COUNT:
                    case AMD_AGG_SUM:
                    case AMD_AGG_COUNT_DISTINCT: v = (int64_t)w0; break;
                    case AMD_AGG_MIN: v = dec_min(w0); break;
                    case AMD_AGG_MAX: v = dec_max(w0); break;
                    default: {  /* AVG */
                        double sum;
                        memcpy(&sum, &w1, 8);
                        double avg = sum / (double)w0;
                        memcpy(&v, &avg, 8);
                        break;
                    }
                    }
                    F.out[col][r] = v;
                }
                F.out[col++][r] = si[0];
                F.out[col++][r] = (int64_t)close;
                F.out[col][r] = (int64_t)close - 1;
            } else {
                if (w != i) {
                    int64_t *sd = rec + (size_t)w * S.sess_w;
                    for (uint32_t k = 0; k < S.sess_w; k++) sd[k] = si[k];
                }
                w++;
            }
        }
        if (w != n) S.ns[slot] = w;
    }
}

struct FireIdxArgs {
    FireArgs f;
    const uint32_t *old_idx;
    const unsigned long long *old_n;
    uint32_t *new_idx;            /* survivors compact here */
    unsigned long long *new_n;
};

__global__ void __launch_bounds__(256)
k_sess_fire_idx(FireIdxArgs X) {
    const FireArgs &F = X.f;
    const Store &S = F.store;
    int64_t total = (int64_t)*X.old_n;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t t = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         t < total; t += stride) {
        int64_t slot = (int64_t)X.old_idx[t];
        uint32_t n = S.ns[slot];
        if (n == 0) continue;     /* emptied by an earlier fire pass */
        int64_t key = slot == (int64_t)S.C ? EMPTY_KEY : S.keys[slot];
        int64_t *rec = S.recs + (size_t)slot * S.rec_w;
        uint32_t w = 0;
        for (uint32_t i = 0; i < n; i++) {
            int64_t *si = rec + (size_t)i * S.sess_w;
            const uint64_t *sist = (const uint64_t *)(si + 2);
            uint64_t close = (uint64_t)(si[1] + (int64_t)F.gap);
            if (close < F.wm) {
                int64_t r = (int64_t)atomicAdd(F.n_out, 1ULL);
                if (r >= F.out_cap) { *F.err = SERR_OUT_CAP; continue; }
                int col = 0;
                if (F.n_keys) F.out[col++][r] = key;
                for (int a = 0; a < F.agg.n_aggs; a++, col++) {
                    uint64_t w0 = sist[2 * a];
                    uint64_t w1 = sist[2 * a + 1];
                    int64_t v;
                    switch (F.agg.op[a]) {
                    case AMD_AGG_COUNT:
                    case AMD_AGG_SUM:
                    case AMD_AGG_COUNT_DISTINCT: v = (int64_t)w0; break;
                    case AMD_AGG_MIN: v = dec_min(w0); break;
                    case AMD_AGG_MAX: v = dec_max(w0); break;
                    default: {  /* AVG */
                        double sum;
                        memcpy(&sum, &w1, 8);
                        double avg = sum / (double)w0;
                        memcpy(&v, &avg, 8);
                        break;
                    }
                    }
                    F.out[col][r] = v;
                }
                F.out[col++][r] = si[0];
                F.out[col++][r] = (int64_t)close;
                F.out[col][r] = (int64_t)close - 1;
            } else {
                if (w != i) {
                    int64_t *sd = rec + (size_t)w * S.sess_w;
                    for (uint32_t k = 0; k < S.sess_w; k++) sd[k] = si[k];
                }
                w++;
            }
        }
        if (w != n) S.ns[slot] = w;
        if (w) {
            unsigned long long p = atomicAdd(X.new_n, 1ULL);
            X.new_idx[p] = (uint32_t)slot;
        }
    }
}

/* checkpoint drain: one thread per key slot appends its live sessions */
struct DrainArgs {
    Store store;
    int64_t *out[AMD_MAX_AGGS * 2 + 3];
    unsigned long long *n_out;
    int64_t out_cap;
    int32_t n_keys, n_aggs;
    int *err;
};

__global__ void __launch_bounds__(256)
k_sess_drain(DrainArgs D) {
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    const Store &S = D.store;
    int sw = 2 * D.n_aggs;
    for (int64_t slot = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         slot <= (int64_t)S.C; slot += stride) {
        uint32_t n = S.ns[slot];
        if (n == 0) continue;
        if (slot < (int64_t)S.C && S.keys[slot] == EMPTY_KEY) continue;
        int64_t key = slot == (int64_t)S.C ? EMPTY_KEY : S.keys[slot];
        const int64_t *rec = S.recs + (size_t)slot * S.rec_w;
        int64_t base = (int64_t)atomicAdd(D.n_out, (unsigned long long)n);
        if (base + n > D.out_cap) { *D.err = SERR_OUT_CAP; continue; }
        for (uint32_t i = 0; i < n; i++) {
            const int64_t *si = rec + (size_t)i * S.sess_w;
            int64_t r = base + i;
            int col = 0;
            if (D.n_keys) D.out[col++][r] = key;
            for (int k = 0; k < sw; k++)
                D.out[col++][r] = si[2 + k];
            D.out[col++][r] = si[0];
            D.out[col][r] = si[1];
        }
    }
}

/* restore: rows pre-grouped by key on the host; one thread per key group
 * replays its partial sessions through the same merge path */
struct RestoreArgs {
    const int64_t *keys;      /* [n] (ignored when n_keys == 0) */
    const uint64_t *st;       /* [n][n_aggs*2] */
    const int64_t *start;     /* [n] */
    const int64_t *end;       /* [n] */
    const int64_t *group_off; /* [n_groups+1] */
    int64_t n_groups;
    Store store;
    uint64_t gap;
    AggSpec agg;
    int32_t n_keys;
    int32_t cd_agg;
    CdPool cd;
    int *err;
};

__global__ void __launch_bounds__(256)
k_sess_restore(RestoreArgs R) {
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int sw = 2 * R.agg.n_aggs;
    MergeArgs M = {};
    M.store = R.store;
    M.gap = R.gap;
    M.agg = R.agg;
    M.cd_agg = R.cd_agg;
    M.cd = R.cd;
    M.err = R.err;
    for (int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < R.n_groups; g += stride) {
        for (int64_t i = R.group_off[g]; i < R.group_off[g + 1]; i++) {
            int64_t key = R.n_keys ? R.keys[i] : 0;
            merge_partial(M, key, R.start[i], R.end[i],
                          R.st + (size_t)i * sw, R.err);
        }
    }
}

/* ---- COUNT DISTINCT checkpoint value stream ----
 * drain: one row per live (session, value): [key?, session_start, value];
 * restore: host-grouped by key, one thread per key re-inserts (the
 * restored sessions' CD words start zeroed; counts rebuild here). */
struct CdDrainArgs {
    Store store;
    AggSpec agg;
    int32_t cd_agg, n_keys;
    CdPool cd;
    int64_t *out[3];
    unsigned long long *n_out;
    int64_t out_cap;
    int *err;
};

__global__ void __launch_bounds__(256)
k_sess_drain_values(CdDrainArgs D) {
    const Store &S = D.store;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int sw = 2 * D.agg.n_aggs;
    for (int64_t slot = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         slot <= (int64_t)S.C; slot += stride) {
        uint32_t n = S.ns[slot];
        if (n == 0) continue;
        if (slot < (int64_t)S.C && S.keys[slot] == EMPTY_KEY) continue;
        int64_t key = slot == (int64_t)S.C ? EMPTY_KEY : S.keys[slot];
        const int64_t *rec = S.recs + (size_t)slot * S.rec_w;
        for (uint32_t i = 0; i < n; i++) {
            const int64_t *si = rec + (size_t)i * S.sess_w;
            uint64_t reg1 = (uint64_t)si[2 + 2 * D.cd_agg + 1];
            if (reg1 == 0) continue;
            const uint64_t *bm =
                D.cd.regions + (size_t)(reg1 - 1) * cd_region_words(D.cd.D);
            const int64_t *vals = (const int64_t *)(bm + D.cd.D / 64);
            for (uint32_t sl = 0; sl < D.cd.D; sl++) {
                if (!(bm[sl >> 6] & (1ULL << (sl & 63)))) continue;
                int64_t r = (int64_t)atomicAdd(D.n_out, 1ULL);
                if (r >= D.out_cap) { *D.err = SERR_OUT_CAP; continue; }
                int col = 0;
                if (D.n_keys) D.out[col++][r] = key;
                D.out[col++][r] = si[0];
                D.out[col][r] = vals[sl];
            }
        }
    }
}

struct CdRestoreArgs {
    const int64_t *keys;
    const int64_t *start;
    const int64_t *value;
    const int64_t *group_off;
    int64_t n_groups;
    Store store;
    AggSpec agg;
    int32_t cd_agg, n_keys;
    CdPool cd;
    int *err;
};

__global__ void __launch_bounds__(256)
k_sess_restore_values(CdRestoreArgs R) {
    const Store &S = R.store;
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    int sw = 2 * R.agg.n_aggs;
    for (int64_t g = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         g < R.n_groups; g += stride) {
        for (int64_t i = R.group_off[g]; i < R.group_off[g + 1]; i++) {
            int64_t key = R.n_keys ? R.keys[i] : 0;
            int64_t slot = key_slot(S.keys, S.C, key, R.err,
                                    SERR_TABLE_FULL);
            if (slot < 0) continue;
            int64_t *rec = S.recs + (size_t)slot * S.rec_w;
            uint32_t n = S.ns[slot];
            for (uint32_t si = 0; si < n; si++) {
                int64_t *sp = rec + (size_t)si * S.sess_w;
                if (sp[0] != R.start[i])
                    continue;
                uint64_t *w = (uint64_t *)(sp + 2) + 2 * R.cd_agg;
                if (w[1] == 0) {
                    int64_t r = cd_alloc(R.cd, R.err);
                    if (r < 0) break;
                    w[1] = (uint64_t)(r + 1);
                }
                w[0] += cd_insert(R.cd, (int64_t)w[1] - 1, R.value[i],
                                  R.err);
                break;
            }
        }
    }
}

}  // namespace sess

using namespace sess;

static char g_sess_err[256];

struct GpuSession {
    AmdSessionConfig cfg;
    AggSpec agg;
    Store store;
    int64_t *bkeys;
    uint64_t *bst;
    uint32_t B;
    /* COUNT DISTINCT state */
    int32_t cd_agg;            /* -1 when absent */
    CdPool cd;
    int64_t *bv_val;
    int32_t *bv_next;
    unsigned long long *bv_cur;
    int64_t *d_out[AMD_MAX_AGGS * 2 + 4];
    unsigned long long *d_n_out;
    int *d_err;
    int64_t *stg_h[12], *stg_d[12];
    int64_t stg_cap;
    int n_in_cols, out_cols, drain_cols;
    int64_t out_cap;
    /* live-slot index flip buffers (see Store.live) */
    uint32_t *live_buf[2];
    unsigned long long *live_nbuf[2];
    int live_cur;
    int use_live;
    int has_wm; uint64_t wm;
    hipStream_t stream;
    char err_msg[512];
};

#define SHIP(o, call)                                                         \
    do {                                                                      \
        hipError_t _e = (call);                                               \
        if (_e != hipSuccess) {                                               \
            snprintf((o)->err_msg, sizeof (o)->err_msg, "%s:%d hip: %s",      \
                     __FILE__, __LINE__, hipGetErrorString(_e));              \
            return 1;                                                         \
        }                                                                     \
    } while (0)

API void *arroyo_amd_session_create(const AmdSessionConfig *cfg) {
    if (!cfg || cfg->n_keys < 0 || cfg->n_keys > 1 || cfg->n_value_cols < 0 ||
        cfg->n_value_cols > 8 || cfg->n_aggs < 1 ||
        cfg->n_aggs > AMD_MAX_AGGS || cfg->gap_nanos == 0) {
        snprintf(g_sess_err, sizeof g_sess_err, "invalid session config");
        return nullptr;
    }
    int cd_agg = -1;
    for (int a = 0; a < cfg->n_aggs; a++)
        if (cfg->agg_ops[a] == AMD_AGG_COUNT_DISTINCT) {
            if (cd_agg >= 0 || cfg->agg_col[a] < 0) {
                snprintf(g_sess_err, sizeof g_sess_err,
                         "at most one COUNT DISTINCT aggregate, over a "
                         "value column");
                return nullptr;
            }
            cd_agg = a;
        }
    GpuSession *o = new GpuSession();
    o->cd_agg = cd_agg;
    o->cfg = *cfg;
    o->agg.n_aggs = cfg->n_aggs;
    for (int i = 0; i < cfg->n_aggs; i++) {
        o->agg.op[i] = cfg->agg_ops[i];
        o->agg.col[i] = cfg->agg_col[i];
    }
    o->store.C = 1u << (cfg->log2_capacity ? cfg->log2_capacity : 16);
    o->store.MS = cfg->max_sessions ? cfg->max_sessions : 8;
    o->B = 1u << (cfg->log2_batch_capacity ? cfg->log2_batch_capacity : 14);
    o->out_cap = 1ll << (cfg->log2_out_cap ? cfg->log2_out_cap : 20);
    o->n_in_cols = cfg->n_keys + cfg->n_value_cols + 1;
    o->out_cols = cfg->n_keys + cfg->n_aggs + 3;
    o->drain_cols = cfg->n_keys + 2 * cfg->n_aggs + 2;
    if (hipSetDevice(cfg->device) != hipSuccess) {
        snprintf(g_sess_err, sizeof g_sess_err,
                 "hipSetDevice(%d) failed: no HIP device (no CPU fallback)",
                 cfg->device);
        delete o;
        return nullptr;
    }
    hipError_t e;
    auto fail = [&](const char *what, hipError_t e2) {
        snprintf(g_sess_err, sizeof g_sess_err, "%s: %s", what,
                 hipGetErrorString(e2));
        delete o;
        return nullptr;
    };
#define SALLOC(p, bytes)                                                      \
    if ((e = hipMalloc((void **)&(p), (bytes))) != hipSuccess)                \
        return fail(#p, e);
    size_t C1 = (size_t)o->store.C + 1;
    size_t MS = o->store.MS, sw = 2 * (size_t)cfg->n_aggs;
    o->store.sess_w = (uint32_t)(2 + sw);
    /* pad records to a 64B multiple so the first session never straddles
     * a line */
    o->store.rec_w = (uint32_t)((MS * (2 + sw) + 7) & ~7u);
    SALLOC(o->store.keys, C1 * 8);
    SALLOC(o->store.ns, C1 * 4);
    SALLOC(o->live_buf[0], C1 * 4);
    SALLOC(o->live_buf[1], C1 * 4);
    SALLOC(o->live_nbuf[0], 8);
    SALLOC(o->live_nbuf[1], 8);
    SALLOC(o->store.recs, C1 * o->store.rec_w * 8);
    SALLOC(o->bkeys, ((size_t)o->B + 1) * 8);
    SALLOC(o->bst, ((size_t)o->B + 1) * (2 + sw) * 8);
    int max_out = o->drain_cols > o->out_cols ? o->drain_cols : o->out_cols;
    for (int i = 0; i < max_out; i++)
        SALLOC(o->d_out[i], (size_t)o->out_cap * 8);
    SALLOC(o->d_n_out, 8);
    SALLOC(o->d_err, 4);
    if (cd_agg >= 0) {
        o->cd.D = 1u << (cfg->log2_distinct ? cfg->log2_distinct : 10);
        o->cd.n_regions =
            1ll << (cfg->log2_cd_regions ? cfg->log2_cd_regions : 14);
        size_t words = (size_t)(o->cd.D / 64 + o->cd.D);
        SALLOC(o->cd.regions, (size_t)o->cd.n_regions * words * 8);
        SALLOC(o->cd.rcur, 8);
        SALLOC(o->bv_val, (size_t)(1 << 20) * 8);
        SALLOC(o->bv_next, (size_t)(1 << 20) * 4);
        SALLOC(o->bv_cur, 8);
        hipMemset(o->cd.regions, 0,
                  (size_t)o->cd.n_regions * words * 8);
        hipMemset(o->cd.rcur, 0, 8);
        hipMemset(o->bv_cur, 0, 8);
    }
#undef SALLOC
    hipMemset(o->store.keys, 0xFF, C1 * 8);
    hipMemset(o->store.ns, 0, C1 * 4);
    hipMemset(o->live_nbuf[0], 0, 8);
    hipMemset(o->live_nbuf[1], 0, 8);
    o->live_cur = 0;
    o->use_live = 1;
    if (const char *ev = getenv("ARROYO_AMD_SESS_LIVE_IDX"))
        o->use_live = atoi(ev) != 0;
    if (o->use_live) {
        o->store.live = o->live_buf[0];
        o->store.live_n = o->live_nbuf[0];
    }
    hipMemset(o->store.recs, 0, C1 * o->store.rec_w * 8);
    hipMemset(o->bkeys, 0xFF, ((size_t)o->B + 1) * 8);
    hipMemset(o->bst, 0, ((size_t)o->B + 1) * (2 + sw) * 8);
    hipMemset(o->d_err, 0, 4);
    hipStreamCreate(&o->stream);
    o->stg_cap = 1 << 20;
    for (int c = 0; c < o->n_in_cols; c++) {
        if (hipHostMalloc((void **)&o->stg_h[c], (size_t)o->stg_cap * 8) !=
                hipSuccess ||
            hipMalloc((void **)&o->stg_d[c], (size_t)o->stg_cap * 8) !=
                hipSuccess) {
            snprintf(g_sess_err, sizeof g_sess_err,
                     "session staging alloc failed");
            delete o;
            return nullptr;
        }
    }
    return o;
}

API const char *arroyo_amd_session_last_error(void *h) {
    return h ? ((GpuSession *)h)->err_msg : g_sess_err;
}

static int sess_check_err(GpuSession *o) {
    int e = 0;
    SHIP(o, hipMemcpyAsync(&e, o->d_err, 4, hipMemcpyDeviceToHost,
                           o->stream));
    SHIP(o, hipStreamSynchronize(o->stream));
    if (!e) return 0;
    const char *msg =
        e == SERR_TABLE_FULL ? "session key table full; raise log2_capacity"
        : e == SERR_BATCH_FULL
            ? "per-batch table full; raise log2_batch_capacity"
        : e == SERR_SESSIONS
            ? "per-key live-session limit hit; raise max_sessions"
        : e == SERR_OUT_CAP ? "output buffer full; raise log2_out_cap"
        : e == SERR_CD_FULL
            ? "per-session distinct set full; raise log2_distinct"
        : e == SERR_CD_REGIONS
            ? "distinct-set region pool full; raise log2_cd_regions"
        : e == SERR_BV_POOL ? "batch value-chain pool full"
                            : "device error";
    snprintf(o->err_msg, sizeof o->err_msg, "%s", msg);
    return 1;
}

static int grid_for(int64_t want_threads) {
    int64_t want = (want_threads + 255) / 256;
    return (int)(want > 4096 ? 4096 : (want < 1 ? 1 : want));
}

/* one sub-batch whose event-time span is < gap */
static int sess_submit(GpuSession *o, const int64_t *const *dcols,
                       int64_t n_rows, uint64_t ts_offset) {
    if (n_rows == 0) return 0;
    /* no per-batch clears: k_sess_merge leaves the table empty behind it */
    UpdateArgs A = {};
    for (int c = 0; c < o->n_in_cols; c++) A.cols[c] = dcols[c];
    A.n_keys = o->cfg.n_keys;
    A.n_vals = o->cfg.n_value_cols;
    A.n_rows = n_rows;
    A.ts_offset = ts_offset;
    A.has_wm = o->has_wm;
    A.wm = o->wm;
    A.bkeys = o->bkeys;
    A.bst = o->bst;
    A.B = o->B;
    A.agg = o->agg;
    A.cd_agg = o->cd_agg;
    if (o->cd_agg >= 0) {
        A.bv_val = o->bv_val;
        A.bv_next = o->bv_next;
        A.bv_cur = o->bv_cur;
        A.bv_cap = 1 << 20;
    }
    A.err = o->d_err;
    size_t shmem = (size_t)SESS_LDS_SLOTS * (2 + 2 * o->cfg.n_aggs) * 8;
    int ublocks = grid_for(n_rows);
    if (const char *ev = getenv("ARROYO_AMD_SESS_BLOCKS"))
        if (atoi(ev) > 0) ublocks = atoi(ev);
    /* quad-batched path: keyed, no COUNT DISTINCT, 16B-aligned columns */
    bool batch = o->cfg.n_keys == 1 && o->cd_agg < 0 && n_rows >= 4;
    for (int c = 0; c < o->n_in_cols && batch; c++)
        batch = ((uintptr_t)dcols[c] & 15) == 0;
    if (const char *ev = getenv("ARROYO_AMD_SESS_BATCH"))
        if (!atoi(ev)) batch = false;
    if (batch) {
        int64_t main_rows = n_rows & ~3ll;
        UpdateArgs B4 = A;
        B4.n_rows = main_rows;
        int64_t want = (main_rows / 4 + 255) / 256;
        int qb = (int)(want > 2048 ? 2048 : (want < 1 ? 1 : want));
        if (const char *ev = getenv("ARROYO_AMD_SESS_BLOCKS"))
            if (atoi(ev) > 0) qb = atoi(ev);
        hipLaunchKernelGGL(k_sess_update_batch, dim3(qb), dim3(256), shmem,
                           o->stream, B4);
        SHIP(o, hipGetLastError());
        int64_t tail = n_rows - main_rows;
        if (tail) {
            UpdateArgs T = A;
            for (int c = 0; c < o->n_in_cols; c++)
                T.cols[c] = dcols[c] + main_rows;
            T.n_rows = tail;
            hipLaunchKernelGGL(k_sess_update, dim3(1), dim3(64), shmem,
                               o->stream, T);
            SHIP(o, hipGetLastError());
        }
    } else {
        hipLaunchKernelGGL(k_sess_update, dim3(ublocks), dim3(256),
                           shmem, o->stream, A);
        SHIP(o, hipGetLastError());
    }
    MergeArgs M = {};
    M.bkeys = o->bkeys;
    M.bst = o->bst;
    M.B = o->B;
    M.store = o->store;
    M.gap = o->cfg.gap_nanos;
    M.agg = o->agg;
    M.cd_agg = o->cd_agg;
    M.cd = o->cd;
    M.bv_val = o->bv_val;
    M.bv_next = o->bv_next;
    M.bv_cur = o->bv_cur;
    M.err = o->d_err;
    hipLaunchKernelGGL(k_sess_merge, dim3(grid_for((int64_t)o->B + 1)),
                       dim3(256), 0, o->stream, M);
    SHIP(o, hipGetLastError());
    return 0;
}

API int arroyo_amd_session_process_batch_device(void *h,
                                                const int64_t *const *dcols,
                                                int32_t n_cols,
                                                int64_t n_rows,
                                                uint64_t ts_offset) {
    GpuSession *o = (GpuSession *)h;
    if (n_cols != o->n_in_cols) {
        snprintf(o->err_msg, sizeof o->err_msg, "expected %d cols, got %d",
                 o->n_in_cols, n_cols);
        return 1;
    }
    /* the caller owns the guarantee that the batch's event-time span is
     * < gap on this path (bench-style streams: ms spans vs second gaps) */
    return sess_submit(o, dcols, n_rows, ts_offset);
}

/* contiguous multi-batch submission (bench replay rings): one C call
 * enqueues `reps` per-batch update+merge rounds -- the per-batch gap
 * logic is unchanged (each batch's table merges into the store before
 * the next batch's update), only the host boundary cost amortizes */
API int arroyo_amd_session_process_batches_device(void *h,
                                                  const int64_t *const *dcols,
                                                  int32_t n_cols,
                                                  int64_t n_rows,
                                                  int32_t reps,
                                                  uint64_t ts_offset) {
    GpuSession *o = (GpuSession *)h;
    if (n_cols != o->n_in_cols) {
        snprintf(o->err_msg, sizeof o->err_msg, "expected %d cols, got %d",
                 o->n_in_cols, n_cols);
        return 1;
    }
    const int64_t *cols[8];
    for (int32_t k = 0; k < reps; k++) {
        for (int c = 0; c < n_cols; c++)
            cols[c] = dcols[c] + (int64_t)k * n_rows;
        if (sess_submit(o, cols, n_rows, ts_offset)) return 1;
    }
    return 0;
}

API int arroyo_amd_session_process_batch(void *h, const int64_t *const *cols,
                                         int32_t n_cols, int64_t n_rows) {
    GpuSession *o = (GpuSession *)h;
    if (n_cols != o->n_in_cols) {
        snprintf(o->err_msg, sizeof o->err_msg, "expected %d cols, got %d",
                 o->n_in_cols, n_cols);
        return 1;
    }
    if (n_rows == 0) return 0;
    const int64_t *ts = cols[n_cols - 1];
    /* bucket rows by floor(ts / (gap/2)) so each sub-batch spans < gap;
     * sub-batch order is irrelevant (the merge is order-independent) */
    uint64_t bw = o->cfg.gap_nanos / 2;
    if (bw == 0) bw = 1;
    int64_t mn = ts[0], mx = ts[0];
    for (int64_t r = 1; r < n_rows; r++) {
        if (ts[r] < mn) mn = ts[r];
        if (ts[r] > mx) mx = ts[r];
    }
    std::vector<int64_t> order(n_rows);
    for (int64_t i = 0; i < n_rows; i++) order[i] = i;
    bool split = (uint64_t)(mx - mn) >= o->cfg.gap_nanos;
    if (split)
        std::stable_sort(order.begin(), order.end(),
                         [&](int64_t a, int64_t b) {
                             return (uint64_t)ts[a] / bw <
                                    (uint64_t)ts[b] / bw;
                         });
    int64_t done = 0;
    while (done < n_rows) {
        int64_t take = n_rows - done;
        if (split) {
            uint64_t b0 = (uint64_t)ts[order[done]] / bw;
            take = 1;
            while (done + take < n_rows &&
                   (uint64_t)ts[order[done + take]] / bw == b0)
                take++;
        }
        int64_t sent = 0;
        while (sent < take) {
            int64_t chunk = take - sent;
            if (chunk > o->stg_cap) chunk = o->stg_cap;
            const int64_t *dcols[12];
            for (int c = 0; c < o->n_in_cols; c++) {
                for (int64_t i = 0; i < chunk; i++)
                    o->stg_h[c][i] = cols[c][order[done + sent + i]];
                SHIP(o, hipMemcpyAsync(o->stg_d[c], o->stg_h[c],
                                       (size_t)chunk * 8,
                                       hipMemcpyHostToDevice, o->stream));
                dcols[c] = o->stg_d[c];
            }
            if (sess_submit(o, dcols, chunk, 0)) return 1;
            SHIP(o, hipStreamSynchronize(o->stream));
            sent += chunk;
        }
        done += take;
    }
    return 0;
}

API int arroyo_amd_session_handle_watermark(void *h, uint64_t wm,
                                            AmdOutBatch *out) {
    GpuSession *o = (GpuSession *)h;
    if (sess_check_err(o)) return 1;
    o->has_wm = 1;
    o->wm = wm;
    SHIP(o, hipMemsetAsync(o->d_n_out, 0, 8, o->stream));
    FireArgs F = {};
    F.store = o->store;
    F.gap = o->cfg.gap_nanos;
    F.wm = wm;
    F.agg = o->agg;
    for (int i = 0; i < o->out_cols; i++) F.out[i] = o->d_out[i];
    F.n_out = o->d_n_out;
    F.out_cap = o->out_cap;
    F.n_keys = o->cfg.n_keys;
    F.err = o->d_err;
    if (o->use_live) {
        /* fire over the live index, compacting survivors into the flip
         * buffer; subsequent merges append there */
        int nxt = 1 - o->live_cur;
        SHIP(o, hipMemsetAsync(o->live_nbuf[nxt], 0, 8, o->stream));
        FireIdxArgs X = {};
        X.f = F;
        X.old_idx = o->live_buf[o->live_cur];
        X.old_n = o->live_nbuf[o->live_cur];
        X.new_idx = o->live_buf[nxt];
        X.new_n = o->live_nbuf[nxt];
        hipLaunchKernelGGL(k_sess_fire_idx, dim3(2048), dim3(256), 0,
                           o->stream, X);
        SHIP(o, hipGetLastError());
        o->live_cur = nxt;
        o->store.live = o->live_buf[nxt];
        o->store.live_n = o->live_nbuf[nxt];
    } else {
        hipLaunchKernelGGL(k_sess_fire,
                           dim3(grid_for((int64_t)o->store.C + 1)),
                           dim3(256), 0, o->stream, F);
        SHIP(o, hipGetLastError());
    }
    unsigned long long n = 0;
    SHIP(o, hipMemcpyAsync(&n, o->d_n_out, 8, hipMemcpyDeviceToHost,
                           o->stream));
    SHIP(o, hipStreamSynchronize(o->stream));
    if (sess_check_err(o)) return 1;
    if (out) {
        memset(out, 0, sizeof *out);
        out->n_rows = (int64_t)n;
        out->n_cols = o->out_cols;
        out->cols = (void **)calloc(o->out_cols, sizeof(void *));
        out->is_f64 = (int32_t *)calloc(o->out_cols, sizeof(int32_t));
        for (int a = 0; a < o->cfg.n_aggs; a++)
            if (o->cfg.agg_ops[a] == AMD_AGG_AVG)
                out->is_f64[o->cfg.n_keys + a] = 1;
        for (int i = 0; i < o->out_cols; i++) {
            out->cols[i] = malloc((size_t)(n ? n : 1) * 8);
            if (n)
                SHIP(o, hipMemcpyAsync(out->cols[i], o->d_out[i],
                                       (size_t)n * 8, hipMemcpyDeviceToHost,
                                       o->stream));
        }
        SHIP(o, hipStreamSynchronize(o->stream));
    }
    return 0;
}

API int arroyo_amd_session_checkpoint_drain(void *h, AmdOutBatch *out) {
    GpuSession *o = (GpuSession *)h;
    if (sess_check_err(o)) return 1;
    SHIP(o, hipMemsetAsync(o->d_n_out, 0, 8, o->stream));
    DrainArgs D = {};
    D.store = o->store;
    for (int i = 0; i < o->drain_cols; i++) D.out[i] = o->d_out[i];
    D.n_out = o->d_n_out;
    D.out_cap = o->out_cap;
    D.n_keys = o->cfg.n_keys;
    D.n_aggs = o->cfg.n_aggs;
    D.err = o->d_err;
    hipLaunchKernelGGL(k_sess_drain,
                       dim3(grid_for((int64_t)o->store.C + 1)), dim3(256), 0,
                       o->stream, D);
    SHIP(o, hipGetLastError());
    unsigned long long n = 0;
    SHIP(o, hipMemcpyAsync(&n, o->d_n_out, 8, hipMemcpyDeviceToHost,
                           o->stream));
    SHIP(o, hipStreamSynchronize(o->stream));
    if (sess_check_err(o)) return 1;
    memset(out, 0, sizeof *out);
    out->n_rows = (int64_t)n;
    out->n_cols = o->drain_cols;
    out->cols = (void **)calloc(o->drain_cols, sizeof(void *));
    out->is_f64 = (int32_t *)calloc(o->drain_cols, sizeof(int32_t));
    for (int i = 0; i < o->drain_cols; i++) {
        out->cols[i] = malloc((size_t)(n ? n : 1) * 8);
        if (n)
            SHIP(o, hipMemcpyAsync(out->cols[i], o->d_out[i], (size_t)n * 8,
                                   hipMemcpyDeviceToHost, o->stream));
    }
    SHIP(o, hipStreamSynchronize(o->stream));
    return 0;
}

API int arroyo_amd_session_restore(void *h, const int64_t *const *cols,
                                   int32_t n_cols, int64_t n_rows) {
    GpuSession *o = (GpuSession *)h;
    if (n_cols != o->drain_cols) {
        snprintf(o->err_msg, sizeof o->err_msg,
                 "restore expects %d cols, got %d", o->drain_cols, n_cols);
        return 1;
    }
    if (n_rows == 0) return 0;
    int sw = 2 * o->cfg.n_aggs;
    /* group rows by key host-side: one device thread per key group */
    std::vector<int64_t> order(n_rows);
    for (int64_t i = 0; i < n_rows; i++) order[i] = i;
    const int64_t *keys = o->cfg.n_keys ? cols[0] : nullptr;
    if (keys)
        std::stable_sort(order.begin(), order.end(),
                         [&](int64_t a, int64_t b) {
                             return keys[a] < keys[b];
                         });
    std::vector<int64_t> h_keys(n_rows), h_start(n_rows), h_end(n_rows);
    std::vector<uint64_t> h_st((size_t)n_rows * sw);
    std::vector<int64_t> off;
    off.push_back(0);
    for (int64_t i = 0; i < n_rows; i++) {
        int64_t r = order[i];
        h_keys[i] = keys ? keys[r] : 0;
        if (i && keys && h_keys[i] != h_keys[i - 1]) off.push_back(i);
        for (int k = 0; k < sw; k++)
            h_st[(size_t)i * sw + k] =
                (uint64_t)cols[o->cfg.n_keys + k][r];
        if (o->cd_agg >= 0) {
            /* drained CD words are (count, region id) from the OLD op;
             * zero them -- restore_values rebuilds count and region */
            h_st[(size_t)i * sw + 2 * o->cd_agg] = 0;
            h_st[(size_t)i * sw + 2 * o->cd_agg + 1] = 0;
        }
        h_start[i] = cols[o->cfg.n_keys + sw][r];
        h_end[i] = cols[o->cfg.n_keys + sw + 1][r];
    }
    off.push_back(n_rows);
    int64_t n_groups = (int64_t)off.size() - 1;
    int64_t *d_keys, *d_start, *d_end, *d_off;
    uint64_t *d_st;
    SHIP(o, hipMalloc((void **)&d_keys, (size_t)n_rows * 8));
    SHIP(o, hipMalloc((void **)&d_start, (size_t)n_rows * 8));
    SHIP(o, hipMalloc((void **)&d_end, (size_t)n_rows * 8));
    SHIP(o, hipMalloc((void **)&d_st, (size_t)n_rows * sw * 8));
    SHIP(o, hipMalloc((void **)&d_off, (size_t)(n_groups + 1) * 8));
    SHIP(o, hipMemcpy(d_keys, h_keys.data(), (size_t)n_rows * 8,
                      hipMemcpyHostToDevice));
    SHIP(o, hipMemcpy(d_start, h_start.data(), (size_t)n_rows * 8,
                      hipMemcpyHostToDevice));
    SHIP(o, hipMemcpy(d_end, h_end.data(), (size_t)n_rows * 8,
                      hipMemcpyHostToDevice));
    SHIP(o, hipMemcpy(d_st, h_st.data(), (size_t)n_rows * sw * 8,
                      hipMemcpyHostToDevice));
    SHIP(o, hipMemcpy(d_off, off.data(), (size_t)(n_groups + 1) * 8,
                      hipMemcpyHostToDevice));
    RestoreArgs R = {};
    R.keys = d_keys;
    R.st = d_st;
    R.start = d_start;
    R.end = d_end;
    R.group_off = d_off;
    R.n_groups = n_groups;
    R.store = o->store;
    R.gap = o->cfg.gap_nanos;
    R.agg = o->agg;
    R.n_keys = o->cfg.n_keys;
    R.cd_agg = o->cd_agg;
    R.cd = o->cd;
    R.err = o->d_err;
    hipLaunchKernelGGL(k_sess_restore, dim3(grid_for(n_groups)), dim3(256),
                       0, o->stream, R);
    SHIP(o, hipGetLastError());
    SHIP(o, hipStreamSynchronize(o->stream));
    hipFree(d_keys);
    hipFree(d_start);
    hipFree(d_end);
    hipFree(d_st);
    hipFree(d_off);
    return sess_check_err(o);
}

API int arroyo_amd_session_drain_values(void *h, AmdOutBatch *out) {
    GpuSession *o = (GpuSession *)h;
    if (sess_check_err(o)) return 1;
    int ncols = o->cfg.n_keys + 2;
    memset(out, 0, sizeof *out);
    out->n_cols = ncols;
    out->cols = (void **)calloc(ncols, sizeof(void *));
    out->is_f64 = (int32_t *)calloc(ncols, sizeof(int32_t));
    if (o->cd_agg < 0) {
        for (int i = 0; i < ncols; i++) out->cols[i] = malloc(8);
        return 0;
    }
    SHIP(o, hipMemsetAsync(o->d_n_out, 0, 8, o->stream));
    CdDrainArgs D = {};
    D.store = o->store;
    D.agg = o->agg;
    D.cd_agg = o->cd_agg;
    D.n_keys = o->cfg.n_keys;
    D.cd = o->cd;
    for (int i = 0; i < ncols; i++) D.out[i] = o->d_out[i];
    D.n_out = o->d_n_out;
    D.out_cap = o->out_cap;
    D.err = o->d_err;
    hipLaunchKernelGGL(k_sess_drain_values,
                       dim3(grid_for((int64_t)o->store.C + 1)), dim3(256), 0,
                       o->stream, D);
    SHIP(o, hipGetLastError());
    unsigned long long n = 0;
    SHIP(o, hipMemcpyAsync(&n, o->d_n_out, 8, hipMemcpyDeviceToHost,
                           o->stream));
    SHIP(o, hipStreamSynchronize(o->stream));
    if (sess_check_err(o)) return 1;
    out->n_rows = (int64_t)n;
    for (int i = 0; i < ncols; i++) {
        out->cols[i] = malloc((size_t)(n ? n : 1) * 8);
        if (n)
            SHIP(o, hipMemcpyAsync(out->cols[i], o->d_out[i], (size_t)n * 8,
                                   hipMemcpyDeviceToHost, o->stream));
    }
    SHIP(o, hipStreamSynchronize(o->stream));
    return 0;
}

API int arroyo_amd_session_restore_values(void *h,
                                          const int64_t *const *cols,
                                          int32_t n_cols, int64_t n_rows) {
    GpuSession *o = (GpuSession *)h;
    int want = o->cfg.n_keys + 2;
    if (o->cd_agg < 0 || n_cols != want) {
        snprintf(o->err_msg, sizeof o->err_msg,
                 "restore_values expects %d cols and a COUNT DISTINCT "
                 "aggregate", want);
        return o->cd_agg < 0 && n_rows == 0 ? 0 : 1;
    }
    if (n_rows == 0) return 0;
    std::vector<int64_t> order(n_rows);
    for (int64_t i = 0; i < n_rows; i++) order[i] = i;
    const int64_t *keys = o->cfg.n_keys ? cols[0] : nullptr;
    if (keys)
        std::stable_sort(order.begin(), order.end(),
                         [&](int64_t a, int64_t b) {
                             return keys[a] < keys[b];
                         });
    std::vector<int64_t> h_keys(n_rows), h_start(n_rows), h_val(n_rows);
    std::vector<int64_t> off;
    off.push_back(0);
    for (int64_t i = 0; i < n_rows; i++) {
        int64_t r = order[i];
        h_keys[i] = keys ? keys[r] : 0;
        if (i && keys && h_keys[i] != h_keys[i - 1]) off.push_back(i);
        h_start[i] = cols[o->cfg.n_keys][r];
        h_val[i] = cols[o->cfg.n_keys + 1][r];
    }
    off.push_back(n_rows);
    int64_t n_groups = (int64_t)off.size() - 1;
    int64_t *d_keys, *d_start, *d_val, *d_off;
    SHIP(o, hipMalloc((void **)&d_keys, (size_t)n_rows * 8));
    SHIP(o, hipMalloc((void **)&d_start, (size_t)n_rows * 8));
    SHIP(o, hipMalloc((void **)&d_val, (size_t)n_rows * 8));
    SHIP(o, hipMalloc((void **)&d_off, (size_t)(n_groups + 1) * 8));
    SHIP(o, hipMemcpy(d_keys, h_keys.data(), (size_t)n_rows * 8,
                      hipMemcpyHostToDevice));
    SHIP(o, hipMemcpy(d_start, h_start.data(), (size_t)n_rows * 8,
                      hipMemcpyHostToDevice));
    SHIP(o, hipMemcpy(d_val, h_val.data(), (size_t)n_rows * 8,
                      hipMemcpyHostToDevice));
    SHIP(o, hipMemcpy(d_off, off.data(), (size_t)(n_groups + 1) * 8,
                      hipMemcpyHostToDevice));
    CdRestoreArgs R = {};
    R.keys = d_keys;
    R.start = d_start;
    R.value = d_val;
    R.group_off = d_off;
    R.n_groups = n_groups;
    R.store = o->store;
    R.agg = o->agg;
    R.cd_agg = o->cd_agg;
    R.n_keys = o->cfg.n_keys;
    R.cd = o->cd;
    R.err = o->d_err;
    hipLaunchKernelGGL(k_sess_restore_values, dim3(grid_for(n_groups)),
                       dim3(256), 0, o->stream, R);
    SHIP(o, hipGetLastError());
    SHIP(o, hipStreamSynchronize(o->stream));
    hipFree(d_keys);
    hipFree(d_start);
    hipFree(d_val);
    hipFree(d_off);
    return sess_check_err(o);
}

API void arroyo_amd_session_destroy(void *h) {
    GpuSession *o = (GpuSession *)h;
    if (!o) return;
    hipStreamSynchronize(o->stream);
    hipFree(o->store.keys);
    hipFree(o->store.ns);
    hipFree(o->live_buf[0]);
    hipFree(o->live_buf[1]);
    hipFree(o->live_nbuf[0]);
    hipFree(o->live_nbuf[1]);
    hipFree(o->store.recs);
    hipFree(o->bkeys);
    hipFree(o->bst);
    int max_out = o->drain_cols > o->out_cols ? o->drain_cols : o->out_cols;
    for (int i = 0; i < max_out; i++) hipFree(o->d_out[i]);
    hipFree(o->d_n_out);
    hipFree(o->d_err);
    if (o->cd_agg >= 0) {
        hipFree(o->cd.regions);
        hipFree(o->cd.rcur);
        hipFree(o->bv_val);
        hipFree(o->bv_next);
        hipFree(o->bv_cur);
    }
    for (int c = 0; c < o->n_in_cols; c++) {
        hipHostFree(o->stg_h[c]);
        hipFree(o->stg_d[c]);
    }
    hipStreamDestroy(o->stream);
    delete o;
}
