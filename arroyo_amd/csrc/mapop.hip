/* arroyo-amd stateless map/filter/projection: MI355X-native (gfx950)
 * equivalent of ValueExecutionOperator / ProjectionOperator /
 * KeyExecutionOperator (crates/arroyo-worker/src/arrow/mod.rs:48-243)
 * behind the arroyo_amd_map_* C ABI (include/arroyo_amd.h).
 *
 * The reference evaluates a serialized DataFusion expression plan per batch
 * (StatelessPhysicalExecutor :245-290).  Here the plan's information
 * content is a register program (AmdMapConfig) interpreted by one fused
 * kernel: registers live in VGPRs, the instruction stream is wave-uniform
 * (no divergence beyond the filter), and the whole op is a bandwidth-bound
 * streaming map.  Filtered output preserves row order exactly like the
 * reference's filter kernels: keep flags -> hipCUB exclusive scan ->
 * ordered scatter, three launches per batch on one stream.
 *
 * Parity is pinned against oracle/arroyo_oracle.c by tests/test_mapop.py
 * (bit-exact, including the f64 bit patterns).
 */
#include <hip/hip_runtime.h>
#include <hipcub/hipcub.hpp>

#include <cstdio>
#include <cstdlib>
#include <cstring>

#include "../../include/arroyo_amd_types.h"

#define API extern "C" __attribute__((visibility("default")))

namespace mapop {

#define MERR_DIV0 1

struct MapArgs {
    const int64_t *cols[AMD_MAP_MAX_REGS];
    int32_t n_in;
    int64_t n_rows;
    AmdMapConfig cfg;          /* program in kernel args (constant memory) */
    int64_t *keep;             /* [n] 0/1 (filter) */
    int64_t *vals[AMD_MAP_MAX_OUT]; /* unfiltered per-row results */
    int *err;
};

__device__ inline void map_eval(const AmdMapConfig &c, int64_t *regs,
                                int *err) {
    for (int i = 0; i < c.n_prog; i++) {
        const AmdMapInstr &in = c.prog[i];
        int64_t a = regs[in.a], b = regs[in.b], v = 0;
        double fa = __longlong_as_double(a), fb = __longlong_as_double(b);
        switch (in.op) {
        case AMD_MOP_CONST: v = in.imm; break;
        case AMD_MOP_ADD: v = a + b; break;
        case AMD_MOP_SUB: v = a - b; break;
        case AMD_MOP_MUL: v = a * b; break;
        case AMD_MOP_DIV:
            if (b == 0) { *err = MERR_DIV0; return; }
            v = a / b;
            break;
        case AMD_MOP_MOD:
            if (b == 0) { *err = MERR_DIV0; return; }
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
        case AMD_MOP_I2F: v = __double_as_longlong((double)a); break;
        case AMD_MOP_F2I: v = (int64_t)fa; break;
        case AMD_MOP_FADD: v = __double_as_longlong(fa + fb); break;
        case AMD_MOP_FSUB: v = __double_as_longlong(fa - fb); break;
        case AMD_MOP_FMUL: v = __double_as_longlong(fa * fb); break;
        case AMD_MOP_FDIV: v = __double_as_longlong(fa / fb); break;
        }
        regs[in.dst] = v;
    }
}

__global__ void __launch_bounds__(256)
k_map_eval(MapArgs A) {
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         r < A.n_rows; r += stride) {
        int64_t regs[AMD_MAP_MAX_REGS];
        for (int i = 0; i < A.n_in; i++) regs[i] = A.cols[i][r];
        map_eval(A.cfg, regs, A.err);
        for (int i = 0; i < A.cfg.n_out; i++)
            A.vals[i][r] = regs[A.cfg.out_reg[i]];
        if (A.keep)
            A.keep[r] = regs[A.cfg.filter_reg] != 0;
    }
}

/* ordered scatter of kept rows: pos = exclusive scan of keep flags */
__global__ void __launch_bounds__(256)
k_map_scatter(const int64_t *keep, const int64_t *pos,
              const int64_t *const vals0, const int64_t *const vals1,
              const int64_t *const vals2, const int64_t *const vals3,
              const int64_t *const vals4, const int64_t *const vals5,
              int64_t *out0, int64_t *out1, int64_t *out2, int64_t *out3,
              int64_t *out4, int64_t *out5, int n_out, int64_t n) {
    const int64_t *vals[6] = {vals0, vals1, vals2, vals3, vals4, vals5};
    int64_t *outs[6] = {out0, out1, out2, out3, out4, out5};
    int64_t stride = (int64_t)gridDim.x * blockDim.x;
    for (int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         r < n; r += stride) {
        if (!keep[r]) continue;
        int64_t p = pos[r];
        for (int i = 0; i < n_out; i++) outs[i][p] = vals[i][r];
    }
}

}  // namespace mapop

using namespace mapop;

static char g_map_err[256];

struct GpuMap {
    AmdMapConfig cfg;
    int64_t cap;               /* per-batch row capacity */
    int64_t *d_vals[AMD_MAP_MAX_OUT];
    int64_t *d_out[AMD_MAP_MAX_OUT];
    int64_t *d_keep, *d_pos;
    void *cub_tmp;
    size_t cub_bytes;
    int *d_err;
    int64_t *stg_h[AMD_MAP_MAX_REGS], *stg_d[AMD_MAP_MAX_REGS];
    hipStream_t stream;
    char err_msg[512];
};

#define MHIP(o, call)                                                         \
    do {                                                                      \
        hipError_t _e = (call);                                               \
        if (_e != hipSuccess) {                                               \
            snprintf((o)->err_msg, sizeof (o)->err_msg, "%s:%d hip: %s",      \
                     __FILE__, __LINE__, hipGetErrorString(_e));              \
            return 1;                                                         \
        }                                                                     \
    } while (0)

API void *arroyo_amd_map_create(const AmdMapConfig *cfg) {
    if (!cfg || cfg->n_in_cols < 1 || cfg->n_in_cols > AMD_MAP_MAX_REGS ||
        cfg->n_prog < 0 || cfg->n_prog > AMD_MAP_MAX_PROG ||
        cfg->n_out < 1 || cfg->n_out > 6 ||
        cfg->filter_reg >= AMD_MAP_MAX_REGS) {
        snprintf(g_map_err, sizeof g_map_err,
                 "invalid map config (n_out <= 6 on the GPU path)");
        return nullptr;
    }
    for (int i = 0; i < cfg->n_prog; i++)
        if (cfg->prog[i].dst < 0 || cfg->prog[i].dst >= AMD_MAP_MAX_REGS ||
            cfg->prog[i].a < 0 || cfg->prog[i].a >= AMD_MAP_MAX_REGS ||
            cfg->prog[i].b < 0 || cfg->prog[i].b >= AMD_MAP_MAX_REGS) {
            snprintf(g_map_err, sizeof g_map_err, "invalid map program");
            return nullptr;
        }
    GpuMap *o = new GpuMap();
    o->cfg = *cfg;
    o->cap = 1 << 20;
    if (hipSetDevice(cfg->device) != hipSuccess) {
        snprintf(g_map_err, sizeof g_map_err,
                 "hipSetDevice(%d) failed: no HIP device (no CPU fallback)",
                 cfg->device);
        delete o;
        return nullptr;
    }
    hipError_t e = hipSuccess;
    for (int i = 0; i < cfg->n_out && e == hipSuccess; i++) {
        e = hipMalloc((void **)&o->d_vals[i], (size_t)o->cap * 8);
        if (e == hipSuccess)
            e = hipMalloc((void **)&o->d_out[i], (size_t)o->cap * 8);
    }
    if (e == hipSuccess) e = hipMalloc((void **)&o->d_keep, (size_t)o->cap * 8);
    if (e == hipSuccess) e = hipMalloc((void **)&o->d_pos, (size_t)o->cap * 8);
    if (e == hipSuccess) e = hipMalloc((void **)&o->d_err, 4);
    for (int c = 0; c < cfg->n_in_cols && e == hipSuccess; c++) {
        if (hipHostMalloc((void **)&o->stg_h[c], (size_t)o->cap * 8) !=
                hipSuccess ||
            hipMalloc((void **)&o->stg_d[c], (size_t)o->cap * 8) !=
                hipSuccess)
            e = hipErrorOutOfMemory;
    }
    if (e != hipSuccess) {
        snprintf(g_map_err, sizeof g_map_err, "map alloc: %s",
                 hipGetErrorString(e));
        delete o;
        return nullptr;
    }
    o->cub_bytes = 0;
    hipcub::DeviceScan::ExclusiveSum(nullptr, o->cub_bytes, o->d_keep,
                                     o->d_pos, (int)o->cap);
    if (hipMalloc(&o->cub_tmp, o->cub_bytes ? o->cub_bytes : 1) !=
        hipSuccess) {
        snprintf(g_map_err, sizeof g_map_err, "map cub alloc failed");
        delete o;
        return nullptr;
    }
    hipMemset(o->d_err, 0, 4);
    hipStreamCreate(&o->stream);
    return o;
}

API const char *arroyo_amd_map_last_error(void *h) {
    return h ? ((GpuMap *)h)->err_msg : g_map_err;
}

static int map_grid(int64_t want) {
    int64_t w = (want + 255) / 256;
    return (int)(w > 2048 ? 2048 : (w < 1 ? 1 : w));
}

static int map_run(GpuMap *o, const int64_t *const *cols, int32_t n_cols,
                   int64_t n_rows, AmdOutBatch *out) {
    const AmdMapConfig &c = o->cfg;
    if (n_cols != c.n_in_cols) {
        snprintf(o->err_msg, sizeof o->err_msg, "expected %d cols, got %d",
                 c.n_in_cols, n_cols);
        return 1;
    }
    memset(out, 0, sizeof *out);
    out->n_cols = c.n_out;
    out->cols = (void **)calloc(c.n_out, sizeof(void *));
    out->is_f64 = (int32_t *)calloc(c.n_out, sizeof(int32_t));
    for (int i = 0; i < c.n_out; i++) out->is_f64[i] = c.out_is_f64[i];
    int64_t emitted = 0;
    int64_t done = 0;
    while (done < n_rows) {
        int64_t take = n_rows - done;
        if (take > o->cap) take = o->cap;
        MapArgs A = {};
        for (int cc = 0; cc < n_cols; cc++) {
            memcpy(o->stg_h[cc], cols[cc] + done, (size_t)take * 8);
            MHIP(o, hipMemcpyAsync(o->stg_d[cc], o->stg_h[cc],
                                   (size_t)take * 8, hipMemcpyHostToDevice,
                                   o->stream));
            A.cols[cc] = o->stg_d[cc];
        }
        A.n_in = n_cols;
        A.n_rows = take;
        A.cfg = c;
        A.keep = c.filter_reg >= 0 ? o->d_keep : nullptr;
        for (int i = 0; i < c.n_out; i++) A.vals[i] = o->d_vals[i];
        A.err = o->d_err;
        hipLaunchKernelGGL(k_map_eval, dim3(map_grid(take)), dim3(256), 0,
                           o->stream, A);
        MHIP(o, hipGetLastError());
        int64_t n_keep = take;
        const int64_t *src[AMD_MAP_MAX_OUT];
        for (int i = 0; i < c.n_out; i++) src[i] = o->d_vals[i];
        if (c.filter_reg >= 0) {
            size_t tmp = o->cub_bytes;
            hipcub::DeviceScan::ExclusiveSum(o->cub_tmp, tmp, o->d_keep,
                                             o->d_pos, (int)take, o->stream);
            hipLaunchKernelGGL(
                k_map_scatter, dim3(map_grid(take)), dim3(256), 0, o->stream,
                o->d_keep, o->d_pos, o->d_vals[0],
                c.n_out > 1 ? o->d_vals[1] : nullptr,
                c.n_out > 2 ? o->d_vals[2] : nullptr,
                c.n_out > 3 ? o->d_vals[3] : nullptr,
                c.n_out > 4 ? o->d_vals[4] : nullptr,
                c.n_out > 5 ? o->d_vals[5] : nullptr, o->d_out[0],
                c.n_out > 1 ? o->d_out[1] : nullptr,
                c.n_out > 2 ? o->d_out[2] : nullptr,
                c.n_out > 3 ? o->d_out[3] : nullptr,
                c.n_out > 4 ? o->d_out[4] : nullptr,
                c.n_out > 5 ? o->d_out[5] : nullptr, c.n_out, take);
            MHIP(o, hipGetLastError());
            /* n_keep = pos[last] + keep[last] */
            int64_t tail[2] = {0, 0};
            MHIP(o, hipMemcpyAsync(&tail[0], o->d_pos + (take - 1), 8,
                                   hipMemcpyDeviceToHost, o->stream));
            MHIP(o, hipMemcpyAsync(&tail[1], o->d_keep + (take - 1), 8,
                                   hipMemcpyDeviceToHost, o->stream));
            MHIP(o, hipStreamSynchronize(o->stream));
            n_keep = tail[0] + tail[1];
            for (int i = 0; i < c.n_out; i++) src[i] = o->d_out[i];
        }
        int derr = 0;
        MHIP(o, hipMemcpyAsync(&derr, o->d_err, 4, hipMemcpyDeviceToHost,
                               o->stream));
        MHIP(o, hipStreamSynchronize(o->stream));
        if (derr) {
            snprintf(o->err_msg, sizeof o->err_msg, "division by zero");
            return 1;
        }
        if (n_keep) {
            for (int i = 0; i < c.n_out; i++) {
                out->cols[i] = realloc(out->cols[i],
                                       (size_t)(emitted + n_keep) * 8);
                MHIP(o, hipMemcpyAsync((int64_t *)out->cols[i] + emitted,
                                       src[i], (size_t)n_keep * 8,
                                       hipMemcpyDeviceToHost, o->stream));
            }
            MHIP(o, hipStreamSynchronize(o->stream));
            emitted += n_keep;
        }
        done += take;
    }
    out->n_rows = emitted;
    for (int i = 0; i < c.n_out; i++)
        if (!out->cols[i]) out->cols[i] = malloc(8);
    return 0;
}

/* device-resident map/filter: input columns already in HBM; output stays
 * in HBM.  Fills d_out_cols[i] with device addresses owned by the
 * operator (valid until the next process_batch* call) and *n_out_rows
 * with the surviving row count.  One launch, so n_rows must fit the
 * operator's chunk capacity (the host path chunks instead). */
API int arroyo_amd_map_process_batch_device(void *h,
                                            const int64_t *const *dcols,
                                            int32_t n_cols, int64_t n_rows,
                                            const int64_t **d_out_cols,
                                            int64_t *n_out_rows) {
    GpuMap *o = (GpuMap *)h;
    const AmdMapConfig &c = o->cfg;
    if (n_cols != c.n_in_cols) {
        snprintf(o->err_msg, sizeof o->err_msg, "expected %d cols, got %d",
                 c.n_in_cols, n_cols);
        return 1;
    }
    if (n_rows > o->cap) {
        snprintf(o->err_msg, sizeof o->err_msg,
                 "device batch of %lld rows exceeds capacity %lld",
                 (long long)n_rows, (long long)o->cap);
        return 1;
    }
    MapArgs A = {};
    for (int cc = 0; cc < n_cols; cc++) A.cols[cc] = dcols[cc];
    A.n_in = n_cols;
    A.n_rows = n_rows;
    A.cfg = c;
    A.keep = c.filter_reg >= 0 ? o->d_keep : nullptr;
    for (int i = 0; i < c.n_out; i++) A.vals[i] = o->d_vals[i];
    A.err = o->d_err;
    hipLaunchKernelGGL(k_map_eval, dim3(map_grid(n_rows)), dim3(256), 0,
                       o->stream, A);
    MHIP(o, hipGetLastError());
    int64_t n_keep = n_rows;
    if (c.filter_reg >= 0 && n_rows > 0) {
        size_t tmp = o->cub_bytes;
        hipcub::DeviceScan::ExclusiveSum(o->cub_tmp, tmp, o->d_keep,
                                         o->d_pos, (int)n_rows, o->stream);
        hipLaunchKernelGGL(
            k_map_scatter, dim3(map_grid(n_rows)), dim3(256), 0, o->stream,
            o->d_keep, o->d_pos, o->d_vals[0],
            c.n_out > 1 ? o->d_vals[1] : nullptr,
            c.n_out > 2 ? o->d_vals[2] : nullptr,
            c.n_out > 3 ? o->d_vals[3] : nullptr,
            c.n_out > 4 ? o->d_vals[4] : nullptr,
            c.n_out > 5 ? o->d_vals[5] : nullptr, o->d_out[0],
            c.n_out > 1 ? o->d_out[1] : nullptr,
            c.n_out > 2 ? o->d_out[2] : nullptr,
            c.n_out > 3 ? o->d_out[3] : nullptr,
            c.n_out > 4 ? o->d_out[4] : nullptr,
            c.n_out > 5 ? o->d_out[5] : nullptr, c.n_out, n_rows);
        MHIP(o, hipGetLastError());
        int64_t tail[2] = {0, 0};
        MHIP(o, hipMemcpyAsync(&tail[0], o->d_pos + (n_rows - 1), 8,
                               hipMemcpyDeviceToHost, o->stream));
        MHIP(o, hipMemcpyAsync(&tail[1], o->d_keep + (n_rows - 1), 8,
                               hipMemcpyDeviceToHost, o->stream));
        MHIP(o, hipStreamSynchronize(o->stream));
        n_keep = tail[0] + tail[1];
    }
    int derr = 0;
    MHIP(o, hipMemcpyAsync(&derr, o->d_err, 4, hipMemcpyDeviceToHost,
                           o->stream));
    MHIP(o, hipStreamSynchronize(o->stream));
    if (derr) {
        snprintf(o->err_msg, sizeof o->err_msg, "division by zero");
        return 1;
    }
    for (int i = 0; i < c.n_out; i++)
        d_out_cols[i] = c.filter_reg >= 0 ? o->d_out[i] : o->d_vals[i];
    *n_out_rows = n_keep;
    return 0;
}

API int arroyo_amd_map_process_batch(void *h, const int64_t *const *cols,
                                     int32_t n_cols, int64_t n_rows,
                                     AmdOutBatch *out) {
    GpuMap *o = (GpuMap *)h;
    memset(out, 0, sizeof *out); /* so early-validation errors leave a
                                  * well-defined (empty) out */
    int rc = map_run(o, cols, n_cols, n_rows, out);
    if (rc && out->cols) {
        /* error exits (device error, HIP failure) must not leak the
         * partially filled output the caller will never free */
        for (int i = 0; i < out->n_cols; i++) free(out->cols[i]);
        free(out->cols);
        free(out->is_f64);
        memset(out, 0, sizeof *out);
    }
    return rc;
}

API void arroyo_amd_map_destroy(void *h) {
    GpuMap *o = (GpuMap *)h;
    if (!o) return;
    hipStreamSynchronize(o->stream);
    for (int i = 0; i < o->cfg.n_out; i++) {
        hipFree(o->d_vals[i]);
        hipFree(o->d_out[i]);
    }
    hipFree(o->d_keep);
    hipFree(o->d_pos);
    hipFree(o->d_err);
    hipFree(o->cub_tmp);
    for (int c = 0; c < o->cfg.n_in_cols; c++) {
        hipHostFree(o->stg_h[c]);
        hipFree(o->stg_d[c]);
    }
    hipStreamDestroy(o->stream);
    delete o;
}
