// Sparse-native batched mini-batch-SGD linear solver — CDNA4 (gfx950).
//
// The dense solver (sgd_kernels.hip) is a dense-MFMA design: right for
// tabular widths, impossible for hashed text (a 1M x 2^20 CSR densifies
// to ~4 TB).  This file is the text-scale path (python driver + eager
// torch reference: skdist_amd/models/_sparse_sgd.py; reference workload:
// sk-dist's HashingVectorizer->LR/SVC pipelines, _defaults.py:91-198).
//
// Shapes: n rows, f features (e.g. 2^20), C = ncols models trained at
// once (candidate x fold x class columns, same column semantics as the
// dense solver).  W is a row-major fp32 table [f][CP] (CP = C padded to
// 64) so a 64-lane group's gather of one feature's weights for 64
// consecutive columns is one coalesced 256 B segment — the dominant
// traffic.  Intercept lives in a separate vector Wb[CP]; L2 decay of the
// 2^20-row W would be the only per-step O(f*C) touch, so it is applied
// LAZILY as a per-column scale factor s[CP]:
//
//     W_true[j,c] = s[c] * W[j,c]
//     per step:  s'[c]    = s[c] * (1 - lr_c * lrs * l2_c)
//                W[j,c]  -= lr_c * lrs * g[j,c] * inv_m / s'[c]   (touched j)
//     => W_true' = (1 - lr*l2) W_true - lr * g * inv_m            (exactly
//        the dense k_reduce_update step, momentum 0)
//
// Determinism: no atomics anywhere.  K1 owns (row, column) — the dot
// product walks the row's nnz in CSR order; K3 owns (feature, column) —
// the gradient walks the feature's batch rows in CSC order; K2 reduces
// fixed 1024-row chunks then one sequential pass.  Same guarantee as the
// dense path's split-K slabs.
//
// Layouts:
//   crow   [n+1]  int64   CSR row pointers (seeded row shuffle applied)
//   cidx   [nnz]  int32   feature indices
//   cval   [nnz]  f32     values
//   W      [f][CP]   f32  scaled master weights
//   Wb,s   [CP]      f32  intercept / lazy-L2 scale
//   G      [m][CP]   bf16 per-batch loss gradients (fits L3 at text scale)
//   batch CSC (built once, shared by every epoch):
//     ufeat [uf]  int32   features present in the batch (sorted, unique)
//     cptr  [uf+1] int64  nnz range of each ufeat slot
//     ridx  [nnz_b] int32 row-local index within the batch
//     bval  [nnz_b] f32   values (CSC order)
#include "common.h"

// threads-per-column-tile: a block of 256 lanes covers ROWS_PER_BLOCK
// work items x TC columns; TC adapts to small column counts so a 2-
// candidate search does not idle 3/4 of each block.
static __device__ __forceinline__ int g_tc(int cp) {
    return cp >= 256 ? 256 : cp;  // cp is a multiple of 64
}

// ---------------------------------------------------------------------- //
// K1: CSR forward + fused loss-gradient epilogue (or raw-z emit)
// grid: (ceil(m / rpb), CP / TC), block 256
// one (row, column) pair is owned by exactly one lane.
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_sp_fwd(
    const long long* __restrict__ crow,
    const int* __restrict__ cidx,
    const float* __restrict__ cval,
    const float* __restrict__ W,    // [f][CP]
    const float* __restrict__ Wb,   // [CP]
    const float* __restrict__ s,    // [CP]
    const float* __restrict__ y,    // [n] (full, shuffled order)
    const int* __restrict__ fold,   // [n]
    const float* __restrict__ row_w,   // [n] or null
    const int* __restrict__ col_class,
    const int* __restrict__ col_fold,
    const int* __restrict__ col_class2,
    const long long* __restrict__ rows,  // [m] explicit row ids, or null
    __bf16* __restrict__ G,         // [m][CP]  (loss-grad mode)
    float* __restrict__ Z,          // [m][CP]  (emit-z mode; G null)
    long long start, long long m, int cp, int loss_id)
{
    const int tc = g_tc(cp);
    const int rpb = 256 / tc;
    const int c = blockIdx.y * tc + (threadIdx.x % tc);
    if (c >= cp) return;
    // grid-stride over rows: short hashed-text rows make one-row-per-
    // block launch/latency-bound (guide Guideline 11)
    const long long stride = (long long)gridDim.x * rpb;
    for (long long ri = (long long)blockIdx.x * rpb + threadIdx.x / tc;
         ri < m; ri += stride) {
    const long long r = (rows != nullptr) ? rows[ri] : start + ri;

    const long long k0 = crow[r], k1 = crow[r + 1];
    // 4 partial accumulators hide the gather latency chain
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    long long k = k0;
    for (; k + 4 <= k1; k += 4) {
        const int j0 = cidx[k], j1 = cidx[k + 1];
        const int j2 = cidx[k + 2], j3 = cidx[k + 3];
        a0 += cval[k]     * W[(long long)j0 * cp + c];
        a1 += cval[k + 1] * W[(long long)j1 * cp + c];
        a2 += cval[k + 2] * W[(long long)j2 * cp + c];
        a3 += cval[k + 3] * W[(long long)j3 * cp + c];
    }
    for (; k < k1; ++k)
        a0 += cval[k] * W[(long long)cidx[k] * cp + c];
    const float z = s[c] * ((a0 + a1) + (a2 + a3)) + Wb[c];

    if (G == nullptr) {            // scoring path: raw decision values
        Z[ri * cp + c] = z;
        continue;
    }
    const float yv = y[r];
    const int cls = col_class[c];
    const float t = (cls < 0) ? yv : (yv == (float)cls ? 1.f : 0.f);
    float g = dloss(loss_id, z, t);
    if (row_w != nullptr) g *= row_w[r];
    bool train = (fold[r] != col_fold[c]);
    const int c2 = col_class2[c];
    if (c2 >= 0)
        train = train && (yv == (float)cls || yv == (float)c2);
    G[ri * cp + c] = f32_to_bf16(train ? g : 0.f);
    }
}

// ---------------------------------------------------------------------- //
// K2a: per-chunk column sums of G, deterministic.  chunk is chosen by
// the host (ceil(m/256) rounded to 32, >= 32) so the grid fills the
// chip even at small column counts; 4 independent accumulators give
// the loads ILP (one-accumulator version measured 231 us: one HBM
// latency per row).
// grid: (ceil(m/chunk), ceil(CP/256)), block 256
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_sp_colsum(
    const __bf16* __restrict__ G, float* __restrict__ part,  // [nch][CP]
    long long m, int cp, int chunk)
{
    const int c = blockIdx.y * 256 + threadIdx.x;
    if (c >= cp) return;
    const long long r0 = (long long)blockIdx.x * chunk;
    const long long r1 = (r0 + chunk < m) ? r0 + chunk : m;
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    long long r = r0;
    for (; r + 4 <= r1; r += 4) {
        a0 += bf16_to_f32(G[r * cp + c]);
        a1 += bf16_to_f32(G[(r + 1) * cp + c]);
        a2 += bf16_to_f32(G[(r + 2) * cp + c]);
        a3 += bf16_to_f32(G[(r + 3) * cp + c]);
    }
    for (; r < r1; ++r)
        a0 += bf16_to_f32(G[r * cp + c]);
    part[(long long)blockIdx.x * cp + c] = ((a0 + a1) + (a2 + a3));
}

// K2b: reduce chunks -> intercept update + lazy-scale advance
// grid: ceil(CP/256), block 256.  With `hb` non-null the intercept step
// is Adagrad-normalized (hb accumulates the squared mean-gradient).
extern "C" __global__ __launch_bounds__(256) void k_sp_bias_scale(
    const float* __restrict__ part, float* __restrict__ Wb,
    float* __restrict__ s, float* __restrict__ hb,
    const float* __restrict__ col_lr, const float* __restrict__ col_l2,
    int n_chunks, int cp, float inv_m, float lr_scale)
{
    const int c = blockIdx.x * 256 + threadIdx.x;
    if (c >= cp) return;
    float g0 = 0.f, g1 = 0.f, g2 = 0.f, g3 = 0.f;
    int z = 0;
    for (; z + 4 <= n_chunks; z += 4) {
        g0 += part[(long long)z * cp + c];
        g1 += part[(long long)(z + 1) * cp + c];
        g2 += part[(long long)(z + 2) * cp + c];
        g3 += part[(long long)(z + 3) * cp + c];
    }
    for (; z < n_chunks; ++z)
        g0 += part[(long long)z * cp + c];
    float g = ((g0 + g1) + (g2 + g3)) * inv_m;
    const float lr = col_lr[c] * lr_scale;
    float step = lr * g;
    if (hb != nullptr) {
        const float h = hb[c] + g * g;
        hb[c] = h;
        step = lr * g * __frsqrt_rn(h + 1e-12f);
    }
    Wb[c] -= step;                            // no L2 on the intercept
    s[c] *= (1.f - lr * col_l2[c]);          // lazy decay of all of W
}

// ---------------------------------------------------------------------- //
// K3: batch-CSC weight update — one (feature, column) per lane
// grid: (ceil(uf / fpb), CP / TC), block 256
// With `h` non-null the data-gradient step is Adagrad-normalized per
// (feature, column): rare text features take full-size first steps
// instead of 1/batch-size ones (the standard sparse-text optimizer);
// the lazy L2 scale path is untouched, so regularization stays exactly
// the dense solver's per-step decay.
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_sp_update(
    const int* __restrict__ ufeat,   // [uf]
    const long long* __restrict__ cptr,  // [uf+1]
    const int* __restrict__ ridx,    // row-local
    const float* __restrict__ bval,
    const __bf16* __restrict__ G,    // [m][CP]
    float* __restrict__ W,           // [f][CP]
    float* __restrict__ h,           // [f][CP] Adagrad accum, or null
    const float* __restrict__ s,     // [CP] (post-K2b value)
    const float* __restrict__ col_lr,
    long long uf, int cp, float inv_m, float lr_scale)
{
    const int tc = g_tc(cp);
    const int fpb = 256 / tc;
    const int c = blockIdx.y * tc + (threadIdx.x % tc);
    if (c >= cp) return;
    // grid-stride over feature slots: most hashed-text features have
    // 1-4 batch rows, so one-slot-per-block is launch/latency-bound
    const long long stride = (long long)gridDim.x * fpb;
    for (long long si = (long long)blockIdx.x * fpb + threadIdx.x / tc;
         si < uf; si += stride) {
    const int j = ufeat[si];
    const long long k0 = cptr[si], k1 = cptr[si + 1];
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    long long k = k0;
    for (; k + 4 <= k1; k += 4) {
        a0 += bval[k]     * bf16_to_f32(G[(long long)ridx[k] * cp + c]);
        a1 += bval[k + 1] * bf16_to_f32(G[(long long)ridx[k + 1] * cp + c]);
        a2 += bval[k + 2] * bf16_to_f32(G[(long long)ridx[k + 2] * cp + c]);
        a3 += bval[k + 3] * bf16_to_f32(G[(long long)ridx[k + 3] * cp + c]);
    }
    for (; k < k1; ++k)
        a0 += bval[k] * bf16_to_f32(G[(long long)ridx[k] * cp + c]);
    const float g = ((a0 + a1) + (a2 + a3)) * inv_m;
    const long long e = (long long)j * cp + c;
    float step = col_lr[c] * lr_scale * g;
    if (h != nullptr) {
        const float hv = h[e] + g * g;
        h[e] = hv;
        step *= __frsqrt_rn(hv + 1e-12f);
    }
    W[e] -= step / s[c];
    }
}

// fold the scale into W when it drifts far from 1 (rare; keeps /s[c]
// well-conditioned). grid: (f, CP/256), block 256
extern "C" __global__ __launch_bounds__(256) void k_sp_renorm(
    float* __restrict__ W, const float* __restrict__ s, long long f,
    int cp)
{
    const int c = blockIdx.y * 256 + threadIdx.x;
    if (c >= cp) return;
    W[(long long)blockIdx.x * cp + c] *= s[c];
}

// ---------------------------------------------------------------------- //
// host launchers
// ---------------------------------------------------------------------- //
static inline void sp_fwd_launch(
    const void* crow, const void* cidx, const void* cval, const void* W,
    const void* Wb, const void* s, const void* y, const void* fold,
    const void* row_w, const void* col_class, const void* col_fold,
    const void* col_class2, const void* rows, void* G, void* Z,
    long long start, long long m, int cp, int loss_id, hipStream_t st)
{
    const int tc = cp >= 256 ? 256 : cp;
    const int rpb = 256 / tc;
    long long gx = (m + rpb - 1) / rpb;
    if (gx > 2048) gx = 2048;
    dim3 grid((unsigned)gx, (unsigned)((cp + tc - 1) / tc));
    hipLaunchKernelGGL(k_sp_fwd, grid, dim3(256), 0, st,
                       (const long long*)crow, (const int*)cidx,
                       (const float*)cval, (const float*)W,
                       (const float*)Wb, (const float*)s, (const float*)y,
                       (const int*)fold, (const float*)row_w,
                       (const int*)col_class, (const int*)col_fold,
                       (const int*)col_class2, (const long long*)rows,
                       (__bf16*)G, (float*)Z, start, m, cp, loss_id);
}

extern "C" hipError_t skdist_sp_sgd_step(
    const void* crow, const void* cidx, const void* cval,
    void* W, void* Wb, void* s, void* h, void* hb, void* G, void* part,
    const void* y, const void* fold, const void* row_w,
    const void* col_class, const void* col_fold, const void* col_class2,
    const void* col_lr, const void* col_l2,
    const void* ufeat, const void* cptr, const void* ridx,
    const void* bval, long long uf,
    long long start, long long m, int cp, int loss_id,
    float inv_m, float lr_scale, hipStream_t stream)
{
    sp_fwd_launch(crow, cidx, cval, W, Wb, s, y, fold, row_w, col_class,
                  col_fold, col_class2, nullptr, G, nullptr, start, m, cp,
                  loss_id, stream);
    HIP_CHECK(hipGetLastError());
    int chunk = (int)((m + 255) / 256);
    chunk = (chunk + 31) / 32 * 32;
    if (chunk < 32) chunk = 32;
    const int nch = (int)((m + chunk - 1) / chunk);
    {
        dim3 grid((unsigned)nch, (unsigned)((cp + 255) / 256));
        hipLaunchKernelGGL(k_sp_colsum, grid, dim3(256), 0, stream,
                           (const __bf16*)G, (float*)part, m, cp, chunk);
        HIP_CHECK(hipGetLastError());
    }
    {
        dim3 grid((unsigned)((cp + 255) / 256));
        hipLaunchKernelGGL(k_sp_bias_scale, grid, dim3(256), 0, stream,
                           (const float*)part, (float*)Wb, (float*)s,
                           (float*)hb,
                           (const float*)col_lr, (const float*)col_l2,
                           nch, cp, inv_m, lr_scale);
        HIP_CHECK(hipGetLastError());
    }
    if (uf > 0) {
        const int tc = cp >= 256 ? 256 : cp;
        const int fpb = 256 / tc;
        long long gx = (uf + fpb - 1) / fpb;
        if (gx > 2048) gx = 2048;
        dim3 grid((unsigned)gx, (unsigned)((cp + tc - 1) / tc));
        hipLaunchKernelGGL(k_sp_update, grid, dim3(256), 0, stream,
                           (const int*)ufeat, (const long long*)cptr,
                           (const int*)ridx, (const float*)bval,
                           (const __bf16*)G, (float*)W, (float*)h,
                           (const float*)s,
                           (const float*)col_lr, uf, cp, inv_m, lr_scale);
        HIP_CHECK(hipGetLastError());
    }
    return hipSuccess;
}

extern "C" hipError_t skdist_sp_forward(
    const void* crow, const void* cidx, const void* cval,
    const void* W, const void* Wb, const void* s, const void* rows,
    void* Z, long long m, int cp, hipStream_t stream)
{
    if (m == 0) return hipSuccess;
    sp_fwd_launch(crow, cidx, cval, W, Wb, s, nullptr, nullptr, nullptr,
                  nullptr, nullptr, nullptr, rows, nullptr, Z, 0, m, cp,
                  0, stream);
    return hipGetLastError();
}

extern "C" hipError_t skdist_sp_renorm(
    void* W, const void* s, long long f, int cp, hipStream_t stream)
{
    dim3 grid((unsigned)f, (unsigned)((cp + 255) / 256));
    hipLaunchKernelGGL(k_sp_renorm, grid, dim3(256), 0, stream,
                       (float*)W, (const float*)s, f, cp);
    return hipGetLastError();
}
