// Batched mini-batch-SGD linear solver — CDNA4 (gfx950) kernels.
//
// One SGD step over a minibatch of m rows trains ncols independent linear
// models (columns) at once (math + torch reference: skdist_amd/models/_sgd.py;
// reference inventory: SURVEY.md §2.4 row 1).
//
//   K1 k_fwd_gt:        GT[c][i] = mask*dloss( (Xs·W)[i][c], target )
//   K2 k_grad_partial:  partial[z] = Xᵀ·G  (split-K, deterministic slabs)
//   K3 k_reduce_update: W -= lr * (Σz partial / m + λ∘W); refresh WbfT
//
// GEMM structure (K1/K2): 128×128 tile, 4 waves (2×2 of 64×64), MFMA bf16
// 16x16x32, BK=32, double-buffered LDS staged by global_load_lds_dwordx4
// (wave-uniform LDS chunk base + lane×16B, per-lane swizzled SOURCE
// address).  LDS image is lane-linear [128][32] bf16; the 16B slot of each
// row is XOR-swizzled by ((row>>2)&3) so ds_read_b128 fragment loads are
// bank-conflict-free (guide §5.4 rule 21 / T2).
//
// Layouts (row-major, K-contiguous):
//   Xs    [n_pad][fa]        bf16  epoch-shuffled, rows n..n_pad zero
//   XsT   [fa_store][n_pad]  bf16  transpose (fa_store = ceil(fa/128)*128)
//   WbfT  [ncols_pad][fa]    bf16
//   W, V  [fa][ncols_pad]    f32
//   GT    [ncols_pad][gts]   bf16
//   partial [SPLITK][fa][ncols_pad] f32
#include "common.h"

#define BM 128
#define BN 128
#define BK 32
#define LDC 132  // LDS row stride for the epilogue transpose tile
// double-buffered LDS tiles carved from the extern-shared base `sm`
// (one __shared__ object only — guide §5 'Three .s-level traps' (a))

// stage one 128x32 bf16 tile (8 KiB) into linear LDS via 8 glds chunks;
// each of the 4 waves issues 2. src_row(row) must return the global
// pointer to the row's k-offset base (element units).
#define STAGE_TILE(ldsbase, SRC_EXPR)                                       \
    do {                                                                    \
        _Pragma("unroll") for (int i = 0; i < 2; ++i) {                     \
            const int c = w + 4 * i;                                        \
            const int row = 16 * c + (lane >> 2);                           \
            const int slot = (lane & 3) ^ ((row >> 2) & 3);                 \
            const __bf16* _src = (SRC_EXPR) + slot * 8;                     \
            __builtin_amdgcn_global_load_lds(                               \
                (const unsigned int*)_src,                                  \
                (unsigned int*)((ldsbase) + c * 512), 16, 0, 0);            \
        }                                                                   \
    } while (0)

// swizzled ds_read_b128 of one 8-elem fragment at (row, k-slot fk/8)
static __device__ __forceinline__ bf16x8
frag_load(const __bf16* lds, int row, int fk) {
    const int phys = (fk >> 3) ^ ((row >> 2) & 3);
    return *(const bf16x8*)(lds + row * BK + phys * 8);
}

// ---------------------------------------------------------------------- //
// K1: fused forward GEMM + loss gradient + transposed store
// grid: (m_pad/BM, ncols_pad/BN), block 256
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_fwd_gt(
    const __bf16* __restrict__ Xs,    // [n_pad][fa]
    const __bf16* __restrict__ WbfT,  // [ncols_pad][fa]
    __bf16* __restrict__ GT,          // [ncols_pad][gt_stride]
    const float* __restrict__ y,      // [n]
    const int* __restrict__ fold,     // [n]
    const int* __restrict__ col_class,
    const int* __restrict__ col_fold,
    const int* __restrict__ col_class2,  // >=0: one-vs-one partner class
    const float* __restrict__ row_w,     // per-row sample weights or null
    int start, int m, long long n, int fa, int ncols_pad, int gt_stride,
    int loss_id)
{
    extern __shared__ __attribute__((aligned(16))) char sm[];
    #define bufA(b) ((__bf16*)(sm + (b) * 16384))
    #define bufB(b) ((__bf16*)(sm + 8192 + (b) * 16384))
    __bf16* ldsC = (__bf16*)sm;             // [BN][LDC], aliases the bufs
    char* meta = sm + BN * LDC * 2;
    float* y_s = (float*)meta;              // [BM]
    int* fold_s = (int*)(meta + 512);       // [BM]
    int* cls_s = (int*)(meta + 1024);       // [BN]
    int* cfold_s = (int*)(meta + 1536);     // [BN]
    int* cls2_s = (int*)(meta + 2048);      // [BN]
    float* rw_s = (float*)(meta + 2560);    // [BM]

    const int tid = threadIdx.x;
    const int bm = blockIdx.x * BM;
    const int bn = blockIdx.y * BN;
    const int lane = tid & 63;
    const int w = tid >> 6;

    if (tid < BM) {
        const long long r = (long long)start + bm + tid;
        const bool ok = r < n;
        y_s[tid] = ok ? y[r] : 0.f;
        fold_s[tid] = ok ? fold[r] : -9;
        rw_s[tid] = (row_w != nullptr && ok) ? row_w[r] : 1.f;
    } else {
        const int c = tid - BM;
        cls_s[c] = col_class[bn + c];
        cfold_s[c] = col_fold[bn + c];
        cls2_s[c] = col_class2[bn + c];
    }

    const __bf16* Abase = Xs + (long long)(start + bm) * fa;
    const __bf16* Bbase = WbfT + (long long)bn * fa;

    const int wr = ((w >> 1) & 1) * 64;
    const int wc = (w & 1) * 64;
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;

    f32x4 acc[4][4] = {};
    const int nk = fa / BK;
    int cur = 0;
    STAGE_TILE(bufA(0), Abase + (long long)row * fa);
    STAGE_TILE(bufB(0), Bbase + (long long)row * fa);
    for (int kt = 0; kt < nk; ++kt) {
        __syncthreads();  // staged tile (glds) complete for buf[cur]
        if (kt + 1 < nk) {
            STAGE_TILE(bufA(cur ^ 1),
                       Abase + (long long)row * fa + (kt + 1) * BK);
            STAGE_TILE(bufB(cur ^ 1),
                       Bbase + (long long)row * fa + (kt + 1) * BK);
        }
        bf16x8 af[4], bfr[4];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            af[mi] = frag_load(bufA(cur), wr + mi * 16 + fr, fk);
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            bfr[ni] = frag_load(bufB(cur), wc + ni * 16 + fr, fk);
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] =
                    MFMA_BF16_16x16x32(af[mi], bfr[ni], acc[mi][ni]);
        cur ^= 1;
    }

    // epilogue: z -> masked dloss -> bf16, transposed through LDS
    __syncthreads();
    const int rw = lane >> 4;
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            const int rowb = wr + mi * 16 + rw * 4;
            const int colb = wc + ni * 16 + fr;
            const int cls = cls_s[colb];
            const int cfo = cfold_s[colb];
            const int c2 = cls2_s[colb];
            short4v g4;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = rowb + r;
                const float z = acc[mi][ni][r];
                const float yv = y_s[row];
                const float t =
                    (cls < 0) ? yv : (yv == (float)cls ? 1.f : 0.f);
                const float g = dloss(loss_id, z, t) * rw_s[row];
                bool train = (fold_s[row] != cfo) && (bm + row < m);
                if (c2 >= 0)  // one-vs-one: only the pair's rows train
                    train = train &&
                            (yv == (float)cls || yv == (float)c2);
                g4[r] = __builtin_bit_cast(short,
                                           f32_to_bf16(train ? g : 0.f));
            }
            *(short4v*)&ldsC[colb * LDC + rowb] = g4;
        }
    }
    __syncthreads();
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
        const int p = tid + 256 * i;
        const int col = p >> 3;
        const int ch = (p & 7) * 16;
        int4v v0 = *(int4v*)&ldsC[col * LDC + ch];
        int4v v1 = *(int4v*)&ldsC[col * LDC + ch + 8];
        __bf16* dst = GT + (long long)(bn + col) * gt_stride + bm + ch;
        *(int4v*)dst = v0;
        *(int4v*)(dst + 8) = v1;
    }
}

// ---------------------------------------------------------------------- //
// K2: Grad partial slabs, split-K over minibatch rows
// grid: (fa_store/BM, ncols_pad/BN, SPLITK), block 256
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_grad_partial(
    const __bf16* __restrict__ XsT,  // [fa_store][n_pad]
    const __bf16* __restrict__ GT,   // [ncols_pad][gt_stride]
    float* __restrict__ partial,     // [SPLITK][fa][ncols_pad]
    int start, int m_pad, long long n_pad, int fa, int ncols_pad,
    int gt_stride, int k_chunk)
{
    extern __shared__ __attribute__((aligned(16))) char sm[];

    const int tid = threadIdx.x;
    const int bf = blockIdx.x * BM;   // feature-row offset
    const int bn = blockIdx.y * BN;   // col offset
    const int z = blockIdx.z;
    const int lane = tid & 63;
    const int w = tid >> 6;

    const __bf16* Abase = XsT + (long long)bf * n_pad + start;
    const __bf16* Bbase = GT + (long long)bn * gt_stride;

    const int k0 = z * k_chunk;
    const int k1 = min(m_pad, k0 + k_chunk);

    const int wr = ((w >> 1) & 1) * 64;
    const int wc = (w & 1) * 64;
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;

    f32x4 acc[4][4] = {};
    int cur = 0;
    if (k0 < k1) {
        STAGE_TILE(bufA(0), Abase + (long long)row * n_pad + k0);
        STAGE_TILE(bufB(0), Bbase + (long long)row * gt_stride + k0);
    }
    for (int kt = k0; kt < k1; kt += BK) {
        __syncthreads();
        if (kt + BK < k1) {
            STAGE_TILE(bufA(cur ^ 1),
                       Abase + (long long)row * n_pad + kt + BK);
            STAGE_TILE(bufB(cur ^ 1),
                       Bbase + (long long)row * gt_stride + kt + BK);
        }
        bf16x8 af[4], bfr[4];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            af[mi] = frag_load(bufA(cur), wr + mi * 16 + fr, fk);
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            bfr[ni] = frag_load(bufB(cur), wc + ni * 16 + fr, fk);
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] =
                    MFMA_BF16_16x16x32(af[mi], bfr[ni], acc[mi][ni]);
        cur ^= 1;
    }

    float* out = partial + (long long)z * fa * ncols_pad;
    const int rw = lane >> 4;
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        const int rowb = bf + wr + mi * 16 + rw * 4;
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            const int col = bn + wc + ni * 16 + fr;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                if (rowb + r < fa)
                    out[(long long)(rowb + r) * ncols_pad + col] =
                        acc[mi][ni][r];
            }
        }
    }
}

// ---------------------------------------------------------------------- //
// K3: reduce partials, update W (+momentum), refresh WbfT
// grid: (ncols_pad/128, fa), block 128
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(128) void k_reduce_update(
    const float* __restrict__ partial,
    float* __restrict__ W,
    float* __restrict__ V,
    __bf16* __restrict__ WbfT,
    const float* __restrict__ col_lr,
    const float* __restrict__ col_l2,
    const unsigned char* __restrict__ fmask,  // [fa][ncols_pad] or null:
    float inv_m, float lr_scale, float momentum,  // 0 pins W to 0 (the
    int intercept_row, int splitk, int fa, int ncols_pad)  // per-column
{                                             // feature-subset device path)
    const int col = blockIdx.x * 128 + threadIdx.x;
    const int row = blockIdx.y;
    const long long e = (long long)row * ncols_pad + col;

    float s = 0.f;
    for (int z = 0; z < splitk; ++z)
        s += partial[(long long)z * fa * ncols_pad + e];

    float wv = W[e];
    const float l2 = (row == intercept_row) ? 0.f : col_l2[col];
    const float grad = s * inv_m + l2 * wv;
    float step = col_lr[col] * lr_scale * grad;
    if (momentum > 0.f) {
        const float v = V[e] * momentum + step;
        V[e] = v;
        step = v;
    }
    wv -= step;
    if (fmask != nullptr && fmask[e] == 0) {
        wv = 0.f;
        if (momentum > 0.f) V[e] = 0.f;
    }
    W[e] = wv;
    WbfT[(long long)col * fa + row] = f32_to_bf16(wv);
}

// ---------------------------------------------------------------------- //
// host launcher
// ---------------------------------------------------------------------- //
extern "C" hipError_t skdist_sgd_step(
    const void* Xs, const void* XsT, const void* WbfT_in,
    void* GT, void* W, void* V, void* WbfT, void* partial,
    const void* y, const void* fold,
    const void* col_class, const void* col_fold, const void* col_class2,
    const void* col_lr, const void* col_l2, const void* fmask,
    const void* row_w, float inv_m,
    long long start, long long m, long long n, long long n_pad,
    long long fa_store, int fa, int ncols_pad,
    int gt_stride, int splitk, int loss_id,
    float lr_scale, float momentum, int intercept_row,
    hipStream_t stream)
{
    const int m_pad = (int)((m + BM - 1) / BM) * BM;
    {
        dim3 grid(m_pad / BM, ncols_pad / BN);
        size_t lds = (size_t)BN * LDC * 2 + 3072;
        hipLaunchKernelGGL(k_fwd_gt, grid, dim3(256), lds, stream,
                           (const __bf16*)Xs, (const __bf16*)WbfT_in,
                           (__bf16*)GT, (const float*)y, (const int*)fold,
                           (const int*)col_class, (const int*)col_fold,
                           (const int*)col_class2, (const float*)row_w,
                           (int)start, (int)m, n, fa, ncols_pad, gt_stride,
                           loss_id);
        HIP_CHECK(hipGetLastError());
    }
    {
        const int k_chunk = (int)((m_pad / splitk + BK - 1) / BK) * BK;
        dim3 grid((unsigned)(fa_store / BM), ncols_pad / BN, splitk);
        size_t lds = 32768;
        hipLaunchKernelGGL(k_grad_partial, grid, dim3(256), lds, stream,
                           (const __bf16*)XsT, (const __bf16*)GT,
                           (float*)partial, (int)start, m_pad, n_pad, fa,
                           ncols_pad, gt_stride, k_chunk);
        HIP_CHECK(hipGetLastError());
    }
    {
        dim3 grid(ncols_pad / 128, fa);
        hipLaunchKernelGGL(k_reduce_update, grid, dim3(128), 0, stream,
                           (const float*)partial, (float*)W, (float*)V,
                           (__bf16*)WbfT, (const float*)col_lr,
                           (const float*)col_l2,
                           (const unsigned char*)fmask, inv_m,
                           lr_scale, momentum, intercept_row, splitk, fa,
                           ncols_pad);
        HIP_CHECK(hipGetLastError());
    }
    return hipSuccess;
}

// ---------------------------------------------------------------------- //
// K4: fused standardize + intercept/pad augment + bf16 cast
// out[i][j] = (X[i][j] - mean[j]) * inv_std[j]   for j <  f
//           = 1.0                                 for j == f (intercept)
//           = 0.0                                 for j >  f (K-pad)
// One pass over X instead of the eager sub/div/cat/cast chain
// (DeviceDataset build, skdist_amd/models/_sgd.py).
// grid: ceil(n*fa/8/256), block 256; each thread emits one 16 B chunk.
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_standardize(
    const float* __restrict__ X, const float* __restrict__ mean,
    const float* __restrict__ inv_std, __bf16* __restrict__ out,
    long long n, int f, int fa)
{
    const long long chunk =
        (long long)blockIdx.x * blockDim.x + threadIdx.x;
    const long long total = n * (long long)fa / 8;
    if (chunk >= total) return;
    const long long e0 = chunk * 8;
    const long long row = e0 / fa;
    const int c0 = (int)(e0 - row * fa);
    const float* xr = X + row * (long long)f;
    short4v lo, hi;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
        const int c = c0 + j;
        float v;
        if (c < f)
            v = (xr[c] - mean[c]) * inv_std[c];
        else
            v = (c == f) ? 1.f : 0.f;
        const short b = __builtin_bit_cast(short, f32_to_bf16(v));
        if (j < 4) lo[j] = b; else hi[j - 4] = b;
    }
    *(short4v*)(out + e0) = lo;
    *(short4v*)(out + e0 + 4) = hi;
}

extern "C" hipError_t skdist_standardize(
    const void* X, const void* mean, const void* inv_std, void* out,
    long long n, int f, int fa, hipStream_t stream)
{
    if (fa % 8 != 0) return hipErrorInvalidValue;
    const long long chunks = n * (long long)fa / 8;
    const int blocks = (int)((chunks + 255) / 256);
    hipLaunchKernelGGL(k_standardize, dim3(blocks), dim3(256), 0, stream,
                       (const float*)X, (const float*)mean,
                       (const float*)inv_std, (__bf16*)out, n, f, fa);
    return hipGetLastError();
}

// ---------------------------------------------------------------------- //
// K5: fused test-fold scoring — GEMM tile + per-column sufficient stats
// (the k-fold CV scoring kernel; SURVEY.md §2.4 "k-fold split + score").
// The caller gathers ONE fold's test rows into a 128-padded bf16 buffer
// (pad rows zero, y pad = -1e30 sentinel -> excluded), so no masks are
// needed: every row scores every column.
//   mode 0 (binary accuracy): out[col*2]   += #(sign(z) == y)
//                             out[col*2+1] += #rows           (valid)
//   mode 1 (regression):      out[col*3]   += (z-y)^2
//                             out[col*3+1] += y, out[col*3+2] += y^2
// grid: (m_pad/128, ncols_pad/128), block 256 (4 waves of 64x64).
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_score(
    const __bf16* __restrict__ Xf,    // [m_pad][fa] gathered fold rows
    const __bf16* __restrict__ WbfT,  // [ncols_pad][fa]
    const float* __restrict__ yf,     // [m_pad] (-1e30 pad sentinel)
    float* __restrict__ out,          // [ncols_pad * (2|3)]
    int fa, int ncols_pad, int mode)
{
    extern __shared__ __attribute__((aligned(16))) char sm[];
    #define sbufA(b) ((__bf16*)(sm + (b) * 16384))
    #define sbufB(b) ((__bf16*)(sm + 8192 + (b) * 16384))
    float* y_s = (float*)(sm + 32768);        // [128]
    float* red = (float*)(sm + 33280);        // [3][132] per-col partials

    const int tid = threadIdx.x;
    const int bm = blockIdx.x * 128;
    const int bn = blockIdx.y * 128;
    const int lane = tid & 63;
    const int w = tid >> 6;
    if (tid < 128) y_s[tid] = yf[bm + tid];

    const __bf16* Abase = Xf + (long long)bm * fa;
    const __bf16* Bbase = WbfT + (long long)bn * fa;
    const int wr = ((w >> 1) & 1) * 64;
    const int wc = (w & 1) * 64;
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;

    f32x4 acc[4][4] = {};
    const int nk = fa / BK;
    int cur = 0;
    {   // stage first tiles (STAGE_TILE expects `row`, `lane`, `w`)
        _Pragma("unroll") for (int i = 0; i < 2; ++i) {
            const int c = w + 4 * i;
            const int row = 16 * c + (lane >> 2);
            const int slot = (lane & 3) ^ ((row >> 2) & 3);
            __builtin_amdgcn_global_load_lds(
                (const unsigned int*)(Abase + (long long)row * fa +
                                      slot * 8),
                (unsigned int*)(sbufA(0) + c * 512), 16, 0, 0);
            __builtin_amdgcn_global_load_lds(
                (const unsigned int*)(Bbase + (long long)row * fa +
                                      slot * 8),
                (unsigned int*)(sbufB(0) + c * 512), 16, 0, 0);
        }
    }
    for (int kt = 0; kt < nk; ++kt) {
        __syncthreads();
        if (kt + 1 < nk) {
            _Pragma("unroll") for (int i = 0; i < 2; ++i) {
                const int c = w + 4 * i;
                const int row = 16 * c + (lane >> 2);
                const int slot = (lane & 3) ^ ((row >> 2) & 3);
                __builtin_amdgcn_global_load_lds(
                    (const unsigned int*)(Abase + (long long)row * fa +
                                          (kt + 1) * BK + slot * 8),
                    (unsigned int*)(sbufA(cur ^ 1) + c * 512), 16, 0, 0);
                __builtin_amdgcn_global_load_lds(
                    (const unsigned int*)(Bbase + (long long)row * fa +
                                          (kt + 1) * BK + slot * 8),
                    (unsigned int*)(sbufB(cur ^ 1) + c * 512), 16, 0, 0);
            }
        }
        bf16x8 af[4], bfr[4];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            af[mi] = frag_load(sbufA(cur), wr + mi * 16 + fr, fk);
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            bfr[ni] = frag_load(sbufB(cur), wc + ni * 16 + fr, fk);
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] =
                    MFMA_BF16_16x16x32(af[mi], bfr[ni], acc[mi][ni]);
        cur ^= 1;
    }
    __syncthreads();

    // per-lane partials over this wave's 16 rows x 4 col-groups
    const int rw = lane >> 4;
    float s0[4] = {}, s1[4] = {}, s2[4] = {};
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = wr + mi * 16 + rw * 4 + r;
                const float yv = y_s[row];
                if (yv < -1e29f) continue;   // pad row
                const float z = acc[mi][ni][r];
                if (mode == 0) {
                    s0[ni] += ((z >= 0.f) == (yv != 0.f)) ? 1.f : 0.f;
                    s1[ni] += 1.f;
                } else {
                    const float e = z - yv;
                    s0[ni] += e * e;
                    s1[ni] += yv;
                    s2[ni] += yv * yv;
                }
            }
        }
    }
    // reduce across the 16 lanes-groups sharing a column (rows axis):
    // lanes with same (lane&15, wc) hold different rows; use LDS.
    const int nstat = (mode == 0) ? 2 : 3;
    for (int st = 0; st < nstat; ++st) {
        __syncthreads();
        for (int i = tid; i < 132; i += 256) red[i] = 0.f;
        __syncthreads();
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            const float v = (st == 0) ? s0[ni] : (st == 1 ? s1[ni]
                                                          : s2[ni]);
            atomicAdd(&red[wc + ni * 16 + fr], v);
        }
        __syncthreads();
        for (int c = tid; c < 128; c += 256)
            atomicAdd(&out[(long long)(bn + c) * nstat + st], red[c]);
    }
}

extern "C" hipError_t skdist_score(
    const void* Xf, const void* WbfT, const void* yf, void* out,
    long long m_pad, int fa, int ncols_pad, int mode,
    hipStream_t stream)
{
    dim3 grid((unsigned)(m_pad / 128), ncols_pad / 128);
    hipLaunchKernelGGL(k_score, grid, dim3(256), 33280 + 132 * 4, stream,
                       (const __bf16*)Xf, (const __bf16*)WbfT,
                       (const float*)yf, (float*)out, fa, ncols_pad,
                       mode);
    return hipGetLastError();
}
