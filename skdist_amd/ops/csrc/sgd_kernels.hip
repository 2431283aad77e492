// Batched mini-batch-SGD linear solver — CDNA4 (gfx950) kernels.
//
// One SGD step over a minibatch of m rows trains ncols independent linear
// models (columns) at once (see skdist_amd/models/_sgd.py for the math and
// reference torch implementation; reference inventory: SURVEY.md §2.4 row 1).
//
//   K1 k_fwd_gt:        GT[c][i] = mask*dloss( (Xs·W)[i][c], target )
//                       MFMA bf16 GEMM (M=m, N=ncols, K=fa) with the loss
//                       gradient fused into the epilogue and the result
//                       transposed through LDS so K2 reads it K-contiguous.
//   K2 k_grad_partial:  partial[z] = XsTᵀ-slab GEMM: Grad = Xᵀ·G, split-K
//                       over the minibatch rows into SPLITK deterministic
//                       partial slabs (no atomics -> bitwise reproducible).
//   K3 k_reduce_update: W -= lr * (Σz partial / m + λ∘W); refreshes the
//                       bf16 transposed weight copy WbfT used by K1.
//
// Data layouts (all row-major, K-contiguous for MFMA fragment loads):
//   Xs    [n][fa]          bf16  epoch-shuffled, fa % 32 == 0
//   XsT   [fa][n]          bf16  transpose of Xs
//   WbfT  [ncols_pad][fa]  bf16  transposed bf16 copy of W
//   W, V  [fa][ncols_pad]  f32   master weights / momentum
//   GT    [ncols_pad][gts] bf16  per-step gradient scratch
//   partial [SPLITK][fa][ncols_pad] f32
#include "common.h"

#define BM 128
#define BN 128
#define BK 32
#define LDA 40  // LDS row stride (elems) for 32-elem K tiles, +8 pad
#define LDC 132 // LDS row stride for the 128-wide epilogue transpose tile

// ---------------------------------------------------------------------- //
// K1: fused forward GEMM + loss gradient + transposed store
// grid: (ceil(m/BM), ncols_pad/BN), block: 256
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_fwd_gt(
    const __bf16* __restrict__ Xs,    // [n][fa]
    const __bf16* __restrict__ WbfT,  // [ncols_pad][fa]
    __bf16* __restrict__ GT,          // [ncols_pad][gt_stride]
    const float* __restrict__ y,      // [n]
    const int* __restrict__ fold,     // [n]
    const int* __restrict__ col_class,
    const int* __restrict__ col_fold,
    int start, int m, long long n, int fa, int ncols_pad, int gt_stride,
    int loss_id)
{
    extern __shared__ __attribute__((aligned(16))) char sm[];
    __bf16* ldsA = (__bf16*)sm;                    // [BM][LDA]
    __bf16* ldsB = (__bf16*)(sm + BM * LDA * 2);   // [BN][LDA]
    __bf16* ldsC = (__bf16*)sm;                    // [BN][LDC] (aliases A+B)
    char* meta = sm + BN * LDC * 2;                // after C region
    float* y_s = (float*)meta;                     // [BM]
    int* fold_s = (int*)(meta + 512);              // [BM]
    int* cls_s = (int*)(meta + 1024);              // [BN]
    int* cfold_s = (int*)(meta + 1536);            // [BN]

    const int tid = threadIdx.x;
    const int bm = blockIdx.x * BM;
    const int bn = blockIdx.y * BN;

    // bounded descriptors: OOB -> 0
    const long long rows_left = n - (long long)(start + bm);
    const int rows_here = rows_left < BM ? (rows_left < 0 ? 0 : (int)rows_left) : BM;
    auto rsrcX = make_rsrc(Xs + (long long)(start + bm) * fa,
                           (unsigned long long)rows_here * fa * 2);
    auto rsrcW = make_rsrc(WbfT + (long long)bn * fa,
                           (unsigned long long)BN * fa * 2);

    // per-block row/col metadata into LDS
    if (tid < BM) {
        int r = start + bm + tid;
        bool ok = tid < rows_here;
        y_s[tid] = ok ? y[r] : 0.f;
        fold_s[tid] = ok ? fold[r] : -9;
    } else {
        int c = tid - BM;  // 0..127
        cls_s[c] = col_class[bn + c];
        cfold_s[c] = col_fold[bn + c];
    }

    const int lane = tid & 63;
    const int w = tid >> 6;
    const int wr = (w >> 1) * 64;
    const int wc = (w & 1) * 64;
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;

    f32x4 acc[4][4] = {};

    const int nk = fa / BK;
    for (int kt = 0; kt < nk; ++kt) {
        __syncthreads();
        // stage A and B tiles: 2 x 16B pieces each per thread
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            int p = tid + 256 * i;
            int row = p >> 2;
            int kp = (p & 3) * 8;
            int4v va = buf_load_dwordx4(
                rsrcX, (row * fa + kt * BK + kp) * 2);
            *(int4v*)&ldsA[row * LDA + kp] = va;
            int4v vb = buf_load_dwordx4(
                rsrcW, (row * fa + kt * BK + kp) * 2);
            *(int4v*)&ldsB[row * LDA + kp] = vb;
        }
        __syncthreads();
        bf16x8 af[4], bf[4];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            af[mi] = *(bf16x8*)&ldsA[(wr + mi * 16 + fr) * LDA + fk];
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            bf[ni] = *(bf16x8*)&ldsB[(wc + ni * 16 + fr) * LDA + fk];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] = MFMA_BF16_16x16x32(af[mi], bf[ni], acc[mi][ni]);
    }

    // epilogue: z -> masked dloss -> bf16, transposed through LDS
    __syncthreads();
    const int rw = lane >> 4;  // 0..3
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            int rowb = wr + mi * 16 + rw * 4;
            int colb = wc + ni * 16 + fr;
            int cls = cls_s[colb];
            int cfo = cfold_s[colb];
            short4v g4;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                int row = rowb + r;
                float z = acc[mi][ni][r];
                float yv = y_s[row];
                float t = (cls < 0) ? yv : (yv == (float)cls ? 1.f : 0.f);
                float g = dloss(loss_id, z, t);
                bool train = (fold_s[row] != cfo) && (bm + row < m);
                __bf16 gb = f32_to_bf16(train ? g : 0.f);
                g4[r] = __builtin_bit_cast(short, gb);
            }
            *(short4v*)&ldsC[colb * LDC + rowb] = g4;
        }
    }
    __syncthreads();
    // cooperative coalesced store: GT[bn+col][bm + ch*16 .. +15]
    #pragma unroll
    for (int i = 0; i < 4; ++i) {
        int p = tid + 256 * i;
        int col = p >> 3;
        int ch = (p & 7) * 16;
        int4v v0 = *(int4v*)&ldsC[col * LDC + ch];
        int4v v1 = *(int4v*)&ldsC[col * LDC + ch + 8];
        __bf16* dst = GT + (long long)(bn + col) * gt_stride + bm + ch;
        *(int4v*)dst = v0;
        *(int4v*)(dst + 8) = v1;
    }
}

// ---------------------------------------------------------------------- //
// K2: Grad partial slabs, split-K over minibatch rows
// grid: (ceil(fa/BM), ncols_pad/BN, SPLITK), block: 256
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_grad_partial(
    const __bf16* __restrict__ XsT,  // [fa][n]
    const __bf16* __restrict__ GT,   // [ncols_pad][gt_stride]
    float* __restrict__ partial,     // [SPLITK][fa][ncols_pad]
    int start, int m_pad, long long n, int fa, int ncols_pad, int gt_stride,
    int k_chunk)
{
    extern __shared__ __attribute__((aligned(16))) char sm[];
    __bf16* ldsA = (__bf16*)sm;                    // [BM][LDA] rows of XsT
    __bf16* ldsB = (__bf16*)(sm + BM * LDA * 2);   // [BN][LDA] rows of GT

    const int tid = threadIdx.x;
    const int bf = blockIdx.x * BM;   // feature-row offset
    const int bn = blockIdx.y * BN;   // col offset
    const int z = blockIdx.z;

    auto rsrcX = make_rsrc(XsT, (unsigned long long)fa * n * 2);
    auto rsrcG = make_rsrc(GT + (long long)bn * gt_stride,
                           (unsigned long long)BN * gt_stride * 2);

    const int k0 = z * k_chunk;
    const int k1 = min(m_pad, k0 + k_chunk);

    const int lane = tid & 63;
    const int w = tid >> 6;
    const int wr = (w >> 1) * 64;
    const int wc = (w & 1) * 64;
    const int fr = lane & 15;
    const int fk = (lane >> 4) * 8;

    f32x4 acc[4][4] = {};

    for (int kt = k0; kt < k1; kt += BK) {
        __syncthreads();
        #pragma unroll
        for (int i = 0; i < 2; ++i) {
            int p = tid + 256 * i;
            int row = p >> 2;
            int kp = (p & 3) * 8;
            // A: XsT[bf+row][start + kt + kp ...]; rows >= fa read 0 (OOB)
            int4v va = buf_load_dwordx4(
                rsrcX,
                (int)(((long long)(bf + row) * n + start + kt + kp) * 2));
            *(int4v*)&ldsA[row * LDA + kp] = va;
            int4v vb = buf_load_dwordx4(
                rsrcG, (row * gt_stride + kt + kp) * 2);
            *(int4v*)&ldsB[row * LDA + kp] = vb;
        }
        __syncthreads();
        bf16x8 af[4], bfr[4];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            af[mi] = *(bf16x8*)&ldsA[(wr + mi * 16 + fr) * LDA + fk];
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni)
            bfr[ni] = *(bf16x8*)&ldsB[(wc + ni * 16 + fr) * LDA + fk];
        #pragma unroll
        for (int mi = 0; mi < 4; ++mi)
            #pragma unroll
            for (int ni = 0; ni < 4; ++ni)
                acc[mi][ni] =
                    MFMA_BF16_16x16x32(af[mi], bfr[ni], acc[mi][ni]);
    }

    // store partial slab (f32, coalesced along cols), guard feature rows
    float* out = partial + (long long)z * fa * ncols_pad;
    const int rw = lane >> 4;
    #pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
        int rowb = bf + wr + mi * 16 + rw * 4;
        #pragma unroll
        for (int ni = 0; ni < 4; ++ni) {
            int col = bn + wc + ni * 16 + fr;
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
// grid: (ncols_pad/128, fa), block: 128
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(128) void k_reduce_update(
    const float* __restrict__ partial,  // [SPLITK][fa][ncols_pad]
    float* __restrict__ W,              // [fa][ncols_pad]
    float* __restrict__ V,              // [fa][ncols_pad] or nullptr
    __bf16* __restrict__ WbfT,          // [ncols_pad][fa]
    const float* __restrict__ col_lr,
    const float* __restrict__ col_l2,
    float inv_m, float lr_scale, float momentum,
    int intercept_row, int splitk, int fa, int ncols_pad)
{
    const int col = blockIdx.x * 128 + threadIdx.x;
    const int row = blockIdx.y;
    const long long e = (long long)row * ncols_pad + col;

    float s = 0.f;
    for (int z = 0; z < splitk; ++z)
        s += partial[(long long)z * fa * ncols_pad + e];

    float wv = W[e];
    float l2 = (row == intercept_row) ? 0.f : col_l2[col];
    float grad = s * inv_m + l2 * wv;
    float step = col_lr[col] * lr_scale * grad;
    if (momentum > 0.f) {
        float v = V[e] * momentum + step;
        V[e] = v;
        step = v;
    }
    wv -= step;
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
    const void* col_class, const void* col_fold,
    const void* col_lr, const void* col_l2,
    long long start, long long m, long long n, int fa, int ncols_pad,
    int gt_stride, int splitk, int loss_id,
    float lr_scale, float momentum, int intercept_row,
    hipStream_t stream)
{
    const int m_pad = (int)((m + BM - 1) / BM) * BM;
    {
        dim3 grid(m_pad / BM, ncols_pad / BN);
        size_t lds = (size_t)BN * LDC * 2 + 2048;
        hipLaunchKernelGGL(k_fwd_gt, grid, dim3(256), lds, stream,
                           (const __bf16*)Xs, (const __bf16*)WbfT_in,
                           (__bf16*)GT, (const float*)y, (const int*)fold,
                           (const int*)col_class, (const int*)col_fold,
                           (int)start, (int)m, n, fa, ncols_pad, gt_stride,
                           loss_id);
        HIP_CHECK(hipGetLastError());
    }
    {
        int k_chunk = ((m_pad / splitk + BK - 1) / BK) * BK;
        dim3 grid((fa + BM - 1) / BM, ncols_pad / BN, splitk);
        size_t lds = (size_t)(BM + BN) * LDA * 2;
        hipLaunchKernelGGL(k_grad_partial, grid, dim3(256), lds, stream,
                           (const __bf16*)XsT, (const __bf16*)GT,
                           (float*)partial, (int)start, m_pad, n, fa,
                           ncols_pad, gt_stride, k_chunk);
        HIP_CHECK(hipGetLastError());
    }
    {
        dim3 grid(ncols_pad / 128, fa);
        hipLaunchKernelGGL(k_reduce_update, grid, dim3(128), 0, stream,
                           (const float*)partial, (float*)W, (float*)V,
                           (__bf16*)WbfT, (const float*)col_lr,
                           (const float*)col_l2, (float)(1.0 / (double)m),
                           lr_scale, momentum, intercept_row, splitk, fa,
                           ncols_pad);
        HIP_CHECK(hipGetLastError());
    }
    return hipSuccess;
}
