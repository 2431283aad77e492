// Batched forest inference — CDNA4 (gfx950) kernels.
//
// Device analog of the reference's executor-side model.predict inside the
// pandas UDF (skdist/distribute/predict.py:160-178) for tree ensembles:
// the fitted forest is flattened into four node arrays + a leaf-value
// table resident in HBM (small enough to sit in L2/L3), and every thread
// walks all trees for one row, accumulating leaf payloads in registers.
//
// Layouts:
//   X       [rows][f]  f32 row-major
//   feat    [total_nodes] int32   (-1 = leaf; leaf's `left` indexes values)
//   thr     [total_nodes] f32     (x[feat] <= thr  → left)
//   left/right [total_nodes] int32 (absolute node ids, pre-offset per tree)
//   roots   [n_trees] int32
//   values  [n_values][vs] f32
#include "common.h"

#define MAXVS 32

extern "C" __global__ __launch_bounds__(256) void k_forest_predict(
    const float* __restrict__ X, const int* __restrict__ feat,
    const float* __restrict__ thr, const int* __restrict__ left,
    const int* __restrict__ right, const int* __restrict__ roots,
    const float* __restrict__ values, float* __restrict__ out,
    long long rows, int f, int n_trees, int vs) {
    const long long r = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= rows) return;
    const float* x = X + r * f;
    float acc[MAXVS];
#pragma unroll
    for (int c = 0; c < MAXVS; ++c) acc[c] = 0.f;
    for (int t = 0; t < n_trees; ++t) {
        int node = roots[t];
        int jf = feat[node];
        while (jf >= 0) {
            node = (x[jf] <= thr[node]) ? left[node] : right[node];
            jf = feat[node];
        }
        const float* v = values + (long long)left[node] * vs;
#pragma unroll
        for (int c = 0; c < MAXVS; ++c)
            if (c < vs) acc[c] += v[c];
    }
    const float inv = 1.f / (float)n_trees;
#pragma unroll
    for (int c = 0; c < MAXVS; ++c)
        if (c < vs) out[r * vs + c] = acc[c] * inv;
}

extern "C" __global__ __launch_bounds__(256) void k_forest_apply(
    const float* __restrict__ X, const int* __restrict__ feat,
    const float* __restrict__ thr, const int* __restrict__ left,
    const int* __restrict__ right, const int* __restrict__ roots,
    int* __restrict__ out_leaf, long long rows, int f, int n_trees) {
    const long long r = (long long)blockIdx.x * blockDim.x + threadIdx.x;
    if (r >= rows) return;
    const float* x = X + r * f;
    for (int t = 0; t < n_trees; ++t) {
        int node = roots[t];
        int jf = feat[node];
        while (jf >= 0) {
            node = (x[jf] <= thr[node]) ? left[node] : right[node];
            jf = feat[node];
        }
        out_leaf[r * n_trees + t] = node;
    }
}

extern "C" hipError_t skdist_forest_predict(
    const void* X, const void* feat, const void* thr, const void* left,
    const void* right, const void* roots, const void* values, void* out,
    long long rows, int f, int n_trees, int vs, hipStream_t stream) {
    if (vs > MAXVS) return hipErrorInvalidValue;
    const int blocks = (int)((rows + 255) / 256);
    hipLaunchKernelGGL(k_forest_predict, dim3(blocks), dim3(256), 0,
                       stream, (const float*)X, (const int*)feat,
                       (const float*)thr, (const int*)left,
                       (const int*)right, (const int*)roots,
                       (const float*)values, (float*)out, rows, f, n_trees,
                       vs);
    return hipGetLastError();
}

extern "C" hipError_t skdist_forest_apply(
    const void* X, const void* feat, const void* thr, const void* left,
    const void* right, const void* roots, void* out_leaf, long long rows,
    int f, int n_trees, hipStream_t stream) {
    const int blocks = (int)((rows + 255) / 256);
    hipLaunchKernelGGL(k_forest_apply, dim3(blocks), dim3(256), 0, stream,
                       (const float*)X, (const int*)feat,
                       (const float*)thr, (const int*)left,
                       (const int*)right, (const int*)roots,
                       (int*)out_leaf, rows, f, n_trees);
    return hipGetLastError();
}
