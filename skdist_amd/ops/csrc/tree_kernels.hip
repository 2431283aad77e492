// Batched histogram decision-tree builder — CDNA4 (gfx950) kernels.
//
// MI355X-native replacement for the per-tree sklearn Cython builder the
// reference fans out (reference worker
// skdist/distribute/ensemble.py:68-109).  A batch of TB trees grows
// level-synchronously against ONE HBM-resident binned copy of X:
//
//   K1 k_tree_hist:    per-(node,feature,bin) weighted stats, LDS-staged
//   K2 k_tree_split:   best (feature,bin) per frontier node — gini /
//                      entropy / mse scan with exact max_features
//                      subsampling and ExtraTrees random-threshold mode
//   K3 k_part_count:   per-chunk left-row counts for the stable partition
//   K4 k_part_scatter: stable partition of each node's sample segment
//
// Host orchestration + torch eager reference: skdist_amd/models/forest.py.
//
// Layouts:
//   codes      [n][fp]   uint8   row-major quantile bin codes (fp = f
//                                     padded to a multiple of 4 for uchar4 loads)
//   sample_idx [TB][n]   int32   per-tree row ids, node segments contiguous
//   weights    [TB][n]   uint8   bootstrap multiplicities (0 = out of bag)
//   hist       [NF][f][nbins][S] f32   S = n_classes (cls) | 3 (w,wy,wyy)
//   chunks     [n_chunks][4] int32     {node_slot, tree_slot, row_start,
//                                       row_count}
//
// Determinism: classification stats are integer-valued f32 atomics (exact,
// order-independent); regression wy/wyy sums are fp32 atomics whose
// rounding depends on arrival order — ties between split candidates are
// broken by (feature, bin) so only near-exact-tie splits can differ.
#include "common.h"

#define WAVE 64

static __device__ __forceinline__ unsigned wang_hash(unsigned s) {
    s = (s ^ 61u) ^ (s >> 16);
    s *= 9u;
    s ^= s >> 4;
    s *= 0x27d4eb2du;
    s ^= s >> 15;
    return s;
}

// ---------------------------------------------------------------------- //
// K1: histogram build
// grid: (n_chunks, n_feature_groups), block 256
// dynamic LDS: fg * nbins * S floats
// ---------------------------------------------------------------------- //
typedef __attribute__((ext_vector_type(4))) unsigned char uchar4v;

extern "C" __global__ __launch_bounds__(256) void k_tree_hist(
    const unsigned char* __restrict__ codes,    // [n][fp] row-major
    const int* __restrict__ y_int,              // [n] (cls) or nullptr
    const float* __restrict__ y_f,              // [n] (reg) or nullptr
    const unsigned char* __restrict__ weights,  // [TB][n]
    const int* __restrict__ sample_idx,         // [TB][n]
    const int* __restrict__ chunks,             // [n_chunks][4]
    float* __restrict__ hist,                   // [NF][f][nbins][S]
    long long n, int f, int fp, int nbins, int S, int is_cls, int fg) {
    extern __shared__ float lds[];
    const int chunk = blockIdx.x;
    const int g0 = blockIdx.y * fg;
    const int nf_g = min(fg, f - g0);
    const int node_slot = chunks[chunk * 4 + 0];
    const int tree_slot = chunks[chunk * 4 + 1];
    const int row_start = chunks[chunk * 4 + 2];
    const int row_count = chunks[chunk * 4 + 3];

    const int lds_sz = nf_g * nbins * S;
    for (int i = threadIdx.x; i < lds_sz; i += blockDim.x) lds[i] = 0.f;
    __syncthreads();

    const int* si = sample_idx + (long long)tree_slot * n + row_start;
    const unsigned char* wrow = weights + (long long)tree_slot * n;
    // vector path: 4 feature codes per uchar4 load (codes rows are
    // padded to fp, a multiple of 4, so over-reads land in zero pad)
    const bool vec = (g0 & 3) == 0;
    for (int r = threadIdx.x; r < row_count; r += blockDim.x) {
        const int i = si[r];
        const float w = (float)wrow[i];
        int stat = 0;
        float v1 = 0.f, v2 = 0.f;
        if (is_cls) {
            stat = y_int[i];
        } else {
            const float yv = y_f[i];
            v1 = w * yv;
            v2 = w * yv * yv;
        }
        const unsigned char* crow = codes + (long long)i * fp + g0;
        if (vec) {
            const uchar4v* c4 = (const uchar4v*)crow;
            for (int q = 0; q * 4 < nf_g; ++q) {
                const uchar4v b4 = c4[q];
                #pragma unroll
                for (int t = 0; t < 4; ++t) {
                    const int jj = q * 4 + t;
                    if (jj >= nf_g) break;
                    float* h = lds + ((jj * nbins + (int)b4[t]) * S);
                    if (is_cls) {
                        atomicAdd(h + stat, w);
                    } else {
                        atomicAdd(h + 0, w);
                        atomicAdd(h + 1, v1);
                        atomicAdd(h + 2, v2);
                    }
                }
            }
        } else {
            for (int jj = 0; jj < nf_g; ++jj) {
                float* h = lds + ((jj * nbins + (int)crow[jj]) * S);
                if (is_cls) {
                    atomicAdd(h + stat, w);
                } else {
                    atomicAdd(h + 0, w);
                    atomicAdd(h + 1, v1);
                    atomicAdd(h + 2, v2);
                }
            }
        }
    }
    __syncthreads();

    float* gh = hist + ((long long)node_slot * f + g0) * nbins * S;
    for (int i = threadIdx.x; i < lds_sz; i += blockDim.x) {
        const float v = lds[i];
        if (v != 0.f) atomicAdd(gh + i, v);
    }
}

// ---------------------------------------------------------------------- //
// K2: split finding — one 256-thread block per frontier node
// ---------------------------------------------------------------------- //
#define CRIT_GINI 0
#define CRIT_ENTROPY 1
#define CRIT_MSE 2

// impurity from a stats vector (cls: weighted class counts; reg: w,wy,wyy)
static __device__ __forceinline__ float
impurity(const float* s, float w, int S, int is_cls, int crit) {
    if (w <= 0.f) return 0.f;
    if (is_cls) {
        if (crit == CRIT_GINI) {
            float q = 0.f;
            for (int c = 0; c < S; ++c) q += s[c] * s[c];
            return 1.f - q / (w * w);
        }
        float e = 0.f;
        for (int c = 0; c < S; ++c) {
            if (s[c] > 0.f) {
                const float p = s[c] / w;
                e -= p * __log2f(p);
            }
        }
        return e;
    }
    // mse: variance
    const float mean = s[1] / w;
    float v = s[2] / w - mean * mean;
    return v > 0.f ? v : 0.f;
}

extern "C" __global__ __launch_bounds__(256) void k_tree_split(
    const float* __restrict__ hist,        // [NF][f][nbins][S]
    const unsigned* __restrict__ node_seed,  // [NF]
    int f, int nbins, int S, int is_cls, int crit, int m_features,
    int extra_mode, float min_leaf_w,
    int* __restrict__ out_feat,   // [NF]  (-1 = no valid split)
    int* __restrict__ out_bin,    // [NF]
    float* __restrict__ out_wl,   // [NF]  left weighted count
    float* __restrict__ out_gain, // [NF]  impurity improvement (unscaled)
    float* __restrict__ out_imp,  // [NF]  parent impurity
    float* __restrict__ out_stats,  // [NF][S] parent stat totals
    float* __restrict__ out_lstats  // [NF][S] winning split's left stats
) {
    const int node = blockIdx.x;
    const unsigned seed = node_seed[node];
    const float* H = hist + (long long)node * f * nbins * S;
    const int tid = threadIdx.x;

    __shared__ float s_parent[36];     // S <= 32 stats + pad
    __shared__ int s_cnt;
    __shared__ float s_best_gain[256 / WAVE];
    __shared__ int s_best_feat[256 / WAVE];
    __shared__ int s_best_bin[256 / WAVE];
    __shared__ float s_best_wl[256 / WAVE];

    // parent totals from feature 0's histogram
    for (int c = tid; c < S; c += blockDim.x) s_parent[c] = 0.f;
    __syncthreads();
    for (int i = tid; i < nbins * S; i += blockDim.x)
        atomicAdd(&s_parent[i % S], H[i]);
    __syncthreads();
    float wp = 0.f;
    if (is_cls) {
        for (int c = 0; c < S; ++c) wp += s_parent[c];
    } else {
        wp = s_parent[0];
    }
    const float imp_p = impurity(s_parent, wp, S, is_cls, crit);

    // exact m-of-f feature subsample: keep features whose hash ranks in
    // the m smallest (binary search for the m-th smallest hash value)
    unsigned h_thresh = 0xffffffffu;
    if (m_features < f) {
        unsigned lo = 0u, hi = 0xffffffffu;
        for (int it = 0; it < 32; ++it) {
            const unsigned mid = lo + ((hi - lo) >> 1);
            if (tid == 0) s_cnt = 0;
            __syncthreads();
            int cnt = 0;
            for (int j = tid; j < f; j += blockDim.x)
                if (wang_hash(seed ^ (unsigned)(j * 2654435761u)) <= mid)
                    ++cnt;
            atomicAdd(&s_cnt, cnt);
            __syncthreads();
            const int total = s_cnt;
            __syncthreads();
            if (total >= m_features) hi = mid; else lo = mid + 1;
            if (lo >= hi) break;
        }
        h_thresh = hi;
    }

    // per-thread best over its features
    float best_gain = -1.f;
    int best_feat = -1, best_bin = -1;
    float best_wl = 0.f;
    float left[36];
    for (int j = tid; j < f; j += blockDim.x) {
        if (m_features < f &&
            wang_hash(seed ^ (unsigned)(j * 2654435761u)) > h_thresh)
            continue;
        const float* Hj = H + (long long)j * nbins * S;
        // ExtraTrees: pick one random bin in the node's occupied range
        int rand_bin = -1;
        if (extra_mode) {
            int lo = -1, hi = -1;
            for (int b = 0; b < nbins; ++b) {
                float wb = 0.f;
                if (is_cls) {
                    for (int c = 0; c < S; ++c) wb += Hj[b * S + c];
                } else {
                    wb = Hj[b * S];
                }
                if (wb > 0.f) {
                    if (lo < 0) lo = b;
                    hi = b;
                }
            }
            if (lo < 0 || hi <= lo) continue;  // constant feature here
            const unsigned r = wang_hash(seed ^ 0x9e3779b9u ^
                                         (unsigned)(j * 40503u));
            rand_bin = lo + (int)(r % (unsigned)(hi - lo));  // in [lo, hi)
        }
        for (int c = 0; c < S; ++c) left[c] = 0.f;
        float wl = 0.f;
        for (int b = 0; b < nbins - 1; ++b) {
            for (int c = 0; c < S; ++c) left[c] += Hj[b * S + c];
            if (is_cls) {
                wl = 0.f;
                for (int c = 0; c < S; ++c) wl += left[c];
            } else {
                wl = left[0];
            }
            if (extra_mode && b != rand_bin) continue;
            const float wr = wp - wl;
            if (wl < min_leaf_w || wr < min_leaf_w) continue;
            const float imp_l = impurity(left, wl, S, is_cls, crit);
            float right[36];
            for (int c = 0; c < S; ++c) right[c] = s_parent[c] - left[c];
            const float imp_r = impurity(right, wr, S, is_cls, crit);
            const float gain = imp_p - (wl * imp_l + wr * imp_r) / wp;
            // deterministic tie-break: larger gain, then lower feature/bin
            if (gain > best_gain + 1e-12f) {
                best_gain = gain;
                best_feat = j;
                best_bin = b;
                best_wl = wl;
            }
        }
    }

    // wave then block argmax reduce (lower (feat,bin) wins ties)
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        const float og = __shfl_down(best_gain, off);
        const int of = __shfl_down(best_feat, off);
        const int ob = __shfl_down(best_bin, off);
        const float owl = __shfl_down(best_wl, off);
        const bool take = (og > best_gain + 1e-12f) ||
                          (og > best_gain - 1e-12f && of >= 0 &&
                           (best_feat < 0 || of < best_feat ||
                            (of == best_feat && ob < best_bin)));
        if (take) {
            best_gain = og; best_feat = of; best_bin = ob; best_wl = owl;
        }
    }
    if (lane == 0) {
        s_best_gain[wid] = best_gain;
        s_best_feat[wid] = best_feat;
        s_best_bin[wid] = best_bin;
        s_best_wl[wid] = best_wl;
    }
    __syncthreads();
    if (tid == 0) {
        for (int w2 = 1; w2 < (int)(blockDim.x / WAVE); ++w2) {
            const float og = s_best_gain[w2];
            const int of = s_best_feat[w2];
            const bool take = (og > s_best_gain[0] + 1e-12f) ||
                              (og > s_best_gain[0] - 1e-12f && of >= 0 &&
                               (s_best_feat[0] < 0 ||
                                of < s_best_feat[0] ||
                                (of == s_best_feat[0] &&
                                 s_best_bin[w2] < s_best_bin[0])));
            if (take) {
                s_best_gain[0] = og;
                s_best_feat[0] = of;
                s_best_bin[0] = s_best_bin[w2];
                s_best_wl[0] = s_best_wl[w2];
            }
        }
        out_feat[node] = s_best_gain[0] > 0.f ? s_best_feat[0] : -1;
        out_bin[node] = s_best_bin[0];
        out_wl[node] = s_best_wl[0];
        out_gain[node] = s_best_gain[0];
        out_imp[node] = imp_p;
    }
    for (int c = tid; c < S; c += blockDim.x)
        out_stats[(long long)node * S + c] = s_parent[c];
    __syncthreads();
    // left stats of the winning split (prefix sum over its bins)
    const int wf = (s_best_gain[0] > 0.f) ? s_best_feat[0] : -1;
    for (int c = tid; c < S; c += blockDim.x) {
        float acc = 0.f;
        if (wf >= 0) {
            const float* Hw = H + (long long)wf * nbins * S;
            const int wb = s_best_bin[0];
            for (int b = 0; b <= wb; ++b) acc += Hw[b * S + c];
        }
        out_lstats[(long long)node * S + c] = acc;
    }
}

// ---------------------------------------------------------------------- //
// K3: per-chunk left counts (split nodes only)
// chunks here carry {part_slot, tree_slot, row_start, row_count}
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_part_count(
    const unsigned char* __restrict__ codes, const int* __restrict__ sample_idx,
    const int* __restrict__ chunks, const int* __restrict__ split_feat,
    const int* __restrict__ split_bin, long long n, int fp,
    int* __restrict__ out_counts) {
    const int chunk = blockIdx.x;
    const int part_slot = chunks[chunk * 4 + 0];
    const int tree_slot = chunks[chunk * 4 + 1];
    const int row_start = chunks[chunk * 4 + 2];
    const int row_count = chunks[chunk * 4 + 3];
    const int jf = split_feat[part_slot];
    const int b = split_bin[part_slot];
    const int* si = sample_idx + (long long)tree_slot * n + row_start;
    __shared__ int s_cnt;
    if (threadIdx.x == 0) s_cnt = 0;
    __syncthreads();
    int cnt = 0;
    for (int r = threadIdx.x; r < row_count; r += blockDim.x)
        if (codes[(long long)si[r] * fp + jf] <= b) ++cnt;
    atomicAdd(&s_cnt, cnt);
    __syncthreads();
    if (threadIdx.x == 0) out_counts[chunk] = s_cnt;
}

// ---------------------------------------------------------------------- //
// K4: stable partition scatter
// left_base/right_base: absolute output offsets (into the tree's row of
// sample_idx_out) for this chunk's first left / right element.
// ---------------------------------------------------------------------- //
extern "C" __global__ __launch_bounds__(256) void k_part_scatter(
    const unsigned char* __restrict__ codes,
    const int* __restrict__ sample_idx_in, const int* __restrict__ chunks,
    const int* __restrict__ split_feat, const int* __restrict__ split_bin,
    const int* __restrict__ left_base, const int* __restrict__ right_base,
    long long n, int fp, int* __restrict__ sample_idx_out) {
    const int chunk = blockIdx.x;
    const int part_slot = chunks[chunk * 4 + 0];
    const int tree_slot = chunks[chunk * 4 + 1];
    const int row_start = chunks[chunk * 4 + 2];
    const int row_count = chunks[chunk * 4 + 3];
    const int jf = split_feat[part_slot];
    const int b = split_bin[part_slot];
    const int* si = sample_idx_in + (long long)tree_slot * n + row_start;
    int* so = sample_idx_out + (long long)tree_slot * n;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;
    __shared__ int s_wave_tot[256 / WAVE];
    __shared__ int s_run[2];  // running left / right within this chunk
    if (tid == 0) { s_run[0] = 0; s_run[1] = 0; }
    __syncthreads();

    for (int tile = 0; tile < row_count; tile += blockDim.x) {
        const int r = tile + tid;
        const bool valid = r < row_count;
        int i = 0;
        bool p = false;
        if (valid) {
            i = si[r];
            p = codes[(long long)i * fp + jf] <= b;
        }
        const unsigned long long mask = __ballot(p);
        const int before =
            __popcll(mask & ((1ull << lane) - 1ull));
        if (lane == WAVE - 1) s_wave_tot[wid] = before + (p ? 1 : 0);
        __syncthreads();
        int wave_off = 0;
        for (int w2 = 0; w2 < wid; ++w2) wave_off += s_wave_tot[w2];
        int tile_l = 0;
        for (int w2 = 0; w2 < (int)(blockDim.x / WAVE); ++w2)
            tile_l += s_wave_tot[w2];
        const int lpre = before + wave_off;
        if (valid) {
            if (p)
                so[left_base[chunk] + s_run[0] + lpre] = i;
            else
                so[right_base[chunk] + s_run[1] + (tid - lpre)] = i;
        }
        __syncthreads();
        if (tid == 0) {
            const int vcnt = min((int)blockDim.x, row_count - tile);
            s_run[0] += tile_l;
            s_run[1] += vcnt - tile_l;
        }
        __syncthreads();
    }
}

// ---------------------------------------------------------------------- //
// C launchers
// ---------------------------------------------------------------------- //
extern "C" hipError_t skdist_tree_hist(
    const void* codes, const void* y_int, const void* y_f,
    const void* weights, const void* sample_idx, const void* chunks,
    void* hist, long long n, int f, int fp, int nbins, int S, int is_cls,
    int fg, int n_chunks, hipStream_t stream) {
    const int n_groups = (f + fg - 1) / fg;
    const size_t lds = (size_t)fg * nbins * S * sizeof(float);
    hipLaunchKernelGGL(k_tree_hist, dim3(n_chunks, n_groups), dim3(256),
                       lds, stream, (const unsigned char*)codes,
                       (const int*)y_int, (const float*)y_f,
                       (const unsigned char*)weights,
                       (const int*)sample_idx, (const int*)chunks,
                       (float*)hist, n, f, fp, nbins, S, is_cls, fg);
    return hipGetLastError();
}

extern "C" hipError_t skdist_tree_split(
    const void* hist, const void* node_seed, int n_frontier, int f,
    int nbins, int S, int is_cls, int crit, int m_features, int extra_mode,
    float min_leaf_w, void* out_feat, void* out_bin, void* out_wl,
    void* out_gain, void* out_imp, void* out_stats, void* out_lstats,
    hipStream_t stream) {
    // per-node parallelism is one thread per feature: size the block to
    // f so low-f datasets don't idle 3/4 of each workgroup
    const int threads = f >= 192 ? 256 : (f >= 96 ? 128 : 64);
    hipLaunchKernelGGL(k_tree_split, dim3(n_frontier), dim3(threads), 0,
                       stream, (const float*)hist,
                       (const unsigned*)node_seed, f, nbins, S, is_cls,
                       crit, m_features, extra_mode, min_leaf_w,
                       (int*)out_feat, (int*)out_bin, (float*)out_wl,
                       (float*)out_gain, (float*)out_imp,
                       (float*)out_stats, (float*)out_lstats);
    return hipGetLastError();
}

extern "C" hipError_t skdist_part_count(
    const void* codes, const void* sample_idx, const void* chunks,
    const void* split_feat, const void* split_bin, long long n, int fp,
    int n_chunks, void* out_counts, hipStream_t stream) {
    hipLaunchKernelGGL(k_part_count, dim3(n_chunks), dim3(256), 0, stream,
                       (const unsigned char*)codes, (const int*)sample_idx,
                       (const int*)chunks, (const int*)split_feat,
                       (const int*)split_bin, n, fp, (int*)out_counts);
    return hipGetLastError();
}

extern "C" hipError_t skdist_part_scatter(
    const void* codes, const void* sample_idx_in, const void* chunks,
    const void* split_feat, const void* split_bin, const void* left_base,
    const void* right_base, long long n, int fp, int n_chunks,
    void* sample_idx_out, hipStream_t stream) {
    hipLaunchKernelGGL(k_part_scatter, dim3(n_chunks), dim3(256), 0,
                       stream, (const unsigned char*)codes,
                       (const int*)sample_idx_in, (const int*)chunks,
                       (const int*)split_feat, (const int*)split_bin,
                       (const int*)left_base, (const int*)right_base, n,
                       fp, (int*)sample_idx_out);
    return hipGetLastError();
}
