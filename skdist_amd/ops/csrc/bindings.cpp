// Python bindings for the skdist_amd HIP kernels (built with hipcc against
// the ROCm torch in this image; no CUDA shims, no hipify).
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <hip/hip_runtime.h>

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
    hipStream_t stream);

namespace {

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on the GPU")
#define CHECK_CONT(t) TORCH_CHECK((t).is_contiguous(), #t " not contiguous")

struct StepArgs {
    int64_t n_pad, fa, fa_store, ncols_pad, gt_stride, splitk, n;
};

StepArgs check_args(torch::Tensor& Xs, torch::Tensor& XsT,
                    torch::Tensor& GT, torch::Tensor& W,
                    torch::Tensor& partial, torch::Tensor& y) {
    StepArgs a;
    a.n_pad = Xs.size(0);
    a.fa = Xs.size(1);
    a.fa_store = XsT.size(0);
    a.ncols_pad = W.size(1);
    a.gt_stride = GT.size(1);
    a.splitk = partial.size(0);
    a.n = y.size(0);
    TORCH_CHECK(a.fa % 32 == 0, "fa must be a multiple of 32");
    TORCH_CHECK(a.n_pad % 128 == 0, "n_pad must be a multiple of 128");
    TORCH_CHECK(a.fa_store % 128 == 0, "fa_store must be a multiple of 128");
    TORCH_CHECK(a.ncols_pad % 128 == 0, "ncols_pad must be a mult of 128");
    TORCH_CHECK(Xs.scalar_type() == torch::kBFloat16, "Xs must be bf16");
    TORCH_CHECK(XsT.size(1) == a.n_pad, "XsT shape mismatch");
    TORCH_CHECK(W.size(0) == a.fa, "W rows != fa");
    return a;
}

void sgd_step(torch::Tensor Xs, torch::Tensor XsT, torch::Tensor GT,
              torch::Tensor W, torch::Tensor V, torch::Tensor WbfT,
              torch::Tensor partial, torch::Tensor y, torch::Tensor fold,
              torch::Tensor col_class, torch::Tensor col_fold,
              torch::Tensor col_class2,
              torch::Tensor col_lr, torch::Tensor col_l2,
              torch::Tensor fmask, torch::Tensor row_w,
              int64_t start, int64_t m, int64_t loss_id, double lr_scale,
              double momentum, int64_t intercept_row, double inv_m) {
    for (auto* t : {&Xs, &XsT, &GT, &W, &WbfT, &partial, &y, &fold,
                    &col_class, &col_fold, &col_lr, &col_l2}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    auto a = check_args(Xs, XsT, GT, W, partial, y);
    const bool has_V = V.numel() > 0;
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_sgd_step(
        Xs.data_ptr(), XsT.data_ptr(), WbfT.data_ptr(), GT.data_ptr(),
        W.data_ptr(), has_V ? V.data_ptr() : nullptr, WbfT.data_ptr(),
        partial.data_ptr(), y.data_ptr(), fold.data_ptr(),
        col_class.data_ptr(), col_fold.data_ptr(), col_class2.data_ptr(),
        col_lr.data_ptr(), col_l2.data_ptr(),
        fmask.numel() ? fmask.data_ptr() : nullptr,
        row_w.numel() ? row_w.data_ptr() : nullptr, (float)inv_m,
        start, m, a.n, a.n_pad, a.fa_store, (int)a.fa,
        (int)a.ncols_pad, (int)a.gt_stride, (int)a.splitk, (int)loss_id,
        (float)lr_scale, (float)momentum, (int)intercept_row, stream);
    TORCH_CHECK(err == hipSuccess, "skdist_sgd_step: ",
                hipGetErrorString(err));
}

// Run `body(stream)` for `epochs` epochs with hipGraph capture: epoch 0
// executes eagerly on a dedicated non-blocking stream (the legacy
// default stream cannot capture), one epoch is then recorded into a
// graph and replayed for the rest — one launch per epoch instead of
// hundreds (the per-epoch launch sequence is identical by construction:
// fixed batches, fixed buffers).  Falls back to the eager loop on any
// capture error.
template <typename Body>
static void run_epochs_graphed(int64_t epochs, hipStream_t torch_stream,
                               Body&& body) {
    if (epochs <= 0) return;
    hipStream_t s2 = nullptr;
    if (hipStreamCreateWithFlags(&s2, hipStreamNonBlocking)
            != hipSuccess || s2 == nullptr) {
        for (int64_t e = 0; e < epochs; ++e) body(torch_stream);
        return;
    }
    hipEvent_t ev = nullptr;
    hipEventCreateWithFlags(&ev, hipEventDisableTiming);
    hipEventRecord(ev, torch_stream);
    hipStreamWaitEvent(s2, ev, 0);

    body(s2);  // epoch 0 executes (and warms any lazy module state)
    int64_t done = 1;
    if (epochs > 1) {
        hipGraph_t graph = nullptr;
        hipGraphExec_t exec = nullptr;
        bool ok = hipStreamBeginCapture(
                      s2, hipStreamCaptureModeThreadLocal) == hipSuccess;
        if (ok) {
            try {
                body(s2);  // recorded, not executed
            } catch (...) {
                // never leave the stream in capture mode
                hipStreamEndCapture(s2, &graph);
                if (graph) hipGraphDestroy(graph);
                hipEventDestroy(ev);
                hipStreamDestroy(s2);
                throw;
            }
            ok = hipStreamEndCapture(s2, &graph) == hipSuccess;
        }
        if (ok)
            ok = hipGraphInstantiate(&exec, graph, nullptr, nullptr, 0)
                 == hipSuccess;
        if (ok) {
            for (int64_t e = 1; e < epochs; ++e)
                hipGraphLaunch(exec, s2);
            done = epochs;
        }
        if (exec) hipGraphExecDestroy(exec);
        if (graph) hipGraphDestroy(graph);
    }
    for (int64_t e = done; e < epochs; ++e) body(s2);  // capture fallback

    hipEventRecord(ev, s2);
    hipStreamWaitEvent(torch_stream, ev, 0);
    hipEventDestroy(ev);
    hipStreamDestroy(s2);
}

// one whole epoch: the minibatch loop runs in C++ so the hot path costs
// one pybind crossing per epoch instead of one per step
void sgd_epoch(torch::Tensor Xs, torch::Tensor XsT, torch::Tensor GT,
               torch::Tensor W, torch::Tensor V, torch::Tensor WbfT,
               torch::Tensor partial, torch::Tensor y, torch::Tensor fold,
               torch::Tensor col_class, torch::Tensor col_fold,
               torch::Tensor col_class2,
               torch::Tensor col_lr, torch::Tensor col_l2,
               torch::Tensor fmask, torch::Tensor row_w,
               torch::Tensor inv_m_cpu,  // [n_batches] f32 on CPU
               int64_t batch_size, int64_t loss_id, double lr_scale,
               double momentum, int64_t intercept_row) {
    auto a = check_args(Xs, XsT, GT, W, partial, y);
    TORCH_CHECK(batch_size % 128 == 0, "batch_size must be a mult of 128");
    const bool has_V = V.numel() > 0;
    TORCH_CHECK(!inv_m_cpu.is_cuda(), "inv_m must be a CPU tensor");
    const float* inv_p = (const float*)inv_m_cpu.data_ptr();
    auto stream = c10::hip::getCurrentHIPStream().stream();
    int64_t bi = 0;
    for (int64_t start = 0; start < a.n; start += batch_size, ++bi) {
        const int64_t m = std::min(batch_size, a.n - start);
        hipError_t err = skdist_sgd_step(
            Xs.data_ptr(), XsT.data_ptr(), WbfT.data_ptr(), GT.data_ptr(),
            W.data_ptr(), has_V ? V.data_ptr() : nullptr, WbfT.data_ptr(),
            partial.data_ptr(), y.data_ptr(), fold.data_ptr(),
            col_class.data_ptr(), col_fold.data_ptr(),
            col_class2.data_ptr(), col_lr.data_ptr(), col_l2.data_ptr(),
            fmask.numel() ? fmask.data_ptr() : nullptr,
            row_w.numel() ? row_w.data_ptr() : nullptr, inv_p[bi],
            start, m, a.n, a.n_pad, a.fa_store,
            (int)a.fa, (int)a.ncols_pad, (int)a.gt_stride, (int)a.splitk,
            (int)loss_id, (float)lr_scale, (float)momentum,
            (int)intercept_row, stream);
        TORCH_CHECK(err == hipSuccess, "sgd_epoch: ", hipGetErrorString(err));
    }
}

// all epochs in one call; lr_scale is constant (lr_decay=0), so every
// epoch launches the identical sequence -> hipGraph replay
void sgd_solve(torch::Tensor Xs, torch::Tensor XsT, torch::Tensor GT,
               torch::Tensor W, torch::Tensor V, torch::Tensor WbfT,
               torch::Tensor partial, torch::Tensor y, torch::Tensor fold,
               torch::Tensor col_class, torch::Tensor col_fold,
               torch::Tensor col_class2,
               torch::Tensor col_lr, torch::Tensor col_l2,
               torch::Tensor fmask, torch::Tensor row_w,
               torch::Tensor inv_m_cpu, int64_t batch_size,
               int64_t loss_id, double momentum, int64_t intercept_row,
               int64_t epochs) {
    auto a = check_args(Xs, XsT, GT, W, partial, y);
    TORCH_CHECK(batch_size % 128 == 0, "batch_size must be a mult of 128");
    const bool has_V = V.numel() > 0;
    const float* inv_p = (const float*)inv_m_cpu.data_ptr();
    auto torch_stream = c10::hip::getCurrentHIPStream().stream();
    auto body = [&](hipStream_t st) {
        int64_t bi = 0;
        for (int64_t start = 0; start < a.n; start += batch_size, ++bi) {
            const int64_t m = std::min(batch_size, a.n - start);
            hipError_t err = skdist_sgd_step(
                Xs.data_ptr(), XsT.data_ptr(), WbfT.data_ptr(),
                GT.data_ptr(), W.data_ptr(),
                has_V ? V.data_ptr() : nullptr, WbfT.data_ptr(),
                partial.data_ptr(), y.data_ptr(), fold.data_ptr(),
                col_class.data_ptr(), col_fold.data_ptr(),
                col_class2.data_ptr(), col_lr.data_ptr(),
                col_l2.data_ptr(),
                fmask.numel() ? fmask.data_ptr() : nullptr,
                row_w.numel() ? row_w.data_ptr() : nullptr, inv_p[bi],
                start, m, a.n, a.n_pad, a.fa_store,
                (int)a.fa, (int)a.ncols_pad, (int)a.gt_stride,
                (int)a.splitk, (int)loss_id, 1.0f, (float)momentum,
                (int)intercept_row, st);
            TORCH_CHECK(err == hipSuccess, "sgd_solve: ",
                        hipGetErrorString(err));
        }
    };
    run_epochs_graphed(epochs, torch_stream, body);
}

}  // namespace

// ------------------------------------------------------------------ //
// sparse text-scale solver (sparse_sgd_kernels.hip)
// ------------------------------------------------------------------ //

extern "C" hipError_t skdist_sp_sgd_step(
    const void* crow, const void* cidx, const void* cval,
    void* W, void* Wb, void* s, void* h, void* hb, void* G, void* part,
    const void* y, const void* fold, const void* row_w,
    const void* col_class, const void* col_fold, const void* col_class2,
    const void* col_lr, const void* col_l2,
    const void* ufeat, const void* cptr, const void* ridx,
    const void* bval, long long uf,
    long long start, long long m, int cp, int loss_id,
    float inv_m, float lr_scale, hipStream_t stream);
extern "C" hipError_t skdist_sp_forward(
    const void* crow, const void* cidx, const void* cval,
    const void* W, const void* Wb, const void* s, const void* rows,
    void* Z, long long m, int cp, hipStream_t stream);
extern "C" hipError_t skdist_sp_renorm(
    void* W, const void* s, long long f, int cp, hipStream_t stream);

namespace {

static void check_cp(int64_t cp) {
    TORCH_CHECK(cp % 64 == 0, "cp must be a multiple of 64");
}

// one epoch of the sparse solver: the per-batch loop runs in C++.
// CSC arrays are the concatenation over batches; ub_ptr (CPU int64,
// [n_batches+1]) slices ufeat/cptr per batch, cptr values are GLOBAL
// offsets into ridx/bval.
void sp_sgd_epoch(torch::Tensor crow, torch::Tensor cidx,
                  torch::Tensor cval, torch::Tensor W, torch::Tensor Wb,
                  torch::Tensor s, torch::Tensor h, torch::Tensor hb,
                  torch::Tensor G, torch::Tensor part,
                  torch::Tensor y, torch::Tensor fold, torch::Tensor row_w,
                  torch::Tensor col_class, torch::Tensor col_fold,
                  torch::Tensor col_class2, torch::Tensor col_lr,
                  torch::Tensor col_l2, torch::Tensor ufeat,
                  torch::Tensor cptr, torch::Tensor ridx,
                  torch::Tensor bval, torch::Tensor ub_ptr_cpu,
                  torch::Tensor inv_m_cpu, int64_t batch_size,
                  int64_t loss_id, double lr_scale) {
    for (auto* t : {&crow, &cidx, &cval, &W, &Wb, &s, &G, &part, &y,
                    &fold, &col_class, &col_fold, &col_class2, &col_lr,
                    &col_l2, &ufeat, &cptr, &ridx, &bval}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    TORCH_CHECK(!ub_ptr_cpu.is_cuda() && !inv_m_cpu.is_cuda(),
                "ub_ptr/inv_m must be CPU tensors");
    TORCH_CHECK(crow.scalar_type() == torch::kInt64, "crow must be i64");
    TORCH_CHECK(cidx.scalar_type() == torch::kInt32, "cidx must be i32");
    TORCH_CHECK(G.scalar_type() == torch::kBFloat16, "G must be bf16");
    const int64_t cp = W.size(1);
    check_cp(cp);
    const int64_t n = crow.size(0) - 1;
    TORCH_CHECK(G.size(0) >= std::min<int64_t>(batch_size, n) &&
                G.size(1) == cp, "G buffer too small");
    const int64_t* ub = (const int64_t*)ub_ptr_cpu.data_ptr();
    const float* inv_p = (const float*)inv_m_cpu.data_ptr();
    auto stream = c10::hip::getCurrentHIPStream().stream();
    int64_t bi = 0;
    for (int64_t start = 0; start < n; start += batch_size, ++bi) {
        const int64_t m = std::min(batch_size, n - start);
        const int64_t ub0 = ub[bi], uf = ub[bi + 1] - ub[bi];
        hipError_t err = skdist_sp_sgd_step(
            crow.data_ptr(), cidx.data_ptr(), cval.data_ptr(),
            W.data_ptr(), Wb.data_ptr(), s.data_ptr(),
            h.numel() ? h.data_ptr() : nullptr,
            hb.numel() ? hb.data_ptr() : nullptr, G.data_ptr(),
            part.data_ptr(), y.data_ptr(), fold.data_ptr(),
            row_w.numel() ? row_w.data_ptr() : nullptr,
            col_class.data_ptr(), col_fold.data_ptr(),
            col_class2.data_ptr(), col_lr.data_ptr(), col_l2.data_ptr(),
            (const int*)ufeat.data_ptr() + ub0,
            (const long long*)cptr.data_ptr() + ub0,
            ridx.data_ptr(), bval.data_ptr(), uf,
            start, m, (int)cp, (int)loss_id, inv_p[bi],
            (float)lr_scale, stream);
        TORCH_CHECK(err == hipSuccess, "sp_sgd_epoch: ",
                    hipGetErrorString(err));
    }
}

// all epochs of the sparse solver in one call (constant lr_scale);
// identical per-epoch launch sequence -> hipGraph replay
void sp_sgd_solve(torch::Tensor crow, torch::Tensor cidx,
                  torch::Tensor cval, torch::Tensor W, torch::Tensor Wb,
                  torch::Tensor s, torch::Tensor h, torch::Tensor hb,
                  torch::Tensor G, torch::Tensor part,
                  torch::Tensor y, torch::Tensor fold, torch::Tensor row_w,
                  torch::Tensor col_class, torch::Tensor col_fold,
                  torch::Tensor col_class2, torch::Tensor col_lr,
                  torch::Tensor col_l2, torch::Tensor ufeat,
                  torch::Tensor cptr, torch::Tensor ridx,
                  torch::Tensor bval, torch::Tensor ub_ptr_cpu,
                  torch::Tensor inv_m_cpu, int64_t batch_size,
                  int64_t loss_id, int64_t epochs) {
    const int64_t cp = W.size(1);
    check_cp(cp);
    const int64_t n = crow.size(0) - 1;
    const int64_t* ub = (const int64_t*)ub_ptr_cpu.data_ptr();
    const float* inv_p = (const float*)inv_m_cpu.data_ptr();
    auto torch_stream = c10::hip::getCurrentHIPStream().stream();
    auto body = [&](hipStream_t st) {
        int64_t bi = 0;
        for (int64_t start = 0; start < n; start += batch_size, ++bi) {
            const int64_t m = std::min(batch_size, n - start);
            const int64_t ub0 = ub[bi], uf = ub[bi + 1] - ub[bi];
            hipError_t err = skdist_sp_sgd_step(
                crow.data_ptr(), cidx.data_ptr(), cval.data_ptr(),
                W.data_ptr(), Wb.data_ptr(), s.data_ptr(),
                h.numel() ? h.data_ptr() : nullptr,
                hb.numel() ? hb.data_ptr() : nullptr, G.data_ptr(),
                part.data_ptr(), y.data_ptr(), fold.data_ptr(),
                row_w.numel() ? row_w.data_ptr() : nullptr,
                col_class.data_ptr(), col_fold.data_ptr(),
                col_class2.data_ptr(), col_lr.data_ptr(),
                col_l2.data_ptr(),
                (const int*)ufeat.data_ptr() + ub0,
                (const long long*)cptr.data_ptr() + ub0,
                ridx.data_ptr(), bval.data_ptr(), uf,
                start, m, (int)cp, (int)loss_id, inv_p[bi], 1.0f, st);
            TORCH_CHECK(err == hipSuccess, "sp_sgd_solve: ",
                        hipGetErrorString(err));
        }
    };
    run_epochs_graphed(epochs, torch_stream, body);
}

void sp_forward(torch::Tensor crow, torch::Tensor cidx,
                torch::Tensor cval, torch::Tensor W, torch::Tensor Wb,
                torch::Tensor s, torch::Tensor rows, torch::Tensor Z) {
    for (auto* t : {&crow, &cidx, &cval, &W, &Wb, &s, &rows, &Z}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    TORCH_CHECK(rows.scalar_type() == torch::kInt64, "rows must be i64");
    const int64_t cp = W.size(1);
    check_cp(cp);
    TORCH_CHECK(Z.size(0) == rows.size(0) && Z.size(1) == cp,
                "Z shape mismatch");
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_sp_forward(
        crow.data_ptr(), cidx.data_ptr(), cval.data_ptr(), W.data_ptr(),
        Wb.data_ptr(), s.data_ptr(), rows.data_ptr(), Z.data_ptr(),
        rows.size(0), (int)cp, stream);
    TORCH_CHECK(err == hipSuccess, "sp_forward: ", hipGetErrorString(err));
}

void sp_renorm(torch::Tensor W, torch::Tensor s) {
    CHECK_DEV(W);
    CHECK_CONT(W);
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_sp_renorm(W.data_ptr(), s.data_ptr(),
                                      W.size(0), (int)W.size(1), stream);
    TORCH_CHECK(err == hipSuccess, "sp_renorm: ", hipGetErrorString(err));
}

}  // namespace

// ------------------------------------------------------------------ //
// histogram tree builder (tree_kernels.hip)
// ------------------------------------------------------------------ //

extern "C" hipError_t skdist_tree_hist(
    const void* codes, const void* y_int, const void* y_f,
    const void* weights, const void* sample_idx, const void* chunks,
    void* hist, long long n, int f, int fp, int nbins, int S, int is_cls,
    int fg, int n_chunks, hipStream_t stream);
extern "C" hipError_t skdist_tree_split(
    const void* hist, const void* node_seed, int n_frontier, int f,
    int nbins, int S, int is_cls, int crit, int m_features, int extra_mode,
    float min_leaf_w, void* out_feat, void* out_bin, void* out_wl,
    void* out_gain, void* out_imp, void* out_stats, void* out_lstats,
    hipStream_t stream);
extern "C" hipError_t skdist_part_count(
    const void* codes, const void* sample_idx, const void* chunks,
    const void* split_feat, const void* split_bin, long long n, int fp,
    int n_chunks, void* out_counts, hipStream_t stream);
extern "C" hipError_t skdist_part_scatter(
    const void* codes, const void* sample_idx_in, const void* chunks,
    const void* split_feat, const void* split_bin, const void* left_base,
    const void* right_base, long long n, int fp, int n_chunks,
    void* sample_idx_out, hipStream_t stream);
extern "C" hipError_t skdist_forest_predict(
    const void* X, const void* feat, const void* thr, const void* left,
    const void* right, const void* roots, const void* values, void* out,
    long long rows, int f, int n_trees, int vs, hipStream_t stream);
extern "C" hipError_t skdist_forest_apply(
    const void* X, const void* feat, const void* thr, const void* left,
    const void* right, const void* roots, void* out_leaf, long long rows,
    int f, int n_trees, hipStream_t stream);
extern "C" hipError_t skdist_score(
    const void* Xf, const void* WbfT, const void* yf, void* out,
    long long m_pad, int fa, int ncols_pad, int mode,
    hipStream_t stream);
extern "C" hipError_t skdist_standardize(
    const void* X, const void* mean, const void* inv_std, void* out,
    long long n, int f, int fa, hipStream_t stream);
extern "C" hipError_t skdist_hash_vectorize(
    const void* bytes, const void* doc_off, long long n_docs, int mode,
    int min_n, int max_n, int n_features, int alt_sign, void* out_keys,
    void* out_vals, void* ctr, long long cap, hipStream_t stream);

namespace {

void tree_hist(torch::Tensor codes, torch::Tensor y_int, torch::Tensor y_f,
               torch::Tensor weights, torch::Tensor sample_idx,
               torch::Tensor chunks, torch::Tensor hist, int64_t n,
               int64_t f, int64_t nbins, int64_t S, int64_t is_cls,
               int64_t fg) {
    const int64_t fp = codes.size(1);
    for (auto* t : {&codes, &weights, &sample_idx, &chunks, &hist}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    TORCH_CHECK(codes.scalar_type() == torch::kUInt8, "codes must be u8");
    TORCH_CHECK(chunks.size(1) == 4, "chunks must be [n_chunks, 4]");
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_tree_hist(
        codes.data_ptr(), is_cls ? y_int.data_ptr() : nullptr,
        is_cls ? nullptr : y_f.data_ptr(), weights.data_ptr(),
        sample_idx.data_ptr(), chunks.data_ptr(), hist.data_ptr(), n,
        (int)f, (int)fp, (int)nbins, (int)S, (int)is_cls, (int)fg,
        (int)chunks.size(0), stream);
    TORCH_CHECK(err == hipSuccess, "tree_hist: ", hipGetErrorString(err));
}

void tree_split(torch::Tensor hist, torch::Tensor node_seed, int64_t f,
                int64_t nbins, int64_t S, int64_t is_cls, int64_t crit,
                int64_t m_features, int64_t extra_mode, double min_leaf_w,
                torch::Tensor out_feat, torch::Tensor out_bin,
                torch::Tensor out_wl, torch::Tensor out_gain,
                torch::Tensor out_imp, torch::Tensor out_stats,
                torch::Tensor out_lstats) {
    CHECK_DEV(hist);
    CHECK_CONT(hist);
    TORCH_CHECK(S <= 32, "tree_split supports <= 32 stats");
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_tree_split(
        hist.data_ptr(), node_seed.data_ptr(), (int)node_seed.size(0),
        (int)f, (int)nbins, (int)S, (int)is_cls, (int)crit,
        (int)m_features, (int)extra_mode, (float)min_leaf_w,
        out_feat.data_ptr(), out_bin.data_ptr(), out_wl.data_ptr(),
        out_gain.data_ptr(), out_imp.data_ptr(), out_stats.data_ptr(),
        out_lstats.data_ptr(), stream);
    TORCH_CHECK(err == hipSuccess, "tree_split: ", hipGetErrorString(err));
}

void part_count(torch::Tensor codes, torch::Tensor sample_idx,
                torch::Tensor chunks, torch::Tensor split_feat,
                torch::Tensor split_bin, int64_t n,
                torch::Tensor out_counts) {
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_part_count(
        codes.data_ptr(), sample_idx.data_ptr(), chunks.data_ptr(),
        split_feat.data_ptr(), split_bin.data_ptr(), n,
        (int)codes.size(1), (int)chunks.size(0), out_counts.data_ptr(),
        stream);
    TORCH_CHECK(err == hipSuccess, "part_count: ", hipGetErrorString(err));
}

void part_scatter(torch::Tensor codes, torch::Tensor sample_idx_in,
                  torch::Tensor chunks, torch::Tensor split_feat,
                  torch::Tensor split_bin, torch::Tensor left_base,
                  torch::Tensor right_base, int64_t n,
                  torch::Tensor sample_idx_out) {
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_part_scatter(
        codes.data_ptr(), sample_idx_in.data_ptr(), chunks.data_ptr(),
        split_feat.data_ptr(), split_bin.data_ptr(), left_base.data_ptr(),
        right_base.data_ptr(), n, (int)codes.size(1),
        (int)chunks.size(0), sample_idx_out.data_ptr(), stream);
    TORCH_CHECK(err == hipSuccess, "part_scatter: ",
                hipGetErrorString(err));
}

void forest_predict(torch::Tensor X, torch::Tensor feat, torch::Tensor thr,
                    torch::Tensor left, torch::Tensor right,
                    torch::Tensor roots, torch::Tensor values,
                    torch::Tensor out) {
    for (auto* t : {&X, &thr, &values, &out}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_forest_predict(
        X.data_ptr(), feat.data_ptr(), thr.data_ptr(), left.data_ptr(),
        right.data_ptr(), roots.data_ptr(), values.data_ptr(),
        out.data_ptr(), X.size(0), (int)X.size(1), (int)roots.size(0),
        (int)values.size(1), stream);
    TORCH_CHECK(err == hipSuccess, "forest_predict: ",
                hipGetErrorString(err));
}

void forest_apply(torch::Tensor X, torch::Tensor feat, torch::Tensor thr,
                  torch::Tensor left, torch::Tensor right,
                  torch::Tensor roots, torch::Tensor out_leaf) {
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_forest_apply(
        X.data_ptr(), feat.data_ptr(), thr.data_ptr(), left.data_ptr(),
        right.data_ptr(), roots.data_ptr(), out_leaf.data_ptr(), X.size(0),
        (int)X.size(1), (int)roots.size(0), stream);
    TORCH_CHECK(err == hipSuccess, "forest_apply: ",
                hipGetErrorString(err));
}

void score_fold(torch::Tensor Xf, torch::Tensor WbfT, torch::Tensor yf,
                torch::Tensor out, int64_t mode) {
    for (auto* t : {&Xf, &WbfT, &yf, &out}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    TORCH_CHECK(Xf.scalar_type() == torch::kBFloat16, "Xf must be bf16");
    TORCH_CHECK(Xf.size(0) % 128 == 0, "m_pad must be a mult of 128");
    TORCH_CHECK(WbfT.size(0) % 128 == 0, "ncols_pad mult of 128");
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_score(
        Xf.data_ptr(), WbfT.data_ptr(), yf.data_ptr(), out.data_ptr(),
        Xf.size(0), (int)Xf.size(1), (int)WbfT.size(0), (int)mode,
        stream);
    TORCH_CHECK(err == hipSuccess, "score_fold: ", hipGetErrorString(err));
}

void standardize(torch::Tensor X, torch::Tensor mean,
                 torch::Tensor inv_std, torch::Tensor out, int64_t f) {
    for (auto* t : {&X, &mean, &inv_std, &out}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    TORCH_CHECK(out.scalar_type() == torch::kBFloat16, "out must be bf16");
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_standardize(
        X.data_ptr(), mean.data_ptr(), inv_std.data_ptr(), out.data_ptr(),
        out.size(0), (int)f, (int)out.size(1), stream);
    TORCH_CHECK(err == hipSuccess, "standardize: ",
                hipGetErrorString(err));
}

void hash_vectorize(torch::Tensor bytes, torch::Tensor doc_off,
                    int64_t mode, int64_t min_n, int64_t max_n,
                    int64_t n_features, int64_t alt_sign,
                    torch::Tensor out_keys, torch::Tensor out_vals,
                    torch::Tensor ctr) {
    for (auto* t : {&bytes, &doc_off, &out_keys, &out_vals, &ctr}) {
        CHECK_DEV(*t);
        CHECK_CONT(*t);
    }
    TORCH_CHECK(bytes.scalar_type() == torch::kUInt8, "bytes must be u8");
    auto stream = c10::hip::getCurrentHIPStream().stream();
    hipError_t err = skdist_hash_vectorize(
        bytes.data_ptr(), doc_off.data_ptr(), doc_off.size(0) - 1,
        (int)mode, (int)min_n, (int)max_n, (int)n_features, (int)alt_sign,
        out_keys.data_ptr(), out_vals.data_ptr(), ctr.data_ptr(),
        out_keys.size(0), stream);
    TORCH_CHECK(err == hipSuccess, "hash_vectorize: ",
                hipGetErrorString(err));
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("sgd_step", &sgd_step, "fused batched SGD step (K1+K2+K3)");
    m.def("sgd_epoch", &sgd_epoch, "one epoch of fused SGD steps");
    m.def("sp_sgd_epoch", &sp_sgd_epoch,
          "one epoch of the sparse text-scale solver");
    m.def("sgd_solve", &sgd_solve,
          "all epochs of the dense solver (hipGraph-replayed)");
    m.def("sp_sgd_solve", &sp_sgd_solve,
          "all epochs of the sparse solver (hipGraph-replayed)");
    m.def("sp_forward", &sp_forward, "sparse decision values (scoring)");
    m.def("sp_renorm", &sp_renorm, "fold the lazy-L2 scale into W");
    m.def("score_fold", &score_fold,
          "fused fold-scoring GEMM + per-column stats");
    m.def("standardize", &standardize,
          "fused standardize + augment + bf16 cast");
    m.def("hash_vectorize", &hash_vectorize,
          "tokenize + murmur3 feature hashing -> COO pairs");
    m.def("tree_hist", &tree_hist, "per-(node,feature,bin) stats");
    m.def("tree_split", &tree_split, "best split per frontier node");
    m.def("part_count", &part_count, "partition phase A: left counts");
    m.def("part_scatter", &part_scatter, "partition phase B: scatter");
    m.def("forest_predict", &forest_predict, "batched forest inference");
    m.def("forest_apply", &forest_apply, "leaf ids per (row, tree)");
}
