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
    const void* col_lr, const void* col_l2,
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
              int64_t start, int64_t m, int64_t loss_id, double lr_scale,
              double momentum, int64_t intercept_row) {
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
        col_lr.data_ptr(),
        col_l2.data_ptr(), start, m, a.n, a.n_pad, a.fa_store, (int)a.fa,
        (int)a.ncols_pad, (int)a.gt_stride, (int)a.splitk, (int)loss_id,
        (float)lr_scale, (float)momentum, (int)intercept_row, stream);
    TORCH_CHECK(err == hipSuccess, "skdist_sgd_step: ",
                hipGetErrorString(err));
}

// one whole epoch: the minibatch loop runs in C++ so the hot path costs
// one pybind crossing per epoch instead of one per step
void sgd_epoch(torch::Tensor Xs, torch::Tensor XsT, torch::Tensor GT,
               torch::Tensor W, torch::Tensor V, torch::Tensor WbfT,
               torch::Tensor partial, torch::Tensor y, torch::Tensor fold,
               torch::Tensor col_class, torch::Tensor col_fold,
               torch::Tensor col_class2,
               torch::Tensor col_lr, torch::Tensor col_l2,
               int64_t batch_size, int64_t loss_id, double lr_scale,
               double momentum, int64_t intercept_row) {
    auto a = check_args(Xs, XsT, GT, W, partial, y);
    TORCH_CHECK(batch_size % 128 == 0, "batch_size must be a mult of 128");
    const bool has_V = V.numel() > 0;
    auto stream = c10::hip::getCurrentHIPStream().stream();
    for (int64_t start = 0; start < a.n; start += batch_size) {
        const int64_t m = std::min(batch_size, a.n - start);
        hipError_t err = skdist_sgd_step(
            Xs.data_ptr(), XsT.data_ptr(), WbfT.data_ptr(), GT.data_ptr(),
            W.data_ptr(), has_V ? V.data_ptr() : nullptr, WbfT.data_ptr(),
            partial.data_ptr(), y.data_ptr(), fold.data_ptr(),
            col_class.data_ptr(), col_fold.data_ptr(),
            col_class2.data_ptr(), col_lr.data_ptr(),
            col_l2.data_ptr(), start, m, a.n, a.n_pad, a.fa_store,
            (int)a.fa, (int)a.ncols_pad, (int)a.gt_stride, (int)a.splitk,
            (int)loss_id, (float)lr_scale, (float)momentum,
            (int)intercept_row, stream);
        TORCH_CHECK(err == hipSuccess, "sgd_epoch: ", hipGetErrorString(err));
    }
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("sgd_step", &sgd_step, "fused batched SGD step (K1+K2+K3)");
    m.def("sgd_epoch", &sgd_epoch, "one epoch of fused SGD steps");
}
