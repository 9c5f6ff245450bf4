// Python bindings for the turboprune_amd gfx950 kernels.
#include <torch/extension.h>

namespace turboprune {
at::Tensor mask_apply(const at::Tensor&, const at::Tensor&, at::ScalarType);
void mask_from_threshold_(at::Tensor, const at::Tensor&, double);
at::Tensor masked_abs_score(const at::Tensor&, const at::Tensor&,
                            const at::Tensor&);
void bernoulli_mask_(at::Tensor, double, int64_t);
at::Tensor sign_abs_(at::Tensor);
at::Tensor colsum_bf16(const at::Tensor&);
at::Tensor gemm_tn_bf16(const at::Tensor&, const at::Tensor&);
void mul_sign_(at::Tensor, const at::Tensor&);
void sgd_step_(at::Tensor, const at::Tensor&, at::Tensor, const at::Tensor&,
               at::Tensor, double, double, double);
double kth_smallest(const at::Tensor&, int64_t);
std::tuple<at::Tensor, at::Tensor> ce_fwd(const at::Tensor&,
                                          const at::Tensor&);
at::Tensor ce_bwd(const at::Tensor&, const at::Tensor&, const at::Tensor&,
                  const at::Tensor&);
at::Tensor accuracy_count(const at::Tensor&, const at::Tensor&);
at::Tensor normalize_u8(const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, const at::Tensor&, at::ScalarType);
// masked MFMA GEMM (gemm_masked.hip)
bool masked_linear_available(const at::Tensor&, const at::Tensor&);
at::Tensor linear_fwd(const at::Tensor&, const at::Tensor&,
                      const c10::optional<at::Tensor>&);
std::tuple<at::Tensor, at::Tensor> linear_bwd(const at::Tensor&,
                                              const at::Tensor&,
                                              const at::Tensor&);
at::Tensor gemm_bf16(const at::Tensor&, const at::Tensor&, bool, bool);
at::Tensor gemm_bt_256(const at::Tensor&, const at::Tensor&,
                       const c10::optional<at::Tensor>&, bool);
void sgd_step_multi_(std::vector<at::Tensor>, std::vector<at::Tensor>,
                     std::vector<at::Tensor>, std::vector<at::Tensor>,
                     std::vector<at::Tensor>, double, double, double);
std::vector<at::Tensor> sgd_multi_plan(std::vector<at::Tensor>,
                                       std::vector<at::Tensor>,
                                       std::vector<at::Tensor>,
                                       std::vector<at::Tensor>,
                                       std::vector<at::Tensor>);
void sgd_step_multi_planned_(const at::Tensor&, const at::Tensor&, bool,
                             bool, bool, bool, bool, double, double,
                             double);
std::tuple<at::Tensor, at::Tensor> attn_fwd(
    const at::Tensor&, const at::Tensor&, const at::Tensor&, double);
std::tuple<at::Tensor, at::Tensor, at::Tensor> attn_bwd(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, double);
at::Tensor conv2d_implicit_gradin(const at::Tensor&, const at::Tensor&,
                                  int64_t, int64_t, int64_t, int64_t);
void schedulefree_step_(at::Tensor, at::Tensor, const at::Tensor&,
                        const at::Tensor&, at::Tensor, double,
                        double, double, double);
// fused NHWC batchnorm (batchnorm.hip)
at::Tensor transpose2d(const at::Tensor&);
at::Tensor conv2d_implicit_fwd(const at::Tensor&, const at::Tensor&,
                               const c10::optional<at::Tensor>&, int64_t,
                               int64_t);
at::Tensor conv2d_implicit_wrw(const at::Tensor&, const at::Tensor&,
                               int64_t, int64_t, int64_t, int64_t);
std::tuple<at::Tensor, at::Tensor, at::Tensor> ln_fwd(
    const at::Tensor&, const at::Tensor&, const at::Tensor&, double);
std::tuple<at::Tensor, at::Tensor, at::Tensor> ln_bwd(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&);
at::Tensor gelu_fwd(const at::Tensor&);
at::Tensor gelu_bwd(const at::Tensor&, const at::Tensor&);
at::Tensor crop_translate(const at::Tensor&, int64_t, const at::Tensor&);
at::Tensor random_resized_crop(const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, int64_t, at::ScalarType);
void cutout_(at::Tensor, const at::Tensor&, int64_t);
std::tuple<at::Tensor, at::Tensor> maxpool_fwd(const at::Tensor&, int, int,
                                               int, int, int, int);
at::Tensor maxpool_bwd(const at::Tensor&, const at::Tensor&, int, int, int,
                       int, int, int, int, int, int, int);
bool bn_fast_path_ok(const at::Tensor&);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_fwd(
    const at::Tensor&, const c10::optional<at::Tensor>&, const at::Tensor&,
    const at::Tensor&, at::Tensor, at::Tensor, bool, double, double, bool);
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_bwd(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, bool, bool);
}  // namespace turboprune

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "turboprune_amd CDNA4 (gfx950) kernels";
  m.def("mask_apply", &turboprune::mask_apply,
        "out = (mask!=0 ? weight : 0).to(dtype)");
  m.def("mask_from_threshold_", &turboprune::mask_from_threshold_,
        "mask = where(score <= thr, 0, 1) in-place");
  m.def("masked_abs_score", &turboprune::masked_abs_score,
        "|w*m| or |w*m*g|");
  m.def("bernoulli_mask_", &turboprune::bernoulli_mask_,
        "Philox Bernoulli(p) mask fill");
  m.def("gemm_tn_bf16", &turboprune::gemm_tn_bf16,
        "C[N,K] = A[M,N]^T B[M,K] — transpose-free deep-K TN GEMM "
        "(tr_b16 fragments)");
  m.def("colsum_bf16", &turboprune::colsum_bf16,
        "fp32 column sums of a bf16 matrix (bias gradients)");
  m.def("sign_abs_", &turboprune::sign_abs_,
        "SynFlow linearize: t = |t| in place, returns int8 signs (K10)");
  m.def("mul_sign_", &turboprune::mul_sign_,
        "SynFlow restore: t *= sign");
  m.def("sgd_step_", &turboprune::sgd_step_,
        "fused SGD momentum+wd step with mask-reapply cache rewrite");
  m.def("sgd_step_multi_", &turboprune::sgd_step_multi_,
        "multi-tensor fused SGD step (one launch over many params; "
        "uniform flags per call — opt-in via TURBOPRUNE_MULTI_SGD=1)");
  m.def("sgd_multi_plan", &turboprune::sgd_multi_plan,
        "build the device descriptor table + block map for the "
        "multi-tensor SGD step (cache keyed by operand data_ptrs)");
  m.def("sgd_step_multi_planned_", &turboprune::sgd_step_multi_planned_,
        "launch the multi-tensor SGD step from a prebuilt plan");
  m.def("schedulefree_step_", &turboprune::schedulefree_step_,
        "fused Schedule-Free SGD step (y/z update + masked-cache "
        "rewrite in one sweep)");
  m.def("attn_fwd", &turboprune::attn_fwd,
        "EXPERIMENTAL fused flash-style attention forward -> (O, lse), "
        "head_dim 64 (opt-in TURBOPRUNE_ATTN=native after validation)");
  m.def("attn_bwd", &turboprune::attn_bwd,
        "EXPERIMENTAL fused attention backward -> (dq, dk_f32, dv_f32)");
  m.def("kth_smallest", &turboprune::kth_smallest,
        "radix-select k-th smallest of a 1-D fp32 tensor");
  m.def("ce_fwd", &turboprune::ce_fwd, "fused CE forward -> (loss, lse)");
  m.def("ce_bwd", &turboprune::ce_bwd, "fused CE backward");
  m.def("accuracy_count", &turboprune::accuracy_count,
        "count(argmax(logits)==target)");
  m.def("normalize_u8", &turboprune::normalize_u8,
        "fused u8->normalized float/bf16 with optional flip");
  m.def("masked_linear_available", &turboprune::masked_linear_available);
  m.def("linear_fwd", &turboprune::linear_fwd, "MFMA GEMM y = x @ w^T + b");
  m.def("linear_bwd", &turboprune::linear_bwd,
        "MFMA GEMM grads (grad_x, grad_w)");
  m.def("gemm_bf16", &turboprune::gemm_bf16,
        "raw MFMA bf16 GEMM (testing entry)");
  m.def("gemm_bt_256", &turboprune::gemm_bt_256,
        "EXPERIMENTAL 256x256 8-phase MFMA GEMM (unwired; needs "
        "on-device race screen before promotion)");
  m.def("ln_fwd", &turboprune::ln_fwd, "fused LayerNorm fwd");
  m.def("ln_bwd", &turboprune::ln_bwd, "fused LayerNorm bwd");
  m.def("gelu_fwd", &turboprune::gelu_fwd, "fused exact GELU fwd");
  m.def("gelu_bwd", &turboprune::gelu_bwd, "fused exact GELU bwd");
  m.def("conv2d_implicit_fwd", &turboprune::conv2d_implicit_fwd,
        "implicit-GEMM NHWC conv forward (experimental)");
  m.def("conv2d_implicit_wrw", &turboprune::conv2d_implicit_wrw,
        "implicit-GEMM NHWC conv weight gradient (experimental)");
  m.def("conv2d_implicit_gradin", &turboprune::conv2d_implicit_gradin,
        "implicit-GEMM NHWC conv input gradient, any stride, fused "
        "dilated gather (experimental)");
  m.def("random_resized_crop", &turboprune::random_resized_crop,
        "fused bilinear RandomResizedCrop + flip + normalize");
  m.def("crop_translate", &turboprune::crop_translate,
        "CIFAR random-translate crop from reflect-padded batch");
  m.def("cutout_", &turboprune::cutout_, "CIFAR cutout fill in-place");
  m.def("maxpool_fwd", &turboprune::maxpool_fwd,
        "NHWC maxpool forward -> (y, argmax bytes)");
  m.def("maxpool_bwd", &turboprune::maxpool_bwd,
        "NHWC maxpool backward (gather)");
  m.def("transpose2d", &turboprune::transpose2d,
        "tiled LDS 2-D transpose (2-byte dtypes)");
  m.def("bn_fast_path_ok", &turboprune::bn_fast_path_ok);
  m.def("bn_fwd", &turboprune::bn_fwd,
        "fused NHWC batchnorm(+relu)(+residual) forward");
  m.def("bn_bwd", &turboprune::bn_bwd,
        "fused NHWC batchnorm(+relu)(+residual) backward");
}
