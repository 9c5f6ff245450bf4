// Fused SGD-momentum + mask-reapply step (SURVEY K5 — the north-star
// optimizer kernel).
//
// One pass over each parameter fuses:
//   d = grad + wd * w
//   buf = momentum * buf + d          (when momentum != 0)
//   w  -= lr * (momentum ? buf : d)   (raw weight: masked entries DO
//                                      receive wd/momentum updates, as in
//                                      the reference — nullified only in
//                                      the compute weight)
//   cache = mask != 0 ? w : 0         (bf16/f32 masked compute weight)
//
// vs. eager torch this replaces 4-6 kernel launches + a separate
// mask-multiply per layer per step with ONE memory-bound sweep:
// reads w, grad, buf, mask; writes w, buf, cache.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

template <typename GradT, typename CacheT, bool kMomentum, bool kMask,
          bool kCache>
__global__ void sgd_step_kernel(float* __restrict__ w,
                                const GradT* __restrict__ g,
                                float* __restrict__ buf,
                                const float* __restrict__ mask,
                                CacheT* __restrict__ cache, float lr,
                                float momentum, float wd, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float wi = w[i];
    float d = to_float<GradT>(g[i]) + wd * wi;
    if (kMomentum) {
      float b = momentum * buf[i] + d;
      buf[i] = b;
      d = b;
    }
    wi -= lr * d;
    w[i] = wi;
    if (kCache) {
      float cv = kMask ? (mask[i] != 0.f ? wi : 0.f) : wi;
      cache[i] = from_float<CacheT>(cv);
    }
  }
}

template <typename GradT, typename CacheT>
static void launch_sgd(at::Tensor& w, const at::Tensor& g, at::Tensor& buf,
                       const at::Tensor& mask, at::Tensor& cache, float lr,
                       float momentum, float wd, hipStream_t stream) {
  int64_t n = w.numel();
  int grid = elementwise_grid(n);
  bool has_m = momentum != 0.f && buf.defined() && buf.numel() > 0;
  bool has_mask = mask.defined() && mask.numel() > 0;
  bool has_cache = cache.defined() && cache.numel() > 0;
  float* bp = has_m ? buf.data_ptr<float>() : nullptr;
  const float* mp = has_mask ? mask.data_ptr<float>() : nullptr;
  CacheT* cp =
      has_cache ? reinterpret_cast<CacheT*>(cache.data_ptr()) : nullptr;

#define TP_SGD(MOM, MASKF, CACHEF)                                        \
  hipLaunchKernelGGL((sgd_step_kernel<GradT, CacheT, MOM, MASKF, CACHEF>), \
                     dim3(grid), dim3(kBlock), 0, stream,                 \
                     w.data_ptr<float>(),                                 \
                     reinterpret_cast<const GradT*>(g.data_ptr()), bp,    \
                     mp, cp, lr, momentum, wd, n)
  if (has_m) {
    if (has_cache && has_mask) TP_SGD(true, true, true);
    else if (has_cache) TP_SGD(true, false, true);
    else TP_SGD(true, false, false);
  } else {
    if (has_cache && has_mask) TP_SGD(false, true, true);
    else if (has_cache) TP_SGD(false, false, true);
    else TP_SGD(false, false, false);
  }
#undef TP_SGD
}

// all operand tensors iterate flat in the WEIGHT's storage order; any
// layout (e.g. channels_last) is fine as long as strides match — tensors
// in another dense layout are re-laid-out once.
static at::Tensor match_layout(const at::Tensor& t, const at::Tensor& like) {
  if (t.strides().vec() == like.strides().vec() &&
      t.is_non_overlapping_and_dense())
    return t;
  auto out = at::empty_strided(like.sizes(), like.strides(), t.options());
  out.copy_(t);
  return out;
}

void sgd_step_(at::Tensor w, const at::Tensor& grad, at::Tensor buf,
               const at::Tensor& mask, at::Tensor cache, double lr,
               double momentum, double wd) {
  TORCH_CHECK(w.is_cuda() && w.is_non_overlapping_and_dense() &&
              w.scalar_type() == at::kFloat,
              "sgd_step_: weight must be dense fp32 on GPU");
  auto g = match_layout(grad, w);
  TORCH_CHECK(g.numel() == w.numel());
  if (buf.defined() && buf.numel() > 0) {
    TORCH_CHECK(buf.scalar_type() == at::kFloat);
    TORCH_CHECK(buf.strides().vec() == w.strides().vec(),
                "sgd_step_: momentum buffer layout must match weight");
  }
  if (mask.defined() && mask.numel() > 0)
    TORCH_CHECK(mask.strides().vec() == w.strides().vec(),
                "sgd_step_: mask layout must match weight");
  if (cache.defined() && cache.numel() > 0)
    TORCH_CHECK(cache.strides().vec() == w.strides().vec(),
                "sgd_step_: cache layout must match weight");
  auto stream = at::hip::getCurrentHIPStream();

  auto cache_t = cache.defined() && cache.numel() > 0
                     ? cache.scalar_type()
                     : at::kFloat;
  if (g.scalar_type() == at::kFloat) {
    if (cache_t == at::kBFloat16)
      launch_sgd<float, __hip_bfloat16>(w, g, buf, mask, cache, lr,
                                        momentum, wd, stream);
    else
      launch_sgd<float, float>(w, g, buf, mask, cache, lr, momentum, wd,
                               stream);
  } else if (g.scalar_type() == at::kBFloat16) {
    if (cache_t == at::kBFloat16)
      launch_sgd<__hip_bfloat16, __hip_bfloat16>(w, g, buf, mask, cache, lr,
                                                 momentum, wd, stream);
    else
      launch_sgd<__hip_bfloat16, float>(w, g, buf, mask, cache, lr,
                                        momentum, wd, stream);
  } else {
    TORCH_CHECK(false, "sgd_step_: unsupported grad dtype");
  }
}

}  // namespace turboprune
