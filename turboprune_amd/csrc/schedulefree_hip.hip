#include "hip/hip_runtime.h"
// Fused Schedule-Free SGD step (Defazio et al. 2024; the y/z/x iterate
// scheme in optim/sgd.py::ScheduleFreeSGD). One pass per parameter:
//
//   g  = grad + wd * y            (weight-decay at the y iterate)
//   y += (z - y) * ckp1           (pull toward the Polyak average)
//   y += g * lr * (beta*(1-ckp1) - 1)
//   z -= lr * g
//   cache = mask ? y : 0          (bf16 masked compute weight)
//
// Mirrors the eager Python math exactly (g computed from the PRE-update
// y). Like sgd_fused.hip, the cache rewrite rides in the same sweep and
// the kernel writes through data_ptr WITHOUT bumping tensor versions —
// the _version staleness guard in mask_layers treats the cache as
// valid, which removes the per-forward mask_apply the eager path causes
// (every eager lerp_/add_ bumps the weight version and forces a cache
// refresh on the next forward of every masked layer).

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

template <typename GradT, typename CacheT, bool kMask, bool kCache>
__global__ void sf_step_kernel(float* __restrict__ y,
                               float* __restrict__ z,
                               const GradT* __restrict__ g,
                               const float* __restrict__ mask,
                               CacheT* __restrict__ cache, float lr,
                               float beta, float ckp1, float wd,
                               int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  float ycoef = lr * (beta * (1.f - ckp1) - 1.f);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float yi = y[i];
    float zi = z[i];
    float gi = to_float<GradT>(g[i]) + wd * yi;
    yi += (zi - yi) * ckp1;
    yi += gi * ycoef;
    zi -= lr * gi;
    y[i] = yi;
    z[i] = zi;
    if (kCache) {
      float cv = kMask ? (mask[i] != 0.f ? yi : 0.f) : yi;
      cache[i] = from_float<CacheT>(cv);
    }
  }
}

void schedulefree_step_(at::Tensor y, at::Tensor z, const at::Tensor& grad,
                        const at::Tensor& mask, at::Tensor cache,
                        double lr, double beta, double ckp1, double wd) {
  TORCH_CHECK(y.is_cuda() && y.scalar_type() == at::kFloat &&
              y.is_non_overlapping_and_dense(),
              "schedulefree_step_: dense fp32 GPU param required");
  TORCH_CHECK(z.strides().vec() == y.strides().vec() &&
              z.scalar_type() == at::kFloat);
  TORCH_CHECK(grad.strides().vec() == y.strides().vec(),
              "schedulefree_step_: grad layout must match param");
  bool has_mask = mask.defined() && mask.numel() > 0;
  bool has_cache = cache.defined() && cache.numel() > 0;
  if (has_mask)
    TORCH_CHECK(mask.strides().vec() == y.strides().vec());
  if (has_cache)
    TORCH_CHECK(cache.strides().vec() == y.strides().vec());
  int64_t n = y.numel();
  int grid = elementwise_grid(n);
  auto stream = at::hip::getCurrentHIPStream();
  const float* mp = has_mask ? mask.data_ptr<float>() : nullptr;

#define TP_SF(GT, CT, MASKF, CACHEF)                                      \
  hipLaunchKernelGGL((sf_step_kernel<GT, CT, MASKF, CACHEF>), dim3(grid), \
                     dim3(kBlock), 0, stream, y.data_ptr<float>(),        \
                     z.data_ptr<float>(),                                 \
                     reinterpret_cast<const GT*>(grad.data_ptr()), mp,    \
                     has_cache ? reinterpret_cast<CT*>(cache.data_ptr())  \
                               : nullptr,                                 \
                     (float)lr, (float)beta, (float)ckp1, (float)wd, n)
#define TP_SF_MC(GT, CT)                                                  \
  do {                                                                    \
    if (has_cache && has_mask) TP_SF(GT, CT, true, true);                 \
    else if (has_cache) TP_SF(GT, CT, false, true);                       \
    else TP_SF(GT, CT, false, false);                                     \
  } while (0)

  bool cache_bf16 =
      has_cache && cache.scalar_type() == at::kBFloat16;
  if (grad.scalar_type() == at::kFloat) {
    if (cache_bf16) TP_SF_MC(float, __hip_bfloat16);
    else TP_SF_MC(float, float);
  } else if (grad.scalar_type() == at::kBFloat16) {
    if (cache_bf16) TP_SF_MC(__hip_bfloat16, __hip_bfloat16);
    else TP_SF_MC(__hip_bfloat16, float);
  } else {
    TORCH_CHECK(false, "schedulefree_step_: unsupported grad dtype");
  }
#undef TP_SF_MC
#undef TP_SF
}

}  // namespace turboprune
