#include "hip/hip_runtime.h"
// Fused NHWC BatchNorm2d (+ReLU) (+residual add) forward/backward.
//
// Motivation (profiles/r01_bench_resnet50_1gpu_baseline.md): MIOpen's BN
// (6 kernels per layer per step) + eager ReLU + eager residual add are
// ~60% of the ResNet50 step's GPU time. These kernels replace them:
//
//   fwd (train): [reduce] per-channel sum/sumsq  ->  [finalize] stats +
//                running-stat update + scale/shift  ->  [apply]
//                y = relu(x*scale + shift (+ res)) — ONE elementwise pass
//   fwd (eval):  [apply] with running stats
//   bwd:         [reduce] per-channel sum(dy_eff), sum(dy_eff * xhat)
//                with dy_eff = dy * (y > 0) when ReLU was fused  ->
//                [finalize] dgamma/dbeta + dx coefficients  ->
//                [apply] dx (+ dres) — ONE elementwise pass
//
// Layout: NHWC (channels_last) — adjacent lanes read adjacent channels,
// fully coalesced; rows (N*H*W) are grid-strided. Stats/params fp32,
// activations bf16 or fp32.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

// ---------------- forward reduce: per-channel sum & sumsq ----------------
template <typename T>
__global__ void bn_reduce_kernel(const T* __restrict__ x, int64_t rows,
                                 int C, float* __restrict__ sum,
                                 float* __restrict__ sumsq) {
  int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float s = 0.f, ss = 0.f;
  for (int64_t r = blockIdx.x; r < rows; r += gridDim.x) {
    float v = to_float<T>(x[r * C + c]);
    s += v;
    ss += v * v;
  }
  atomicAdd(&sum[c], s);
  atomicAdd(&sumsq[c], ss);
}

// ---------------- forward finalize (one small launch) --------------------
__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float* __restrict__ save_mean,
                                   float* __restrict__ save_rstd,
                                   float* __restrict__ scale,
                                   float* __restrict__ shift, int C,
                                   float inv_n, float unbiased_factor,
                                   float momentum, float eps,
                                   bool update_running) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = sum[c] * inv_n;
  float var = fmaxf(sumsq[c] * inv_n - mean * mean, 0.f);
  float rstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_rstd[c] = rstd;
  if (update_running) {
    running_mean[c] += momentum * (mean - running_mean[c]);
    running_var[c] += momentum * (var * unbiased_factor - running_var[c]);
  }
  float sc = gamma[c] * rstd;
  scale[c] = sc;
  shift[c] = beta[c] - mean * sc;
}

__global__ void bn_eval_coeffs_kernel(const float* __restrict__ running_mean,
                                      const float* __restrict__ running_var,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ beta,
                                      float* __restrict__ scale,
                                      float* __restrict__ shift, int C,
                                      float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float rstd = rsqrtf(running_var[c] + eps);
  float sc = gamma[c] * rstd;
  scale[c] = sc;
  shift[c] = beta[c] - running_mean[c] * sc;
}

// ---------------- forward apply: y = relu?(x*scale+shift (+res)) ---------
template <typename T, bool RELU, bool RES>
__global__ void bn_apply_kernel(const T* __restrict__ x,
                                const T* __restrict__ res,
                                T* __restrict__ y,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                int64_t rows, int C) {
  int64_t total = rows * C;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    float v = to_float<T>(x[i]) * scale[c] + shift[c];
    if (RES) v += to_float<T>(res[i]);
    if (RELU) v = fmaxf(v, 0.f);
    y[i] = from_float<T>(v);
  }
}

// ---------------- backward reduce ----------------------------------------
template <typename T, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ x,
                                     const T* __restrict__ y,
                                     const T* __restrict__ dy, int64_t rows,
                                     int C,
                                     const float* __restrict__ save_mean,
                                     const float* __restrict__ save_rstd,
                                     float* __restrict__ sum_dy,
                                     float* __restrict__ sum_dy_xhat) {
  int c = blockIdx.y * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = save_mean[c], rstd = save_rstd[c];
  float s = 0.f, sx = 0.f;
  for (int64_t r = blockIdx.x; r < rows; r += gridDim.x) {
    int64_t i = r * C + c;
    float g = to_float<T>(dy[i]);
    if (RELU && to_float<T>(y[i]) <= 0.f) g = 0.f;
    s += g;
    sx += g * (to_float<T>(x[i]) - mean) * rstd;
  }
  atomicAdd(&sum_dy[c], s);
  atomicAdd(&sum_dy_xhat[c], sx);
}

__global__ void bn_bwd_finalize_kernel(const float* __restrict__ sum_dy,
                                       const float* __restrict__ sum_dy_xhat,
                                       const float* __restrict__ gamma,
                                       const float* __restrict__ save_rstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ c_dy,
                                       float* __restrict__ c_xhat,
                                       float* __restrict__ c_const, int C,
                                       float inv_n) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  dgamma[c] = sum_dy_xhat[c];
  dbeta[c] = sum_dy[c];
  // dx = g*rstd * (dy_eff - inv_n*sum_dy - xhat*inv_n*sum_dy_xhat)
  float gr = gamma[c] * save_rstd[c];
  c_dy[c] = gr;
  c_xhat[c] = -gr * inv_n * sum_dy_xhat[c];
  c_const[c] = -gr * inv_n * sum_dy[c];
}

template <typename T, bool RELU, bool RES>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x,
                                    const T* __restrict__ y,
                                    const T* __restrict__ dy,
                                    T* __restrict__ dx,
                                    T* __restrict__ dres,
                                    const float* __restrict__ save_mean,
                                    const float* __restrict__ save_rstd,
                                    const float* __restrict__ c_dy,
                                    const float* __restrict__ c_xhat,
                                    const float* __restrict__ c_const,
                                    int64_t rows, int C) {
  int64_t total = rows * C;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c = (int)(i % C);
    float g = to_float<T>(dy[i]);
    if (RELU && to_float<T>(y[i]) <= 0.f) g = 0.f;
    if (RES) dres[i] = from_float<T>(g);
    float xhat = (to_float<T>(x[i]) - save_mean[c]) * save_rstd[c];
    dx[i] = from_float<T>(c_dy[c] * g + c_xhat[c] * xhat + c_const[c]);
  }
}

// ---------------------------------------------------------------- host
struct BNShape {
  int64_t rows;
  int C;
};

static BNShape bn_shape(const at::Tensor& x) {
  TORCH_CHECK(x.dim() == 4, "expected NCHW tensor (channels_last layout)");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "fused BN requires channels_last activations");
  return {x.size(0) * x.size(2) * x.size(3), (int)x.size(1)};
}

static dim3 reduce_grid(int64_t rows, int C) {
  int cy = (C + kBlock - 1) / kBlock;
  int rx = (int)std::min<int64_t>((rows + 63) / 64, 1024);
  return dim3(rx, cy);
}

// returns (y, save_mean, save_rstd)
std::tuple<at::Tensor, at::Tensor, at::Tensor> bn_fwd(
    const at::Tensor& x, const c10::optional<at::Tensor>& residual,
    const at::Tensor& gamma, const at::Tensor& beta,
    at::Tensor running_mean, at::Tensor running_var, bool training,
    double momentum, double eps, bool relu) {
  auto sh = bn_shape(x);
  auto fopt = x.options().dtype(at::kFloat);
  auto scale = at::empty({sh.C}, fopt);
  auto shift = at::empty({sh.C}, fopt);
  auto save_mean = at::empty({sh.C}, fopt);
  auto save_rstd = at::empty({sh.C}, fopt);
  auto stream = at::hip::getCurrentHIPStream();
  auto g = gamma.contiguous().to(at::kFloat);
  auto b = beta.contiguous().to(at::kFloat);

  int cblocks = (sh.C + kBlock - 1) / kBlock;

#define BN_DISPATCH_T(fn, ...)                                           \
  if (x.scalar_type() == at::kBFloat16) {                                \
    fn(__hip_bfloat16, __VA_ARGS__);                                     \
  } else {                                                               \
    fn(float, __VA_ARGS__);                                              \
  }

  if (training) {
    auto sum = at::zeros({sh.C}, fopt);
    auto sumsq = at::zeros({sh.C}, fopt);
#define BN_RED(T, ...)                                                    \
    hipLaunchKernelGGL((bn_reduce_kernel<T>), reduce_grid(sh.rows, sh.C), \
                       dim3(kBlock), 0, stream,                           \
                       reinterpret_cast<const T*>(x.data_ptr()), sh.rows, \
                       sh.C, sum.data_ptr<float>(), sumsq.data_ptr<float>())
    BN_DISPATCH_T(BN_RED, )
#undef BN_RED
    float inv_n = 1.0f / (float)sh.rows;
    float unbiased = sh.rows > 1 ? (float)sh.rows / (sh.rows - 1) : 1.0f;
    hipLaunchKernelGGL(bn_finalize_kernel, dim3(cblocks), dim3(kBlock), 0,
                       stream, sum.data_ptr<float>(),
                       sumsq.data_ptr<float>(), g.data_ptr<float>(),
                       b.data_ptr<float>(),
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(),
                       save_mean.data_ptr<float>(),
                       save_rstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       sh.C, inv_n, unbiased, (float)momentum, (float)eps,
                       true);
  } else {
    hipLaunchKernelGGL(bn_eval_coeffs_kernel, dim3(cblocks), dim3(kBlock),
                       0, stream, running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), g.data_ptr<float>(),
                       b.data_ptr<float>(), scale.data_ptr<float>(),
                       shift.data_ptr<float>(), sh.C, (float)eps);
  }

  auto y = at::empty_like(x);
  bool has_res = residual.has_value() && residual->defined();
  at::Tensor res;
  if (has_res) {
    res = residual->contiguous(at::MemoryFormat::ChannelsLast);
    TORCH_CHECK(res.sizes() == x.sizes());
  }
  int64_t total = sh.rows * sh.C;
  int grid = elementwise_grid(total, kBlock, 8);
#define BN_APPLY(T, RELUF, RESF)                                          \
  hipLaunchKernelGGL((bn_apply_kernel<T, RELUF, RESF>), dim3(grid),       \
                     dim3(kBlock), 0, stream,                             \
                     reinterpret_cast<const T*>(x.data_ptr()),            \
                     RESF ? reinterpret_cast<const T*>(res.data_ptr())    \
                          : nullptr,                                      \
                     reinterpret_cast<T*>(y.data_ptr()),                  \
                     scale.data_ptr<float>(), shift.data_ptr<float>(),    \
                     sh.rows, sh.C)
#define BN_APPLY_D(T, unused)                                             \
  if (relu) { if (has_res) BN_APPLY(T, true, true);                       \
              else BN_APPLY(T, true, false); }                            \
  else { if (has_res) BN_APPLY(T, false, true);                           \
         else BN_APPLY(T, false, false); }
  BN_DISPATCH_T(BN_APPLY_D, )
#undef BN_APPLY_D
#undef BN_APPLY
  return {y, save_mean, save_rstd};
}

// returns (dx, dgamma, dbeta, dres?) — dres empty when no residual
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_bwd(
    const at::Tensor& x, const at::Tensor& y, const at::Tensor& dy_in,
    const at::Tensor& gamma, const at::Tensor& save_mean,
    const at::Tensor& save_rstd, bool relu, bool has_res) {
  auto sh = bn_shape(x);
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto fopt = x.options().dtype(at::kFloat);
  auto stream = at::hip::getCurrentHIPStream();
  auto g = gamma.contiguous().to(at::kFloat);

  auto sum_dy = at::zeros({sh.C}, fopt);
  auto sum_dy_xhat = at::zeros({sh.C}, fopt);
#define BN_BRED(T, RELUF)                                                  \
  hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, RELUF>),                     \
                     reduce_grid(sh.rows, sh.C), dim3(kBlock), 0, stream,  \
                     reinterpret_cast<const T*>(x.data_ptr()),             \
                     reinterpret_cast<const T*>(y.data_ptr()),             \
                     reinterpret_cast<const T*>(dy.data_ptr()), sh.rows,   \
                     sh.C, save_mean.data_ptr<float>(),                    \
                     save_rstd.data_ptr<float>(),                          \
                     sum_dy.data_ptr<float>(),                            \
                     sum_dy_xhat.data_ptr<float>())
#define BN_BRED_D(T, unused)                                               \
  if (relu) BN_BRED(T, true); else BN_BRED(T, false);
  BN_DISPATCH_T(BN_BRED_D, )
#undef BN_BRED_D
#undef BN_BRED

  auto dgamma = at::empty({sh.C}, fopt);
  auto dbeta = at::empty({sh.C}, fopt);
  auto c_dy = at::empty({sh.C}, fopt);
  auto c_xhat = at::empty({sh.C}, fopt);
  auto c_const = at::empty({sh.C}, fopt);
  int cblocks = (sh.C + kBlock - 1) / kBlock;
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3(cblocks), dim3(kBlock), 0,
                     stream, sum_dy.data_ptr<float>(),
                     sum_dy_xhat.data_ptr<float>(), g.data_ptr<float>(),
                     save_rstd.data_ptr<float>(),
                     dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                     c_dy.data_ptr<float>(), c_xhat.data_ptr<float>(),
                     c_const.data_ptr<float>(), sh.C,
                     1.0f / (float)sh.rows);

  auto dx = at::empty_like(x);
  at::Tensor dres;
  if (has_res) dres = at::empty_like(x);
  int64_t total = sh.rows * sh.C;
  int grid = elementwise_grid(total, kBlock, 8);
#define BN_BAPPLY(T, RELUF, RESF)                                          \
  hipLaunchKernelGGL((bn_bwd_apply_kernel<T, RELUF, RESF>), dim3(grid),    \
                     dim3(kBlock), 0, stream,                              \
                     reinterpret_cast<const T*>(x.data_ptr()),             \
                     reinterpret_cast<const T*>(y.data_ptr()),             \
                     reinterpret_cast<const T*>(dy.data_ptr()),            \
                     reinterpret_cast<T*>(dx.data_ptr()),                  \
                     RESF ? reinterpret_cast<T*>(dres.data_ptr())          \
                          : nullptr,                                       \
                     save_mean.data_ptr<float>(),                          \
                     save_rstd.data_ptr<float>(), c_dy.data_ptr<float>(),  \
                     c_xhat.data_ptr<float>(), c_const.data_ptr<float>(),  \
                     sh.rows, sh.C)
#define BN_BAPPLY_D(T, unused)                                             \
  if (relu) { if (has_res) BN_BAPPLY(T, true, true);                       \
              else BN_BAPPLY(T, true, false); }                            \
  else { if (has_res) BN_BAPPLY(T, false, true);                           \
         else BN_BAPPLY(T, false, false); }
  BN_DISPATCH_T(BN_BAPPLY_D, )
#undef BN_BAPPLY_D
#undef BN_BAPPLY
  return {dx, dgamma, dbeta, dres};
}

}  // namespace turboprune
