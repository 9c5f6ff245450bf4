#include "hip/hip_runtime.h"
// Fused LayerNorm (last-dim) and exact-GELU fwd/bwd for the ViT path.
//
// LayerNorm: one wave per token row (C = 384..1536, 16B octet loads),
// shuffle reductions, saved (mean, rstd) per row; backward writes dx in
// the same pass and accumulates per-channel dgamma/dbeta into per-block
// partial buffers (summed by a finalize kernel — same pattern as the BN
// reduces; a global atomic accumulator serializes per channel).
// GELU: erf-exact (torch default), vectorized elementwise fwd/bwd.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

void colsum_atomic(const at::Tensor&, at::Tensor);   // elementwise.hip
void colsum2_atomic(const at::Tensor&, const at::Tensor&, at::Tensor,
                    at::Tensor);  // elementwise.hip

template <typename T>
struct LnVec;
template <>
struct LnVec<__hip_bfloat16> { static constexpr int kN = 8; };
template <>
struct LnVec<float> { static constexpr int kN = 4; };

template <typename T>
TP_DEVICE void ln_load(const T* p, float* v) {
  constexpr int VN = LnVec<T>::kN;
  uint4 raw = *reinterpret_cast<const uint4*>(p);
  const T* e = reinterpret_cast<const T*>(&raw);
#pragma unroll
  for (int j = 0; j < VN; ++j) v[j] = to_float<T>(e[j]);
}

template <typename T>
TP_DEVICE void ln_store(T* p, const float* v) {
  constexpr int VN = LnVec<T>::kN;
  uint4 raw;
  T* e = reinterpret_cast<T*>(&raw);
#pragma unroll
  for (int j = 0; j < VN; ++j) e[j] = from_float<T>(v[j]);
  *reinterpret_cast<uint4*>(p) = raw;
}

// ---------------- LayerNorm forward --------------------------------------
// block = 256 (4 waves); wave w handles rows blockIdx.x*4 + w, grid-stride.
// Each lane caches its <=MAXPOS octets in registers: x is read ONCE.
constexpr int kLnMaxPos = 4;  // supports C <= 64*kLnMaxPos*VN

template <typename T>
__global__ void ln_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out, int64_t rows,
                              int C, float eps) {
  constexpr int VN = LnVec<T>::kN;
  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int n_oct = C / VN;
  for (int64_t row = (int64_t)blockIdx.x * 4 + wid; row < rows;
       row += (int64_t)gridDim.x * 4) {
    const T* xr = x + row * C;
    float v[kLnMaxPos][VN];
    float s = 0.f, ss = 0.f;
#pragma unroll
    for (int p = 0; p < kLnMaxPos; ++p) {
      int o = lane + p * kWave;
      if (o < n_oct) {
        ln_load(xr + o * VN, v[p]);
#pragma unroll
        for (int j = 0; j < VN; ++j) {
          s += v[p][j];
          ss += v[p][j] * v[p][j];
        }
      }
    }
    s = wave_reduce_sum(s);
    ss = wave_reduce_sum(ss);
    s = __shfl(s, 0, kWave);
    ss = __shfl(ss, 0, kWave);
    float mean = s / C;
    float var = fmaxf(ss / C - mean * mean, 0.f);
    float rstd = rsqrtf(var + eps);
    if (lane == 0) { mean_out[row] = mean; rstd_out[row] = rstd; }
    T* yr = y + row * C;
#pragma unroll
    for (int p = 0; p < kLnMaxPos; ++p) {
      int o = lane + p * kWave;
      if (o < n_oct) {
        float out[VN];
#pragma unroll
        for (int j = 0; j < VN; ++j) {
          int c = o * VN + j;
          out[j] = (v[p][j] - mean) * rstd * gamma[c] + beta[c];
        }
        ln_store(yr + o * VN, out);
      }
    }
  }
}

// ---------------- LayerNorm backward --------------------------------------
// dx = rstd * gamma .* dy - rstd/C * (sum(dy.*gamma) + xhat*sum(dy.*gamma.*xhat))
// Each lane owns fixed channel octets (o = lane + p*64), so dgamma/dbeta
// accumulate in REGISTERS across the wave's rows — no atomics; each WAVE
// writes its own partial row, summed by the finalize kernel.
template <typename T>
__global__ void ln_bwd_kernel(const T* __restrict__ x,
                              const T* __restrict__ dy, T* __restrict__ dx,
                              const float* __restrict__ gamma,
                              const float* __restrict__ mean_in,
                              const float* __restrict__ rstd_in,
                              float* __restrict__ partial_dg,
                              float* __restrict__ partial_db, int64_t rows,
                              int C) {
  constexpr int VN = LnVec<T>::kN;
  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int n_oct = C / VN;
  float adg[kLnMaxPos][VN], adb[kLnMaxPos][VN];
#pragma unroll
  for (int p = 0; p < kLnMaxPos; ++p)
#pragma unroll
    for (int j = 0; j < VN; ++j) adg[p][j] = adb[p][j] = 0.f;

  for (int64_t row = (int64_t)blockIdx.x * 4 + wid; row < rows;
       row += (int64_t)gridDim.x * 4) {
    const T* xr = x + row * C;
    const T* gr = dy + row * C;
    float mean = mean_in[row], rstd = rstd_in[row];
    float v[kLnMaxPos][VN], g[kLnMaxPos][VN];
    float s1 = 0.f, s2 = 0.f;  // sum(dy*gamma), sum(dy*gamma*xhat)
#pragma unroll
    for (int p = 0; p < kLnMaxPos; ++p) {
      int o = lane + p * kWave;
      if (o < n_oct) {
        ln_load(xr + o * VN, v[p]);
        ln_load(gr + o * VN, g[p]);
#pragma unroll
        for (int j = 0; j < VN; ++j) {
          int c = o * VN + j;
          float xhat = (v[p][j] - mean) * rstd;
          float dgj = g[p][j] * gamma[c];
          s1 += dgj;
          s2 += dgj * xhat;
          adg[p][j] += g[p][j] * xhat;
          adb[p][j] += g[p][j];
        }
      }
    }
    s1 = wave_reduce_sum(s1);
    s2 = wave_reduce_sum(s2);
    s1 = __shfl(s1, 0, kWave) / C;
    s2 = __shfl(s2, 0, kWave) / C;
    T* dxr = dx + row * C;
#pragma unroll
    for (int p = 0; p < kLnMaxPos; ++p) {
      int o = lane + p * kWave;
      if (o < n_oct) {
        float out[VN];
#pragma unroll
        for (int j = 0; j < VN; ++j) {
          int c = o * VN + j;
          float xhat = (v[p][j] - mean) * rstd;
          out[j] = rstd * (g[p][j] * gamma[c] - s1 - xhat * s2);
        }
        ln_store(dxr + o * VN, out);
      }
    }
  }
  // per-wave partial rows: slot = blockIdx.x*4 + wid
  int64_t slot = (int64_t)blockIdx.x * 4 + wid;
#pragma unroll
  for (int p = 0; p < kLnMaxPos; ++p) {
    int o = lane + p * kWave;
    if (o < n_oct) {
#pragma unroll
      for (int j = 0; j < VN; ++j) {
        partial_dg[slot * C + o * VN + j] = adg[p][j];
        partial_db[slot * C + o * VN + j] = adb[p][j];
      }
    }
  }
}

// ---------------------------------------------------------------- host LN
static int ln_grid(int64_t rows) {
  int64_t b = (rows + 3) / 4;
  return (int)std::min<int64_t>(b, 1024);
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> ln_fwd(
    const at::Tensor& x_in, const at::Tensor& gamma, const at::Tensor& beta,
    double eps) {
  auto x = x_in.contiguous();
  int C = (int)x.size(-1);
  int64_t rows = x.numel() / C;
  int vn = x.scalar_type() == at::kBFloat16 ? 8 : 4;
  TORCH_CHECK(C % vn == 0, "ln_fwd: C % ", vn, " != 0");
  TORCH_CHECK(C / vn <= 64 * 4, "ln_fwd: C too large for register cache");
  auto fopt = x.options().dtype(at::kFloat);
  auto y = at::empty_like(x);
  auto mean = at::empty({rows}, fopt);
  auto rstd = at::empty({rows}, fopt);
  auto g = gamma.contiguous().to(at::kFloat);
  auto b = beta.contiguous().to(at::kFloat);
  auto stream = at::hip::getCurrentHIPStream();
  int grid = ln_grid(rows);
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(ln_fwd_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                       g.data_ptr<float>(), b.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), rows,
                       C, (float)eps);
  } else {
    TORCH_CHECK(x.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(ln_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       stream, x.data_ptr<float>(), y.data_ptr<float>(),
                       g.data_ptr<float>(), b.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(), rows,
                       C, (float)eps);
  }
  return {y, mean, rstd};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> ln_bwd(
    const at::Tensor& x_in, const at::Tensor& dy_in,
    const at::Tensor& gamma, const at::Tensor& mean,
    const at::Tensor& rstd) {
  auto x = x_in.contiguous();
  auto dy = dy_in.contiguous();
  int C = (int)x.size(-1);
  int64_t rows = x.numel() / C;
  auto fopt = x.options().dtype(at::kFloat);
  auto dx = at::empty_like(x);
  auto g = gamma.contiguous().to(at::kFloat);
  int grid = ln_grid(rows);
  int slots = grid * 4;  // one partial row per wave
  auto partial_dg = at::empty({slots, C}, fopt);
  auto partial_db = at::empty({slots, C}, fopt);
  auto stream = at::hip::getCurrentHIPStream();
  size_t lds_bytes = 0;
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(ln_bwd_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), lds_bytes, stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()),
                       g.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), partial_dg.data_ptr<float>(),
                       partial_db.data_ptr<float>(), rows, C);
  } else {
    hipLaunchKernelGGL(ln_bwd_kernel<float>, dim3(grid), dim3(kBlock),
                       lds_bytes, stream, x.data_ptr<float>(),
                       dy.data_ptr<float>(), dx.data_ptr<float>(),
                       g.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), partial_dg.data_ptr<float>(),
                       partial_db.data_ptr<float>(), rows, C);
  }
  auto dgamma = at::zeros({C}, fopt);
  auto dbeta = at::zeros({C}, fopt);
  colsum2_atomic(partial_dg, partial_db, dgamma, dbeta);
  return {dx, dgamma, dbeta};
}

// ---------------- GELU (erf-exact) ----------------------------------------
template <typename T>
__global__ void gelu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                int64_t n_oct) {
  constexpr int VN = LnVec<T>::kN;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float inv_sqrt2 = 0.70710678118654752f;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n_oct;
       i += stride) {
    float v[VN], out[VN];
    ln_load(x + i * VN, v);
#pragma unroll
    for (int j = 0; j < VN; ++j)
      out[j] = 0.5f * v[j] * (1.f + erff(v[j] * inv_sqrt2));
    ln_store(y + i * VN, out);
  }
}

template <typename T>
__global__ void gelu_bwd_kernel(const T* __restrict__ x,
                                const T* __restrict__ dy,
                                T* __restrict__ dx, int64_t n_oct) {
  constexpr int VN = LnVec<T>::kN;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const float inv_sqrt2 = 0.70710678118654752f;
  const float inv_sqrt2pi = 0.39894228040143268f;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n_oct;
       i += stride) {
    float v[VN], g[VN], out[VN];
    ln_load(x + i * VN, v);
    ln_load(dy + i * VN, g);
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float cdf = 0.5f * (1.f + erff(v[j] * inv_sqrt2));
      float pdf = inv_sqrt2pi * __expf(-0.5f * v[j] * v[j]);
      out[j] = g[j] * (cdf + v[j] * pdf);
    }
    ln_store(dx + i * VN, out);
  }
}

at::Tensor gelu_fwd(const at::Tensor& x_in) {
  auto x = x_in.contiguous();
  int vn = x.scalar_type() == at::kBFloat16 ? 8 : 4;
  TORCH_CHECK(x.numel() % vn == 0);
  auto y = at::empty_like(x);
  int64_t n_oct = x.numel() / vn;
  auto stream = at::hip::getCurrentHIPStream();
  int grid = elementwise_grid(n_oct, kBlock, 4);
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(gelu_fwd_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(y.data_ptr()),
                       n_oct);
  } else {
    TORCH_CHECK(x.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(gelu_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       stream, x.data_ptr<float>(), y.data_ptr<float>(),
                       n_oct);
  }
  return y;
}

at::Tensor gelu_bwd(const at::Tensor& x_in, const at::Tensor& dy_in) {
  auto x = x_in.contiguous();
  auto dy = dy_in.contiguous();
  int vn = x.scalar_type() == at::kBFloat16 ? 8 : 4;
  auto dx = at::empty_like(x);
  int64_t n_oct = x.numel() / vn;
  auto stream = at::hip::getCurrentHIPStream();
  int grid = elementwise_grid(n_oct, kBlock, 4);
  if (x.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(gelu_bwd_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(dy.data_ptr()),
                       reinterpret_cast<__hip_bfloat16*>(dx.data_ptr()),
                       n_oct);
  } else {
    hipLaunchKernelGGL(gelu_bwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       stream, x.data_ptr<float>(), dy.data_ptr<float>(),
                       dx.data_ptr<float>(), n_oct);
  }
  return dx;
}

}  // namespace turboprune
