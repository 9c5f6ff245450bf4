// Fused cross-entropy: one kernel fwd (log-sum-exp + NLL,
// mean reduction), one kernel bwd (softmax - onehot, scaled).
// Logits (B, C) bf16 or fp32, C up to a few thousand (ImageNet: 1000).
// One wave per row: vectorized loads, shuffle reductions — replaces the
// eager log_softmax + nll_loss + their backward launches.
//
// Also here: accuracy_count — argmax == target reduction.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              float* __restrict__ loss_sum,
                              float* __restrict__ lse, int B, int C) {
  // one wave per row; blockDim.x == kBlock → 4 rows per block
  int row = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  int lane = threadIdx.x & (kWave - 1);
  if (row >= B) return;
  const T* x = logits + (int64_t)row * C;

  float m = -INFINITY;
  for (int c = lane; c < C; c += kWave) m = fmaxf(m, to_float<T>(x[c]));
  m = wave_reduce_max(m);
  m = __shfl(m, 0, kWave);

  float s = 0.f;
  for (int c = lane; c < C; c += kWave) s += expf(to_float<T>(x[c]) - m);
  s = wave_reduce_sum(s);
  s = __shfl(s, 0, kWave);

  if (lane == 0) {
    float l = m + logf(s);
    lse[row] = l;
    float nll = l - to_float<T>(x[target[row]]);
    atomicAdd(loss_sum, nll / B);
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ grad_out,
                              T* __restrict__ grad_logits, int B, int C) {
  int row = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  int lane = threadIdx.x & (kWave - 1);
  if (row >= B) return;
  const T* x = logits + (int64_t)row * C;
  T* g = grad_logits + (int64_t)row * C;
  float l = lse[row];
  float scale = grad_out[0] / B;
  int64_t t = target[row];
  for (int c = lane; c < C; c += kWave) {
    float p = expf(to_float<T>(x[c]) - l);
    float gv = (p - (c == t ? 1.f : 0.f)) * scale;
    g[c] = from_float<T>(gv);
  }
}

std::tuple<at::Tensor, at::Tensor> ce_fwd(const at::Tensor& logits,
                                          const at::Tensor& target) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  TORCH_CHECK(target.scalar_type() == at::kLong);
  int B = logits.size(0), C = logits.size(1);
  auto loss = at::zeros({}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({B}, logits.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  int rows_per_block = kBlock / kWave;
  int grid = (B + rows_per_block - 1) / rows_per_block;
  auto tgt = target.contiguous();
  if (logits.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(ce_fwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       stream, logits.data_ptr<float>(),
                       tgt.data_ptr<int64_t>(), loss.data_ptr<float>(),
                       lse.data_ptr<float>(), B, C);
  } else if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(ce_fwd_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(
                           logits.data_ptr()),
                       tgt.data_ptr<int64_t>(), loss.data_ptr<float>(),
                       lse.data_ptr<float>(), B, C);
  } else {
    TORCH_CHECK(false, "ce_fwd: unsupported dtype");
  }
  return {loss, lse};
}

at::Tensor ce_bwd(const at::Tensor& logits, const at::Tensor& target,
                  const at::Tensor& lse, const at::Tensor& grad_out) {
  int B = logits.size(0), C = logits.size(1);
  auto grad = at::empty_like(logits);
  auto stream = at::hip::getCurrentHIPStream();
  int rows_per_block = kBlock / kWave;
  int grid = (B + rows_per_block - 1) / rows_per_block;
  auto tgt = target.contiguous();
  auto go = grad_out.contiguous().to(at::kFloat);
  if (logits.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(ce_bwd_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       stream, logits.data_ptr<float>(),
                       tgt.data_ptr<int64_t>(), lse.data_ptr<float>(),
                       go.data_ptr<float>(), grad.data_ptr<float>(), B, C);
  } else {
    hipLaunchKernelGGL(ce_bwd_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(
                           logits.data_ptr()),
                       tgt.data_ptr<int64_t>(), lse.data_ptr<float>(),
                       go.data_ptr<float>(),
                       reinterpret_cast<__hip_bfloat16*>(grad.data_ptr()),
                       B, C);
  }
  return grad;
}

// ---- accuracy: count(argmax(logits) == target) ---------------------------
template <typename T>
__global__ void accuracy_kernel(const T* __restrict__ logits,
                                const int64_t* __restrict__ target,
                                int64_t* __restrict__ count, int B, int C) {
  int row = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  int lane = threadIdx.x & (kWave - 1);
  if (row >= B) return;
  const T* x = logits + (int64_t)row * C;
  float best = -INFINITY;
  int best_c = 0;
  for (int c = lane; c < C; c += kWave) {
    float v = to_float<T>(x[c]);
    if (v > best) { best = v; best_c = c; }
  }
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) {
    float ob = __shfl_down(best, off, kWave);
    int oc = __shfl_down(best_c, off, kWave);
    // tie-break toward the lower index (torch argmax semantics)
    if (ob > best || (ob == best && oc < best_c)) { best = ob; best_c = oc; }
  }
  if (lane == 0 && best_c == (int)target[row])
    atomicAdd(reinterpret_cast<unsigned long long*>(count), 1ull);
}

at::Tensor accuracy_count(const at::Tensor& logits, const at::Tensor& target) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 && logits.is_contiguous());
  int B = logits.size(0), C = logits.size(1);
  auto count = at::zeros({}, logits.options().dtype(at::kLong));
  auto stream = at::hip::getCurrentHIPStream();
  int rows_per_block = kBlock / kWave;
  int grid = (B + rows_per_block - 1) / rows_per_block;
  auto tgt = target.contiguous();
  if (logits.scalar_type() == at::kFloat) {
    hipLaunchKernelGGL(accuracy_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       stream, logits.data_ptr<float>(),
                       tgt.data_ptr<int64_t>(), count.data_ptr<int64_t>(),
                       B, C);
  } else if (logits.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(accuracy_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(
                           logits.data_ptr()),
                       tgt.data_ptr<int64_t>(), count.data_ptr<int64_t>(),
                       B, C);
  } else {
    TORCH_CHECK(false, "accuracy_count: unsupported dtype");
  }
  return count;
}

}  // namespace turboprune
