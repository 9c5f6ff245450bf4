// Elementwise tier: mask application, score construction, mask rewrite,
// Bernoulli mask generation.
//
// All kernels are memory-bound grid-stride loops vectorized to 16 B/lane
// (guide §6 G13: 64-lane waves, float4 loads); masks/weights are fp32,
// outputs fp32 or bf16 (the compute dtype).

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include <algorithm>

#include "common.h"

namespace turboprune {

// ---- mask_apply: out = (mask != 0) ? w : 0, cast to OutT ----------------
template <typename OutT>
__global__ void mask_apply_kernel_v4(const float* __restrict__ w,
                                     const float* __restrict__ m,
                                     OutT* __restrict__ out, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 wv = reinterpret_cast<const float4*>(w)[i];
    float4 mv = reinterpret_cast<const float4*>(m)[i];
    OutT o[4];
    o[0] = from_float<OutT>(mv.x != 0.f ? wv.x : 0.f);
    o[1] = from_float<OutT>(mv.y != 0.f ? wv.y : 0.f);
    o[2] = from_float<OutT>(mv.z != 0.f ? wv.z : 0.f);
    o[3] = from_float<OutT>(mv.w != 0.f ? wv.w : 0.f);
    *reinterpret_cast<uint4*>(&out[i * 4]) =
        *reinterpret_cast<const uint4*>(&o[0]);
  }
}

template <typename OutT>
__global__ void mask_apply_kernel(const float* __restrict__ w,
                                  const float* __restrict__ m,
                                  OutT* __restrict__ out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    out[i] = from_float<OutT>(m[i] != 0.f ? w[i] : 0.f);
}

// OutT must be 4 bytes for the v4 uint4 store trick to hold 16 B; for
// bf16 out we store 4x2 B = 8 B via uint2.
template <>
__global__ void mask_apply_kernel_v4<__hip_bfloat16>(
    const float* __restrict__ w, const float* __restrict__ m,
    __hip_bfloat16* __restrict__ out, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    float4 wv = reinterpret_cast<const float4*>(w)[i];
    float4 mv = reinterpret_cast<const float4*>(m)[i];
    __hip_bfloat16 o[4];
    o[0] = __float2bfloat16(mv.x != 0.f ? wv.x : 0.f);
    o[1] = __float2bfloat16(mv.y != 0.f ? wv.y : 0.f);
    o[2] = __float2bfloat16(mv.z != 0.f ? wv.z : 0.f);
    o[3] = __float2bfloat16(mv.w != 0.f ? wv.w : 0.f);
    *reinterpret_cast<uint2*>(&out[i * 4]) =
        *reinterpret_cast<const uint2*>(&o[0]);
  }
}

// re-layout t to match `like`'s strides (dense); no-op when it already does
static at::Tensor to_layout_of(const at::Tensor& t, const at::Tensor& like) {
  if (t.strides().vec() == like.strides().vec() &&
      t.is_non_overlapping_and_dense())
    return t;
  auto out = at::empty_strided(like.sizes(), like.strides(), t.options());
  out.copy_(t);
  return out;
}

at::Tensor mask_apply(const at::Tensor& weight, const at::Tensor& mask,
                      at::ScalarType out_dtype) {
  TORCH_CHECK(weight.is_cuda() && mask.is_cuda(), "expected GPU tensors");
  // preserve the weight's (possibly channels_last) layout: all tensors
  // iterate flat in storage order with matching strides
  auto w = weight.is_non_overlapping_and_dense()
               ? weight.to(at::kFloat)
               : weight.contiguous().to(at::kFloat);
  auto m = to_layout_of(mask.to(at::kFloat), w);
  TORCH_CHECK(w.numel() == m.numel(), "weight/mask numel mismatch");
  auto out = at::empty_strided(w.sizes(), w.strides(),
                               w.options().dtype(out_dtype));
  int64_t n = w.numel();
  auto stream = at::hip::getCurrentHIPStream();
  bool vec4 = (n % 4 == 0);
  if (out_dtype == at::kFloat) {
    if (vec4) {
      hipLaunchKernelGGL(mask_apply_kernel_v4<float>,
                         dim3(elementwise_grid(n / 4)), dim3(kBlock), 0,
                         stream, w.data_ptr<float>(), m.data_ptr<float>(),
                         out.data_ptr<float>(), n / 4);
    } else {
      hipLaunchKernelGGL(mask_apply_kernel<float>,
                         dim3(elementwise_grid(n)), dim3(kBlock), 0, stream,
                         w.data_ptr<float>(), m.data_ptr<float>(),
                         out.data_ptr<float>(), n);
    }
  } else if (out_dtype == at::kBFloat16) {
    auto* op = reinterpret_cast<__hip_bfloat16*>(out.data_ptr());
    if (vec4) {
      hipLaunchKernelGGL(mask_apply_kernel_v4<__hip_bfloat16>,
                         dim3(elementwise_grid(n / 4)), dim3(kBlock), 0,
                         stream, w.data_ptr<float>(), m.data_ptr<float>(),
                         op, n / 4);
    } else {
      hipLaunchKernelGGL(mask_apply_kernel<__hip_bfloat16>,
                         dim3(elementwise_grid(n)), dim3(kBlock), 0, stream,
                         w.data_ptr<float>(), m.data_ptr<float>(), op, n);
    }
  } else {
    TORCH_CHECK(false, "mask_apply: unsupported out dtype");
  }
  return out;
}

// ---- mask rewrite: mask = (score <= thr) ? 0 : 1 ------------------------
__global__ void mask_from_threshold_kernel(float* __restrict__ mask,
                                           const float* __restrict__ score,
                                           float thr, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    mask[i] = score[i] <= thr ? 0.f : 1.f;
}

void mask_from_threshold_(at::Tensor mask, const at::Tensor& score,
                          double thr) {
  TORCH_CHECK(mask.is_cuda() && score.is_cuda());
  TORCH_CHECK(mask.is_non_overlapping_and_dense());
  auto s = to_layout_of(score.to(at::kFloat), mask);
  int64_t n = mask.numel();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mask_from_threshold_kernel, dim3(elementwise_grid(n)),
                     dim3(kBlock), 0, stream, mask.data_ptr<float>(),
                     s.data_ptr<float>(), (float)thr, n);
}

// ---- score construction: |w*m| or |w*m*g| --------------------------------
__global__ void masked_abs_score_kernel(const float* __restrict__ w,
                                        const float* __restrict__ m,
                                        const float* __restrict__ g,
                                        float* __restrict__ out, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float v = w[i] * m[i];
    if (g != nullptr) v *= g[i];
    out[i] = fabsf(v);
  }
}

at::Tensor masked_abs_score(const at::Tensor& weight, const at::Tensor& mask,
                            const at::Tensor& other) {
  auto w = weight.is_non_overlapping_and_dense()
               ? weight.to(at::kFloat)
               : weight.contiguous().to(at::kFloat);
  auto m = to_layout_of(mask.to(at::kFloat), w);
  const float* g = nullptr;
  at::Tensor o;
  if (other.defined() && other.numel() > 0) {
    o = to_layout_of(other.to(at::kFloat), w);
    g = o.data_ptr<float>();
  }
  auto out = at::empty_strided(w.sizes(), w.strides(), w.options());
  int64_t n = w.numel();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(masked_abs_score_kernel, dim3(elementwise_grid(n)),
                     dim3(kBlock), 0, stream, w.data_ptr<float>(),
                     m.data_ptr<float>(), g, out.data_ptr<float>(), n);
  return out;
}

// ---- Bernoulli mask fill (Philox, K9) ------------------------------------
__global__ void bernoulli_mask_kernel(float* __restrict__ mask, float p,
                                      uint64_t seed, int64_t n4) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    Philox4 r = philox4x32(seed, (uint64_t)i);
    float4 v;
    v.x = u32_to_uniform(r.x) <= p ? 1.f : 0.f;
    v.y = u32_to_uniform(r.y) <= p ? 1.f : 0.f;
    v.z = u32_to_uniform(r.z) <= p ? 1.f : 0.f;
    v.w = u32_to_uniform(r.w) <= p ? 1.f : 0.f;
    int64_t base = i * 4;
    // n4 counts ceil(n/4) quads; guard the tail
    float* out = mask + base;
    int64_t remain = (n4 * 4 - base);  // >= 1
    if (remain >= 4) {
      *reinterpret_cast<float4*>(out) = v;
    } else {
      out[0] = v.x;
      if (remain > 1) out[1] = v.y;
      if (remain > 2) out[2] = v.z;
    }
  }
}

__global__ void bernoulli_tail_kernel(float* __restrict__ m, float p,
                                      uint64_t seed, int64_t base,
                                      int64_t n) {
  int64_t i = base + threadIdx.x;
  if (i < n) {
    Philox4 r = philox4x32(seed, (uint64_t)(1ull << 62) + i);
    m[i] = u32_to_uniform(r.x) <= p ? 1.f : 0.f;
  }
}

void bernoulli_mask_(at::Tensor mask, double p, int64_t seed) {
  TORCH_CHECK(mask.is_cuda() && mask.is_non_overlapping_and_dense() &&
              mask.scalar_type() == at::kFloat);
  int64_t n = mask.numel();
  // pad to quads; kernel guards tail against n4*4 (== padded length),
  // so pass exact quad count with tail handling below
  int64_t n4 = n / 4;
  auto stream = at::hip::getCurrentHIPStream();
  if (n4 > 0)
    hipLaunchKernelGGL(bernoulli_mask_kernel, dim3(elementwise_grid(n4)),
                       dim3(kBlock), 0, stream, mask.data_ptr<float>(),
                       (float)p, (uint64_t)seed, n4);
  int64_t tail = n - n4 * 4;
  if (tail > 0)
    hipLaunchKernelGGL(bernoulli_tail_kernel, dim3(1), dim3(4), 0, stream,
                       mask.data_ptr<float>(), (float)p, (uint64_t)seed,
                       n4 * 4, n);
}

// ---- parallel column sum: out[c] += sum_r in[r][C] -----------------------
// 2D grid (channel blocks x row chunks), one atomicAdd per thread per
// chunk (R/chunk atomics per channel). Replaces serial per-channel loops
// over partial-reduction buffers (a C-thread kernel looping 4k rows was
// 1.2 ms).
__global__ void colsum_atomic_kernel(const float* __restrict__ in,
                                     float* __restrict__ out, int64_t R,
                                     int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  int64_t r0 = (int64_t)blockIdx.y * 128;
  int64_t r1 = r0 + 128 < R ? r0 + 128 : R;
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  int64_t r = r0;
  for (; r + 3 < r1; r += 4) {
    s0 += in[r * C + c];
    s1 += in[(r + 1) * C + c];
    s2 += in[(r + 2) * C + c];
    s3 += in[(r + 3) * C + c];
  }
  for (; r < r1; ++r) s0 += in[r * C + c];
  atomicAdd(&out[c], (s0 + s1) + (s2 + s3));
}

void colsum_atomic(const at::Tensor& partial, at::Tensor out) {
  int64_t R = partial.size(0);
  int C = (int)partial.size(1);
  int cblocks = (C + kBlock - 1) / kBlock;
  int rchunks = (int)((R + 127) / 128);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(colsum_atomic_kernel, dim3(cblocks, rchunks),
                     dim3(kBlock), 0, stream, partial.data_ptr<float>(),
                     out.data_ptr<float>(), R, C);
}

// fused pair variant: one launch for the (sum, sumsq) / (g, gx) /
// (dgamma, dbeta) partial pairs the BN/LN finalizes always consume
// together (212 single-array launches per ResNet50 step otherwise)
__global__ void colsum2_atomic_kernel(const float* __restrict__ in_a,
                                      const float* __restrict__ in_b,
                                      float* __restrict__ out_a,
                                      float* __restrict__ out_b, int64_t R,
                                      int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  int64_t r0 = (int64_t)blockIdx.y * 128;
  int64_t r1 = r0 + 128 < R ? r0 + 128 : R;
  float a0 = 0.f, a1 = 0.f, b0 = 0.f, b1 = 0.f;
  int64_t r = r0;
  for (; r + 1 < r1; r += 2) {
    a0 += in_a[r * C + c];
    a1 += in_a[(r + 1) * C + c];
    b0 += in_b[r * C + c];
    b1 += in_b[(r + 1) * C + c];
  }
  for (; r < r1; ++r) {
    a0 += in_a[r * C + c];
    b0 += in_b[r * C + c];
  }
  atomicAdd(&out_a[c], a0 + a1);
  atomicAdd(&out_b[c], b0 + b1);
}

void colsum2_atomic(const at::Tensor& pa, const at::Tensor& pb,
                    at::Tensor oa, at::Tensor ob) {
  int64_t R = pa.size(0);
  int C = (int)pa.size(1);
  int cblocks = (C + kBlock - 1) / kBlock;
  int rchunks = (int)((R + 127) / 128);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(colsum2_atomic_kernel, dim3(cblocks, rchunks),
                     dim3(kBlock), 0, stream, pa.data_ptr<float>(),
                     pb.data_ptr<float>(), oa.data_ptr<float>(),
                     ob.data_ptr<float>(), R, C);
}

// ---- bf16 column sum (bias gradients) -----------------------------------
// grad_b = sum_m gy[m, n]: one streaming pass (the eager torch reduce
// was 5.5% of the DeiT step, r2w). bn_reduce-style grid: 8 column
// octets x 32 row lanes per block, LDS tree, one atomicAdd per column
// per block (fp32 atomics — same determinism class as the bn dgamma
// path).
__global__ void colsum_bf16_kernel(const __hip_bfloat16* __restrict__ A,
                                   float* __restrict__ out, int64_t rows,
                                   int C) {
  __shared__ float ls[8][8][32];
  int oct_in_blk = threadIdx.x & 7;
  int lane = threadIdx.x >> 3;
  int oct = blockIdx.y * 8 + oct_in_blk;
  int c0 = oct * 8;
  float s[8] = {};
  if (c0 < C) {
    for (int64_t r = (int64_t)blockIdx.x * 32 + lane; r < rows;
         r += (int64_t)gridDim.x * 32) {
      uint4 v = *reinterpret_cast<const uint4*>(A + r * C + c0);
      const __hip_bfloat16* p =
          reinterpret_cast<const __hip_bfloat16*>(&v);
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += __bfloat162float(p[j]);
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) ls[oct_in_blk][j][lane] = s[j];
  __syncthreads();
  for (int step = 16; step > 0; step >>= 1) {
    if (lane < step) {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ls[oct_in_blk][j][lane] += ls[oct_in_blk][j][lane + step];
    }
    __syncthreads();
  }
  if (lane == 0 && c0 < C) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      atomicAdd(&out[c0 + j], ls[oct_in_blk][j][0]);
  }
}

at::Tensor colsum_bf16(const at::Tensor& A_in) {
  auto A = A_in.contiguous();
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              A.dim() == 2 && A.size(1) % 8 == 0);
  int64_t rows = A.size(0);
  int C = (int)A.size(1);
  auto out = at::zeros({C}, A.options().dtype(at::kFloat));
  int cb = (C / 8 + 7) / 8;
  int64_t rb_want = (rows + 32 * 16 - 1) / (32 * 16);
  int rb = (int)std::min<int64_t>(std::max<int64_t>(rb_want, 1),
                                  std::max<int64_t>(4096 / cb, 8));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(colsum_bf16_kernel, dim3(rb, cb), dim3(256), 0,
                     stream,
                     reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                     out.data_ptr<float>(), rows, C);
  return out;
}

// ---- SynFlow linearize / restore ---------------------------
// One fused pass each way: linearize emits the sign as int8 (4x smaller
// than the reference's fp32 sign tensors, pruning_utils.py:223-248) and
// takes |t| in place; restore multiplies the sign back.
__global__ void sign_abs_kernel(float* __restrict__ t,
                                int8_t* __restrict__ sign, int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    float v = t[i];
    sign[i] = v > 0.f ? 1 : (v < 0.f ? -1 : 0);
    t[i] = fabsf(v);
  }
}

__global__ void mul_sign_kernel(float* __restrict__ t,
                                const int8_t* __restrict__ sign,
                                int64_t n) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    t[i] *= (float)sign[i];
}

at::Tensor sign_abs_(at::Tensor t) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kFloat &&
              t.is_non_overlapping_and_dense(),
              "sign_abs_: dense fp32 GPU tensor expected");
  auto sign = at::empty_like(t, t.options().dtype(at::kChar));
  int64_t n = t.numel();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(sign_abs_kernel, dim3(elementwise_grid(n)),
                     dim3(kBlock), 0, stream, t.data_ptr<float>(),
                     sign.data_ptr<int8_t>(), n);
  return sign;
}

void mul_sign_(at::Tensor t, const at::Tensor& sign) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kFloat &&
              sign.scalar_type() == at::kChar &&
              t.numel() == sign.numel());
  int64_t n = t.numel();
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mul_sign_kernel, dim3(elementwise_grid(n)),
                     dim3(kBlock), 0, stream, t.data_ptr<float>(),
                     sign.data_ptr<int8_t>(), n);
}

}  // namespace turboprune
