// Common helpers for the turboprune_amd gfx950 (CDNA4) kernels.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include <cstdint>

#define TP_HOST_DEVICE __host__ __device__ __forceinline__
#define TP_DEVICE __device__ __forceinline__

namespace turboprune {

constexpr int kWave = 64;           // CDNA wavefront width
constexpr int kBlock = 256;         // default block: 4 waves
// MI355X: 256 CUs; memory-bound grids want >> 256 workgroups but capped
// so grid-stride loops amortize launch setup (guide §6 G11).
constexpr int kMaxGrid = 2048;

inline int elementwise_grid(int64_t n, int block = kBlock,
                            int64_t items_per_thread = 1) {
  int64_t blocks =
      (n + (int64_t)block * items_per_thread - 1) /
      ((int64_t)block * items_per_thread);
  if (blocks > kMaxGrid) blocks = kMaxGrid;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

// ---- dtype conversion helpers -------------------------------------------
template <typename T>
TP_DEVICE float to_float(T v);
template <>
TP_DEVICE float to_float<float>(float v) { return v; }
template <>
TP_DEVICE float to_float<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <>
TP_DEVICE float to_float<_Float16>(_Float16 v) { return (float)v; }

template <typename T>
TP_DEVICE T from_float(float v);
template <>
TP_DEVICE float from_float<float>(float v) { return v; }
template <>
TP_DEVICE __hip_bfloat16 from_float<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <>
TP_DEVICE _Float16 from_float<_Float16>(float v) { return (_Float16)v; }

// ---- wave / block reductions --------------------------------------------
TP_DEVICE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, kWave);
  return v;  // valid in lane 0
}

TP_DEVICE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, kWave));
  return v;
}

// Block reduction via LDS; one value per block in thread 0.
template <int BLOCK>
TP_DEVICE float block_reduce_sum(float v, float* lds /* BLOCK/kWave */) {
  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  if (wid == 0) {
    v = (lane < BLOCK / kWave) ? lds[lane] : 0.f;
    v = wave_reduce_sum(v);
  }
  return v;
}

template <int BLOCK>
TP_DEVICE float block_reduce_max(float v, float* lds) {
  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  if (wid == 0) {
    v = (lane < BLOCK / kWave) ? lds[lane] : -INFINITY;
    v = wave_reduce_max(v);
  }
  return v;
}

// ---- philox4x32-10 counter-based RNG ------------------------------------
struct Philox4 {
  uint32_t x, y, z, w;
};

TP_DEVICE uint32_t mulhi32(uint32_t a, uint32_t b) {
  return (uint32_t)(((uint64_t)a * b) >> 32);
}

TP_DEVICE Philox4 philox4x32(uint64_t seed, uint64_t counter) {
  constexpr uint32_t kPhiloxM0 = 0xD2511F53u, kPhiloxM1 = 0xCD9E8D57u;
  constexpr uint32_t kPhiloxW0 = 0x9E3779B9u, kPhiloxW1 = 0xBB67AE85u;
  uint32_t c0 = (uint32_t)counter, c1 = (uint32_t)(counter >> 32);
  uint32_t c2 = 0, c3 = 0;
  uint32_t k0 = (uint32_t)seed, k1 = (uint32_t)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint32_t lo0 = kPhiloxM0 * c0, hi0 = mulhi32(kPhiloxM0, c0);
    uint32_t lo1 = kPhiloxM1 * c2, hi1 = mulhi32(kPhiloxM1, c2);
    uint32_t n0 = hi1 ^ c1 ^ k0, n1 = lo1;
    uint32_t n2 = hi0 ^ c3 ^ k1, n3 = lo0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += kPhiloxW0; k1 += kPhiloxW1;
  }
  return {c0, c1, c2, c3};
}

TP_DEVICE float u32_to_uniform(uint32_t v) {
  // (0, 1] uniform from 32 random bits
  return (v >> 8) * (1.0f / 16777216.0f);
}

}  // namespace turboprune
