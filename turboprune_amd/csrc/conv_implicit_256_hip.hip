#include "hip/hip_runtime.h"
// EXPERIMENTAL — implicit-GEMM NHWC conv forward / grad_input on the
// 256x256 8-phase MFMA structure. Composes two independently verified
// pieces:
//   - the 8-phase schedule of gemm_256_8phase.hip (staging rotation,
//     vmcnt(4) boundaries + tail drain, register-resident B fragments;
//     machine-verified in tests/test_gemm256_{addressing,schedule,
//     dataflow}.py — the staging/compute structure here is IDENTICAL),
//   - the im2col gather coordinates of conv_implicit.hip (zero-page
//     OOB/padding/dilation; pinned by tests/test_conv_gradin_coords.py).
// Differences from gemm256: the A half-tile source is the on-the-fly
// im2col gather (per-lane addresses are free in glds), the B source is
// the tap-major channels_last weight with a Cout guard, and the
// epilogue stores to the UNPADDED (M, Cout) output with bias.
//
// Routing: conv2d_implicit_fwd/gradin (conv_implicit.hip) call into
// this variant when TURBOPRUNE_CONV256=1 and the shape fits
// (Cout >= 256 after padding considerations, >= 2 K-tiles). Default
// OFF until scripts/round2_device_checks.sh validates it on hardware.
//
// Requirements: Cin % 64 == 0 (BK tile inside one tap), bf16 NHWC.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace c256 {
constexpr int BM = 256, BN = 256, BK = 64;
constexpr int WARPS_N = 4;
constexpr int WM = 128, WN = 64;
constexpr int MREP = 8, NREP = 4;
constexpr int kHalfBytes = 128 * BK * 2;
TP_DEVICE char* slot(char* smem, int buf, int which) {
  return smem + (buf * 4 + which) * kHalfBytes;
}
TP_DEVICE int swz(int rel) { return rel ^ (((rel >> 9) & 1) << 5); }
}  // namespace c256

template <int DIL>
__launch_bounds__(512, 1)
__global__ void conv256_fwd_kernel(
    const __hip_bfloat16* __restrict__ x,   // NHWC compact (N,Hc,Wc,Cin)
    const __hip_bfloat16* __restrict__ wt,  // (Cout, KH*KW*Cin) tap-major
    __hip_bfloat16* __restrict__ y,         // (M, Cout) row-major
    const float* __restrict__ bias,
    const __hip_bfloat16* __restrict__ zero_page, int Nb, int Hi, int Wi,
    int Cin, int Cout, int Ho, int Wo, int KH, int KW, int stride, int pad,
    int grid_n, int Hc, int Wc, int has_bias) {
  using namespace c256;
  extern __shared__ char smem[];

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  int tile_m = (wg / grid_n) * BM;
  int tile_n = (wg % grid_n) * BN;
  int64_t M = (int64_t)Nb * Ho * Wo;
  int cb_per_tap = Cin / BK;
  int K = KH * KW * Cin;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int wr = wid / WARPS_N, wc = wid % WARPS_N;
  int ahalf = wr;
  int bhalf = wc >> 1;
  int bcol0 = (wc & 1) * 64;

  // stage half `which` of K-tile kt: which 0/1 = A halves (im2col
  // gather), 2/3 = B halves (weight rows). Same lane-linear dest +
  // source-swizzle scheme as gemm256 (verified); only the SOURCE
  // address computation differs.
  auto stage = [&](int kt, int which) {
    int buf = kt & 1;
    char* dst = slot(smem, buf, which);
    int tap = kt / cb_per_tap;
    int dh = tap / KW, dw = tap % KW;
    int cin0 = (kt % cb_per_tap) * BK;
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      int rel = ((g * 8 + wid) * 8) * 128 + lane * 16;
      int lg = swz(rel);
      int row = lg >> 7;        // logical row within the half-tile
      int kb = lg & 127;        // byte within the 64-elem k row
      const char* src;
      if (which < 2) {
        // A: output pixel gather
        int64_t opix = tile_m + which * 128 + row;
        src = reinterpret_cast<const char*>(zero_page);
        if (opix < M) {
          int wo = (int)(opix % Wo);
          int64_t r2 = opix / Wo;
          int ho = (int)(r2 % Ho);
          int n = (int)(r2 / Ho);
          int hi = ho * stride - pad + dh;
          int wi = wo * stride - pad + dw;
          bool ok = hi >= 0 && hi < Hi && wi >= 0 && wi < Wi;
          if (DIL > 1 && ok) {
            if (hi % DIL || wi % DIL) {
              ok = false;
            } else {
              hi /= DIL;
              wi /= DIL;
            }
          }
          if (ok)
            src = reinterpret_cast<const char*>(
                      x + (((int64_t)n * Hc + hi) * Wc + wi) * Cin + cin0) +
                  kb;
        }
      } else {
        // B: weight row (Cout guard -> zero page)
        int brow = tile_n + (which - 2) * 128 + row;
        src = brow < Cout
                  ? reinterpret_cast<const char*>(
                        wt + (int64_t)brow * K + (int64_t)kt * BK) + kb
                  : reinterpret_cast<const char*>(zero_page);
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              dst + ((g * 8 + wid) * 8) * 128),
          16, 0, 0);
    }
  };

  auto read_a = [&](int buf, int mi, int kh) -> bf16x8 {
    int row = mi * 16 + (lane & 15);
    int kf = kh * 32 + (lane >> 4) * 8;
    return *reinterpret_cast<const bf16x8*>(slot(smem, buf, ahalf) +
                                            swz(row * 128 + kf * 2));
  };
  auto read_b = [&](int buf, int ni, int kh) -> bf16x8 {
    int col = bcol0 + ni * 16 + (lane & 15);
    int kf = kh * 32 + (lane >> 4) * 8;
    return *reinterpret_cast<const bf16x8*>(slot(smem, buf, 2 + bhalf) +
                                            swz(col * 128 + kf * 2));
  };

  f32x4 acc[MREP][NREP] = {};
  bf16x8 afr[4][2];
  bf16x8 bfr[4][2];
  int total_kt = K / BK;  // >= 2 (host guarantees)

  // prologue (identical order to gemm256): B0(0) B1(0) A0(0) A1(0)
  // B0(1) B1(1)
  stage(0, 2); stage(0, 3); stage(0, 0); stage(0, 1);
  if (total_kt > 1) { stage(1, 2); stage(1, 3); }
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < total_kt; ++t) {
    int buf = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      if (p == 0) {
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int kh = 0; kh < 2; ++kh) {
            afr[i][kh] = read_a(buf, i, kh);
            bfr[i][kh] = read_b(buf, i, kh);
          }
      } else if (p == 2) {
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int kh = 0; kh < 2; ++kh)
            afr[i][kh] = read_a(buf, 4 + i, kh);
      }
      if (p < 2) {
        if (t + 1 < total_kt) stage(t + 1, p);
      } else {
        if (t + 2 < total_kt) stage(t + 2, p);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
      {
        int msub = (p == 2 || p == 3) ? 1 : 0;
        int nsub = (p == 1 || p == 2) ? 1 : 0;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
#pragma unroll
          for (int ni = 0; ni < 2; ++ni)
#pragma unroll
            for (int kh = 0; kh < 2; ++kh)
              acc[msub * 4 + mi][nsub * 2 + ni] =
                  __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                      afr[mi][kh], bfr[nsub * 2 + ni][kh],
                      acc[msub * 4 + mi][nsub * 2 + ni], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      if (p == 3) {
        if (t + 2 < total_kt)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // epilogue: LDS-staged C (256x256 bf16 fits the dynamic buffer), then
  // guarded 16B stores to the unpadded (M, Cout) output
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  __hip_bfloat16* cs = reinterpret_cast<__hip_bfloat16*>(smem);
#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = wc * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = wr * WM + mi * 16 + (lane >> 4) * 4 + j;
        float v = acc[mi][ni][j];
        if (has_bias && tile_n + col < Cout) v += bias[tile_n + col];
        cs[row * BN + col] = __float2bfloat16(v);
      }
    }
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < BM * (BN / 8); idx += blockDim.x) {
    int r = idx / (BN / 8);
    int cc = (idx % (BN / 8)) * 8;
    int64_t orow = tile_m + r;
    if (orow >= M) continue;
    if (tile_n + cc + 8 <= Cout) {
      *reinterpret_cast<uint4*>(&y[orow * Cout + tile_n + cc]) =
          *reinterpret_cast<const uint4*>(&cs[r * BN + cc]);
    } else {
      for (int e = 0; e < 8; ++e)
        if (tile_n + cc + e < Cout)
          y[orow * Cout + tile_n + cc + e] = cs[r * BN + cc + e];
    }
  }
}

// host: shape-eligibility + launch. Called from conv_implicit.hip's
// dispatchers (not bound directly).
bool conv256_eligible(int Cin, int Cout, int KH, int KW) {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("TURBOPRUNE_CONV256");
    v = (e && e[0] == '1') ? 1 : 0;
  }
  if (v != 1) return false;
  if (Cin % 64) return false;
  int K = KH * KW * Cin;
  // >= 2 K-tiles; Cout >= 192 so the 256-col tile wastes < 25%
  return K >= 2 * c256::BK && Cout >= 192;
}

void conv256_launch(const __hip_bfloat16* x, const __hip_bfloat16* wt,
                    __hip_bfloat16* y, const float* bias,
                    const __hip_bfloat16* zero_page, int Nb, int Hi,
                    int Wi, int Cin, int Cout, int Ho, int Wo, int KH,
                    int KW, int stride, int pad, int Hc, int Wc,
                    bool has_bias, int dil, hipStream_t stream) {
  using namespace c256;
  int64_t M = (int64_t)Nb * Ho * Wo;
  int grid_m = (int)((M + BM - 1) / BM);
  int grid_n = (Cout + BN - 1) / BN;
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute(
        reinterpret_cast<const void*>(conv256_fwd_kernel<1>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
    hipFuncSetAttribute(
        reinterpret_cast<const void*>(conv256_fwd_kernel<2>),
        hipFuncAttributeMaxDynamicSharedMemorySize, 128 * 1024);
    attr_set = true;
  }
  dim3 grid(grid_m * grid_n);
  if (dil == 1)
    hipLaunchKernelGGL((conv256_fwd_kernel<1>), grid, dim3(512),
                       128 * 1024, stream, x, wt, y, bias, zero_page, Nb,
                       Hi, Wi, Cin, Cout, Ho, Wo, KH, KW, stride, pad,
                       grid_n, Hc, Wc, has_bias ? 1 : 0);
  else
    hipLaunchKernelGGL((conv256_fwd_kernel<2>), grid, dim3(512),
                       128 * 1024, stream, x, wt, y, bias, zero_page, Nb,
                       Hi, Wi, Cin, Cout, Ho, Wo, KH, KW, stride, pad,
                       grid_n, Hc, Wc, has_bias ? 1 : 0);
}

}  // namespace turboprune
