// Implicit-GEMM NHWC conv FORWARD on MFMA (experimental):
// y[n,ho,wo,co] = sum_{dh,dw,ci} x[n, ho*s-p+dh, wo*s-p+dw, ci] * w[co,ci,dh,dw]
//
// Same structure as gemm_bt_kernel (128x128 C-tile, BK=64,
// v_mfma_f32_16x16x32_bf16, glds staging, XOR swizzle, LDS-staged
// epilogue) with the A operand staged as im2col ON THE FLY: each glds
// lane's source address is the (output-pixel, kernel-tap, channel-block)
// it covers; out-of-bounds taps (padding) and row tails point at a
// 16-byte ZERO PAGE instead — glds has no predication, but per-lane
// addresses are free.
//
// K-order is (dh, dw, ci) with ci fastest, which is exactly the memory
// order of a channels_last weight viewed as (Cout, KH*KW*Cin) — the bf16
// masked-weight cache feeds B with zero copies.
//
// Requirements: Cin % 64 == 0 (so a BK=64 tile stays inside one tap),
// NHWC bf16. Round-1 scope: forward only, correctness + measurement
// (dispatch stays on MIOpen until fwd+bwd beat it per shape).

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

// 8-phase 256x256 variant (conv_implicit_256.hip, TURBOPRUNE_CONV256=1)
bool conv256_eligible(int Cin, int Cout, int KH, int KW);
void conv256_launch(const __hip_bfloat16*, const __hip_bfloat16*,
                    __hip_bfloat16*, const float*, const __hip_bfloat16*,
                    int, int, int, int, int, int, int, int, int, int,
                    int, int, int, bool, int, hipStream_t);

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace conv_ig {
constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WARPS_N = 2;
constexpr int WM = 64, WN = 64;
constexpr int MREP = 4, NREP = 4;

TP_DEVICE int lds_byte(int row, int k) {
  int blk = (k >> 3) ^ (row & 7);
  return row * (BK * 2) + blk * 16 + (k & 7) * 2;
}
}  // namespace conv_ig

// DIL: zero-insertion factor of the LOGICAL input image (fractional
// stride). DIL=1 is the plain forward. For DIL>1 the input bounds
// (Hi, Wi) are the DILATED dims; (Hc, Wc) are the compact tensor's dims
// and only coordinates divisible by DIL hold data — everything else
// gathers the zero page. This runs grad_input of a stride-DIL conv as
// one fused forward conv of the output gradient with the
// rotated-transposed weight (ops/conv_backward.py math), with no
// materialized dilated tensor.
// BN_T: the Cout-tile width. 128 is the default; the 64-wide variant
// removes the half-empty N-tiles of Cout=64 layers (layer1 3x3 ran at
// 554 us vs MIOpen 375 with BN=128 purely from tile padding waste).
template <typename OutT, bool HAS_BIAS, int DIL = 1, int BN_T = 128>
__launch_bounds__(256)
__global__ void conv3x3_fwd_kernel(
    const __hip_bfloat16* __restrict__ x,   // NHWC (N,Hc,Wc,Cin) compact
    const __hip_bfloat16* __restrict__ wt,  // (Cout, KH*KW*Cin) tap-major
    OutT* __restrict__ y,                   // (M, Cout) row-major
    const float* __restrict__ bias,
    const __hip_bfloat16* __restrict__ zero_page, int Nb, int Hi, int Wi,
    int Cin, int Cout, int Ho, int Wo, int KH, int KW, int stride, int pad,
    int grid_n, int Hc, int Wc) {
  using namespace conv_ig;
  constexpr int WN_T = BN_T / WARPS_N;
  constexpr int NREP_T = WN_T / 16;
  __shared__ char smem[2 * 2 * BM * BK * 2];
  const int kTileBytes = BM * BK * 2;
  auto sA = [&](int buf) -> char* { return smem + buf * 2 * kTileBytes; };
  auto sB = [&](int buf) -> char* {
    return smem + buf * 2 * kTileBytes + kTileBytes;
  };

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  int tile_m = (wg / grid_n) * BM;
  int tile_n = (wg % grid_n) * BN_T;
  int64_t M = (int64_t)Nb * Ho * Wo;
  int cb_per_tap = Cin / BK;  // K-tiles per kernel tap

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int wr = wid / WARPS_N, wc = wid % WARPS_N;

  auto stage = [&](int buf, int kt) {
    int tap = kt / cb_per_tap;
    int dh = tap / KW, dw = tap % KW;
    int cin0_base = (kt % cb_per_tap) * BK;
    int lrow = lane >> 3;
    int lblk = lane & 7;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wid * 32 + i * 8 + lrow;
      int src_blk = lblk ^ (row & 7);
      // --- A: im2col gather -----------------------------------------
      int64_t opix = tile_m + row;
      const char* srcA;
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
            ok = false;  // zero-inserted coordinate
          } else {
            hi /= DIL;
            wi /= DIL;
          }
        }
        if (ok) {
          int cin = cin0_base + src_blk * 8;
          srcA = reinterpret_cast<const char*>(
                     x + (((int64_t)n * Hc + hi) * Wc + wi) * Cin + cin);
        } else {
          srcA = reinterpret_cast<const char*>(zero_page);
        }
      } else {
        srcA = reinterpret_cast<const char*>(zero_page);
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)srcA,
          (__attribute__((address_space(3))) void*)(
              sA(buf) + (wid * 32 + i * 8) * (BK * 2)),
          16, 0, 0);
      // --- B: weight rows (tap-major K) ------------------------------
      if (row < BN_T) {
        int brow = tile_n + row;
        const char* srcB =
            brow < Cout
                ? reinterpret_cast<const char*>(
                      wt + (int64_t)brow * (KH * KW * Cin) +
                      (int64_t)kt * BK) + src_blk * 16
                : reinterpret_cast<const char*>(zero_page);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)srcB,
            (__attribute__((address_space(3))) void*)(
                sB(buf) + (wid * 32 + i * 8) * (BK * 2)),
            16, 0, 0);
      }
    }
  };

  f32x4 acc[MREP][NREP_T] = {};
  int nt = (KH * KW * Cin) / BK;
  stage(0, 0);
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < nt; ++t) {
    if (t + 1 < nt) stage(cur ^ 1, t + 1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[MREP], b_frag[NREP_T];
      int kf = ks * 32 + (lane >> 4) * 8;
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            sA(cur) + lds_byte(wr * WM + mi * 16 + rowf, kf));
#pragma unroll
      for (int ni = 0; ni < NREP_T; ++ni)
        b_frag[ni] = *reinterpret_cast<const bf16x8*>(
            sB(cur) + lds_byte(wc * WN_T + ni * 16 + rowf, kf));
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
#pragma unroll
        for (int ni = 0; ni < NREP_T; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
    cur ^= 1;
  }

  // LDS-staged epilogue with row/col guards (output is UNPADDED (M,Cout))
  __syncthreads();
  OutT* cs = reinterpret_cast<OutT*>(smem);
#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP_T; ++ni) {
      int col = wc * WN_T + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = wr * WM + mi * 16 + (lane >> 4) * 4 + j;
        float v = acc[mi][ni][j];
        if (HAS_BIAS && tile_n + col < Cout) v += bias[tile_n + col];
        cs[row * BN_T + col] = from_float<OutT>(v);
      }
    }
  }
  __syncthreads();
  constexpr int EV = 16 / sizeof(OutT);
  int chunks_per_row = BN_T / EV;
  for (int idx = threadIdx.x; idx < BM * chunks_per_row;
       idx += blockDim.x) {
    int r = idx / chunks_per_row;
    int cc = (idx % chunks_per_row) * EV;
    int64_t orow = tile_m + r;
    if (orow >= M) continue;
    if (tile_n + cc + EV <= Cout) {
      *reinterpret_cast<uint4*>(&y[orow * Cout + tile_n + cc]) =
          *reinterpret_cast<const uint4*>(&cs[r * BN_T + cc]);
    } else {
      for (int e = 0; e < EV; ++e)
        if (tile_n + cc + e < Cout)
          y[orow * Cout + tile_n + cc + e] = cs[r * BN_T + cc + e];
    }
  }
}

// x: channels_last NCHW logical (N,Cin,Hi,Wi); weight: channels_last
// (Cout,Cin,KH,KW) (i.e. the bf16 masked cache); returns channels_last y.
at::Tensor conv2d_implicit_fwd(const at::Tensor& x_in,
                               const at::Tensor& weight,
                               const c10::optional<at::Tensor>& bias,
                               int64_t stride, int64_t pad) {
  TORCH_CHECK(x_in.is_cuda() && x_in.scalar_type() == at::kBFloat16);
  TORCH_CHECK(weight.scalar_type() == at::kBFloat16);
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto w = weight.contiguous(at::MemoryFormat::ChannelsLast);
  int Nb = x.size(0), Cin = x.size(1), Hi = x.size(2), Wi = x.size(3);
  int Cout = w.size(0), KH = w.size(2), KW = w.size(3);
  TORCH_CHECK(Cin % 64 == 0, "conv_implicit: Cin % 64 != 0");
  int Ho = (Hi + 2 * (int)pad - KH) / (int)stride + 1;
  int Wo = (Wi + 2 * (int)pad - KW) / (int)stride + 1;
  int64_t M = (int64_t)Nb * Ho * Wo;

  auto y = at::empty({Nb, Cout, Ho, Wo},
                     x.options().memory_format(
                         at::MemoryFormat::ChannelsLast));
  static at::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = at::zeros({64}, x.options());  // 128 B of bf16 zeros

  at::Tensor biasp;
  bool has_bias = bias.has_value() && bias->defined();
  if (has_bias) biasp = bias->contiguous().to(at::kFloat);

  bool narrow = (Cout == 64);  // BN_T=64: no half-empty N-tiles
  int bn = narrow ? 64 : conv_ig::BN;
  int grid_m = (int)((M + conv_ig::BM - 1) / conv_ig::BM);
  int grid_n = (Cout + bn - 1) / bn;
  dim3 grid(grid_m * grid_n);
  auto stream = at::hip::getCurrentHIPStream();
  auto* xp = reinterpret_cast<const __hip_bfloat16*>(x.data_ptr());
  // channels_last weight memory IS (Cout, KH*KW*Cin) tap-major
  auto* wp = reinterpret_cast<const __hip_bfloat16*>(w.data_ptr());
  auto* yp = reinterpret_cast<__hip_bfloat16*>(y.data_ptr());
  auto* zp = reinterpret_cast<const __hip_bfloat16*>(zero_page.data_ptr());

  if (conv256_eligible(Cin, Cout, KH, KW)) {
    conv256_launch(xp, wp, yp,
                   has_bias ? biasp.data_ptr<float>() : nullptr, zp, Nb,
                   Hi, Wi, Cin, Cout, Ho, Wo, KH, KW, (int)stride,
                   (int)pad, Hi, Wi, has_bias, 1, stream);
    return y;
  }
#define TP_FWD_LAUNCH(BIAS, BNT, BPTR)                                   \
  hipLaunchKernelGGL((conv3x3_fwd_kernel<__hip_bfloat16, BIAS, 1, BNT>), \
                     grid, dim3(256), 0, stream, xp, wp, yp, BPTR, zp,   \
                     Nb, Hi, Wi, Cin, Cout, Ho, Wo, KH, KW, (int)stride, \
                     (int)pad, grid_n, Hi, Wi)
  if (has_bias) {
    if (narrow)
      TP_FWD_LAUNCH(true, 64, biasp.data_ptr<float>());
    else
      TP_FWD_LAUNCH(true, 128, biasp.data_ptr<float>());
  } else {
    if (narrow)
      TP_FWD_LAUNCH(false, 64, nullptr);
    else
      TP_FWD_LAUNCH(false, 128, nullptr);
  }
#undef TP_FWD_LAUNCH
  return y;
}

// grad_input of conv(x, w, stride, pad) as ONE fused kernel:
//   gx = conv( dilate_stride(gy), rot180(w)^T, 1, k-1-pad )
// with the zero-inserted gy coordinates resolved inside the im2col
// gather (no materialized dilated tensor; ops/conv_backward.py is the
// CPU-tested math oracle). gy: (N, Cout, Ho, Wo) channels_last bf16;
// w: the ORIGINAL (Cout, Cin, KH, KW) channels_last weight.
// Requires Cout % 64 == 0 (gy's channels are the contraction).
at::Tensor conv2d_implicit_gradin(const at::Tensor& gy_in,
                                  const at::Tensor& weight, int64_t Hi_l,
                                  int64_t Wi_l, int64_t stride,
                                  int64_t pad) {
  TORCH_CHECK(gy_in.is_cuda() && gy_in.scalar_type() == at::kBFloat16);
  auto gy = gy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto w = weight.contiguous(at::MemoryFormat::ChannelsLast);
  int Nb = gy.size(0), Cout = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  int Cin = w.size(1), KH = w.size(2), KW = w.size(3);
  TORCH_CHECK(Cout % 64 == 0, "conv_gradin: Cout % 64 != 0");
  TORCH_CHECK(KH == KW && stride >= 1);
  // rotated-transposed weight, channels_last => (Cin, KH*KW*Cout) tap-major
  auto w_rt = at::flip(w, {2, 3}).permute({1, 0, 2, 3})
                  .contiguous(at::MemoryFormat::ChannelsLast);
  int Hi = (int)Hi_l, Wi = (int)Wi_l;
  int Hd = Hi + 2 * (int)pad - KH + 1;  // dilated image dims
  int Wd = Wi + 2 * (int)pad - KW + 1;
  TORCH_CHECK((Ho - 1) * stride + 1 <= Hd && (Wo - 1) * stride + 1 <= Wd,
              "conv_gradin: inconsistent geometry");
  int64_t M = (int64_t)Nb * Hi * Wi;

  auto gx = at::empty({Nb, Cin, Hi, Wi},
                      gy.options().memory_format(
                          at::MemoryFormat::ChannelsLast));
  static at::Tensor zero_page2;
  if (!zero_page2.defined() || zero_page2.device() != gy.device())
    zero_page2 = at::zeros({64}, gy.options());

  bool narrow = (Cin == 64);  // output-channel role
  int bn = narrow ? 64 : conv_ig::BN;
  int grid_m = (int)((M + conv_ig::BM - 1) / conv_ig::BM);
  int grid_n = (Cin + bn - 1) / bn;
  dim3 grid(grid_m * grid_n);
  auto stream = at::hip::getCurrentHIPStream();
  auto* gp = reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr());
  auto* wp = reinterpret_cast<const __hip_bfloat16*>(w_rt.data_ptr());
  auto* xp = reinterpret_cast<__hip_bfloat16*>(gx.data_ptr());
  auto* zp = reinterpret_cast<const __hip_bfloat16*>(zero_page2.data_ptr());
  int new_pad = KH - 1 - (int)pad;
  // roles: "x" = gy (dilated bounds Hd,Wd; compact Ho,Wo), "Cin" = Cout,
  // "Cout" = Cin, output pixels = Hi x Wi, stride 1.
  if (stride <= 2 && conv256_eligible(Cout, Cin, KH, KW)) {
    conv256_launch(gp, wp, xp, nullptr, zp, Nb, Hd, Wd, Cout, Cin, Hi,
                   Wi, KH, KW, 1, new_pad, Ho, Wo, false, (int)stride,
                   stream);
    return gx;
  }
#define TP_GI_LAUNCH(DILV, BNT)                                          \
  hipLaunchKernelGGL(                                                    \
      (conv3x3_fwd_kernel<__hip_bfloat16, false, DILV, BNT>), grid,      \
      dim3(256), 0, stream, gp, wp, xp, nullptr, zp, Nb, Hd, Wd, Cout,   \
      Cin, Hi, Wi, KH, KW, 1, new_pad, grid_n, Ho, Wo)
  if (stride == 1) {
    if (narrow)
      TP_GI_LAUNCH(1, 64);
    else
      TP_GI_LAUNCH(1, 128);
  } else if (stride == 2) {
    if (narrow)
      TP_GI_LAUNCH(2, 64);
    else
      TP_GI_LAUNCH(2, 128);
  } else {
    TORCH_CHECK(false, "conv_gradin: stride > 2 unsupported");
  }
#undef TP_GI_LAUNCH
  return gx;
}

}  // namespace turboprune
