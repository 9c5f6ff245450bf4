// MFMA bf16 GEMM for the masked Linear / Conv1d (1x1) path
// (SURVEY K3/K4): y = x @ w^T (+ bias) with w the bf16 masked compute
// weight staged through LDS.
//
// Structure (guide §5: canonical CDNA GEMM, T3 minimal 2-phase recipe):
//   - 128x128 C-tile per 256-thread block (4 waves as 2x2, 64x64/wave),
//     BK = 64, v_mfma_f32_16x16x32_bf16 fragments (4x4 per wave);
//   - global->LDS staging via __builtin_amdgcn_global_load_lds width 16
//     (lane-linear dest), double-buffered, ONE barrier per K-tile;
//   - both operands staged K-major ([row][k] / [col][k]) so every MFMA
//     fragment read is one aligned 16-byte ds_read;
//   - XOR swizzle ((row&7) on the 16B block index) applied on the SOURCE
//     address and the LDS read (guide §5.4 rule 21) to break the
//     stride-128B bank conflict;
//   - XCD-aware bijective blockIdx swizzle (T1) for L2 locality.
//
// Both operands must be K-major: A (M,K) row-major, B (N,K) row-major
// (for a linear layer w IS (N,K) row-major already). M,N multiples of
// 128 and K a multiple of 64 — the host wrapper pads.
//
// Fragment lane maps (v_mfma_f32_16x16x32_bf16, gfx950):
//   A: lane l holds A[l%16][(l/16)*8 + j], j=0..7   (one 16B read)
//   B: lane l holds B[(l/16)*8 + j][l%16]           (staged as [col][k])
//   C: lane l, reg j -> row (l/16)*4 + j, col l%16  (guide §3)

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

at::Tensor gemm_tn_bf16(const at::Tensor&, const at::Tensor&);  // conv_wrw.hip

at::Tensor transpose2d(const at::Tensor&);  // transpose.hip

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WARPS_M = 2, WARPS_N = 2;
constexpr int WM = BM / WARPS_M;  // 64
constexpr int WN = BN / WARPS_N;  // 64
constexpr int MREP = WM / 16, NREP = WN / 16;  // 4 x 4 fragments per wave

// LDS tile: [128 rows][64 k] bf16, row stride 128 B, 16B blocks
// swizzled: physical block b_phys = b_logical ^ (row & 7).
TP_DEVICE int lds_byte(int row, int k) {
  int blk = (k >> 3) ^ (row & 7);
  return row * (BK * 2) + blk * 16 + (k & 7) * 2;
}

// gridDim.y > 1 = split-K: slab blockIdx.y computes K-tiles
// [y*per, min((y+1)*per, K/BK)) into C + y*M*N (fp32, no bias); a reduce
// kernel sums the slabs. Small-output deep-K GEMMs (grad_w: e.g. DeiT
// 1152x384 over K=50k = 27 tiles) otherwise strand 90% of the CUs.
template <typename OutT, bool HAS_BIAS>
__launch_bounds__(256)
__global__ void gemm_bt_kernel(const __hip_bfloat16* __restrict__ A,
                               const __hip_bfloat16* __restrict__ B,
                               OutT* __restrict__ C,
                               const float* __restrict__ bias, int M, int N,
                               int K, int grid_n) {
  __shared__ char smem[2 * 2 * BM * BK * 2];  // 2 buf x (A,B) x 16 KiB
  // LDS layout: [buf][A|B] — byte offsets computed, pointer arrays of
  // LDS create unsupported constant addrspacecasts
  const int kTileBytes = BM * BK * 2;
  auto sA = [&](int buf) -> char* { return smem + buf * 2 * kTileBytes; };
  auto sB = [&](int buf) -> char* {
    return smem + buf * 2 * kTileBytes + kTileBytes;
  };

  // XCD-aware bijective block swizzle (guide T1)
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  int tile_m = (wg / grid_n) * BM;
  int tile_n = (wg % grid_n) * BN;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int wr = wid / WARPS_N, wc = wid % WARPS_N;

  // ---- staging helper: each wave stages 32 rows of A and 32 of B ------
  // per glds: 64 lanes x 16B = 8 rows (8 x 16B blocks per row)
  auto stage = [&](int buf, int kt) {
    const __hip_bfloat16* gA =
        A + (int64_t)(tile_m) * K + (int64_t)kt * BK;
    const __hip_bfloat16* gB =
        B + (int64_t)(tile_n) * K + (int64_t)kt * BK;
    int lrow = lane >> 3;          // 0..7 within the 8-row group
    int lblk = lane & 7;           // 16B block 0..7
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wid * 32 + i * 8 + lrow;
      int src_blk = lblk ^ (row & 7);
      // A row `row`, bytes src_blk*16 .. +16
      const char* srcA = reinterpret_cast<const char*>(gA) +
                         (int64_t)row * K * 2 + src_blk * 16;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)srcA,
          (__attribute__((address_space(3))) void*)(
              sA(buf) + (wid * 32 + i * 8) * (BK * 2)),
          16, 0, 0);
      const char* srcB = reinterpret_cast<const char*>(gB) +
                         (int64_t)row * K * 2 + src_blk * 16;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)srcB,
          (__attribute__((address_space(3))) void*)(
              sB(buf) + (wid * 32 + i * 8) * (BK * 2)),
          16, 0, 0);
    }
  };

  f32x4 acc[MREP][NREP] = {};

  int total_kt = K / BK;
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  C += (int64_t)blockIdx.y * M * N;

  if (kt0 < kt1) stage(0, kt0);  // empty tail slab writes zeros
  __syncthreads();  // carries vmcnt(0): glds drained

  int cur = 0;
  for (int t = kt0; t < kt1; ++t) {
    if (t + 1 < kt1) stage(cur ^ 1, t + 1);

    // compute on buf `cur`: 2 k-steps of 32
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[MREP], b_frag[NREP];
      int kf = ks * 32 + (lane >> 4) * 8;  // this lane's k0
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi) {
        int row = wr * WM + mi * 16 + rowf;
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            sA(cur) + lds_byte(row, kf));
      }
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni) {
        int col = wc * WN + ni * 16 + rowf;
        b_frag[ni] = *reinterpret_cast<const bf16x8*>(
            sB(cur) + lds_byte(col, kf));
      }
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
#pragma unroll
        for (int ni = 0; ni < NREP; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // waves done reading `cur`; next glds drained
    cur ^= 1;
  }

  // ---- epilogue: stage the C tile in LDS, then coalesced 16B row
  // stores (the MFMA C fragment is column-per-lane: direct stores are
  // 2-byte scattered — issue-bound AND line-wasteful, guide T21) -------
  __syncthreads();  // all waves done reading sA/sB
  OutT* cs = reinterpret_cast<OutT*>(smem);  // 128x128 OutT fits 64 KiB
#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = wc * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = wr * WM + mi * 16 + (lane >> 4) * 4 + j;
        float v = acc[mi][ni][j];
        if (HAS_BIAS) v += bias[tile_n + col];
        cs[row * BN + col] = from_float<OutT>(v);
      }
    }
  }
  __syncthreads();
  constexpr int EV = 16 / sizeof(OutT);  // elems per 16B chunk
  int chunks_per_row = BN / EV;
  int total_chunks = BM * chunks_per_row;
  for (int idx = threadIdx.x; idx < total_chunks; idx += blockDim.x) {
    int r = idx / chunks_per_row;
    int cc = (idx % chunks_per_row) * EV;
    *reinterpret_cast<uint4*>(&C[(int64_t)(tile_m + r) * N + tile_n + cc]) =
        *reinterpret_cast<const uint4*>(&cs[r * BN + cc]);
  }
}

// split-K slab reduction: out[i] = sum_s partial[s][i] (+bias) -> OutT
template <typename OutT, bool HAS_BIAS>
__global__ void splitk_reduce_kernel(const float* __restrict__ partial,
                                     OutT* __restrict__ out,
                                     const float* __restrict__ bias,
                                     int64_t mn, int N, int slabs) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < mn;
       i += stride) {
    float v = 0.f;
    for (int s = 0; s < slabs; ++s) v += partial[(int64_t)s * mn + i];
    if (HAS_BIAS) v += bias[i % N];
    out[i] = from_float<OutT>(v);
  }
}

// ---------------------------------------------------------------- host
static at::Tensor pad_to(const at::Tensor& t, int64_t r, int64_t c) {
  if (t.size(0) == r && t.size(1) == c) return t.contiguous();
  auto out = at::zeros({r, c}, t.options());
  out.narrow(0, 0, t.size(0)).narrow(1, 0, t.size(1)).copy_(t);
  return out;
}

static inline int64_t round_up(int64_t v, int64_t m) {
  return (v + m - 1) / m * m;
}

// C(M,N) = A(M,K) @ B(N,K)^T, both K-major bf16. out fp32 when
// out_fp32, else bf16.
at::Tensor gemm_bt(const at::Tensor& A, const at::Tensor& B,
                   const c10::optional<at::Tensor>& bias, bool out_fp32) {
  TORCH_CHECK(A.is_cuda() && B.is_cuda());
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16,
              "gemm_bt: bf16 operands required");
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && A.size(1) == B.size(1),
              "gemm_bt: shape mismatch");
  int64_t M = A.size(0), K = A.size(1), N = B.size(0);
  int64_t Mp = round_up(M, BM), Np = round_up(N, BN), Kp = round_up(K, BK);
  auto Ap = pad_to(A, Mp, Kp);
  auto Bp = pad_to(B, Np, Kp);
  at::Tensor biasp;
  bool has_bias = bias.has_value() && bias->defined();
  if (has_bias) {
    biasp = at::zeros({Np}, bias->options().dtype(at::kFloat));
    biasp.narrow(0, 0, N).copy_(bias->to(at::kFloat));
  }
  auto out = at::empty({Mp, Np},
                       A.options().dtype(out_fp32 ? at::kFloat
                                                  : at::kBFloat16));
  int grid_m = Mp / BM, grid_n = Np / BN;
  int tiles = grid_m * grid_n;
  int total_kt = (int)(Kp / BK);
  // split-K when the tile grid underfills the 256-CU chip and K is deep
  int splitk = 1;
  while (tiles * splitk < 384 && splitk * 2 * 2 <= total_kt &&
         splitk < 32)
    splitk *= 2;
  auto stream = at::hip::getCurrentHIPStream();
  auto* ap = reinterpret_cast<const __hip_bfloat16*>(Ap.data_ptr());
  auto* bp = reinterpret_cast<const __hip_bfloat16*>(Bp.data_ptr());
  const float* biasptr = has_bias ? biasp.data_ptr<float>() : nullptr;

  if (splitk > 1) {
    auto partial = at::empty({splitk, Mp, Np},
                             A.options().dtype(at::kFloat));
    dim3 grid(tiles, splitk);
    hipLaunchKernelGGL((gemm_bt_kernel<float, false>), grid, dim3(256), 0,
                       stream, ap, bp, partial.data_ptr<float>(), nullptr,
                       (int)Mp, (int)Np, (int)Kp, grid_n);
    int64_t mn = Mp * Np;
    int rgrid = elementwise_grid(mn, kBlock, 4);
#define TP_RED(OutT, HB)                                                   \
    hipLaunchKernelGGL((splitk_reduce_kernel<OutT, HB>), dim3(rgrid),      \
                       dim3(kBlock), 0, stream,                            \
                       partial.data_ptr<float>(),                          \
                       reinterpret_cast<OutT*>(out.data_ptr()), biasptr,   \
                       mn, (int)Np, splitk)
    if (out_fp32) {
      if (has_bias) TP_RED(float, true); else TP_RED(float, false);
    } else {
      if (has_bias) TP_RED(__hip_bfloat16, true);
      else TP_RED(__hip_bfloat16, false);
    }
#undef TP_RED
  } else {
    dim3 grid(tiles, 1);
#define TP_GEMM(OutT, HB)                                                  \
    hipLaunchKernelGGL((gemm_bt_kernel<OutT, HB>), grid, dim3(256), 0,     \
                       stream, ap, bp,                                     \
                       reinterpret_cast<OutT*>(out.data_ptr()), biasptr,   \
                       (int)Mp, (int)Np, (int)Kp, grid_n)
    if (out_fp32) {
      if (has_bias) TP_GEMM(float, true); else TP_GEMM(float, false);
    } else {
      if (has_bias) TP_GEMM(__hip_bfloat16, true);
      else TP_GEMM(__hip_bfloat16, false);
    }
#undef TP_GEMM
  }
  if (Mp != M || Np != N)
    return out.narrow(0, 0, M).narrow(1, 0, N).contiguous();
  return out;
}

at::Tensor gemm_bf16(const at::Tensor& A, const at::Tensor& B, bool /*ta*/,
                     bool /*tb*/) {
  return gemm_bt(A, B, c10::nullopt, /*out_fp32=*/false);
}

// Opt-in routing to the EXPERIMENTAL 256^2 8-phase kernel
// (gemm_256_8phase.hip) — promotion path for round 2: flip
// TURBOPRUNE_GEMM256=1 after scripts/validate_gemm256.py passes on
// device. Large bf16-out shapes only; everything else stays on the
// validated 128^2 kernel.
at::Tensor gemm_bt_256(const at::Tensor&, const at::Tensor&,
                       const c10::optional<at::Tensor>&, bool);

static bool use_g256() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("TURBOPRUNE_GEMM256");
    v = (e && e[0] == '1') ? 1 : 0;
  }
  return v == 1;
}

static at::Tensor gemm_route(const at::Tensor& A, const at::Tensor& B,
                             const c10::optional<at::Tensor>& bias,
                             bool out_fp32) {
  if (use_g256() && !out_fp32 && A.size(0) >= 256 && B.size(0) >= 256 &&
      A.size(1) >= 256)
    return gemm_bt_256(A, B, bias, out_fp32);
  return gemm_bt(A, B, bias, out_fp32);
}

bool masked_linear_available(const at::Tensor& x, const at::Tensor& w) {
  if (!x.is_cuda() || x.scalar_type() != at::kBFloat16 ||
      w.scalar_type() != at::kBFloat16)
    return false;
  if (w.dim() != 2) return false;
  int64_t M = 1;
  for (int i = 0; i + 1 < x.dim(); ++i) M *= x.size(i);
  // worth it only when padding overhead is modest
  int64_t K = x.size(-1), N = w.size(0);
  if (K < 64 || N < 64 || M < 16) return false;
  return true;
}

at::Tensor linear_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias) {
  auto sizes = x.sizes().vec();
  int64_t K = sizes.back();
  int64_t M = x.numel() / K;
  auto x2 = x.reshape({M, K}).contiguous();
  auto y = gemm_route(x2, w.contiguous(), bias, /*out_fp32=*/false);
  sizes.back() = w.size(0);
  return y.reshape(sizes);
}

std::tuple<at::Tensor, at::Tensor> linear_bwd(const at::Tensor& grad_y,
                                              const at::Tensor& x,
                                              const at::Tensor& w) {
  auto gy_sizes = grad_y.sizes().vec();
  int64_t N = gy_sizes.back();
  int64_t M = grad_y.numel() / N;
  int64_t K = x.size(-1);
  auto gy2 = grad_y.reshape({M, N}).contiguous();
  auto x2 = x.reshape({M, K}).contiguous();
  // transposes via the tiled LDS transpose kernel (eager .t().contiguous()
  // measured ~1.1 TB/s and dominated the DeiT step)
  // grad_x (M,K) = gy (M,N) @ w (N,K): B_t = w^T (K,N) K-major in N
  auto wT = transpose2d(w.contiguous());
  auto gx = gemm_route(gy2, wT, c10::nullopt, false);
  // grad_w (N,K) = gy^T (N,M) @ x (M,K): the transpose-free TN GEMM
  // (tr_b16 fragments, conv_wrw.hip) replaces the former
  // transpose2d+gemm_bt composition — measured 1.3-1.8x faster on
  // every DeiT training shape (r2t); TURBOPRUNE_TN=0 restores the old
  // path for A/B.
  static int use_tn = -1;
  if (use_tn < 0) {
    const char* e = getenv("TURBOPRUNE_TN");
    use_tn = (e && e[0] == '0') ? 0 : 1;
  }
  auto gw = use_tn
                ? gemm_tn_bf16(gy2, x2)
                : gemm_route(transpose2d(gy2), transpose2d(x2),
                             c10::nullopt, false);
  auto x_sizes = x.sizes().vec();
  return {gx.reshape(x_sizes), gw};
}

}  // namespace turboprune
