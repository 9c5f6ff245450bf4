// EXPERIMENTAL — 256x256 8-phase MFMA GEMM (guide §5 template ladder,
// top rung). NOT wired into any model path yet: sync-structure kernels
// require on-device race screening before use (guide §5 "two-lane
// discipline"); round 2 A/Bs this against gemm_bt and promotes it via
// TURBOPRUNE_GEMM256=1. Exposed as gemm_bt_256 (same contract as
// gemm_bt: C(M,N) = A(M,K) @ B(N,K)^T, K-major bf16 operands).
//
// Structure (vs gemm_bt's 128^2/2-phase, ~900 TF @8k):
//   - 256x256 C tile, 512 threads = 8 waves (2M x 4N), 128x64 per wave,
//     BK=64, acc[8][4] f32x4 (128 VGPR);
//   - LDS 128 KiB dynamic: 8 half-tile slots (A0/A1/B0/B1 x 2 K-tile
//     buffers), each 128 rows x 64 k bf16 (16 KiB);
//   - st_16x32 swizzle: phys = rel ^ (((rel>>9)&1)<<5) within the
//     half-tile image (bank-conflict 8-way -> 4-way on ds_read_b128),
//     applied on the glds SOURCE address and the LDS read address
//     (glds dest is lane-linear; the map is an involution);
//   - K-loop: 2 K-tiles / iteration, 8 phases. Each phase = {ds_read
//     one operand sub-tile} {2 x glds prefetch} {raw s_barrier}
//     {s_waitcnt lgkmcnt(0)} {setprio(1)} {16 x MFMA} {setprio(0)}
//     {raw s_barrier}. Per-wave register residency: all 4 B n-frags
//     (x2 k-halves) live across the tile, A m-sub frags live 2 phases —
//     so each half-tile is ds_read ONCE per tile: phase0 reads A-m0 +
//     B-n0 (12 x ds_read_b128), phase1 B-n1 (4), phase2 A-m1 (8),
//     phase3 none;
//   - counted vmcnt ONLY at tile boundaries (end of phase 3, before
//     the closing barrier). This draft uses vmcnt(4) (2 half-tiles in
//     flight): with the staging rotation below every operand of the
//     next tile is then retired in all waves, and every glds lands in a
//     slot whose last ds_read completed at least a phase earlier:
//       p0 of tile t stages A0(t+1), p1 A1(t+1)  [A slots of the OTHER
//         buffer; A(t-1) was ds_read at p0/p2 of t-1]
//       p2 stages B0(t+2), p3 B1(t+2)            [B slots of THIS
//         buffer; B(t) was fully ds_read at p0, lives in registers]
//     (guide quotes vmcnt(6)/3-deep; revisit on-device — deeper flight
//     needs a staging order this reconstruction could not make safe.)
//
// M,N padded to 256, K to 128 by the host; K must give >= 2 K-tiles
// (host falls back to gemm_bt below that or when out_fp32/split-K
// shapes are requested).

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

at::Tensor gemm_bt(const at::Tensor&, const at::Tensor&,
                   const c10::optional<at::Tensor>&, bool);

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace g256 {
constexpr int BM = 256, BN = 256, BK = 64;
constexpr int WARPS_M = 2, WARPS_N = 4;   // 8 waves
constexpr int WM = 128, WN = 64;          // per-wave C
constexpr int MREP = 8, NREP = 4;
constexpr int kHalfBytes = 128 * BK * 2;  // 16 KiB half-tile image
// slot order in LDS: [buf][A0 A1 B0 B1]
TP_DEVICE char* slot(char* smem, int buf, int which) {
  return smem + (buf * 4 + which) * kHalfBytes;
}
// st_16x32 swizzle inside a half-tile image (rows of 128 B)
TP_DEVICE int swz(int rel) { return rel ^ (((rel >> 9) & 1) << 5); }
}  // namespace g256

__launch_bounds__(512, 1)
__global__ void gemm256_kernel(const __hip_bfloat16* __restrict__ A,
                               const __hip_bfloat16* __restrict__ B,
                               __hip_bfloat16* __restrict__ C,
                               const float* __restrict__ bias, int M, int N,
                               int K, int grid_n, int has_bias) {
  using namespace g256;
  extern __shared__ char smem[];

  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {  // XCD-aware bijective swizzle (guide T1)
    int q = nwg / 8, r = nwg % 8;
    int xcd = wg % 8, idx = wg / 8;
    wg = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  int tile_m = (wg / grid_n) * BM;
  int tile_n = (wg % grid_n) * BN;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;              // 0..7
  int wr = wid / WARPS_N, wc = wid % WARPS_N;
  int ahalf = wr;                             // the A half this wave reads
  int bhalf = wc >> 1;                        // the B half this wave reads
  int bcol0 = (wc & 1) * 64;                  // col offset inside the half

  // ---- staging: one half-tile = 2 collective glds (8 waves x 64 lanes
  // x 16 B = 8 KiB each). Lane-linear dest; source address carries the
  // swizzle: lane L of glds g covers dest rel = (g*64 + wid*8)*128 +
  // L*16; content belongs at logical swz(rel).
  auto stage_half = [&](int buf, int which, const __hip_bfloat16* gbase,
                        int ldk, int kt) {
    char* dst = slot(smem, buf, which);
#pragma unroll
    for (int g = 0; g < 2; ++g) {
      int rel = ((g * 8 + wid) * 8) * 128 + lane * 16;
      int lg = swz(rel);
      int row = lg >> 7;                      // logical row in half-tile
      int kb = lg & 127;                      // byte within the 64-k row
      const char* src = reinterpret_cast<const char*>(gbase) +
                        (int64_t)row * ldk * 2 + (int64_t)kt * BK * 2 + kb;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              dst + ((g * 8 + wid) * 8) * 128),
          16, 0, 0);
    }
  };
  // which -> global row base for this C tile
  const __hip_bfloat16* baseA[2] = {A + (int64_t)tile_m * K,
                                    A + (int64_t)(tile_m + 128) * K};
  const __hip_bfloat16* baseB[2] = {B + (int64_t)tile_n * K,
                                    B + (int64_t)(tile_n + 128) * K};
  auto stage = [&](int t, int which) {  // stage half `which` of K-tile t
    stage_half(t & 1, which,
               which < 2 ? baseA[which] : baseB[which - 2], K, t);
  };

  // ---- fragment reads (one ds_read_b128 each) ------------------------
  auto read_a = [&](int buf, int mi, int kh) -> bf16x8 {
    int row = mi * 16 + (lane & 15);
    int kf = kh * 32 + (lane >> 4) * 8;
    int rel = row * 128 + kf * 2;
    return *reinterpret_cast<const bf16x8*>(slot(smem, buf, ahalf) +
                                            swz(rel));
  };
  auto read_b = [&](int buf, int ni, int kh) -> bf16x8 {
    int col = bcol0 + ni * 16 + (lane & 15);
    int kf = kh * 32 + (lane >> 4) * 8;
    int rel = col * 128 + kf * 2;
    return *reinterpret_cast<const bf16x8*>(slot(smem, buf, 2 + bhalf) +
                                            swz(rel));
  };

  f32x4 acc[MREP][NREP] = {};
  bf16x8 afr[4][2];   // current m-sub
  bf16x8 bfr[4][2];   // ALL n-frags of this wave's B half (resident)

  int total_kt = K / BK;  // even, >= 4 (host guarantees)

  // ---- prologue: B0(0) B1(0) A0(0) A1(0) B0(1) B1(1); tile 0's A/B all
  // among the first 4 stagings -> retired by vmcnt(4) below
  stage(0, 2); stage(0, 3); stage(0, 0); stage(0, 1);
  stage(1, 2); stage(1, 3);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < total_kt; ++t) {
    int buf = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      // -- ds_read this phase's new operand sub-tile (safe: the
      //    previous tile ended with [vmcnt(4); barrier], so every
      //    operand of tile t is retired in all waves before any p0 read)
      if (p == 0) {
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int kh = 0; kh < 2; ++kh) {
            afr[i][kh] = read_a(buf, i, kh);      // m-sub 0
            bfr[i][kh] = read_b(buf, i, kh);      // all 4 resident n-frags
          }
      } else if (p == 2) {
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int kh = 0; kh < 2; ++kh)
            afr[i][kh] = read_a(buf, 4 + i, kh);  // m-sub 1
      }
      // -- glds prefetch rotation (see header): 1 half-tile / phase.
      //    A halves ONE tile ahead (other buffer, last read p0 of t-1);
      //    B halves TWO tiles ahead (this buffer, fully read at p0).
      if (p < 2) {
        if (t + 1 < total_kt) stage(t + 1, p);    // p0:A0' p1:A1'
      } else {
        if (t + 2 < total_kt) stage(t + 2, p);    // p2:B0'' p3:B1''
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
      // -- 16 MFMA: quadrant (msub = p>>1, nsub = p0,p3->0? gray:
      //    p0:(m0,n0) p1:(m0,n1) p2:(m1,n1) p3:(m1,n0)
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
      // -- tile-boundary retire: all but the 2 newest half-tile
      //    stagings complete in this wave; the barrier makes that hold
      //    across waves before the next tile's p0 ds_reads. TAIL: when
      //    this tile staged no B half-tiles (t+2 past the end), the
      //    newest 2 loads ARE the next tile's A halves — drain fully
      //    (caught by tests/test_gemm256_schedule.py's adversarial
      //    timing model).
      if (p == 3) {
        if (t + 2 < total_kt)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue: stage C through LDS (256x256 bf16 = 128 KiB fits the
  // dynamic buffer), then coalesced 16B row stores (guide T21)
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
        if (has_bias) v += bias[tile_n + col];
        cs[row * BN + col] = __float2bfloat16(v);
      }
    }
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < BM * (BN / 8); idx += blockDim.x) {
    int r = idx / (BN / 8);
    int cc = (idx % (BN / 8)) * 8;
    *reinterpret_cast<uint4*>(&C[(int64_t)(tile_m + r) * N + tile_n + cc]) =
        *reinterpret_cast<const uint4*>(&cs[r * BN + cc]);
  }
}

// ---------------------------------------------------------------- host
static at::Tensor pad256(const at::Tensor& t, int64_t r, int64_t c) {
  if (t.size(0) == r && t.size(1) == c) return t.contiguous();
  auto out = at::zeros({r, c}, t.options());
  out.narrow(0, 0, t.size(0)).narrow(1, 0, t.size(1)).copy_(t);
  return out;
}

at::Tensor gemm_bt_256(const at::Tensor& A, const at::Tensor& B,
                       const c10::optional<at::Tensor>& bias,
                       bool out_fp32) {
  using namespace g256;
  TORCH_CHECK(A.is_cuda() && B.is_cuda());
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16, "gemm_bt_256: bf16 only");
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && A.size(1) == B.size(1));
  int64_t M = A.size(0), K = A.size(1), N = B.size(0);
  int64_t Kp = (K + 127) / 128 * 128;
  if (out_fp32 || Kp < 4 * BK) return gemm_bt(A, B, bias, out_fp32);
  int64_t Mp = (M + BM - 1) / BM * BM, Np = (N + BN - 1) / BN * BN;
  auto Ap = pad256(A, Mp, Kp);
  auto Bp = pad256(B, Np, Kp);
  at::Tensor biasp;
  bool has_bias = bias.has_value() && bias->defined();
  if (has_bias) {
    biasp = at::zeros({Np}, bias->options().dtype(at::kFloat));
    biasp.narrow(0, 0, N).copy_(bias->to(at::kFloat));
  }
  auto Cp = at::empty({Mp, Np}, A.options());
  int grid_n = (int)(Np / BN);
  int tiles = (int)(Mp / BM) * grid_n;
  static bool attr_set = false;
  if (!attr_set) {
    hipFuncSetAttribute(reinterpret_cast<const void*>(gemm256_kernel),
                        hipFuncAttributeMaxDynamicSharedMemorySize,
                        128 * 1024);
    attr_set = true;
  }
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(gemm256_kernel, dim3(tiles), dim3(512), 128 * 1024,
                     stream,
                     reinterpret_cast<const __hip_bfloat16*>(Ap.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(Bp.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(Cp.data_ptr()),
                     has_bias ? biasp.data_ptr<float>() : nullptr, (int)Mp,
                     (int)Np, (int)Kp, grid_n, has_bias ? 1 : 0);
  if (Mp != M || Np != N) return Cp.narrow(0, 0, M).narrow(1, 0, N);
  return Cp;
}

}  // namespace turboprune
