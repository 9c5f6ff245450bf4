// Implicit-GEMM conv WEIGHT gradient (wrw) on MFMA.
//
//   gw[co][dh,dw,ci] = sum_opix gy[opix][co] * x[n, ho*s-p+dh, wo*s-p+dw, ci]
//
// A TN GEMM: both operands are M-major over output pixels (the deep
// contraction, N*Ho*Wo up to ~1.6M) — the transposed-staging problem
// that drove a seven-generation engineering campaign, fully measured
// in profiles/r02_summary.md:
//
//   v1  128x128x64 register staging, ds_write_b16 scatter (shipped r01)
//   v2  BK=128, b128-grouped register transpose, 1024-workgroup splitK
//   v3  wide-N 64x576 tiles (operands read ~once), 32-bit coords
//   v4  global_load_lds raw staging + in-LDS transpose
//   v5  persistent ring window of x rows (opt-in, TURBOPRUNE_WRW=5)
//   two-stage slab-parallel splitK reduce (hidden 450 us at splitk=512)
//   v6  ds_read_b64_tr_b16 hardware-transpose fragments straight from
//       the glds-blocked raw image (DEFAULT for 3x3): no transpose
//       pass, double-buffered raw, one barrier/tile, pipelined counted
//       lgkm waits, incremental 32-bit corner offsets. Layer1
//       (64,64,3,1): 309 us vs MIOpen 477 — dispatched native.
//   v7  ring window + window-direct tr fragments (opt-in, =7)
//
// All generations remain selectable via TURBOPRUNE_WRW for A/B.
// Split-K over opix chunks (gridDim.y slabs -> fp32 partials -> the
// two-stage reduce), since the output (Cout x 9Cin) is tiny.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include <algorithm>

#include "common.h"

namespace turboprune {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace conv_wrw {
constexpr int BM = 128, BN = 128, BK = 64;
constexpr int WARPS_N = 2;
constexpr int WM = 64, WN = 64;
constexpr int MREP = 4, NREP = 4;

TP_DEVICE int lds_byte(int row, int k) {
  int blk = (k >> 3) ^ (row & 7);
  return row * (BK * 2) + blk * 16 + (k & 7) * 2;
}
}  // namespace conv_wrw

// one slab computes K-tiles [kt0, kt1) of the opix axis into
// partial[blockIdx.y * (Mp*Np) + ...] (fp32)
__global__ __launch_bounds__(256) void conv_wrw_kernel(
    const __hip_bfloat16* __restrict__ gy,  // (M, Cout) row-major
    const __hip_bfloat16* __restrict__ x,   // NHWC
    float* __restrict__ partial,            // (slabs, Mp, Np)
    int Nb, int Hi, int Wi, int Cin, int Cout, int Ho, int Wo, int KH,
    int KW, int stride, int pad, int Mp, int Np, int grid_n) {
  using namespace conv_wrw;
  // LDS: 2 x (A,B) tiles of 128x64 bf16 = 64 KiB (single-buffered pairs,
  // one barrier per K-tile; staging is register->ds_write)
  __shared__ char smem[2 * BM * BK * 2];
  char* sA = smem;                     // [co][opix]
  char* sB = smem + BM * BK * 2;       // [tapci][opix]

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;     // co tile
  int tile_n = (wg % grid_n) * BN;     // tapci tile
  int64_t M = (int64_t)Nb * Ho * Wo;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int wr = wid / WARPS_N, wc = wid % WARPS_N;

  int total_kt = (int)((M + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  f32x4 acc[MREP][NREP] = {};

  for (int kt = kt0; kt < kt1; ++kt) {
    int64_t opix0 = (int64_t)kt * BK;
    // ---- stage A: gy^T tile [128 co][64 opix] -------------------------
    // 256 threads x 2 iters: each thread loads 8 co (16B) of one opix,
    // then scatters 8 ds_write_b16 into [co][opix].
    {
      int op = threadIdx.x & 63;        // opix within tile
      int co8 = threadIdx.x >> 6;       // 0..3
      // 4 iterations x 4 co8 = 16 co-octets = all 128 rows
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int co0 = (co8 + it * 4) * 8;
        int64_t opix = opix0 + op;
        __hip_bfloat16 vals[8];
        if (opix < M && tile_m + co0 < Cout) {
          const __hip_bfloat16* src = gy + opix * Cout + tile_m + co0;
          *reinterpret_cast<uint4*>(vals) =
              *reinterpret_cast<const uint4*>(src);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) vals[j] = __float2bfloat16(0.f);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int row = co0 + j;
          *reinterpret_cast<__hip_bfloat16*>(sA + lds_byte(row, op)) =
              vals[j];
        }
      }
    }
    // ---- stage B: im2col^T tile [128 tapci][64 opix] ------------------
    {
      int op = threadIdx.x & 63;
      int tc8 = threadIdx.x >> 6;
#pragma unroll
      for (int it = 0; it < 4; ++it) {
        int tc0 = (tc8 + it * 4) * 8;   // tapci octet start
        int64_t opix = opix0 + op;
        __hip_bfloat16 vals[8];
        bool ok = false;
        if (opix < M && tile_n + tc0 < KH * KW * Cin) {
          int tapci = tile_n + tc0;
          int tap = tapci / Cin;        // Cin % 8 == 0 keeps octet in tap
          int ci = tapci % Cin;
          int dh = tap / KW, dw = tap % KW;
          int wo = (int)(opix % Wo);
          int64_t r2 = opix / Wo;
          int ho = (int)(r2 % Ho);
          int n = (int)(r2 / Ho);
          int hi = ho * stride - pad + dh;
          int wi = wo * stride - pad + dw;
          if (hi >= 0 && hi < Hi && wi >= 0 && wi < Wi) {
            const __hip_bfloat16* src =
                x + (((int64_t)n * Hi + hi) * Wi + wi) * Cin + ci;
            *reinterpret_cast<uint4*>(vals) =
                *reinterpret_cast<const uint4*>(src);
            ok = true;
          }
        }
        if (!ok) {
#pragma unroll
          for (int j = 0; j < 8; ++j) vals[j] = __float2bfloat16(0.f);
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          *reinterpret_cast<__hip_bfloat16*>(sB + lds_byte(tc0 + j, op)) =
              vals[j];
        }
      }
    }
    __syncthreads();
    // ---- MFMA over the 64-opix tile ----------------------------------
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[MREP], b_frag[NREP];
      int kf = ks * 32 + (lane >> 4) * 8;
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            sA + lds_byte(wr * WM + mi * 16 + rowf, kf));
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni)
        b_frag[ni] = *reinterpret_cast<const bf16x8*>(
            sB + lds_byte(wc * WN + ni * 16 + rowf, kf));
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
#pragma unroll
        for (int ni = 0; ni < NREP; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  // write the fp32 slab (direct; Mp/Np padded so no guards)
#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wc * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + wr * WM + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

// Double-buffered variant (TURBOPRUNE_WRW_DB=1, round-2 A/B): tile
// t+1's global loads issue before tile t's MFMAs (latency hidden under
// compute), the ds_write transpose lands in the OTHER buffer, one
// barrier per tile. The single-buffered kernel above stages
// synchronously — the guide's slowest staging regime. Same addresses
// and math; only buffering/order differ. 64 KiB dynamic LDS (2 x A+B).
__global__ __launch_bounds__(256) void conv_wrw_db_kernel(
    const __hip_bfloat16* __restrict__ gy, const __hip_bfloat16* __restrict__ x,
    float* __restrict__ partial, int Nb, int Hi, int Wi, int Cin, int Cout,
    int Ho, int Wo, int KH, int KW, int stride, int pad, int Mp, int Np,
    int grid_n) {
  using namespace conv_wrw;
  extern __shared__ char smem[];
  const int kPair = 2 * BM * BK * 2;  // A+B images of one buffer
  auto sA = [&](int buf) -> char* { return smem + buf * kPair; };
  auto sB = [&](int buf) -> char* { return smem + buf * kPair + BM * BK * 2; };

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;
  int tile_n = (wg % grid_n) * BN;
  int64_t M = (int64_t)Nb * Ho * Wo;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int wr = wid / WARPS_N, wc = wid % WARPS_N;

  int total_kt = (int)((M + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  int op = threadIdx.x & 63;
  int g8 = threadIdx.x >> 6;  // 0..3

  // registers for one staged tile (A: gy^T rows, B: im2col^T rows)
  __hip_bfloat16 va[4][8], vb[4][8];

  auto load_tile = [&](int kt) {
    int64_t opix0 = (int64_t)kt * BK;
    int64_t opix = opix0 + op;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int co0 = (g8 + it * 4) * 8;
      if (opix < M && tile_m + co0 < Cout) {
        *reinterpret_cast<uint4*>(va[it]) = *reinterpret_cast<const uint4*>(
            gy + opix * Cout + tile_m + co0);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) va[it][j] = __float2bfloat16(0.f);
      }
      int tc0 = (g8 + it * 4) * 8;
      bool ok = false;
      if (opix < M && tile_n + tc0 < KH * KW * Cin) {
        int tapci = tile_n + tc0;
        int tap = tapci / Cin;
        int ci = tapci % Cin;
        int dh = tap / KW, dw = tap % KW;
        int wo = (int)(opix % Wo);
        int64_t r2 = opix / Wo;
        int ho = (int)(r2 % Ho);
        int n = (int)(r2 / Ho);
        int hi = ho * stride - pad + dh;
        int wi = wo * stride - pad + dw;
        if (hi >= 0 && hi < Hi && wi >= 0 && wi < Wi) {
          *reinterpret_cast<uint4*>(vb[it]) =
              *reinterpret_cast<const uint4*>(
                  x + (((int64_t)n * Hi + hi) * Wi + wi) * Cin + ci);
          ok = true;
        }
      }
      if (!ok) {
#pragma unroll
        for (int j = 0; j < 8; ++j) vb[it][j] = __float2bfloat16(0.f);
      }
    }
  };
  auto write_tile = [&](int buf) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      int r0 = (g8 + it * 4) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *reinterpret_cast<__hip_bfloat16*>(sA(buf) + lds_byte(r0 + j, op)) =
            va[it][j];
        *reinterpret_cast<__hip_bfloat16*>(sB(buf) + lds_byte(r0 + j, op)) =
            vb[it][j];
      }
    }
  };

  f32x4 acc[MREP][NREP] = {};
  if (kt0 < kt1) {
    load_tile(kt0);
    write_tile(0);
  }
  __syncthreads();

  int cur = 0;
  for (int t = kt0; t < kt1; ++t) {
    if (t + 1 < kt1) load_tile(t + 1);  // global loads in flight
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a_frag[MREP], b_frag[NREP];
      int kf = ks * 32 + (lane >> 4) * 8;
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            sA(cur) + lds_byte(wr * WM + mi * 16 + rowf, kf));
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni)
        b_frag[ni] = *reinterpret_cast<const bf16x8*>(
            sB(cur) + lds_byte(wc * WN + ni * 16 + rowf, kf));
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
#pragma unroll
        for (int ni = 0; ni < NREP; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    if (t + 1 < kt1) write_tile(cur ^ 1);  // other buffer: no race with
                                           // this tile's reads
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wc * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + wr * WM + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

// ---- v2 (default): BK=128 opix tiles, b128-grouped LDS staging -------
//
// Why v1 lost to MIOpen 3x at bs512 (profiles/r02_conv_dispatch.md):
//   - 32 scalar ds_write_b16 per thread per operand per 64-opix tile —
//     the slowest staging regime (guide §5.4);
//   - 16 div/mod opix decompositions per thread per tile;
//   - split-K targeted only 384 workgroups (256-CU chip wants >>256);
//   - 2 barriers per 64-opix tile.
// v2: each thread loads an 8-opix x 8-row patch (8 x uint4), transposes
// in registers, and writes 8 x ds_write_b128 (one full 16B LDS block
// each — 8x fewer LDS instructions per element than v1); the opix
// decomposition is done once per tile and carried incrementally
// (no div/mod per element); BK=128 halves the barrier count; global
// loads for tile t+1 issue before tile t's MFMAs (register prefetch,
// single 64 KiB LDS buffer keeps 2 blocks/CU).
namespace conv_wrw2 {
constexpr int BM = 128, BN = 128, BK = 128;
constexpr int WARPS_N = 2;
constexpr int WM = 64, WN = 64;
constexpr int MREP = 4, NREP = 4;
// [row][k] bf16 image, 256 B rows, XOR-swizzled 16 B blocks.
// The 256 B row stride aliases the 64 LDS banks exactly, so the
// swizzle must spread ROWS across the 16 blocks: fold row bits 0-2 AND
// bit 3 into the block index (16 blocks per row with BK=128 — bit 3 is
// available). Together with the rotated write order in write_tile this
// makes both the b128 staging writes and the b128 fragment reads
// bank-conflict-free (lanes that differ only in row hit distinct
// blocks).
TP_DEVICE int lds_byte(int row, int k) {
  int blk = (k >> 3) ^ (row & 7) ^ (((row >> 3) & 1) << 3);
  return row * (BK * 2) + blk * 16 + (k & 7) * 2;
}
}  // namespace conv_wrw2

__global__ __launch_bounds__(256) void conv_wrw2_kernel(
    const __hip_bfloat16* __restrict__ gy,  // (M, Cout) row-major
    const __hip_bfloat16* __restrict__ x,   // NHWC
    float* __restrict__ partial,            // (slabs, Mp, Np)
    int Nb, int Hi, int Wi, int Cin, int Cout, int Ho, int Wo, int KH,
    int KW, int stride, int pad, int Mp, int Np, int grid_n) {
  using namespace conv_wrw2;
  __shared__ char smem[2 * BM * BK * 2];  // A + B images, 64 KiB
  char* sA = smem;
  char* sB = smem + BM * BK * 2;

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;  // co tile
  int tile_n = (wg % grid_n) * BN;  // tapci tile
  int64_t M = (int64_t)Nb * Ho * Wo;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int wr = wid / WARPS_N, wc = wid % WARPS_N;

  int total_kt = (int)((M + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  int kgrp = threadIdx.x & 15;   // 8-opix group within the tile
  int ro8 = threadIdx.x >> 4;    // row octet (A: co, B: tapci)

  // B octet geometry is tile-invariant: tapci -> (tap, ci) -> (dh, dw)
  int tapci = tile_n + ro8 * 8;
  int Ktot = KH * KW * Cin;
  bool b_row_ok = tapci < Ktot;
  int tap = b_row_ok ? tapci / Cin : 0;
  int ci = b_row_ok ? tapci % Cin : 0;
  int dh = tap / KW, dw = tap % KW;
  int co0 = tile_m + ro8 * 8;
  bool a_row_ok = co0 < Cout;

  __hip_bfloat16 va[8][8], vb[8][8];  // [j over 8 opix][octet elem]

  auto load_tile = [&](int kt) {
    int64_t base = (int64_t)kt * BK + kgrp * 8;
    // decompose ONCE, carry incrementally over the 8 opix
    int wo_ = (int)(base % Wo);
    int64_t r2 = base / Wo;
    int ho_ = (int)(r2 % Ho);
    int n_ = (int)(r2 / Ho);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int64_t opix = base + j;
      if (a_row_ok && opix < M) {
        *reinterpret_cast<uint4*>(va[j]) = *reinterpret_cast<const uint4*>(
            gy + opix * Cout + co0);
      } else {
        *reinterpret_cast<uint4*>(va[j]) = uint4{0, 0, 0, 0};
      }
      int hi = ho_ * stride - pad + dh;
      int wi = wo_ * stride - pad + dw;
      if (b_row_ok && opix < M && hi >= 0 && hi < Hi && wi >= 0 &&
          wi < Wi) {
        *reinterpret_cast<uint4*>(vb[j]) = *reinterpret_cast<const uint4*>(
            x + (((int64_t)n_ * Hi + hi) * Wi + wi) * Cin + ci);
      } else {
        *reinterpret_cast<uint4*>(vb[j]) = uint4{0, 0, 0, 0};
      }
      if (++wo_ == Wo) {
        wo_ = 0;
        if (++ho_ == Ho) {
          ho_ = 0;
          ++n_;
        }
      }
    }
  };

  auto write_tile = [&]() {
    // transpose [8 opix][8 rows] -> per row a 16 B run of 8 opix
    // (k = kgrp*8 is block-aligned: one ds_write_b128 per row).
    // Indices stay compile-time constant — runtime-rotated register
    // indexing spills the staging array to scratch (measured r2d),
    // and the residual write conflicts sit below the MFMA issue time
    // anyway.
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      __hip_bfloat16 ra[8], rb[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        ra[j] = va[j][c];
        rb[j] = vb[j][c];
      }
      *reinterpret_cast<uint4*>(sA + lds_byte(ro8 * 8 + c, kgrp * 8)) =
          *reinterpret_cast<const uint4*>(ra);
      *reinterpret_cast<uint4*>(sB + lds_byte(ro8 * 8 + c, kgrp * 8)) =
          *reinterpret_cast<const uint4*>(rb);
    }
  };

  f32x4 acc[MREP][NREP] = {};
  if (kt0 < kt1) load_tile(kt0);
  for (int t = kt0; t < kt1; ++t) {
    write_tile();
    __syncthreads();
    if (t + 1 < kt1) load_tile(t + 1);  // global loads fly under MFMA
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      bf16x8 a_frag[MREP], b_frag[NREP];
      int kf = ks * 32 + (lane >> 4) * 8;
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            sA + lds_byte(wr * WM + mi * 16 + rowf, kf));
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni)
        b_frag[ni] = *reinterpret_cast<const bf16x8*>(
            sB + lds_byte(wc * WN + ni * 16 + rowf, kf));
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
#pragma unroll
        for (int ni = 0; ni < NREP; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // LDS consumed; next write_tile may overwrite
  }

#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wc * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + wr * WM + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

// ---- v3: wide-N tiles for the 3x3 shapes --------------------------------
//
// PMC finding (profiles/r02: pmc_wrw): v2 is neither MFMA- nor
// LDS-bound — it is TRAFFIC-bound. With 128x128 tiles a 64x576-output
// problem (layer1 3x3) re-reads gy gridN times and x gridM times:
// ~4 GB of DRAM for a 410 MB problem, a ~500 us floor — exactly
// MIOpen's time, unreachable to beat from that tiling.
//
// v3 observation: the wrw OUTPUT is tiny (Cout x 9Cin), so make the
// block tile as WIDE as a whole K-slab: 64 co x 576 tapci per block
// (wave w owns tapci [w*144,(w+1)*144): MREP=4 x NREP=9 = 144 f32 acc
// regs). grid_n = K/576 (1 for Cin=64 — gy and x are then each read
// exactly ONCE), BK=32 opix per iteration, split-K fills the chip.
// The 64 B LDS row stride also kills the v2 bank-aliasing problem
// (64 B = 16 banks, not 0 mod 64).
namespace conv_wrw3 {
constexpr int BM = 64;      // co rows per block
constexpr int BN = 576;     // tapci rows per block (4 waves x 144)
constexpr int BK = 32;      // opix per iteration
constexpr int WN = 144;     // tapci per wave
constexpr int MREP = 4, NREP = 9;
// [row][k] bf16 image, 64 B rows (4 x 16 B blocks), XOR swizzle on the
// low 2 row bits
TP_DEVICE int lds_byte(int row, int k) {
  int blk = ((k >> 3) ^ row) & 3;
  return row * (BK * 2) + blk * 16 + (k & 7) * 2;
}
}  // namespace conv_wrw3

__global__ __launch_bounds__(256) void conv_wrw3_kernel(
    const __hip_bfloat16* __restrict__ gy,  // (M, Cout) row-major
    const __hip_bfloat16* __restrict__ x,   // NHWC
    float* __restrict__ partial,            // (slabs, Mp, Np)
    int Nb, int Hi, int Wi, int Cin, int Cout, int Ho, int Wo, int KH,
    int KW, int stride, int pad, int Mp, int Np, int grid_n) {
  using namespace conv_wrw3;
  // A [64][32] + B [576][32] bf16 = 40 KiB
  __shared__ char smem[(BM + BN) * BK * 2];
  char* sA = smem;
  char* sB = smem + BM * BK * 2;

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;  // co tile
  int tile_n = (wg % grid_n) * BN;  // tapci tile
  int64_t M = (int64_t)Nb * Ho * Wo;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;

  int total_kt = (int)((M + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  // staging slots: 0..31 = A (ro8 = s>>2 of 8 co rows, kg = s&3 of
  // 8 opix), 32..319 = B (same split over 72 row-octets). Threads
  // 0..63 take a second slot (s + 256). All slot state lives in plain
  // scalar locals and the staging arrays are captured (not passed) by
  // the lambdas — aggregates passed by reference defeat SROA and land
  // the hot-loop state in scratch (measured r2e: 120 scratch ops, 6x
  // slowdown).
  int Ktot = KH * KW * Cin;

  // decode slot s -> (is_b, row0, kg, row_ok, src geometry)
  int s0 = threadIdx.x;
  bool b0_is_b = s0 >= 32;
  int p0_ro8 = b0_is_b ? (s0 - 32) >> 2 : s0 >> 2;
  int p0_kg = b0_is_b ? (s0 - 32) & 3 : s0 & 3;
  int p0_row0 = p0_ro8 * 8;
  int p0_tapci = tile_n + p0_row0;
  bool p0_ok = b0_is_b ? (p0_tapci < Ktot) : (tile_m + p0_row0 < Cout);
  int p0_tap = (b0_is_b && p0_ok) ? p0_tapci / Cin : 0;
  int p0_ci = (b0_is_b && p0_ok) ? p0_tapci % Cin : 0;
  int p0_dh = p0_tap / KW, p0_dw = p0_tap % KW;
  const __hip_bfloat16* p0_gy = gy + tile_m + p0_row0;

  int s1 = threadIdx.x + 256;      // overflow slot (threads 0..63): B
  bool has1 = threadIdx.x < 64;
  int p1_ro8 = (s1 - 32) >> 2;
  int p1_kg = (s1 - 32) & 3;
  int p1_row0 = p1_ro8 * 8;
  int p1_tapci = tile_n + p1_row0;
  bool p1_ok = has1 && p1_tapci < Ktot;
  int p1_tap = p1_ok ? p1_tapci / Cin : 0;
  int p1_ci = p1_ok ? p1_tapci % Cin : 0;
  int p1_dh = p1_tap / KW, p1_dw = p1_tap % KW;

  __hip_bfloat16 v[8][8];  // primary slot: [j over 8 opix][octet elem]

  // All coordinate state is 32-bit and advanced INCREMENTALLY across
  // iterations: the original per-tile int64 divmods expanded to ~150
  // soft-division instructions each and made the loop body ~3000 ISA
  // instructions against 36 MFMAs (r2f measurement). One divmod at
  // kt0, then carry chains only. M, opix and byte offsets all fit in
  // 32 bits for every supported shape (M <= 512*112*112 < 2^31).
  int Mi = (int)M;

  struct Coord {
    int wo, ho, n;
  };
  auto coord_init = [&](int opix0) -> Coord {
    Coord c;
    c.wo = opix0 % Wo;
    int r2 = opix0 / Wo;
    c.ho = r2 % Ho;
    c.n = r2 / Ho;
    return c;
  };
  auto coord_step = [&](Coord& c, int by) {
    c.wo += by;
    while (c.wo >= Wo) {
      c.wo -= Wo;
      if (++c.ho == Ho) {
        c.ho = 0;
        ++c.n;
      }
    }
  };

  int p0_base = kt0 * BK + p0_kg * 8;  // this slot's first opix
  // clamped inits keep the divmod benign for out-of-range slots; every
  // load is guarded by opix < Mi anyway
  Coord c0 = coord_init(p0_base < Mi ? p0_base : 0);
  int p1_base = kt0 * BK + p1_kg * 8;
  Coord c1 = coord_init((has1 && p1_base < Mi) ? p1_base : 0);

  auto load_primary = [&](int opix_b, const Coord& cb) {
    int wo_ = cb.wo, ho_ = cb.ho, n_ = cb.n;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int opix = opix_b + j;
      bool ok = p0_ok && opix < Mi;
      if (!b0_is_b) {
        if (ok) {
          *reinterpret_cast<uint4*>(v[j]) =
              *reinterpret_cast<const uint4*>(p0_gy + (int64_t)opix * Cout);
        } else {
          *reinterpret_cast<uint4*>(v[j]) = uint4{0, 0, 0, 0};
        }
      } else {
        int hi = ho_ * stride - pad + p0_dh;
        int wi = wo_ * stride - pad + p0_dw;
        if (ok && hi >= 0 && hi < Hi && wi >= 0 && wi < Wi) {
          *reinterpret_cast<uint4*>(v[j]) =
              *reinterpret_cast<const uint4*>(
                  x + ((int64_t)(n_ * Hi + hi) * Wi + wi) * Cin + p0_ci);
        } else {
          *reinterpret_cast<uint4*>(v[j]) = uint4{0, 0, 0, 0};
        }
        if (++wo_ == Wo) {
          wo_ = 0;
          if (++ho_ == Ho) {
            ho_ = 0;
            ++n_;
          }
        }
      }
    }
  };

  auto write_primary = [&]() {
    char* img = b0_is_b ? sB : sA;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      __hip_bfloat16 r[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = v[j][c];
      *reinterpret_cast<uint4*>(img + lds_byte(p0_row0 + c, p0_kg * 8)) =
          *reinterpret_cast<const uint4*>(r);
    }
  };

  // overflow B slot (threads 0..63): load + write 8 opix with transient
  // registers, one pass
  auto stage1 = [&](int opix_b, const Coord& cb) {
    __hip_bfloat16 tmp[8][8];
    int wo_ = cb.wo, ho_ = cb.ho, n_ = cb.n;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int opix = opix_b + j;
      int hi = ho_ * stride - pad + p1_dh;
      int wi = wo_ * stride - pad + p1_dw;
      if (p1_ok && opix < Mi && hi >= 0 && hi < Hi && wi >= 0 &&
          wi < Wi) {
        *reinterpret_cast<uint4*>(tmp[j]) =
            *reinterpret_cast<const uint4*>(
                x + ((int64_t)(n_ * Hi + hi) * Wi + wi) * Cin + p1_ci);
      } else {
        *reinterpret_cast<uint4*>(tmp[j]) = uint4{0, 0, 0, 0};
      }
      if (++wo_ == Wo) {
        wo_ = 0;
        if (++ho_ == Ho) {
          ho_ = 0;
          ++n_;
        }
      }
    }
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      __hip_bfloat16 r[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = tmp[j][c];
      *reinterpret_cast<uint4*>(
          sB + lds_byte(p1_row0 + c, p1_kg * 8)) =
          *reinterpret_cast<const uint4*>(r);
    }
  };

  f32x4 acc[MREP][NREP] = {};
  if (kt0 < kt1) load_primary(p0_base, c0);
  for (int t = kt0; t < kt1; ++t) {
    write_primary();
    if (has1) stage1(p1_base, c1);
    __syncthreads();
    if (t + 1 < kt1) {
      p0_base += BK;
      coord_step(c0, BK);
      p1_base += BK;
      coord_step(c1, BK);
      load_primary(p0_base, c0);  // loads fly under MFMA
    }
    {
      // B fragments are read one at a time inside the ni loop so only
      // one is live at once (holding all 9 costs 36 VGPRs and drops
      // occupancy to 1 wave/SIMD with the 144-reg accumulator)
      bf16x8 a_frag[MREP];
      int kf = (lane >> 4) * 8;
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            sA + lds_byte(mi * 16 + rowf, kf));
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni) {
        bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            sB + lds_byte(wid * WN + ni * 16 + rowf, kf));
#pragma unroll
        for (int mi = 0; mi < MREP; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wid * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

// ---- v4: glds raw staging + in-LDS transpose ----------------------------
//
// What v2/v3 established (profiles/r02_summary.md): a winning wrw must
// simultaneously (a) keep operand re-reads ~1x (v3's 64x576 tile),
// (b) keep >=10 KB of global loads in flight per CU (v3's
// register-held gathers managed 8x16 B per wave at occupancy 1:
// ~0.9 TB/s), (c) spill nothing, (d) spend ~0 registers on staging.
// The only structure that satisfies all four is the fwd kernel's:
// `global_load_lds` — per-lane gathered addresses, zero destination
// registers, fire-and-forget depth limited only by vmcnt. glds cannot
// transpose, so the tile lands in LDS in gather order ([opix][c]) and
// a separate LDS->LDS transpose stage builds the [row][k] MFMA images.
//
// Per 64co x 576tapci x 32opix tile:
//   glds:  A raw 4 KB in 4 instrs + B raw 36 KB in 36 instrs
//          (each instr: 64 lanes x 16 B -> 1 KB contiguous deposit;
//          per-lane bounds select to a zero page, like conv_implicit)
//   MFMA(t) runs while tile t+1's glds are in flight
//   transpose: each thread moves ~1.25 row-octets: 8 strided
//          ds_read_b128 from raw + register transpose + 8 ds_write_b128
//          into the swizzled image (short-lived registers)
// LDS: rawA 4K + rawB 36K + imgA 4K + imgB 36K = 80 KiB dynamic
// -> exactly 2 blocks/CU.
namespace conv_wrw4 {
constexpr int BM = 64, BN = 576, BK = 32;
constexpr int WN = 144;
constexpr int MREP = 4, NREP = 9;
constexpr int RAW_A = 0;                  // [32 opix][64 co]   4 KiB
constexpr int RAW_B = 4 * 1024;           // [9 chunk][32 opix][64 ci]
constexpr int IMG_A = RAW_A + 40 * 1024;  // swizzled [64 co][32 k]
constexpr int IMG_B = IMG_A + 4 * 1024;   // swizzled [576][32 k]
TP_DEVICE int lds_byte(int row, int k) {  // same image as v3
  int blk = ((k >> 3) ^ row) & 3;
  return row * (BK * 2) + blk * 16 + (k & 7) * 2;
}
}  // namespace conv_wrw4

__global__ __launch_bounds__(256) void conv_wrw4_kernel(
    const __hip_bfloat16* __restrict__ gy,  // (M, Cout) row-major
    const __hip_bfloat16* __restrict__ x,   // NHWC
    const __hip_bfloat16* __restrict__ zero_page,  // >=128 B of zeros
    float* __restrict__ partial,            // (slabs, Mp, Np)
    int Nb, int Hi, int Wi, int Cin, int Cout, int Ho, int Wo, int KH,
    int KW, int stride, int pad, int Mp, int Np, int grid_n) {
  using namespace conv_wrw4;
  extern __shared__ char smem[];

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;  // co tile
  int tile_n = (wg % grid_n) * BN;  // tapci tile
  int64_t M64 = (int64_t)Nb * Ho * Wo;
  int Mi = (int)M64;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;

  int total_kt = (int)((M64 + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  // ---- glds staging ----------------------------------------------------
  // wave w issues: 1 A-instr (opix rows 8w..8w+7) + 9 B-instrs
  // (chunk c = w + 4*i, og cycling). Per-lane opix for an instr with
  // opix-group og: og*8 + (lane>>3). Each lane tracks FOUR coordinate
  // sets (its opix offset + {0,8,16,24}), advanced by +BK per tile with
  // carry loops — no divisions in the loop.
  int lrow = lane >> 3;   // 0..7: opix within group
  int loct = lane & 7;    // 16 B octet within the 128 B segment

  int Ktot = KH * KW * Cin;
  // B chunk geometry (9 chunks of 64 tapci): tap + ci0 per chunk
  // (Cin % 64 == 0 keeps each chunk inside one tap)
  int cdh[9], cdw[9], cci[9];
  bool cok[9];
#pragma unroll
  for (int c = 0; c < 9; ++c) {
    int tapci = tile_n + c * 64;
    cok[c] = tapci < Ktot;
    int tap = cok[c] ? tapci / Cin : 0;
    cci[c] = cok[c] ? tapci % Cin : 0;
    cdh[c] = tap / KW;
    cdw[c] = tap % KW;
  }

  // per-lane coordinate sets for opix = kt*BK + og*8 + lrow, og=0..3
  int wo4[4], ho4[4], n4[4], op4[4];
#pragma unroll
  for (int og = 0; og < 4; ++og) {
    int opix = kt0 * BK + og * 8 + lrow;
    op4[og] = opix;
    int o = opix < Mi ? opix : 0;
    wo4[og] = o % Wo;
    int r2 = o / Wo;
    ho4[og] = r2 % Ho;
    n4[og] = r2 / Ho;
  }
  auto advance = [&]() {
#pragma unroll
    for (int og = 0; og < 4; ++og) {
      op4[og] += BK;
      wo4[og] += BK;
      while (wo4[og] >= Wo) {
        wo4[og] -= Wo;
        if (++ho4[og] == Ho) {
          ho4[og] = 0;
          ++n4[og];
        }
      }
    }
  };

  auto issue_glds = [&]() {
    // A: this wave's 8 opix rows (og = wid & 3 happens to be wid)
    {
      int og = wid;
      const char* src;
      if (tile_m + loct * 8 < Cout && op4[og] < Mi) {
        src = reinterpret_cast<const char*>(
            gy + (int64_t)op4[og] * Cout + tile_m + loct * 8);
      } else {
        src = reinterpret_cast<const char*>(zero_page) + loct * 16;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              smem + RAW_A + og * 8 * 128),
          16, 0, 0);
    }
    // B: 9 instrs per wave: (chunk, og) pairs; c fixed per instr,
    // og = instr index & 3 keeps all four coord sets busy
#pragma unroll
    for (int i = 0; i < 9; ++i) {
      int c = i;           // chunk
      int og = (i + wid) & 3;
      int hi = ho4[og] * stride - pad + cdh[c];
      int wi = wo4[og] * stride - pad + cdw[c];
      bool ok = cok[c] && op4[og] < Mi && hi >= 0 && hi < Hi &&
                wi >= 0 && wi < Wi;
      const char* src =
          ok ? reinterpret_cast<const char*>(
                   x + ((int64_t)(n4[og] * Hi + hi) * Wi + wi) * Cin +
                   cci[c] + loct * 8)
             : reinterpret_cast<const char*>(zero_page) + loct * 16;
      // each WAVE must deposit a distinct 1 KB run; runs are indexed
      // by (chunk, og): raw_b[chunk][og*8 + lrow][loct]
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              smem + RAW_B + (c * 32 + og * 8) * 128),
          16, 0, 0);
    }
  };

  // ---- transpose raw -> swizzled images --------------------------------
  // slots: 0..31 A (co-octet ro=s>>2, kg=s&3), 32..319 B (tapci-octet).
  // Threads 0..63 take a second slot.
  int s0 = threadIdx.x;
  int s1 = threadIdx.x + 256;
  bool has1 = threadIdx.x < 64;

  auto do_slot = [&](int s) {
    bool is_b = s >= 32;
    int ro = is_b ? (s - 32) >> 2 : s >> 2;  // row-octet
    int kg = is_b ? (s - 32) & 3 : s & 3;    // 8-opix group
    // raw read base: A raw [opix][co]: byte = opix*128 + ro*16
    //                B raw chunk = ro>>3, ci-octet = ro&7
    int raw_base = is_b ? RAW_B + (ro >> 3) * 4096 + (ro & 7) * 16
                        : RAW_A + ro * 16;
    char* img = smem + (is_b ? IMG_B : IMG_A);
    __hip_bfloat16 t8[8][8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<uint4*>(t8[j]) = *reinterpret_cast<const uint4*>(
          smem + raw_base + (kg * 8 + j) * 128);
    }
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      __hip_bfloat16 r[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = t8[j][c];
      *reinterpret_cast<uint4*>(img + lds_byte(ro * 8 + c, kg * 8)) =
          *reinterpret_cast<const uint4*>(r);
    }
  };

  f32x4 acc[MREP][NREP] = {};
  if (kt0 < kt1) {
    issue_glds();   // tile kt0 in flight
    __syncthreads();  // carries vmcnt(0): deposits landed (guide §5.4)
  }
  for (int t = kt0; t < kt1; ++t) {
    // raw(t) is complete (drained by the previous barrier): transpose
    do_slot(s0);
    if (has1) do_slot(s1);
    __syncthreads();  // images ready; raw free
    // issue tile t+1's gathers — they fly under the MFMAs and are
    // drained by the loop-end barrier
    if (t + 1 < kt1) {
      advance();
      issue_glds();
    }
    {
      bf16x8 a_frag[MREP];
      int kf = (lane >> 4) * 8;
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            smem + IMG_A + lds_byte(mi * 16 + rowf, kf));
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni) {
        bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            smem + IMG_B + lds_byte(wid * WN + ni * 16 + rowf, kf));
#pragma unroll
        for (int mi = 0; mi < MREP; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    // images consumed + next tile's glds drained before re-transpose
    __syncthreads();
  }

#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wid * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

// ---- v5: persistent ring window (3x3 stride-1, large Wo) ---------------
//
// v4's residual wall is gather THROUGHPUT: the im2col gather moves 9x
// the x bytes through L2 at 16 B granularity (r2l PMC). Observation:
// at BK=32 opix a tile advances only ~0.5-1 OUTPUT ROW, and for
// k3/s1/p1 the x rows a tile needs are the CONTIGUOUS flat-row range
// [R0-1, R1+1] (Hi == Ho, flat row F = n*Hi + hi = R + dh - 1; the
// cross-image bleed rows are masked at transpose time). So keep a
// 5-row ring window of x RESIDENT in LDS and, per tile, load only the
// 0-2 NEW rows — pure contiguous 1 KB glds (7 per row), x read ~once
// total. The 9-tap resolution happens during the LDS transpose (window
// reads), which is on-chip. Covers the shapes where the window (5 rows
// x Wi x Cin x 2 B = 35.8 KiB) fits and rows are 7 x 1 KiB exactly:
// layer1 (56²,Cin64) and layer2 (28²,Cin128) — the two biggest MIOpen
// wrw entries. Others fall back to v4.
namespace conv_wrw5 {
constexpr int BM = 64, BN = 576, BK = 32;
constexpr int WN = 144;
constexpr int MREP = 4, NREP = 9;
constexpr int WROWS = 5;
constexpr int RAW_A = 0;                     //  4 KiB
constexpr int WIN = 4 * 1024;                // 36 KiB (5 x 7168 B ring)
constexpr int IMG_A = WIN + 36 * 1024;       //  4 KiB
constexpr int IMG_B = IMG_A + 4 * 1024;      // 36 KiB
TP_DEVICE int lds_byte(int row, int k) {
  int blk = ((k >> 3) ^ row) & 3;
  return row * (BK * 2) + blk * 16 + (k & 7) * 2;
}
}  // namespace conv_wrw5

// MODE: 0 = normal; diagnostic ablations (results invalid):
// 1 = no MFMA, 2 = no transpose, 3 = no glds,
// 4 = barriers+bookkeeping only, 5 = bookkeeping only
// (TURBOPRUNE_WRW5_MODE)
template <int MODE = 0>
__global__ __launch_bounds__(256) void conv_wrw5_kernel(
    const __hip_bfloat16* __restrict__ gy,  // (M, Cout) row-major
    const __hip_bfloat16* __restrict__ x,   // NHWC
    float* __restrict__ partial,            // (slabs, Mp, Np)
    int Nb, int HiWi, int Cin, int Cout, int HoWo, int Mp, int Np,
    int grid_n) {
  // HiWi == HoWo (k3 s1 p1); square images: Hi=Wi=HoWo? No — HiWi is
  // the SIDE (Hi == Wi == Ho == Wo for the gated shapes).
  using namespace conv_wrw5;
  extern __shared__ char smem[];
  const int Wo = HoWo, Hi = HiWi, Wi = HiWi, Ho = HoWo;
  const int rowpitch = Wi * Cin * 2;        // == 7168 by the host gate

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;  // co tile
  int tile_n = (wg % grid_n) * BN;  // tapci tile
  int64_t M64 = (int64_t)Nb * Ho * Wo;
  int Mi = (int)M64;
  int Frows = Nb * Hi;              // total flat x rows

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;

  int total_kt = (int)((M64 + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  // ---- uniform per-tile scalar state -----------------------------------
  int opix0 = kt0 * BK;
  int wo0 = opix0 % Wo;             // one division at entry only
  int R0 = opix0 / Wo;              // flat output row of the first opix
  int hoabs0 = R0 % Ho;
  int m5 = R0 % WROWS;
  int Floaded = R0 - 2;             // highest flat x row in the ring

  // B chunk geometry (9 chunks of 64 tapci)
  int Ktot = 9 * Cin;
  int cdh[9], cdw[9], cci[9];
  bool cok[9];
#pragma unroll
  for (int c = 0; c < 9; ++c) {
    int tapci = tile_n + c * 64;
    cok[c] = tapci < Ktot;
    int tap = cok[c] ? tapci / Cin : 0;
    cci[c] = cok[c] ? tapci % Cin : 0;
    cdh[c] = tap / 3;
    cdw[c] = tap % 3;
  }

  // A glds (identical to v4): this wave's 8 opix rows
  int lrow = lane >> 3;
  int loct = lane & 7;
  int opA = opix0 + wid * 8 + lrow;   // per-lane A opix

  auto issue_A = [&]() {
    const char* src;
    if (opA < Mi) {
      src = reinterpret_cast<const char*>(
          gy + (int64_t)opA * Cout + tile_m + loct * 8);
    } else {
      src = reinterpret_cast<const char*>(x);  // any valid memory
    }
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(
            smem + RAW_A + wid * 8 * 128),
        16, 0, 0);
  };

  // window loader: bring flat rows (Floaded, target] into the ring.
  // Row F -> ring slot F mod 5; 7 contiguous 1 KB glds per row,
  // instr i of the row handled by wave (i % 4).
  auto load_rows = [&](int target) {
    for (int F = Floaded + 1; F <= target; ++F) {
      int Fc = F < 0 ? 0 : (F >= Frows ? Frows - 1 : F);
      int slot = F % WROWS;
      if (slot < 0) slot += WROWS;
      const char* base =
          reinterpret_cast<const char*>(x) + (int64_t)Fc * rowpitch;
      char* dst = smem + WIN + slot * rowpitch;
      int nq = rowpitch >> 10;  // 1 KiB chunks per row (host gate: %1024==0)
      for (int q = wid; q < nq; q += 4) {
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)(
                base + q * 1024 + lane * 16),
            (__attribute__((address_space(3))) void*)(dst + q * 1024),
            16, 0, 0);
      }
    }
    Floaded = target;
  };

  // ---- transposes -------------------------------------------------------
  auto do_A = [&](int s) {  // s < 32: (ro = s>>2 co-octet, kg = s&3)
    int ro = s >> 2, kg = s & 3;
    __hip_bfloat16 t8[8][8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // mask tail opix here (the glds fallback source is arbitrary
      // valid memory, not zeros)
      if (opix0 + kg * 8 + j < Mi) {
        *reinterpret_cast<uint4*>(t8[j]) =
            *reinterpret_cast<const uint4*>(
                smem + RAW_A + (kg * 8 + j) * 128 + ro * 16);
      } else {
        *reinterpret_cast<uint4*>(t8[j]) = uint4{0, 0, 0, 0};
      }
    }
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      __hip_bfloat16 r[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = t8[j][c];
      *reinterpret_cast<uint4*>(
          smem + IMG_A + lds_byte(ro * 8 + c, kg * 8)) =
          *reinterpret_cast<const uint4*>(r);
    }
  };

  auto do_B = [&](int s) {  // s in [0,288): ro = s>>2 tapci-octet, kg
    int ro = s >> 2, kg = s & 3;
    int c = ro >> 3;           // chunk 0..8
    int dh = cdh[c], dw = cdw[c];
    int ci0 = cci[c] + (ro & 7) * 8;
    bool rok = cok[c];
    // thread-local opix coords for its 8 opix (kg*8 + j)
    int wo_ = wo0 + kg * 8;
    int hr_ = 0;               // ho_rel (rows advanced past R0)
    int ha_ = hoabs0;
    while (wo_ >= Wo) {        // <= 1 iteration (Wo >= 28, kg*8 <= 24)
      wo_ -= Wo;
      ++hr_;
      if (++ha_ == Ho) ha_ = 0;
    }
    int op_ = opix0 + kg * 8;
    __hip_bfloat16 t8[8][8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int hi = ha_ + dh - 1;
      int wi = wo_ + dw - 1;
      bool ok = rok && op_ + j < Mi && (unsigned)hi < (unsigned)Hi &&
                (unsigned)wi < (unsigned)Wi;
      int r5 = m5 + hr_ + dh - 1;
      if (r5 < 0) r5 += WROWS;
      if (r5 >= WROWS) r5 -= WROWS;
      if (r5 >= WROWS) r5 -= WROWS;  // hr_+dh-1 can reach +5
      const char* src = smem + WIN + r5 * rowpitch +
                        (wi * Cin + ci0) * 2;
      if (ok) {
        *reinterpret_cast<uint4*>(t8[j]) =
            *reinterpret_cast<const uint4*>(src);
      } else {
        *reinterpret_cast<uint4*>(t8[j]) = uint4{0, 0, 0, 0};
      }
      if (++wo_ == Wo) {
        wo_ = 0;
        ++hr_;
        if (++ha_ == Ho) ha_ = 0;
      }
    }
#pragma unroll
    for (int cc = 0; cc < 8; ++cc) {
      __hip_bfloat16 r[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = t8[j][cc];
      *reinterpret_cast<uint4*>(
          smem + IMG_B + lds_byte(ro * 8 + cc, kg * 8)) =
          *reinterpret_cast<const uint4*>(r);
    }
  };

  auto advance_tile = [&]() {
    opix0 += BK;
    opA += BK;
    wo0 += BK;
    while (wo0 >= Wo) {
      wo0 -= Wo;
      ++R0;
      if (++hoabs0 == Ho) hoabs0 = 0;
      if (++m5 == WROWS) m5 = 0;
    }
  };

  f32x4 acc[MREP][NREP] = {};
  if (kt0 < kt1) {
    issue_A();
    // initial window: rows [R0-1 .. R1+1]
    int R1 = R0 + (wo0 + BK - 1) / Wo;
    load_rows(R1 + 1);
    __syncthreads();  // drains glds (vmcnt0 implied)
  }
  for (int t = kt0; t < kt1; ++t) {
    // transpose tile t out of rawA + ring window
    if (MODE != 2 && MODE < 4) {
      int s = threadIdx.x;
      if (s < 32) do_A(s);
      else do_B(s - 32);
      if (threadIdx.x < 64) do_B(256 - 32 + threadIdx.x);
    }
    if (MODE != 5) __syncthreads();
    if (t + 1 < kt1) {
      advance_tile();
      if (MODE != 3 && MODE < 4) {
        issue_A();
        int R1 = R0 + (wo0 + BK - 1) / Wo;
        load_rows(R1 + 1);  // 0-2 new rows, fly under the MFMAs
      }
    }
    if (MODE != 1 && MODE < 4) {
      bf16x8 a_frag[MREP];
      int kf = (lane >> 4) * 8;
      int rowf = lane & 15;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = *reinterpret_cast<const bf16x8*>(
            smem + IMG_A + lds_byte(mi * 16 + rowf, kf));
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni) {
        bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            smem + IMG_B + lds_byte(wid * WN + ni * 16 + rowf, kf));
#pragma unroll
        for (int mi = 0; mi < MREP; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    if (MODE != 5) __syncthreads();  // images consumed + glds drained
  }

#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wid * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

// fp32 output: the split-K accumulation is fp32 and the consumer is the
// fp32 master-weight gradient — rounding to bf16 here would be a
// systematic numerics divergence vs the reference autocast path
// (ADVICE r01). The output is only Cout x (KH*KW*Cin), so the extra
// bytes are negligible next to the partial slabs.
// ---- v6: glds-blocked image + ds_read_b64_tr_b16 fragments -------------
//
// v4/v5's measured wall is the LDS pipe: the raw->img transpose pass
// moves ~140 KiB/tile through the 128 B/cycle port. CDNA4's
// ds_read_b64_tr_b16 deletes that pass: each 16-lane quarter-group
// reads one contiguous [4 k][16 col] bf16 block and receives its
// TRANSPOSE (lane = column, 4 regs = rows; semantics probed on device,
// scripts/probe_tr16.hip). Because the staging is a per-lane GATHER,
// the glds deposit builds the blocked image directly (permuting which
// (opix, ci-octet) each lane loads is free), so the kernel becomes:
//
//   glds(t+1 -> other raw buffer)  |  MFMA(t) with tr_b16 fragment
//   reads straight from raw        |  one barrier per tile
//
// LDS: 2 x (A 4K + B 36K) = 80 KiB -> 2 blocks/CU. No transpose
// stage, no img buffers, per-lane zero-page bounds as v4.
//
// Image layout (B): block(kq, cs) at ((chunk*8 + kq)*4 + cs)*128 B,
// kq = opix quad (8 per 32-opix tile), cs = 16-tapci subtile within
// the 64-tapci chunk; one glds instr (chunk, og) deposits its 8
// blocks contiguously at (chunk*32 + og*8)*128 — identical dst
// arithmetic to v4, different internal granule order. A analogous.

// two tr_b16 reads (k-quads h2=0,1 at +512 B) -> one MFMA operand.
// Split issue/wait so reads pipeline under MFMAs: tr_issue starts the
// pair, tr_wait drains lgkm and carries a register dependency so the
// compiler cannot move uses above it.
struct TrPair {
  unsigned long long r0, r1;
};
TP_DEVICE void tr_issue(unsigned a0, TrPair& p) {
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %2 offset:512"
      : "=v"(p.r0), "=v"(p.r1)
      : "v"(a0));
}
// N = DS ops allowed to stay outstanding (DS completes in order, so
// lgkmcnt(2) proves an older pair landed while a newer pair flies)
template <int N>
TP_DEVICE bf16x8 tr_wait(TrPair& p) {
  asm volatile("s_waitcnt lgkmcnt(%c2)"
               : "+v"(p.r0), "+v"(p.r1)
               : "i"(N));
  union {
    unsigned long long u[2];
    bf16x8 v;
  } c;
  c.u[0] = p.r0;
  c.u[1] = p.r1;
  return c.v;
}
TP_DEVICE bf16x8 tr_frag16(unsigned a0) {
  TrPair p;
  tr_issue(a0, p);
  return tr_wait<0>(p);
}

// single-read variants (one k-quad = 4 k per lane)
TP_DEVICE void tr_issue1(unsigned a0, unsigned long long& r) {
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(r) : "v"(a0));
}
template <int N>
TP_DEVICE unsigned long long tr_wait1(unsigned long long& r) {
  asm volatile("s_waitcnt lgkmcnt(%c1)" : "+v"(r) : "i"(N));
  return r;
}


// ---- v7: ring window + tr_b16 fragments straight from the window -------
//
// The tr_b16 "run" (4 contiguous bf16 at a per-lane address) for a
// B-fragment element is 4 consecutive ci of one (opix, tap) — which
// exists VERBATIM inside the v5 ring window (x rows resident in LDS).
// So for the large-Wo stride-1 3x3 shapes the kernel needs neither a
// blocked image nor a transpose: contiguous ~7 KiB row loads per tile
// (x read ~once) and the MFMA B-operands gather from the window with
// per-lane run addresses. A stays on the v6 gather+blocked-tr path
// (tiny). One barrier per tile; out-of-range taps select a zeroed LDS
// block. WROWS=7 makes the t+1 row prefetch provably non-colliding
// with tile t's reads (max flat-row advance is 2 at Wo>=28).
namespace conv_wrw7 {
constexpr int BM = 64, BN = 576, BK = 32;
constexpr int WN = 144;
constexpr int MREP = 4, NREP = 9;
constexpr int WROWS = 7;
constexpr int MAXROW = 7168;                  // bytes (gate)
constexpr int WIN = 0;                        // 7 x rowpitch
constexpr int RAW_A0 = WROWS * MAXROW;        // 2 x 4 KiB
constexpr int ZBLK = RAW_A0 + 2 * 4096;       // 128 B zeros
constexpr int LDS_BYTES = ZBLK + 128;
}  // namespace conv_wrw7

__global__ __launch_bounds__(256) void conv_wrw7_kernel(
    const __hip_bfloat16* __restrict__ gy,  // (M, Cout) row-major
    const __hip_bfloat16* __restrict__ x,   // NHWC
    const __hip_bfloat16* __restrict__ zero_page,
    float* __restrict__ partial,            // (slabs, Mp, Np)
    int Nb, int HW, int Cin, int Cout, int Mp, int Np, int grid_n) {
  using namespace conv_wrw7;
  extern __shared__ char smem[];
  const int Wo = HW, Ho = HW, Hi = HW, Wi = HW;  // k3 s1 p1 square
  const int rowpitch = Wi * Cin * 2;

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;
  int tile_n = (wg % grid_n) * BN;
  int64_t M64 = (int64_t)Nb * Ho * Wo;
  int Mi = (int)M64;
  int Frows = Nb * Hi;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int g = lane >> 4;
  int l15 = lane & 15;
  int jl = l15 >> 2;   // fragment row within quad
  int ml = l15 & 3;    // run (4-ci quad) within subtile

  int total_kt = (int)((M64 + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  // zero the LDS zero-block (any wave; barrier below orders it)
  if (threadIdx.x < 16)
    reinterpret_cast<long long*>(smem + ZBLK)[threadIdx.x] = 0;

  // per-wave per-ni B constants: 16-tapci subtile -> (dh, dw, ci0);
  // noff folds (dw-1)*Cin + ci0 + 4*ml into one byte constant
  int tdh[NREP], tdw[NREP];
  int noff[NREP];
  int Ktot = 9 * Cin;
  bool nok[NREP];
#pragma unroll
  for (int ni = 0; ni < NREP; ++ni) {
    int tapci0 = (wid * 9 + ni) * 16;
    nok[ni] = tile_n + tapci0 < Ktot;
    int tap = nok[ni] ? (tile_n + tapci0) / Cin : 0;
    int ci0 = nok[ni] ? (tile_n + tapci0) % Cin : 0;
    tdh[ni] = tap / 3;
    tdw[ni] = tap % 3;
    noff[ni] = ((tdw[ni] - 1) * Cin + ci0 + 4 * ml) * 2;
  }

  // per-lane opix streams s=0,1: opix = kt*BK + g*8 + s*4 + jl
  int wo_[2], hrel_[2], habs_[2], op_[2], wob_[2];
  int m5 = 0;  // (R0 mod WROWS); R0 = flat output row of opix0
  int wo0, hoabs0, R0;
  {
    int opix0 = kt0 * BK;
    wo0 = opix0 % Wo;
    R0 = opix0 / Wo;
    hoabs0 = R0 % Ho;
    m5 = R0 % WROWS;
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      int off = g * 8 + s * 4 + jl;
      int o = opix0 + off;
      op_[s] = o;
      int w = wo0 + off;
      int hr = 0, ha = hoabs0;
      while (w >= Wo) {
        w -= Wo;
        ++hr;
        if (++ha == Ho) ha = 0;
      }
      wo_[s] = w;
      hrel_[s] = hr;
      habs_[s] = ha;
      wob_[s] = w * Cin * 2;
    }
  }
  int Floaded = R0 - 2;

  auto advance = [&]() {
    // uniform trackers
    wo0 += BK;
    while (wo0 >= Wo) {
      wo0 -= Wo;
      ++R0;
      if (++hoabs0 == Ho) hoabs0 = 0;
      if (++m5 == WROWS) m5 = 0;
    }
#pragma unroll
    for (int s = 0; s < 2; ++s) {
      op_[s] += BK;
      wo_[s] += BK;
      while (wo_[s] >= Wo) {
        wo_[s] -= Wo;
        ++hrel_[s];
        if (++habs_[s] == Ho) habs_[s] = 0;
      }
      wob_[s] = wo_[s] * Cin * 2;
    }
    // hrel_ is relative to the OLD R0 after uniform advance moved it;
    // re-anchor: hrel_new = (old hrel + old R0) - new R0 handled by
    // tracking against R0 directly below (hrel_ -= deltaR0 done in
    // caller via ranchor)
  };

  auto issue_A = [&](int buf) {
    // v6-style blocked A deposit (one instr per wave)
    int cs6 = (lane >> 3) & 3;
    int h6 = lane & 1;
    int q6 = (lane >> 5) & 1;
    int r26 = (lane & 7) >> 1;
    int opoff = q6 * 4 + r26;
    int opix = (op_[0] - (g * 8 + jl)) + wid * 8 + opoff;
    // (op_[0] - lane B-offset) = opix0 of this tile
    int co8 = cs6 * 2 + h6;
    const char* src;
    if (opix < Mi) {
      src = reinterpret_cast<const char*>(
          gy + (int64_t)opix * Cout + tile_m + co8 * 8);
    } else {
      src = reinterpret_cast<const char*>(zero_page) + co8 * 16;
    }
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(
            smem + RAW_A0 + buf * 4096 + wid * 1024),
        16, 0, 0);
  };

  auto load_rows = [&](int target) {
    for (int F = Floaded + 1; F <= target; ++F) {
      int Fc = F < 0 ? 0 : (F >= Frows ? Frows - 1 : F);
      int slot = F % WROWS;
      if (slot < 0) slot += WROWS;
      const char* base =
          reinterpret_cast<const char*>(x) + (int64_t)Fc * rowpitch;
      char* dst = smem + WIN + slot * rowpitch;
      int nq = rowpitch >> 10;
      for (int q = wid; q < nq; q += 4) {
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)(
                base + q * 1024 + lane * 16),
            (__attribute__((address_space(3))) void*)(dst + q * 1024),
            16, 0, 0);
      }
    }
    Floaded = target;
  };

  f32x4 acc[MREP][NREP] = {};
  int cur = 0;
  if (kt0 < kt1) {
    issue_A(0);
    int R1 = R0 + (wo0 + BK - 1) / Wo;
    load_rows(R1 + 1);
    __syncthreads();
  }
  for (int t = kt0; t < kt1; ++t) {
    // --- snapshot tile-t fragment address bases ------------------------
    // per s, per dh: window row byte (ring slot) + validity
    int rbase[2][3];
    bool rok[2][3];
#pragma unroll
    for (int s = 0; s < 2; ++s) {
#pragma unroll
      for (int dh = 0; dh < 3; ++dh) {
        int hi = habs_[s] + dh - 1;
        rok[s][dh] = (unsigned)hi < (unsigned)Hi && op_[s] < Mi;
        int r5 = m5 + hrel_[s] + dh - 1;
        while (r5 >= WROWS) r5 -= WROWS;
        if (r5 < 0) r5 += WROWS;
        rbase[s][dh] = WIN + r5 * rowpitch;
      }
    }
    int wob_s[2] = {wob_[0], wob_[1]};
    int wo_s[2] = {wo_[0], wo_[1]};

    // --- issue tile t+1 loads (fly under the MFMAs) --------------------
    if (t + 1 < kt1) {
      int oldR0 = R0;
      advance();
      int dR = R0 - oldR0;
      hrel_[0] -= dR;
      hrel_[1] -= dR;
      issue_A(cur ^ 1);
      int R1 = R0 + (wo0 + BK - 1) / Wo;
      load_rows(R1 + 1);
    }

    // --- MFMA: A from blocked raw (tr), B straight from the window -----
    {
      char* rawA = smem + RAW_A0 + cur * 4096;
      TrPair pa[MREP];
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        tr_issue((unsigned)(size_t)(
                     rawA + ((2 * g) * 4 + mi) * 128 + l15 * 8),
                 pa[mi]);
      bf16x8 a_frag[MREP];
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi) a_frag[mi] = tr_wait<0>(pa[mi]);
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni) {
        // per-lane run addresses for the two k-quads (s = 0, 1)
        unsigned ab[2];
#pragma unroll
        for (int s = 0; s < 2; ++s) {
          int dh = tdh[ni];
          bool ok = nok[ni] && rok[s][dh] &&
                    (unsigned)(wo_s[s] + tdw[ni] - 1) < (unsigned)Wi;
          int a = rbase[s][dh] + wob_s[s] + noff[ni];
          ab[s] = ok ? (unsigned)a : (unsigned)(ZBLK + 8 * ml);
        }
        unsigned base32 = (unsigned)(size_t)smem;
        unsigned long long q0, q1;
        tr_issue1(base32 + ab[0], q0);
        tr_issue1(base32 + ab[1], q1);
        union {
          unsigned long long u[2];
          bf16x8 v;
        } cb;
        cb.u[0] = tr_wait1<1>(q0);
        cb.u[1] = tr_wait1<0>(q1);
        bf16x8 b_frag = cb.v;
#pragma unroll
        for (int mi = 0; mi < MREP; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wid * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

namespace conv_wrw6 {
constexpr int BM = 64, BN = 576, BK = 32;
constexpr int WN = 144;
constexpr int MREP = 4, NREP = 9;
constexpr int RAW_A0 = 0;            //  4 KiB
constexpr int RAW_B0 = 4 * 1024;     // 36 KiB
constexpr int BUF_STRIDE = 40 * 1024;
}  // namespace conv_wrw6

__global__ __launch_bounds__(256) void conv_wrw6_kernel(
    const __hip_bfloat16* __restrict__ gy,  // (M, Cout) row-major
    const __hip_bfloat16* __restrict__ x,   // NHWC
    const __hip_bfloat16* __restrict__ zero_page,  // >=128 B zeros
    float* __restrict__ partial,            // (slabs, Mp, Np)
    int Nb, int Hi, int Wi, int Cin, int Cout, int Ho, int Wo, int KH,
    int KW, int stride, int pad, int Mp, int Np, int grid_n) {
  using namespace conv_wrw6;
  extern __shared__ char smem[];

  int wg = blockIdx.x;
  int tile_m = (wg / grid_n) * BM;  // co tile
  int tile_n = (wg % grid_n) * BN;  // tapci tile
  int64_t M64 = (int64_t)Nb * Ho * Wo;
  int Mi = (int)M64;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;

  int total_kt = (int)((M64 + BK - 1) / BK);
  int per = (total_kt + gridDim.y - 1) / gridDim.y;
  int kt0 = blockIdx.y * per;
  int kt1 = min(kt0 + per, total_kt);
  partial += (int64_t)blockIdx.y * Mp * Np;

  // per-lane deposit geometry: granule l -> block b = l>>3 of the
  // instr's 1 KiB (b = quad-half q * 4 + col-subtile cs), row-in-block
  // r2 = (l&7)>>1, column-octet-half h = l&1. Lane's opix offset
  // within the instr's 8-opix group: q*4 + r2.
  int q_ = (lane >> 5) & 1;
  int cs_ = (lane >> 3) & 3;
  int r2_ = (lane & 7) >> 1;
  int h_ = lane & 1;
  int opoff = q_ * 4 + r2_;

  int Ktot = KH * KW * Cin;
  // per-chunk constants: tap byte offset into x relative to the
  // (n, ho*stride-pad, wo*stride-pad) corner, ci-chunk byte offset,
  // and the (dh, dw) needed for bounds
  int cdh[9], cdw[9];
  int coff[9];  // ((dh*Wi + dw)*Cin + ci0) * 2
  bool cok[9];
#pragma unroll
  for (int c = 0; c < 9; ++c) {
    int tapci = tile_n + c * 64;
    cok[c] = tapci < Ktot;
    int tap = cok[c] ? tapci / Cin : 0;
    int ci0 = cok[c] ? tapci % Cin : 0;
    cdh[c] = tap / KW;
    cdw[c] = tap % KW;
    coff[c] = ((cdh[c] * Wi + cdw[c]) * Cin + ci0) * 2;
  }

  // per-lane coordinate sets for opix = kt*BK + og*8 + opoff, og=0..3.
  // Besides (wo, ho, n) each og carries the 32-bit BYTE offset of its
  // corner x element ((n*Hi + ho*s - p)*Wi + wo*s - p)*Cin*2 — updated
  // incrementally so the per-instr address math is one add
  // (PMC r2q: the per-instr 64-bit address formation made
  // SQ_INSTS_VALU ~= MFMA busy cycles).
  int wo4[4], ho4[4], n4[4], op4[4], xo4[4];
  const int stepw = stride * Cin * 2;                    // wo+1
  const int rowfix_num = (stride * Wi - 0) * Cin * 2;    // see advance
#pragma unroll
  for (int og = 0; og < 4; ++og) {
    int opix = kt0 * BK + og * 8 + opoff;
    op4[og] = opix;
    int o = opix < Mi ? opix : 0;
    wo4[og] = o % Wo;
    int r2v = o / Wo;
    ho4[og] = r2v % Ho;
    n4[og] = r2v / Ho;
    xo4[og] = (((n4[og] * Hi + ho4[og] * stride - pad) * Wi) +
               wo4[og] * stride - pad) * Cin * 2;
  }
  auto advance = [&]() {
#pragma unroll
    for (int og = 0; og < 4; ++og) {
      op4[og] += BK;
      wo4[og] += BK;
      xo4[og] += BK * stepw;
      while (wo4[og] >= Wo) {
        wo4[og] -= Wo;
        ho4[og] += 1;
        // corner moved: -Wo columns, +1 output row (stride rows of x)
        xo4[og] += rowfix_num - Wo * stepw;
        if (ho4[og] == Ho) {
          ho4[og] = 0;
          ++n4[og];
          // -Ho output rows, +1 image
          xo4[og] += (Hi - Ho * stride) * Wi * Cin * 2;
        }
      }
    }
  };

  auto issue_glds = [&](int buf) {
    char* rawA = smem + buf * BUF_STRIDE + RAW_A0;
    char* rawB = smem + buf * BUF_STRIDE + RAW_B0;
    // A: wave wid's 8-opix group; granule = (co-octet cs_*2+h_,
    // opix base + opoff)
    {
      int og = wid;
      int co8 = cs_ * 2 + h_;
      const char* src;
      if (op4[og] < Mi) {
        src = reinterpret_cast<const char*>(
            gy + (int64_t)op4[og] * Cout + tile_m + co8 * 8);
      } else {
        src = reinterpret_cast<const char*>(zero_page) + co8 * 16;
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(rawA + og * 1024),
          16, 0, 0);
    }
    const char* xb = reinterpret_cast<const char*>(x) +
                     (int64_t)(cs_ * 2 + h_) * 16;
    const char* zb = reinterpret_cast<const char*>(zero_page) +
                     (cs_ * 2 + h_) * 16;
#pragma unroll
    for (int i = 0; i < 9; ++i) {
      int c = i;
      int og = (i + wid) & 3;
      int hi = ho4[og] * stride - pad + cdh[c];
      int wi = wo4[og] * stride - pad + cdw[c];
      bool ok = cok[c] && op4[og] < Mi && (unsigned)hi < (unsigned)Hi &&
                (unsigned)wi < (unsigned)Wi;
      const char* src = ok ? xb + xo4[og] + coff[c] : zb;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              rawB + (c * 32 + og * 8) * 128),
          16, 0, 0);
    }
  };

  f32x4 acc[MREP][NREP] = {};
  int cur = 0;
  if (kt0 < kt1) {
    issue_glds(0);
    __syncthreads();  // carries vmcnt(0): deposits landed
  }
  for (int t = kt0; t < kt1; ++t) {
    if (t + 1 < kt1) {
      advance();
      issue_glds(cur ^ 1);  // next tile into the other buffer
    }
    {
      char* rawA = smem + cur * BUF_STRIDE + RAW_A0;
      char* rawB = smem + cur * BUF_STRIDE + RAW_B0;
      int g = lane >> 4;
      int l15 = lane & 15;
      auto baddr = [&](int ni) {
        int cs_abs = wid * 9 + ni;          // 16-tapci subtile index
        int c = cs_abs >> 2, cs = cs_abs & 3;
        return (unsigned)(size_t)(
            rawB + ((c * 8 + 2 * g) * 4 + cs) * 128 + l15 * 8);
      };
      // pipeline: all 4 A pairs + B0 issued up front (one drain), each
      // B(ni+1) issued before ni's MFMAs so its latency hides
      TrPair pa[MREP], pb0, pb1;
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        tr_issue((unsigned)(size_t)(
                     rawA + ((2 * g) * 4 + mi) * 128 + l15 * 8),
                 pa[mi]);
      tr_issue(baddr(0), pb0);
      bf16x8 a_frag[MREP];
#pragma unroll
      for (int mi = 0; mi < MREP; ++mi)
        a_frag[mi] = tr_wait<2>(pa[mi]);  // pb0 may stay in flight
#pragma unroll
      for (int ni = 0; ni < NREP; ++ni) {
        if (ni + 1 < NREP) tr_issue(baddr(ni + 1), pb1);
        bf16x8 b_frag =
            ni + 1 < NREP ? tr_wait<2>(pb0) : tr_wait<0>(pb0);
#pragma unroll
        for (int mi = 0; mi < MREP; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag, acc[mi][ni], 0, 0, 0);
        pb0 = pb1;
      }
    }
    __syncthreads();  // raw consumed + next deposits drained
    cur ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MREP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NREP; ++ni) {
      int col = tile_n + wid * WN + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = tile_m + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Np + col] = acc[mi][ni][j];
      }
    }
  }
}

// Slab-parallel two-stage reduce: the single-stage kernel below sizes
// its grid by mn (the OUTPUT, e.g. 36,864 elements for layer1) — at
// splitk=512 that is 36 blocks streaming 75 MB with 512 strided reads
// per element: ~450 us of pure latency, 52% of the whole v5 wrw call
// (r2o mode-5 ablation: the "empty" kernel + reduce cost 491 us).
// Stage 1 spreads GROUPS of slabs across gridDim.y (full chip);
// stage 2 folds the GROUPS (deterministic, no atomics).
__global__ void wrw_reduce_stage1_kernel(
    const float* __restrict__ partial, float* __restrict__ mid,
    int64_t mn_padded, int64_t slab_stride, int slabs, int groups) {
  int g = blockIdx.y;
  int s0 = (int)((int64_t)g * slabs / groups);
  int s1 = (int)((int64_t)(g + 1) * slabs / groups);
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < mn_padded;
       i += stride) {
    float v = 0.f;
    for (int s = s0; s < s1; ++s) v += partial[s * slab_stride + i];
    mid[(int64_t)g * mn_padded + i] = v;
  }
}

__global__ void wrw_reduce_stage2_kernel(
    const float* __restrict__ mid, float* __restrict__ out, int64_t mn,
    int64_t mn_padded, int groups, int Np, int64_t out_cols) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < mn;
       i += stride) {
    int64_t row = i / out_cols;
    int64_t col = i % out_cols;
    int64_t src = row * Np + col;
    float v = 0.f;
    for (int g = 0; g < groups; ++g) v += mid[(int64_t)g * mn_padded + src];
    out[i] = v;
  }
}

__global__ void wrw_reduce_kernel(const float* __restrict__ partial,
                                  float* __restrict__ out,
                                  int64_t mn, int64_t slab_stride,
                                  int slabs, int Np, int N_real,
                                  int64_t out_cols) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < mn;
       i += stride) {
    int64_t row = i / out_cols;
    int64_t col = i % out_cols;
    float v = 0.f;
    for (int s = 0; s < slabs; ++s)
      v += partial[(int64_t)s * slab_stride + row * Np + col];
    out[i] = v;
  }
}

// launches the right reduce for (slabs, mn): slab-parallel two-stage
// when the single-stage grid would underfill the chip
static void launch_wrw_reduce(const at::Tensor& partial, at::Tensor& gw,
                              int64_t Mp, int64_t Np, int slabs, int K,
                              int Cout, hipStream_t stream) {
  int64_t mn = (int64_t)Cout * K;
  int64_t mn_padded = Mp * Np;
  int rgrid = elementwise_grid(mn, kBlock, 4);
  if (slabs >= 32 && rgrid < 512) {
    int groups = std::min(32, slabs);
    auto mid = at::empty({groups, mn_padded},
                         partial.options().dtype(at::kFloat));
    int g1 = elementwise_grid(mn_padded, kBlock, 4);
    hipLaunchKernelGGL(wrw_reduce_stage1_kernel, dim3(g1, groups),
                       dim3(kBlock), 0, stream,
                       partial.data_ptr<float>(), mid.data_ptr<float>(),
                       mn_padded, mn_padded, slabs, groups);
    hipLaunchKernelGGL(wrw_reduce_stage2_kernel, dim3(rgrid),
                       dim3(kBlock), 0, stream, mid.data_ptr<float>(),
                       gw.data_ptr<float>(), mn, mn_padded, groups,
                       (int)Np, (int64_t)K);
  } else {
    hipLaunchKernelGGL(wrw_reduce_kernel, dim3(rgrid), dim3(kBlock), 0,
                       stream, partial.data_ptr<float>(),
                       gw.data_ptr<float>(), mn, mn_padded, slabs,
                       (int)Np, K, (int64_t)K);
  }
}


// ---- transpose-free TN GEMM: C[N,K] = sum_m A[m,N] * B[m,K] ------------
//
// The deep-K backward-weight GEMM (linear_bwd grad_w, 1x1-conv wrw):
// both operands are m-major, so the classic path materializes TWO
// transposes first (transpose2d + gemm_bt — ~2 extra full passes).
// Here both tiles stage ROW-MAJOR via glds in tr_b16-blocked form
// ([4 m][16 col] 128 B blocks — the per-lane source choice builds the
// blocking for free) and the MFMA fragments come from
// ds_read_b64_tr_b16 (semantics probed in scripts/probe_tr16.hip).
// 128x128 C tiles (2x2 waves), BKM=32 m per iteration, double-buffered
// 32 KiB LDS, split-K with the two-stage slab reduce.
namespace gemm_tn {
constexpr int BN = 128, BKC = 128, BKM = 32;
constexpr int WARPS = 2;  // 2x2
constexpr int WT = 64;    // per-wave C tile side
constexpr int REP = 4;
constexpr int TILE_B = BKM * BN * 2;  // 8 KiB per operand tile
}  // namespace gemm_tn

__global__ __launch_bounds__(256) void gemm_tn_kernel(
    const __hip_bfloat16* __restrict__ A,  // (M, N) row-major
    const __hip_bfloat16* __restrict__ B,  // (M, K) row-major
    const __hip_bfloat16* __restrict__ zero_page,
    float* __restrict__ partial,           // (slabs, Np, Kp)
    int64_t M, int N, int K, int Np, int Kp, int grid_k) {
  using namespace gemm_tn;
  __shared__ char smem[2 * 2 * TILE_B];  // [buf][A|B]
  auto sA = [&](int b) { return smem + b * 2 * TILE_B; };
  auto sB = [&](int b) { return smem + b * 2 * TILE_B + TILE_B; };

  int wg = blockIdx.x;
  int n0 = (wg / grid_k) * BN;
  int k0 = (wg % grid_k) * BKC;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;
  int wr = wid >> 1, wc = wid & 1;
  int g = lane >> 4;
  int l15 = lane & 15;

  int total_mt = (int)((M + BKM - 1) / BKM);
  int per = (total_mt + gridDim.y - 1) / gridDim.y;
  int mt0 = blockIdx.y * per;
  int mt1 = min(mt0 + per, total_mt);
  partial += (int64_t)blockIdx.y * Np * Kp;

  // staging: per wave 4 instrs (2 per operand): instr covers m-quad
  // (wid half? distribute 8 A + 8 B instrs over 4 waves: wave w does
  // A instrs {w, w+4} ... simpler: wave w stages A quads {2w, 2w+1}
  // via 2 instrs and B the same.
  // per-lane: ns = l>>3 (col subtile), r = (l&7)>>1, h = l&1
  int ns_ = lane >> 3;        // 0..7
  int r_ = (lane & 7) >> 1;   // m row within quad
  int h_ = lane & 1;          // 8-col half of the 16-col subtile

  auto stage = [&](int buf, int mt) {
    int64_t m_base = (int64_t)mt * BKM;
#pragma unroll
    for (int ii = 0; ii < 2; ++ii) {
      int mq = wid * 2 + ii;          // m quad 0..7
      int64_t m = m_base + mq * 4 + r_;
      int ca = n0 + ns_ * 16 + h_ * 8;
      const char* srcA =
          (m < M && ca + 8 <= N)
              ? reinterpret_cast<const char*>(A + m * N + ca)
              : reinterpret_cast<const char*>(zero_page) +
                    (ns_ & 7) * 16;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)srcA,
          (__attribute__((address_space(3))) void*)(sA(buf) + mq * 1024),
          16, 0, 0);
      int cb = k0 + ns_ * 16 + h_ * 8;
      const char* srcB =
          (m < M && cb + 8 <= K)
              ? reinterpret_cast<const char*>(B + m * K + cb)
              : reinterpret_cast<const char*>(zero_page) +
                    (ns_ & 7) * 16;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)srcB,
          (__attribute__((address_space(3))) void*)(sB(buf) + mq * 1024),
          16, 0, 0);
    }
  };

  f32x4 acc[REP][REP] = {};
  int cur = 0;
  if (mt0 < mt1) {
    stage(0, mt0);
    __syncthreads();
  }
  for (int t = mt0; t < mt1; ++t) {
    if (t + 1 < mt1) stage(cur ^ 1, t + 1);
    {
      // fragments: [4 m][16 col] blocks at (mq*8 + ns)*128;
      // wave row range: wr*64 + mi*16; col range: wc*64 + ni*16
      // NOTE the k-quad pair (h2 = 0,1 -> mq = 2g, 2g+1) sits 1024 B
      // apart in this 8-subtile layout, so paired tr_issue (+512)
      // does not apply — single reads with explicit offsets.
      unsigned long long qa[REP][2];
#pragma unroll
      for (int mi = 0; mi < REP; ++mi) {
        int ns = (wr * 64 + mi * 16) >> 4;
        unsigned a0 = (unsigned)(size_t)(
            sA(cur) + ((2 * g) * 8 + ns) * 128 + l15 * 8);
        tr_issue1(a0, qa[mi][0]);
        tr_issue1(a0 + 1024, qa[mi][1]);
      }
      bf16x8 a_frag[REP];
#pragma unroll
      for (int mi = 0; mi < REP; ++mi) {
        union {
          unsigned long long u[2];
          bf16x8 v;
        } ca;
        ca.u[0] = tr_wait1<0>(qa[mi][0]);
        ca.u[1] = tr_wait1<0>(qa[mi][1]);
        a_frag[mi] = ca.v;
      }
#pragma unroll
      for (int ni = 0; ni < REP; ++ni) {
        int ns = (wc * 64 + ni * 16) >> 4;
        unsigned b0a = (unsigned)(size_t)(
            sB(cur) + ((2 * g) * 8 + ns) * 128 + l15 * 8);
        unsigned long long qb0, qb1;
        tr_issue1(b0a, qb0);
        tr_issue1(b0a + 1024, qb1);
        union {
          unsigned long long u[2];
          bf16x8 v;
        } cb;
        cb.u[0] = tr_wait1<1>(qb0);
        cb.u[1] = tr_wait1<0>(qb1);
        bf16x8 b_frag = cb.v;
#pragma unroll
        for (int mi = 0; mi < REP; ++mi)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[mi], b_frag, acc[mi][ni], 0, 0, 0);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < REP; ++mi) {
#pragma unroll
    for (int ni = 0; ni < REP; ++ni) {
      int col = k0 + wc * WT + ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = n0 + wr * WT + mi * 16 + (lane >> 4) * 4 + j;
        partial[(int64_t)row * Kp + col] = acc[mi][ni][j];
      }
    }
  }
}

// C[N,K] = A[M,N]^T B[M,K], bf16 in, bf16 out (fp32 accumulate).
at::Tensor gemm_tn_bf16(const at::Tensor& A_in, const at::Tensor& B_in) {
  using namespace gemm_tn;
  auto A = A_in.contiguous();
  auto B = B_in.contiguous();
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16);
  TORCH_CHECK(A.dim() == 2 && B.dim() == 2 && A.size(0) == B.size(0));
  TORCH_CHECK(A.size(1) % 8 == 0 && B.size(1) % 8 == 0,
              "gemm_tn: N, K must be octet-aligned");
  int64_t M = A.size(0);
  int N = (int)A.size(1), K = (int)B.size(1);
  int gn = (N + BN - 1) / BN, gk = (K + BKC - 1) / BKC;
  int Np = gn * BN, Kp = gk * BKC;
  int tiles = gn * gk;
  int total_mt = (int)((M + BKM - 1) / BKM);
  int splitk = 1;
  while (tiles * splitk < 1024 && splitk * 2 <= total_mt && splitk < 512)
    splitk *= 2;
  auto stream = at::hip::getCurrentHIPStream();
  auto partial = at::empty({splitk, (int64_t)Np, (int64_t)Kp},
                           A.options().dtype(at::kFloat));
  static at::Tensor zp;
  if (!zp.defined() || zp.device() != A.device())
    zp = at::zeros({64}, A.options());
  hipLaunchKernelGGL(gemm_tn_kernel, dim3(tiles, splitk), dim3(256), 0,
                     stream,
                     reinterpret_cast<const __hip_bfloat16*>(A.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(B.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(zp.data_ptr()),
                     partial.data_ptr<float>(), M, N, K, Np, Kp, gk);
  // reduce slabs -> bf16 C
  auto Cf = at::empty({N, K}, A.options().dtype(at::kFloat));
  launch_wrw_reduce(partial, Cf, Np, Kp, splitk, K, N, stream);
  return Cf.to(at::kBFloat16);
}

// gy: (N, Cout, Ho, Wo) channels_last; x: (N, Cin, Hi, Wi) channels_last.
// Returns grad_weight (Cout, Cin, KH, KW) channels_last bf16.
at::Tensor conv2d_implicit_wrw(const at::Tensor& gy_in,
                               const at::Tensor& x_in, int64_t KH,
                               int64_t KW, int64_t stride, int64_t pad) {
  using namespace conv_wrw;
  auto gy = gy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto x = x_in.contiguous(at::MemoryFormat::ChannelsLast);
  int Nb = x.size(0), Cin = x.size(1), Hi = x.size(2), Wi = x.size(3);
  int Cout = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  TORCH_CHECK(Cin % 8 == 0, "wrw: Cin % 8 != 0");
  int64_t M = (int64_t)Nb * Ho * Wo;
  int K = (int)(KH * KW) * Cin;
  int Mp = (Cout + BM - 1) / BM * BM;
  int Np = (K + BN - 1) / BN * BN;

  int grid_m = Mp / BM, grid_n = Np / BN;
  int tiles = grid_m * grid_n;

  // kernel version: auto = v4 for 3x3 (glds staging + in-LDS
  // transpose), v2 otherwise; TURBOPRUNE_WRW=3 forces v3, =2 v2,
  // =1 v1, TURBOPRUNE_WRW_DB=1 the v1 double-buffered variant
  static int use_db = -1, use_v1 = -1, force_v2 = -1, force_v3 = -1;
  if (use_db < 0) {
    const char* e = getenv("TURBOPRUNE_WRW_DB");
    use_db = (e && e[0] == '1') ? 1 : 0;
    e = getenv("TURBOPRUNE_WRW");
    use_v1 = (e && e[0] == '1') ? 1 : 0;
    force_v2 = (e && e[0] == '2') ? 1 : 0;
    force_v3 = (e && e[0] == '3') ? 1 : 0;
  }
  auto stream = at::hip::getCurrentHIPStream();

  static int force_v4 = -1;
  if (force_v4 < 0) {
    const char* e = getenv("TURBOPRUNE_WRW");
    force_v4 = (e && e[0] == '4') ? 1 : 0;
  }
  static int force_v5 = -1, force_v6 = -1;
  if (force_v5 < 0) {
    const char* e = getenv("TURBOPRUNE_WRW");
    force_v5 = (e && e[0] == '5') ? 1 : 0;
    force_v6 = (e && e[0] == '6') ? 1 : 0;
  }
  static int force_v7 = -1;
  if (force_v7 < 0) {
    const char* e = getenv("TURBOPRUNE_WRW");
    force_v7 = (e && e[0] == '7') ? 1 : 0;
  }
  bool v7_ok = KH == 3 && KW == 3 && stride == 1 && pad == 1 &&
               Hi == Wi && Ho == Wo && Hi == Ho && Wo >= 28 &&
               (Wi * Cin * 2) % 1024 == 0 &&
               Wi * Cin * 2 <= conv_wrw7::MAXROW;
  // v7 measured SLOWER than v6 on its gated shapes (415/459 vs
  // 309/341 us) with a 4x larger (still in-tolerance) error — kept as
  // TURBOPRUNE_WRW=7 for round-3 work, not dispatched by default.
  if (force_v7 && v7_ok) {
    // v7: ring window + direct tr_b16 window fragments
    constexpr int BM7 = conv_wrw7::BM, BN7 = conv_wrw7::BN,
                  BK7 = conv_wrw7::BK;
    int gm = (Cout + BM7 - 1) / BM7;
    int gn = (K + BN7 - 1) / BN7;
    int Mp7 = gm * BM7;
    int Np7 = gn * BN7;
    int tiles7 = gm * gn;
    int total_kt = (int)((M + BK7 - 1) / BK7);
    int splitk = 1;
    while (tiles7 * splitk < 1024 && splitk * 2 <= total_kt &&
           splitk < 512)
      splitk *= 2;
    auto partial = at::empty({splitk, (int64_t)Mp7, (int64_t)Np7},
                             x.options().dtype(at::kFloat));
    static at::Tensor zp7;
    if (!zp7.defined() || zp7.device() != x.device())
      zp7 = at::zeros({64}, x.options());
    constexpr int kLds7 = conv_wrw7::LDS_BYTES;
    static bool attr7 = false;
    if (!attr7) {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(conv_wrw7_kernel),
          hipFuncAttributeMaxDynamicSharedMemorySize, kLds7);
      attr7 = true;
    }
    hipLaunchKernelGGL(conv_wrw7_kernel, dim3(tiles7, splitk), dim3(256),
                       kLds7, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(
                           zp7.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Cin, Cout,
                       Mp7, Np7, gn);
    auto gw = at::empty({Cout, Cin, KH, KW},
                        gy.options().dtype(at::kFloat).memory_format(
                            at::MemoryFormat::ChannelsLast));
    launch_wrw_reduce(partial, gw, Mp7, Np7, splitk, K, Cout, stream);
    return gw;
  }

  if ((force_v6 ||
       (!use_db && !use_v1 && !force_v2 && !force_v3 && !force_v4 &&
        !force_v5)) &&
      KH == 3 && KW == 3) {
    // v6: tr_b16 fragments straight from the glds-blocked raw image
    constexpr int BM6 = conv_wrw6::BM, BN6 = conv_wrw6::BN,
                  BK6 = conv_wrw6::BK;
    int gm = (Cout + BM6 - 1) / BM6;
    int gn = (K + BN6 - 1) / BN6;
    int Mp6 = gm * BM6;
    int Np6 = gn * BN6;
    int tiles6 = gm * gn;
    int total_kt = (int)((M + BK6 - 1) / BK6);
    int splitk = 1;
    while (tiles6 * splitk < 1024 && splitk * 2 <= total_kt &&
           splitk < 512)
      splitk *= 2;
    auto partial = at::empty({splitk, (int64_t)Mp6, (int64_t)Np6},
                             x.options().dtype(at::kFloat));
    static at::Tensor zp6;
    if (!zp6.defined() || zp6.device() != x.device())
      zp6 = at::zeros({64}, x.options());
    constexpr int kLds6 = 80 * 1024;
    static bool attr6 = false;
    if (!attr6) {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(conv_wrw6_kernel),
          hipFuncAttributeMaxDynamicSharedMemorySize, kLds6);
      attr6 = true;
    }
    hipLaunchKernelGGL(conv_wrw6_kernel, dim3(tiles6, splitk), dim3(256),
                       kLds6, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(
                           zp6.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Wi, Cin, Cout,
                       Ho, Wo, (int)KH, (int)KW, (int)stride, (int)pad,
                       Mp6, Np6, gn);
    auto gw = at::empty({Cout, Cin, KH, KW},
                        gy.options().dtype(at::kFloat).memory_format(
                            at::MemoryFormat::ChannelsLast));
    launch_wrw_reduce(partial, gw, Mp6, Np6, splitk, K, Cout, stream);
    return gw;
  }

  bool v5_ok = KH == 3 && KW == 3 && stride == 1 && pad == 1 &&
               Hi == Wi && Ho == Wo && Hi == Ho && Wo >= 28 &&
               (Wi * Cin * 2) % 1024 == 0 && Wi * Cin * 2 <= 7168;
  if (!use_db && !use_v1 && !force_v2 && !force_v3 && !force_v4 &&
      force_v5 && v5_ok) {
    // v5: persistent-ring-window (contiguous glds, x read ~once)
    constexpr int BM5 = conv_wrw5::BM, BN5 = conv_wrw5::BN,
                  BK5 = conv_wrw5::BK;
    int gm = (Cout + BM5 - 1) / BM5;
    int gn = (K + BN5 - 1) / BN5;
    int Mp5 = gm * BM5;
    int Np5 = gn * BN5;
    int tiles5 = gm * gn;
    int total_kt = (int)((M + BK5 - 1) / BK5);
    int splitk = 1;
    while (tiles5 * splitk < 1024 && splitk * 2 <= total_kt &&
           splitk < 512)
      splitk *= 2;
    auto partial = at::empty({splitk, (int64_t)Mp5, (int64_t)Np5},
                             x.options().dtype(at::kFloat));
    constexpr int kLds5 = 80 * 1024;
    static int mode5 = -1;
    if (mode5 < 0) {
      const char* e = getenv("TURBOPRUNE_WRW5_MODE");
      mode5 = e ? atoi(e) : 0;
      for (auto* f : {reinterpret_cast<const void*>(conv_wrw5_kernel<0>),
                      reinterpret_cast<const void*>(conv_wrw5_kernel<1>),
                      reinterpret_cast<const void*>(conv_wrw5_kernel<2>),
                      reinterpret_cast<const void*>(conv_wrw5_kernel<3>),
                      reinterpret_cast<const void*>(conv_wrw5_kernel<4>),
                      reinterpret_cast<const void*>(conv_wrw5_kernel<5>)})
        (void)hipFuncSetAttribute(
            f, hipFuncAttributeMaxDynamicSharedMemorySize, kLds5);
    }
    auto* kfn = mode5 == 1 ? conv_wrw5_kernel<1>
                : mode5 == 2 ? conv_wrw5_kernel<2>
                : mode5 == 3 ? conv_wrw5_kernel<3>
                : mode5 == 4 ? conv_wrw5_kernel<4>
                : mode5 == 5 ? conv_wrw5_kernel<5>
                             : conv_wrw5_kernel<0>;
    hipLaunchKernelGGL(kfn, dim3(tiles5, splitk), dim3(256),
                       kLds5, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Cin, Cout, Ho,
                       Mp5, Np5, gn);
    auto gw = at::empty({Cout, Cin, KH, KW},
                        gy.options().dtype(at::kFloat).memory_format(
                            at::MemoryFormat::ChannelsLast));
    launch_wrw_reduce(partial, gw, Mp5, Np5, splitk, K, Cout, stream);
    return gw;
  }

  if (!use_db && !use_v1 && !force_v2 && !force_v3 && KH == 3 &&
      KW == 3) {
    // v4: same tiling/split-K as v3, glds-staged
    constexpr int BM4 = conv_wrw4::BM, BN4 = conv_wrw4::BN,
                  BK4 = conv_wrw4::BK;
    int gm = (Cout + BM4 - 1) / BM4;
    int gn = (K + BN4 - 1) / BN4;
    int Mp4 = gm * BM4;
    int Np4 = gn * BN4;
    int tiles4 = gm * gn;
    int total_kt = (int)((M + BK4 - 1) / BK4);
    int splitk = 1;
    while (tiles4 * splitk < 1024 && splitk * 2 <= total_kt &&
           splitk < 512)
      splitk *= 2;
    auto partial = at::empty({splitk, (int64_t)Mp4, (int64_t)Np4},
                             x.options().dtype(at::kFloat));
    static at::Tensor zero_page;
    if (!zero_page.defined() || zero_page.device() != x.device())
      zero_page = at::zeros({64}, x.options());
    constexpr int kLds = 80 * 1024;
    static bool attr_set = false;
    if (!attr_set) {
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(conv_wrw4_kernel),
          hipFuncAttributeMaxDynamicSharedMemorySize, kLds);
      attr_set = true;
    }
    hipLaunchKernelGGL(conv_wrw4_kernel, dim3(tiles4, splitk), dim3(256),
                       kLds, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(
                           zero_page.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Wi, Cin, Cout,
                       Ho, Wo, (int)KH, (int)KW, (int)stride, (int)pad,
                       Mp4, Np4, gn);
    auto gw = at::empty({Cout, Cin, KH, KW},
                        gy.options().dtype(at::kFloat).memory_format(
                            at::MemoryFormat::ChannelsLast));
    launch_wrw_reduce(partial, gw, Mp4, Np4, splitk, K, Cout, stream);
    return gw;
  }

  if (!use_db && !use_v1 && !force_v2 && KH == 3 && KW == 3) {
    // v3: 64 x 576 tiles — gy re-read ceil(K/576)x, x re-read
    // ceil(Cout/64)x only via L2-absorbed tap overlap
    int gm = (Cout + conv_wrw3::BM - 1) / conv_wrw3::BM;
    int gn = (K + conv_wrw3::BN - 1) / conv_wrw3::BN;
    int Mp3 = gm * conv_wrw3::BM;
    int Np3 = gn * conv_wrw3::BN;
    int tiles3 = gm * gn;
    int total_kt = (int)((M + conv_wrw3::BK - 1) / conv_wrw3::BK);
    int splitk = 1;
    while (tiles3 * splitk < 1024 && splitk * 2 <= total_kt &&
           splitk < 512)
      splitk *= 2;
    auto partial = at::empty({splitk, (int64_t)Mp3, (int64_t)Np3},
                             x.options().dtype(at::kFloat));
    hipLaunchKernelGGL(conv_wrw3_kernel, dim3(tiles3, splitk), dim3(256),
                       0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Wi, Cin, Cout,
                       Ho, Wo, (int)KH, (int)KW, (int)stride, (int)pad,
                       Mp3, Np3, gn);
    auto gw = at::empty({Cout, Cin, KH, KW},
                        gy.options().dtype(at::kFloat).memory_format(
                            at::MemoryFormat::ChannelsLast));
    launch_wrw_reduce(partial, gw, Mp3, Np3, splitk, K, Cout, stream);
    return gw;
  }

  if (!use_db && !use_v1) {  // v2: BK=128, >=1024-workgroup split-K
    int total_kt = (int)((M + conv_wrw2::BK - 1) / conv_wrw2::BK);
    int splitk = 1;
    while (tiles * splitk < 1024 && splitk * 2 <= total_kt &&
           splitk < 256)
      splitk *= 2;
    auto partial = at::empty({splitk, (int64_t)Mp, (int64_t)Np},
                             x.options().dtype(at::kFloat));
    hipLaunchKernelGGL(conv_wrw2_kernel, dim3(tiles, splitk), dim3(256),
                       0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Wi, Cin, Cout,
                       Ho, Wo, (int)KH, (int)KW, (int)stride, (int)pad, Mp,
                       Np, grid_n);
    auto gw = at::empty({Cout, Cin, KH, KW},
                        gy.options().dtype(at::kFloat).memory_format(
                            at::MemoryFormat::ChannelsLast));
    launch_wrw_reduce(partial, gw, Mp, Np, splitk, K, Cout, stream);
    return gw;
  }

  int total_kt = (int)((M + BK - 1) / BK);
  int splitk = 1;
  while (tiles * splitk < 384 && splitk * 4 <= total_kt && splitk < 64)
    splitk *= 2;

  auto partial = at::empty({splitk, (int64_t)Mp, (int64_t)Np},
                           x.options().dtype(at::kFloat));
  dim3 grid(tiles, splitk);
  if (use_db) {
    static bool attr_set = false;
    if (!attr_set) {
      hipFuncSetAttribute(reinterpret_cast<const void*>(conv_wrw_db_kernel),
                          hipFuncAttributeMaxDynamicSharedMemorySize,
                          2 * 2 * BM * BK * 2);
      attr_set = true;
    }
    hipLaunchKernelGGL(conv_wrw_db_kernel, grid, dim3(256),
                       2 * 2 * BM * BK * 2, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Wi, Cin, Cout,
                       Ho, Wo, (int)KH, (int)KW, (int)stride, (int)pad, Mp,
                       Np, grid_n);
  } else {
    hipLaunchKernelGGL(conv_wrw_kernel, grid, dim3(256), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(gy.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       partial.data_ptr<float>(), Nb, Hi, Wi, Cin, Cout,
                       Ho, Wo, (int)KH, (int)KW, (int)stride, (int)pad, Mp,
                       Np, grid_n);
  }

  // reduce slabs -> (Cout, K) fp32, channels_last weight memory order
  auto gw = at::empty({Cout, Cin, KH, KW},
                      gy.options().dtype(at::kFloat).memory_format(
                          at::MemoryFormat::ChannelsLast));
  int64_t mn = (int64_t)Cout * K;
  int rgrid = elementwise_grid(mn, kBlock, 4);
  hipLaunchKernelGGL(wrw_reduce_kernel, dim3(rgrid), dim3(kBlock), 0,
                     stream, partial.data_ptr<float>(),
                     gw.data_ptr<float>(), mn,
                     (int64_t)Mp * Np, splitk, Np, K, (int64_t)K);
  return gw;
}

}  // namespace turboprune
