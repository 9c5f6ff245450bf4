#include "hip/hip_runtime.h"
// EXPERIMENTAL — fused attention forward (flash-style online softmax)
// for the DeiT path: head_dim 64, non-causal, no dropout. NOT wired by
// default (ops/attention.py gates on TURBOPRUNE_ATTN=native after
// on-device validation, scripts/validate_attention.py).
//
// Plain-HIP ladder structure (guide §B: the pre-T16 rung — LDS-staged
// tiles, wave-parallel softmax, no glds/counted-vmcnt machinery; the
// tuned 8-wave register-staged combo is a round-3 target):
//   - one block (4 waves, 256 thr) per (batch*head, 64-row Q tile);
//     each wave owns 16 q rows end to end (its S/P/O fragments share
//     the same C-layout row group, so softmax stats never cross waves);
//   - K tiles of 64 keys staged to LDS like gemm_bt's B operand
//     ([key][d], K-major over d); S_t = Q K_t^T on MFMA, NREP=4;
//   - row max/sum via __shfl_xor over the 16-lane row group (guide:
//     wave-parallel softmax, NOT if(lane<16) serial loops);
//   - P_t (bf16) staged through LDS to become the next MFMA's A
//     operand; V_t transposed on staging (ds_write_b16 scatter ->
//     [d][key] image, the conv_wrw idiom) to be the B operand;
//   - O accumulates in f32 C fragments, rescaled by exp(m_old-m_new)
//     per tile; epilogue divides by l and stores bf16.
//
// All LDS images use the gemm XOR block swizzle (blk ^= row&7).

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

namespace attn {
constexpr int D = 64;      // head dim (fixed)
constexpr int BQ = 64;     // q rows per block
constexpr int BK = 64;     // keys per tile
// LDS image [64 rows][64 cols] bf16, 128 B rows, XOR-swizzled 16B blocks
TP_DEVICE int img(int row, int col) {
  int blk = (col >> 3) ^ (row & 7);
  return row * 128 + blk * 16 + (col & 7) * 2;
}
}  // namespace attn

__launch_bounds__(256)
__global__ void attn_fwd_kernel(const __hip_bfloat16* __restrict__ Q,
                                const __hip_bfloat16* __restrict__ K,
                                const __hip_bfloat16* __restrict__ V,
                                __hip_bfloat16* __restrict__ O,
                                float* __restrict__ Lse,  // (BH, S) m+log l
                                int S, float scale, int n_qtiles) {
  using namespace attn;
  // LDS: Q (8K) + K (8K) + Vt (8K) + P (8K)
  __shared__ char smem[4 * BQ * 128];
  char* sQ = smem;
  char* sK = smem + BQ * 128;
  char* sVt = smem + 2 * BQ * 128;
  char* sP = smem + 3 * BQ * 128;

  int bh = blockIdx.x / n_qtiles;          // batch*head
  int q0 = (blockIdx.x % n_qtiles) * BQ;   // first q row
  const __hip_bfloat16* q = Q + (int64_t)bh * S * D;
  const __hip_bfloat16* k = K + (int64_t)bh * S * D;
  const __hip_bfloat16* v = V + (int64_t)bh * S * D;
  __hip_bfloat16* o = O + (int64_t)bh * S * D;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;           // 0..3: q-row group

  // ---- stage Q tile [64 q][64 d] (once) ------------------------------
  // 256 thr x 16B: thread covers q row (tid>>3), d-octet (tid&7)... two
  // rows per thread iteration: 64 rows x 8 octets = 512 = 2 x 256.
  {
    for (int it = threadIdx.x; it < BQ * 8; it += blockDim.x) {
      int row = it >> 3, oc = it & 7;
      __hip_bfloat16 vals[8];
      if (q0 + row < S) {
        *reinterpret_cast<uint4*>(vals) = *reinterpret_cast<const uint4*>(
            q + (int64_t)(q0 + row) * D + oc * 8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[j] = __float2bfloat16(0.f);
      }
      *reinterpret_cast<uint4*>(sQ + img(row, oc * 8)) =
          *reinterpret_cast<const uint4*>(vals);
    }
  }
  __syncthreads();

  // per-lane softmax state: 4 q rows (C-layout j=0..3 of this lane)
  float m_i[4], l_i[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    m_i[j] = -1e30f;
    l_i[j] = 0.f;
  }
  f32x4 o_acc[4] = {};  // 16 q x 64 d per wave: NREP=4 over d

  // preload this wave's Q fragments (row group = wid*16)
  bf16x8 q_frag[2];  // k halves of d
#pragma unroll
  for (int kh = 0; kh < 2; ++kh)
    q_frag[kh] = *reinterpret_cast<const bf16x8*>(
        sQ + img(wid * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));

  for (int kt0 = 0; kt0 < S; kt0 += BK) {
    // ---- stage K tile [64 key][64 d] and V^T tile [64 d][64 key] -----
    for (int it = threadIdx.x; it < BK * 8; it += blockDim.x) {
      int row = it >> 3, oc = it & 7;
      __hip_bfloat16 vals[8];
      bool in = kt0 + row < S;
      if (in) {
        *reinterpret_cast<uint4*>(vals) = *reinterpret_cast<const uint4*>(
            k + (int64_t)(kt0 + row) * D + oc * 8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vals[j] = __float2bfloat16(0.f);
      }
      *reinterpret_cast<uint4*>(sK + img(row, oc * 8)) =
          *reinterpret_cast<const uint4*>(vals);
      // V row -> transposed scatter into [d][key]
      if (in) {
        *reinterpret_cast<uint4*>(vals) = *reinterpret_cast<const uint4*>(
            v + (int64_t)(kt0 + row) * D + oc * 8);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<__hip_bfloat16*>(sVt + img(oc * 8 + j, row)) =
            in ? vals[j] : __float2bfloat16(0.f);
    }
    __syncthreads();

    // ---- S_t = scale * Q K^T : 16 q x 64 key per wave ----------------
    f32x4 s_acc[4] = {};
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            sK + img(ni * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
        s_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[kh], kf, s_acc[ni], 0, 0, 0);
      }

    // mask the key tail and apply scale
    int valid = S - kt0;  // >= 1
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int key = ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        s_acc[ni][j] = key < valid ? s_acc[ni][j] * scale : -1e30f;
    }

    // ---- online softmax (wave-parallel row reduce) -------------------
    float m_new[4], p_sum[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float mx = -1e30f;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) mx = fmaxf(mx, s_acc[ni][j]);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        mx = fmaxf(mx, __shfl_xor(mx, off, kWave));
      m_new[j] = fmaxf(m_i[j], mx);
      float ps = 0.f;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        float e = __builtin_expf(s_acc[ni][j] - m_new[j]);
        s_acc[ni][j] = e;  // reuse as P
        ps += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        ps += __shfl_xor(ps, off, kWave);
      p_sum[j] = ps;
    }

    // ---- stage P (bf16) for the PV MFMA ------------------------------
    // C layout: lane holds rows (lane>>4)*4+j, col lane&15 -> scatter
    __syncthreads();  // sK/sVt reads done... (sP disjoint; this barrier
                      // orders the PREVIOUS pv reads of sP vs rewrite)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = wid * 16 + (lane >> 4) * 4 + j;
        int col = ni * 16 + (lane & 15);
        *reinterpret_cast<__hip_bfloat16*>(sP + img(row, col)) =
            __float2bfloat16(s_acc[ni][j]);
      }
    __syncthreads();

    // ---- O = O * exp(m_old - m_new) + P V ----------------------------
    float rescale[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      rescale[j] = __builtin_expf(m_i[j] - m_new[j]);
      l_i[j] = l_i[j] * rescale[j] + p_sum[j];
      m_i[j] = m_new[j];
    }
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int j = 0; j < 4; ++j) o_acc[ni][j] *= rescale[j];

    bf16x8 p_frag[2];
#pragma unroll
    for (int kh = 0; kh < 2; ++kh)
      p_frag[kh] = *reinterpret_cast<const bf16x8*>(
          sP + img(wid * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            sVt + img(ni * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
        o_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            p_frag[kh], vf, o_acc[ni], 0, 0, 0);
      }
    __syncthreads();  // done with sK/sVt/sP for this tile
  }

  // ---- epilogue: O / l, stage through sQ image, 16B stores -----------
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int row = wid * 16 + (lane >> 4) * 4 + j;
      int col = ni * 16 + (lane & 15);
      float denom = l_i[j] > 0.f ? l_i[j] : 1.f;
      *reinterpret_cast<__hip_bfloat16*>(sQ + img(row, col)) =
          __float2bfloat16(o_acc[ni][j] / denom);
    }
  // per-row logsumexp for the backward (every lane of a row group holds
  // the same stats; lane&15 == 0 writes)
  if (Lse != nullptr && (lane & 15) == 0) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int row = wid * 16 + (lane >> 4) * 4 + j;
      if (q0 + row < S)
        Lse[(int64_t)bh * S + q0 + row] =
            m_i[j] + __builtin_logf(l_i[j] > 0.f ? l_i[j] : 1.f);
    }
  }
  __syncthreads();
  for (int it = threadIdx.x; it < BQ * 8; it += blockDim.x) {
    int row = it >> 3, oc = it & 7;
    if (q0 + row < S) {
      __hip_bfloat16 vals[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vals[j] = *reinterpret_cast<const __hip_bfloat16*>(
            sQ + img(row, oc * 8 + j));
      *reinterpret_cast<uint4*>(o + (int64_t)(q0 + row) * D + oc * 8) =
          *reinterpret_cast<const uint4*>(vals);
    }
  }
}

// ---------------------------------------------------------------------
// Backward (flash-attention identities, recompute-P from the saved
// logsumexp):
//   P  = exp(scale*QK^T - Lse)          (zero for masked key tail)
//   dV += P^T dO                        (atomic fp32 across q-tiles)
//   dP = dO V^T
//   D  = rowsum(dO * O)
//   dS = scale * P ∘ (dP - D)
//   dQ = dS K                           (private per q-tile block)
//   dK += dS^T Q                        (atomic fp32)
// One block per (bh, 64-q-row tile); same 4-wave/16-rows-per-wave
// partition as the forward. Transposed operands (Q^T, dO^T, K^T, P^T /
// dS^T) are scatter-staged into their own LDS images so every MFMA
// fragment read stays one 16B ds_read.
__launch_bounds__(256)
__global__ void attn_bwd_kernel(const __hip_bfloat16* __restrict__ Q,
                                const __hip_bfloat16* __restrict__ K,
                                const __hip_bfloat16* __restrict__ V,
                                const __hip_bfloat16* __restrict__ O,
                                const __hip_bfloat16* __restrict__ dO,
                                const float* __restrict__ Lse,
                                __hip_bfloat16* __restrict__ dQ,
                                float* __restrict__ dK,
                                float* __restrict__ dV, int S, float scale,
                                int n_qtiles) {
  using namespace attn;
  // LDS: 9 images of 8 KiB + D row sums
  __shared__ char smem[9 * BQ * 128];
  __shared__ float sD[BQ];
  char* sQ = smem;
  char* sQt = smem + 1 * BQ * 128;
  char* sdO = smem + 2 * BQ * 128;
  char* sdOt = smem + 3 * BQ * 128;
  char* sK = smem + 4 * BQ * 128;
  char* sKt = smem + 5 * BQ * 128;
  char* sV = smem + 6 * BQ * 128;
  char* sP = smem + 7 * BQ * 128;   // [q][key]: dS image
  char* sPt = smem + 8 * BQ * 128;  // [key][q]: P^T then dS^T

  int bh = blockIdx.x / n_qtiles;
  int q0 = (blockIdx.x % n_qtiles) * BQ;
  const __hip_bfloat16* q = Q + (int64_t)bh * S * D;
  const __hip_bfloat16* k = K + (int64_t)bh * S * D;
  const __hip_bfloat16* v = V + (int64_t)bh * S * D;
  const __hip_bfloat16* o = O + (int64_t)bh * S * D;
  const __hip_bfloat16* go = dO + (int64_t)bh * S * D;
  __hip_bfloat16* gq = dQ + (int64_t)bh * S * D;
  float* gk = dK + (int64_t)bh * S * D;
  float* gv = dV + (int64_t)bh * S * D;

  int lane = threadIdx.x & (kWave - 1);
  int wid = threadIdx.x / kWave;

  // ---- stage Q, Q^T, dO, dO^T once; compute D = rowsum(dO * O) -------
  for (int it = threadIdx.x; it < BQ * 8; it += blockDim.x) {
    int row = it >> 3, oc = it & 7;
    __hip_bfloat16 qv[8], dv8[8];
    bool in = q0 + row < S;
    if (in) {
      *reinterpret_cast<uint4*>(qv) = *reinterpret_cast<const uint4*>(
          q + (int64_t)(q0 + row) * D + oc * 8);
      *reinterpret_cast<uint4*>(dv8) = *reinterpret_cast<const uint4*>(
          go + (int64_t)(q0 + row) * D + oc * 8);
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) qv[j] = dv8[j] = __float2bfloat16(0.f);
    }
    *reinterpret_cast<uint4*>(sQ + img(row, oc * 8)) =
        *reinterpret_cast<const uint4*>(qv);
    *reinterpret_cast<uint4*>(sdO + img(row, oc * 8)) =
        *reinterpret_cast<const uint4*>(dv8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<__hip_bfloat16*>(sQt + img(oc * 8 + j, row)) =
          qv[j];
      *reinterpret_cast<__hip_bfloat16*>(sdOt + img(oc * 8 + j, row)) =
          dv8[j];
    }
  }
  // D: 4 threads per row, 16 d each, pairwise shfl reduce
  {
    int row = threadIdx.x >> 2;     // 0..63
    int part = threadIdx.x & 3;
    float acc = 0.f;
    if (q0 + row < S) {
      const __hip_bfloat16* dor = go + (int64_t)(q0 + row) * D + part * 16;
      const __hip_bfloat16* orow = o + (int64_t)(q0 + row) * D + part * 16;
#pragma unroll
      for (int j = 0; j < 16; ++j)
        acc += __bfloat162float(dor[j]) * __bfloat162float(orow[j]);
    }
    acc += __shfl_xor(acc, 1, kWave);
    acc += __shfl_xor(acc, 2, kWave);
    if (part == 0) sD[row] = acc;
  }
  __syncthreads();

  // per-lane L (row stats) and dQ accumulator
  float l_row[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int row = wid * 16 + (lane >> 4) * 4 + j;
    l_row[j] = (q0 + row < S) ? Lse[(int64_t)bh * S + q0 + row] : 1e30f;
  }
  f32x4 dq_acc[4] = {};

  bf16x8 q_frag[2], do_frag[2];
#pragma unroll
  for (int kh = 0; kh < 2; ++kh) {
    q_frag[kh] = *reinterpret_cast<const bf16x8*>(
        sQ + img(wid * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
    do_frag[kh] = *reinterpret_cast<const bf16x8*>(
        sdO + img(wid * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
  }

  for (int kt0 = 0; kt0 < S; kt0 += BK) {
    // ---- stage K, K^T, V ---------------------------------------------
    for (int it = threadIdx.x; it < BK * 8; it += blockDim.x) {
      int row = it >> 3, oc = it & 7;
      __hip_bfloat16 kv8[8], vv8[8];
      bool in = kt0 + row < S;
      if (in) {
        *reinterpret_cast<uint4*>(kv8) = *reinterpret_cast<const uint4*>(
            k + (int64_t)(kt0 + row) * D + oc * 8);
        *reinterpret_cast<uint4*>(vv8) = *reinterpret_cast<const uint4*>(
            v + (int64_t)(kt0 + row) * D + oc * 8);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) kv8[j] = vv8[j] = __float2bfloat16(0.f);
      }
      *reinterpret_cast<uint4*>(sK + img(row, oc * 8)) =
          *reinterpret_cast<const uint4*>(kv8);
      *reinterpret_cast<uint4*>(sV + img(row, oc * 8)) =
          *reinterpret_cast<const uint4*>(vv8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *reinterpret_cast<__hip_bfloat16*>(sKt + img(oc * 8 + j, row)) =
            kv8[j];
    }
    __syncthreads();

    // ---- P = exp(scale * Q K^T - L), dP = dO V^T ---------------------
    f32x4 p_acc[4] = {}, dp_acc[4] = {};
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            sK + img(ni * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
        p_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[kh], kf, p_acc[ni], 0, 0, 0);
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            sV + img(ni * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
        dp_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            do_frag[kh], vf, dp_acc[ni], 0, 0, 0);
      }
    int valid = S - kt0;
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      int key = ni * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float p = (key < valid)
                      ? __builtin_expf(p_acc[ni][j] * scale - l_row[j])
                      : 0.f;
        p_acc[ni][j] = p;  // now holds P
      }
    }

    // ---- scatter P^T, dV += P^T dO -----------------------------------
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = wid * 16 + (lane >> 4) * 4 + j;
        int key = ni * 16 + (lane & 15);
        *reinterpret_cast<__hip_bfloat16*>(sPt + img(key, row)) =
            __float2bfloat16(p_acc[ni][j]);
      }
    __syncthreads();
    {
      f32x4 dv_acc[4] = {};
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int kh = 0; kh < 2; ++kh) {
          bf16x8 pf = *reinterpret_cast<const bf16x8*>(
              sPt + img(wid * 16 + (lane & 15),
                        kh * 32 + (lane >> 4) * 8));
          bf16x8 dof = *reinterpret_cast<const bf16x8*>(
              sdOt + img(ni * 16 + (lane & 15),
                         kh * 32 + (lane >> 4) * 8));
          dv_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              pf, dof, dv_acc[ni], 0, 0, 0);
        }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int key = wid * 16 + (lane >> 4) * 4 + j;
          int col = ni * 16 + (lane & 15);
          if (kt0 + key < S)
            atomicAdd(&gv[(int64_t)(kt0 + key) * D + col], dv_acc[ni][j]);
        }
    }

    // ---- dS = scale * P (dP - D); scatter to sP and (after the dV
    //      reads are done) sPt --------------------------------------
    __syncthreads();  // all waves done reading sPt (and sP free)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int row = wid * 16 + (lane >> 4) * 4 + j;
        int key = ni * 16 + (lane & 15);
        float ds = scale * p_acc[ni][j] * (dp_acc[ni][j] - sD[row]);
        __hip_bfloat16 b = __float2bfloat16(ds);
        *reinterpret_cast<__hip_bfloat16*>(sP + img(row, key)) = b;
        *reinterpret_cast<__hip_bfloat16*>(sPt + img(key, row)) = b;
      }
    __syncthreads();

    // ---- dQ += dS K (B from K^T image); dK += dS^T Q (B from Q^T) ----
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        bf16x8 dsf = *reinterpret_cast<const bf16x8*>(
            sP + img(wid * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
        bf16x8 ktf = *reinterpret_cast<const bf16x8*>(
            sKt + img(ni * 16 + (lane & 15), kh * 32 + (lane >> 4) * 8));
        dq_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf, ktf, dq_acc[ni], 0, 0, 0);
      }
    {
      f32x4 dk_acc[4] = {};
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int kh = 0; kh < 2; ++kh) {
          bf16x8 dstf = *reinterpret_cast<const bf16x8*>(
              sPt + img(wid * 16 + (lane & 15),
                        kh * 32 + (lane >> 4) * 8));
          bf16x8 qtf = *reinterpret_cast<const bf16x8*>(
              sQt + img(ni * 16 + (lane & 15),
                        kh * 32 + (lane >> 4) * 8));
          dk_acc[ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              dstf, qtf, dk_acc[ni], 0, 0, 0);
        }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          int key = wid * 16 + (lane >> 4) * 4 + j;
          int col = ni * 16 + (lane & 15);
          if (kt0 + key < S)
            atomicAdd(&gk[(int64_t)(kt0 + key) * D + col], dk_acc[ni][j]);
        }
    }
    __syncthreads();  // done with sK/sKt/sV/sP/sPt for this tile
  }

  // ---- epilogue: dQ through the sP image, 16B stores -----------------
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int row = wid * 16 + (lane >> 4) * 4 + j;
      int col = ni * 16 + (lane & 15);
      *reinterpret_cast<__hip_bfloat16*>(sP + img(row, col)) =
          __float2bfloat16(dq_acc[ni][j]);
    }
  __syncthreads();
  for (int it = threadIdx.x; it < BQ * 8; it += blockDim.x) {
    int row = it >> 3, oc = it & 7;
    if (q0 + row < S) {
      __hip_bfloat16 vals[8];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        vals[j] = *reinterpret_cast<const __hip_bfloat16*>(
            sP + img(row, oc * 8 + j));
      *reinterpret_cast<uint4*>(gq + (int64_t)(q0 + row) * D + oc * 8) =
          *reinterpret_cast<const uint4*>(vals);
    }
  }
}

// (dq bf16, dk fp32, dv fp32) — caller casts dk/dv.
std::tuple<at::Tensor, at::Tensor, at::Tensor> attn_bwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& o, const at::Tensor& grad_o, const at::Tensor& lse,
    double scale) {
  using namespace attn;
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 &&
              q.dim() == 4 && q.size(3) == D);
  auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
  auto oc = o.contiguous(), doc = grad_o.contiguous();
  auto lc = lse.contiguous();
  int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  auto dq = at::empty_like(qc);
  auto dk = at::zeros({B, H, S, (int64_t)D},
                      q.options().dtype(at::kFloat));
  auto dv = at::zeros_like(dk);
  int n_qtiles = (int)((S + BQ - 1) / BQ);
  int blocks = (int)(B * H) * n_qtiles;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(
      attn_bwd_kernel, dim3(blocks), dim3(256), 0, stream,
      reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(oc.data_ptr()),
      reinterpret_cast<const __hip_bfloat16*>(doc.data_ptr()),
      lc.data_ptr<float>(),
      reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()),
      dk.data_ptr<float>(), dv.data_ptr<float>(), (int)S, (float)scale,
      n_qtiles);
  return {dq, dk, dv};
}

// q, k, v: (B, H, S, 64) contiguous bf16 -> (O same shape,
// Lse (B, H, S) fp32 row logsumexp for the backward).
std::tuple<at::Tensor, at::Tensor> attn_fwd(const at::Tensor& q,
                                            const at::Tensor& k,
                                            const at::Tensor& v,
                                            double scale) {
  using namespace attn;
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == at::kBFloat16 &&
              q.dim() == 4 && q.size(3) == D,
              "attn_fwd: (B,H,S,64) bf16 required");
  auto qc = q.contiguous(), kc = k.contiguous(), vc = v.contiguous();
  int64_t B = q.size(0), H = q.size(1), S = q.size(2);
  TORCH_CHECK(k.sizes() == q.sizes() && v.sizes() == q.sizes());
  auto out = at::empty_like(qc);
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  int n_qtiles = (int)((S + BQ - 1) / BQ);
  int blocks = (int)(B * H) * n_qtiles;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(attn_fwd_kernel, dim3(blocks), dim3(256), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()),
                     reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                     lse.data_ptr<float>(), (int)S, (float)scale,
                     n_qtiles);
  return {out, lse};
}

}  // namespace turboprune
