// Fused NHWC BatchNorm2d (+ReLU) (+residual add) forward/backward.
//
// Motivation (profiles/r01_bench_resnet50_1gpu_baseline.md): MIOpen's BN
// (6 kernels per layer per step) + eager ReLU + eager residual add are
// ~60% of the ResNet50 step's GPU time. These kernels replace them:
//
//   fwd (train): [reduce] per-channel sum/sumsq  ->  [finalize] stats +
//                running-stat update + scale/shift  ->  [apply]
//                y = relu(x*scale + shift (+ res)) — ONE elementwise pass
//   fwd (eval):  [apply] with running stats
//   bwd:         [reduce] per-channel sum(dy_eff), sum(dy_eff * xhat)
//                with dy_eff = dy * (y > 0) when ReLU was fused  ->
//                [finalize] dgamma/dbeta + dx coefficients  ->
//                [apply] dx (+ dres) — ONE elementwise pass
//
// Layout: NHWC (channels_last): the channel axis is fastest, so every
// access is a 16-byte (8 x bf16 / 4 x f32 "octet") vector load — scalar
// bf16 loads are 2-2.5x slower on CDNA4 (guide §6 G13). Reductions use
// a (row-blocks x channel-blocks) grid, 8 channel-octets x 32 row-lanes
// per 256-thread block, LDS tree over row-lanes, one atomicAdd per
// channel per block. Elementwise passes are flat 16B grid-stride with
// per-channel coefficients cached in LDS. Stats/params fp32.
//
// Fast path requires C % 8 == 0 (bf16) / C % 4 == 0 (f32) — true for
// every model in the zoo; other C fall back to torch composed ops at the
// Python layer.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

void colsum_atomic(const at::Tensor&, at::Tensor);   // elementwise.hip
void colsum2_atomic(const at::Tensor&, const at::Tensor&, at::Tensor,
                    at::Tensor);  // elementwise.hip

// element vector: 16 bytes of activations
template <typename T>
struct Octet;
template <>
struct Octet<__hip_bfloat16> {
  static constexpr int kN = 8;
  uint4 raw;
  TP_DEVICE float get(int j) const {
    const __hip_bfloat16* p = reinterpret_cast<const __hip_bfloat16*>(&raw);
    return __bfloat162float(p[j]);
  }
  TP_DEVICE void set(int j, float v) {
    __hip_bfloat16* p = reinterpret_cast<__hip_bfloat16*>(&raw);
    p[j] = __float2bfloat16(v);
  }
};
template <>
struct Octet<float> {
  static constexpr int kN = 4;
  uint4 raw;
  TP_DEVICE float get(int j) const {
    return reinterpret_cast<const float*>(&raw)[j];
  }
  TP_DEVICE void set(int j, float v) {
    reinterpret_cast<float*>(&raw)[j] = v;
  }
};

template <typename T>
TP_DEVICE Octet<T> load_octet(const T* p) {
  Octet<T> o;
  o.raw = *reinterpret_cast<const uint4*>(p);
  return o;
}
template <typename T>
TP_DEVICE void store_octet(T* p, const Octet<T>& o) {
  *reinterpret_cast<uint4*>(p) = o.raw;
}

// ---------------- forward reduce ----------------------------------------
// grid: (row_blocks, channel_blocks); block: 256 = 8 octets x 32 lanes.
//
// VALU budget: at 6.3 TB/s each CU must retire a 16B octet every ~6 VALU
// issue slots, so the inner loop uses explicit bit ops (bf16 -> f32 is a
// shift / a mask, no v_cvt) and packed f32x2 accumulators (v_pk_add_f32
// / v_pk_fma_f32): 2 shifts + 1 pk_add + 1 pk_fma per dword (2 elems).
typedef __attribute__((ext_vector_type(2))) float f32x2;

TP_DEVICE f32x2 bf16pair_to_f32x2(uint32_t d) {
  f32x2 v;
  v[0] = __uint_as_float(d << 16);
  v[1] = __uint_as_float(d & 0xffff0000u);
  return v;
}

// per-dword accumulate helpers (sum += v, acc2 += v*w)
template <typename T>
struct OctetAcc;
template <>
struct OctetAcc<__hip_bfloat16> {
  static constexpr int kP = 4;  // f32x2 pairs per octet
  TP_DEVICE static void unpack(const Octet<__hip_bfloat16>& o,
                               f32x2 (&v)[4]) {
    const uint32_t* d = reinterpret_cast<const uint32_t*>(&o.raw);
#pragma unroll
    for (int p = 0; p < 4; ++p) v[p] = bf16pair_to_f32x2(d[p]);
  }
  // channel index of pair p, slot s: 2*p + s
};
template <>
struct OctetAcc<float> {
  static constexpr int kP = 2;
  TP_DEVICE static void unpack(const Octet<float>& o, f32x2 (&v)[2]) {
    const float* f = reinterpret_cast<const float*>(&o.raw);
    v[0][0] = f[0]; v[0][1] = f[1];
    v[1][0] = f[2]; v[1][1] = f[3];
  }
};

// writes per-block partial sums to partial_sum/partial_ss[blockIdx.x][C]
// (plain stores; a single global accumulator serializes gridDim.x
// atomicAdds per channel — a ~400 us tail at 4096 blocks).
template <typename T>
__global__ void bn_reduce_kernel(const T* __restrict__ x, int64_t rows,
                                 int C, float* __restrict__ partial_sum,
                                 float* __restrict__ partial_ss) {
  constexpr int VN = Octet<T>::kN;
  constexpr int P = OctetAcc<T>::kP;
  __shared__ float lsum[8][Octet<T>::kN][32];
  __shared__ float lss[8][Octet<T>::kN][32];
  int oct_in_blk = threadIdx.x & 7;   // 8 octets per block
  int lane = threadIdx.x >> 3;        // 32 row-lanes
  int oct = blockIdx.y * 8 + oct_in_blk;
  int c0 = oct * VN;
  f32x2 s2[P], ss2[P];
#pragma unroll
  for (int p = 0; p < P; ++p) s2[p] = ss2[p] = f32x2{0.f, 0.f};
  if (c0 < C) {
    int64_t r = (int64_t)blockIdx.x * 32 + lane;
    int64_t rstep = (int64_t)gridDim.x * 32;
    for (; r + rstep < rows; r += 2 * rstep) {
      Octet<T> o0 = load_octet(x + r * C + c0);
      Octet<T> o1 = load_octet(x + (r + rstep) * C + c0);
      f32x2 v0[P], v1[P];
      OctetAcc<T>::unpack(o0, v0);
      OctetAcc<T>::unpack(o1, v1);
#pragma unroll
      for (int p = 0; p < P; ++p) {
        s2[p] += v0[p] + v1[p];
        ss2[p] += v0[p] * v0[p] + v1[p] * v1[p];
      }
    }
    for (; r < rows; r += rstep) {
      Octet<T> o = load_octet(x + r * C + c0);
      f32x2 v[P];
      OctetAcc<T>::unpack(o, v);
#pragma unroll
      for (int p = 0; p < P; ++p) {
        s2[p] += v[p];
        ss2[p] += v[p] * v[p];
      }
    }
  }
  float s[VN], ss[VN];
#pragma unroll
  for (int p = 0; p < P; ++p) {
    s[2 * p] = s2[p][0]; s[2 * p + 1] = s2[p][1];
    ss[2 * p] = ss2[p][0]; ss[2 * p + 1] = ss2[p][1];
  }
#pragma unroll
  for (int j = 0; j < VN; ++j) {
    lsum[oct_in_blk][j][lane] = s[j];
    lss[oct_in_blk][j][lane] = ss[j];
  }
  __syncthreads();
  // tree over the 32 lanes: threads with lane<16 fold
  for (int step = 16; step > 0; step >>= 1) {
    if (lane < step) {
#pragma unroll
      for (int j = 0; j < VN; ++j) {
        lsum[oct_in_blk][j][lane] += lsum[oct_in_blk][j][lane + step];
        lss[oct_in_blk][j][lane] += lss[oct_in_blk][j][lane + step];
      }
    }
    __syncthreads();
  }
  if (lane == 0 && c0 < C) {
    int64_t row = (int64_t)blockIdx.x * C;
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      partial_sum[row + c0 + j] = lsum[oct_in_blk][j][0];
      partial_ss[row + c0 + j] = lss[oct_in_blk][j][0];
    }
  }
}

// ---------------- forward finalize (one small launch) --------------------
__global__ void bn_finalize_kernel(const float* __restrict__ sum_in,
                                   const float* __restrict__ ss_in,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float* __restrict__ save_mean,
                                   float* __restrict__ save_rstd,
                                   float* __restrict__ scale,
                                   float* __restrict__ shift, int C,
                                   float inv_n, float unbiased_factor,
                                   float momentum, float eps,
                                   bool update_running) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean = sum_in[c] * inv_n;
  float var = fmaxf(ss_in[c] * inv_n - mean * mean, 0.f);
  float rstd = rsqrtf(var + eps);
  save_mean[c] = mean;
  save_rstd[c] = rstd;
  if (update_running) {
    running_mean[c] += momentum * (mean - running_mean[c]);
    running_var[c] += momentum * (var * unbiased_factor - running_var[c]);
  }
  float sc = gamma[c] * rstd;
  scale[c] = sc;
  shift[c] = beta[c] - mean * sc;
}

__global__ void bn_eval_coeffs_kernel(const float* __restrict__ running_mean,
                                      const float* __restrict__ running_var,
                                      const float* __restrict__ gamma,
                                      const float* __restrict__ beta,
                                      float* __restrict__ scale,
                                      float* __restrict__ shift, int C,
                                      float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float rstd = rsqrtf(running_var[c] + eps);
  float sc = gamma[c] * rstd;
  scale[c] = sc;
  shift[c] = beta[c] - running_mean[c] * sc;
}

// ---------------- forward apply ------------------------------------------
// flat grid-stride over 16B octets; scale/shift cached in LDS (C <= 4096)
// WM: also emit a per-octet ReLU bitmask (bit j = element j survived) so
// the backward never re-reads y (1/16 of the bytes).
template <typename T, bool RELU, bool RES, bool WM>
__global__ void bn_apply_kernel(const T* __restrict__ x,
                                const T* __restrict__ res,
                                T* __restrict__ y,
                                uint8_t* __restrict__ rmask,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                int64_t rows, int C) {
  constexpr int VN = Octet<T>::kN;
  extern __shared__ __attribute__((aligned(16))) float lds_coeff[];
  float* lscale = lds_coeff;
  float* lshift = lds_coeff + C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    lscale[c] = scale[c];
    lshift[c] = shift[c];
  }
  __syncthreads();
  int n_oct = C / VN;
  int64_t total = rows * n_oct;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c0 = (int)(i % n_oct) * VN;
    int64_t base = (i / n_oct) * C + c0;
    Octet<T> o = load_octet(x + base);
    Octet<T> r;
    if (RES) r = load_octet(res + base);
    Octet<T> out;
    unsigned mbits = 0;
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      float v = o.get(j) * lscale[c0 + j] + lshift[c0 + j];
      if (RES) v += r.get(j);
      if (RELU) {
        if (WM && v > 0.f) mbits |= (1u << j);
        v = fmaxf(v, 0.f);
      }
      out.set(j, v);
    }
    store_octet(y + base, out);
    if (RELU && WM) rmask[i] = (uint8_t)mbits;
  }
}

// ---------------- backward reduce ----------------------------------------
// accumulates sum_g = Σ dy_eff and sum_gx = Σ dy_eff * x per channel
// (the xhat projection Σ g·xhat = rstd·(sum_gx − mean·sum_g) is applied
// in the finalize kernel — keeps the hot loop at ~3 packed VALU/dword).
template <typename T, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ x,
                                     const uint8_t* __restrict__ rmask,
                                     const T* __restrict__ dy, int64_t rows,
                                     int C,
                                     float* __restrict__ partial_g,
                                     float* __restrict__ partial_gx) {
  constexpr int VN = Octet<T>::kN;
  constexpr int P = OctetAcc<T>::kP;
  __shared__ float lsum[8][Octet<T>::kN][32];
  __shared__ float lsx[8][Octet<T>::kN][32];
  int oct_in_blk = threadIdx.x & 7;
  int lane = threadIdx.x >> 3;
  int oct = blockIdx.y * 8 + oct_in_blk;
  int c0 = oct * VN;
  f32x2 s2[P], sx2[P];
#pragma unroll
  for (int p = 0; p < P; ++p) s2[p] = sx2[p] = f32x2{0.f, 0.f};
  if (c0 < C) {
    int64_t r = (int64_t)blockIdx.x * 32 + lane;
    int64_t rstep = (int64_t)gridDim.x * 32;
    int n_oct = C / VN;
    for (; r + rstep < rows; r += 2 * rstep) {
      int64_t b0 = r * C + c0, b1 = (r + rstep) * C + c0;
      Octet<T> ox0 = load_octet(x + b0), ox1 = load_octet(x + b1);
      Octet<T> og0 = load_octet(dy + b0), og1 = load_octet(dy + b1);
      unsigned m0 = 0xffu, m1 = 0xffu;
      if (RELU) {
        m0 = rmask[r * n_oct + oct];
        m1 = rmask[(r + rstep) * n_oct + oct];
      }
      f32x2 vx0[P], vg0[P], vx1[P], vg1[P];
      OctetAcc<T>::unpack(ox0, vx0);
      OctetAcc<T>::unpack(og0, vg0);
      OctetAcc<T>::unpack(ox1, vx1);
      OctetAcc<T>::unpack(og1, vg1);
#pragma unroll
      for (int p = 0; p < P; ++p) {
        f32x2 g0 = vg0[p], g1 = vg1[p];
        if (RELU) {
          g0[0] = (m0 >> (2 * p)) & 1 ? g0[0] : 0.f;
          g0[1] = (m0 >> (2 * p + 1)) & 1 ? g0[1] : 0.f;
          g1[0] = (m1 >> (2 * p)) & 1 ? g1[0] : 0.f;
          g1[1] = (m1 >> (2 * p + 1)) & 1 ? g1[1] : 0.f;
        }
        s2[p] += g0 + g1;
        sx2[p] += g0 * vx0[p] + g1 * vx1[p];
      }
    }
    for (; r < rows; r += rstep) {
      int64_t base = r * C + c0;
      Octet<T> ox = load_octet(x + base);
      Octet<T> og = load_octet(dy + base);
      unsigned mb = RELU ? rmask[r * n_oct + oct] : 0xffu;
      f32x2 vx[P], vg[P];
      OctetAcc<T>::unpack(ox, vx);
      OctetAcc<T>::unpack(og, vg);
#pragma unroll
      for (int p = 0; p < P; ++p) {
        f32x2 g = vg[p];
        if (RELU) {
          g[0] = (mb >> (2 * p)) & 1 ? g[0] : 0.f;
          g[1] = (mb >> (2 * p + 1)) & 1 ? g[1] : 0.f;
        }
        s2[p] += g;
        sx2[p] += g * vx[p];
      }
    }
  }
  float s[VN], sx[VN];
#pragma unroll
  for (int p = 0; p < P; ++p) {
    s[2 * p] = s2[p][0]; s[2 * p + 1] = s2[p][1];
    sx[2 * p] = sx2[p][0]; sx[2 * p + 1] = sx2[p][1];
  }
#pragma unroll
  for (int j = 0; j < VN; ++j) {
    lsum[oct_in_blk][j][lane] = s[j];
    lsx[oct_in_blk][j][lane] = sx[j];
  }
  __syncthreads();
  for (int step = 16; step > 0; step >>= 1) {
    if (lane < step) {
#pragma unroll
      for (int j = 0; j < VN; ++j) {
        lsum[oct_in_blk][j][lane] += lsum[oct_in_blk][j][lane + step];
        lsx[oct_in_blk][j][lane] += lsx[oct_in_blk][j][lane + step];
      }
    }
    __syncthreads();
  }
  if (lane == 0 && c0 < C) {
    int64_t row = (int64_t)blockIdx.x * C;
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      partial_g[row + c0 + j] = lsum[oct_in_blk][j][0];
      partial_gx[row + c0 + j] = lsx[oct_in_blk][j][0];
    }
  }
}

__global__ void bn_bwd_finalize_kernel(const float* __restrict__ sum_g_in,
                                       const float* __restrict__ sum_gx_in,
                                       const float* __restrict__ gamma,
                                       const float* __restrict__ save_mean,
                                       const float* __restrict__ save_rstd,
                                       float* __restrict__ dgamma,
                                       float* __restrict__ dbeta,
                                       float* __restrict__ c_dy,
                                       float* __restrict__ c_xhat,
                                       float* __restrict__ c_const, int C,
                                       float inv_n) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float sum_dy = sum_g_in[c];
  float sum_gx_c = sum_gx_in[c];
  // Σ g·xhat = rstd · (Σ g·x − mean·Σ g)
  float sum_dy_xhat = save_rstd[c] * (sum_gx_c - save_mean[c] * sum_dy);
  dgamma[c] = sum_dy_xhat;
  dbeta[c] = sum_dy;
  // dx = g*rstd * (dy_eff - inv_n*sum_dy - xhat*inv_n*sum_dy_xhat)
  float gr = gamma[c] * save_rstd[c];
  c_dy[c] = gr;
  c_xhat[c] = -gr * inv_n * sum_dy_xhat;
  c_const[c] = -gr * inv_n * sum_dy;
}

// ---------------- backward apply -----------------------------------------
// LDS caches 5 coeff arrays: mean, rstd, c_dy, c_xhat, c_const
template <typename T, bool RELU, bool RES>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ x,
                                    const uint8_t* __restrict__ rmask,
                                    const T* __restrict__ dy,
                                    T* __restrict__ dx,
                                    T* __restrict__ dres,
                                    const float* __restrict__ save_mean,
                                    const float* __restrict__ save_rstd,
                                    const float* __restrict__ c_dy,
                                    const float* __restrict__ c_xhat,
                                    const float* __restrict__ c_const,
                                    int64_t rows, int C) {
  constexpr int VN = Octet<T>::kN;
  extern __shared__ __attribute__((aligned(16))) float lds_coeff[];
  float* lmean = lds_coeff;
  float* lrstd = lds_coeff + C;
  float* lcdy = lds_coeff + 2 * C;
  float* lcx = lds_coeff + 3 * C;
  float* lcc = lds_coeff + 4 * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    lmean[c] = save_mean[c];
    lrstd[c] = save_rstd[c];
    lcdy[c] = c_dy[c];
    lcx[c] = c_xhat[c];
    lcc[c] = c_const[c];
  }
  __syncthreads();
  int n_oct = C / VN;
  int64_t total = rows * n_oct;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int c0 = (int)(i % n_oct) * VN;
    int64_t base = (i / n_oct) * C + c0;
    Octet<T> ox = load_octet(x + base);
    Octet<T> og = load_octet(dy + base);
    unsigned mb = RELU ? rmask[i] : 0xffu;
    Octet<T> odx, odr;
#pragma unroll
    for (int j = 0; j < VN; ++j) {
      int c = c0 + j;
      float g = og.get(j);
      if (RELU && !((mb >> j) & 1)) g = 0.f;
      if (RES) odr.set(j, g);
      float xhat = (ox.get(j) - lmean[c]) * lrstd[c];
      odx.set(j, lcdy[c] * g + lcx[c] * xhat + lcc[c]);
    }
    store_octet(dx + base, odx);
    if (RES) store_octet(dres + base, odr);
  }
}

// ---------------------------------------------------------------- host
struct BNShape {
  int64_t rows;
  int C;
};

static BNShape bn_shape(const at::Tensor& x) {
  TORCH_CHECK(x.dim() == 4, "expected NCHW tensor (channels_last layout)");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "fused BN requires channels_last activations");
  return {x.size(0) * x.size(2) * x.size(3), (int)x.size(1)};
}

// grid-shape knobs (A/B sweep on hardware; defaults = r01 tuning)
static int bn_env(const char* name, int dflt) {
  const char* e = getenv(name);
  return e && e[0] ? atoi(e) : dflt;
}

template <typename T>
static dim3 reduce_grid(int64_t rows, int C) {
  static int rpt = bn_env("TURBOPRUNE_BN_ROWS", 16);
  static int cap = bn_env("TURBOPRUNE_BN_RBCAP", 4096);
  int cb = (C / Octet<T>::kN + 7) / 8;   // channel blocks (8 octets each)
  int64_t rb_want = (rows + 32 * rpt - 1) / (32 * rpt);
  int rb = (int)std::min<int64_t>(std::max<int64_t>(rb_want, 1),
                                  std::max<int64_t>(cap / cb, 8));
  return dim3(rb, cb);
}

static int apply_grid(int64_t rows, int C, int vn) {
  // octets per thread + block cap: full-chip TLP to hide HBM latency
  // (a 2048-block cap left these passes latency-bound in r01)
  static int opt = bn_env("TURBOPRUNE_BN_OCT", 2);
  static int cap = bn_env("TURBOPRUNE_BN_BCAP", 16384);
  int64_t total = rows * (C / vn);
  int64_t blocks = (total + kBlock * opt - 1) / (kBlock * opt);
  return (int)std::min<int64_t>(std::max<int64_t>(blocks, 1), cap);
}

bool bn_fast_path_ok(const at::Tensor& x) {
  int C = (int)x.size(1);
  int vn = x.scalar_type() == at::kBFloat16 ? 8 : 4;
  return C % vn == 0 && C <= 4096;
}

// returns (y, save_mean, save_rstd, relu_mask-or-undef)
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_fwd(
    const at::Tensor& x, const c10::optional<at::Tensor>& residual,
    const at::Tensor& gamma, const at::Tensor& beta,
    at::Tensor running_mean, at::Tensor running_var, bool training,
    double momentum, double eps, bool relu) {
  auto sh = bn_shape(x);
  TORCH_CHECK(bn_fast_path_ok(x), "bn_fwd: C must be a multiple of the "
              "16B vector and <= 4096");
  auto fopt = x.options().dtype(at::kFloat);
  auto scale = at::empty({sh.C}, fopt);
  auto shift = at::empty({sh.C}, fopt);
  auto save_mean = at::empty({sh.C}, fopt);
  auto save_rstd = at::empty({sh.C}, fopt);
  auto stream = at::hip::getCurrentHIPStream();
  auto g = gamma.contiguous().to(at::kFloat);
  auto b = beta.contiguous().to(at::kFloat);

  int cblocks = (sh.C + kBlock - 1) / kBlock;

#define BN_DISPATCH_T(fn, ...)                                           \
  if (x.scalar_type() == at::kBFloat16) {                                \
    using T = __hip_bfloat16;                                            \
    fn(__VA_ARGS__);                                                     \
  } else {                                                               \
    using T = float;                                                     \
    fn(__VA_ARGS__);                                                     \
  }

  if (training) {
    dim3 rg;
    if (x.scalar_type() == at::kBFloat16)
      rg = reduce_grid<__hip_bfloat16>(sh.rows, sh.C);
    else
      rg = reduce_grid<float>(sh.rows, sh.C);
    auto partial_sum = at::empty({(int64_t)rg.x, sh.C}, fopt);
    auto partial_ss = at::empty({(int64_t)rg.x, sh.C}, fopt);
#define BN_RED()                                                          \
    hipLaunchKernelGGL((bn_reduce_kernel<T>), rg, dim3(kBlock), 0,        \
                       stream, reinterpret_cast<const T*>(x.data_ptr()),  \
                       sh.rows, sh.C, partial_sum.data_ptr<float>(),      \
                       partial_ss.data_ptr<float>())
    BN_DISPATCH_T(BN_RED)
#undef BN_RED
    float inv_n = 1.0f / (float)sh.rows;
    float unbiased = sh.rows > 1 ? (float)sh.rows / (sh.rows - 1) : 1.0f;
    auto sum = at::zeros({sh.C}, fopt);
    auto sumsq = at::zeros({sh.C}, fopt);
    colsum2_atomic(partial_sum, partial_ss, sum, sumsq);
    hipLaunchKernelGGL(bn_finalize_kernel, dim3(cblocks), dim3(kBlock), 0,
                       stream, sum.data_ptr<float>(),
                       sumsq.data_ptr<float>(),
                       g.data_ptr<float>(),
                       b.data_ptr<float>(),
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(),
                       save_mean.data_ptr<float>(),
                       save_rstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       sh.C, inv_n, unbiased, (float)momentum, (float)eps,
                       true);
  } else {
    hipLaunchKernelGGL(bn_eval_coeffs_kernel, dim3(cblocks), dim3(kBlock),
                       0, stream, running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), g.data_ptr<float>(),
                       b.data_ptr<float>(), scale.data_ptr<float>(),
                       shift.data_ptr<float>(), sh.C, (float)eps);
  }

  auto y = at::empty_like(x);
  bool has_res = residual.has_value() && residual->defined();
  at::Tensor res;
  if (has_res) {
    res = residual->contiguous(at::MemoryFormat::ChannelsLast);
    TORCH_CHECK(res.sizes() == x.sizes());
  }
  // ReLU bitmask (1 bit/elem, 1 byte/octet): written in training so the
  // backward never re-reads y
  int vn = x.scalar_type() == at::kBFloat16 ? 8 : 4;
  at::Tensor rmask;
  bool write_mask = relu && training;
  uint8_t* rmask_ptr = nullptr;
  if (write_mask) {
    rmask = at::empty({sh.rows * (sh.C / vn)}, x.options().dtype(at::kByte));
    rmask_ptr = rmask.data_ptr<uint8_t>();
  }
  size_t lds_bytes = 2 * sh.C * sizeof(float);
#define BN_APPLY(RELUF, RESF, WMF)                                        \
  hipLaunchKernelGGL(                                                     \
      (bn_apply_kernel<T, RELUF, RESF, WMF>),                             \
      dim3(apply_grid(sh.rows, sh.C, Octet<T>::kN)), dim3(kBlock),        \
      lds_bytes, stream, reinterpret_cast<const T*>(x.data_ptr()),        \
      RESF ? reinterpret_cast<const T*>(res.data_ptr()) : nullptr,        \
      reinterpret_cast<T*>(y.data_ptr()), rmask_ptr,                      \
      scale.data_ptr<float>(),                                            \
      shift.data_ptr<float>(), sh.rows, sh.C)
#define BN_APPLY_D()                                                      \
  if (relu && write_mask) {                                               \
    if (has_res) BN_APPLY(true, true, true);                              \
    else BN_APPLY(true, false, true);                                     \
  } else if (relu) {                                                      \
    if (has_res) BN_APPLY(true, true, false);                             \
    else BN_APPLY(true, false, false);                                    \
  } else {                                                                \
    if (has_res) BN_APPLY(false, true, false);                            \
    else BN_APPLY(false, false, false);                                   \
  }
  BN_DISPATCH_T(BN_APPLY_D)
#undef BN_APPLY_D
#undef BN_APPLY
  return {y, save_mean, save_rstd, rmask};
}

// returns (dx, dgamma, dbeta, dres?) — dres empty when no residual
std::tuple<at::Tensor, at::Tensor, at::Tensor, at::Tensor> bn_bwd(
    const at::Tensor& x, const at::Tensor& relu_mask,
    const at::Tensor& dy_in,
    const at::Tensor& gamma, const at::Tensor& save_mean,
    const at::Tensor& save_rstd, bool relu, bool has_res) {
  TORCH_CHECK(!relu || relu_mask.defined(),
              "bn_bwd: relu requires the forward's relu_mask");
  const uint8_t* rmask_ptr =
      relu ? relu_mask.data_ptr<uint8_t>() : nullptr;
  auto sh = bn_shape(x);
  auto dy = dy_in.contiguous(at::MemoryFormat::ChannelsLast);
  auto fopt = x.options().dtype(at::kFloat);
  auto stream = at::hip::getCurrentHIPStream();
  auto g = gamma.contiguous().to(at::kFloat);

  dim3 rg;
  if (x.scalar_type() == at::kBFloat16)
    rg = reduce_grid<__hip_bfloat16>(sh.rows, sh.C);
  else
    rg = reduce_grid<float>(sh.rows, sh.C);
  auto sum_dy = at::empty({(int64_t)rg.x, sh.C}, fopt);      // partial_g
  auto sum_dy_xhat = at::empty({(int64_t)rg.x, sh.C}, fopt); // partial_gx
#define BN_BRED(RELUF)                                                     \
  hipLaunchKernelGGL((bn_bwd_reduce_kernel<T, RELUF>), rg, dim3(kBlock),  \
                     0,                                                    \
                     stream, reinterpret_cast<const T*>(x.data_ptr()),     \
                     rmask_ptr,                                            \
                     reinterpret_cast<const T*>(dy.data_ptr()), sh.rows,   \
                     sh.C, sum_dy.data_ptr<float>(),                       \
                     sum_dy_xhat.data_ptr<float>())
#define BN_BRED_D()                                                        \
  if (relu) BN_BRED(true); else BN_BRED(false);
  BN_DISPATCH_T(BN_BRED_D)
#undef BN_BRED_D
#undef BN_BRED

  auto dgamma = at::empty({sh.C}, fopt);
  auto dbeta = at::empty({sh.C}, fopt);
  auto c_dy = at::empty({sh.C}, fopt);
  auto c_xhat = at::empty({sh.C}, fopt);
  auto c_const = at::empty({sh.C}, fopt);
  int cblocks = (sh.C + kBlock - 1) / kBlock;
  auto sg = at::zeros({sh.C}, fopt);
  auto sgx = at::zeros({sh.C}, fopt);
  colsum2_atomic(sum_dy, sum_dy_xhat, sg, sgx);
  hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3(cblocks), dim3(kBlock), 0,
                     stream, sg.data_ptr<float>(),
                     sgx.data_ptr<float>(),
                     g.data_ptr<float>(),
                     save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),
                     dgamma.data_ptr<float>(), dbeta.data_ptr<float>(),
                     c_dy.data_ptr<float>(), c_xhat.data_ptr<float>(),
                     c_const.data_ptr<float>(), sh.C,
                     1.0f / (float)sh.rows);

  auto dx = at::empty_like(x);
  at::Tensor dres;
  if (has_res) dres = at::empty_like(x);
  size_t lds_bytes = 5 * sh.C * sizeof(float);
#define BN_BAPPLY(RELUF, RESF)                                             \
  hipLaunchKernelGGL(                                                      \
      (bn_bwd_apply_kernel<T, RELUF, RESF>),                               \
      dim3(apply_grid(sh.rows, sh.C, Octet<T>::kN)), dim3(kBlock),         \
      lds_bytes, stream, reinterpret_cast<const T*>(x.data_ptr()),         \
      rmask_ptr,                                                           \
      reinterpret_cast<const T*>(dy.data_ptr()),                           \
      reinterpret_cast<T*>(dx.data_ptr()),                                 \
      RESF ? reinterpret_cast<T*>(dres.data_ptr()) : nullptr,              \
      save_mean.data_ptr<float>(), save_rstd.data_ptr<float>(),            \
      c_dy.data_ptr<float>(), c_xhat.data_ptr<float>(),                    \
      c_const.data_ptr<float>(), sh.rows, sh.C)
#define BN_BAPPLY_D()                                                      \
  if (relu) { if (has_res) BN_BAPPLY(true, true);                          \
              else BN_BAPPLY(true, false); }                               \
  else { if (has_res) BN_BAPPLY(false, true);                              \
         else BN_BAPPLY(false, false); }
  BN_DISPATCH_T(BN_BAPPLY_D)
#undef BN_BAPPLY_D
#undef BN_BAPPLY
  return {dx, dgamma, dbeta, dres};
}

}  // namespace turboprune
