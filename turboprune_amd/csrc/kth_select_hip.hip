#include "hip/hip_runtime.h"
// Radix-select k-th smallest of an fp32 vector (the pruning
// threshold over up to ~25.6M scores) — no full sort.
//
// MSB-first 8-bit radix: 4 histogram sweeps over the data, each a
// memory-bound grid-stride pass (LDS-privatized 256-bin histograms,
// one global atomicAdd per bin per block), with a single-block scan
// kernel between passes keeping {prefix, k} state ON DEVICE — no host
// round-trips until the final 4-byte result copy.
//
// Floats are mapped to order-preserving uint32 keys
// (negative: ~u, else u | 0x80000000), so the selected 32-bit pattern
// after 4 passes IS the k-th value exactly (ties included).

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

TP_DEVICE uint32_t float_to_key(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

static inline float key_to_float_host(uint32_t m) {
  uint32_t u = (m & 0x80000000u) ? (m ^ 0x80000000u) : ~m;
  union { uint32_t u; float f; } cvt;
  cvt.u = u;
  return cvt.f;
}

// state layout: [0] = prefix (bits above shift+8), [1] = k (1-based)
__global__ void radix_hist_kernel(const float* __restrict__ vals, int64_t n,
                                  const uint64_t* __restrict__ state,
                                  unsigned int* __restrict__ hist,
                                  int shift) {
  __shared__ unsigned int lh[256];
  for (int t = threadIdx.x; t < 256; t += blockDim.x) lh[t] = 0;
  __syncthreads();
  uint32_t prefix = (uint32_t)state[0];
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint32_t key = float_to_key(vals[i]);
    bool match = (shift == 24) || ((key >> (shift + 8)) == prefix);
    if (match) atomicAdd(&lh[(key >> shift) & 255u], 1u);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < 256; t += blockDim.x)
    if (lh[t]) atomicAdd(&hist[t], lh[t]);
}

__global__ void radix_scan_kernel(unsigned int* __restrict__ hist,
                                  uint64_t* __restrict__ state, int shift) {
  // single block of 256 threads: exclusive scan + bin pick
  __shared__ unsigned long long cum[257];
  int t = threadIdx.x;
  cum[t + 1] = hist[t];
  if (t == 0) cum[0] = 0;
  __syncthreads();
  if (t == 0) {  // 256-step serial scan: trivial vs the sweep cost
    for (int i = 1; i <= 256; ++i) cum[i] += cum[i - 1];
    uint64_t k = state[1];
    int bin = 0;
    while (bin < 255 && cum[bin + 1] < k) ++bin;
    state[0] = (state[0] << 8) | (uint64_t)bin;
    state[1] = k - cum[bin];
  }
  __syncthreads();
  hist[t] = 0;  // ready for the next pass
}

double kth_smallest(const at::Tensor& values, int64_t k) {
  TORCH_CHECK(values.is_cuda() && values.dim() == 1);
  auto v = values.contiguous().to(at::kFloat);
  int64_t n = v.numel();
  TORCH_CHECK(1 <= k && k <= n, "kth_smallest: k out of range");
  auto stream = at::hip::getCurrentHIPStream();

  auto opts = v.options().dtype(at::kLong);
  auto state = at::zeros({2}, opts);
  state[1] = k;  // small H2D is fine pre-loop
  auto hist = at::zeros({256}, v.options().dtype(at::kInt));

  int grid = elementwise_grid(n, kBlock, 8);
  for (int pass = 0; pass < 4; ++pass) {
    int shift = 24 - 8 * pass;
    hipLaunchKernelGGL(radix_hist_kernel, dim3(grid), dim3(kBlock), 0,
                       stream, v.data_ptr<float>(), n,
                       reinterpret_cast<uint64_t*>(state.data_ptr<int64_t>()),
                       reinterpret_cast<unsigned int*>(hist.data_ptr<int>()),
                       shift);
    hipLaunchKernelGGL(radix_scan_kernel, dim3(1), dim3(256), 0, stream,
                       reinterpret_cast<unsigned int*>(hist.data_ptr<int>()),
                       reinterpret_cast<uint64_t*>(state.data_ptr<int64_t>()),
                       shift);
  }
  uint64_t key = (uint64_t)state[0].item<int64_t>();
  return (double)key_to_float_host((uint32_t)key);
}

}  // namespace turboprune
