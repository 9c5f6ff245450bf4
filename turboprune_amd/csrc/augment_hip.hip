#include "hip/hip_runtime.h"
// GPU data-pipeline kernels (SURVEY K11/K12 post-decode stage):
// fused uint8 -> normalized float/bf16 with optional horizontal flip.
// One pass over the batch: replaces to(float) + div + sub + div + flip
// (5 eager kernels) with one memory-bound sweep.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

template <typename OutT>
__global__ void normalize_u8_kernel(const uint8_t* __restrict__ in,
                                    OutT* __restrict__ out,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ std,
                                    const bool* __restrict__ flip, int N,
                                    int C, int H, int W) {
  int64_t total = (int64_t)N * C * H * W;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int w = i % W;
    int64_t rest = i / W;
    int h = rest % H;
    int64_t rest2 = rest / H;
    int c = rest2 % C;
    int n = rest2 / C;
    int src_w = (flip != nullptr && flip[n]) ? (W - 1 - w) : w;
    int64_t src = (((int64_t)n * C + c) * H + h) * W + src_w;
    float v = (in[src] * (1.0f / 255.0f) - mean[c]) / std[c];
    out[i] = from_float<OutT>(v);
  }
}

at::Tensor normalize_u8(const at::Tensor& images, const at::Tensor& mean,
                        const at::Tensor& std, const at::Tensor& flip,
                        at::ScalarType out_dtype) {
  TORCH_CHECK(images.is_cuda() && images.dim() == 4 &&
              images.scalar_type() == at::kByte && images.is_contiguous());
  int N = images.size(0), C = images.size(1), H = images.size(2),
      W = images.size(3);
  auto m = mean.contiguous().to(images.device(), at::kFloat);
  auto s = std.contiguous().to(images.device(), at::kFloat);
  const bool* fp = nullptr;
  at::Tensor f;
  if (flip.defined() && flip.numel() > 0) {
    f = flip.contiguous().to(at::kBool);
    fp = f.data_ptr<bool>();
  }
  auto out = at::empty({N, C, H, W}, images.options().dtype(out_dtype));
  int64_t total = (int64_t)N * C * H * W;
  auto stream = at::hip::getCurrentHIPStream();
  int grid = elementwise_grid(total, kBlock, 4);
  if (out_dtype == at::kFloat) {
    hipLaunchKernelGGL(normalize_u8_kernel<float>, dim3(grid), dim3(kBlock),
                       0, stream, images.data_ptr<uint8_t>(),
                       out.data_ptr<float>(), m.data_ptr<float>(),
                       s.data_ptr<float>(), fp, N, C, H, W);
  } else if (out_dtype == at::kBFloat16) {
    hipLaunchKernelGGL(normalize_u8_kernel<__hip_bfloat16>, dim3(grid),
                       dim3(kBlock), 0, stream, images.data_ptr<uint8_t>(),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       m.data_ptr<float>(), s.data_ptr<float>(), fp, N, C, H,
                       W);
  } else {
    TORCH_CHECK(false, "normalize_u8: unsupported out dtype");
  }
  return out;
}

}  // namespace turboprune
