#include "hip/hip_runtime.h"
// CIFAR GPU augmentation kernels: random-translate crop
// from reflect-padded images, and cutout fill. Whole-epoch tensors stay
// GPU-resident (airbench design); these replace the torch-op versions.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

// out[n,c,y,x] = padded[n,c,y+sy[n],x+sx[n]]  (NCHW fp32, 32x32 out)
__global__ void crop_translate_kernel(const float* __restrict__ padded,
                                      float* __restrict__ out,
                                      const int64_t* __restrict__ shifts,
                                      int N, int C, int HP, int WP,
                                      int HO, int WO) {
  int64_t total = (int64_t)N * C * HO * WO;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int x = (int)(i % WO);
    int64_t r = i / WO;
    int y = (int)(r % HO);
    int64_t r2 = r / HO;
    int c = (int)(r2 % C);
    int n = (int)(r2 / C);
    int sy = (int)shifts[n * 2];
    int sx = (int)shifts[n * 2 + 1];
    out[i] = padded[(((int64_t)n * C + c) * HP + y + sy) * WP + x + sx];
  }
}

at::Tensor crop_translate(const at::Tensor& padded, int64_t out_size,
                          const at::Tensor& shifts) {
  TORCH_CHECK(padded.is_cuda() && padded.dim() == 4 &&
              padded.scalar_type() == at::kFloat && padded.is_contiguous());
  TORCH_CHECK(shifts.scalar_type() == at::kLong);
  int N = padded.size(0), C = padded.size(1), HP = padded.size(2),
      WP = padded.size(3);
  auto out = at::empty({N, C, out_size, out_size}, padded.options());
  auto sh = shifts.contiguous();
  int64_t total = (int64_t)N * C * out_size * out_size;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(crop_translate_kernel,
                     dim3(elementwise_grid(total, kBlock, 4)), dim3(kBlock),
                     0, stream, padded.data_ptr<float>(),
                     out.data_ptr<float>(), sh.data_ptr<int64_t>(), N, C,
                     HP, WP, (int)out_size, (int)out_size);
  return out;
}

// zero a size x size square centered at centers[n] in every channel
__global__ void cutout_kernel(float* __restrict__ images,
                              const int64_t* __restrict__ centers, int N,
                              int C, int H, int W, int half) {
  int64_t total = (int64_t)N * C * H * W;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int x = (int)(i % W);
    int64_t r = i / W;
    int y = (int)(r % H);
    int n = (int)(r / H / C);
    int cy = (int)centers[n * 2];
    int cx = (int)centers[n * 2 + 1];
    if (y >= cy - half && y <= cy + half && x >= cx - half &&
        x <= cx + half)
      images[i] = 0.f;
  }
}

void cutout_(at::Tensor images, const at::Tensor& centers, int64_t size) {
  TORCH_CHECK(images.is_cuda() && images.dim() == 4 &&
              images.scalar_type() == at::kFloat && images.is_contiguous());
  int N = images.size(0), C = images.size(1), H = images.size(2),
      W = images.size(3);
  auto ctr = centers.contiguous();
  int64_t total = (int64_t)N * C * H * W;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(cutout_kernel,
                     dim3(elementwise_grid(total, kBlock, 4)), dim3(kBlock),
                     0, stream, images.data_ptr<float>(),
                     ctr.data_ptr<int64_t>(), N, C, H, W, (int)(size / 2));
}

}  // namespace turboprune
