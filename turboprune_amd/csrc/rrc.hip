// Fused RandomResizedCrop + horizontal flip + normalize (SURVEY K12:
// the post-decode train augmentation of the FFCV ImageNet pipeline,
// reference dataset.py:385-392). One pass: uint8 NCHW source -> bilinear
// sample of a per-image crop box -> optional flip -> (x/255-mean)/std ->
// bf16/f32 NCHW output.

#include <ATen/ATen.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace turboprune {

// boxes: int64 [N,4] = (top, left, height, width) per image
template <typename OutT>
__global__ void rrc_kernel(const uint8_t* __restrict__ in,
                           OutT* __restrict__ out,
                           const int64_t* __restrict__ boxes,
                           const bool* __restrict__ flip,
                           const float* __restrict__ mean,
                           const float* __restrict__ std, int N, int C,
                           int Hi, int Wi, int Ho, int Wo) {
  int64_t total = (int64_t)N * C * Ho * Wo;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int x = (int)(i % Wo);
    int64_t r = i / Wo;
    int y = (int)(r % Ho);
    int64_t r2 = r / Ho;
    int c = (int)(r2 % C);
    int n = (int)(r2 / C);

    int top = (int)boxes[n * 4];
    int left = (int)boxes[n * 4 + 1];
    int ch = (int)boxes[n * 4 + 2];
    int cw = (int)boxes[n * 4 + 3];
    int xo = (flip != nullptr && flip[n]) ? (Wo - 1 - x) : x;

    // bilinear sample in CROP space (align_corners=False); clamp to the
    // crop edges like F.interpolate on the extracted crop would
    float sy = (float)ch / Ho, sx = (float)cw / Wo;
    float fy = ((float)y + 0.5f) * sy - 0.5f;
    float fx = ((float)xo + 0.5f) * sx - 0.5f;
    int y0 = (int)floorf(fy), x0 = (int)floorf(fx);
    float wy = fy - y0, wx = fx - x0;
    int y1 = y0 + 1, x1 = x0 + 1;
    y0 = min(max(y0, 0), ch - 1) + top;
    y1 = min(max(y1, 0), ch - 1) + top;
    x0 = min(max(x0, 0), cw - 1) + left;
    x1 = min(max(x1, 0), cw - 1) + left;
    const uint8_t* src = in + ((int64_t)n * C + c) * Hi * Wi;
    float v00 = src[(int64_t)y0 * Wi + x0];
    float v01 = src[(int64_t)y0 * Wi + x1];
    float v10 = src[(int64_t)y1 * Wi + x0];
    float v11 = src[(int64_t)y1 * Wi + x1];
    float v = (v00 * (1 - wx) + v01 * wx) * (1 - wy) +
              (v10 * (1 - wx) + v11 * wx) * wy;
    v = (v * (1.0f / 255.0f) - mean[c]) / std[c];
    out[i] = from_float<OutT>(v);
  }
}

at::Tensor random_resized_crop(const at::Tensor& images,
                               const at::Tensor& boxes,
                               const at::Tensor& flip,
                               const at::Tensor& mean,
                               const at::Tensor& std, int64_t out_size,
                               at::ScalarType out_dtype) {
  TORCH_CHECK(images.is_cuda() && images.dim() == 4 &&
              images.scalar_type() == at::kByte && images.is_contiguous());
  TORCH_CHECK(boxes.scalar_type() == at::kLong && boxes.size(1) == 4);
  int N = images.size(0), C = images.size(1), Hi = images.size(2),
      Wi = images.size(3);
  auto out = at::empty({N, C, out_size, out_size},
                       images.options().dtype(out_dtype));
  auto bx = boxes.contiguous();
  auto m = mean.contiguous().to(images.device(), at::kFloat);
  auto s = std.contiguous().to(images.device(), at::kFloat);
  const bool* fp = nullptr;
  at::Tensor f;
  if (flip.defined() && flip.numel() > 0) {
    f = flip.contiguous().to(at::kBool);
    fp = f.data_ptr<bool>();
  }
  int64_t total = (int64_t)N * C * out_size * out_size;
  int grid = elementwise_grid(total, kBlock, 4);
  auto stream = at::hip::getCurrentHIPStream();
  if (out_dtype == at::kBFloat16) {
    hipLaunchKernelGGL(rrc_kernel<__hip_bfloat16>, dim3(grid), dim3(kBlock),
                       0, stream, images.data_ptr<uint8_t>(),
                       reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                       bx.data_ptr<int64_t>(), fp, m.data_ptr<float>(),
                       s.data_ptr<float>(), N, C, Hi, Wi, (int)out_size,
                       (int)out_size);
  } else {
    TORCH_CHECK(out_dtype == at::kFloat);
    hipLaunchKernelGGL(rrc_kernel<float>, dim3(grid), dim3(kBlock), 0,
                       stream, images.data_ptr<uint8_t>(),
                       out.data_ptr<float>(), bx.data_ptr<int64_t>(), fp,
                       m.data_ptr<float>(), s.data_ptr<float>(), N, C, Hi,
                       Wi, (int)out_size, (int)out_size);
  }
  return out;
}

}  // namespace turboprune
