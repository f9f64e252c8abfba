#include "hip/hip_runtime.h"
// Data-augmentation pipeline: fused RandomResizedCrop +
// horizontal flip + normalize, decoding CPU-provided uint8 HWC images into
// the NHWC compute tensor in one kernel. Crop geometry and the flip coin are
// drawn on the host (per-sample RNG parity with the reference's torchvision
// transforms, utils.py:131-137); the kernel does the bilinear resample.
//
// meta per image (int32 x 8): [byte_offset/4? no: elem offset, H, W, crop_y,
// crop_x, crop_h, crop_w, flip]
#include "common_hip.h"

namespace {

template <typename T>
__global__ void aug_kernel(const unsigned char* __restrict__ raw,
                           const int* __restrict__ meta, T* __restrict__ out,
                           int N, int S, float m0, float m1, float m2,
                           float s0, float s1, float s2) {
  const float mean[3] = {m0, m1, m2};
  const float stdi[3] = {1.f / s0, 1.f / s1, 1.f / s2};
  const int64_t total = (int64_t)N * S * S;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int x = i % S;
    int64_t t = i / S;
    const int y = t % S;
    const int n = t / S;
    const int* md = meta + n * 8;
    const int64_t off = *(const int*)&md[0];
    const int H = md[1], W = md[2];
    const int cy = md[3], cx = md[4], ch = md[5], cw = md[6], flip = md[7];
    const int xo = flip ? (S - 1 - x) : x;
    // bilinear sample of crop pixel (y,xo) scaled from (ch, cw) -> (S, S)
    const float sy = (float)ch / S, sx = (float)cw / S;
    float fy = (y + 0.5f) * sy - 0.5f + cy;
    float fx = (xo + 0.5f) * sx - 0.5f + cx;
    fy = fminf(fmaxf(fy, 0.f), H - 1.f);
    fx = fminf(fmaxf(fx, 0.f), W - 1.f);
    const int y0 = (int)fy, x0 = (int)fx;
    const int y1 = min(y0 + 1, H - 1), x1 = min(x0 + 1, W - 1);
    const float wy = fy - y0, wx = fx - x0;
    const unsigned char* img = raw + off;
#pragma unroll
    for (int c = 0; c < 3; ++c) {
      const float p00 = img[(y0 * W + x0) * 3 + c];
      const float p01 = img[(y0 * W + x1) * 3 + c];
      const float p10 = img[(y1 * W + x0) * 3 + c];
      const float p11 = img[(y1 * W + x1) * 3 + c];
      const float v = (p00 * (1 - wx) + p01 * wx) * (1 - wy) +
                      (p10 * (1 - wx) + p11 * wx) * wy;
      out[(((int64_t)n * S + y) * S + x) * 3 + c] =
          from_f32<T>((v / 255.f - mean[c]) * stdi[c]);
    }
  }
}

}  // namespace

// raw: flat uint8 buffer of concatenated HWC images; meta int32 [N, 8];
// returns NHWC (channels_last) [N, 3, S, S] in `dtype`.
at::Tensor aug_crop_flip_norm(at::Tensor raw, at::Tensor meta, int64_t S,
                              std::vector<double> mean,
                              std::vector<double> std, at::ScalarType dtype) {
  CHECK_GPU(raw);
  const int N = meta.size(0);
  auto out = at::empty({N, 3, S, S},
                       raw.options()
                           .dtype(dtype)
                           .memory_format(at::MemoryFormat::ChannelsLast));
  const int64_t total = (int64_t)N * S * S;
  DISPATCH_FLOAT_AND_BF16(dtype, "aug", [&] {
    hipLaunchKernelGGL((aug_kernel<scalar_t>), dim3(grid_1d(total, 256)),
                       dim3(256), 0, cur_stream(),
                       raw.data_ptr<unsigned char>(), meta.data_ptr<int>(),
                       (scalar_t*)out.data_ptr(), N, S, (float)mean[0],
                       (float)mean[1], (float)mean[2], (float)std[0],
                       (float)std[1], (float)std[2]);
  });
  return out;
}
