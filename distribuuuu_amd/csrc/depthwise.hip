// Depthwise convolution: groups == C, one filter tap set per
// channel. No MFMA (K-dim per group = R*S): memory-bound elementwise-style
// kernels over NHWC with 8-channel packs; weights [C,1,R,S] (channels_last ->
// physical [C][R][S]) are L2-resident.
#include "common.h"

namespace {

template <typename T>
__global__ void dwconv_fwd_kernel(const T* __restrict__ x,
                                  const T* __restrict__ w, T* __restrict__ y,
                                  int N, int H, int W, int C, int Ho, int Wo,
                                  int R, int S, int sh, int sw, int ph, int pw) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpacks = C / V;
  const int64_t total = (int64_t)N * Ho * Wo * cpacks;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int cp = i % cpacks;
    int64_t t = i / cpacks;
    const int wo = t % Wo;
    t /= Wo;
    const int ho = t % Ho;
    const int n = t / Ho;
    float acc[V] = {};
    const int h0 = ho * sh - ph, w0 = wo * sw - pw;
    for (int r = 0; r < R; ++r) {
      const int h = h0 + r;
      if (h < 0 || h >= H) continue;
      for (int s = 0; s < S; ++s) {
        const int w_ = w0 + s;
        if (w_ < 0 || w_ >= W) continue;
        P px = *reinterpret_cast<const P*>(
            x + (((int64_t)n * H + h) * W + w_) * C + cp * V);
#pragma unroll
        for (int j = 0; j < V; ++j) {
          const int c = cp * V + j;
          acc[j] += to_f32(px.v[j]) * to_f32(w[(c * R + r) * S + s]);
        }
      }
    }
    P py;
#pragma unroll
    for (int j = 0; j < V; ++j) py.v[j] = from_f32<T>(acc[j]);
    *reinterpret_cast<P*>(y + (((int64_t)n * Ho + ho) * Wo + wo) * C +
                          cp * V) = py;
  }
}

template <typename T>
__global__ void dwconv_dgrad_kernel(const T* __restrict__ gy,
                                    const T* __restrict__ w,
                                    T* __restrict__ gx, int N, int H, int W,
                                    int C, int Ho, int Wo, int R, int S,
                                    int sh, int sw, int ph, int pw) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpacks = C / V;
  const int64_t total = (int64_t)N * H * W * cpacks;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int cp = i % cpacks;
    int64_t t = i / cpacks;
    const int w_ = t % W;
    t /= W;
    const int h = t % H;
    const int n = t / H;
    float acc[V] = {};
    for (int r = 0; r < R; ++r) {
      const int hn = h + ph - r;
      if (hn < 0 || hn % sh != 0) continue;
      const int ho = hn / sh;
      if (ho >= Ho) continue;
      for (int s = 0; s < S; ++s) {
        const int wn = w_ + pw - s;
        if (wn < 0 || wn % sw != 0) continue;
        const int wo = wn / sw;
        if (wo >= Wo) continue;
        P pg = *reinterpret_cast<const P*>(
            gy + (((int64_t)n * Ho + ho) * Wo + wo) * C + cp * V);
#pragma unroll
        for (int j = 0; j < V; ++j) {
          const int c = cp * V + j;
          acc[j] += to_f32(pg.v[j]) * to_f32(w[(c * R + r) * S + s]);
        }
      }
    }
    P px;
#pragma unroll
    for (int j = 0; j < V; ++j) px.v[j] = from_f32<T>(acc[j]);
    *reinterpret_cast<P*>(gx + (((int64_t)n * H + h) * W + w_) * C + cp * V) =
        px;
  }
}

// wgrad: per (c, r, s) reduce over pixels; block-local LDS tree then one
// fp32 atomic per (c,r,s) per block.
template <typename T>
__global__ void dwconv_wgrad_kernel(const T* __restrict__ gy,
                                    const T* __restrict__ x,
                                    float* __restrict__ gw, int N, int H,
                                    int W, int C, int Ho, int Wo, int R,
                                    int S, int sh, int sw, int ph, int pw,
                                    int64_t pix_per_block) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  __shared__ float red[256 * (16 / sizeof(T) > 8 ? 16 / sizeof(T) : 8)];
  const int cpacks = C / V;
  const int rs = blockIdx.y;  // tap index
  const int r = rs / S, s = rs % S;
  const int64_t pixels = (int64_t)N * Ho * Wo;
  const int64_t p0 = (int64_t)blockIdx.x * pix_per_block;
  const int64_t p1 = min(p0 + pix_per_block, pixels);
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = 1 << (31 - __builtin_clz((int)blockDim.x / ncp));
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float acc[V] = {};
    if (active) {
      for (int64_t pix = p0 + rl; pix < p1; pix += nrl) {
        const int wo = pix % Wo;
        int64_t t2 = pix / Wo;
        const int ho = t2 % Ho;
        const int n = t2 / Ho;
        const int h = ho * sh - ph + r;
        const int w_ = wo * sw - pw + s;
        if (h < 0 || h >= H || w_ < 0 || w_ >= W) continue;
        P pg = *reinterpret_cast<const P*>(
            gy + (((int64_t)n * Ho + ho) * Wo + wo) * C + cp * V);
        P px = *reinterpret_cast<const P*>(
            x + (((int64_t)n * H + h) * W + w_) * C + cp * V);
#pragma unroll
        for (int j = 0; j < V; ++j)
          acc[j] += to_f32(pg.v[j]) * to_f32(px.v[j]);
      }
    }
    float* slot = &red[(rl * ncp + cp0) * V];
    if (active) {
#pragma unroll
      for (int j = 0; j < V; ++j) slot[j] = acc[j];
    }
    __syncthreads();
    for (int st = nrl >> 1; st > 0; st >>= 1) {
      if (active && rl < st) {
        const float* other = &red[((rl + st) * ncp + cp0) * V];
#pragma unroll
        for (int j = 0; j < V; ++j) slot[j] += other[j];
      }
      __syncthreads();
    }
    if (active && rl == 0) {
#pragma unroll
      for (int j = 0; j < V; ++j)
        atomicAdd(&gw[((int64_t)(cp * V + j) * R + r) * S + s], slot[j]);
    }
    __syncthreads();
  }
}

__global__ void cast_f32_kernel(const float* __restrict__ a,
                                __hip_bfloat16* __restrict__ b, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    b[i] = from_f32<__hip_bfloat16>(a[i]);
}

}  // namespace

at::Tensor dwconv_fwd(at::Tensor x, at::Tensor w, int64_t sh, int64_t sw,
                      int64_t ph, int64_t pw) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int R = w.size(2), S = w.size(3);
  const int Ho = (H + 2 * ph - R) / sh + 1, Wo = (W + 2 * pw - S) / sw + 1;
  auto y = at::empty({N, C, Ho, Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "dwconv_fwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C % ", V, " != 0");
    int64_t total = (int64_t)N * Ho * Wo * (C / V);
    hipLaunchKernelGGL((dwconv_fwd_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)w.data_ptr(), (scalar_t*)y.data_ptr(),
                       N, H, W, C, Ho, Wo, R, S, sh, sw, ph, pw);
  });
  return y;
}

at::Tensor dwconv_dgrad(at::Tensor gy, at::Tensor w, int64_t H, int64_t W,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int N = gy.size(0), C = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  const int R = w.size(2), S = w.size(3);
  auto gx = at::empty({N, C, H, W},
                      gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  DISPATCH_FLOAT_AND_BF16(gy.scalar_type(), "dwconv_dgrad", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    int64_t total = (int64_t)N * H * W * (C / V);
    hipLaunchKernelGGL((dwconv_dgrad_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)gy.data_ptr(),
                       (const scalar_t*)w.data_ptr(), (scalar_t*)gx.data_ptr(),
                       N, H, W, C, Ho, Wo, R, S, sh, sw, ph, pw);
  });
  return gx;
}

at::Tensor dwconv_wgrad(at::Tensor gy, at::Tensor x, int64_t R, int64_t S,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int Ho = gy.size(2), Wo = gy.size(3);
  auto acc = at::empty({C, 1, (int64_t)R, (int64_t)S},
                       x.options().dtype(at::kFloat));
  hipMemsetAsync(acc.data_ptr(), 0, acc.numel() * 4, cur_stream());
  const int64_t pixels = (int64_t)N * Ho * Wo;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "dwconv_wgrad", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    const int cpacks = C / V;
    const int nrl = std::max(256 / std::min(cpacks, 256), 1);
    int64_t ppb = std::max<int64_t>(ceil_div(pixels, 256), nrl);
    int grid = (int)ceil_div(pixels, ppb);
    hipLaunchKernelGGL((dwconv_wgrad_kernel<scalar_t>),
                       dim3(grid, R * S), dim3(256), 0, cur_stream(),
                       (const scalar_t*)gy.data_ptr(),
                       (const scalar_t*)x.data_ptr(), acc.data_ptr<float>(),
                       N, H, W, C, Ho, Wo, R, S, sh, sw, ph, pw, ppb);
  });
  if (x.scalar_type() == at::kBFloat16) {
    auto gw = at::empty({C, 1, (int64_t)R, (int64_t)S},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
    int64_t n = acc.numel();
    hipLaunchKernelGGL(cast_f32_kernel, dim3(grid_1d(n, 256)), dim3(256), 0,
                       cur_stream(), acc.data_ptr<float>(),
                       (__hip_bfloat16*)gw.data_ptr(), n);
    return gw;
  }
  return acc.contiguous(at::MemoryFormat::ChannelsLast);
}
