// Pooling kernels: NHWC max-pool with argmax indices,
// global average pool (one-workgroup-per-(n,c-slab) reduction), avg-pool 2x2.
#include "common.h"

namespace {

// ---- max pool fwd: thread per (n,ho,wo,c-pack of 8), vectorized -----------
// argmax saved as a WINDOW-LOCAL byte offset (kh*K + kw): the earlier int32
// flat index cost 4x the idx bytes on both sides (the backward gather reads
// idx for up to 4 windows per input element)
template <typename T>
__global__ void maxpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   unsigned char* __restrict__ idx, int N,
                                   int H, int W, int C, int Ho, int Wo, int K,
                                   int S, int P) {
  constexpr int V = 16 / sizeof(T);
  using Pk = Pack<T, V>;
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
    const int h0 = ho * S - P, w0 = wo * S - P;
    float best[V];
    __align__(8) unsigned char bidx[V];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      best[j] = -INFINITY;
      bidx[j] = 0;
    }
    for (int kh = 0; kh < K; ++kh) {
      const int h = h0 + kh;
      if (h < 0 || h >= H) continue;
      for (int kw = 0; kw < K; ++kw) {
        const int w = w0 + kw;
        if (w < 0 || w >= W) continue;
        Pk p = *reinterpret_cast<const Pk*>(
            x + (((int64_t)n * H + h) * W + w) * C + cp * V);
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float v = to_f32(p.v[j]);
          if (v > best[j]) {
            best[j] = v;
            bidx[j] = (unsigned char)(kh * K + kw);
          }
        }
      }
    }
    const int64_t out = (((int64_t)n * Ho + ho) * Wo + wo) * C + cp * V;
    Pk py;
#pragma unroll
    for (int j = 0; j < V; ++j) py.v[j] = from_f32<T>(best[j]);
    *reinterpret_cast<Pk*>(y + out) = py;
    if (V == 8)
      *reinterpret_cast<uint2*>(idx + out) =
          *reinterpret_cast<uint2*>(&bidx[0]);
    else
      *reinterpret_cast<unsigned int*>(idx + out) =
          *reinterpret_cast<unsigned int*>(&bidx[0]);
  }
}

// ---- max pool bwd: gather per input c-pack (no atomics) -------------------
template <typename T>
__global__ void maxpool_bwd_kernel(const T* __restrict__ gy,
                                   const unsigned char* __restrict__ idx,
                                   T* __restrict__ gx, int N, int H, int W,
                                   int C, int Ho, int Wo, int K, int S, int P) {
  constexpr int V = 16 / sizeof(T);
  using Pk = Pack<T, V>;
  const int cpacks = C / V;
  const int64_t total = (int64_t)N * H * W * cpacks;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int cp = i % cpacks;
    int64_t t = i / cpacks;
    const int w = t % W;
    t /= W;
    const int h = t % H;
    const int n = t / H;
    float acc[V] = {};
    const int ho_lo = max(0, (h + P - K + S) / S), ho_hi = min(Ho - 1, (h + P) / S);
    const int wo_lo = max(0, (w + P - K + S) / S), wo_hi = min(Wo - 1, (w + P) / S);
    for (int ho = ho_lo; ho <= ho_hi; ++ho)
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        // window-local offset this input element would need to have won
        const int want = (h - (ho * S - P)) * K + (w - (wo * S - P));
        const int64_t o = (((int64_t)n * Ho + ho) * Wo + wo) * C + cp * V;
        __align__(8) unsigned char bi[V];
        if (V == 8)
          *reinterpret_cast<uint2*>(&bi[0]) =
              *reinterpret_cast<const uint2*>(idx + o);
        else
          *reinterpret_cast<unsigned int*>(&bi[0]) =
              *reinterpret_cast<const unsigned int*>(idx + o);
        Pk pg = *reinterpret_cast<const Pk*>(gy + o);
#pragma unroll
        for (int j = 0; j < V; ++j)
          if (bi[j] == want) acc[j] += to_f32(pg.v[j]);
      }
    Pk out;
#pragma unroll
    for (int j = 0; j < V; ++j) out.v[j] = from_f32<T>(acc[j]);
    *reinterpret_cast<Pk*>(gx + (((int64_t)n * H + h) * W + w) * C + cp * V) =
        out;
  }
}

// ---- specialized stem maxpool bwd: K=3 S=2 P=1 ------------------------------
// Each input element receives from at most 2x2 output windows; all four
// idx/gy loads are issued up-front (the generic nested runtime loop
// serialized them: 301 us for ~1 GB of traffic). Row-based grid kills the
// 64-bit div/mod chain of the flat form.
template <typename T>
__global__ void maxpool_bwd_k3s2_kernel(const T* __restrict__ gy,
                                        const unsigned char* __restrict__ idx,
                                        T* __restrict__ gx, int N, int H,
                                        int W, int C, int Ho, int Wo) {
  constexpr int V = 16 / sizeof(T);
  using Pk = Pack<T, V>;
  const int cpacks = C / V;
  const int rowpacks = W * cpacks;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= rowpacks) return;
  const int row = blockIdx.y;  // n * H + h
  const int n = row / H, h = row - n * H;
  const int w = i / cpacks;
  const int cp = i - w * cpacks;
  const int ho0 = max(0, h >> 1), ho1 = min(Ho - 1, (h + 1) >> 1);
  const int wo0 = max(0, w >> 1), wo1 = min(Wo - 1, (w + 1) >> 1);
  const bool h2 = ho1 > ho0, w2 = wo1 > wo0;
  const int64_t base =
      (((int64_t)n * Ho + ho0) * Wo + wo0) * C + (int64_t)cp * V;
  const int64_t dW = (int64_t)C * Wo;  // one output row in elements
  Pk pg[4];
  __align__(8) unsigned char bi[4][V];
  auto ld = [&](int k, int64_t o) {
    pg[k] = *reinterpret_cast<const Pk*>(gy + o);
    if (V == 8)
      *reinterpret_cast<uint2*>(&bi[k][0]) =
          *reinterpret_cast<const uint2*>(idx + o);
    else
      *reinterpret_cast<unsigned int*>(&bi[k][0]) =
          *reinterpret_cast<const unsigned int*>(idx + o);
  };
  ld(0, base);
  if (w2) ld(1, base + C);
  if (h2) ld(2, base + dW);
  if (h2 && w2) ld(3, base + dW + C);
  float acc[V] = {};
  auto add = [&](int k, int ho, int wo) {
    const int want = (h - (ho * 2 - 1)) * 3 + (w - (wo * 2 - 1));
#pragma unroll
    for (int j = 0; j < V; ++j)
      if (bi[k][j] == want) acc[j] += to_f32(pg[k].v[j]);
  };
  add(0, ho0, wo0);
  if (w2) add(1, ho0, wo1);
  if (h2) add(2, ho1, wo0);
  if (h2 && w2) add(3, ho1, wo1);
  Pk out;
#pragma unroll
  for (int j = 0; j < V; ++j) out.v[j] = from_f32<T>(acc[j]);
  *reinterpret_cast<Pk*>(gx + ((int64_t)row * W + w) * C +
                         (int64_t)cp * V) = out;
}

// ---- global average pool: block per (n, 32 c-packs); 8 row lanes ----------
// (the original thread-per-channel form walked H*W scalar loads serially:
// 180 us on RegNetY's SE pools)
template <typename T>
__global__ void gap_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               int HW, int C, float inv_hw) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  __shared__ float red[256 * (16 / sizeof(T) > 8 ? 16 / sizeof(T) : 8)];
  const int cpacks = C / V;
  const int n = blockIdx.x;
  const int cpl = threadIdx.x & 31;
  const int rl = threadIdx.x >> 5;  // 8 row lanes
  const int cp = blockIdx.y * 32 + cpl;
  float acc[V] = {};
  if (cp < cpacks) {
    const P* xq = reinterpret_cast<const P*>(x) +
                  ((int64_t)n * HW + rl) * cpacks + cp;
    const int64_t rstep = (int64_t)8 * cpacks;
    int r = rl;
    for (; r + 24 < HW; r += 32) {
      P pk[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) pk[u] = xq[u * rstep];
      xq += 4 * rstep;
#pragma unroll
      for (int u = 0; u < 4; ++u)
#pragma unroll
        for (int j = 0; j < V; ++j) acc[j] += to_f32(pk[u].v[j]);
    }
    for (; r < HW; r += 8) {
      P pk = xq[0];
      xq += rstep;
#pragma unroll
      for (int j = 0; j < V; ++j) acc[j] += to_f32(pk.v[j]);
    }
  }
  float* slot = &red[threadIdx.x * V];
#pragma unroll
  for (int j = 0; j < V; ++j) slot[j] = acc[j];
  __syncthreads();
  if (rl != 0 || cp >= cpacks) return;
#pragma unroll
  for (int t = 1; t < 8; ++t) {
    const float* o = &red[(t * 32 + cpl) * V];
#pragma unroll
    for (int j = 0; j < V; ++j) acc[j] += o[j];
  }
  P out;
#pragma unroll
  for (int j = 0; j < V; ++j) out.v[j] = from_f32<T>(acc[j] * inv_hw);
  reinterpret_cast<P*>(y)[(int64_t)n * cpacks + cp] = out;
}

// scalar fallback for C not a multiple of the pack width
template <typename T>
__global__ void gap_fwd_scalar_kernel(const T* __restrict__ x,
                                      T* __restrict__ y, int HW, int C,
                                      float inv_hw) {
  const int n = blockIdx.x;
  for (int c = blockIdx.y * blockDim.x + threadIdx.x; c < C;
       c += gridDim.y * blockDim.x) {
    const T* base = x + (int64_t)n * HW * C + c;
    float acc = 0.f;
    for (int i = 0; i < HW; ++i) acc += to_f32(base[(int64_t)i * C]);
    y[(int64_t)n * C + c] = from_f32<T>(acc * inv_hw);
  }
}

template <typename T>
__global__ void gap_bwd_kernel(const T* __restrict__ gy, T* __restrict__ gx,
                               int64_t total, int HW, int C, float inv_hw) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c = i % C;
    const int64_t n = i / ((int64_t)HW * C);
    gx[i] = from_f32<T>(to_f32(gy[n * C + c]) * inv_hw);
  }
}

// ---- avg pool KxK stride S (no padding; densenet/botnet use 2x2 s2) -------
template <typename T>
__global__ void avgpool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   int N, int H, int W, int C, int Ho, int Wo,
                                   int K, int S, float inv_kk) {
  const int64_t total = (int64_t)N * Ho * Wo * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c = i % C;
    int64_t t = i / C;
    const int wo = t % Wo;
    t /= Wo;
    const int ho = t % Ho;
    const int n = t / Ho;
    float acc = 0.f;
    for (int kh = 0; kh < K; ++kh) {
      const int h = ho * S + kh;
      if (h >= H) continue;
      for (int kw = 0; kw < K; ++kw) {
        const int w = wo * S + kw;
        if (w >= W) continue;
        acc += to_f32(x[(((int64_t)n * H + h) * W + w) * C + c]);
      }
    }
    y[i] = from_f32<T>(acc * inv_kk);
  }
}

template <typename T>
__global__ void avgpool_bwd_kernel(const T* __restrict__ gy, T* __restrict__ gx,
                                   int N, int H, int W, int C, int Ho, int Wo,
                                   int K, int S, float inv_kk) {
  const int64_t total = (int64_t)N * H * W * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c = i % C;
    int64_t t = i / C;
    const int w = t % W;
    t /= W;
    const int h = t % H;
    const int n = t / H;
    float acc = 0.f;
    const int ho_lo = max(0, (h - K + S) / S), ho_hi = min(Ho - 1, h / S);
    const int wo_lo = max(0, (w - K + S) / S), wo_hi = min(Wo - 1, w / S);
    for (int ho = ho_lo; ho <= ho_hi; ++ho) {
      if (h - ho * S >= K) continue;
      for (int wo = wo_lo; wo <= wo_hi; ++wo) {
        if (w - wo * S >= K) continue;
        acc += to_f32(gy[(((int64_t)n * Ho + ho) * Wo + wo) * C + c]);
      }
    }
    gx[i] = from_f32<T>(acc * inv_kk);
  }
}

}  // namespace

std::vector<at::Tensor> maxpool_fwd(at::Tensor x, int64_t K, int64_t S,
                                    int64_t P) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int Ho = (H + 2 * P - K) / S + 1, Wo = (W + 2 * P - K) / S + 1;
  auto y = at::empty({N, C, Ho, Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  auto idx = at::empty({N, C, Ho, Wo},
                       x.options()
                           .dtype(at::kByte)
                           .memory_format(at::MemoryFormat::ChannelsLast));
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "maxpool_fwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C % ", V, " != 0");
    int64_t total = (int64_t)N * Ho * Wo * (C / V);
    hipLaunchKernelGGL((maxpool_fwd_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(), (scalar_t*)y.data_ptr(),
                       idx.data_ptr<unsigned char>(), N, H, W, C, Ho, Wo, K, S, P);
  });
  return {y, idx};
}

at::Tensor maxpool_bwd(at::Tensor gy, at::Tensor idx, int64_t H, int64_t W,
                       int64_t K, int64_t S, int64_t P) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int N = gy.size(0), C = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  auto gx = at::empty({N, C, H, W},
                      gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  DISPATCH_FLOAT_AND_BF16(gy.scalar_type(), "maxpool_bwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    if (K == 3 && S == 2 && P == 1 && C % V == 0) {
      const int rowpacks = (int)W * (C / V);
      hipLaunchKernelGGL((maxpool_bwd_k3s2_kernel<scalar_t>),
                         dim3((rowpacks + 255) / 256, N * H), dim3(256), 0,
                         cur_stream(), (const scalar_t*)gy.data_ptr(),
                         idx.data_ptr<unsigned char>(),
                         (scalar_t*)gx.data_ptr(), N, H, W, C, Ho, Wo);
      return;
    }
    int64_t total = (int64_t)N * H * W * (C / V);
    hipLaunchKernelGGL((maxpool_bwd_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)gy.data_ptr(), idx.data_ptr<unsigned char>(),
                       (scalar_t*)gx.data_ptr(), N, H, W, C, Ho, Wo, K, S, P);
  });
  return gx;
}

at::Tensor gap_fwd(at::Tensor x) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto y = at::empty({N, C, 1, 1}, x.options());
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "gap_fwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    if (C % V == 0) {
      const int cblocks = (int)ceil_div(C / V, 32);
      hipLaunchKernelGGL((gap_fwd_kernel<scalar_t>), dim3(N, cblocks),
                         dim3(256), 0, cur_stream(),
                         (const scalar_t*)x.data_ptr(),
                         (scalar_t*)y.data_ptr(), H * W, C, 1.f / (H * W));
    } else {
      const int cblocks = (int)std::min<int64_t>(ceil_div(C, 256), 8);
      hipLaunchKernelGGL((gap_fwd_scalar_kernel<scalar_t>), dim3(N, cblocks),
                         dim3(256), 0, cur_stream(),
                         (const scalar_t*)x.data_ptr(),
                         (scalar_t*)y.data_ptr(), H * W, C, 1.f / (H * W));
    }
  });
  return y;
}

at::Tensor gap_bwd(at::Tensor gy, int64_t H, int64_t W) {
  CHECK_GPU(gy);
  const int N = gy.size(0), C = gy.size(1);
  auto gx = at::empty({N, C, H, W},
                      gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)N * H * W * C;
  DISPATCH_FLOAT_AND_BF16(gy.scalar_type(), "gap_bwd", [&] {
    hipLaunchKernelGGL((gap_bwd_kernel<scalar_t>), dim3(grid_1d(total, 256)),
                       dim3(256), 0, cur_stream(),
                       (const scalar_t*)gy.data_ptr(), (scalar_t*)gx.data_ptr(),
                       total, H * W, C, 1.f / (H * W));
  });
  return gx;
}

at::Tensor avgpool_fwd(at::Tensor x, int64_t K, int64_t S) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int Ho = (H - K) / S + 1, Wo = (W - K) / S + 1;
  auto y = at::empty({N, C, Ho, Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)N * Ho * Wo * C;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "avgpool_fwd", [&] {
    hipLaunchKernelGGL((avgpool_fwd_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(), (scalar_t*)y.data_ptr(),
                       N, H, W, C, Ho, Wo, K, S, 1.f / (K * K));
  });
  return y;
}

at::Tensor avgpool_bwd(at::Tensor gy, int64_t K, int64_t S, int64_t H,
                       int64_t W) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int N = gy.size(0), C = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  auto gx = at::empty({N, C, H, W},
                      gy.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)N * H * W * C;
  DISPATCH_FLOAT_AND_BF16(gy.scalar_type(), "avgpool_bwd", [&] {
    hipLaunchKernelGGL((avgpool_bwd_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)gy.data_ptr(), (scalar_t*)gx.data_ptr(),
                       N, H, W, C, Ho, Wo, K, S, 1.f / (K * K));
  });
  return gx;
}
