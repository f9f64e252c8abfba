// Elementwise kernels: ReLU fwd/bwd, fused residual
// add+ReLU. Memory-bound: 16 B/lane packed loads (Guideline 13), grid-stride.
#include "common.h"

namespace {

template <typename T, int V>
__global__ void relu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                int64_t npacks) {
  using P = Pack<T, V>;
  const P* xp = reinterpret_cast<const P*>(x);
  P* yp = reinterpret_cast<P*>(y);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < npacks;
       i += (int64_t)gridDim.x * blockDim.x) {
    P p = xp[i];
#pragma unroll
    for (int j = 0; j < V; ++j) p.v[j] = from_f32<T>(fmaxf(to_f32(p.v[j]), 0.f));
    yp[i] = p;
  }
}

template <typename T, int V>
__global__ void add_relu_fwd_kernel(const T* __restrict__ a,
                                    const T* __restrict__ b, T* __restrict__ y,
                                    int64_t npacks) {
  using P = Pack<T, V>;
  const P* ap = reinterpret_cast<const P*>(a);
  const P* bp = reinterpret_cast<const P*>(b);
  P* yp = reinterpret_cast<P*>(y);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < npacks;
       i += (int64_t)gridDim.x * blockDim.x) {
    P pa = ap[i], pb = bp[i];
#pragma unroll
    for (int j = 0; j < V; ++j)
      pa.v[j] = from_f32<T>(fmaxf(to_f32(pa.v[j]) + to_f32(pb.v[j]), 0.f));
    yp[i] = pa;
  }
}

// gx = gy where y > 0 (y is the post-ReLU output)
template <typename T, int V>
__global__ void relu_bwd_kernel(const T* __restrict__ gy,
                                const T* __restrict__ y, T* __restrict__ gx,
                                int64_t npacks) {
  using P = Pack<T, V>;
  const P* gp = reinterpret_cast<const P*>(gy);
  const P* yp = reinterpret_cast<const P*>(y);
  P* op = reinterpret_cast<P*>(gx);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < npacks;
       i += (int64_t)gridDim.x * blockDim.x) {
    P pg = gp[i], py = yp[i];
#pragma unroll
    for (int j = 0; j < V; ++j)
      pg.v[j] = (to_f32(py.v[j]) > 0.f) ? pg.v[j] : from_f32<T>(0.f);
    op[i] = pg;
  }
}

template <typename T>
void launch_ew(const at::Tensor& out_like, int64_t numel, const void* a,
               const void* b, void* y, int which) {
  constexpr int V = 16 / sizeof(T);
  TORCH_CHECK(numel % V == 0, "numel must be divisible by ", V);
  int64_t npacks = numel / V;
  int block = 256;
  int grid = grid_1d(npacks, block);
  auto s = cur_stream();
  if (which == 0)
    hipLaunchKernelGGL((relu_fwd_kernel<T, V>), dim3(grid), dim3(block), 0, s,
                       (const T*)a, (T*)y, npacks);
  else if (which == 1)
    hipLaunchKernelGGL((add_relu_fwd_kernel<T, V>), dim3(grid), dim3(block), 0,
                       s, (const T*)a, (const T*)b, (T*)y, npacks);
  else
    hipLaunchKernelGGL((relu_bwd_kernel<T, V>), dim3(grid), dim3(block), 0, s,
                       (const T*)a, (const T*)b, (T*)y, npacks);
}

}  // namespace

at::Tensor relu_fwd(at::Tensor x) {
  CHECK_GPU(x);
  auto y = at::empty_like(x);
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "relu_fwd", [&] {
    launch_ew<scalar_t>(x, x.numel(), x.data_ptr(), nullptr, y.data_ptr(), 0);
  });
  return y;
}

at::Tensor add_relu_fwd(at::Tensor a, at::Tensor b) {
  CHECK_GPU(a);
  TORCH_CHECK(a.sizes() == b.sizes(), "shape mismatch");
  auto y = at::empty_like(a);
  DISPATCH_FLOAT_AND_BF16(a.scalar_type(), "add_relu_fwd", [&] {
    launch_ew<scalar_t>(a, a.numel(), a.data_ptr(), b.data_ptr(), y.data_ptr(),
                        1);
  });
  return y;
}

at::Tensor relu_bwd(at::Tensor gy, at::Tensor y) {
  CHECK_GPU(gy);
  auto gx = at::empty_like(gy);
  DISPATCH_FLOAT_AND_BF16(gy.scalar_type(), "relu_bwd", [&] {
    launch_ew<scalar_t>(gy, gy.numel(), gy.data_ptr(), y.data_ptr(),
                        gx.data_ptr(), 2);
  });
  return gx;
}

// ---------------------------------------------------------------------------
// SE channel scaling (K5): y[n, :, c] = x * s[n, c], and the fused backward
// gx = gy * s, gs[n, c] = sum_hw(gy * x) in ONE pass (ATen needed two
// broadcast muls plus a separate reduction). One block per image n: the
// per-(n, c) scales hoist once and the gs reduction never crosses blocks.
// ---------------------------------------------------------------------------
namespace {

DEV_INLINE int p2f_se(int v) { return 1 << (31 - __builtin_clz(v)); }

template <typename T>
__global__ __launch_bounds__(256) void se_scale_fwd_kernel(
    const T* __restrict__ x, const T* __restrict__ s, T* __restrict__ y,
    int HW, int C, int rows_per_chunk) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int n = blockIdx.y;
  const int row_lo = blockIdx.x * rows_per_chunk;
  const int row_hi = min(row_lo + rows_per_chunk, HW);
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2f_se(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  if (rl >= nrl) return;
  const P* xp = reinterpret_cast<const P*>(x) + (int64_t)n * HW * cpacks;
  P* yp = reinterpret_cast<P*>(y) + (int64_t)n * HW * cpacks;
  const P* sp = reinterpret_cast<const P*>(s) + (int64_t)n * cpacks;
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float sv[V];
    const P sq = sp[cp];
#pragma unroll
    for (int j = 0; j < V; ++j) sv[j] = to_f32(sq.v[j]);
    const int64_t rstep = (int64_t)nrl * cpacks;
    const P* xq = xp + (int64_t)(row_lo + rl) * cpacks + cp;
    P* yq = yp + (int64_t)(row_lo + rl) * cpacks + cp;
    for (int row = row_lo + rl; row < row_hi; row += nrl) {
      P px = xq[0];
#pragma unroll
      for (int j = 0; j < V; ++j)
        px.v[j] = from_f32<T>(to_f32(px.v[j]) * sv[j]);
      yq[0] = px;
      xq += rstep;
      yq += rstep;
    }
  }
}

// gs accumulates with fp32 atomics across row-chunk blocks (same
// determinism tradeoff as the split-m wgrad; ~1e-3 relative run-to-run)
template <typename T>
__global__ __launch_bounds__(256) void se_scale_bwd_kernel(
    const T* __restrict__ gy, const T* __restrict__ x,
    const T* __restrict__ s, T* __restrict__ gx, float* __restrict__ gs,
    int HW, int C, int rows_per_chunk) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  __shared__ float red[256 * (16 / sizeof(T) > 8 ? 16 / sizeof(T) : 8)];
  const int n = blockIdx.y;
  const int row_lo = blockIdx.x * rows_per_chunk;
  const int row_hi = min(row_lo + rows_per_chunk, HW);
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2f_se(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const P* gp = reinterpret_cast<const P*>(gy) + (int64_t)n * HW * cpacks;
  const P* xp = reinterpret_cast<const P*>(x) + (int64_t)n * HW * cpacks;
  P* oxp = reinterpret_cast<P*>(gx) + (int64_t)n * HW * cpacks;
  const P* sp = reinterpret_cast<const P*>(s) + (int64_t)n * cpacks;
  float* gsp = gs + (int64_t)n * C;
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float acc[V] = {};
    if (active) {
      float sv[V];
      const P sq = sp[cp];
#pragma unroll
      for (int j = 0; j < V; ++j) sv[j] = to_f32(sq.v[j]);
      const int64_t rstep = (int64_t)nrl * cpacks;
      const P* gq = gp + (int64_t)(row_lo + rl) * cpacks + cp;
      const P* xq = xp + (int64_t)(row_lo + rl) * cpacks + cp;
      P* oq = oxp + (int64_t)(row_lo + rl) * cpacks + cp;
      for (int row = row_lo + rl; row < row_hi; row += nrl) {
        P pg = gq[0], px = xq[0], ox;
#pragma unroll
        for (int j = 0; j < V; ++j) {
          const float g = to_f32(pg.v[j]);
          acc[j] += g * to_f32(px.v[j]);
          ox.v[j] = from_f32<T>(g * sv[j]);
        }
        oq[0] = ox;
        gq += rstep;
        xq += rstep;
        oq += rstep;
      }
    }
    float* slot = &red[(rl * ncp + cp0) * V];
    if (active)
#pragma unroll
      for (int j = 0; j < V; ++j) slot[j] = acc[j];
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
        atomicAdd(&gsp[cp * V + j], slot[j]);
    }
    __syncthreads();
  }
}

}  // namespace

at::Tensor se_scale_fwd(at::Tensor x, at::Tensor s) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1);
  const int HW = x.size(2) * x.size(3);
  auto y = at::empty_like(x);
  const int chunks = std::max<int>(1, (int)ceil_div(768, N));
  const int rpc = (int)ceil_div(HW, chunks);
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "se_scale_fwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "se_scale: C % ", V);
    hipLaunchKernelGGL((se_scale_fwd_kernel<scalar_t>),
                       dim3((int)ceil_div(HW, rpc), N), dim3(256), 0,
                       cur_stream(), (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)s.data_ptr(),
                       (scalar_t*)y.data_ptr(), HW, C, rpc);
  });
  return y;
}

std::vector<at::Tensor> se_scale_bwd(at::Tensor gy, at::Tensor x,
                                     at::Tensor s) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int N = x.size(0), C = x.size(1);
  const int HW = x.size(2) * x.size(3);
  auto gx = at::empty_like(x);
  auto gsf = at::empty({N, C}, x.options().dtype(at::kFloat));
  hipMemsetAsync(gsf.data_ptr(), 0, gsf.numel() * 4, cur_stream());
  const int chunks = std::max<int>(1, (int)ceil_div(768, N));
  const int rpc = (int)ceil_div(HW, chunks);
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "se_scale_bwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "se_scale: C % ", V);
    hipLaunchKernelGGL((se_scale_bwd_kernel<scalar_t>),
                       dim3((int)ceil_div(HW, rpc), N), dim3(256), 0,
                       cur_stream(), (const scalar_t*)gy.data_ptr(),
                       (const scalar_t*)x.data_ptr(),
                       (const scalar_t*)s.data_ptr(),
                       (scalar_t*)gx.data_ptr(), gsf.data_ptr<float>(),
                       HW, C, rpc);
  });
  return {gx, gsf.to(x.scalar_type()).reshape({N, C, 1, 1})};
}

// ---------------------------------------------------------------------------
// Dropout (K19): counter-based RNG (SplitMix64 on (seed, index)) so the mask
// is a pure function of (seed, position) — no state, replay-stable under
// hipGraph, and backward regenerates the mask instead of storing it.
// ---------------------------------------------------------------------------
namespace {

DEV_INLINE float u01_from(uint64_t z) {
  z += 0x9e3779b97f4a7c15ULL;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ULL;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebULL;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.0f / 16777216.0f);  // 24-bit mantissa
}

template <typename T, bool BWD>
__global__ void dropout_kernel(const T* __restrict__ x, T* __restrict__ y,
                               int64_t total, float p, float inv_keep,
                               uint64_t seed) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const bool keep = u01_from(seed ^ (uint64_t)i) >= p;
    y[i] = keep ? from_f32<T>(to_f32(x[i]) * inv_keep) : from_f32<T>(0.f);
  }
}

}  // namespace

at::Tensor dropout_fwd(at::Tensor x, double p, int64_t seed) {
  CHECK_GPU(x);
  auto y = at::empty_like(x);
  const int64_t total = x.numel();
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "dropout_fwd", [&] {
    hipLaunchKernelGGL((dropout_kernel<scalar_t, false>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(),
                       (scalar_t*)y.data_ptr(), total, (float)p,
                       1.f / (1.f - (float)p), (uint64_t)seed);
  });
  return y;
}
