#include "hip/hip_runtime.h"
// Elementwise kernels: ReLU fwd/bwd, fused residual
// add+ReLU. Memory-bound: 16 B/lane packed loads (Guideline 13), grid-stride.
#include "common_hip.h"

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
