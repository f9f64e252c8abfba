// BatchNorm kernels: NHWC per-channel statistics reduction,
// fused normalize(+residual)(+activation) apply, and backward reductions.
// Stats and parameters are fp32 regardless of the compute dtype (bf16-accuracy
// requirement, SURVEY.md §7 hard-part 4).
#include "common.h"

namespace {

// act ids: 0 none, 1 relu, 2 silu, 3 sigmoid
DEV_INLINE float act_apply(float z, int act) {
  switch (act) {
    case 1: return fmaxf(z, 0.f);
    case 2: return z / (1.f + __expf(-z));
    case 3: return 1.f / (1.f + __expf(-z));
    default: return z;
  }
}

// dAct/dz expressed with what each case can reach:
//   relu: from post-act y; silu/sigmoid: from pre-act z.
DEV_INLINE float act_grad(float y, float z, int act) {
  switch (act) {
    case 1: return y > 0.f ? 1.f : 0.f;
    case 2: {
      float s = 1.f / (1.f + __expf(-z));
      return s * (1.f + z * (1.f - s));
    }
    case 3: {
      float s = 1.f / (1.f + __expf(-z));
      return s * (1.f - s);
    }
    default: return 1.f;
  }
}

// ---- per-channel sum / sum-of-squares over N*H*W rows (NHWC: C fastest) ----
// Each block owns a slab of rows × all C channels; fp32 atomics merge blocks.
template <typename T>
__global__ void bn_sums_kernel(const T* __restrict__ x, float* __restrict__ s,
                               float* __restrict__ ss, int64_t rows, int C,
                               int rows_per_block) {
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    float acc = 0.f, acc2 = 0.f;
    for (int64_t r = row0; r < row1; ++r) {
      float v = to_f32(x[r * C + c]);
      acc += v;
      acc2 += v * v;
    }
    atomicAdd(&s[c], acc);
    atomicAdd(&ss[c], acc2);
  }
}

// ---- fused apply: y = act(x*scale[c] + shift[c] (+ res)) -------------------
template <typename T, int V, bool HAS_RES>
__global__ void bn_apply_kernel(const T* __restrict__ x,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                const T* __restrict__ res, T* __restrict__ y,
                                int64_t npacks, int cpacks, int act) {
  using P = Pack<T, V>;
  const P* xp = reinterpret_cast<const P*>(x);
  const P* rp = reinterpret_cast<const P*>(res);
  P* yp = reinterpret_cast<P*>(y);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < npacks;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c0 = (int)(i % cpacks) * V;
    P px = xp[i];
    P pr;
    if (HAS_RES) pr = rp[i];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      float z = to_f32(px.v[j]) * scale[c0 + j] + shift[c0 + j];
      if (HAS_RES) z += to_f32(pr.v[j]);
      px.v[j] = from_f32<T>(act_apply(z, act));
    }
    yp[i] = px;
  }
}

// ---- backward reduction: per-channel sum(g), sum(g * xhat) -----------------
// g = gy * act'(...); xhat = (x - mean) * rstd
template <typename T, bool HAS_RES>
__global__ void bn_bwd_reduce_kernel(
    const T* __restrict__ gy, const T* __restrict__ x, const T* __restrict__ y,
    const T* __restrict__ res, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ sum_g,
    float* __restrict__ sum_gxh, int64_t rows, int C, int rows_per_block,
    int act) {
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const float m = mean[c], r = rstd[c], w = gamma[c], b = beta[c];
    float acc = 0.f, acc2 = 0.f;
    for (int64_t i = row0; i < row1; ++i) {
      const int64_t off = i * C + c;
      float xh = (to_f32(x[off]) - m) * r;
      float g = to_f32(gy[off]);
      if (act != 0) {
        float z = xh * w + b;
        if (HAS_RES) z += to_f32(res[off]);
        g *= act_grad(to_f32(y[off]), z, act);
      }
      acc += g;
      acc2 += g * xh;
    }
    atomicAdd(&sum_g[c], acc);
    atomicAdd(&sum_gxh[c], acc2);
  }
}

// ---- backward apply: gx (+ gres) -------------------------------------------
// training: gx = gamma*rstd * (g - (sum_g + xhat*sum_gxh)/count)
// eval:     gx = gamma*rstd * g
template <typename T, int V, bool HAS_RES, bool TRAINING>
__global__ void bn_bwd_dx_kernel(
    const T* __restrict__ gy, const T* __restrict__ x, const T* __restrict__ y,
    const T* __restrict__ res, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, const float* __restrict__ sum_g,
    const float* __restrict__ sum_gxh, T* __restrict__ gx,
    T* __restrict__ gres, int64_t npacks, int cpacks, int act, float inv_cnt) {
  using P = Pack<T, V>;
  const P* gp = reinterpret_cast<const P*>(gy);
  const P* xp = reinterpret_cast<const P*>(x);
  const P* ypk = reinterpret_cast<const P*>(y);
  const P* rp = reinterpret_cast<const P*>(res);
  P* oxp = reinterpret_cast<P*>(gx);
  P* orp = reinterpret_cast<P*>(gres);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < npacks;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c0 = (int)(i % cpacks) * V;
    P pg = gp[i], px = xp[i], py, pr;
    if (act != 0) py = ypk[i];
    if (HAS_RES && act != 0) pr = rp[i];
    P ox, orr;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const int c = c0 + j;
      const float m = mean[c], r = rstd[c], w = gamma[c];
      float xh = (to_f32(px.v[j]) - m) * r;
      float g = to_f32(pg.v[j]);
      if (act != 0) {
        float z = xh * w + beta[c];
        if (HAS_RES) z += to_f32(pr.v[j]);
        g *= act_grad(to_f32(py.v[j]), z, act);
      }
      if (HAS_RES) orr.v[j] = from_f32<T>(g);
      float v;
      if (TRAINING)
        v = w * r * (g - (sum_g[c] + xh * sum_gxh[c]) * inv_cnt);
      else
        v = w * r * g;
      ox.v[j] = from_f32<T>(v);
    }
    oxp[i] = ox;
    if (HAS_RES) orp[i] = orr;
  }
}

int pick_rows_per_block(int64_t rows) {
  // target ~1024 reduction blocks
  return (int)std::max<int64_t>(ceil_div(rows, 1024), 8);
}

template <typename scalar_t, int V, bool HR, bool TR>
void launch_dx(const at::Tensor& gy, const at::Tensor& x, const at::Tensor& y,
               const scalar_t* resp, const at::Tensor& mean,
               const at::Tensor& rstd, const at::Tensor& gamma,
               const at::Tensor& beta, const at::Tensor& sum_g,
               const at::Tensor& sum_gxh, at::Tensor& gx, scalar_t* gresp,
               int64_t npacks, int cpacks, int act, float inv_cnt, int grid,
               hipStream_t stream) {
  hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, V, HR, TR>), dim3(grid),
                     dim3(256), 0, stream, (const scalar_t*)gy.data_ptr(),
                     (const scalar_t*)x.data_ptr(),
                     (const scalar_t*)y.data_ptr(), resp,
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     gamma.data_ptr<float>(), beta.data_ptr<float>(),
                     sum_g.data_ptr<float>(), sum_gxh.data_ptr<float>(),
                     (scalar_t*)gx.data_ptr(), gresp, npacks, cpacks, act,
                     inv_cnt);
}

}  // namespace

std::vector<at::Tensor> bn_sums(at::Tensor x) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int C = x.size(1);
  const int64_t rows = x.numel() / C;
  auto opts = x.options().dtype(at::kFloat);
  auto s = at::zeros({C}, opts);
  auto ss = at::zeros({C}, opts);
  int rpb = pick_rows_per_block(rows);
  int grid = (int)ceil_div(rows, rpb);
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_sums", [&] {
    hipLaunchKernelGGL((bn_sums_kernel<scalar_t>), dim3(grid), dim3(256), 0,
                       cur_stream(), (const scalar_t*)x.data_ptr(),
                       s.data_ptr<float>(), ss.data_ptr<float>(), rows, C, rpb);
  });
  return {s, ss};
}

at::Tensor bn_apply_act(at::Tensor x, at::Tensor scale, at::Tensor shift,
                        int64_t act, c10::optional<at::Tensor> res) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int C = x.size(1);
  auto y = at::empty_like(x);
  const bool has_res = res.has_value();
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_apply_act", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    int64_t npacks = x.numel() / V;
    int cpacks = C / V;
    int grid = grid_1d(npacks, 256);
    auto stream = cur_stream();
    if (has_res)
      hipLaunchKernelGGL((bn_apply_kernel<scalar_t, V, true>), dim3(grid),
                         dim3(256), 0, stream, (const scalar_t*)x.data_ptr(),
                         scale.data_ptr<float>(), shift.data_ptr<float>(),
                         (const scalar_t*)res->data_ptr(),
                         (scalar_t*)y.data_ptr(), npacks, cpacks, (int)act);
    else
      hipLaunchKernelGGL((bn_apply_kernel<scalar_t, V, false>), dim3(grid),
                         dim3(256), 0, stream, (const scalar_t*)x.data_ptr(),
                         scale.data_ptr<float>(), shift.data_ptr<float>(),
                         (const scalar_t*)nullptr, (scalar_t*)y.data_ptr(),
                         npacks, cpacks, (int)act);
  });
  return y;
}

std::vector<at::Tensor> bn_bwd(at::Tensor gy, at::Tensor x, at::Tensor y,
                               c10::optional<at::Tensor> res, at::Tensor mean,
                               at::Tensor rstd, at::Tensor gamma,
                               at::Tensor beta, int64_t act, bool training,
                               bool need_gres) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int C = x.size(1);
  const int64_t rows = x.numel() / C;
  auto fopts = x.options().dtype(at::kFloat);
  auto sum_g = at::zeros({C}, fopts);
  auto sum_gxh = at::zeros({C}, fopts);
  const bool has_res = res.has_value();
  auto gx = at::empty_like(x);
  auto gres = need_gres ? at::empty_like(x) : at::Tensor();

  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_bwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    auto stream = cur_stream();
    int rpb = pick_rows_per_block(rows);
    int rgrid = (int)ceil_div(rows, rpb);
    const scalar_t* resp =
        has_res ? (const scalar_t*)res->data_ptr() : nullptr;
    {  // grad-stat reduction runs in train AND eval (gw/gb need it)
      if (has_res)
        hipLaunchKernelGGL((bn_bwd_reduce_kernel<scalar_t, true>), dim3(rgrid),
                           dim3(256), 0, stream, (const scalar_t*)gy.data_ptr(),
                           (const scalar_t*)x.data_ptr(),
                           (const scalar_t*)y.data_ptr(), resp,
                           mean.data_ptr<float>(), rstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), beta.data_ptr<float>(),
                           sum_g.data_ptr<float>(), sum_gxh.data_ptr<float>(),
                           rows, C, rpb, (int)act);
      else
        hipLaunchKernelGGL((bn_bwd_reduce_kernel<scalar_t, false>), dim3(rgrid),
                           dim3(256), 0, stream, (const scalar_t*)gy.data_ptr(),
                           (const scalar_t*)x.data_ptr(),
                           (const scalar_t*)y.data_ptr(), resp,
                           mean.data_ptr<float>(), rstd.data_ptr<float>(),
                           gamma.data_ptr<float>(), beta.data_ptr<float>(),
                           sum_g.data_ptr<float>(), sum_gxh.data_ptr<float>(),
                           rows, C, rpb, (int)act);
    }
    int64_t npacks = x.numel() / V;
    int cpacks = C / V;
    int grid = grid_1d(npacks, 256);
    float inv_cnt = 1.f / (float)rows;
    scalar_t* gresp = need_gres ? (scalar_t*)gres.data_ptr() : nullptr;
    if (training) {
      if (has_res)
        launch_dx<scalar_t, V, true, true>(gy, x, y, resp, mean, rstd, gamma,
                                           beta, sum_g, sum_gxh, gx, gresp,
                                           npacks, cpacks, (int)act, inv_cnt,
                                           grid, stream);
      else
        launch_dx<scalar_t, V, false, true>(gy, x, y, resp, mean, rstd, gamma,
                                            beta, sum_g, sum_gxh, gx, gresp,
                                            npacks, cpacks, (int)act, inv_cnt,
                                            grid, stream);
    } else {
      if (has_res)
        launch_dx<scalar_t, V, true, false>(gy, x, y, resp, mean, rstd, gamma,
                                            beta, sum_g, sum_gxh, gx, gresp,
                                            npacks, cpacks, (int)act, inv_cnt,
                                            grid, stream);
      else
        launch_dx<scalar_t, V, false, false>(gy, x, y, resp, mean, rstd, gamma,
                                             beta, sum_g, sum_gxh, gx, gresp,
                                             npacks, cpacks, (int)act, inv_cnt,
                                             grid, stream);
    }
  });
  return {gx, sum_gxh, sum_g, gres};
}
