// BatchNorm kernels: NHWC per-channel statistics reduction,
// fused normalize(+residual)(+activation) apply, and backward reductions.
// Stats and parameters are fp32 regardless of the compute dtype (bf16-accuracy
// requirement, SURVEY.md §7 hard-part 4).
//
// Structure (all deterministic — no fp32 atomics):
//   fwd:  bn_sums_kernel (per-block partials, 8-deep MLP unroll)
//         -> bn_finalize_kernel (partial reduce + mean/rstd/scale/shift +
//            running-stat update, one launch)
//         -> bn_apply_kernel (y = act(x*scale + shift + res))
//   bwd:  bn_bwd_reduce_kernel (RAW per-channel sums sum(g), sum(g*x);
//         no per-channel parameter gathers on the hot path)
//         -> bn_bwd_finalize_kernel (tiny: -> gw, gb and the dx coefficients
//            P1, P2, P3 with gx = P1*g + P3*x + P2)
//         -> bn_bwd_dx_kernel (3 coefficient gathers per pack)
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

// act' from the RECOMPUTED pre-activation z = x*scale + shift (+res): the
// forward's y stream is never read in backward (F3a of
// docs/DESIGN_bn_conv_fusion.md — one full tensor stream dropped from both
// backward passes). relu's mask z > 0 is bit-identical to forward's y > 0
// (same fp32 math on the same bf16 inputs).
DEV_INLINE float act_grad(float z, int act) {
  switch (act) {
    case 1: return z > 0.f ? 1.f : 0.f;
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

DEV_INLINE int p2_floor(int v) { return 1 << (31 - __builtin_clz(v)); }

// ---- fwd stage 1: per-channel sum / sum-of-squares partials ----------------
// thread -> (pack column cp, row lane rl); 8-deep unrolled row loop for MLP;
// LDS tree over row lanes; block writes its partial row [2C].
template <typename T>
__global__ void bn_sums_kernel(const T* __restrict__ x,
                               float* __restrict__ part, int64_t rows, int C,
                               int64_t rows_per_block) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  __shared__ float red[256 * 2 * (16 / sizeof(T) > 8 ? 16 / sizeof(T) : 8)];
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2_floor(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  const P* xp = reinterpret_cast<const P*>(x);
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float acc[V] = {}, acc2[V] = {};
    if (active) {
      int64_t r = row0 + rl;
      const int64_t rstep = (int64_t)nrl * cpacks;
      const P* xq = xp + r * cpacks + cp;
      for (; r + 7 * (int64_t)nrl < row1; r += 8 * (int64_t)nrl) {
        P pk[8];
#pragma unroll
        for (int u = 0; u < 8; ++u) pk[u] = xq[u * rstep];
        xq += 8 * rstep;
#pragma unroll
        for (int u = 0; u < 8; ++u)
#pragma unroll
          for (int j = 0; j < V; ++j) {
            float v = to_f32(pk[u].v[j]);
            acc[j] += v;
            acc2[j] += v * v;
          }
      }
      for (; r < row1; r += nrl) {
        P p = xq[0];
        xq += rstep;
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float v = to_f32(p.v[j]);
          acc[j] += v;
          acc2[j] += v * v;
        }
      }
    }
    float* slot = &red[(rl * ncp + cp0) * 2 * V];
    if (active) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        slot[j] = acc[j];
        slot[V + j] = acc2[j];
      }
    }
    __syncthreads();
    for (int st = nrl >> 1; st > 0; st >>= 1) {
      if (active && rl < st) {
        const float* other = &red[((rl + st) * ncp + cp0) * 2 * V];
#pragma unroll
        for (int j = 0; j < 2 * V; ++j) slot[j] += other[j];
      }
      __syncthreads();
    }
    if (active && rl == 0) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        part[(int64_t)blockIdx.x * 2 * C + cp * V + j] = slot[j];
        part[(int64_t)blockIdx.x * 2 * C + C + cp * V + j] = slot[V + j];
      }
    }
    __syncthreads();
  }
}

// ---- generic stage 2: column-reduce partials [nblocks, width] -> [width] ---
// 256 threads = 64 channels x 4 row-lanes.
__global__ void reduce_partials_kernel(const float* __restrict__ part,
                                       float* __restrict__ out, int width,
                                       int nblocks) {
  __shared__ float red[256];
  const int c = blockIdx.x * 64 + (threadIdx.x & 63);
  const int rlane = threadIdx.x >> 6;
  float acc = 0.f;
  if (c < width) {
    int b = rlane;
    for (; b + 12 < nblocks; b += 16) {
      acc += part[(int64_t)b * width + c] +
             part[(int64_t)(b + 4) * width + c] +
             part[(int64_t)(b + 8) * width + c] +
             part[(int64_t)(b + 12) * width + c];
    }
    for (; b < nblocks; b += 4) acc += part[(int64_t)b * width + c];
  }
  red[threadIdx.x] = acc;
  __syncthreads();
  if (rlane == 0 && c < width)
    out[c] = red[threadIdx.x] + red[threadIdx.x + 64] +
             red[threadIdx.x + 128] + red[threadIdx.x + 192];
}

// ---- fwd finalize: partial reduce + mean/rstd/scale/shift + running update -
__global__ void bn_finalize_kernel(const float* __restrict__ part, int nblocks,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   float* __restrict__ mean,
                                   float* __restrict__ rstd,
                                   float* __restrict__ scale,
                                   float* __restrict__ shift, int C,
                                   float inv_cnt, float unbias, float momentum,
                                   float eps, int update_running) {
  // 16 channels x 16 row-lanes, 4-tap unroll: the previous 64x4 mapping left
  // one load-latency chain of nblocks/4 iterations per thread (~36us per
  // launch at nblocks=512 — stage 2 cost MORE than the bandwidth-bound
  // stage 1).
  __shared__ float red[512];
  const int ci = threadIdx.x & 15;
  const int rlane = threadIdx.x >> 4;
  const int c = blockIdx.x * 16 + ci;
  float sum = 0.f, sumsq = 0.f;
  if (c < C) {
    int b = rlane;
    for (; b + 48 < nblocks; b += 64) {
      sum += (part[(int64_t)b * 2 * C + c] +
              part[(int64_t)(b + 16) * 2 * C + c]) +
             (part[(int64_t)(b + 32) * 2 * C + c] +
              part[(int64_t)(b + 48) * 2 * C + c]);
      sumsq += (part[(int64_t)b * 2 * C + C + c] +
                part[(int64_t)(b + 16) * 2 * C + C + c]) +
               (part[(int64_t)(b + 32) * 2 * C + C + c] +
                part[(int64_t)(b + 48) * 2 * C + C + c]);
    }
    for (; b < nblocks; b += 16) {
      sum += part[(int64_t)b * 2 * C + c];
      sumsq += part[(int64_t)b * 2 * C + C + c];
    }
  }
  red[threadIdx.x] = sum;
  red[256 + threadIdx.x] = sumsq;
  __syncthreads();
  if (rlane != 0 || c >= C) return;
#pragma unroll
  for (int t = 1; t < 16; ++t) {
    sum += red[t * 16 + ci];
    sumsq += red[256 + t * 16 + ci];
  }
  const float m = sum * inv_cnt;
  float v = sumsq * inv_cnt - m * m;
  v = fmaxf(v, 0.f);
  const float r = rsqrtf(v + eps);
  mean[c] = m;
  rstd[c] = r;
  const float sc = gamma[c] * r;
  scale[c] = sc;
  shift[c] = beta[c] - m * sc;
  if (update_running) {
    running_mean[c] = running_mean[c] * (1.f - momentum) + m * momentum;
    running_var[c] = running_var[c] * (1.f - momentum) + v * unbias * momentum;
  }
}

// eval-mode prep: scale/shift from running stats
__global__ void bn_eval_prep_kernel(const float* __restrict__ running_mean,
                                    const float* __restrict__ running_var,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ beta,
                                    float* __restrict__ mean,
                                    float* __restrict__ rstd,
                                    float* __restrict__ scale,
                                    float* __restrict__ shift, int C,
                                    float eps) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  const float m = running_mean[c];
  const float r = rsqrtf(running_var[c] + eps);
  mean[c] = m;
  rstd[c] = r;
  const float sc = gamma[c] * r;
  scale[c] = sc;
  shift[c] = beta[c] - m * sc;
}

// ---- fused apply: y = act(x*scale[c] + shift[c] (+ res)) -------------------
// Row-walk structure (same as bn_bwd_reduce/dx): each thread owns a fixed
// channel pack so scale/shift load ONCE per column, and rows advance with
// pointer increments (no per-iteration modulo or table gathers — the
// previous channel-incrementing grid-stride form re-gathered the float4
// tables every pack).
// EMITMASK (HAS_RES relu only): one act'(z) bit per element, a byte per
// V-pack at [row * cpacks + cp] — the backward kernels then skip the res
// stream AND the z recompute (docs/ARCHITECTURE.md round-2 status).
template <typename T, bool HAS_RES, bool EMITMASK = false>
__global__ __launch_bounds__(256, 2) void bn_apply_kernel(
    const T* __restrict__ x, const float* __restrict__ scale,
    const float* __restrict__ shift, const T* __restrict__ res,
    T* __restrict__ y, int64_t rows, int C, int64_t rows_per_block, int act,
    unsigned char* __restrict__ mask = nullptr) {
  static_assert(!(EMITMASK && !HAS_RES), "mask emission is for residual BNs");
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2_floor(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  const P* xp = reinterpret_cast<const P*>(x);
  const P* rp = reinterpret_cast<const P*>(res);
  P* yp = reinterpret_cast<P*>(y);
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float sc[V], sh[V];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      sc[j] = scale[cp * V + j];
      sh[j] = shift[cp * V + j];
    }
    if (!active) continue;
    int64_t row = row0 + rl;
    const int64_t rstep = (int64_t)nrl * cpacks;
    const P* xq = xp + row * cpacks + cp;
    const P* rq = rp + row * cpacks + cp;
    P* yq = yp + row * cpacks + cp;
    unsigned char* mq = EMITMASK ? mask + row * cpacks + cp : nullptr;
    for (; row + 3 * (int64_t)nrl < row1; row += 4 * (int64_t)nrl) {
      P px4[4], pr4[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        px4[u] = xq[u * rstep];
        if (HAS_RES) pr4[u] = rq[u * rstep];
      }
      xq += 4 * rstep;
      rq += 4 * rstep;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        unsigned int mb = 0;
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float z = to_f32(px4[u].v[j]) * sc[j] + sh[j];
          if (HAS_RES) z += to_f32(pr4[u].v[j]);
          if (EMITMASK && z > 0.f) mb |= 1u << j;
          px4[u].v[j] = from_f32<T>(act_apply(z, act));
        }
        yq[u * rstep] = px4[u];
        if (EMITMASK) mq[u * rstep] = (unsigned char)mb;
      }
      yq += 4 * rstep;
      if (EMITMASK) mq += 4 * rstep;
    }
    for (; row < row1; row += nrl) {
      P px = xq[0], pr;
      if (HAS_RES) pr = rq[0];
      xq += rstep;
      rq += rstep;
      unsigned int mb = 0;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float z = to_f32(px.v[j]) * sc[j] + sh[j];
        if (HAS_RES) z += to_f32(pr.v[j]);
        if (EMITMASK && z > 0.f) mb |= 1u << j;
        px.v[j] = from_f32<T>(act_apply(z, act));
      }
      yq[0] = px;
      if (EMITMASK) {
        mq[0] = (unsigned char)mb;
        mq += rstep;
      }
      yq += rstep;
    }
  }
}

// ---- bwd stage 1: RAW per-channel sums sum(g), sum(g*x) --------------------
// g = gy * act'; act' recomputes z = scale*x+shift(+res) for every act, so
// the y stream is never read. No mean/rstd/gamma gathers on the hot path.
template <typename T, bool HAS_RES, int ACT, bool MASKED = false>
__global__ __launch_bounds__(256, 2) void bn_bwd_reduce_kernel(
    const T* __restrict__ gy, const T* __restrict__ x,
    const T* __restrict__ res, const float* __restrict__ scale,
    const float* __restrict__ shift, float* __restrict__ part, int64_t rows,
    int C, int64_t rows_per_block,
    const unsigned char* __restrict__ mask = nullptr) {
  static_assert(!(MASKED && !(HAS_RES && ACT == 1)),
                "mask path covers residual relu BNs only");
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  __shared__ float red[256 * 2 * (16 / sizeof(T) > 8 ? 16 / sizeof(T) : 8)];
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2_floor(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  const P* gp = reinterpret_cast<const P*>(gy);
  const P* xp = reinterpret_cast<const P*>(x);
  const P* rp = reinterpret_cast<const P*>(res);
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float sc[V], sh[V];
    if (ACT != 0) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        sc[j] = scale[cp * V + j];
        sh[j] = shift[cp * V + j];
      }
    }
    float accg[V] = {}, accgx[V] = {};
    if (active) {
      int64_t row = row0 + rl;
      // pointer-increment addressing: a per-access (row*cpacks + cp) 64-bit
      // multiply throttled this loop to ~1 TB/s (same diagnosis as
      // bn_apply_kernel above)
      const int64_t rstep = (int64_t)nrl * cpacks;
      const P* gq = gp + row * cpacks + cp;
      const P* xq = xp + row * cpacks + cp;
      const P* rq = rp + row * cpacks + cp;
      const unsigned char* mq = MASKED ? mask + row * cpacks + cp : nullptr;
      for (; row + 3 * (int64_t)nrl < row1; row += 4 * (int64_t)nrl) {
        P pg4[4], px4[4], pr4[4];
        unsigned char mb4[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          pg4[u] = gq[u * rstep];
          px4[u] = xq[u * rstep];
          if (MASKED)
            mb4[u] = mq[u * rstep];
          else if (HAS_RES && ACT != 0)
            pr4[u] = rq[u * rstep];
        }
        gq += 4 * rstep;
        xq += 4 * rstep;
        rq += 4 * rstep;
        if (MASKED) mq += 4 * rstep;
#pragma unroll
        for (int u = 0; u < 4; ++u)
#pragma unroll
          for (int j = 0; j < V; ++j) {
            float xv = to_f32(px4[u].v[j]);
            float g = to_f32(pg4[u].v[j]);
            if (MASKED) {
              g = (mb4[u] >> j) & 1 ? g : 0.f;
            } else if (ACT != 0) {
              float z = xv * sc[j] + sh[j];
              if (HAS_RES) z += to_f32(pr4[u].v[j]);
              g *= act_grad(z, ACT);
            }
            accg[j] += g;
            accgx[j] += g * xv;
          }
      }
      for (; row < row1; row += nrl) {
        P pg = gq[0], px = xq[0], prr;
        unsigned char mb = 0;
        if (MASKED)
          mb = mq[0];
        else if (HAS_RES && ACT != 0)
          prr = rq[0];
        gq += rstep;
        xq += rstep;
        rq += rstep;
        if (MASKED) mq += rstep;
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float xv = to_f32(px.v[j]);
          float g = to_f32(pg.v[j]);
          if (MASKED) {
            g = (mb >> j) & 1 ? g : 0.f;
          } else if (ACT != 0) {
            float z = xv * sc[j] + sh[j];
            if (HAS_RES) z += to_f32(prr.v[j]);
            g *= act_grad(z, ACT);
          }
          accg[j] += g;
          accgx[j] += g * xv;
        }
      }
    }
    float* slot = &red[(rl * ncp + cp0) * 2 * V];
    if (active) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        slot[j] = accg[j];
        slot[V + j] = accgx[j];
      }
    }
    __syncthreads();
    for (int st = nrl >> 1; st > 0; st >>= 1) {
      if (active && rl < st) {
        const float* other = &red[((rl + st) * ncp + cp0) * 2 * V];
#pragma unroll
        for (int j = 0; j < 2 * V; ++j) slot[j] += other[j];
      }
      __syncthreads();
    }
    if (active && rl == 0) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        part[(int64_t)blockIdx.x * 2 * C + cp * V + j] = slot[j];
        part[(int64_t)blockIdx.x * 2 * C + C + cp * V + j] = slot[V + j];
      }
    }
    __syncthreads();
  }
}

// ---- bwd finalize: partials -> gw, gb and the dx coefficients --------------
// sum_gxh = rstd*(sum(g*x) - mean*sum(g)); gw = sum_gxh, gb = sum(g)
// training: gx = w*r*(g - (sum_g + xhat*sum_gxh)/cnt) = P1*g + P3*x + P2 with
//   P1 = w*r; P3 = -w*r^2*sum_gxh/cnt; P2 = -P1*sum_g/cnt - P3*mean
// eval: P1 = w*r; P2 = P3 = 0.
__global__ void bn_bwd_finalize_kernel(
    const float* __restrict__ part, int nblocks,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, float* __restrict__ gw,
    float* __restrict__ gb, float* __restrict__ P1, float* __restrict__ P2,
    float* __restrict__ P3, int C, float inv_cnt, int training) {
  __shared__ float red[512];
  const int ci = threadIdx.x & 15;
  const int rlane = threadIdx.x >> 4;
  const int c = blockIdx.x * 16 + ci;
  float sg = 0.f, sgx = 0.f;
  if (c < C) {
    int b = rlane;
    for (; b + 48 < nblocks; b += 64) {
      sg += (part[(int64_t)b * 2 * C + c] +
             part[(int64_t)(b + 16) * 2 * C + c]) +
            (part[(int64_t)(b + 32) * 2 * C + c] +
             part[(int64_t)(b + 48) * 2 * C + c]);
      sgx += (part[(int64_t)b * 2 * C + C + c] +
              part[(int64_t)(b + 16) * 2 * C + C + c]) +
             (part[(int64_t)(b + 32) * 2 * C + C + c] +
              part[(int64_t)(b + 48) * 2 * C + C + c]);
    }
    for (; b < nblocks; b += 16) {
      sg += part[(int64_t)b * 2 * C + c];
      sgx += part[(int64_t)b * 2 * C + C + c];
    }
  }
  red[threadIdx.x] = sg;
  red[256 + threadIdx.x] = sgx;
  __syncthreads();
  if (rlane != 0 || c >= C) return;
#pragma unroll
  for (int t = 1; t < 16; ++t) {
    sg += red[t * 16 + ci];
    sgx += red[256 + t * 16 + ci];
  }
  const float m = mean[c], r = rstd[c], w = gamma[c];
  const float sum_gxh = r * (sgx - m * sg);
  gw[c] = sum_gxh;
  gb[c] = sg;
  const float p1 = w * r;
  P1[c] = p1;
  if (training) {
    const float p3 = -w * r * r * sum_gxh * inv_cnt;
    P3[c] = p3;
    P2[c] = -p1 * sg * inv_cnt - p3 * m;
  } else {
    P3[c] = 0.f;
    P2[c] = 0.f;
  }
}

// ---- bwd dx: gx = P1*g + P3*x + P2 (+ gres = g) ----------------------------
// act mask recomputed from z = x*scale+shift(+res); no y stream (F3a).
// Row-walk structure (same as bn_bwd_reduce): each thread owns a fixed
// channel pack, so the 3-5 per-channel coefficient tables are loaded ONCE
// per column instead of per grid-stride iteration — the flat grid-stride
// form re-gathered them every pack and cost ~0.5 ms/step on ResNet-50.
template <typename T, bool HAS_RES, int ACT, bool MASKED = false>
__global__ __launch_bounds__(256, 2) void bn_bwd_dx_kernel(
    const T* __restrict__ gy, const T* __restrict__ x,
    const T* __restrict__ res, const float* __restrict__ scale,
    const float* __restrict__ shift, const float* __restrict__ P1c,
    const float* __restrict__ P2c, const float* __restrict__ P3c,
    T* __restrict__ gx, T* __restrict__ gres, int64_t rows, int C,
    int64_t rows_per_block,
    const unsigned char* __restrict__ mask = nullptr) {
  static_assert(!(MASKED && !(HAS_RES && ACT == 1)),
                "mask path covers residual relu BNs only");
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2_floor(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  const P* gp = reinterpret_cast<const P*>(gy);
  const P* xp = reinterpret_cast<const P*>(x);
  const P* rp = reinterpret_cast<const P*>(res);
  P* oxp = reinterpret_cast<P*>(gx);
  P* orp = reinterpret_cast<P*>(gres);
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float sc[V], sh[V], a1[V], a2[V], a3[V];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const int cc = cp * V + j;
      a1[j] = P1c[cc];
      a2[j] = P2c[cc];
      a3[j] = P3c[cc];
      if (ACT != 0) {
        sc[j] = scale[cc];
        sh[j] = shift[cc];
      }
    }
    if (!active) continue;
    int64_t row = row0 + rl;
    const int64_t rstep = (int64_t)nrl * cpacks;
    const P* gq = gp + row * cpacks + cp;
    const P* xq = xp + row * cpacks + cp;
    const P* rq = rp + row * cpacks + cp;
    const unsigned char* mq = MASKED ? mask + row * cpacks + cp : nullptr;
    P* oq = oxp + row * cpacks + cp;
    P* orq = orp + row * cpacks + cp;
    for (; row + 3 * (int64_t)nrl < row1; row += 4 * (int64_t)nrl) {
      P pg4[4], px4[4], pr4[4];
      unsigned char mb4[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        pg4[u] = gq[u * rstep];
        px4[u] = xq[u * rstep];
        if (MASKED)
          mb4[u] = mq[u * rstep];
        else if (HAS_RES && ACT != 0)
          pr4[u] = rq[u * rstep];
      }
      gq += 4 * rstep;
      xq += 4 * rstep;
      rq += 4 * rstep;
      if (MASKED) mq += 4 * rstep;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        P ox, orr;
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float xv = to_f32(px4[u].v[j]);
          float g = to_f32(pg4[u].v[j]);
          if (MASKED) {
            g = (mb4[u] >> j) & 1 ? g : 0.f;
          } else if (ACT != 0) {
            float z = xv * sc[j] + sh[j];
            if (HAS_RES) z += to_f32(pr4[u].v[j]);
            g *= act_grad(z, ACT);
          }
          if (HAS_RES) orr.v[j] = from_f32<T>(g);
          ox.v[j] = from_f32<T>(a1[j] * g + a3[j] * xv + a2[j]);
        }
        oq[u * rstep] = ox;
        if (HAS_RES) orq[u * rstep] = orr;
      }
      oq += 4 * rstep;
      orq += 4 * rstep;
    }
    for (; row < row1; row += nrl) {
      P pg = gq[0], px = xq[0], pr;
      unsigned char mb = 0;
      if (MASKED)
        mb = mq[0];
      else if (HAS_RES && ACT != 0)
        pr = rq[0];
      gq += rstep;
      xq += rstep;
      rq += rstep;
      if (MASKED) mq += rstep;
      P ox, orr;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xv = to_f32(px.v[j]);
        float g = to_f32(pg.v[j]);
        if (MASKED) {
          g = (mb >> j) & 1 ? g : 0.f;
        } else if (ACT != 0) {
          float z = xv * sc[j] + sh[j];
          if (HAS_RES) z += to_f32(pr.v[j]);
          g *= act_grad(z, ACT);
        }
        if (HAS_RES) orr.v[j] = from_f32<T>(g);
        ox.v[j] = from_f32<T>(a1[j] * g + a3[j] * xv + a2[j]);
      }
      oq[0] = ox;
      if (HAS_RES) orq[0] = orr;
      oq += rstep;
      orq += rstep;
    }
  }
}

int64_t pick_rows_per_block(int64_t rows, int rows_per_iter) {
  // stage-1 block budget: BW-saturating with the MLP unroll while keeping the
  // stage-2 partial reduction short (env override for tuning probes)
  const char* s = getenv("DISTRIBUUUU_BN_S1GRID");  // re-read: probe sweeps
  // measured (tools/probes/bnbwd_ab.py): 512 beats 1024/2048/4096 across the
  // full ResNet-50 shape mix once the reduce kernel stopped spilling
  const int cap = s ? atoi(s) : 512;
  int64_t rpb = std::max<int64_t>(ceil_div(rows, cap), rows_per_iter);
  return ceil_div(rpb, rows_per_iter) * rows_per_iter;
}

// act is runtime but templated into the kernel: a unified body kept the
// silu/sigmoid scale/shift register arrays live on every path and spilled
// 90-186 VGPRs (scratch traffic capped the reduce at ~1.5 TB/s).
template <typename scalar_t>
void launch_bwd_reduce(const scalar_t* gy, const scalar_t* x,
                       const scalar_t* resp,
                       const float* scale, const float* shift, float* part,
                       int64_t rows, int C, int64_t rpb, int act, int rgrid,
                       hipStream_t stream,
                       const unsigned char* maskp = nullptr) {
#define BR_CASE(HR, A)                                                      \
  hipLaunchKernelGGL((bn_bwd_reduce_kernel<scalar_t, HR, A>), dim3(rgrid),  \
                     dim3(256), 0, stream, gy, x, resp, scale, shift,       \
                     part, rows, C, rpb)
  if (maskp != nullptr) {
    TORCH_CHECK(act == 1 && resp != nullptr, "mask path is residual relu");
    hipLaunchKernelGGL((bn_bwd_reduce_kernel<scalar_t, true, 1, true>),
                       dim3(rgrid), dim3(256), 0, stream, gy, x, resp, scale,
                       shift, part, rows, C, rpb, maskp);
  } else if (resp != nullptr) {
    switch (act) {
      case 0: BR_CASE(true, 0); break;
      case 1: BR_CASE(true, 1); break;
      case 2: BR_CASE(true, 2); break;
      default: BR_CASE(true, 3); break;
    }
  } else {
    switch (act) {
      case 0: BR_CASE(false, 0); break;
      case 1: BR_CASE(false, 1); break;
      case 2: BR_CASE(false, 2); break;
      default: BR_CASE(false, 3); break;
    }
  }
#undef BR_CASE
}


template <typename scalar_t>
void launch_dx(const at::Tensor& gy, const at::Tensor& x,
               const scalar_t* resp, const at::Tensor& scale,
               const at::Tensor& shift, const at::Tensor& P1,
               const at::Tensor& P2, const at::Tensor& P3, at::Tensor& gx,
               scalar_t* gresp, int64_t rows, int C, int64_t rpb, int act,
               int rgrid, hipStream_t stream,
               const unsigned char* maskp = nullptr) {
#define DX_CASE(HR, A)                                                       \
  hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, HR, A>), dim3(rgrid),       \
                     dim3(256), 0, stream, (const scalar_t*)gy.data_ptr(),   \
                     (const scalar_t*)x.data_ptr(), resp,                    \
                     scale.data_ptr<float>(), shift.data_ptr<float>(),       \
                     P1.data_ptr<float>(), P2.data_ptr<float>(),             \
                     P3.data_ptr<float>(), (scalar_t*)gx.data_ptr(), gresp,  \
                     rows, C, rpb)
  if (maskp != nullptr) {
    TORCH_CHECK(act == 1 && resp != nullptr, "mask path is residual relu");
    hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, true, 1, true>),
                       dim3(rgrid), dim3(256), 0, stream,
                       (const scalar_t*)gy.data_ptr(),
                       (const scalar_t*)x.data_ptr(), resp,
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       P1.data_ptr<float>(), P2.data_ptr<float>(),
                       P3.data_ptr<float>(), (scalar_t*)gx.data_ptr(), gresp,
                       rows, C, rpb, maskp);
  } else if (resp != nullptr) {
    switch (act) {
      case 0: DX_CASE(true, 0); break;
      case 1: DX_CASE(true, 1); break;
      case 2: DX_CASE(true, 2); break;
      default: DX_CASE(true, 3); break;
    }
  } else {
    switch (act) {
      case 0: DX_CASE(false, 0); break;
      case 1: DX_CASE(false, 1); break;
      case 2: DX_CASE(false, 2); break;
      default: DX_CASE(false, 3); break;
    }
  }
#undef DX_CASE
}

// Stage-A collapse for conv-epilogue partial matrices (F1) with many rows:
// big 56^2 layers produce ~12k per-wave partial rows, and feeding those to
// the 16-row-lane finalize kernel directly leaves one ~800-iteration load
// chain per thread on a near-idle grid. Collapse to 64 rows first with a
// (width/256 x 64) grid, then finalize.
__global__ void collapse_partials_kernel(const float* __restrict__ part,
                                         float* __restrict__ out, int width,
                                         int nblocks, int rows_per) {
  const int c = blockIdx.x * 256 + threadIdx.x;
  if (c >= width) return;
  const int r0 = blockIdx.y * rows_per;
  const int r1 = min(r0 + rows_per, nblocks);
  float acc = 0.f;
  int r = r0;
  for (; r + 3 < r1; r += 4)
    acc += (part[(int64_t)r * width + c] +
            part[(int64_t)(r + 1) * width + c]) +
           (part[(int64_t)(r + 2) * width + c] +
            part[(int64_t)(r + 3) * width + c]);
  for (; r < r1; ++r) acc += part[(int64_t)r * width + c];
  out[(int64_t)blockIdx.y * width + c] = acc;
}

at::Tensor maybe_collapse_partials(at::Tensor part) {
  const int nblocks = part.size(0);
  if (nblocks <= 512) return part;
  const int width = part.size(1);
  constexpr int CROWS = 64;
  auto out = at::empty({CROWS, width}, part.options());
  const int rows_per = (int)ceil_div(nblocks, CROWS);
  dim3 grid((int)ceil_div(width, 256), CROWS);
  hipLaunchKernelGGL(collapse_partials_kernel, grid, dim3(256), 0,
                     cur_stream(), part.data_ptr<float>(),
                     out.data_ptr<float>(), width, nblocks, rows_per);
  return out;
}

// stage-1 only helper: per-block partials [grid, 2C]
std::pair<at::Tensor, int> bn_partials(const at::Tensor& x) {
  const int C = x.size(1);
  const int64_t rows = x.numel() / C;
  auto opts = x.options().dtype(at::kFloat);
  at::Tensor part;
  int grid_out = 0;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_partials", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    const int cpacks = C / V;
    const int nrl = std::max(256 / cpacks, 1);
    const int64_t rpb = pick_rows_per_block(rows, nrl);
    const int grid = (int)ceil_div(rows, rpb);
    part = at::empty({grid, 2 * C}, opts);
    hipLaunchKernelGGL((bn_sums_kernel<scalar_t>), dim3(grid), dim3(256), 0,
                       cur_stream(), (const scalar_t*)x.data_ptr(),
                       part.data_ptr<float>(), rows, C, rpb);
    grid_out = grid;
  });
  return {part, grid_out};
}

}  // namespace

// Column-reduce an [nblocks, width] fp32 partial matrix to [width] — used by
// SyncBN to turn conv-epilogue partials (F1) into the local [2C] sums it
// all-reduces across ranks.
at::Tensor bn_reduce_partials(at::Tensor part) {
  CHECK_GPU(part);
  TORCH_CHECK(part.dim() == 2 && part.scalar_type() == at::kFloat,
              "bn_reduce_partials: [nblocks, width] fp32");
  part = maybe_collapse_partials(part);
  const int width = part.size(1);
  const int nblocks = part.size(0);
  auto out = at::empty({width}, part.options());
  hipLaunchKernelGGL(reduce_partials_kernel, dim3((int)ceil_div(width, 64)),
                     dim3(256), 0, cur_stream(), part.data_ptr<float>(),
                     out.data_ptr<float>(), width, nblocks);
  return out;
}

std::vector<at::Tensor> bn_sums(at::Tensor x) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int C = x.size(1);
  auto pg = bn_partials(x);
  auto both = at::empty({2 * C}, x.options().dtype(at::kFloat));
  hipLaunchKernelGGL(reduce_partials_kernel, dim3((int)ceil_div(2 * C, 64)),
                     dim3(256), 0, cur_stream(), pg.first.data_ptr<float>(),
                     both.data_ptr<float>(), 2 * C, pg.second);
  return {both.narrow(0, 0, C), both.narrow(0, C, C)};
}

// One-shot stats: partials -> (mean, rstd, scale, shift) + running update.
// part_opt: per-block [nblocks, 2C] sum/sumsq partials already produced by
// the PRODUCING conv's epilogue (F1 of docs/DESIGN_bn_conv_fusion.md) —
// when given, the bn_sums pass over x is skipped entirely.
std::vector<at::Tensor> bn_stats(at::Tensor x, at::Tensor gamma,
                                 at::Tensor beta,
                                 c10::optional<at::Tensor> rm_opt,
                                 c10::optional<at::Tensor> rv_opt,
                                 double momentum, double eps, bool training,
                                 c10::optional<at::Tensor> part_opt) {
  at::Tensor running_mean = rm_opt.has_value() ? *rm_opt : at::Tensor();
  at::Tensor running_var = rv_opt.has_value() ? *rv_opt : at::Tensor();
  const int C = x.size(1);
  auto opts = x.options().dtype(at::kFloat);
  auto mean = at::empty({C}, opts);
  auto rstd = at::empty({C}, opts);
  auto scale = at::empty({C}, opts);
  auto shift = at::empty({C}, opts);
  if (training) {
    std::pair<at::Tensor, int> pg;
    if (part_opt.has_value()) {
      TORCH_CHECK(part_opt->size(1) == 2 * C, "bn_stats: partials width");
      auto part = maybe_collapse_partials(*part_opt);
      pg = {part, (int)part.size(0)};
    } else {
      pg = bn_partials(x);
    }
    const int64_t rows = x.numel() / C;
    const float inv_cnt = 1.f / (float)rows;
    const float unbias = rows > 1 ? (float)rows / (float)(rows - 1) : 1.f;
    hipLaunchKernelGGL(bn_finalize_kernel, dim3((int)ceil_div(C, 16)),
                       dim3(256), 0, cur_stream(), pg.first.data_ptr<float>(),
                       pg.second, gamma.data_ptr<float>(),
                       beta.data_ptr<float>(),
                       running_mean.defined() ? running_mean.data_ptr<float>()
                                              : nullptr,
                       running_var.defined() ? running_var.data_ptr<float>()
                                             : nullptr,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(), C,
                       inv_cnt, unbias, (float)momentum, (float)eps,
                       running_mean.defined() ? 1 : 0);
  } else {
    hipLaunchKernelGGL(bn_eval_prep_kernel, dim3((int)ceil_div(C, 256)),
                       dim3(256), 0, cur_stream(),
                       running_mean.data_ptr<float>(),
                       running_var.data_ptr<float>(), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), scale.data_ptr<float>(),
                       shift.data_ptr<float>(), C, (float)eps);
  }
  return {mean, rstd, scale, shift};
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
    const int64_t rows = x.numel() / C;
    const int cpacks = C / V;
    const int nrl = std::max(256 / cpacks, 1);
    const int64_t rpb = pick_rows_per_block(rows, nrl);
    const int rgrid = (int)ceil_div(rows, rpb);
    auto stream = cur_stream();
    if (has_res)
      hipLaunchKernelGGL((bn_apply_kernel<scalar_t, true>), dim3(rgrid),
                         dim3(256), 0, stream, (const scalar_t*)x.data_ptr(),
                         scale.data_ptr<float>(), shift.data_ptr<float>(),
                         (const scalar_t*)res->data_ptr(),
                         (scalar_t*)y.data_ptr(), rows, C, rpb, (int)act);
    else
      hipLaunchKernelGGL((bn_apply_kernel<scalar_t, false>), dim3(rgrid),
                         dim3(256), 0, stream, (const scalar_t*)x.data_ptr(),
                         scale.data_ptr<float>(), shift.data_ptr<float>(),
                         (const scalar_t*)nullptr, (scalar_t*)y.data_ptr(),
                         rows, C, rpb, (int)act);
  });
  return y;
}

// ---- padded-output BN apply (bn1 -> 3x3 v2 conv fusion) --------------------
// Writes y into a [N, C, H+2ph, W+2pw] canvas at (h+ph, w+pw) so the
// consuming v2 conv reads it pad-free and the per-step pad_image pass
// disappears. No residual (bn1/bn2 only). The interior walk advances
// (n, h, w) incrementally (requires nrl <= W, host-gated).
template <typename T>
__global__ __launch_bounds__(256, 2) void bn_apply_pad_kernel(
    const T* __restrict__ x, const float* __restrict__ scale,
    const float* __restrict__ shift, T* __restrict__ y, int N, int H, int W,
    int C, int ph, int pw, int64_t rows_per_block, int act) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2_floor(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const int64_t rows = (int64_t)N * H * W;
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  const int Hp = H + 2 * ph, Wp = W + 2 * pw;
  const P* xp = reinterpret_cast<const P*>(x);
  P* yp = reinterpret_cast<P*>(y);
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float sc[V], sh[V];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      sc[j] = scale[cp * V + j];
      sh[j] = shift[cp * V + j];
    }
    if (!active) continue;
    int64_t row = row0 + rl;
    if (row >= row1) continue;
    const int64_t rstep = (int64_t)nrl * cpacks;
    const P* xq = xp + row * cpacks + cp;
    // interior (n, h, w) from the flat row, then incremental advance
    int n = (int)(row / ((int64_t)H * W));
    int rem = (int)(row - (int64_t)n * H * W);
    int h = rem / W, w = rem - (rem / W) * W;
    for (; row + 3 * (int64_t)nrl < row1; row += 4 * (int64_t)nrl) {
      P px4[4];
      int64_t yo[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        px4[u] = xq[u * rstep];
        yo[u] = ((((int64_t)n * Hp + h + ph) * Wp) + w + pw) * cpacks + cp;
        w += nrl;
        if (w >= W) {
          w -= W;
          if (++h == H) {
            h = 0;
            ++n;
          }
        }
      }
      xq += 4 * rstep;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
#pragma unroll
        for (int j = 0; j < V; ++j) {
          const float z = to_f32(px4[u].v[j]) * sc[j] + sh[j];
          px4[u].v[j] = from_f32<T>(act_apply(z, act));
        }
        yp[yo[u]] = px4[u];
      }
    }
    for (; row < row1; row += nrl) {
      P px = xq[0];
      xq += rstep;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        const float z = to_f32(px.v[j]) * sc[j] + sh[j];
        px.v[j] = from_f32<T>(act_apply(z, act));
      }
      yp[((((int64_t)n * Hp + h + ph) * Wp) + w + pw) * cpacks + cp] = px;
      w += nrl;
      if (w >= W) {
        w -= W;
        if (++h == H) {
          h = 0;
          ++n;
        }
      }
    }
  }
}

// zero the pad ring of a [N, C, Hp, Wp] canvas (top/bottom rows full,
// left/right strips on interior rows)
template <typename T>
__global__ void pad_ring_zero_kernel(T* __restrict__ y, int N, int H, int W,
                                     int C, int ph, int pw) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpacks = C / V;
  const int Hp = H + 2 * ph, Wp = W + 2 * pw;
  const int row = blockIdx.y;  // n * Hp + hp
  const int hp_ = row % Hp;
  const bool full = hp_ < ph || hp_ >= H + ph;
  const int strip = 2 * pw * cpacks;  // packs in the two side strips
  const int npacks = full ? Wp * cpacks : strip;
  P z = {};
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < npacks;
       i += gridDim.x * blockDim.x) {
    int col;
    if (full) {
      col = i;
    } else {
      const int half = i / (pw * cpacks);
      const int off = i - half * (pw * cpacks);
      col = half == 0 ? off : (W + pw) * cpacks + off;
    }
    reinterpret_cast<P*>(y)[(int64_t)row * Wp * cpacks + col] = z;
  }
}

// pad-aware gy addressing for BN backward (bn1's gy arrives as the
// consuming conv's padded-canvas dgrad): gy[(h+ph, w+pw)] of [Hp, Wp],
// x/gx stay dense. No residual, no mask.
template <typename T, int ACT>
__global__ __launch_bounds__(256, 2) void bn_bwd_reduce_pad_kernel(
    const T* __restrict__ gy, const T* __restrict__ x,
    const float* __restrict__ scale, const float* __restrict__ shift,
    float* __restrict__ part, int N, int H, int W, int C, int ph, int pw,
    int64_t rows_per_block) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  __shared__ float red[256 * 2 * (16 / sizeof(T) > 8 ? 16 / sizeof(T) : 8)];
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2_floor(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const int64_t rows = (int64_t)N * H * W;
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  const int Hp = H + 2 * ph, Wp = W + 2 * pw;
  const P* gp = reinterpret_cast<const P*>(gy);
  const P* xp = reinterpret_cast<const P*>(x);
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float sc[V], sh[V];
    if (ACT != 0) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        sc[j] = scale[cp * V + j];
        sh[j] = shift[cp * V + j];
      }
    }
    float accg[V] = {}, accgx[V] = {};
    if (active && row0 + rl < row1) {
      int64_t row = row0 + rl;
      const int64_t rstep = (int64_t)nrl * cpacks;
      const P* xq = xp + row * cpacks + cp;
      int n = (int)(row / ((int64_t)H * W));
      int rem = (int)(row - (int64_t)n * H * W);
      int h = rem / W, w = rem - (rem / W) * W;
      for (; row + 3 * (int64_t)nrl < row1; row += 4 * (int64_t)nrl) {
        P px4[4], pg4[4];
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          px4[u] = xq[u * rstep];
          pg4[u] =
              gp[((((int64_t)n * Hp + h + ph) * Wp) + w + pw) * cpacks + cp];
          w += nrl;
          if (w >= W) {
            w -= W;
            if (++h == H) {
              h = 0;
              ++n;
            }
          }
        }
        xq += 4 * rstep;
#pragma unroll
        for (int u = 0; u < 4; ++u)
#pragma unroll
          for (int j = 0; j < V; ++j) {
            float xv = to_f32(px4[u].v[j]);
            float g = to_f32(pg4[u].v[j]);
            if (ACT != 0) g *= act_grad(xv * sc[j] + sh[j], ACT);
            accg[j] += g;
            accgx[j] += g * xv;
          }
      }
      for (; row < row1; row += nrl) {
        P px = xq[0];
        xq += rstep;
        P pg = gp[((((int64_t)n * Hp + h + ph) * Wp) + w + pw) * cpacks + cp];
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float xv = to_f32(px.v[j]);
          float g = to_f32(pg.v[j]);
          if (ACT != 0) g *= act_grad(xv * sc[j] + sh[j], ACT);
          accg[j] += g;
          accgx[j] += g * xv;
        }
        w += nrl;
        if (w >= W) {
          w -= W;
          if (++h == H) {
            h = 0;
            ++n;
          }
        }
      }
    }
    float* slot = &red[(rl * ncp + cp0) * 2 * V];
    if (active) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        slot[j] = accg[j];
        slot[V + j] = accgx[j];
      }
    }
    __syncthreads();
    for (int st = nrl >> 1; st > 0; st >>= 1) {
      if (active && rl < st) {
        const float* other = &red[((rl + st) * ncp + cp0) * 2 * V];
#pragma unroll
        for (int j = 0; j < 2 * V; ++j) slot[j] += other[j];
      }
      __syncthreads();
    }
    if (active && rl == 0) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        part[(int64_t)blockIdx.x * 2 * C + cp * V + j] = slot[j];
        part[(int64_t)blockIdx.x * 2 * C + C + cp * V + j] = slot[V + j];
      }
    }
    __syncthreads();
  }
}

template <typename T, int ACT>
__global__ __launch_bounds__(256, 2) void bn_bwd_dx_pad_kernel(
    const T* __restrict__ gy, const T* __restrict__ x,
    const float* __restrict__ scale, const float* __restrict__ shift,
    const float* __restrict__ P1c, const float* __restrict__ P2c,
    const float* __restrict__ P3c, T* __restrict__ gx, int N, int H, int W,
    int C, int ph, int pw, int64_t rows_per_block) {
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpacks = C / V;
  const int ncp = min(cpacks, (int)blockDim.x);
  const int nrl = p2_floor(blockDim.x / ncp);
  const int cp0 = threadIdx.x % ncp;
  const int rl = threadIdx.x / ncp;
  const bool active = rl < nrl;
  const int64_t rows = (int64_t)N * H * W;
  const int64_t row0 = (int64_t)blockIdx.x * rows_per_block;
  const int64_t row1 = min(row0 + rows_per_block, rows);
  const int Hp = H + 2 * ph, Wp = W + 2 * pw;
  const P* gp = reinterpret_cast<const P*>(gy);
  const P* xp = reinterpret_cast<const P*>(x);
  P* oxp = reinterpret_cast<P*>(gx);
  for (int cp = cp0; cp < cpacks; cp += ncp) {
    float sc[V], sh[V], a1[V], a2[V], a3[V];
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const int cc = cp * V + j;
      a1[j] = P1c[cc];
      a2[j] = P2c[cc];
      a3[j] = P3c[cc];
      if (ACT != 0) {
        sc[j] = scale[cc];
        sh[j] = shift[cc];
      }
    }
    if (!active) continue;
    int64_t row = row0 + rl;
    if (row >= row1) continue;
    const int64_t rstep = (int64_t)nrl * cpacks;
    const P* xq = xp + row * cpacks + cp;
    P* oq = oxp + row * cpacks + cp;
    int n = (int)(row / ((int64_t)H * W));
    int rem = (int)(row - (int64_t)n * H * W);
    int h = rem / W, w = rem - (rem / W) * W;
    for (; row + 3 * (int64_t)nrl < row1; row += 4 * (int64_t)nrl) {
      P px4[4], pg4[4];
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        px4[u] = xq[u * rstep];
        pg4[u] =
            gp[((((int64_t)n * Hp + h + ph) * Wp) + w + pw) * cpacks + cp];
        w += nrl;
        if (w >= W) {
          w -= W;
          if (++h == H) {
            h = 0;
            ++n;
          }
        }
      }
      xq += 4 * rstep;
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        P ox;
#pragma unroll
        for (int j = 0; j < V; ++j) {
          float xv = to_f32(px4[u].v[j]);
          float g = to_f32(pg4[u].v[j]);
          if (ACT != 0) g *= act_grad(xv * sc[j] + sh[j], ACT);
          ox.v[j] = from_f32<T>(a1[j] * g + a3[j] * xv + a2[j]);
        }
        oq[u * rstep] = ox;
      }
      oq += 4 * rstep;
    }
    for (; row < row1; row += nrl) {
      P px = xq[0];
      xq += rstep;
      P pg = gp[((((int64_t)n * Hp + h + ph) * Wp) + w + pw) * cpacks + cp];
      P ox;
#pragma unroll
      for (int j = 0; j < V; ++j) {
        float xv = to_f32(px.v[j]);
        float g = to_f32(pg.v[j]);
        if (ACT != 0) g *= act_grad(xv * sc[j] + sh[j], ACT);
        ox.v[j] = from_f32<T>(a1[j] * g + a3[j] * xv + a2[j]);
      }
      oq[0] = ox;
      oq += rstep;
      w += nrl;
      if (w >= W) {
        w -= W;
        if (++h == H) {
          h = 0;
          ++n;
        }
      }
    }
  }
}

// Residual relu BN apply that ALSO emits the act' bitmask (one byte per
// V-pack) so the backward kernels can drop the res stream and the z
// recompute. Returns {y, mask}.
std::vector<at::Tensor> bn_apply_act_mask(at::Tensor x, at::Tensor scale,
                                          at::Tensor shift, int64_t act,
                                          at::Tensor res) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  TORCH_CHECK(act == 1, "mask emission: relu only");
  const int C = x.size(1);
  auto y = at::empty_like(x);
  at::Tensor mask;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_apply_act_mask", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    const int64_t rows = x.numel() / C;
    const int cpacks = C / V;
    mask = at::empty({rows * cpacks}, x.options().dtype(at::kByte));
    const int nrl = std::max(256 / cpacks, 1);
    const int64_t rpb = pick_rows_per_block(rows, nrl);
    const int rgrid = (int)ceil_div(rows, rpb);
    hipLaunchKernelGGL((bn_apply_kernel<scalar_t, true, true>), dim3(rgrid),
                       dim3(256), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       (const scalar_t*)res.data_ptr(),
                       (scalar_t*)y.data_ptr(), rows, C, rpb, (int)act,
                       (unsigned char*)mask.data_ptr());
  });
  return {y, mask};
}

// bn1 apply writing into a padded canvas (consumer 3x3 v2 conv reads it
// pad-free). Returns y [N, C, H+2ph, W+2pw] with a zeroed ring.
at::Tensor bn_apply_act_pad(at::Tensor x, at::Tensor scale, at::Tensor shift,
                            int64_t act, int64_t ph, int64_t pw) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int Hp = H + 2 * (int)ph, Wp = W + 2 * (int)pw;
  auto y = at::empty({N, C, Hp, Wp},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_apply_act_pad", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    const int cpacks = C / V;
    int nrl = 1;
    while (nrl * 2 <= std::max(256 / cpacks, 1)) nrl *= 2;
    TORCH_CHECK(nrl <= W, "bn pad apply: nrl > W (host gate missed)");
    const int64_t rows = (int64_t)N * H * W;
    const int64_t rpb = pick_rows_per_block(rows, std::max(256 / cpacks, 1));
    const int rgrid = (int)ceil_div(rows, rpb);
    auto stream = cur_stream();
    const int ring_max = std::max(Wp * cpacks, 2 * (int)pw * cpacks);
    hipLaunchKernelGGL((pad_ring_zero_kernel<scalar_t>),
                       dim3((ring_max + 255) / 256, N * Hp), dim3(256), 0,
                       stream, (scalar_t*)y.data_ptr(), N, H, W, C, (int)ph,
                       (int)pw);
    hipLaunchKernelGGL((bn_apply_pad_kernel<scalar_t>), dim3(rgrid),
                       dim3(256), 0, stream, (const scalar_t*)x.data_ptr(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       (scalar_t*)y.data_ptr(), N, H, W, C, (int)ph, (int)pw,
                       rpb, (int)act);
  });
  return y;
}

template <typename scalar_t>
void launch_bwd_reduce_pad(const at::Tensor& gy, const at::Tensor& x,
                           const at::Tensor& scale, const at::Tensor& shift,
                           at::Tensor& part, int N, int H, int W, int C,
                           int ph, int pw, int64_t rpb, int act, int rgrid,
                           hipStream_t stream) {
#define BRP_CASE(A)                                                         \
  hipLaunchKernelGGL((bn_bwd_reduce_pad_kernel<scalar_t, A>), dim3(rgrid),  \
                     dim3(256), 0, stream,                                  \
                     (const scalar_t*)gy.data_ptr(),                        \
                     (const scalar_t*)x.data_ptr(),                         \
                     scale.data_ptr<float>(), shift.data_ptr<float>(),      \
                     part.data_ptr<float>(), N, H, W, C, ph, pw, rpb)
  switch (act) {
    case 0: BRP_CASE(0); break;
    case 1: BRP_CASE(1); break;
    case 2: BRP_CASE(2); break;
    default: BRP_CASE(3); break;
  }
#undef BRP_CASE
}

template <typename scalar_t>
void launch_dx_pad(const at::Tensor& gy, const at::Tensor& x,
                   const at::Tensor& scale, const at::Tensor& shift,
                   const at::Tensor& P1, const at::Tensor& P2,
                   const at::Tensor& P3, at::Tensor& gx, int N, int H, int W,
                   int C, int ph, int pw, int64_t rpb, int act, int rgrid,
                   hipStream_t stream) {
#define BDP_CASE(A)                                                        \
  hipLaunchKernelGGL((bn_bwd_dx_pad_kernel<scalar_t, A>), dim3(rgrid),     \
                     dim3(256), 0, stream,                                 \
                     (const scalar_t*)gy.data_ptr(),                       \
                     (const scalar_t*)x.data_ptr(),                        \
                     scale.data_ptr<float>(), shift.data_ptr<float>(),     \
                     P1.data_ptr<float>(), P2.data_ptr<float>(),           \
                     P3.data_ptr<float>(), (scalar_t*)gx.data_ptr(), N, H, \
                     W, C, ph, pw, rpb)
  switch (act) {
    case 0: BDP_CASE(0); break;
    case 1: BDP_CASE(1); break;
    case 2: BDP_CASE(2); break;
    default: BDP_CASE(3); break;
  }
#undef BDP_CASE
}

// BN backward when gy arrives in the padded canvas layout (x, gx dense).
std::vector<at::Tensor> bn_bwd_pad(at::Tensor gy, at::Tensor x,
                                   at::Tensor mean, at::Tensor rstd,
                                   at::Tensor gamma, at::Tensor scale,
                                   at::Tensor shift, int64_t act,
                                   bool training, int64_t ph, int64_t pw) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  TORCH_CHECK(gy.size(2) == H + 2 * ph && gy.size(3) == W + 2 * pw,
              "bn_bwd_pad: gy canvas shape");
  const int64_t rows = (int64_t)N * H * W;
  auto fopts = x.options().dtype(at::kFloat);
  auto gx = at::empty_like(x);
  auto gw = at::empty({C}, fopts);
  auto gb = at::empty({C}, fopts);
  auto P1 = at::empty({C}, fopts);
  auto P2 = at::empty({C}, fopts);
  auto P3 = at::empty({C}, fopts);
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_bwd_pad", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    auto stream = cur_stream();
    const int cpacks = C / V;
    const int nrl = std::max(256 / cpacks, 1);
    int nrl_p2 = 1;
    while (nrl_p2 * 2 <= nrl) nrl_p2 *= 2;
    TORCH_CHECK(nrl_p2 <= W, "bn_bwd_pad: nrl > W");
    const int64_t rpb = pick_rows_per_block(rows, nrl);
    const int rgrid = (int)ceil_div(rows, rpb);
    auto part = at::empty({rgrid, 2 * C}, fopts);
    launch_bwd_reduce_pad<scalar_t>(gy, x, scale, shift, part, N, H, W, C,
                                    (int)ph, (int)pw, rpb, (int)act, rgrid,
                                    stream);
    const float inv_cnt = 1.f / (float)rows;
    hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((int)ceil_div(C, 16)),
                       dim3(256), 0, stream, part.data_ptr<float>(), rgrid,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), gw.data_ptr<float>(),
                       gb.data_ptr<float>(), P1.data_ptr<float>(),
                       P2.data_ptr<float>(), P3.data_ptr<float>(), C, inv_cnt,
                       training ? 1 : 0);
    launch_dx_pad<scalar_t>(gy, x, scale, shift, P1, P2, P3, gx, N, H, W,
                            C, (int)ph, (int)pw, rpb, (int)act, rgrid,
                            stream);
  });
  return {gx, gw, gb};
}

// Local raw grad-stat sums [2C] = [sum(g), sum(g*x)] (stage 1 + stage 2);
// used standalone by SyncBN (which all-reduces the result across ranks).
at::Tensor bn_bwd_stats(at::Tensor gy, at::Tensor x,
                        c10::optional<at::Tensor> res, at::Tensor scale,
                        at::Tensor shift, int64_t act) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int C = x.size(1);
  const int64_t rows = x.numel() / C;
  auto fopts = x.options().dtype(at::kFloat);
  const bool has_res = res.has_value();
  at::Tensor both;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_bwd_stats", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    auto stream = cur_stream();
    const int cpacks = C / V;
    const int nrl = std::max(256 / cpacks, 1);
    const int64_t rpb = pick_rows_per_block(rows, nrl);
    const int rgrid = (int)ceil_div(rows, rpb);
    auto part = at::empty({rgrid, 2 * C}, fopts);
    const scalar_t* resp =
        has_res ? (const scalar_t*)res->data_ptr() : nullptr;
    launch_bwd_reduce<scalar_t>(
        (const scalar_t*)gy.data_ptr(), (const scalar_t*)x.data_ptr(),
        resp, scale.data_ptr<float>(),
        shift.data_ptr<float>(), part.data_ptr<float>(), rows, C, rpb,
        (int)act, rgrid, stream);
    both = at::empty({2 * C}, fopts);
    hipLaunchKernelGGL(reduce_partials_kernel, dim3((int)ceil_div(2 * C, 64)),
                       dim3(256), 0, stream, part.data_ptr<float>(),
                       both.data_ptr<float>(), 2 * C, rgrid);
  });
  return both;
}

// dx/gw/gb given (possibly cross-rank-reduced) raw sums and the TOTAL count.
std::vector<at::Tensor> bn_bwd_apply(at::Tensor gy, at::Tensor x,
                                     c10::optional<at::Tensor> res,
                                     at::Tensor mean, at::Tensor rstd,
                                     at::Tensor gamma, at::Tensor scale,
                                     at::Tensor shift, at::Tensor sums,
                                     double total_count, int64_t act,
                                     bool training, bool need_gres) {
  const int C = x.size(1);
  const int64_t rows = x.numel() / C;
  auto fopts = x.options().dtype(at::kFloat);
  const bool has_res = res.has_value();
  auto gx = at::empty_like(x);
  auto gres = has_res ? at::empty_like(x) : at::Tensor();
  auto gw = at::empty({C}, fopts);
  auto gb = at::empty({C}, fopts);
  auto P1 = at::empty({C}, fopts);
  auto P2 = at::empty({C}, fopts);
  auto P3 = at::empty({C}, fopts);
  (void)need_gres;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_bwd_apply", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    auto stream = cur_stream();
    const float inv_cnt = 1.f / (float)total_count;
    // sums is a flat [2C] row => treat as a 1-block partial
    hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((int)ceil_div(C, 16)),
                       dim3(256), 0, stream, sums.data_ptr<float>(), 1,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), gw.data_ptr<float>(),
                       gb.data_ptr<float>(), P1.data_ptr<float>(),
                       P2.data_ptr<float>(), P3.data_ptr<float>(), C, inv_cnt,
                       training ? 1 : 0);
    const int cpacks = C / V;
    const int nrl = std::max(256 / cpacks, 1);
    const int64_t rpb = pick_rows_per_block(rows, nrl);
    const int rgrid = (int)ceil_div(rows, rpb);
    const scalar_t* resp =
        has_res ? (const scalar_t*)res->data_ptr() : nullptr;
    scalar_t* gresp = has_res ? (scalar_t*)gres.data_ptr() : nullptr;
    launch_dx<scalar_t>(gy, x, resp, scale, shift, P1, P2, P3, gx, gresp,
                        rows, C, rpb, (int)act, rgrid, stream);
  });
  return {gx, gw, gb, gres};
}

// Full backward. scale/shift are the forward's fused coefficients (needed to
// recompute z for the activation backward; the y stream is never read).
std::vector<at::Tensor> bn_bwd(at::Tensor gy, at::Tensor x,
                               c10::optional<at::Tensor> res, at::Tensor mean,
                               at::Tensor rstd, at::Tensor gamma,
                               at::Tensor scale, at::Tensor shift,
                               int64_t act, bool training, bool need_gres,
                               c10::optional<at::Tensor> mask) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int C = x.size(1);
  const int64_t rows = x.numel() / C;
  auto fopts = x.options().dtype(at::kFloat);
  const bool has_res = res.has_value();
  auto gx = at::empty_like(x);
  auto gres = has_res ? at::empty_like(x) : at::Tensor();
  auto gw = at::empty({C}, fopts);
  auto gb = at::empty({C}, fopts);
  auto P1 = at::empty({C}, fopts);
  auto P2 = at::empty({C}, fopts);
  auto P3 = at::empty({C}, fopts);
  (void)need_gres;

  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "bn_bwd", [&] {
    constexpr int V = 16 / sizeof(scalar_t);
    TORCH_CHECK(C % V == 0, "C must be divisible by ", V);
    auto stream = cur_stream();
    const int cpacks = C / V;
    const int nrl = std::max(256 / cpacks, 1);
    const int64_t rpb = pick_rows_per_block(rows, nrl);
    const int rgrid = (int)ceil_div(rows, rpb);
    auto part = at::empty({rgrid, 2 * C}, fopts);
    const scalar_t* resp =
        has_res ? (const scalar_t*)res->data_ptr() : nullptr;
    const unsigned char* maskp =
        mask.has_value() ? (const unsigned char*)mask->data_ptr() : nullptr;
    launch_bwd_reduce<scalar_t>(
        (const scalar_t*)gy.data_ptr(), (const scalar_t*)x.data_ptr(),
        resp, scale.data_ptr<float>(),
        shift.data_ptr<float>(), part.data_ptr<float>(), rows, C, rpb,
        (int)act, rgrid, stream, maskp);
    const float inv_cnt = 1.f / (float)rows;
    hipLaunchKernelGGL(bn_bwd_finalize_kernel, dim3((int)ceil_div(C, 16)),
                       dim3(256), 0, stream, part.data_ptr<float>(), rgrid,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), gw.data_ptr<float>(),
                       gb.data_ptr<float>(), P1.data_ptr<float>(),
                       P2.data_ptr<float>(), P3.data_ptr<float>(), C, inv_cnt,
                       training ? 1 : 0);
    scalar_t* gresp = has_res ? (scalar_t*)gres.data_ptr() : nullptr;
    launch_dx<scalar_t>(gy, x, resp, scale, shift, P1, P2, P3, gx, gresp,
                        rows, C, rpb, (int)act, rgrid, stream, maskp);
  });
  return {gx, gw, gb, gres};
}
