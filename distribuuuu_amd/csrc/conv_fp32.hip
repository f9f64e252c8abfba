// fp32 implicit-GEMM convolution for gfx950 (reference-default precision —
// the reference trains fp32 throughout, utils.py:355 there). Exact fp32:
// v_mfma_f32_16x16x4_f32 (f32 in / f32 accumulate, 157 TF peak = the fp32
// vector rate; no xf32 on gfx950). The bf16 path (conv.hip/conv2.hip) stays
// the flagship; this file makes fp32 training run NATIVE (no MIOpen) with
// the same NHWC layout, padding machinery and autograd wrappers.
//
// Kernels:
//   conv_igemm_fp32_kernel  — fwd/dgrad 128x128x16 LDS-staged tile, 4 waves
//                             (2x2), 4x4 fragments of mfma_f32_16x16x4f32;
//                             handles groups, strided/offset output scatter
//                             (dgrad parity/stride paths) and the optional
//                             BN-partials epilogue (F1).
//   conv_wgrad_fp32_kernel  — 64x64 k x rsc tile, m-steps of 16 staged
//                             m-major in LDS and read column-wise (pad
//                             breaks conflicts), split-m fp32 atomics.
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4f;

namespace {

constexpr int FBM = 128, FBN = 128, FBK = 16;
constexpr int FLK = FBK + 1;  // +1 f32 pad: conflict-free column reads

struct ConvF32Params {
  const float* x;  // [N,H,W,Ct] NHWC
  const float* w;  // [Kt,R,S,Cg]
  float* y;        // [N,HoA,WoA,Kt]
  int N, H, W, Ct, Kt;
  int R, S, Cg, Kg;
  int sh, sw, ph, pw, dh, dw;
  int Ho, Wo;
  int HoA, WoA, osh, osw, oh0, ow0;
  int M, nspan, ksteps;
  int tiles_m;
  float* part;  // EMIT: [tiles_m*2, 2*Kt]
};

template <bool EMIT>
__global__ __launch_bounds__(256) void conv_igemm_fp32_kernel(
    ConvF32Params p) {
  const int g = blockIdx.z;
  int tile_m = blockIdx.x, tile_n = blockIdx.y;
  {  // XCD-aware bijective remap over m-tiles
    const int nwg = p.tiles_m;
    const int q = nwg / 8, r8 = nwg % 8;
    const int xcd = tile_m % 8, idx = tile_m / 8;
    tile_m = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q) + idx;
  }

  __shared__ float ldsA[2][FBM][FLK];
  __shared__ float ldsB[2][FBN][FLK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int il = lane & 15, kg = lane >> 4;

  // staging: thread t loads 8 floats (2 float4s) of row t>>1 at col (t&1)*8
  const int srow = tid >> 1;
  const int scol8 = (tid & 1) << 3;
  int an, ahbase, awbase;
  bool arow_ok;
  {
    const int m = tile_m * FBM + srow;
    arow_ok = m < p.M;
    const int mm = arow_ok ? m : 0;
    const int n = mm / (p.Ho * p.Wo);
    const int rem = mm % (p.Ho * p.Wo);
    an = n;
    ahbase = (rem / p.Wo) * p.sh - p.ph;
    awbase = (rem % p.Wo) * p.sw - p.pw;
  }
  const int SCg = p.S * p.Cg;
  int bk;
  bool brow_ok;
  {
    const int k = tile_n * FBN + srow;
    brow_ok = k < p.Kg;
    bk = g * p.Kg + (brow_ok ? k : 0);
  }

  uint4 regA[2], regB[2];
  auto stage_load = [&](int ks) {
    const int span0 = (ks % p.nspan) * FBK;
    const int r = ks / p.nspan;
    const int span = span0 + scol8;
    const bool span_ok = span < SCg;
    const int s = span_ok ? span / p.Cg : 0;
    const int c = span - s * p.Cg;
    const int h = ahbase + r * p.dh;
    const int w_ = awbase + s * p.dw;
    const bool ok =
        arow_ok && span_ok && h >= 0 && h < p.H && w_ >= 0 && w_ < p.W;
    const float* ap = p.x +
        (((int64_t)an * p.H + h) * p.W + w_) * p.Ct + g * p.Cg + c;
    regA[0] = ok ? *reinterpret_cast<const uint4*>(ap) : uint4{0, 0, 0, 0};
    regA[1] = ok ? *reinterpret_cast<const uint4*>(ap + 4) : uint4{0, 0, 0, 0};
    const bool okb = brow_ok && span_ok;
    const float* bp = p.w + ((int64_t)bk * p.R + r) * SCg + span;
    regB[0] = okb ? *reinterpret_cast<const uint4*>(bp) : uint4{0, 0, 0, 0};
    regB[1] = okb ? *reinterpret_cast<const uint4*>(bp + 4) : uint4{0, 0, 0, 0};
  };
  auto stage_write = [&](int buf) {
    *reinterpret_cast<uint4*>(&ldsA[buf][srow][scol8]) = regA[0];
    *reinterpret_cast<uint4*>(&ldsA[buf][srow][scol8 + 4]) = regA[1];
    *reinterpret_cast<uint4*>(&ldsB[buf][srow][scol8]) = regB[0];
    *reinterpret_cast<uint4*>(&ldsB[buf][srow][scol8 + 4]) = regB[1];
  };

  f32x4f acc[4][4] = {};
  stage_load(0);
  stage_write(0);
  __syncthreads();
  if (p.ksteps > 1) stage_load(1);

  int cur = 0;
  for (int ks = 0; ks < p.ksteps; ++ks) {
#pragma unroll
    for (int t4 = 0; t4 < 4; ++t4) {
      // A[l&15][k=l>>4] / B[k=l>>4][l&15] operand maps (one f32 per lane)
      float af[4], bf[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        af[mi] = ldsA[cur][wm * 64 + mi * 16 + il][t4 * 4 + kg];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bf[ni] = ldsB[cur][wn * 64 + ni * 16 + il][t4 * 4 + kg];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
    if (ks + 1 < p.ksteps) {
      stage_write(cur ^ 1);
      if (ks + 2 < p.ksteps) stage_load(ks + 2);
      __syncthreads();
    }
    cur ^= 1;
  }

  // epilogue: per-wave fp32 slab restage -> 16-wide float stores
  __syncthreads();
  float* slab = &ldsA[0][0][0] + wid * (16 * 68);
  const int HoWo = p.Ho * p.Wo;
  const int er = lane >> 2;
  const int ec = (lane & 3) << 4;
  float ps = 0.f, pq = 0.f;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr)
        slab[(kg * 4 + rr) * 68 + ni * 16 + il] = acc[mi][ni][rr];
    __builtin_amdgcn_wave_barrier();
    const int m = tile_m * FBM + wm * 64 + mi * 16 + er;
    if (m < p.M) {
      const int n = m / HoWo;
      const int rem = m - n * HoWo;
      const int64_t obase =
          (((int64_t)n * p.HoA + (rem / p.Wo) * p.osh + p.oh0) * p.WoA +
           (rem % p.Wo) * p.osw + p.ow0) * p.Kt + g * p.Kg;
      const int k0 = tile_n * FBN + wn * 64 + ec;
      if (k0 + 16 <= p.Kg) {
#pragma unroll
        for (int q = 0; q < 4; ++q) {
          uint4 u;
          u.x = __float_as_uint(slab[er * 68 + ec + q * 4 + 0]);
          u.y = __float_as_uint(slab[er * 68 + ec + q * 4 + 1]);
          u.z = __float_as_uint(slab[er * 68 + ec + q * 4 + 2]);
          u.w = __float_as_uint(slab[er * 68 + ec + q * 4 + 3]);
          *reinterpret_cast<uint4*>(&p.y[obase + k0 + q * 4]) = u;
        }
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j)
          if (k0 + j < p.Kg) p.y[obase + k0 + j] = slab[er * 68 + ec + j];
      }
    }
    if (EMIT) {
      // fp32 outputs are stored unrounded: sum the slab values directly
      const int base = tile_m * FBM + wm * 64 + mi * 16;
      const int rows_valid = (int)min((int64_t)16, (int64_t)p.M - base);
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        if (r < rows_valid) {
          const float v = slab[r * 68 + lane];
          ps += v;
          pq += v * v;
        }
      }
    }
    __builtin_amdgcn_wave_barrier();
  }
  if (EMIT) {
    const int kbase = tile_n * FBN + wn * 64;
    bn_partial_store(p.part,
                     (int64_t)(tile_m * 2 + wm) * 2 * p.Kt + g * p.Kg + kbase,
                     p.Kt, lane, min(64, p.Kg - kbase), ps, pq);
  }
}

// ---------------------------------------------------------------------------
// fp32 wgrad: gw[k][rsc] = sum_m gy[m][k] * patch(x)[m][rsc].
// 64x64 (k x rsc) tile per block, m-steps of 16 staged m-major in LDS
// ([16][64+pad] per operand) and read column-wise for the
// mfma_f32_16x16x4f32 operand maps; split-m chunks accumulate with fp32
// atomics directly into gw (no cast pass — gw IS fp32).
// ---------------------------------------------------------------------------
constexpr int WF_BM = 64, WF_BN = 64, WF_BK = 16;
constexpr int WF_LD = WF_BN + 4;        // column-read pad
constexpr int WF_CHUNK = 64;            // m-steps per block chunk (1024 px)

struct WgradF32Params {
  const float* x;   // [N,H,W,Ct]
  const float* gy;  // [N,Ho,Wo,Kt]
  float* gw;        // [Kt, R*S*Cg] zeroed fp32
  int N, H, W, Ct, Kt;
  int R, S, Cg, Kg;
  int sh, sw, ph, pw, dh, dw;
  int Ho, Wo;
  int M, RSC;
  int ktiles, ntiles, chunks;
  unsigned long long magicHoWo, magicWo;
};

DEV_INLINE int magic_div_f(int m, unsigned long long magic) {
  return (int)(((unsigned long long)(unsigned)m * magic) >> 47);
}

__global__ __launch_bounds__(256) void conv_wgrad_fp32_kernel(
    WgradF32Params p) {
  const int g = blockIdx.z;
  const int ktile = blockIdx.x % p.ktiles;
  const int ntile = blockIdx.x / p.ktiles;
  const int chunk = blockIdx.y;

  __shared__ float ldsG[2][WF_BK][WF_LD];
  __shared__ float ldsX[2][WF_BK][WF_LD];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int il = lane & 15, kq = lane >> 4;

  // staging: thread t loads 4 floats of m-row t>>4 at col (t&15)*4
  const int sml = tid >> 4;         // 0..15
  const int scol4 = (tid & 15) << 2;
  const int SCg = p.S * p.Cg;
  const int rsc = ntile * WF_BN + scol4;
  const bool rsc_ok = rsc < p.RSC;
  const int br = rsc_ok ? rsc / SCg : 0;
  const int brem = rsc - br * SCg;
  const int bs = brem / p.Cg;
  const int bc = brem - bs * p.Cg;
  const int kcol = ktile * WF_BM + scol4;
  const bool k_ok = kcol < p.Kg;

  const int m0 = chunk * (WF_CHUNK * WF_BK);
  const int HoWo = p.Ho * p.Wo;
  const int msteps = min(WF_CHUNK, (int)((p.M - m0 + WF_BK - 1) / WF_BK));

  uint4 regG, regX;
  auto stage_load = [&](int ms) {
    const int m = m0 + ms * WF_BK + sml;
    const bool m_ok = m < p.M;
    const int mm = m_ok ? m : 0;
    const int n = magic_div_f(mm, p.magicHoWo);
    const int rem = mm - n * HoWo;
    const int ho = magic_div_f(rem, p.magicWo);
    const int wo = rem - ho * p.Wo;
    regG = (m_ok && k_ok)
               ? *reinterpret_cast<const uint4*>(
                     p.gy + (int64_t)mm * p.Kt + g * p.Kg + kcol)
               : uint4{0, 0, 0, 0};
    const int h = ho * p.sh - p.ph + br * p.dh;
    const int w_ = wo * p.sw - p.pw + bs * p.dw;
    regX = (m_ok && rsc_ok && h >= 0 && h < p.H && w_ >= 0 && w_ < p.W)
               ? *reinterpret_cast<const uint4*>(
                     p.x + (((int64_t)n * p.H + h) * p.W + w_) * p.Ct +
                     g * p.Cg + bc)
               : uint4{0, 0, 0, 0};
  };
  auto stage_write = [&](int buf) {
    *reinterpret_cast<uint4*>(&ldsG[buf][sml][scol4]) = regG;
    *reinterpret_cast<uint4*>(&ldsX[buf][sml][scol4]) = regX;
  };

  f32x4f acc[2][2] = {};
  stage_load(0);
  stage_write(0);
  __syncthreads();
  if (msteps > 1) stage_load(1);

  int cur = 0;
  for (int ms = 0; ms < msteps; ++ms) {
#pragma unroll
    for (int mc = 0; mc < 4; ++mc) {
      float af[2], bf[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        af[mi] = ldsG[cur][mc * 4 + kq][wm * 32 + mi * 16 + il];
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        bf[ni] = ldsX[cur][mc * 4 + kq][wn * 32 + ni * 16 + il];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
    if (ms + 1 < msteps) {
      stage_write(cur ^ 1);
      if (ms + 2 < msteps) stage_load(ms + 2);
      __syncthreads();
    }
    cur ^= 1;
  }

  // D: col = lane&15, row = (lane>>4)*4 + rr
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int k = ktile * WF_BM + wm * 32 + mi * 16 + kq * 4 + rr;
      if (k >= p.Kg) continue;
      const int64_t rowbase = (int64_t)(g * p.Kg + k) * p.RSC;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int col = ntile * WF_BN + wn * 32 + ni * 16 + il;
        if (col < p.RSC) atomicAdd(&p.gw[rowbase + col], acc[mi][ni][rr]);
      }
    }
  }
}

}  // namespace

at::Tensor conv2d_fwd_into_fp32(at::Tensor x, at::Tensor w, at::Tensor y,
                                int64_t Ho, int64_t Wo, int64_t sh, int64_t sw,
                                int64_t ph, int64_t pw, int64_t dh, int64_t dw,
                                int64_t groups, int64_t osh, int64_t osw,
                                int64_t oh0, int64_t ow0,
                                at::Tensor* part_out) {
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == at::kFloat, "conv2d_fwd_fp32: fp32 only");
  check_nhwc(x, "x");
  check_nhwc(w, "w");
  const int N = x.size(0), Ct = x.size(1), H = x.size(2), W = x.size(3);
  const int Kt = w.size(0), Cg = w.size(1), R = w.size(2), S = w.size(3);
  TORCH_CHECK(Ct == Cg * groups, "channel/group mismatch");
  TORCH_CHECK(Cg % 8 == 0, "conv2d_fwd_fp32: per-group C must be 8-aligned");
  const int Kg = Kt / groups;
  ConvF32Params p;
  p.x = (const float*)x.data_ptr();
  p.w = (const float*)w.data_ptr();
  p.y = (float*)y.data_ptr();
  p.N = N; p.H = H; p.W = W; p.Ct = Ct; p.Kt = Kt;
  p.R = R; p.S = S; p.Cg = Cg; p.Kg = Kg;
  p.sh = sh; p.sw = sw; p.ph = ph; p.pw = pw; p.dh = dh; p.dw = dw;
  p.Ho = Ho; p.Wo = Wo;
  p.HoA = y.size(2); p.WoA = y.size(3);
  p.osh = osh; p.osw = osw; p.oh0 = oh0; p.ow0 = ow0;
  p.M = N * Ho * Wo;
  p.nspan = (S * Cg + FBK - 1) / FBK;
  p.ksteps = R * p.nspan;
  p.tiles_m = (p.M + FBM - 1) / FBM;
  p.part = nullptr;
  dim3 grid(p.tiles_m, (Kg + FBN - 1) / FBN, groups);
  if (part_out != nullptr) {
    *part_out = at::empty({(int64_t)p.tiles_m * 2, (int64_t)2 * Kt},
                          x.options().dtype(at::kFloat));
    p.part = part_out->data_ptr<float>();
    hipLaunchKernelGGL(conv_igemm_fp32_kernel<true>, grid, dim3(256), 0,
                       cur_stream(), p);
  } else {
    hipLaunchKernelGGL(conv_igemm_fp32_kernel<false>, grid, dim3(256), 0,
                       cur_stream(), p);
  }
  return y;
}

at::Tensor conv2d_wgrad_fp32(at::Tensor gy, at::Tensor x, int64_t R, int64_t S,
                             int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                             int64_t dh, int64_t dw, int64_t groups) {
  CHECK_GPU(gy);
  TORCH_CHECK(gy.scalar_type() == at::kFloat, "wgrad_fp32: fp32 only");
  check_nhwc(gy, "gy");
  check_nhwc(x, "x");
  const int N = x.size(0), Ct = x.size(1), H = x.size(2), W = x.size(3);
  const int Kt = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  const int Cg = Ct / groups, Kg = Kt / groups;
  TORCH_CHECK(Cg % 4 == 0 && Kg % 4 == 0, "wgrad_fp32: Cg/Kg 4-aligned");
  WgradF32Params p;
  p.x = (const float*)x.data_ptr();
  p.gy = (const float*)gy.data_ptr();
  p.N = N; p.H = H; p.W = W; p.Ct = Ct; p.Kt = Kt;
  p.R = R; p.S = S; p.Cg = Cg; p.Kg = Kg;
  p.sh = sh; p.sw = sw; p.ph = ph; p.pw = pw; p.dh = dh; p.dw = dw;
  p.Ho = Ho; p.Wo = Wo;
  p.M = N * Ho * Wo;
  p.RSC = R * S * Cg;
  p.ktiles = (Kg + WF_BM - 1) / WF_BM;
  p.ntiles = (p.RSC + WF_BN - 1) / WF_BN;
  p.chunks = (p.M + WF_CHUNK * WF_BK - 1) / (WF_CHUNK * WF_BK);
  p.magicHoWo = ((1ULL << 47) / ((unsigned long long)Ho * Wo)) + 1;
  p.magicWo = ((1ULL << 47) / (unsigned long long)Wo) + 1;
  auto gwf = at::zeros({(int64_t)Kt, p.RSC}, x.options());
  p.gw = gwf.data_ptr<float>();
  dim3 grid(p.ktiles * p.ntiles, p.chunks, groups);
  hipLaunchKernelGGL(conv_wgrad_fp32_kernel, grid, dim3(256), 0, cur_stream(),
                     p);
  // [Kt, RSC] rows are already [Kt][R][S][Cg] channels_last order
  return gwf.reshape({Kt, (int64_t)R, (int64_t)S, Cg})
      .permute({0, 3, 1, 2})
      .contiguous(at::MemoryFormat::ChannelsLast);
}
