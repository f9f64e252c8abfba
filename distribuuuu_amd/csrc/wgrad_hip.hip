#include "hip/hip_runtime.h"
// Weight-gradient implicit GEMM: gw[k][r][s][c] =
// sum_m gy[m][k] * patch(x)[m][r][s][c], reduction over all output pixels m.
//
// Both operands are m-major in memory while the MFMA wants the reduction (m)
// per-lane-contiguous. v2: tiles are staged in their NATURAL [m][col] order
// with conflict-free 16-byte LDS writes into [32 m][16 col] subtiles, and the
// MFMA fragments are read TRANSPOSED with ds_read_b64_tr_b16 (hardware
// 4x4-word transpose within 16-lane groups; semantics verified by
// tools/probes/tr_probe.hip on gfx950: out-lane o = q*4+e of a 16-lane group
// receives element e of the 8-byte words addressed by source lanes
// q, q+4, q+8, q+12). The v1 element-scatter staging was 8-way bank-conflict
// bound (SQ_LDS_BANK_CONFLICT = 83% of LDS cycles).
//
// Split-m: each block owns a pixel chunk and atomically accumulates its fp32
// partial tile into gw_accum[Kt][R*S*Cg]; a final cast kernel emits bf16.
#include "common_hip.h"

typedef __attribute__((ext_vector_type(4))) float f32x4w;
typedef __bf16 bf16x8w __attribute__((ext_vector_type(8)));

namespace {

constexpr int WBM = 64;   // k rows per tile
constexpr int WBN = 64;   // rsc cols per tile
constexpr int WBK = 64;   // pixels per k-step
constexpr int CHUNK_STEPS = 64;  // k-steps per block chunk (4096 pixels)
// [32 m][16 col] subtile: 32 rows x 32 B = 1024 B, padded to 1056 B so the
// 8-lane b128 write groups land on distinct banks (264 dwords % 32 = 8).
constexpr int SUBT = 1056;
constexpr int TILE_BYTES = 8 * SUBT;  // 2 m-subtiles x 4 col-subtiles

struct WgradParams {
  const __hip_bfloat16* x;   // [N,H,W,Ct]
  const __hip_bfloat16* gy;  // [N,Ho,Wo,Kt]
  float* acc;                // [Kt, R*S*Cg] zeroed
  int N, H, W, Ct, Kt;
  int R, S, Cg, Kg;
  int sh, sw, ph, pw, dh, dw;
  int Ho, Wo;
  int M, RSC;
  int ktiles, ntiles, chunks;
};

// LDS byte offset of element (m, col) in a [64 m][64 col] tile stored as
// [m/32][col/16] subtiles of [32][16].
DEV_INLINE int lds_off(int m, int col) {
  return ((m >> 5) * 4 + (col >> 4)) * SUBT + (m & 31) * 32 + (col & 15) * 2;
}

// per-lane tr_b16 source address for a fragment whose out-lane l wants
// column (l&15) of a 16-col subtile and m-rows (l>>4)*8 + jj*4 + (0..3):
// source lane s=l&15 contributes the word at row (l>>4)*8 + (s>>2) + jj*4,
// byte (s&3)*8 of the subtile.
DEV_INLINE int tr_addr(int subtile_base, int lane, int jj) {
  const int s = lane & 15;
  const int g = lane >> 4;
  return subtile_base + (g * 8 + (s >> 2) + jj * 4) * 32 + (s & 3) * 8;
}

DEV_INLINE bf16x8w tr_read_frag(unsigned lds_base, int subtile_base,
                                int lane) {
  union {
    uint2 h[2];
    bf16x8w v;
  } u;
  const unsigned a0 = lds_base + tr_addr(subtile_base, lane, 0);
  const unsigned a1 = lds_base + tr_addr(subtile_base, lane, 1);
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=&v"(u.h[0]), "=&v"(u.h[1])
      : "v"(a0), "v"(a1)
      : "memory");
  __builtin_amdgcn_sched_barrier(0);  // rule 18: fence MFMA below the wait
  return u.v;
}

__global__ __launch_bounds__(256) void conv_wgrad_kernel(WgradParams p) {
  const int g = blockIdx.z;
  const int ktile = blockIdx.x % p.ktiles;
  const int ntile = blockIdx.x / p.ktiles;
  const int chunk = blockIdx.y;

  // ONE shared object (a second one forces vmcnt(0) drains before ds_reads)
  __shared__ __align__(16) char smem[2 * 2 * TILE_BYTES];
  auto ldsA = [&](int buf) -> char* { return smem + buf * 2 * TILE_BYTES; };
  auto ldsB = [&](int buf) -> char* {
    return smem + buf * 2 * TILE_BYTES + TILE_BYTES;
  };
  const unsigned smem_base = (unsigned)(unsigned long long)&smem[0];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int il = lane & 15, kq = lane >> 4;

  // staging: thread t loads 8 contiguous cols at pixel m_local:
  //   m_local = t/8 (+32), col8 = (t%8)*8
  const int sml = tid >> 3;        // 0..31
  const int scol8 = (tid & 7) << 3;
  const int SCg = p.S * p.Cg;
  const int rsc = ntile * WBN + scol8;
  const bool rsc_ok = rsc < p.RSC;
  const int br = rsc_ok ? rsc / SCg : 0;
  const int brem = rsc - br * SCg;
  const int bs = brem / p.Cg;
  const int bc = brem - bs * p.Cg;
  const int kcol = ktile * WBM + scol8;
  const bool k_ok = kcol < p.Kg;
  const int wroff = lds_off(sml, scol8);  // same for both halves' subtile rows
  const int wroff2 = lds_off(sml + 32, scol8);

  const int m0 = chunk * (CHUNK_STEPS * WBK);
  const int HoWo = p.Ho * p.Wo;
  const int ksteps = min(CHUNK_STEPS, (int)((p.M - m0 + WBK - 1) / WBK));

  uint4 regA[2], regB[2];
  auto stage_load = [&](int ks) {
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int m = m0 + ks * WBK + sml + half * 32;
      const bool m_ok = m < p.M;
      const int mm = m_ok ? m : 0;
      const int n = mm / HoWo;
      const int rem = mm - n * HoWo;
      const int ho = rem / p.Wo, wo = rem - (rem / p.Wo) * p.Wo;
      if (m_ok && k_ok) {
        regA[half] = *reinterpret_cast<const uint4*>(
            p.gy + ((int64_t)mm * p.Kt) + g * p.Kg + kcol);
      } else {
        regA[half] = uint4{0, 0, 0, 0};
      }
      const int h = ho * p.sh - p.ph + br * p.dh;
      const int w_ = wo * p.sw - p.pw + bs * p.dw;
      if (m_ok && rsc_ok && h >= 0 && h < p.H && w_ >= 0 && w_ < p.W) {
        regB[half] = *reinterpret_cast<const uint4*>(
            p.x + (((int64_t)n * p.H + h) * p.W + w_) * p.Ct + g * p.Cg + bc);
      } else {
        regB[half] = uint4{0, 0, 0, 0};
      }
    }
  };
  auto stage_write = [&](int buf) {
    *reinterpret_cast<uint4*>(ldsA(buf) + wroff) = regA[0];
    *reinterpret_cast<uint4*>(ldsA(buf) + wroff2) = regA[1];
    *reinterpret_cast<uint4*>(ldsB(buf) + wroff) = regB[0];
    *reinterpret_cast<uint4*>(ldsB(buf) + wroff2) = regB[1];
  };

  f32x4w accv[2][2] = {};

  stage_load(0);
  stage_write(0);
  __syncthreads();
  if (ksteps > 1) stage_load(1);

  int cur = 0;
  for (int ks = 0; ks < ksteps; ++ks) {
    const unsigned abase = smem_base + (unsigned)(cur * 2 * TILE_BYTES);
    const unsigned bbase = abase + TILE_BYTES;
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {  // two 32-deep m sub-steps
      bf16x8w afrag[2], bfrag[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        afrag[mi] =
            tr_read_frag(abase, (mc * 4 + (wm * 2 + mi)) * SUBT, lane);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        bfrag[ni] =
            tr_read_frag(bbase, (mc * 4 + (wn * 2 + ni)) * SUBT, lane);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          accv[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], accv[mi][ni], 0, 0, 0);
    }
    __syncthreads();
    if (ks + 1 < ksteps) {
      stage_write(cur ^ 1);
      if (ks + 2 < ksteps) stage_load(ks + 2);
      __syncthreads();
    }
    cur ^= 1;
  }

  // epilogue: fp32 atomic accumulate (D: col=lane&15, row=(lane>>4)*4+rr)
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int k = ktile * WBM + wm * 32 + mi * 16 + kq * 4 + rr;
      if (k >= p.Kg) continue;
      const int64_t rowbase = (int64_t)(g * p.Kg + k) * p.RSC;
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int col = ntile * WBN + wn * 32 + ni * 16 + il;
        if (col < p.RSC) atomicAdd(&p.acc[rowbase + col], accv[mi][ni][rr]);
      }
    }
  }
}

__global__ void cast_acc_kernel(const float* __restrict__ acc,
                                __hip_bfloat16* __restrict__ gw,
                                int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x)
    gw[i] = from_f32<__hip_bfloat16>(acc[i]);
}

}  // namespace

at::Tensor conv2d_wgrad(at::Tensor gy, at::Tensor x, int64_t R, int64_t S,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                        int64_t dh, int64_t dw, int64_t groups) {
  CHECK_GPU(gy);
  TORCH_CHECK(gy.scalar_type() == at::kBFloat16, "wgrad: bf16 only");
  check_nhwc(gy, "gy");
  check_nhwc(x, "x");
  const int N = x.size(0), Ct = x.size(1), H = x.size(2), W = x.size(3);
  const int Kt = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  const int Cg = Ct / groups, Kg = Kt / groups;
  TORCH_CHECK(Cg % 8 == 0 && Kg % 8 == 0, "wgrad: Cg/Kg must be multiples of 8");
  WgradParams p;
  p.x = (const __hip_bfloat16*)x.data_ptr();
  p.gy = (const __hip_bfloat16*)gy.data_ptr();
  p.N = N; p.H = H; p.W = W; p.Ct = Ct; p.Kt = Kt;
  p.R = R; p.S = S; p.Cg = Cg; p.Kg = Kg;
  p.sh = sh; p.sw = sw; p.ph = ph; p.pw = pw; p.dh = dh; p.dw = dw;
  p.Ho = Ho; p.Wo = Wo;
  p.M = N * Ho * Wo;
  p.RSC = R * S * Cg;
  p.ktiles = (Kg + WBM - 1) / WBM;
  p.ntiles = (p.RSC + WBN - 1) / WBN;
  p.chunks = (p.M + CHUNK_STEPS * WBK - 1) / (CHUNK_STEPS * WBK);

  auto accbuf = at::empty({(int64_t)Kt, p.RSC}, x.options().dtype(at::kFloat));
  accbuf.zero_();
  p.acc = accbuf.data_ptr<float>();
  dim3 grid(p.ktiles * p.ntiles, p.chunks, groups);
  hipLaunchKernelGGL(conv_wgrad_kernel, grid, dim3(256), 0, cur_stream(), p);

  auto gw = at::empty({Kt, Cg, (int64_t)R, (int64_t)S},
                      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const int64_t total = (int64_t)Kt * p.RSC;
  hipLaunchKernelGGL(cast_acc_kernel, dim3(grid_1d(total, 256)), dim3(256), 0,
                     cur_stream(), p.acc, (__hip_bfloat16*)gw.data_ptr(),
                     total);
  return gw;
}
