// Implicit-GEMM convolution for gfx950: NHWC bf16, MFMA
// 16x16x32, LDS-staged double-buffered tiles, fp32 accumulation.
//
// GEMM view of conv fwd: C[M, Kg] = A[M, R*S*Cg] @ B[Kg, R*S*Cg]^T per group,
// where M = N*Ho*Wo output pixels, A rows are gathered input patches (NHWC
// makes each (r, s-span) contiguous in memory) and B is the weight tensor in
// its [K][R][S][C] channels_last layout (already reduction-contiguous).
//
// The same kernel computes dgrad: conv_fwd(dilate(gy), flipT(w)) — the Python
// wrapper materializes the transformed weight [C,R,S,K] and zero-dilated gy,
// and the kernel writes into a larger (HoA, WoA)-strided output region to
// honor output_padding. Linear layers are the R=S=1, H=W=1 case.
//
// Tile: BM=128 x BN=128 x BK=32, 4 waves (2x2), each wave 64x64 via 4x4
// fragments of mfma_f32_16x16x32_bf16. Register-staged global->LDS with
// boundary predication (zero-fill); one-step software pipeline (T14 G15:
// loads for step k+1 issued before compute of step k, LDS write after the
// barrier).
#include "common.h"

// v2 minimum reduction depth R*S*Cg (DISTRIBUUUU_V2_MINRSC, default 512):
// below this the 3-slot glds ring has too few k-steps to spin up.
static int v2_minrsc() {
  static const int v = []() {
    const char* e = getenv("DISTRIBUUUU_V2_MINRSC");
    return e ? atoi(e) : 512;
  }();
  return v;
}

// separate depth floor for the scatter/accumulate fwd_into route (proj and
// residual-fork dgrads) so it can be swept independently of the main
// forward dispatch
static int v2_into_minrsc() {
  static const int v = []() {
    const char* e = getenv("DISTRIBUUUU_V2INTO_MINRSC");
    return e ? atoi(e) : v2_minrsc();
  }();
  return v;
}


typedef __attribute__((ext_vector_type(8))) short short8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));

namespace {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int LDS_K = BK + 8;  // +8 bf16 pad: break 64B-row bank alignment

struct ConvParams {
  const __hip_bfloat16* x;  // [N,H,W,Ct] NHWC
  const __hip_bfloat16* w;  // [Kt,R,S,Cg]
  __hip_bfloat16* y;        // [N,HoA,WoA,Kt] NHWC
  int N, H, W, Ct;          // Ct = total input channels
  int Kt;                   // total output channels
  int R, S, Cg, Kg;         // per-group
  int sh, sw, ph, pw, dh, dw;
  int Ho, Wo;               // logical output (M = N*Ho*Wo)
  int HoA, WoA;             // allocated output strides (>= Ho,Wo)
  int osh, osw;             // output position multipliers (dgrad-s2 scatter)
  int oh0, ow0;             // output position base offsets (parity classes)
  int M, nspan, ksteps;     // nspan = ceil(S*Cg/BK), ksteps = R*nspan
  int tiles_m;              // for XCD swizzle
  float* part;              // EMIT: [tiles_m*2, 2*Kt] BN sum/sumsq partials
  // EMODE 2 (dgrad -> BN-backward stats): this conv's output IS the
  // consuming BatchNorm's upstream grad gy; the epilogue also reads the
  // BN's input x at the SAME positions and emits [sum(g), sum(g*x)]
  // partials with g = act'(x*scale+shift) * gy — the standalone
  // bn_bwd_reduce pass over the whole tensor disappears.
  const __hip_bfloat16* bnx;
  const float* bnscale;
  const float* bnshift;
  int bnact;                // 0 none, 1 relu
};

// EMODE: 0 plain, 1 forward BN sum/sumsq partials, 2 backward masked stats.
// ACC: epilogue accumulates into y (residual-fork dgrad into gres).
template <int EMODE, bool ACC = false>
__global__ __launch_bounds__(256) void conv_igemm_fwd_kernel(ConvParams p) {
  static_assert(!(ACC && EMODE != 0), "ACC only with plain epilogue");
  const int g = blockIdx.z;
  // XCD-aware swizzle over m-tiles (T1; bijective form)
  int tile_m = blockIdx.x, tile_n = blockIdx.y;
  {
    const int nwg = p.tiles_m;
    const int q = nwg / 8, r8 = nwg % 8;
    const int xcd = tile_m % 8, idx = tile_m / 8;
    tile_m = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q) + idx;
  }

  __shared__ __hip_bfloat16 ldsA[2][BM][LDS_K];
  __shared__ __hip_bfloat16 ldsB[2][BN][LDS_K];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;  // 2x2 wave grid
  const int il = lane & 15, kg = lane >> 4;

  // ---- per-thread staging coordinates (fixed across k-steps) ----
  // Each thread stages 2 rows of A and 2 rows of B per k-step (rows t/4 and
  // t/4+64), 8 contiguous elements each at span offset (t%4)*8.
  const int srow = tid >> 2;           // 0..63
  const int scol8 = (tid & 3) << 3;    // 0,8,16,24
  int an[2], ahbase[2], awbase[2];
  bool arow_ok[2];
#pragma unroll
  for (int pss = 0; pss < 2; ++pss) {
    const int m = tile_m * BM + srow + pss * 64;
    arow_ok[pss] = m < p.M;
    const int mm = arow_ok[pss] ? m : 0;
    const int n = mm / (p.Ho * p.Wo);
    const int rem = mm % (p.Ho * p.Wo);
    an[pss] = n;
    ahbase[pss] = (rem / p.Wo) * p.sh - p.ph;
    awbase[pss] = (rem % p.Wo) * p.sw - p.pw;
  }
  const int SCg = p.S * p.Cg;
  int bk[2];
  bool brow_ok[2];
#pragma unroll
  for (int pss = 0; pss < 2; ++pss) {
    const int k = tile_n * BN + srow + pss * 64;
    brow_ok[pss] = k < p.Kg;
    bk[pss] = g * p.Kg + (brow_ok[pss] ? k : 0);
  }

  // ---- staging helpers -----------------------------------------------------
  // loads for k-step ks into registers
  uint4 regA[2], regB[2];
  auto stage_load = [&](int ks) {
    const int r = ks / p.nspan;
    const int span0 = (ks % p.nspan) * BK;
    const int span = span0 + scol8;
    const bool span_ok = span < SCg;
    const int s = span_ok ? span / p.Cg : 0;
    const int c = span - s * p.Cg;
#pragma unroll
    for (int pss = 0; pss < 2; ++pss) {
      // A: x[n, hbase + r*dh, wbase + s*dw, gCg + c .. +8]
      const int h = ahbase[pss] + r * p.dh;
      const int w_ = awbase[pss] + s * p.dw;
      const bool ok = arow_ok[pss] && span_ok && h >= 0 && h < p.H &&
                      w_ >= 0 && w_ < p.W;
      regA[pss] = ok ? *reinterpret_cast<const uint4*>(
                           p.x + (((int64_t)an[pss] * p.H + h) * p.W + w_) *
                                     p.Ct +
                           g * p.Cg + c)
                     : uint4{0, 0, 0, 0};
      // B: w[k, r, span .. +8]
      const bool okb = brow_ok[pss] && span_ok;
      regB[pss] = okb ? *reinterpret_cast<const uint4*>(
                            p.w + ((int64_t)bk[pss] * p.R + r) * SCg + span)
                      : uint4{0, 0, 0, 0};
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int pss = 0; pss < 2; ++pss) {
      *reinterpret_cast<uint4*>(&ldsA[buf][srow + pss * 64][scol8]) = regA[pss];
      *reinterpret_cast<uint4*>(&ldsB[buf][srow + pss * 64][scol8]) = regB[pss];
    }
  };

  // ---- main loop -----------------------------------------------------------
  f32x4 acc[4][4] = {};
  stage_load(0);
  stage_write(0);
  __syncthreads();
  if (p.ksteps > 1) stage_load(1);

  int cur = 0;
  for (int ks = 0; ks < p.ksteps; ++ks) {
    // fragments: A rows wm*64 + mi*16 + il, k = kg*8; B rows wn*64 + ni*16+il
    bf16x8 afrag[4], bfrag[4];
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      afrag[mi] = *reinterpret_cast<const bf16x8*>(
          &ldsA[cur][wm * 64 + mi * 16 + il][kg * 8]);
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      bfrag[ni] = *reinterpret_cast<const bf16x8*>(
          &ldsB[cur][wn * 64 + ni * 16 + il][kg * 8]);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    __syncthreads();
    if (ks + 1 < p.ksteps) {
      stage_write(cur ^ 1);
      if (ks + 2 < p.ksteps) stage_load(ks + 2);
      __syncthreads();
    }
    cur ^= 1;
  }

  // ---- epilogue: LDS-restage then 16-wide bf16 stores ----------------------
  // Scalar 2-byte stores (64/lane) are store-issue-bound (T21 diagnosis);
  // instead each wave round-trips its 64x64 tile through a private fp32 LDS
  // slab 16 rows at a time, then stores 16 consecutive channels per lane.
  __syncthreads();  // everyone done reading the final A/B tiles
  float* slab = reinterpret_cast<float*>(&ldsA[0][0][0]) + wid * (16 * 68);
  const int HoWo = p.Ho * p.Wo;
  const int er = lane >> 2;          // 0..15 row within the 16-row stripe
  const int ec = (lane & 3) << 4;    // 0,16,32,48 col
  float ps = 0.f, pq = 0.f;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    // scatter this mi-stripe's fragments to the slab (conflict-free b32)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr)
        slab[(kg * 4 + rr) * 68 + ni * 16 + il] = acc[mi][ni][rr];
    __builtin_amdgcn_wave_barrier();
    const int m = tile_m * BM + wm * 64 + mi * 16 + er;
    if (m < p.M) {
      const int n = m / HoWo;
      const int rem = m - n * HoWo;
      const int64_t obase =
          (((int64_t)n * p.HoA + (rem / p.Wo) * p.osh + p.oh0) * p.WoA +
           (rem % p.Wo) * p.osw + p.ow0) * p.Kt + g * p.Kg;
      const int k0 = tile_n * BN + wn * 64 + ec;
      union {
        __hip_bfloat16 b[16];
        uint4 q[2];
      } u;
#pragma unroll
      for (int j = 0; j < 16; ++j)
        u.b[j] = from_f32<__hip_bfloat16>(slab[er * 68 + ec + j]);
      if (ACC) {
        if (k0 + 16 <= p.Kg) {
          union {
            __hip_bfloat16 b[16];
            uint4 q[2];
          } old;
          old.q[0] = *reinterpret_cast<const uint4*>(&p.y[obase + k0]);
          old.q[1] = *reinterpret_cast<const uint4*>(&p.y[obase + k0 + 8]);
#pragma unroll
          for (int j = 0; j < 16; ++j)
            u.b[j] = from_f32<__hip_bfloat16>(to_f32(u.b[j]) +
                                              to_f32(old.b[j]));
          *reinterpret_cast<uint4*>(&p.y[obase + k0]) = u.q[0];
          *reinterpret_cast<uint4*>(&p.y[obase + k0 + 8]) = u.q[1];
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j)
            if (k0 + j < p.Kg)
              p.y[obase + k0 + j] = from_f32<__hip_bfloat16>(
                  to_f32(u.b[j]) + to_f32(p.y[obase + k0 + j]));
        }
      } else if (k0 + 16 <= p.Kg) {
        *reinterpret_cast<uint4*>(&p.y[obase + k0]) = u.q[0];
        *reinterpret_cast<uint4*>(&p.y[obase + k0 + 8]) = u.q[1];
      } else {
#pragma unroll
        for (int j = 0; j < 16; ++j)
          if (k0 + j < p.Kg) p.y[obase + k0 + j] = u.b[j];
      }
    }
    if (EMODE == 1) {
      const int base = tile_m * BM + wm * 64 + mi * 16;
      bn_partial_col_accum(slab, ps, pq,
                           (int)min((int64_t)16, (int64_t)p.M - base), lane);
    }
    if (EMODE == 2) {
      // load x at the SAME positions, build g = act'(x*scale+shift)*gy,
      // run the two-pass slab column accumulation (g, then g*x)
      union {
        __hip_bfloat16 b[16];
        uint4 q[2];
      } xv;
      const int k0 = tile_n * BN + wn * 64 + ec;
      bool valid = m < p.M;
      if (valid) {
        const int n = m / HoWo;
        const int rem = m - n * HoWo;
        const int64_t obase =
            (((int64_t)n * p.HoA + (rem / p.Wo) * p.osh + p.oh0) * p.WoA +
             (rem % p.Wo) * p.osw + p.ow0) * p.Kt + g * p.Kg;
        if (k0 + 16 <= p.Kg) {
          xv.q[0] = *reinterpret_cast<const uint4*>(&p.bnx[obase + k0]);
          xv.q[1] = *reinterpret_cast<const uint4*>(&p.bnx[obase + k0 + 8]);
        } else {
          xv.q[0] = uint4{0, 0, 0, 0};
          xv.q[1] = uint4{0, 0, 0, 0};
#pragma unroll
          for (int j = 0; j < 16; ++j)
            if (k0 + j < p.Kg) xv.b[j] = p.bnx[obase + k0 + j];
        }
      }
      __builtin_amdgcn_wave_barrier();  // everyone done with the y stripe
      // pass 1: g   (invalid rows contribute zeros)
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        float gj = 0.f;
        if (valid) {
          const int cc = g * p.Kg + k0 + j;
          const float yj = to_f32(from_f32<__hip_bfloat16>(
              slab[er * 68 + ec + j]));  // the ROUNDED stored gy
          gj = yj;
          if (p.bnact == 1 &&
              to_f32(xv.b[j]) * p.bnscale[cc] + p.bnshift[cc] <= 0.f)
            gj = 0.f;
        }
        slab[er * 68 + ec + j] = gj;
      }
      __builtin_amdgcn_wave_barrier();
#pragma unroll
      for (int r = 0; r < 16; ++r) ps += slab[r * 68 + lane];
      __builtin_amdgcn_wave_barrier();
      // pass 2: g*x (read own g back, multiply by x)
#pragma unroll
      for (int j = 0; j < 16; ++j)
        slab[er * 68 + ec + j] *= valid ? to_f32(xv.b[j]) : 0.f;
      __builtin_amdgcn_wave_barrier();
#pragma unroll
      for (int r = 0; r < 16; ++r) pq += slab[r * 68 + lane];
    }
    __builtin_amdgcn_wave_barrier();
  }
  if (EMODE == 1 || EMODE == 2) {
    // cross-wave (wm) combine via LDS halves the partial rows
    float* xarea = reinterpret_cast<float*>(&ldsB[0][0][0]);
    if (wm == 1) {
      xarea[(wn * 2 + 0) * 64 + lane] = ps;
      xarea[(wn * 2 + 1) * 64 + lane] = pq;
    }
    __syncthreads();
    if (wm == 0) {
      ps += xarea[(wn * 2 + 0) * 64 + lane];
      pq += xarea[(wn * 2 + 1) * 64 + lane];
      const int kbase = tile_n * BN + wn * 64;
      bn_partial_store(p.part,
                       (int64_t)tile_m * 2 * p.Kt + g * p.Kg + kbase,
                       p.Kt, lane, min(64, p.Kg - kbase), ps, pq);
    }
  }
}


// ---------------------------------------------------------------------------
// Small 1x1 GEMM: y[M, Kt] = x[M, C] @ w[Kt, C]^T for the shallow shapes
// (C <= 256, Kt <= 512) where the 128x128x32 tile kernel spends most of its
// time on pipeline fill/drain (2-8 k-steps) and half-empty B tiles (K = 64).
// Both MFMA operands load DIRECT from global: w (<= 256 KB) is L2-resident
// across the whole launch, and each x row is L1-reused by the wave's four
// B fragments. Each wave owns 64 m-rows and loops over 64-wide k chunks.
// ---------------------------------------------------------------------------
struct SmallGemmParams {
  const __hip_bfloat16* x;
  const __hip_bfloat16* w;
  __hip_bfloat16* y;
  int C, Kt;
  int64_t M;
  float* part;  // EMIT: [ceil(M/256)*4, 2*Kt] BN partials
};

template <bool EMIT>
__global__ __launch_bounds__(256) void conv_gemm_small_kernel(
    SmallGemmParams p) {
  __shared__ float slab4[4][16 * 68];
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int il = lane & 15, kq = lane >> 4;
  const int64_t mbase = (int64_t)blockIdx.x * 256 + wid * 64;
  const __hip_bfloat16* xrow[4];
  bool m_ok[4];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    const int64_t m = mbase + mi * 16 + il;
    m_ok[mi] = m < p.M;
    xrow[mi] = p.x + (m_ok[mi] ? m : 0) * p.C + kq * 8;
  }
  float* slab = slab4[wid];
  const int er = lane >> 2, ec = (lane & 3) << 4;
  for (int k0 = 0; k0 < p.Kt; k0 += 64) {
    const __hip_bfloat16* wrow[4];
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      wrow[ni] = p.w + (int64_t)(k0 + ni * 16 + il) * p.C + kq * 8;
    f32x4 acc[4][4] = {};
    for (int c0 = 0; c0 < p.C; c0 += 32) {
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
        afrag[mi] = m_ok[mi] ? *reinterpret_cast<const bf16x8*>(xrow[mi] + c0)
                             : bf16x8{};
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bfrag[ni] = *reinterpret_cast<const bf16x8*>(wrow[ni] + c0);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
    // per-wave LDS-restage epilogue (16-wide bf16 stores; T21)
    float ps = 0.f, pq = 0.f;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr)
          slab[(kq * 4 + rr) * 68 + ni * 16 + il] = acc[mi][ni][rr];
      __builtin_amdgcn_wave_barrier();
      const int64_t m = mbase + mi * 16 + er;
      if (m < p.M) {
        union {
          __hip_bfloat16 b[16];
          uint4 q[2];
        } u;
#pragma unroll
        for (int j = 0; j < 16; ++j)
          u.b[j] = from_f32<__hip_bfloat16>(slab[er * 68 + ec + j]);
        __hip_bfloat16* yp = p.y + m * p.Kt + k0 + ec;
        *reinterpret_cast<uint4*>(yp) = u.q[0];
        *reinterpret_cast<uint4*>(yp + 8) = u.q[1];
      }
      if (EMIT)
        bn_partial_col_accum(
            slab, ps, pq,
            (int)min((int64_t)16, p.M - (mbase + mi * 16)), lane);
      __builtin_amdgcn_wave_barrier();
    }
    if (EMIT) {
      __shared__ float xarea[3][2][64];
      if (wid > 0) {
        xarea[wid - 1][0][lane] = ps;
        xarea[wid - 1][1][lane] = pq;
      }
      __syncthreads();
      if (wid == 0) {
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          ps += xarea[j][0][lane];
          pq += xarea[j][1][lane];
        }
        bn_partial_store(p.part, (int64_t)blockIdx.x * 2 * p.Kt + k0,
                         p.Kt, lane, min(64, p.Kt - k0), ps, pq);
      }
      __syncthreads();  // xarea reused next k0 chunk
    }
  }
}


// C <= 64 variant: the whole per-wave A stripe (64 rows x C) is register-
// cached across k chunks, so x is read exactly once regardless of Kt (the
// generic kernel re-reads A per 64-wide k chunk, which loses for Kt > 64).
// min 3 blocks/CU (3 waves/SIMD -> <=170 VGPRs): the EMIT variant's natural
// allocation is 180, which silently dropped smallc to 2 waves/SIMD
template <int CS, bool EMIT>  // CS = C / 32
__global__ __launch_bounds__(256, 3) void conv_gemm_smallc_kernel(
    SmallGemmParams p) {
  __shared__ float slab4[4][16 * 68];
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int il = lane & 15, kq = lane >> 4;
  const int64_t mbase = (int64_t)blockIdx.x * 256 + wid * 64;
  float* slab = slab4[wid];
  const int er = lane >> 2, ec = (lane & 3) << 4;
  bf16x8 afr[CS][4];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    const int64_t m = mbase + mi * 16 + il;
    const __hip_bfloat16* xr = p.x + (m < p.M ? m : 0) * p.C + kq * 8;
#pragma unroll
    for (int cs = 0; cs < CS; ++cs)
      afr[cs][mi] = m < p.M ? *reinterpret_cast<const bf16x8*>(xr + cs * 32)
                            : bf16x8{};
  }
  for (int k0 = 0; k0 < p.Kt; k0 += 64) {
    const __hip_bfloat16* wrow[4];
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      wrow[ni] = p.w + (int64_t)(k0 + ni * 16 + il) * p.C + kq * 8;
    f32x4 acc[4][4] = {};
#pragma unroll
    for (int cs = 0; cs < CS; ++cs) {
      bf16x8 bfrag[4];
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bfrag[ni] = *reinterpret_cast<const bf16x8*>(wrow[ni] + cs * 32);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[cs][mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
    float ps = 0.f, pq = 0.f;
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr)
          slab[(kq * 4 + rr) * 68 + ni * 16 + il] = acc[mi][ni][rr];
      __builtin_amdgcn_wave_barrier();
      const int64_t m = mbase + mi * 16 + er;
      if (m < p.M) {
        union {
          __hip_bfloat16 b[16];
          uint4 q[2];
        } u;
#pragma unroll
        for (int j = 0; j < 16; ++j)
          u.b[j] = from_f32<__hip_bfloat16>(slab[er * 68 + ec + j]);
        __hip_bfloat16* yp = p.y + m * p.Kt + k0 + ec;
        *reinterpret_cast<uint4*>(yp) = u.q[0];
        *reinterpret_cast<uint4*>(yp + 8) = u.q[1];
      }
      if (EMIT)
        bn_partial_col_accum(
            slab, ps, pq,
            (int)min((int64_t)16, p.M - (mbase + mi * 16)), lane);
      __builtin_amdgcn_wave_barrier();
    }
    if (EMIT) {
      __shared__ float xarea[3][2][64];
      if (wid > 0) {
        xarea[wid - 1][0][lane] = ps;
        xarea[wid - 1][1][lane] = pq;
      }
      __syncthreads();
      if (wid == 0) {
#pragma unroll
        for (int j = 0; j < 3; ++j) {
          ps += xarea[j][0][lane];
          pq += xarea[j][1][lane];
        }
        bn_partial_store(p.part, (int64_t)blockIdx.x * 2 * p.Kt + k0,
                         p.Kt, lane, min(64, p.Kt - k0), ps, pq);
      }
      __syncthreads();  // xarea reused next k0 chunk
    }
  }
}


// ---------------------------------------------------------------------------
// K=64 small-C patch conv (stem 7x7 s2 C=8, 3x3 64->64): one 64-col B tile
// means half of the 128-wide tile kernel idles. Here both operands load
// direct from global with full per-lane patch predication: each lane's 16 B
// A fragment is one (r, s, c-pack) patch element (magic-div span
// decomposition), B is the span-padded [64, RSCp] weight (L2-resident).
// Block = 4 waves x 64 m-rows, each wave owns all 64 output channels.
// ---------------------------------------------------------------------------
struct SmallConvParams {
  const __hip_bfloat16* x;  // [N,H,W,C]
  const __hip_bfloat16* w;  // [64, RSCp] span-padded flat
  const __hip_bfloat16* zbuf;  // 16 zero bytes (clamp target for OOB loads)
  __hip_bfloat16* y;        // [N,Ho,Wo,64]
  int C, H, W;
  int R, S, SC;
  int sh, sw, ph, pw;
  int Ho, Wo;
  int RSC, RSCp;
  int64_t M;
  unsigned long long magicHoWo, magicWo, magicSC, magicC;
  float* part;  // EMIT: [ceil(M/256)*4, 2*64] BN partials
};

DEV_INLINE int magic_div2(int m, unsigned long long magic) {
  return (int)(((unsigned long long)(unsigned)m * magic) >> 47);
}

template <bool EMIT>
__global__ __launch_bounds__(256) void conv_smallk_kernel(SmallConvParams p) {
  __shared__ float slab4[4][16 * 68];
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int il = lane & 15, kq = lane >> 4;
  const int64_t mbase = (int64_t)blockIdx.x * 256 + wid * 64;
  int pn[4], prow[4], pcol[4];
  bool mok[4];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    const int64_t m = mbase + mi * 16 + il;
    mok[mi] = m < p.M;
    const int mm = (int)(mok[mi] ? m : 0);
    const int n = magic_div2(mm, p.magicHoWo);
    const int rem = mm - n * (p.Ho * p.Wo);
    const int ho = magic_div2(rem, p.magicWo);
    const int wo = rem - ho * p.Wo;
    pn[mi] = n;
    prow[mi] = ho * p.sh - p.ph;
    pcol[mi] = wo * p.sw - p.pw;
  }
  const __hip_bfloat16* wr[4];
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
    wr[ni] = p.w + (int64_t)(ni * 16 + il) * p.RSCp + kq * 8;
  f32x4 acc[4][4] = {};
  if (p.C % 32 == 0) {
    // tap-ordered loop (C a multiple of 32): the per-tap bounds check and
    // base address are computed once per (r, s) instead of per k-step —
    // the k-step-ordered form below was VALU-bound on its per-step
    // magic-div + predication chain (~80 VALU per 16 MFMAs).
    const int csteps = p.C / 32;
    for (int r = 0; r < p.R; ++r) {
#pragma unroll 1
      for (int sidx = 0; sidx < p.S; ++sidx) {
        const __hip_bfloat16* ab[4];
        bool ok[4];
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          const int h = prow[mi] + r;
          const int w_ = pcol[mi] + sidx;
          ok[mi] = mok[mi] && h >= 0 && h < p.H && w_ >= 0 && w_ < p.W;
          ab[mi] = p.x +
                   (((int64_t)pn[mi] * p.H + h) * p.W + w_) * p.C + kq * 8;
        }
        const int e0 = (r * p.S + sidx) * p.C;
        for (int cc = 0; cc < csteps; ++cc) {
          bf16x8 afrag[4], bfrag[4];
#pragma unroll
          for (int mi = 0; mi < 4; ++mi)
            afrag[mi] = ok[mi] ? *reinterpret_cast<const bf16x8*>(ab[mi] +
                                                                  cc * 32)
                               : bf16x8{};
#pragma unroll
          for (int ni = 0; ni < 4; ++ni)
            bfrag[ni] =
                *reinterpret_cast<const bf16x8*>(wr[ni] + e0 + cc * 32);
#pragma unroll
          for (int mi = 0; mi < 4; ++mi)
#pragma unroll
            for (int ni = 0; ni < 4; ++ni)
              acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
        }
      }
    }
  } else {
    const int ksteps = p.RSCp / 32;
    for (int t = 0; t < ksteps; ++t) {
      const int e = t * 32 + kq * 8;
      const int r = magic_div2(e, p.magicSC);
      const int rem2 = e - r * p.SC;
      const int sidx = magic_div2(rem2, p.magicC);
      const int c = rem2 - sidx * p.C;
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int h = prow[mi] + r;
        const int w_ = pcol[mi] + sidx;
        const bool ok = mok[mi] && e < p.RSC && h >= 0 && h < p.H &&
                        w_ >= 0 && w_ < p.W;
        afrag[mi] =
            ok ? *reinterpret_cast<const bf16x8*>(
                     p.x + (((int64_t)pn[mi] * p.H + h) * p.W + w_) * p.C + c)
               : bf16x8{};
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        bfrag[ni] = *reinterpret_cast<const bf16x8*>(wr[ni] + t * 32);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], acc[mi][ni], 0, 0, 0);
    }
  }
  float* slab = slab4[wid];
  const int er = lane >> 2, ec = (lane & 3) << 4;
  float ps = 0.f, pq = 0.f;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr)
        slab[(kq * 4 + rr) * 68 + ni * 16 + il] = acc[mi][ni][rr];
    __builtin_amdgcn_wave_barrier();
    const int64_t m = mbase + mi * 16 + er;
    if (m < p.M) {
      union {
        __hip_bfloat16 b[16];
        uint4 q[2];
      } u;
#pragma unroll
      for (int j = 0; j < 16; ++j)
        u.b[j] = from_f32<__hip_bfloat16>(slab[er * 68 + ec + j]);
      __hip_bfloat16* yp = p.y + m * 64 + ec;
      *reinterpret_cast<uint4*>(yp) = u.q[0];
      *reinterpret_cast<uint4*>(yp + 8) = u.q[1];
    }
    if (EMIT)
      bn_partial_col_accum(slab, ps, pq,
                           (int)min((int64_t)16, p.M - (mbase + mi * 16)),
                           lane);
    __builtin_amdgcn_wave_barrier();
  }
  if (EMIT) {
    __shared__ float xarea[3][2][64];
    if (wid > 0) {
      xarea[wid - 1][0][lane] = ps;
      xarea[wid - 1][1][lane] = pq;
    }
    __syncthreads();
    if (wid == 0) {
#pragma unroll
      for (int j = 0; j < 3; ++j) {
        ps += xarea[j][0][lane];
        pq += xarea[j][1][lane];
      }
      bn_partial_store(p.part, (int64_t)blockIdx.x * 128, 64, lane, 64, ps,
                       pq);
    }
  }
}


// Pipelined variant (C % 32 == 0): PMC showed the plain kernel waits on
// memory ~85% of its cycles (SQ busy 12%) — each k-step's 8 loads are
// consumed by the very next MFMAs. This one (a) selects the ADDRESS
// (OOB -> a 16-byte zero buffer) instead of the loaded value, so no
// post-load cndmask forces an early waitcnt, and (b) prefetches step i+1's
// fragments while step i's MFMAs run.
template <int CSTEPS, bool EMIT>  // CSTEPS = C / 32
__global__ __launch_bounds__(256) void conv_smallk_pipe_kernel(
    SmallConvParams p) {
  __shared__ float slab4[4][16 * 68];
  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int il = lane & 15, kq = lane >> 4;
  const int64_t mbase = (int64_t)blockIdx.x * 256 + wid * 64;
  int pn[4], prow[4], pcol[4];
  bool mok[4];
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
    const int64_t m = mbase + mi * 16 + il;
    mok[mi] = m < p.M;
    const int mm = (int)(mok[mi] ? m : 0);
    const int n = magic_div2(mm, p.magicHoWo);
    const int rem = mm - n * (p.Ho * p.Wo);
    const int ho = magic_div2(rem, p.magicWo);
    const int wo = rem - ho * p.Wo;
    pn[mi] = n;
    prow[mi] = ho * p.sh - p.ph;
    pcol[mi] = wo * p.sw - p.pw;
  }
  const __hip_bfloat16* wr[4];
#pragma unroll
  for (int ni = 0; ni < 4; ++ni)
    wr[ni] = p.w + (int64_t)(ni * 16 + il) * p.RSCp + kq * 8;
  const int S = p.S;
  const int steps = p.R * S * CSTEPS;
  int r = 0, si = 0, cc = 0;
  bf16x8 Af[2][4], Bf[2][4];
  auto issue = [&](int buf) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const int h = prow[mi] + r;
      const int w_ = pcol[mi] + si;
      const bool ok = mok[mi] && h >= 0 && h < p.H && w_ >= 0 && w_ < p.W;
      const __hip_bfloat16* ap =
          ok ? p.x + (((int64_t)pn[mi] * p.H + h) * p.W + w_) * p.C +
                   cc * 32 + kq * 8
             : p.zbuf;
      Af[buf][mi] = *reinterpret_cast<const bf16x8*>(ap);
    }
    const int e0 = (r * S + si) * p.C + cc * 32;
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
      Bf[buf][ni] = *reinterpret_cast<const bf16x8*>(wr[ni] + e0);
    ++cc;
    if (cc == CSTEPS) {
      cc = 0;
      ++si;
      if (si == S) {
        si = 0;
        ++r;
      }
    }
  };
  f32x4 acc[4][4] = {};
  // 2x-unrolled so the double-buffer index is a compile-time constant
  // (a runtime `i & 1` index on the fragment arrays spills them to
  // scratch — rule 20 — measured 60x slower)
  issue(0);
  for (int i = 0; i < steps; i += 2) {
    if (i + 1 < steps) issue(1);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
#pragma unroll
      for (int ni = 0; ni < 4; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            Af[0][mi], Bf[0][ni], acc[mi][ni], 0, 0, 0);
    if (i + 1 < steps) {
      if (i + 2 < steps) issue(0);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              Af[1][mi], Bf[1][ni], acc[mi][ni], 0, 0, 0);
    }
  }
  float* slab = slab4[wid];
  const int er = lane >> 2, ec = (lane & 3) << 4;
  float ps = 0.f, pq = 0.f;
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr)
        slab[(kq * 4 + rr) * 68 + ni * 16 + il] = acc[mi][ni][rr];
    __builtin_amdgcn_wave_barrier();
    const int64_t m = mbase + mi * 16 + er;
    if (m < p.M) {
      union {
        __hip_bfloat16 b[16];
        uint4 q[2];
      } u;
#pragma unroll
      for (int j = 0; j < 16; ++j)
        u.b[j] = from_f32<__hip_bfloat16>(slab[er * 68 + ec + j]);
      __hip_bfloat16* yp = p.y + m * 64 + ec;
      *reinterpret_cast<uint4*>(yp) = u.q[0];
      *reinterpret_cast<uint4*>(yp + 8) = u.q[1];
    }
    if (EMIT)
      bn_partial_col_accum(slab, ps, pq,
                           (int)min((int64_t)16, p.M - (mbase + mi * 16)),
                           lane);
    __builtin_amdgcn_wave_barrier();
  }
  if (EMIT) {
    __shared__ float xarea[3][2][64];
    if (wid > 0) {
      xarea[wid - 1][0][lane] = ps;
      xarea[wid - 1][1][lane] = pq;
    }
    __syncthreads();
    if (wid == 0) {
#pragma unroll
      for (int j = 0; j < 3; ++j) {
        ps += xarea[j][0][lane];
        pq += xarea[j][1][lane];
      }
      bn_partial_store(p.part, (int64_t)blockIdx.x * 128, 64, lane, 64, ps,
                       pq);
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------
// fp32 kernel path (conv_fp32.hip)
at::Tensor conv2d_fwd_into_fp32(at::Tensor x, at::Tensor w, at::Tensor y,
                                int64_t Ho, int64_t Wo, int64_t sh, int64_t sw,
                                int64_t ph, int64_t pw, int64_t dh, int64_t dw,
                                int64_t groups, int64_t osh, int64_t osw,
                                int64_t oh0, int64_t ow0,
                                at::Tensor* part_out);

// BN-backward stat emission (EMODE 2) descriptor: the conv being launched is
// a DGRAD whose output is the gy of the BatchNorm that produced this conv's
// logical input; x/scale/shift/act describe that BN.
struct BnBwdEmit {
  const __hip_bfloat16* x;
  const float* scale;
  const float* shift;
  int act;
};

at::Tensor conv2d_fwd_v2_into(at::Tensor x, at::Tensor w, at::Tensor y,
                              int64_t Ho, int64_t Wo, int64_t groups,
                              int64_t osh, int64_t osw, int64_t oh0,
                              int64_t ow0, bool acc = false);

at::Tensor conv2d_fwd_into(at::Tensor x, at::Tensor w, at::Tensor y, int64_t Ho,
                           int64_t Wo, int64_t sh, int64_t sw, int64_t ph,
                           int64_t pw, int64_t dh, int64_t dw, int64_t groups,
                           int64_t osh = 1, int64_t osw = 1, int64_t oh0 = 0,
                           int64_t ow0 = 0, at::Tensor* part_out = nullptr,
                           const BnBwdEmit* bemit = nullptr,
                           bool acc = false) {
  CHECK_GPU(x);
  if (x.scalar_type() == at::kFloat)
    return conv2d_fwd_into_fp32(x, w, y, Ho, Wo, sh, sw, ph, pw, dh, dw,
                                groups, osh, osw, oh0, ow0,
                                bemit ? nullptr : part_out);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "conv2d_fwd: bf16/fp32 only");
  {
    // deep pad-free windows (parity-decomposed / strided dgrads) ride the
    // v2 glds ring with the scatter epilogue instead of the v1 tile kernel
    static const bool v2i_off = []() {
      const char* e = getenv("DISTRIBUUUU_CONV_V2INTO");
      return e && e[0] == '0';
    }();
    const int Cgi = w.size(1), Ri = w.size(2), Si = w.size(3);
    const int Kgi = (int)(w.size(0) / groups);
    if (!v2i_off && part_out == nullptr && bemit == nullptr && ph == 0 &&
        pw == 0 && sh == 1 && sw == 1 && dh == 1 && dw == 1 &&
        (Si * Cgi) % 64 == 0 && Kgi >= 96 &&
        (int64_t)Ri * Si * Cgi >= v2_into_minrsc() && Cgi % 8 == 0 &&
        Ho + Ri - 1 <= x.size(2) && Wo + Si - 1 <= x.size(3))
      return conv2d_fwd_v2_into(x, w, y, Ho, Wo, groups, osh, osw, oh0, ow0,
                                acc);
  }
  check_nhwc(x, "x");
  check_nhwc(w, "w");
  const int N = x.size(0), Ct = x.size(1), H = x.size(2), W = x.size(3);
  const int Kt = w.size(0), Cg = w.size(1), R = w.size(2), S = w.size(3);
  TORCH_CHECK(Ct == Cg * groups, "channel/group mismatch");
  TORCH_CHECK(Cg % 8 == 0, "conv2d_fwd: per-group C must be a multiple of 8");
  const int Kg = Kt / groups;
  ConvParams p;
  p.x = (const __hip_bfloat16*)x.data_ptr();
  p.w = (const __hip_bfloat16*)w.data_ptr();
  p.y = (__hip_bfloat16*)y.data_ptr();
  p.N = N; p.H = H; p.W = W; p.Ct = Ct; p.Kt = Kt;
  p.R = R; p.S = S; p.Cg = Cg; p.Kg = Kg;
  p.sh = sh; p.sw = sw; p.ph = ph; p.pw = pw; p.dh = dh; p.dw = dw;
  p.Ho = Ho; p.Wo = Wo;
  p.HoA = y.size(2); p.WoA = y.size(3);
  p.osh = osh; p.osw = osw;
  p.oh0 = oh0; p.ow0 = ow0;
  p.M = N * Ho * Wo;
  p.nspan = (S * Cg + BK - 1) / BK;
  p.ksteps = R * p.nspan;
  p.tiles_m = (p.M + BM - 1) / BM;
  p.part = nullptr;
  p.bnx = nullptr;
  p.bnscale = nullptr;
  p.bnshift = nullptr;
  p.bnact = 0;
  dim3 grid(p.tiles_m, (Kg + BN - 1) / BN, groups);
  if (bemit != nullptr) {
    *part_out = at::empty({(int64_t)p.tiles_m, (int64_t)2 * Kt},
                          x.options().dtype(at::kFloat));
    p.part = part_out->data_ptr<float>();
    p.bnx = bemit->x;
    p.bnscale = bemit->scale;
    p.bnshift = bemit->shift;
    p.bnact = bemit->act;
    hipLaunchKernelGGL(conv_igemm_fwd_kernel<2>, grid, dim3(256), 0,
                       cur_stream(), p);
  } else if (part_out != nullptr) {
    *part_out = at::empty({(int64_t)p.tiles_m, (int64_t)2 * Kt},
                          x.options().dtype(at::kFloat));
    p.part = part_out->data_ptr<float>();
    hipLaunchKernelGGL(conv_igemm_fwd_kernel<1>, grid, dim3(256), 0,
                       cur_stream(), p);
  } else if (acc) {
    hipLaunchKernelGGL((conv_igemm_fwd_kernel<0, true>), grid, dim3(256), 0,
                       cur_stream(), p);
  } else {
    hipLaunchKernelGGL(conv_igemm_fwd_kernel<0>, grid, dim3(256), 0,
                       cur_stream(), p);
  }
  return y;
}

at::Tensor conv2d_fwd_v2p(at::Tensor x, at::Tensor w, int64_t sh, int64_t sw,
                          int64_t ph, int64_t pw, int64_t dh, int64_t dw,
                          int64_t groups, at::Tensor* part_out,
                          const BnBwdEmit* bemit = nullptr);
at::Tensor conv2d_fwd_v2_flat(at::Tensor x, at::Tensor wp, int64_t Kt_,
                              int64_t Cg_, int64_t R_, int64_t S_, int64_t sh,
                              int64_t sw, int64_t ph, int64_t pw, int64_t dh,
                              int64_t dw, int64_t groups,
                              at::Tensor* part_out, const BnBwdEmit* bemit,
                              at::Tensor* acc_into = nullptr);

static at::Tensor conv2d_fwd_impl(at::Tensor x, at::Tensor w, int64_t sh,
                                  int64_t sw, int64_t ph, int64_t pw,
                                  int64_t dh, int64_t dw, int64_t groups,
                                  at::Tensor* part_out,
                                  const BnBwdEmit* bemit = nullptr) {
  const int N = x.size(0), H = x.size(2), W = x.size(3);
  const int Kt = w.size(0), R = w.size(2), S = w.size(3);
  // v2 (3-slot glds ring) wins when the reduction is deep enough to fill
  // its pipeline; v1 handles shallow-K, small-K and grouped shapes.
  static const bool v2_off = []() {
    const char* e = getenv("DISTRIBUUUU_CONV_V2");
    return e && e[0] == '0';
  }();
  const int C_ = x.size(1);
  // per-group geometry decides the ring gate: deep-enough reduction fills
  // the 3-slot pipeline (RegNetY's 232-wide groups qualify: Kg=232,
  // R*S*Cg=2088)
  const int v2Kg = Kt / (int)groups;
  const int v2Cg = C_ / (int)groups;
  static const int v2_mink = []() {
    const char* e = getenv("DISTRIBUUUU_V2_MINK");
    return e ? atoi(e) : 192;
  }();
  if (!v2_off && v2Kg >= v2_mink &&
      (int64_t)R * S * v2Cg >= v2_minrsc() &&
      v2Cg % 8 == 0 && x.scalar_type() == at::kBFloat16)
    return conv2d_fwd_v2p(x, w, sh, sw, ph, pw, dh, dw, groups, part_out,
                          bemit);
  static const bool small_off = []() {
    const char* e = getenv("DISTRIBUUUU_CONV_SMALL");
    return e && e[0] == '0';
  }();
  if (!small_off && groups == 1 && R == 1 && S == 1 && sh == 1 && sw == 1 &&
      C_ % 32 == 0 && Kt % 64 == 0 && x.scalar_type() == at::kBFloat16 &&
      ((C_ <= 64 && Kt <= 512) || (C_ <= 256 && Kt <= 64))) {
    const int64_t M = (int64_t)N * H * W;
    auto y = at::empty({N, Kt, H, W},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
    SmallGemmParams sp;
    sp.x = (const __hip_bfloat16*)x.data_ptr();
    sp.w = (const __hip_bfloat16*)w.data_ptr();
    sp.y = (__hip_bfloat16*)y.data_ptr();
    sp.C = C_; sp.Kt = Kt; sp.M = M;
    sp.part = nullptr;
    const dim3 sg((int)((M + 255) / 256));
    if (bemit != nullptr) part_out = nullptr;  // small 1x1: no bwd emission
    if (part_out != nullptr) {
      *part_out = at::empty({(int64_t)sg.x, (int64_t)2 * Kt},
                            x.options().dtype(at::kFloat));
      sp.part = part_out->data_ptr<float>();
      if (C_ == 64)
        hipLaunchKernelGGL((conv_gemm_smallc_kernel<2, true>), sg, dim3(256),
                           0, cur_stream(), sp);
      else if (C_ == 32)
        hipLaunchKernelGGL((conv_gemm_smallc_kernel<1, true>), sg, dim3(256),
                           0, cur_stream(), sp);
      else
        hipLaunchKernelGGL(conv_gemm_small_kernel<true>, sg, dim3(256), 0,
                           cur_stream(), sp);
    } else if (C_ == 64)
      hipLaunchKernelGGL((conv_gemm_smallc_kernel<2, false>), sg, dim3(256),
                         0, cur_stream(), sp);
    else if (C_ == 32)
      hipLaunchKernelGGL((conv_gemm_smallc_kernel<1, false>), sg, dim3(256),
                         0, cur_stream(), sp);
    else
      hipLaunchKernelGGL(conv_gemm_small_kernel<false>, sg, dim3(256), 0,
                         cur_stream(), sp);
    return y;
  }
  const int Ho = (H + 2 * ph - dh * (R - 1) - 1) / sh + 1;
  const int Wo = (W + 2 * pw - dw * (S - 1) - 1) / sw + 1;
  if (!small_off && groups == 1 && Kt == 64 && C_ <= 64 && C_ % 8 == 0 &&
      dh == 1 && dw == 1 && R * S > 1 && R * S * C_ <= 1024 &&
      x.scalar_type() == at::kBFloat16) {
    const int RSC = R * S * C_;
    const int RSCp = (RSC + 31) / 32 * 32;
    at::Tensor wsp = w;
    if (RSCp != RSC) {
      wsp = at::zeros({(int64_t)Kt, RSCp}, w.options());
      wsp.narrow(1, 0, RSC).copy_(
          w.permute({0, 2, 3, 1}).reshape({(int64_t)Kt, RSC}));
    }
    auto y = at::empty({N, Kt, Ho, Wo},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
    SmallConvParams sp;
    sp.x = (const __hip_bfloat16*)x.data_ptr();
    sp.w = (const __hip_bfloat16*)wsp.data_ptr();
    sp.y = (__hip_bfloat16*)y.data_ptr();
    sp.C = C_; sp.H = H; sp.W = W;
    sp.R = R; sp.S = S; sp.SC = S * C_;
    sp.sh = sh; sp.sw = sw; sp.ph = ph; sp.pw = pw;
    sp.Ho = Ho; sp.Wo = Wo;
    sp.RSC = RSC; sp.RSCp = RSCp;
    sp.M = (int64_t)N * Ho * Wo;
    sp.magicHoWo = ((1ULL << 47) / ((unsigned long long)Ho * Wo)) + 1;
    sp.magicWo = ((1ULL << 47) / (unsigned long long)Wo) + 1;
    sp.magicSC = ((1ULL << 47) / (unsigned long long)(S * C_)) + 1;
    sp.magicC = ((1ULL << 47) / (unsigned long long)C_) + 1;
    static at::Tensor zbuf;  // 16 zero bytes; allocated outside any capture
    if (!zbuf.defined() || zbuf.device() != x.device())
      zbuf = at::zeros({8}, x.options());
    sp.zbuf = (const __hip_bfloat16*)zbuf.data_ptr();
    sp.part = nullptr;
    const dim3 skg((int)((sp.M + 255) / 256));
    if (bemit != nullptr) part_out = nullptr;  // smallk: no bwd emission
    if (part_out != nullptr) {
      *part_out = at::empty({(int64_t)skg.x, 128},
                            x.options().dtype(at::kFloat));
      sp.part = part_out->data_ptr<float>();
      if (C_ == 64)
        hipLaunchKernelGGL((conv_smallk_pipe_kernel<2, true>), skg, dim3(256),
                           0, cur_stream(), sp);
      else if (C_ == 32)
        hipLaunchKernelGGL((conv_smallk_pipe_kernel<1, true>), skg, dim3(256),
                           0, cur_stream(), sp);
      else
        hipLaunchKernelGGL(conv_smallk_kernel<true>, skg, dim3(256), 0,
                           cur_stream(), sp);
    } else if (C_ == 64)
      hipLaunchKernelGGL((conv_smallk_pipe_kernel<2, false>), skg, dim3(256),
                         0, cur_stream(), sp);
    else if (C_ == 32)
      hipLaunchKernelGGL((conv_smallk_pipe_kernel<1, false>), skg, dim3(256),
                         0, cur_stream(), sp);
    else
      hipLaunchKernelGGL(conv_smallk_kernel<false>, skg, dim3(256), 0,
                         cur_stream(), sp);
    return y;
  }
  auto y = at::empty({N, Kt, Ho, Wo},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  return conv2d_fwd_into(x, w, y, Ho, Wo, sh, sw, ph, pw, dh, dw, groups, 1,
                         1, 0, 0, part_out, bemit);
}


// Probe/testing entry: force the v1 predicated kernel (bypasses the
// small-family routing in conv2d_fwd_impl).
at::Tensor conv2d_fwd_v1(at::Tensor x, at::Tensor w, at::Tensor y,
                         int64_t Ho, int64_t Wo, int64_t sh, int64_t sw,
                         int64_t ph, int64_t pw, int64_t dh, int64_t dw,
                         int64_t groups) {
  return conv2d_fwd_into(x, w, y, Ho, Wo, sh, sw, ph, pw, dh, dw, groups);
}

at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w, int64_t sh, int64_t sw,
                      int64_t ph, int64_t pw, int64_t dh, int64_t dw,
                      int64_t groups) {
  return conv2d_fwd_impl(x, w, sh, sw, ph, pw, dh, dw, groups, nullptr);
}

// conv forward + BN sum/sumsq partials emitted from the epilogue (F1):
// returns {y, part} with part a [nblocks, 2*Kt] fp32 tensor consumable by
// bn_stats(part_opt=...). Stats are over the bf16-rounded stored y, exactly
// matching what bn_sums(y) would compute.
std::vector<at::Tensor> conv2d_fwd_bn(at::Tensor x, at::Tensor w, int64_t sh,
                                      int64_t sw, int64_t ph, int64_t pw,
                                      int64_t dh, int64_t dw, int64_t groups) {
  at::Tensor part;
  auto y = conv2d_fwd_impl(x, w, sh, sw, ph, pw, dh, dw, groups, &part);
  TORCH_CHECK(part.defined(), "conv2d_fwd_bn: dispatch path without partials");
  return {y, part};
}

// NT GEMM through the conv kernel: y[M, N] = a[M, K] @ b[N, K]^T
at::Tensor gemm_nt(at::Tensor a, at::Tensor b) {
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "gemm_nt expects 2-D");
  const int M = a.size(0), K = a.size(1), Nc = b.size(0);
  // NHWC 4-D with H=W=1: logical [M,K,1,1] channels_last == [M][K] rows
  auto a_cl = a.reshape({M, K, 1, 1}).contiguous(at::MemoryFormat::ChannelsLast);
  auto b_cl = b.reshape({Nc, K, 1, 1}).contiguous(at::MemoryFormat::ChannelsLast);
  // route through the dispatcher so deep-K GEMMs hit the v2 ring kernel
  auto y = conv2d_fwd(a_cl, b_cl, 1, 1, 0, 0, 1, 1, 1);
  return y.reshape({M, Nc});
}

// ---------------------------------------------------------------------------
// dgrad transforms
// ---------------------------------------------------------------------------
namespace {

// wt[g*Cg + c][r][s][Kg] = w[g*Kg + k][R-1-r][S-1-s][c]  (per group)
template <typename T>
__global__ void weight_flip_t_kernel(const T* __restrict__ w, T* __restrict__ wt,
                                     int Kg, int Cg, int R, int S, int G) {
  const int64_t total = (int64_t)G * Cg * R * S * Kg;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = i;
    const int k = t % Kg;
    t /= Kg;
    const int s = t % S;
    t /= S;
    const int r = t % R;
    t /= R;
    const int c = t % Cg;
    const int g = t / Cg;
    wt[i] = w[((((int64_t)(g * Kg + k) * R) + (R - 1 - r)) * S + (S - 1 - s)) *
                  Cg +
              c];
  }
}

// zero-dilate NHWC: out[n, ho*sh, wo*sw, c] = x[n, ho, wo, c]
template <typename T>
__global__ void dilate_nhwc_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   int N, int Ho, int Wo, int C, int sh,
                                   int sw, int Hd, int Wd) {
  const int64_t total = (int64_t)N * Ho * Wo * C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c = i % C;
    int64_t t = i / C;
    const int wo = t % Wo;
    t /= Wo;
    const int ho = t % Ho;
    const int n = t / Ho;
    y[(((int64_t)n * Hd + ho * sh) * Wd + wo * sw) * C + c] = x[i];
  }
}

}  // namespace

namespace {
// fused dgrad weight transform: flipT + span-pad in ONE pass. Output flat
// [Ct, R, SPAN64] rows for the v2 ring (sp < S*Kg: w[gKg+k][R-1-r][S-1-s][c],
// else 0) — replaces the separate flip write + pad read/write passes.
template <typename T>
__global__ void weight_flip_t_span_kernel(const T* __restrict__ w,
                                          T* __restrict__ o, int Kg, int Cg,
                                          int R, int S, int G, int SPAN64) {
  const int64_t total = (int64_t)G * Cg * R * SPAN64;
  const int SKg = S * Kg;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int sp = i % SPAN64;
    int64_t t = i / SPAN64;
    const int r = t % R;
    t /= R;
    const int c = t % Cg;
    const int g = t / Cg;
    T v = from_f32<T>(0.f);
    if (sp < SKg) {
      const int sidx = sp / Kg;
      const int k = sp - sidx * Kg;
      v = w[((((int64_t)(g * Kg + k) * R) + (R - 1 - r)) * S +
             (S - 1 - sidx)) * Cg + c];
    }
    o[i] = v;
  }
}
}  // namespace

at::Tensor weight_flip_t_span(at::Tensor w, int64_t groups, int64_t SPAN64) {
  CHECK_GPU(w);
  check_nhwc(w, "w");
  const int Kt = w.size(0), Cg = w.size(1), R = w.size(2), S = w.size(3);
  const int Kg = Kt / groups;
  auto o = at::empty({(int64_t)Cg * groups * R * SPAN64}, w.options());
  const int64_t total = o.numel();
  DISPATCH_FLOAT_AND_BF16(w.scalar_type(), "weight_flip_t_span", [&] {
    hipLaunchKernelGGL((weight_flip_t_span_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)w.data_ptr(), (scalar_t*)o.data_ptr(),
                       Kg, Cg, R, S, (int)groups, (int)SPAN64);
  });
  return o;
}

// weight transform for dgrad: [Kt,Cg,R,S]cl -> [Ct,Kg,R,S]cl flipped
at::Tensor weight_flip_t(at::Tensor w, int64_t groups) {
  CHECK_GPU(w);
  check_nhwc(w, "w");
  const int Kt = w.size(0), Cg = w.size(1), R = w.size(2), S = w.size(3);
  const int Kg = Kt / groups;
  auto wt = at::empty({Cg * groups, Kg, R, S},
                      w.options().memory_format(at::MemoryFormat::ChannelsLast));
  int64_t total = (int64_t)Kt * Cg * R * S;
  DISPATCH_FLOAT_AND_BF16(w.scalar_type(), "weight_flip_t", [&] {
    hipLaunchKernelGGL((weight_flip_t_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)w.data_ptr(), (scalar_t*)wt.data_ptr(),
                       Kg, Cg, R, S, (int)groups);
  });
  return wt;
}

at::Tensor dilate_nhwc(at::Tensor x, int64_t sh, int64_t sw) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), Ho = x.size(2), Wo = x.size(3);
  const int Hd = (Ho - 1) * sh + 1, Wd = (Wo - 1) * sw + 1;
  auto y = at::empty({N, C, Hd, Wd},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  y.zero_();
  int64_t total = (int64_t)N * Ho * Wo * C;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "dilate_nhwc", [&] {
    hipLaunchKernelGGL((dilate_nhwc_kernel<scalar_t>),
                       dim3(grid_1d(total, 256)), dim3(256), 0, cur_stream(),
                       (const scalar_t*)x.data_ptr(), (scalar_t*)y.data_ptr(),
                       N, Ho, Wo, C, sh, sw, Hd, Wd);
  });
  return y;
}

// pad the channel dim (NHWC innermost) with zeros: stem C=3 -> 8 so the
// MFMA kernel's Cg%8 requirement holds
namespace {
template <typename T>
__global__ void pad_channels_kernel(const T* __restrict__ x, T* __restrict__ y,
                                    int64_t rows, int C, int Cn) {
  constexpr int V = 16 / sizeof(T);
  if (Cn == V) {
    // one 16-byte output pack per thread (stem 3->8): the per-element
    // modulo form ran at 2.8 TB/s
    union {
      T b[V];
      uint4 q;
    } u;
    for (int64_t r = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
         r < rows; r += (int64_t)gridDim.x * blockDim.x) {
#pragma unroll
      for (int j = 0; j < V; ++j)
        u.b[j] = j < C ? x[r * C + j] : from_f32<T>(0.f);
      *reinterpret_cast<uint4*>(y + r * V) = u.q;
    }
    return;
  }
  const int64_t total = rows * Cn;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int c = i % Cn;
    const int64_t r = i / Cn;
    y[i] = (c < C) ? x[r * C + c] : from_f32<T>(0.f);
  }
}
}  // namespace

at::Tensor pad_channels(at::Tensor x, int64_t Cn) {
  CHECK_GPU(x);
  check_nhwc(x, "x");
  const int N = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  auto y = at::empty({N, Cn, H, W},
                     x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const int64_t rows = (int64_t)N * H * W;
  DISPATCH_FLOAT_AND_BF16(x.scalar_type(), "pad_channels", [&] {
    hipLaunchKernelGGL((pad_channels_kernel<scalar_t>),
                       dim3(grid_1d(rows * Cn, 256)), dim3(256), 0,
                       cur_stream(), (const scalar_t*)x.data_ptr(),
                       (scalar_t*)y.data_ptr(), rows, C, (int)Cn);
  });
  return y;
}

// dgrad: gx = conv(dilate(gy), flipT(w)) written into a zeroed [H, W] canvas
static at::Tensor conv2d_dgrad_impl(at::Tensor gy, at::Tensor w, int64_t H,
                                    int64_t W, int64_t sh, int64_t sw,
                                    int64_t ph, int64_t pw, int64_t dh,
                                    int64_t dw, int64_t groups,
                                    at::Tensor* part_out,
                                    const BnBwdEmit* bemit,
                                    const at::Tensor* pre = nullptr,
                                    int64_t pre_kind = -1) {
  CHECK_GPU(gy);
  check_nhwc(gy, "gy");
  const int N = gy.size(0);
  const int Cg = w.size(1), R = w.size(2), S = w.size(3);
  const int Ct = Cg * groups;
  {
    // same-size dgrad on the v2 ring: ONE fused flip+span-pad weight pass
    // (the generic path below would flip then pad separately)
    const int Kt0 = gy.size(1);
    const int dKg0 = Ct / (int)groups;
    const int dCg0 = Kt0 / (int)groups;
    static const int v2mink2 = []() {
      const char* e = getenv("DISTRIBUUUU_V2_MINK");
      return e ? atoi(e) : 192;
    }();
    static const bool v2off2 = []() {
      const char* e = getenv("DISTRIBUUUU_CONV_V2");
      return e && e[0] == '0';
    }();
    // any ph <= dh*(R-1) works at stride 1: the as-fwd conv with pad
    // dh*(R-1)-ph lands on the conv-input size exactly (ph = 0 is the
    // padded-canvas BN fusion's dgrad)
    if (sh == 1 && sw == 1 && dh * (R - 1) >= ph &&
        dw * (S - 1) >= pw && Kt0 % (8 * (int)groups) == 0 && !v2off2 &&
        dKg0 >= v2mink2 && (int64_t)R * S * dCg0 >= v2_minrsc() &&
        dCg0 % 8 == 0 &&
        gy.scalar_type() == at::kBFloat16) {
      const int SPAN64 = ((int)(S * dCg0) + 63) / 64 * 64;
      auto wsp = (pre != nullptr && pre_kind == 1)
                     ? *pre
                     : weight_flip_t_span(w, groups, SPAN64);
      return conv2d_fwd_v2_flat(gy, wsp, Ct, dCg0, R, S, 1, 1,
                                dh * (R - 1) - ph, dw * (S - 1) - pw, dh, dw,
                                groups, part_out, bemit);
    }
  }
  auto wt = (pre != nullptr && pre_kind == 0)
                ? *pre
                : weight_flip_t(w, groups);  // [Ct, Kg, R, S] cl
  if (R == 1 && S == 1 && (sh > 1 || sw > 1) && ph == 0 && pw == 0) {
    // strided-output GEMM: gx[ho*sh, wo*sw] = gy[ho, wo] @ w^T, rest zero.
    // The zeroed gap positions contribute g = 0 to the BN-backward sums, so
    // partials over written positions alone are the full sums.
    auto gx1 = at::empty({N, Ct, H, W}, gy.options().memory_format(
                                            at::MemoryFormat::ChannelsLast));
    gx1.zero_();
    conv2d_fwd_into(gy, wt, gx1, gy.size(2), gy.size(3), 1, 1, 0, 0, 1, 1,
                    groups, sh, sw, 0, 0, part_out, bemit);
    return gx1;
  }
  if (R == 3 && S == 3 && sh == 2 && sw == 2 && dh == 1 && dw == 1 &&
      ph == 1 && pw == 1) {
    // parity decomposition: for stride-2 with pad 1, each output-parity
    // class (h%2, w%2) of gx receives only the taps of matching parity —
    // four small {1,2}x{1,2}-tap convs over gy scattered with stride 2
    // instead of a 3x3 conv over 2x-dilated (3/4-zero) gy. Derivation: for
    // h = 2u + h0, ho = u + e with e = (h0 + 1 - r)/2, so the e-ordered tap
    // weights are flipT(w) rows (1 - h0)::2.
    auto gx = at::empty({N, Ct, H, W}, gy.options().memory_format(
                                           at::MemoryFormat::ChannelsLast));
    std::vector<at::Tensor> parts;
    for (int h0 = 0; h0 < 2; ++h0) {
      const int64_t hu = (H - h0 + 1) >> 1;
      auto wr = (h0 == 0) ? wt.slice(2, 1, 2) : wt.slice(2, 0, 3, 2);
      for (int w0 = 0; w0 < 2; ++w0) {
        const int64_t wu = (W - w0 + 1) >> 1;
        auto wsub = ((w0 == 0) ? wr.slice(3, 1, 2) : wr.slice(3, 0, 3, 2))
                        .contiguous(at::MemoryFormat::ChannelsLast);
        at::Tensor sub_part;
        conv2d_fwd_into(gy, wsub, gx, hu, wu, 1, 1, 0, 0, 1, 1, groups, 2, 2,
                        h0, w0, bemit ? &sub_part : nullptr, bemit);
        if (bemit && sub_part.defined()) parts.push_back(sub_part);
      }
    }
    if (bemit && part_out && parts.size() == 4)
      *part_out = at::cat(parts, 0);
    return gx;
  }
  const int Kt_ = gy.size(1);
  if (sh == 1 && sw == 1 && dh * (R - 1) >= ph && dw * (S - 1) >= pw &&
      (R > 1 || (ph == 0 && pw == 0)) &&
      (groups == 1 || Kt_ % (8 * groups) == 0)) {
    // same-size conv: the plain fwd path applies (the v2-eligible shapes
    // were already taken above with the fused weight transform)
    return conv2d_fwd_impl(gy, wt, 1, 1, dh * (R - 1) - ph,
                           dw * (S - 1) - pw, dh, dw, groups, part_out,
                           bemit);
  }
  (void)Kt_;
  auto gyd = (sh == 1 && sw == 1) ? gy : dilate_nhwc(gy, sh, sw);
  const int Hd = gyd.size(2), Wd = gyd.size(3);
  const int pph = dh * (R - 1) - ph, ppw = dw * (S - 1) - pw;
  TORCH_CHECK(pph >= 0 && ppw >= 0, "dgrad pad underflow");
  // Every input row/col receives its own output: run the conv at logical
  // output size exactly (H, W); the kernel's bounds predication supplies the
  // zero-padding of the dilated gy on both ends (incl. rows past the last
  // forward window, where forward's floor division dropped input).
  (void)Hd; (void)Wd;
  auto gx = at::empty({N, Ct, H, W}, gy.options().memory_format(
                                           at::MemoryFormat::ChannelsLast));
  conv2d_fwd_into(gyd, wt, gx, H, W, 1, 1, pph, ppw, dh, dw, groups, 1, 1, 0,
                  0, part_out, bemit);
  return gx;
}

at::Tensor conv2d_dgrad(at::Tensor gy, at::Tensor w, int64_t H, int64_t W,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                        int64_t dh, int64_t dw, int64_t groups) {
  return conv2d_dgrad_impl(gy, w, H, W, sh, sw, ph, pw, dh, dw, groups,
                           nullptr, nullptr);
}

// Forward-time dgrad weight-transform precompute (launched on a side
// stream during the forward pass so backward's critical path skips the
// flip kernels). Returns {transformed_weight, kind}: kind 1 = span-padded
// flat flip for the same-size v2 ring, kind 0 = plain flipT. The gate
// mirrors conv2d_dgrad_impl's v2 dispatch exactly.
std::vector<at::Tensor> conv2d_dgrad_prep(at::Tensor w, int64_t Kt0,
                                          int64_t sh, int64_t sw, int64_t ph,
                                          int64_t pw, int64_t dh, int64_t dw,
                                          int64_t groups) {
  const int Cg = w.size(1), R = w.size(2), S = w.size(3);
  const int Ct = Cg * (int)groups;
  const int dKg0 = Ct / (int)groups;
  const int dCg0 = (int)(Kt0 / groups);
  static const int v2mink4 = []() {
    const char* e = getenv("DISTRIBUUUU_V2_MINK");
    return e ? atoi(e) : 192;
  }();
  static const bool v2off4 = []() {
    const char* e = getenv("DISTRIBUUUU_CONV_V2");
    return e && e[0] == '0';
  }();
  int64_t kind = 0;
  at::Tensor out;
  if (sh == 1 && sw == 1 && dh * (R - 1) >= ph &&
      dw * (S - 1) >= pw && Kt0 % (8 * groups) == 0 && !v2off4 &&
      dKg0 >= v2mink4 && (int64_t)R * S * dCg0 >= v2_minrsc() &&
      dCg0 % 8 == 0 &&
      w.scalar_type() == at::kBFloat16) {
    const int SPAN64 = ((int)(S * dCg0) + 63) / 64 * 64;
    out = weight_flip_t_span(w, groups, SPAN64);
    kind = 1;
  } else {
    out = weight_flip_t(w, groups);
  }
  return {out, at::scalar_tensor(kind)};
}

at::Tensor conv2d_dgrad_pre(at::Tensor gy, at::Tensor w, int64_t H,
                            int64_t W, int64_t sh, int64_t sw, int64_t ph,
                            int64_t pw, int64_t dh, int64_t dw,
                            int64_t groups, at::Tensor pre,
                            int64_t pre_kind) {
  return conv2d_dgrad_impl(gy, w, H, W, sh, sw, ph, pw, dh, dw, groups,
                           nullptr, nullptr, &pre, pre_kind);
}

// dgrad + BN-backward stat partials (docs/DESIGN_bn_conv_fusion.md, the
// dgrad-side analogue of conv2d_fwd_bn): bnx/scale/shift/act describe the
// BatchNorm whose gy this dgrad produces. Returns {gx, part}; part is an
// EMPTY tensor when the dispatched kernel family doesn't emit (caller falls
// back to the standalone reduce).
std::vector<at::Tensor> conv2d_dgrad_bn(at::Tensor gy, at::Tensor w,
                                        int64_t H, int64_t W, int64_t sh,
                                        int64_t sw, int64_t ph, int64_t pw,
                                        int64_t dh, int64_t dw,
                                        int64_t groups, at::Tensor bnx,
                                        at::Tensor bnscale, at::Tensor bnshift,
                                        int64_t bnact) {
  BnBwdEmit em;
  em.x = (const __hip_bfloat16*)bnx.data_ptr();
  em.scale = bnscale.data_ptr<float>();
  em.shift = bnshift.data_ptr<float>();
  em.act = (int)bnact;
  at::Tensor part;
  auto gx = conv2d_dgrad_impl(gy, w, H, W, sh, sw, ph, pw, dh, dw, groups,
                              &part, &em);
  // (pre-transform not threaded here: the EMODE-2 path is opt-in/off)
  if (!part.defined()) part = at::empty({0, 0}, bnscale.options());
  return {gx, part};
}

// Residual-fork gradient accumulation (docs/ARCHITECTURE.md "Round-2
// status"): compute this conv's dgrad and ADD it into `into` — the other
// fork branch's already-computed gradient — inside the dgrad epilogue
// (+1 read stream) instead of autograd's separate 2-read/1-write
// elementwise add over the whole gx. Returns {tensor, flag}: flag 1 means
// `into` now holds the sum; flag 0 means the dispatched route has no ACC
// epilogue and the returned tensor is a FRESH gx (caller must add).
std::tuple<at::Tensor, int64_t> conv2d_dgrad_acc(
    at::Tensor gy, at::Tensor w, int64_t H, int64_t W, int64_t sh, int64_t sw,
    int64_t ph, int64_t pw, int64_t dh, int64_t dw, int64_t groups,
    at::Tensor into, c10::optional<at::Tensor> pre_opt = c10::nullopt,
    int64_t pre_kind = -1) {
  const at::Tensor* pre = pre_opt.has_value() ? &*pre_opt : nullptr;
  CHECK_GPU(gy);
  const int N = gy.size(0);
  const int Cg = w.size(1), R = w.size(2), S = w.size(3);
  const int Ct = Cg * groups;
  const int Kt0 = gy.size(1);
  const int dKg0 = Ct / (int)groups;
  const int dCg0 = Kt0 / (int)groups;
  const bool same_size = sh == 1 && sw == 1 && dh * (R - 1) >= ph &&
                         dw * (S - 1) >= pw &&
                         (R > 1 || (ph == 0 && pw == 0));
  static const int v2mink3 = []() {
    const char* e = getenv("DISTRIBUUUU_V2_MINK");
    return e ? atoi(e) : 192;
  }();
  static const bool v2off3 = []() {
    const char* e = getenv("DISTRIBUUUU_CONV_V2");
    return e && e[0] == '0';
  }();
  if (gy.scalar_type() == at::kBFloat16 && groups == 1 && dCg0 % 8 == 0 &&
      into.scalar_type() == at::kBFloat16 &&
      into.is_contiguous(at::MemoryFormat::ChannelsLast) &&
      into.sizes() == at::IntArrayRef({(int64_t)N, (int64_t)Ct, H, W})) {
    if (same_size && !v2off3 && Kt0 % 8 == 0 && dKg0 >= v2mink3 &&
        (int64_t)R * S * dCg0 >= v2_minrsc()) {
      const int SPAN64 = ((int)(S * dCg0) + 63) / 64 * 64;
      auto wsp = (pre != nullptr && pre_kind == 1)
                     ? *pre
                     : weight_flip_t_span(w, 1, SPAN64);
      conv2d_fwd_v2_flat(gy, wsp, Ct, dCg0, R, S, 1, 1, dh * (R - 1) - ph,
                         dw * (S - 1) - pw, dh, dw, 1, nullptr, nullptr,
                         &into);
      return {into, 1};
    }
    if (same_size) {
      auto wt = (pre != nullptr && pre_kind == 0) ? *pre
                                                   : weight_flip_t(w, 1);
      conv2d_fwd_into(gy, wt, into, H, W, 1, 1, dh * (R - 1) - ph,
                      dw * (S - 1) - pw, dh, dw, 1, 1, 1, 0, 0, nullptr,
                      nullptr, true);
      return {into, 1};
    }
    if (R == 1 && S == 1 && (sh > 1 || sw > 1) && ph == 0 && pw == 0) {
      // strided 1x1 proj: scatter-accumulate at (ho*sh, wo*sw); the gap
      // positions receive no main-branch gradient, so `into` already holds
      // their final values.
      auto wt = (pre != nullptr && pre_kind == 0) ? *pre
                                                   : weight_flip_t(w, 1);
      conv2d_fwd_into(gy, wt, into, gy.size(2), gy.size(3), 1, 1, 0, 0, 1, 1,
                      1, sh, sw, 0, 0, nullptr, nullptr, true);
      return {into, 1};
    }
    if (R == 3 && S == 3 && sh == 2 && sw == 2 && dh == 1 && dw == 1 &&
        ph == 1 && pw == 1) {
      // parity decomposition (see conv2d_dgrad_impl): the four parity
      // classes tile gx disjointly, so each sub-conv scatter-accumulates
      // its own positions exactly once.
      auto wt = (pre != nullptr && pre_kind == 0) ? *pre
                                                   : weight_flip_t(w, 1);
      for (int h0 = 0; h0 < 2; ++h0) {
        const int64_t hu = (H - h0 + 1) >> 1;
        auto wr = (h0 == 0) ? wt.slice(2, 1, 2) : wt.slice(2, 0, 3, 2);
        for (int w0 = 0; w0 < 2; ++w0) {
          const int64_t wu = (W - w0 + 1) >> 1;
          auto wsub = ((w0 == 0) ? wr.slice(3, 1, 2) : wr.slice(3, 0, 3, 2))
                          .contiguous(at::MemoryFormat::ChannelsLast);
          conv2d_fwd_into(gy, wsub, into, hu, wu, 1, 1, 0, 0, 1, 1, 1, 2, 2,
                          h0, w0, nullptr, nullptr, true);
        }
      }
      return {into, 1};
    }
  }
  auto gx = conv2d_dgrad_impl(gy, w, H, W, sh, sw, ph, pw, dh, dw, groups,
                              nullptr, nullptr, pre, pre_kind);
  return {gx, 0};
}
