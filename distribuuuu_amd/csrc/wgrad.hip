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
#include "common.h"

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
  // magic-multiply division: q = (m * magic) >> 47 == m / d exactly for
  // m*d < 2^47 (here m < 2^22, d < 2^15). Runtime u32 division in the
  // per-k-step staging path costs ~30 VALU cycles each.
  unsigned long long magicHoWo, magicWo;
};

DEV_INLINE int magic_div(int m, unsigned long long magic) {
  return (int)(((unsigned long long)(unsigned)m * magic) >> 47);
}

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

// issue-only variant: the reads are NOT awaited here — callers batch many
// issues and place one counted s_waitcnt before consumption. (The original
// wait-per-fragment tr_read_frag serialized 8 full LDS latencies per k-step
// against idle MFMAs — measured ~12% MFMA utilization.)
union TrFrag {
  uint2 h[2];
  bf16x8w v;
};

DEV_INLINE void tr_read_issue(unsigned a0, unsigned a1, TrFrag& f) {
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3"
      : "=&v"(f.h[0]), "=&v"(f.h[1])
      : "v"(a0), "v"(a1)
      : "memory");
}

__global__ __launch_bounds__(256) void conv_wgrad_kernel(WgradParams p) {
  const int g = blockIdx.z;
  int ktile, ntile, chunk;
  const int KN = p.ktiles * p.ntiles;
  if (KN <= 12) {
    // XCD grouping for small tile counts (stem, 3x3 64ch): one chunk's
    // <=12 blocks all land on ONE XCD (block b runs on XCD b%8) so the
    // chunk's x/gy windows are read into one L2 instead of 7-8. (Grouping
    // every shape this way measured slower — big-KN chunks overflow the
    // 4 MiB XCD L2; host pads chunks to a multiple of 8 for the 1-D grid.)
    const int xcd = blockIdx.x & 7, seq = blockIdx.x >> 3;
    chunk = (seq / KN) * 8 + xcd;
    if (chunk >= p.chunks) return;
    const int kn = seq % KN;
    ktile = kn % p.ktiles;
    ntile = kn / p.ktiles;
  } else {
    ktile = blockIdx.x % p.ktiles;
    ntile = blockIdx.x / p.ktiles;
    chunk = blockIdx.y;
  }

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
      const int n = magic_div(mm, p.magicHoWo);
      const int rem = mm - n * HoWo;
      const int ho = magic_div(rem, p.magicWo);
      const int wo = rem - ho * p.Wo;
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
    // issue ALL 16 tr reads for both 32-deep m sub-steps, then consume with
    // counted waits: mc=0's MFMAs run while mc=1's reads are still in flight
    TrFrag fa[2][2], fb[2][2];
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
#pragma unroll
      for (int mi = 0; mi < 2; ++mi) {
        const int sb = (mc * 4 + (wm * 2 + mi)) * SUBT;
        tr_read_issue(abase + tr_addr(sb, lane, 0),
                      abase + tr_addr(sb, lane, 1), fa[mc][mi]);
      }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int sb = (mc * 4 + (wn * 2 + ni)) * SUBT;
        tr_read_issue(bbase + tr_addr(sb, lane, 0),
                      bbase + tr_addr(sb, lane, 1), fb[mc][ni]);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(8)" ::: "memory");  // mc=0's 8 reads done
    __builtin_amdgcn_sched_barrier(0);
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        accv[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            fa[0][mi].v, fb[0][ni].v, accv[mi][ni], 0, 0, 0);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        accv[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            fa[1][mi].v, fb[1][ni].v, accv[mi][ni], 0, 0, 0);
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

// ---------------------------------------------------------------------------
// 128x128 TR-STAGED wgrad: the 64x64 kernel issues 16 tr-read pairs for 8
// MFMAs per k-step per wave (LDS-read bound at ~350 TF); this tile gives
// each wave a 64x64 output (16 tr-read pairs for 32 MFMAs) while KEEPING
// the full per-thread boundary predication the glds ring lacks — so dense
// 3x3 shapes qualify with no pad pass. Staging uses the same
// [32 m][16 col] 1056-B subtiles and b128 writes.
// ---------------------------------------------------------------------------
constexpr int T128_BM = 128;            // k rows per tile
constexpr int T128_BN = 128;            // rsc cols per tile
constexpr int T128_TILE = 16 * SUBT;    // one operand tile (16.5 KB)

struct WgradT128Params {
  const __hip_bfloat16* x;   // [N,H,W,Ct]
  const __hip_bfloat16* gy;  // [N,Ho,Wo,Kt]
  float* acc;                // [Kt, R*S*Cg] zeroed
  int N, H, W, Ct, Kt;
  int R, S, Cg, Kg;
  int sh, sw, ph, pw, dh, dw;
  int Ho, Wo;
  int M, RSC;
  int ktiles, ntiles, chunks, csteps;
  int xcd_remap;
  unsigned long long magicHoWo, magicWo;
};

__global__ __launch_bounds__(256) void conv_wgrad_tr128_kernel(
    WgradT128Params p) {
  const int g = blockIdx.z;
  int tile = blockIdx.x;
  {  // XCD-aware bijective remap (same form as conv2.hip T1): each XCD's
     // dispatch subsequence (blockIdx.x % 8) gets a CONTIGUOUS tile range,
     // decomposed ntile-fastest below, so the staged A (gy) chunk-slab is
     // re-served from that XCD's L2 across its ntile walk instead of
     // re-read from HBM ntiles times.
    const int tiles = p.ktiles * p.ntiles;
    const int q = tiles >> 3, r8 = tiles & 7;
    const int xcd = tile % 8, idx = tile / 8;
    if (p.xcd_remap)
      tile = (xcd < r8 ? xcd * (q + 1) : r8 * (q + 1) + (xcd - r8) * q) + idx;
  }
  const int ktile = p.xcd_remap ? tile / p.ntiles : tile % p.ktiles;
  const int ntile = p.xcd_remap ? tile % p.ntiles : tile / p.ktiles;
  const int chunk = blockIdx.y;

  __shared__ __align__(16) char smem[2 * 2 * T128_TILE];
  auto ldsA = [&](int buf) -> char* { return smem + buf * 2 * T128_TILE; };
  auto ldsB = [&](int buf) -> char* {
    return smem + buf * 2 * T128_TILE + T128_TILE;
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int il = lane & 15, kq = lane >> 4;

  // staging: slot s = tid + it*256 (it<4): col8 = (s&7)*8 + (s>>9)*64,
  // m = (s>>3)&63 — 8 consecutive lanes span 4 subtiles at +4-dword steps
  // and each next 8-lane group rotates +8 dwords (m+1), so every 8-lane
  // b128 write phase hits distinct banks (the 64x64 kernel's property; the
  // first s>>4 mapping wrapped banks at lane 8 and cost ~2e9 conflict
  // cycles in the PMC capture)
  int s_m[4], s_col[4], s_off[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int sl = tid + it * 256;
    s_m[it] = (sl >> 3) & 63;
    s_col[it] = ((sl & 7) << 3) + ((sl >> 9) << 6);
    s_off[it] = ((s_m[it] >> 5) * 8 + (s_col[it] >> 4)) * SUBT +
                (s_m[it] & 31) * 32 + (s_col[it] & 15) * 2;
  }
  const int SCg = p.S * p.Cg;
  // B (x) rsc decomposition per slot (fixed)
  int b_r[4], b_s[4], b_c[4];
  bool b_ok[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int rsc = ntile * T128_BN + s_col[it];
    b_ok[it] = rsc < p.RSC;
    const int rr = b_ok[it] ? rsc / SCg : 0;
    const int rem = (b_ok[it] ? rsc : 0) - rr * SCg;
    b_r[it] = rr;
    b_s[it] = rem / p.Cg;
    b_c[it] = rem - b_s[it] * p.Cg;
  }
  int a_k[4];
  bool a_ok[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int k = ktile * T128_BM + s_col[it];
    a_ok[it] = k < p.Kg;
    a_k[it] = g * p.Kg + (a_ok[it] ? k : 0);
  }

  const int m0 = chunk * (p.csteps * WBK);
  const int HoWo = p.Ho * p.Wo;
  const int ksteps = min(p.csteps, (int)((p.M - m0 + WBK - 1) / WBK));

  uint4 regA[4], regB[4];
  auto stage_load = [&](int ks) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int m = m0 + ks * WBK + s_m[it];
      const bool m_ok = m < p.M;
      const int mm = m_ok ? m : 0;
      const int n = magic_div(mm, p.magicHoWo);
      const int rem = mm - n * HoWo;
      const int ho = magic_div(rem, p.magicWo);
      const int wo = rem - ho * p.Wo;
      regA[it] = (m_ok && a_ok[it])
                     ? *reinterpret_cast<const uint4*>(
                           p.gy + ((int64_t)mm * p.Kt) + a_k[it])
                     : uint4{0, 0, 0, 0};
      const int h = ho * p.sh - p.ph + b_r[it] * p.dh;
      const int w_ = wo * p.sw - p.pw + b_s[it] * p.dw;
      regB[it] =
          (m_ok && b_ok[it] && h >= 0 && h < p.H && w_ >= 0 && w_ < p.W)
              ? *reinterpret_cast<const uint4*>(
                    p.x + (((int64_t)n * p.H + h) * p.W + w_) * p.Ct +
                    g * p.Cg + b_c[it])
              : uint4{0, 0, 0, 0};
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      *reinterpret_cast<uint4*>(ldsA(buf) + s_off[it]) = regA[it];
      *reinterpret_cast<uint4*>(ldsB(buf) + s_off[it]) = regB[it];
    }
  };

  f32x4w accv[4][4] = {};
  stage_load(0);
  stage_write(0);
  __syncthreads();
  if (ksteps > 1) stage_load(1);

  int cur = 0;
  for (int ks = 0; ks < ksteps; ++ks) {
    const unsigned abase =
        (unsigned)(unsigned long long)(ldsA(cur));
    const unsigned bbase =
        (unsigned)(unsigned long long)(ldsB(cur));
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
      TrFrag fa[4], fb[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int sb = (mc * 8 + wm * 4 + mi) * SUBT;
        tr_read_issue(abase + tr_addr(sb, lane, 0),
                      abase + tr_addr(sb, lane, 1), fa[mi]);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int sb = (mc * 8 + wn * 4 + ni) * SUBT;
        tr_read_issue(bbase + tr_addr(sb, lane, 0),
                      bbase + tr_addr(sb, lane, 1), fb[ni]);
      }
      asm volatile("s_waitcnt lgkmcnt(6)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        if (ni == 1) {
          asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
        }
        if (ni == 2) {
          asm volatile("s_waitcnt lgkmcnt(2)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
        }
        if (ni == 3) {
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          accv[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa[mi].v, fb[ni].v, accv[mi][ni], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
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
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int k = ktile * T128_BM + wm * 64 + mi * 16 + kq * 4 + rr;
      if (k >= p.Kg) continue;
      const int64_t rowbase = (int64_t)(g * p.Kg + k) * p.RSC;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int col = ntile * T128_BN + wn * 64 + ni * 16 + il;
        if (col < p.RSC) atomicAdd(&p.acc[rowbase + col], accv[mi][ni][rr]);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Ring-staged wgrad (the conv-v2 structure applied to wgrad): packed 1 KiB
// [32 m][16 col] subtiles filled by global_load_lds (lane-linear), tr_b16
// fragment reads, 3-slot LDS ring with counted vmcnt across raw barriers.
// Requirements (host-checked): M%64==0, Kg%64==0, (S*Cg)%64==0, Cg%8==0 and
// a pre-padded image when ph/pw>0 (no predication anywhere; rsc/k-tail
// contributions land in discarded output rows/cols).
// ---------------------------------------------------------------------------
constexpr int RSUBT = 1024;               // packed subtile
constexpr int RTILE = 8 * RSUBT;          // one operand tile (8 KB)
constexpr int RSLOT = 2 * RTILE;          // A + B per ring slot (16 KB)

struct WgradRingParams {
  const __hip_bfloat16* x;   // padded [N, Hp, Wp, Ct] when ph/pw > 0
  const __hip_bfloat16* gy;  // [M, Kt]
  float* acc;                // [Kt, R*S*Cg] zeroed
  int Hp, Wp, Ct, Kt;
  int R, S, Cg, Kg;
  int sh, sw, dh, dw;
  int Ho, Wo;
  int M, RSC;
  int ktiles, ntiles;
};

#define RWAITVM(N) asm volatile("s_waitcnt vmcnt(" #N ")" ::: "memory")

__global__ __launch_bounds__(256) void conv_wgrad_ring_kernel(
    WgradRingParams p) {
  const int g = blockIdx.z;
  const int ktile = blockIdx.x % p.ktiles;
  const int ntile = blockIdx.x / p.ktiles;
  const int chunk = blockIdx.y;

  __shared__ __align__(16) char smem[3 * RSLOT];
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int il = lane & 15, kq = lane >> 4;

  // staging geometry: slot s = tid + it*256 (it<2); subtile = s/64,
  // msub = subtile/4, colsub = subtile%4, inner = s%64:
  //   mloc = msub*32 + inner/2 ; col = colsub*16 + (inner%2)*8
  int s_mloc[2], s_col[2];
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    const int s = tid + it * 256;
    const int sub = s >> 6, inner = s & 63;
    s_mloc[it] = (sub >> 2) * 32 + (inner >> 1);
    s_col[it] = (sub & 3) * 16 + (inner & 1) * 8;
  }
  // B (x) rsc decomposition per thread (fixed): clamp tail to 0 (its output
  // column >= RSC is never written)
  const int SCg = p.S * p.Cg;
  int b_r[2], b_s[2], b_c[2];
#pragma unroll
  for (int it = 0; it < 2; ++it) {
    int rsc = ntile * WBN + s_col[it];
    if (rsc >= p.RSC) rsc = 0;
    b_r[it] = rsc / SCg;
    const int rem = rsc - b_r[it] * SCg;
    b_s[it] = rem / p.Cg;
    b_c[it] = rem - b_s[it] * p.Cg;
  }
  const int m0 = chunk * (CHUNK_STEPS * WBK);
  const int HoWo = p.Ho * p.Wo;
  const int nt = min(CHUNK_STEPS, (int)((p.M - m0 + WBK - 1) / WBK));

  auto stage = [&](int buf, int ks) {
    char* base = smem + buf * RSLOT + wid * 1024;
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int m = m0 + ks * WBK + s_mloc[it];  // < M (M%64==0)
      // A: gy[m][ktile*64 + col]
      const __hip_bfloat16* asrc =
          p.gy + (int64_t)m * p.Kt + g * p.Kg + ktile * WBM + s_col[it];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)asrc,
          (__attribute__((address_space(3))) uint32_t*)(base + it * 4096),
          16, 0, 0);
      // B: x[pixel(m) shifted by (r, s)][c]
      const int n = m / HoWo;
      const int rem = m - n * HoWo;
      const int h = (rem / p.Wo) * p.sh + b_r[it] * p.dh;
      const int w_ = (rem % p.Wo) * p.sw + b_s[it] * p.dw;
      const __hip_bfloat16* bsrc =
          p.x + (((int64_t)n * p.Hp + h) * p.Wp + w_) * p.Ct + g * p.Cg +
          b_c[it];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)bsrc,
          (__attribute__((address_space(3))) uint32_t*)(base + RTILE +
                                                        it * 4096),
          16, 0, 0);
    }
  };

  // tr_b16 fragment read (packed subtiles, stride 1024)
  auto tr_frag = [&](const char* tile, int colsub, int mc) {
    const int s = lane & 15, gq_ = lane >> 4;
    const int base_off = (mc * 4 + colsub) * RSUBT + (gq_ * 8 + (s >> 2)) * 32 +
                         (s & 3) * 8;
    const unsigned a0 = (unsigned)(unsigned long long)(tile + base_off);
    const unsigned a1 = a0 + 128;  // +4 m-rows
    union {
      uint2 hh[2];
      bf16x8w vv;
    } uu;
    asm volatile(
        "ds_read_b64_tr_b16 %0, %2\n\t"
        "ds_read_b64_tr_b16 %1, %3\n\t"
        "s_waitcnt lgkmcnt(0)"
        : "=&v"(uu.hh[0]), "=&v"(uu.hh[1])
        : "v"(a0), "v"(a1)
        : "memory");
    __builtin_amdgcn_sched_barrier(0);
    return uu.vv;
  };

  f32x4w accv[2][2] = {};

  stage(0, 0);
  if (nt > 1) stage(1, 1);
  if (nt > 2) stage(2, 2);

  int bufsel = 0;
  for (int t = 0; t < nt; ++t) {
    const int infl = (nt - 1 - t) >= 2 ? 2 : (nt - 1 - t);
    if (infl == 2) {
      RWAITVM(8);
    } else if (infl == 1) {
      RWAITVM(4);
    } else {
      RWAITVM(0);
    }
    __builtin_amdgcn_s_barrier();
    const char* ta = smem + bufsel * RSLOT;
    const char* tb = ta + RTILE;
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
      bf16x8w afrag[2], bfrag[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        afrag[mi] = tr_frag(ta, wm * 2 + mi, mc);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        bfrag[ni] = tr_frag(tb, wn * 2 + ni, mc);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          accv[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[mi], bfrag[ni], accv[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (t + 3 < nt) stage(bufsel, t + 3);
    bufsel = bufsel == 2 ? 0 : bufsel + 1;
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

// ---------------------------------------------------------------------------
// 128x128x64 ring wgrad: the 64x64 tile runs out of math to cover its
// staging (8 MFMAs per wave per k-step vs 16 tr-reads + 8 glds + 2 barriers
// — measured ~330 TF ceiling). This tile gives each of the 4 waves a 64x64
// output (32 MFMAs per k-step vs 16 tr-read pairs), staged by
// global_load_lds into packed [32 m][16 col] subtiles in a 3-slot 96 KB
// dynamic-LDS ring with counted vmcnt waits (the conv2.hip idiom).
// Host-guaranteed: Kg%128==0, M%64==0, Cg%8==0, image pre-padded when ph>0;
// rsc column tails clamp their load to col 0 and are dropped in the epilogue.
// ---------------------------------------------------------------------------
constexpr int R128_BM = 128;            // k rows per tile
constexpr int R128_BN = 128;            // rsc cols per tile
constexpr int R128_TILE = 16 * RSUBT;   // one operand tile (16 KB)
constexpr int R128_SLOT = 2 * R128_TILE;

struct Wgrad128Params {
  const __hip_bfloat16* x;   // padded [N, Hp, Wp, Ct] when ph/pw > 0
  const __hip_bfloat16* gy;  // [M, Kt]
  float* acc;                // [Kt, R*S*Cg] zeroed
  int Hp, Wp, Ct, Kt;
  int R, S, Cg, Kg;
  int sh, sw, dh, dw;
  int Ho, Wo;
  int M, RSC;
  int ktiles, ntiles, csteps;
  unsigned long long magicHoWo, magicWo;
};

__global__ __launch_bounds__(256) void conv_wgrad_ring128_kernel(
    Wgrad128Params p) {
  extern __shared__ __align__(16) char smem[];  // 3 * R128_SLOT
  const int g = blockIdx.z;
  const int ktile = blockIdx.x % p.ktiles;
  const int ntile = blockIdx.x / p.ktiles;
  const int chunk = blockIdx.y;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1, wn = wid & 1;
  const int il = lane & 15, kq = lane >> 4;

  // staging: subtile s = it*4 + wid (it<4) per operand; within a subtile
  // lane l fills bytes l*16 = element (m = l>>1, col8 = (l&1)*8) — the glds
  // lane-linear image IS the packed [32 m][16 col] subtile layout.
  int s_mloc[4], s_col[4], s_kcol[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int sub = it * 4 + wid;
    s_mloc[it] = (sub >> 3) * 32 + (lane >> 1);
    s_col[it] = (sub & 7) * 16 + (lane & 1) * 8;
    int kc = ktile * R128_BM + s_col[it];
    if (kc >= p.Kg) kc = 0;  // Kg tail: its output row is discarded
    s_kcol[it] = kc;
  }
  const int SCg = p.S * p.Cg;
  int b_r[4], b_s[4], b_c[4];
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    int rsc = ntile * R128_BN + s_col[it];
    if (rsc >= p.RSC) rsc = 0;  // clamped tail: its output col is discarded
    b_r[it] = rsc / SCg;
    const int rem = rsc - b_r[it] * SCg;
    b_s[it] = rem / p.Cg;
    b_c[it] = rem - b_s[it] * p.Cg;
  }
  const int m0 = chunk * (p.csteps * WBK);
  const int HoWo = p.Ho * p.Wo;
  const int nt = min(p.csteps, (int)((p.M - m0 + WBK - 1) / WBK));

  auto stage = [&](int buf, int ks) {
    char* base = smem + buf * R128_SLOT + wid * 1024;
#pragma unroll
    for (int it = 0; it < 4; ++it) {
      const int m = m0 + ks * WBK + s_mloc[it];  // < M (M%64==0)
      const __hip_bfloat16* asrc =
          p.gy + (int64_t)m * p.Kt + g * p.Kg + s_kcol[it];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)asrc,
          (__attribute__((address_space(3))) uint32_t*)(base + it * 4096), 16,
          0, 0);
      const int n = magic_div(m, p.magicHoWo);
      const int rem = m - n * HoWo;
      const int ho = magic_div(rem, p.magicWo);
      const int wo = rem - ho * p.Wo;
      const int h = ho * p.sh + b_r[it] * p.dh;
      const int w_ = wo * p.sw + b_s[it] * p.dw;
      const __hip_bfloat16* bsrc =
          p.x + (((int64_t)n * p.Hp + h) * p.Wp + w_) * p.Ct + g * p.Cg +
          b_c[it];
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)bsrc,
          (__attribute__((address_space(3))) uint32_t*)(base + R128_TILE +
                                                        it * 4096),
          16, 0, 0);
    }
  };

  f32x4w accv[4][4] = {};

  stage(0, 0);
  if (nt > 1) stage(1, 1);
  if (nt > 2) stage(2, 2);

  int bufsel = 0;
  for (int t = 0; t < nt; ++t) {
    const int infl = (nt - 1 - t) >= 2 ? 2 : (nt - 1 - t);
    if (infl == 2) {
      RWAITVM(16);
    } else if (infl == 1) {
      RWAITVM(8);
    } else {
      RWAITVM(0);
    }
    __builtin_amdgcn_s_barrier();
    const char* ta = smem + bufsel * R128_SLOT;
    const char* tb = ta + R128_TILE;
#pragma unroll
    for (int mc = 0; mc < 2; ++mc) {
      // issue all 16 tr-read pairs of this m-half, then consume under one
      // counted wait so the MFMAs overlap the remaining reads
      TrFrag fa[4], fb[4];
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const int sb = (mc * 8 + wm * 4 + mi) * RSUBT;
        const unsigned a0 = (unsigned)(unsigned long long)(ta + sb) +
                            (unsigned)((kq * 8 + ((lane & 15) >> 2)) * 32 +
                                       (lane & 3) * 8);
        tr_read_issue(a0, a0 + 128, fa[mi]);
      }
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int sb = (mc * 8 + wn * 4 + ni) * RSUBT;
        const unsigned b0 = (unsigned)(unsigned long long)(tb + sb) +
                            (unsigned)((kq * 8 + ((lane & 15) >> 2)) * 32 +
                                       (lane & 3) * 8);
        tr_read_issue(b0, b0 + 128, fb[ni]);
      }
      // LDS FIFO is in-order: 16 reads issued as fa0..3, fb0..3 (2 each);
      // lgkmcnt(6) => all fa + fb[0] complete, then drop 2 per ni step
      asm volatile("s_waitcnt lgkmcnt(6)" ::: "memory");
      __builtin_amdgcn_sched_barrier(0);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        if (ni == 1) {
          asm volatile("s_waitcnt lgkmcnt(4)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
        }
        if (ni == 2) {
          asm volatile("s_waitcnt lgkmcnt(2)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
        }
        if (ni == 3) {
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          __builtin_amdgcn_sched_barrier(0);
        }
#pragma unroll
        for (int mi = 0; mi < 4; ++mi)
          accv[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa[mi].v, fb[ni].v, accv[mi][ni], 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (t + 3 < nt) stage(bufsel, t + 3);
    bufsel = bufsel == 2 ? 0 : bufsel + 1;
  }

  // epilogue: fp32 atomic accumulate (D: col=lane&15, row=(lane>>4)*4+rr)
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int k = ktile * R128_BM + wm * 64 + mi * 16 + kq * 4 + rr;
      if (k >= p.Kg) continue;  // Kg tail (grouped shapes)
      const int64_t rowbase = (int64_t)(g * p.Kg + k) * p.RSC;
#pragma unroll
      for (int ni = 0; ni < 4; ++ni) {
        const int col = ntile * R128_BN + wn * 64 + ni * 16 + il;
        if (col < p.RSC) atomicAdd(&p.acc[rowbase + col], accv[mi][ni][rr]);
      }
    }
  }
}

template <typename T>
__global__ void wg_pad_image_kernel(const T* __restrict__ x, T* __restrict__ y,
                                 int N, int H, int W, int C, int Hp, int Wp,
                                 int ph, int pw) {
  // row-per-block-y, pack-vectorized (the grid-stride per-element form with
  // its / and % chains ran at 4.6 TB/s; see "Elementwise rules")
  constexpr int V = 16 / sizeof(T);
  using P = Pack<T, V>;
  const int cpk = C / V;
  const int row = blockIdx.y;  // n * Hp + hp
  const int n = row / Hp, hp_ = row - n * Hp;
  const int h = hp_ - ph;
  const int rowpacks = Wp * cpk;
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= rowpacks) return;
  const int wp_ = i / cpk;
  const int cp = i - wp_ * cpk;
  const int w = wp_ - pw;
  P v = {};
  if (h >= 0 && h < H && w >= 0 && w < W)
    v = reinterpret_cast<const P*>(
        x)[(((int64_t)n * H + h) * W + w) * cpk + cp];
  reinterpret_cast<P*>(y)[(int64_t)row * rowpacks + i] = v;
}

__global__ void cast_acc_kernel(const float* __restrict__ acc,
                                __hip_bfloat16* __restrict__ gw,
                                int64_t total) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x)
    gw[i] = from_f32<__hip_bfloat16>(acc[i]);
}

}  // namespace

at::Tensor conv2d_wgrad_fp32(at::Tensor gy, at::Tensor x, int64_t R, int64_t S,
                             int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                             int64_t dh, int64_t dw, int64_t groups);

at::Tensor conv2d_wgrad(at::Tensor gy, at::Tensor x, int64_t R, int64_t S,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                        int64_t dh, int64_t dw, int64_t groups) {
  CHECK_GPU(gy);
  if (gy.scalar_type() == at::kFloat)
    return conv2d_wgrad_fp32(gy, x, R, S, sh, sw, ph, pw, dh, dw, groups);
  TORCH_CHECK(gy.scalar_type() == at::kBFloat16, "wgrad: bf16/fp32 only");
  check_nhwc(gy, "gy");
  check_nhwc(x, "x");
  const int N = x.size(0), Ct = x.size(1), H = x.size(2), W = x.size(3);
  const int Kt = gy.size(1), Ho = gy.size(2), Wo = gy.size(3);
  const int Cg = Ct / groups, Kg = Kt / groups;
  TORCH_CHECK(Cg % 8 == 0 && Kg % 8 == 0, "wgrad: Cg/Kg must be multiples of 8");
  const int64_t M64 = (int64_t)N * Ho * Wo;
  const bool ring_ok = (M64 % 64 == 0) && (Kg % 64 == 0) &&
                       ((S * Cg) % 64 == 0) && (Cg % 8 == 0) &&
                       ((R * S * Cg) % 16 == 0);
  // measured: the ring variant trails the tr-staged kernel on most shapes
  // (its 8-MFMA k-steps don't cover the pipeline); keep it opt-in.
  const char* ring_env = getenv("DISTRIBUUUU_WGRAD_RING");
  const bool ring_on = ring_env && ring_env[0] == '1';
  if (ring_ok && ring_on) {
    at::Tensor xin = x;
    int Hp = H, Wp = W;
    if (ph > 0 || pw > 0) {
      Hp = H + 2 * ph;
      Wp = W + 2 * pw;
      auto xp = at::empty({(int64_t)N * Hp * Wp * Ct}, x.options());
      int64_t total = xp.numel();
      hipLaunchKernelGGL((wg_pad_image_kernel<__hip_bfloat16>),
                         dim3((Wp * Ct / 8 + 255) / 256, N * Hp), dim3(256),
                         0, cur_stream(), (const __hip_bfloat16*)x.data_ptr(),
                         (__hip_bfloat16*)xp.data_ptr(), N, H, W, Ct, Hp, Wp,
                         ph, pw);
      xin = xp;
    }
    WgradRingParams q;
    q.x = (const __hip_bfloat16*)xin.data_ptr();
    q.gy = (const __hip_bfloat16*)gy.data_ptr();
    q.Hp = Hp; q.Wp = Wp; q.Ct = Ct; q.Kt = Kt;
    q.R = R; q.S = S; q.Cg = Cg; q.Kg = Kg;
    q.sh = sh; q.sw = sw; q.dh = dh; q.dw = dw;
    q.Ho = Ho; q.Wo = Wo;
    q.M = (int)M64;
    q.RSC = R * S * Cg;
    q.ktiles = Kg / WBM;
    q.ntiles = (q.RSC + WBN - 1) / WBN;
    const int chunks = (int)((M64 + CHUNK_STEPS * WBK - 1) / (CHUNK_STEPS * WBK));
    auto accbuf2 = at::empty({(int64_t)Kt, q.RSC}, x.options().dtype(at::kFloat));
    hipMemsetAsync(accbuf2.data_ptr(), 0, accbuf2.numel() * 4, cur_stream());
    q.acc = accbuf2.data_ptr<float>();
    dim3 grid2(q.ktiles * q.ntiles, chunks, groups);
    hipLaunchKernelGGL(conv_wgrad_ring_kernel, grid2, dim3(256), 0,
                       cur_stream(), q);
    auto gw2 = at::empty({Kt, Cg, (int64_t)R, (int64_t)S},
                         x.options().memory_format(at::MemoryFormat::ChannelsLast));
    const int64_t total2 = (int64_t)Kt * q.RSC;
    hipLaunchKernelGGL(cast_acc_kernel, dim3(grid_1d(total2, 256)), dim3(256),
                       0, cur_stream(), q.acc,
                       (__hip_bfloat16*)gw2.data_ptr(), total2);
    return gw2;
  }

  // 128x128 ring kernel where it measured faster (tools/probes/convmap.py):
  // 1x1 convolutions (no pad pass, B columns are plain channels) whose
  // launch has enough tiles+chunks to fill the 256 CUs, and GROUPED deep
  // 3x3s (RegNetY 232-wide groups: the 64x64 kernel runs out of math to
  // cover its staging there too — Kg tails are clamped/dropped). Dense 3x3
  // keeps the 64x64 kernel: pad pass + tap-spanning B columns cost more
  // than the bigger tile saves.
  const int64_t blocks64 = (((int64_t)Kg + 127) / 128) *
                           (((int64_t)R * S * Cg + 127) / 128) *
                           ((M64 + CHUNK_STEPS * WBK - 1) / (CHUNK_STEPS * WBK));
  const bool dense1x1 = R == 1 && S == 1 && Kg % 128 == 0 && Cg >= 128;
  // M >= 8192: measured crossover (tools/probes/wgrad_group_ab.py) — the
  // 7x7 grouped shapes stay on the 64x64 kernel (85 vs 118 us)
  const bool grouped_deep = groups > 1 && Kg >= 96 &&
                            (int64_t)R * S * Cg >= 512 && M64 >= 8192;
  // DISTRIBUUUU_WGRAD_128: 1 forces the ring128 route (A/B probes, tests of
  // the Kg-tail path on small shapes), 0 disables it; re-read per call
  const char* f128 = getenv("DISTRIBUUUU_WGRAD_128");
  const int force128 = f128 ? atoi(f128) : -1;
  const bool gate_ok = (dense1x1 || grouped_deep) &&
                       blocks64 * groups >= 320;
  if (force128 != 0 && M64 % 64 == 0 && Cg % 8 == 0 &&
      (gate_ok || force128 == 1)) {
    at::Tensor xin = x;
    int Hp = H, Wp = W;
    if (ph > 0 || pw > 0) {
      Hp = H + 2 * ph;
      Wp = W + 2 * pw;
      auto xp = at::empty({(int64_t)N * Hp * Wp * Ct}, x.options());
      int64_t total = xp.numel();
      hipLaunchKernelGGL((wg_pad_image_kernel<__hip_bfloat16>),
                         dim3((Wp * Ct / 8 + 255) / 256, N * Hp), dim3(256),
                         0, cur_stream(), (const __hip_bfloat16*)x.data_ptr(),
                         (__hip_bfloat16*)xp.data_ptr(), N, H, W, Ct, Hp, Wp,
                         ph, pw);
      xin = xp;
    }
    Wgrad128Params q;
    q.x = (const __hip_bfloat16*)xin.data_ptr();
    q.gy = (const __hip_bfloat16*)gy.data_ptr();
    q.Hp = Hp; q.Wp = Wp; q.Ct = Ct; q.Kt = Kt;
    q.R = R; q.S = S; q.Cg = Cg; q.Kg = Kg;
    q.sh = sh; q.sw = sw; q.dh = dh; q.dw = dw;
    q.Ho = Ho; q.Wo = Wo;
    q.M = (int)M64;
    q.RSC = R * S * Cg;
    q.ktiles = (Kg + R128_BM - 1) / R128_BM;
    q.ntiles = (q.RSC + R128_BN - 1) / R128_BN;
    q.magicHoWo = ((1ULL << 47) / ((unsigned long long)Ho * Wo)) + 1;
    q.magicWo = ((1ULL << 47) / (unsigned long long)Wo) + 1;
    // shrink the split-m chunk until the launch fills the 256 CUs
    int csteps = CHUNK_STEPS;
    int chunks = (int)((M64 + (int64_t)csteps * WBK - 1) / (csteps * WBK));
    while (csteps > 8 && (int64_t)q.ktiles * q.ntiles * chunks * groups < 512) {
      csteps /= 2;
      chunks = (int)((M64 + (int64_t)csteps * WBK - 1) / (csteps * WBK));
    }
    q.csteps = csteps;
    auto accbuf = at::empty({(int64_t)Kt, q.RSC},
                            x.options().dtype(at::kFloat));
    hipMemsetAsync(accbuf.data_ptr(), 0, accbuf.numel() * 4, cur_stream());
    q.acc = accbuf.data_ptr<float>();
    static bool attr_done = false;
    if (!attr_done) {
      hipFuncSetAttribute((const void*)conv_wgrad_ring128_kernel,
                          hipFuncAttributeMaxDynamicSharedMemorySize,
                          3 * R128_SLOT);
      attr_done = true;
    }
    dim3 grid(q.ktiles * q.ntiles, chunks, groups);
    hipLaunchKernelGGL(conv_wgrad_ring128_kernel, grid, dim3(256),
                       3 * R128_SLOT, cur_stream(), q);
    auto gw = at::empty({Kt, Cg, (int64_t)R, (int64_t)S},
                        x.options().memory_format(at::MemoryFormat::ChannelsLast));
    const int64_t total = (int64_t)Kt * q.RSC;
    hipLaunchKernelGGL(cast_acc_kernel, dim3(grid_1d(total, 256)), dim3(256),
                       0, cur_stream(), q.acc, (__hip_bfloat16*)gw.data_ptr(),
                       total);
    return gw;
  }

  // 128x128 tr-staged tile (A/B via DISTRIBUUUU_WGRAD_T128: 1 force,
  // 0 off, default auto-gate from the probe measurements)
  {
    const char* e = getenv("DISTRIBUUUU_WGRAD_T128");
    const int t128 = e ? atoi(e) : -1;
    const int64_t t128_blocks =
        (int64_t)((Kg + 127) / 128) * (((int64_t)R * S * Cg + 127) / 128) *
        ((M64 + CHUNK_STEPS * WBK - 1) / (CHUNK_STEPS * WBK)) * groups;
    // measured (tools/probes/wgrad_dense_ab.py): wins/ties every dense 3x3
    // with Kg >= 128 (512-deep: -13%), loses at Kg = 64 (half-empty tiles)
    const bool t128_gate = Kg >= 128 && (int64_t)R * S * Cg >= 128;
    if (t128 == 1 || (t128 == -1 && t128_gate)) {
      WgradT128Params q;
      q.x = (const __hip_bfloat16*)x.data_ptr();
      q.gy = (const __hip_bfloat16*)gy.data_ptr();
      q.N = N; q.H = H; q.W = W; q.Ct = Ct; q.Kt = Kt;
      q.R = R; q.S = S; q.Cg = Cg; q.Kg = Kg;
      q.sh = sh; q.sw = sw; q.ph = ph; q.pw = pw; q.dh = dh; q.dw = dw;
      q.Ho = Ho; q.Wo = Wo;
      q.M = (int)M64;
      q.RSC = R * S * Cg;
      q.ktiles = (Kg + T128_BM - 1) / T128_BM;
      q.ntiles = (q.RSC + T128_BN - 1) / T128_BN;
      {
        const char* xr = getenv("DISTRIBUUUU_WGRAD_XCD");
        q.xcd_remap = xr && xr[0] == '0' ? 0 : 1;
      }
      q.magicHoWo = ((1ULL << 47) / ((unsigned long long)Ho * Wo)) + 1;
      q.magicWo = ((1ULL << 47) / (unsigned long long)Wo) + 1;
      int csteps = CHUNK_STEPS;
      int chunks = (int)((M64 + (int64_t)csteps * WBK - 1) / (csteps * WBK));
      while (csteps > 8 &&
             (int64_t)q.ktiles * q.ntiles * chunks * groups < 512) {
        csteps /= 2;
        chunks = (int)((M64 + (int64_t)csteps * WBK - 1) / (csteps * WBK));
      }
      q.csteps = csteps;
      q.chunks = chunks;
      auto accb = at::empty({(int64_t)Kt, q.RSC},
                            x.options().dtype(at::kFloat));
      hipMemsetAsync(accb.data_ptr(), 0, accb.numel() * 4, cur_stream());
      q.acc = accb.data_ptr<float>();
      dim3 grid(q.ktiles * q.ntiles, chunks, groups);
      hipLaunchKernelGGL(conv_wgrad_tr128_kernel, grid, dim3(256), 0,
                         cur_stream(), q);
      auto gw = at::empty({Kt, Cg, (int64_t)R, (int64_t)S},
                          x.options().memory_format(
                              at::MemoryFormat::ChannelsLast));
      const int64_t tot = (int64_t)Kt * q.RSC;
      hipLaunchKernelGGL(cast_acc_kernel, dim3(grid_1d(tot, 256)), dim3(256),
                         0, cur_stream(), q.acc,
                         (__hip_bfloat16*)gw.data_ptr(), tot);
      return gw;
    }
    (void)t128_blocks;
  }

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
  p.magicHoWo = ((1ULL << 47) / ((unsigned long long)Ho * Wo)) + 1;
  p.magicWo = ((1ULL << 47) / (unsigned long long)Wo) + 1;

  auto accbuf = at::empty({(int64_t)Kt, p.RSC}, x.options().dtype(at::kFloat));
  hipMemsetAsync(accbuf.data_ptr(), 0, accbuf.numel() * 4, cur_stream());
  p.acc = accbuf.data_ptr<float>();
  dim3 grid(p.ktiles * p.ntiles, p.chunks, groups);
  if (p.ktiles * p.ntiles <= 12)  // XCD-grouped 1-D grid (see kernel)
    grid = dim3(p.ktiles * p.ntiles * ((p.chunks + 7) / 8 * 8), 1, groups);
  hipLaunchKernelGGL(conv_wgrad_kernel, grid, dim3(256), 0, cur_stream(), p);

  auto gw = at::empty({Kt, Cg, (int64_t)R, (int64_t)S},
                      x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const int64_t total = (int64_t)Kt * p.RSC;
  hipLaunchKernelGGL(cast_acc_kernel, dim3(grid_1d(total, 256)), dim3(256), 0,
                     cur_stream(), p.acc, (__hip_bfloat16*)gw.data_ptr(),
                     total);
  return gw;
}
