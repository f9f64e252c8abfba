// Fused BoTNet MHSA forward: per (batch*head, q-tile)
// block computes S = q@k^T + rel-pos logits, row softmax, O = P@v — one
// kernel, score tile resident in LDS (L = H*W = 196 fits; SURVEY.md §5.7:
// no sequence partitioning warranted).
//
// Inputs (bf16): q pre-scaled [B, L, D], k [B, L, D], vT [B, D, L] (the v
// operand is consumed l-contiguously as the MFMA B operand), plus the
// per-row relative-logit tables RW = q@rel_w^T [B, L, 2W-1] and
// RH = q@rel_h^T [B, L, 2H-1] (fp32, computed by two small gemm_nt calls).
// rel logits: S[qi, kj] += RW[qi][wj-wi+W-1] + RH[qi][hj-hi+H-1].
//
// Output: O [B, L, D] bf16; when a pout tensor is passed, the softmax
// probs P [B, L, L] are also written (bf16) so backward is a short
// analytic bmm chain instead of a full torch recompute (at L = 196 the
// probs are 39 MB/layer — "flash-style never materialize" buys nothing).
#include "common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4a;
typedef __bf16 bf16x8a __attribute__((ext_vector_type(8)));

namespace {

DEV_INLINE bf16x8a zero8() {
  union { uint4 u; bf16x8a v; } z;
  z.u = uint4{0, 0, 0, 0};
  return z.v;
}

constexpr int QT = 64;  // q rows per block

// q/k/v/o rows address as base + ((n*L + l) * pix + hd * D): with heads=1
// and pix=D this is the dense [B, L, D] layout; with heads>1 it reads the
// qkv convs' channels_last output IN PLACE ([N, L, heads*D] physical — no
// chunk/transpose/contiguous copies) and writes o back the same way.
// `scale` folds the q pre-scale into the logits (and rel tables).
struct MhsaParams {
  const __hip_bfloat16* q;
  const __hip_bfloat16* k;
  const __hip_bfloat16* vt;  // [B, D, L] (transposed operand, dense)
  const float* rw;           // [B, L, 2W-1]
  const float* rh;           // [B, L, 2H-1]
  __hip_bfloat16* o;
  __hip_bfloat16* pout;      // optional [B, L, L] softmax probs (backward)
  int B, L, D, H, W;
  int heads;
  int64_t qpix, kpix, vpix, opix;  // elements per pixel in each tensor
  float scale;
  int ltiles16;  // ceil(L/16)
  int lpad;      // LDS row width for S (multiple of 16 + pad)
};

DEV_INLINE int64_t strided_row(int b, int l, int L, int heads, int64_t pix,
                               int D) {
  const int n = b / heads, hd = b - n * heads;
  return ((int64_t)n * L + l) * pix + (int64_t)hd * D;
}

// Logits live in REGISTERS (LT tiles x 4 rows per lane); LDS holds only the
// bf16 P tile for the O pass. The round-1 version kept a [QT][lpad] fp32
// score slab in LDS (~86 KB -> 1 block/CU -> 1 wave/SIMD, nothing to hide
// latency with); register logits cut LDS to 28 KB -> 5 blocks/CU.
template <int LT>
__global__ __launch_bounds__(256) void mhsa_fwd_kernel(MhsaParams p) {
  const int b = blockIdx.x;
  const int q0 = blockIdx.y * QT;
  extern __shared__ __align__(16) char smem[];
  __hip_bfloat16* P = reinterpret_cast<__hip_bfloat16*>(smem);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // wave -> 16 q-rows
  const int il = lane & 15, kq = lane >> 4;
  const int qrow_base = q0 + wid * 16;

  // ---- load this wave's q fragments (A operand, reused across k tiles) ----
  const int nd = p.D / 32;
  bf16x8a qfrag[4];
  {
    const int qr = qrow_base + il;
    const bool ok = qr < p.L;
    const __hip_bfloat16* qp =
        p.q + strided_row(b, ok ? qr : 0, p.L, p.heads, p.qpix, p.D);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (c < nd && ok)
        qfrag[c] = *reinterpret_cast<const bf16x8a*>(qp + c * 32 + kq * 8);
      else
        qfrag[c] = zero8();
    }
  }

  // ---- S = q@k^T + rel in registers ---------------------------------------
  // D layout per tile: col kpos = nt*16 + il, row = kq*4 + rr
  f32x4a sacc[LT];
#pragma unroll
  for (int nt = 0; nt < LT; ++nt) {
    f32x4a acc = {0.f, 0.f, 0.f, 0.f};
    if (nt < p.ltiles16) {
      const int kr = nt * 16 + il;
      const bool kok = kr < p.L;
      const __hip_bfloat16* kp =
          p.k + strided_row(b, kok ? kr : 0, p.L, p.heads, p.kpix, p.D);
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        if (c >= nd) break;
        bf16x8a kfrag =
            kok ? *reinterpret_cast<const bf16x8a*>(kp + c * 32 + kq * 8)
                : zero8();
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[c], kfrag, acc, 0,
                                                      0, 0);
      }
    }
    sacc[nt] = acc;
  }
  // rel-logit add + L-tail mask (per lane: 4 rows x LT cols)
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int qr = qrow_base + kq * 4 + rr;
    const bool qok = qr < p.L;
    const int wi = qok ? qr % p.W : 0;
    const int hi = qok ? qr / p.W : 0;
    const float* rwp = p.rw + ((int64_t)b * p.L + (qok ? qr : 0)) * (2 * p.W - 1);
    const float* rhp = p.rh + ((int64_t)b * p.L + (qok ? qr : 0)) * (2 * p.H - 1);
#pragma unroll
    for (int nt = 0; nt < LT; ++nt) {
      const int kpos = nt * 16 + il;
      float v = sacc[nt][rr] * p.scale;
      if (qok && kpos < p.L) {
        const int dy = (kpos % p.W) - wi + p.W - 1;
        const int dx = (kpos / p.W) - hi + p.H - 1;
        v += rwp[dy] + rhp[dx];
      } else {
        v = -INFINITY;
      }
      sacc[nt][rr] = v;
    }
  }

  // ---- register softmax: reduce over the 16 il-lanes of each kq group -----
  float rmax[4], rsum[4];
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    float m = -INFINITY;
#pragma unroll
    for (int nt = 0; nt < LT; ++nt) m = fmaxf(m, sacc[nt][rr]);
#pragma unroll
    for (int s = 1; s < 16; s <<= 1) m = fmaxf(m, __shfl_xor(m, s));
    rmax[rr] = m;
  }
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int qr = qrow_base + kq * 4 + rr;
    float s = 0.f;
#pragma unroll
    for (int nt = 0; nt < LT; ++nt) {
      const float e = (qr < p.L && nt * 16 + il < p.L)
                          ? __expf(sacc[nt][rr] - rmax[rr]) : 0.f;
      sacc[nt][rr] = e;  // exp stored back (0 for pad cols/garbage rows)
      s += e;
    }
#pragma unroll
    for (int st = 1; st < 16; st <<= 1) s += __shfl_xor(s, st);
    rsum[rr] = (qr < p.L) ? 1.f / s : 0.f;
  }
  // write P to LDS (+ optional global), zero-fill the lpad tail
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int row = wid * 16 + kq * 4 + rr;
    const int qr = q0 + row;
    __hip_bfloat16* prow = &P[row * p.lpad];
    __hip_bfloat16* gp = (p.pout != nullptr && qr < p.L)
                             ? p.pout + ((int64_t)b * p.L + qr) * p.L
                             : nullptr;
#pragma unroll
    for (int nt = 0; nt < LT; ++nt) {
      const int kpos = nt * 16 + il;
      const __hip_bfloat16 pv =
          from_f32<__hip_bfloat16>(sacc[nt][rr] * rsum[rr]);
      prow[kpos] = pv;
      if (gp && kpos < p.L) gp[kpos] = pv;
    }
    // lpad tail beyond LT*16 (LT*16 >= ltiles16*16 covers it; pad cols in
    // written tiles already got exp=0)
    for (int j = LT * 16 + il; j < p.lpad; j += 16)
      prow[j] = from_f32<__hip_bfloat16>(0.f);
  }
  __builtin_amdgcn_wave_barrier();

  // ---- O = P @ vT: A = P rows (LDS, l-contiguous), B = vT rows ------------
  const int lchunks = p.lpad / 32;
  const int dtiles = p.D / 16;
#pragma unroll 1
  for (int dt = 0; dt < dtiles; ++dt) {
    f32x4a acc = {0.f, 0.f, 0.f, 0.f};
    const int dr = dt * 16 + il;  // vT row (= output channel)
    const __hip_bfloat16* vp = p.vt + ((int64_t)b * p.D + dr) * p.L;
    for (int lc = 0; lc < lchunks; ++lc) {
      bf16x8a pfrag = *reinterpret_cast<const bf16x8a*>(
          &P[(wid * 16 + il) * p.lpad + lc * 32 + kq * 8]);
      // vT: need B[n=d][kk=l] = vt[dr][lc*32 + kq*8 .. +8]; pad l>=L -> P=0
      const int l0 = lc * 32 + kq * 8;
      bf16x8a vfrag;
      if (l0 + 8 <= p.L) {
        vfrag = *reinterpret_cast<const bf16x8a*>(vp + l0);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          vfrag[j] = (l0 + j < p.L) ? (__bf16)vp[l0 + j] : __bf16(0.f);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pfrag, vfrag, acc, 0, 0,
                                                    0);
    }
    // D layout: col d = dt*16 + lane%16, row q = (lane/16)*4+rr
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int qr = qrow_base + kq * 4 + rr;
      if (qr < p.L)
        p.o[strided_row(b, qr, p.L, p.heads, p.opix, p.D) + dt * 16 + il] =
            from_f32<__hip_bfloat16>(acc[rr]);
    }
  }
}

// ---------------------------------------------------------------------------
// Backward (SURVEY.md K13/K14 backward): three kernels replace the round-1
// ATen bmm/elementwise chain (~3 ms/step on BoTNet-50).
//
// mhsa_bwd_q_kernel — per (b, q-tile): dP = dO @ v^T (MFMA, like the forward
//   S pass), dS = P (dP - rowdot) with rowdot a wave reduction, dQ = dS @ kT
//   (like the forward O pass) with the relative-position contribution
//   dQ += dRW @ rel_w + dRH @ rel_h folded into the same epilogue; the
//   per-row dRW/dRH gather-sums come straight out of the LDS dS tile.
// mhsa_bwd_kv_kernel — per (b, k-tile): dK = dS^T q, dV = P^T dO by VALU
//   outer-product accumulation over 16-row q chunks staged in LDS (total
//   FLOPs are tiny next to the q-side GEMMs; MFMA transposes buy nothing).
// mhsa_rel_grad_kernel — grw[m, d] = sum_rows dRW[row, m] q[row, d] via
//   per-block register accumulation + one atomic flush.
// ---------------------------------------------------------------------------
struct MhsaBwdQParams {
  const __hip_bfloat16* dO;  // strided rows (opix)
  const __hip_bfloat16* P;   // [B, L, L]
  const __hip_bfloat16* v;   // strided rows (vpix; B-operand of dP)
  const __hip_bfloat16* kt;  // [B, D, L] (B-operand of dQ, dense)
  const __hip_bfloat16* rw;  // rel_w [2W-1, D]
  const __hip_bfloat16* rh;  // rel_h [2H-1, D]
  __hip_bfloat16* dq;        // strided rows (qpix)
  __hip_bfloat16* ds;        // [B, L, L]
  float* drw;                // [B, L, 2W-1]
  float* drh;                // [B, L, 2H-1]
  int B, L, D, H, W;
  int heads;
  int64_t qpix, vpix, opix;
  float scale;               // dq = scale * (dS kT + rel fold)
  int ltiles16, lpad;
};

template <int LT>
__global__ __launch_bounds__(256) void mhsa_bwd_q_kernel(MhsaBwdQParams p) {
  const int b = blockIdx.x;
  const int q0 = blockIdx.y * QT;
  extern __shared__ __align__(16) char smem[];
  __hip_bfloat16* DS = reinterpret_cast<__hip_bfloat16*>(smem);
  // rel scratch in bf16: the LDS copy only feeds the dq fold (the fp32
  // copies for the rel-grad kernel go to global); 8 KB saved puts the
  // block at 4 waves-groups/CU instead of 3
  __hip_bfloat16* RWs =
      reinterpret_cast<__hip_bfloat16*>(smem + QT * p.lpad * 2);  // [QT][32]
  __hip_bfloat16* RHs = RWs + QT * 32;                            // [QT][32]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int il = lane & 15, kq = lane >> 4;
  const int qrow_base = q0 + wid * 16;
  const int nd = p.D / 32;

  // A fragments of dO (like the forward's qfrag)
  bf16x8a dofrag[4];
  {
    const int qr = qrow_base + il;
    const bool ok = qr < p.L;
    const __hip_bfloat16* dop =
        p.dO + strided_row(b, ok ? qr : 0, p.L, p.heads, p.opix, p.D);
#pragma unroll
    for (int c = 0; c < 4; ++c)
      dofrag[c] = (c < nd && ok)
                      ? *reinterpret_cast<const bf16x8a*>(dop + c * 32 + kq * 8)
                      : zero8();
  }

  // ---- dP = dO @ v^T in registers (same layout as the forward logits) ----
  f32x4a dacc[LT];
#pragma unroll
  for (int nt = 0; nt < LT; ++nt) {
    f32x4a acc = {0.f, 0.f, 0.f, 0.f};
    if (nt < p.ltiles16) {
      const int kr = nt * 16 + il;
      const bool kok = kr < p.L;
      const __hip_bfloat16* vp =
          p.v + strided_row(b, kok ? kr : 0, p.L, p.heads, p.vpix, p.D);
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        if (c >= nd) break;
        bf16x8a vfrag =
            kok ? *reinterpret_cast<const bf16x8a*>(vp + c * 32 + kq * 8)
                : zero8();
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dofrag[c], vfrag, acc,
                                                      0, 0, 0);
      }
    }
    dacc[nt] = acc;
  }

  // ---- rowdot = sum(dP * P) via register pass + lane shuffles -------------
  float dot[4];
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int qr = qrow_base + kq * 4 + rr;
    const __hip_bfloat16* prow =
        p.P + ((int64_t)b * p.L + (qr < p.L ? qr : 0)) * p.L;
    float s = 0.f;
#pragma unroll
    for (int nt = 0; nt < LT; ++nt) {
      const int kpos = nt * 16 + il;
      if (qr < p.L && kpos < p.L)
        s += dacc[nt][rr] * to_f32(prow[kpos]);
    }
#pragma unroll
    for (int st = 1; st < 16; st <<= 1) s += __shfl_xor(s, st);
    dot[rr] = s;
  }

  // ---- dS = P (dP - dot): write LDS bf16 + global (P re-read from L1) -----
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const int row = wid * 16 + kq * 4 + rr;
    const int qr = q0 + row;
    const __hip_bfloat16* prow =
        p.P + ((int64_t)b * p.L + (qr < p.L ? qr : 0)) * p.L;
    __hip_bfloat16* dsrow = &DS[row * p.lpad];
    __hip_bfloat16* gds =
        (qr < p.L) ? p.ds + ((int64_t)b * p.L + qr) * p.L : nullptr;
#pragma unroll
    for (int nt = 0; nt < LT; ++nt) {
      const int kpos = nt * 16 + il;
      const bool ok = qr < p.L && kpos < p.L;
      const float dsv =
          ok ? to_f32(prow[kpos]) * (dacc[nt][rr] - dot[rr]) : 0.f;
      const __hip_bfloat16 dsb = from_f32<__hip_bfloat16>(dsv);
      dsrow[kpos] = dsb;
      if (gds && kpos < p.L) gds[kpos] = dsb;
    }
    for (int j = LT * 16 + il; j < p.lpad; j += 16)
      dsrow[j] = from_f32<__hip_bfloat16>(0.f);
  }
  __builtin_amdgcn_wave_barrier();

  // ---- dRW / dRH row gather-sums from the LDS dS tile ---------------------
  // dRW[qr][m] = sum_h' dS[qr][h'*W + (m + wi - (W-1))]; dRH mirrored
  for (int r = 0; r < 16; ++r) {
    const int qr = qrow_base + r;
    const __hip_bfloat16* dsrow = &DS[(wid * 16 + r) * p.lpad];
    const int wi = (qr < p.L) ? qr % p.W : 0;
    const int hi = (qr < p.L) ? qr / p.W : 0;
    if (lane < 2 * p.W - 1) {
      const int wj = lane + wi - (p.W - 1);
      float s = 0.f;
      if (qr < p.L && wj >= 0 && wj < p.W)
        for (int hj = 0; hj < p.H; ++hj) s += to_f32(dsrow[hj * p.W + wj]);
      RWs[(wid * 16 + r) * 32 + lane] = from_f32<__hip_bfloat16>(s);
      if (qr < p.L) p.drw[((int64_t)b * p.L + qr) * (2 * p.W - 1) + lane] = s;
    }
    if (lane < 2 * p.H - 1) {
      const int hj = lane + hi - (p.H - 1);
      float s = 0.f;
      if (qr < p.L && hj >= 0 && hj < p.H)
        for (int wj = 0; wj < p.W; ++wj) s += to_f32(dsrow[hj * p.W + wj]);
      RHs[(wid * 16 + r) * 32 + lane] = from_f32<__hip_bfloat16>(s);
      if (qr < p.L) p.drh[((int64_t)b * p.L + qr) * (2 * p.H - 1) + lane] = s;
    }
  }
  __builtin_amdgcn_wave_barrier();

  // ---- dQ = dS @ kT + rel fold --------------------------------------------
  const int lchunks = p.lpad / 32;
  const int dtiles = p.D / 16;
#pragma unroll 1
  for (int dt = 0; dt < dtiles; ++dt) {
    f32x4a acc = {0.f, 0.f, 0.f, 0.f};
    const int dr = dt * 16 + il;
    const __hip_bfloat16* kp = p.kt + ((int64_t)b * p.D + dr) * p.L;
    for (int lc = 0; lc < lchunks; ++lc) {
      bf16x8a dsfrag = *reinterpret_cast<const bf16x8a*>(
          &DS[(wid * 16 + il) * p.lpad + lc * 32 + kq * 8]);
      const int l0 = lc * 32 + kq * 8;
      bf16x8a kfrag;
      if (l0 + 8 <= p.L) {
        kfrag = *reinterpret_cast<const bf16x8a*>(kp + l0);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          kfrag[j] = (l0 + j < p.L) ? (__bf16)kp[l0 + j] : __bf16(0.f);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dsfrag, kfrag, acc, 0, 0,
                                                    0);
    }
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int rloc = kq * 4 + rr;
      const int qr = qrow_base + rloc;
      if (qr < p.L) {
        const int d = dt * 16 + il;
        float out = acc[rr];
        const __hip_bfloat16* rws = &RWs[(wid * 16 + rloc) * 32];
        const __hip_bfloat16* rhs = &RHs[(wid * 16 + rloc) * 32];
        for (int m = 0; m < 2 * p.W - 1; ++m)
          out += to_f32(rws[m]) * to_f32(p.rw[m * p.D + d]);
        for (int m = 0; m < 2 * p.H - 1; ++m)
          out += to_f32(rhs[m]) * to_f32(p.rh[m * p.D + d]);
        p.dq[strided_row(b, qr, p.L, p.heads, p.qpix, p.D) + d] =
            from_f32<__hip_bfloat16>(out * p.scale);
      }
    }
  }
}

// ---- dK = dS^T q, dV = P^T dO ---------------------------------------------
struct MhsaBwdKVParams {
  const __hip_bfloat16* ds;  // [B, L, L]
  const __hip_bfloat16* q;   // strided rows (qpix)
  const __hip_bfloat16* P;   // [B, L, L]
  const __hip_bfloat16* dO;  // strided rows (opix)
  __hip_bfloat16* dk;        // strided rows (kpix)
  __hip_bfloat16* dv;        // strided rows (vpix)
  int B, L, D;
  int heads;
  int64_t qpix, kpix, vpix, opix;
  float scale;               // dk = scale * (dS^T q)
};

__global__ __launch_bounds__(256) void mhsa_bwd_kv_kernel(MhsaBwdKVParams p) {
  const int b = blockIdx.x;
  const int k0 = blockIdx.y * 64;
  // LDS: q16/do16 staged as FP32 [16][D] (the bf16 form cost 64
  // v_cvt per j-iteration in the FMA loop); ds16/p16 stay bf16
  __shared__ float q16[16][128];
  __shared__ float do16[16][128];
  __shared__ __hip_bfloat16 ds16[16][64];
  __shared__ __hip_bfloat16 p16[16][64];
  const int tid = threadIdx.x;
  const int kj = k0 + (tid >> 2);          // this thread's k row
  const int d0 = (tid & 3) * 32;           // 32 d-columns
  float acck[32] = {}, accv[32] = {};
  const int nchunk = (p.L + 15) / 16;
  // double-buffered staging through registers: chunk ch+1's global loads
  // are issued right after the barrier and complete under chunk ch's
  // compute loop (the write->barrier->load->barrier form exposed the full
  // global latency every chunk: 397 us/call for a ~33 us-of-traffic op)
  bf16x8a rq, rdo, rds, rp;
  auto stage_issue = [&](int ch) {
    {
      const int r = tid >> 4, c8 = (tid & 15) << 3;
      const int qi = ch * 16 + r;
      const bool ok = qi < p.L && c8 < p.D;
      const int qic = qi < p.L ? qi : 0;
      rq = ok ? *reinterpret_cast<const bf16x8a*>(
                    p.q + strided_row(b, qic, p.L, p.heads, p.qpix, p.D) + c8)
              : zero8();
      rdo = ok ? *reinterpret_cast<const bf16x8a*>(
                     p.dO + strided_row(b, qic, p.L, p.heads, p.opix, p.D) +
                     c8)
               : zero8();
    }
    if (tid < 128) {
      const int r = tid >> 3, c8 = (tid & 7) << 3;
      const int qi = ch * 16 + r;
      const int64_t soff = ((int64_t)b * p.L + (qi < p.L ? qi : 0)) * p.L + k0;
      rds = zero8();
      rp = zero8();
      if (qi < p.L) {
        if (k0 + c8 + 8 <= p.L) {
          rds = *reinterpret_cast<const bf16x8a*>(p.ds + soff + c8);
          rp = *reinterpret_cast<const bf16x8a*>(p.P + soff + c8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            rds[j] = (k0 + c8 + j < p.L) ? (__bf16)p.ds[soff + c8 + j]
                                         : __bf16(0.f);
            rp[j] = (k0 + c8 + j < p.L) ? (__bf16)p.P[soff + c8 + j]
                                        : __bf16(0.f);
          }
        }
      }
    }
  };
  auto stage_write = [&]() {
    {
      const int r = tid >> 4, c8 = (tid & 15) << 3;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        q16[r][c8 + j] = to_f32((__hip_bfloat16)rq[j]);
        do16[r][c8 + j] = to_f32((__hip_bfloat16)rdo[j]);
      }
    }
    if (tid < 128) {
      const int r = tid >> 3, c8 = (tid & 7) << 3;
      *reinterpret_cast<bf16x8a*>(&ds16[r][c8]) = rds;
      *reinterpret_cast<bf16x8a*>(&p16[r][c8]) = rp;
    }
  };
  stage_issue(0);
  for (int ch = 0; ch < nchunk; ++ch) {
    stage_write();
    __syncthreads();
    if (ch + 1 < nchunk) stage_issue(ch + 1);
    const int jmax = min(16, p.L - ch * 16);
    for (int j = 0; j < jmax; ++j) {
      const float sv = to_f32(ds16[j][kj - k0]);
      const float pvv = to_f32(p16[j][kj - k0]);
      // vectorized fp32 LDS reads of the 32-wide d slice
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        const float4 qv = *reinterpret_cast<const float4*>(&q16[j][d0 + c * 4]);
        const float4 dov =
            *reinterpret_cast<const float4*>(&do16[j][d0 + c * 4]);
        acck[c * 4 + 0] += sv * qv.x;
        acck[c * 4 + 1] += sv * qv.y;
        acck[c * 4 + 2] += sv * qv.z;
        acck[c * 4 + 3] += sv * qv.w;
        accv[c * 4 + 0] += pvv * dov.x;
        accv[c * 4 + 1] += pvv * dov.y;
        accv[c * 4 + 2] += pvv * dov.z;
        accv[c * 4 + 3] += pvv * dov.w;
      }
    }
    __syncthreads();
  }
  if (kj < p.L && d0 < p.D) {
    __hip_bfloat16* dkp =
        p.dk + strided_row(b, kj, p.L, p.heads, p.kpix, p.D) + d0;
    __hip_bfloat16* dvp =
        p.dv + strided_row(b, kj, p.L, p.heads, p.vpix, p.D) + d0;
#pragma unroll
    for (int u = 0; u < 32; ++u) {
      dkp[u] = from_f32<__hip_bfloat16>(acck[u] * p.scale);
      dvp[u] = from_f32<__hip_bfloat16>(accv[u]);
    }
  }
}

// ---- grw[m, d] = sum_rows dRW[row, m] * q[row, d] -------------------------
struct MhsaRelGradParams {
  const float* dr;          // [rows, M] (dRW or dRH)
  const __hip_bfloat16* q;  // strided rows
  float* out;               // [M, D]
  int64_t rows;
  int M, D, rows_per_block;
  int heads, L;
  int64_t qpix;
  float scale;              // grw = scale * (dRW^T q)
};

__global__ __launch_bounds__(256) void mhsa_rel_grad_kernel(
    MhsaRelGradParams p) {
  // sync-free: thread (half, d) accumulates its half of the m-range for one
  // d column; q[r, d] loads coalesce across the 128 d-threads, the shared
  // dr[r, m] scalars broadcast within each wave (same address per lane
  // group). The first LDS-staged version synced twice per row and was
  // latency-bound at ~2 ms/step for a 350 MFLOP reduction.
  const int tid = threadIdx.x;
  const int d = tid & 127;
  const int half = tid >> 7;
  const int m0 = half * ((p.M + 1) / 2);
  const int mn = min(p.M - m0, (p.M + 1) / 2);
  if (d >= p.D) return;
  float acc[16] = {};
  const int64_t r0 = (int64_t)blockIdx.x * p.rows_per_block;
  const int64_t r1 = min(r0 + p.rows_per_block, p.rows);
  // division-free row->(batch, pixel) advance + 4-row load batching: the
  // one-row-at-a-time form paid an int64 div and a serial scalar-gather
  // latency per row (302 us/call for a ~36 MB reduction)
  int bb = (int)(r0 / p.L), ll = (int)(r0 - (int64_t)(r0 / p.L) * p.L);
  int64_t r = r0;
  for (; r + 3 < r1; r += 4) {
    float qv[4];
    const float* drp[4];
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      qv[u] =
          to_f32(p.q[strided_row(bb, ll, p.L, p.heads, p.qpix, p.D) + d]);
      drp[u] = p.dr + (r + u) * p.M + m0;
      if (++ll == p.L) {
        ll = 0;
        ++bb;
      }
    }
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      float v[16];
#pragma unroll
      for (int m = 0; m < 16; ++m) v[m] = (m < mn) ? drp[u][m] : 0.f;
#pragma unroll
      for (int m = 0; m < 16; ++m) acc[m] += v[m] * qv[u];
    }
  }
  for (; r < r1; ++r) {
    const float qv =
        to_f32(p.q[strided_row(bb, ll, p.L, p.heads, p.qpix, p.D) + d]);
    const float* dr = p.dr + r * p.M + m0;
#pragma unroll 7
    for (int m = 0; m < mn; ++m) acc[m] += dr[m] * qv;
    if (++ll == p.L) {
      ll = 0;
      ++bb;
    }
  }
  for (int m = 0; m < mn; ++m)
    atomicAdd(&p.out[(int64_t)(m0 + m) * p.D + d], acc[m] * p.scale);
}

// ---- per-row rel-logit tables: out[row, m] = sum_d q[row, d] rel[m, d] ----
// fp32 accumulate AND fp32 output (a bf16 GEMM output rounds the table and
// measurably shifts the softmax when logits are large).
struct RelTabParams {
  const __hip_bfloat16* q;    // strided rows (qpix/heads; dense when heads=1)
  const __hip_bfloat16* rel;  // [M, D]
  float* out;                 // [rows, M]
  int64_t rows;
  int M, D;
  int heads, L;
  int64_t qpix;
  float scale;                // tables of the PRE-SCALED q
};

__global__ __launch_bounds__(256) void mhsa_rel_tables_kernel(
    RelTabParams p) {
  __shared__ __hip_bfloat16 rel_s[32 * 128];
  __shared__ __hip_bfloat16 q_s[8][128];
  for (int i = threadIdx.x; i < p.M * p.D; i += 256) rel_s[i] = p.rel[i];
  __syncthreads();
  const int m = threadIdx.x & 31;
  const int rl = threadIdx.x >> 5;  // 8 rows per iteration
  for (int64_t row0 = (int64_t)blockIdx.x * 8; row0 < p.rows;
       row0 += (int64_t)gridDim.x * 8) {
    if (threadIdx.x < 16 * (p.D / 8)) {
      const int rr = threadIdx.x / (p.D / 8);
      const int c8 = (threadIdx.x % (p.D / 8)) * 8;
      if (rr < 8 && row0 + rr < p.rows) {
        const int64_t ro = row0 + rr;
        const int bb = (int)(ro / p.L), ll = (int)(ro - (int64_t)bb * p.L);
        *reinterpret_cast<bf16x8a*>(&q_s[rr][c8]) =
            *reinterpret_cast<const bf16x8a*>(
                p.q + strided_row(bb, ll, p.L, p.heads, p.qpix, p.D) + c8);
      }
    }
    __syncthreads();
    if (m < p.M && row0 + rl < p.rows) {
      float acc = 0.f;
      for (int dd = 0; dd < p.D; ++dd)
        acc += to_f32(q_s[rl][dd]) * to_f32(rel_s[m * p.D + dd]);
      p.out[(row0 + rl) * p.M + m] = acc * p.scale;
    }
    __syncthreads();
  }
}

}  // namespace

// rw = (scale*q) @ rel^T, fp32 out [rows, M]; q rows address as
// base + ((row/L)*L + row%L ... ) * qpix + (b%heads)*D (dense when heads=1,
// qpix=D, L=1).
at::Tensor mhsa_rel_tables(at::Tensor q, at::Tensor rel, int64_t rows,
                           int64_t heads, int64_t L, int64_t qpix,
                           double scale) {
  CHECK_GPU(q);
  const int D = rel.size(1), M = rel.size(0);
  TORCH_CHECK(M <= 32 && D <= 128 && D % 8 == 0, "rel tables: M<=32 D<=128");
  auto out = at::empty({rows, (int64_t)M}, q.options().dtype(at::kFloat));
  RelTabParams p;
  p.q = (const __hip_bfloat16*)q.data_ptr();
  p.rel = (const __hip_bfloat16*)rel.data_ptr();
  p.out = out.data_ptr<float>();
  p.rows = rows;
  p.M = M;
  p.D = D;
  p.heads = (int)heads;
  p.L = (int)L;
  p.qpix = qpix;
  p.scale = (float)scale;
  hipLaunchKernelGGL(mhsa_rel_tables_kernel,
                     dim3((int)std::min<int64_t>(ceil_div(rows, 8), 2048)),
                     dim3(256), 0, cur_stream(), p);
  return out;
}

// Backward: dO + saved (P, q, k, v) -> dq (incl. rel-pos fold), dk, dv,
// grw, grh. kt/vt are the transposed operands ([B, D, L]).
// Strided/NHWC backward: q/dO/v (and the dq/dk/dv outputs) live in the qkv
// convs' channels_last layouts; dqk_out is one [N, 2*heads*D, H, W] tensor
// receiving dq and dk halves (exactly what the qk conv's backward wants).
// heads=1 + pix=D degenerates to the dense layout.
std::vector<at::Tensor> mhsa_bwd(at::Tensor dO, at::Tensor P, at::Tensor q,
                                 at::Tensor kt, at::Tensor v,
                                 at::Tensor rel_w, at::Tensor rel_h,
                                 int64_t H, int64_t W, int64_t heads,
                                 int64_t qpix, int64_t kqoff, int64_t vpix,
                                 int64_t opix, double scale,
                                 at::Tensor dqk_out, at::Tensor dv_out) {
  CHECK_GPU(dO);
  const int B = P.size(0), L = P.size(1);
  const int D = rel_w.size(1);
  const int MW = 2 * W - 1, MH = 2 * H - 1;
  TORCH_CHECK(MW <= 32 && MH <= 32, "mhsa_bwd: rel table > 32");
  auto ds = at::empty_like(P);
  auto fopts = P.options().dtype(at::kFloat);
  auto drw = at::empty({(int64_t)B * L, MW}, fopts);
  auto drh = at::empty({(int64_t)B * L, MH}, fopts);
  const int lpad = ((L + 31) / 32) * 32;
  __hip_bfloat16* dq_ptr = (__hip_bfloat16*)dqk_out.data_ptr();
  __hip_bfloat16* dk_ptr = dq_ptr + kqoff;
  MhsaBwdQParams p;
  p.dO = (const __hip_bfloat16*)dO.data_ptr();
  p.P = (const __hip_bfloat16*)P.data_ptr();
  p.v = (const __hip_bfloat16*)v.data_ptr();
  p.kt = (const __hip_bfloat16*)kt.data_ptr();
  p.rw = (const __hip_bfloat16*)rel_w.data_ptr();
  p.rh = (const __hip_bfloat16*)rel_h.data_ptr();
  p.dq = dq_ptr;
  p.ds = (__hip_bfloat16*)ds.data_ptr();
  p.drw = drw.data_ptr<float>();
  p.drh = drh.data_ptr<float>();
  p.B = B; p.L = L; p.D = D; p.H = H; p.W = W;
  p.heads = (int)heads;
  p.qpix = qpix; p.vpix = vpix; p.opix = opix;
  p.scale = (float)scale;
  p.ltiles16 = (L + 15) / 16;
  p.lpad = lpad;
  TORCH_CHECK(p.ltiles16 <= 16, "mhsa_bwd: L must be <= 256");
  const int smem_bytes = QT * lpad * 2 + QT * 32 * 4;
  dim3 grid(B, (L + QT - 1) / QT);
  if (p.ltiles16 <= 4)
    hipLaunchKernelGGL(mhsa_bwd_q_kernel<4>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);
  else if (p.ltiles16 <= 8)
    hipLaunchKernelGGL(mhsa_bwd_q_kernel<8>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);
  else if (p.ltiles16 <= 13)
    hipLaunchKernelGGL(mhsa_bwd_q_kernel<13>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);
  else
    hipLaunchKernelGGL(mhsa_bwd_q_kernel<16>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);

  MhsaBwdKVParams kv;
  kv.ds = (const __hip_bfloat16*)ds.data_ptr();
  kv.q = (const __hip_bfloat16*)q.data_ptr();
  kv.P = (const __hip_bfloat16*)P.data_ptr();
  kv.dO = (const __hip_bfloat16*)dO.data_ptr();
  kv.dk = dk_ptr;
  kv.dv = (__hip_bfloat16*)dv_out.data_ptr();
  kv.B = B; kv.L = L; kv.D = D;
  kv.heads = (int)heads;
  kv.qpix = qpix; kv.kpix = qpix; kv.vpix = vpix; kv.opix = opix;
  kv.scale = (float)scale;
  dim3 gkv(B, (L + 63) / 64);
  hipLaunchKernelGGL(mhsa_bwd_kv_kernel, gkv, dim3(256), 0, cur_stream(), kv);

  auto grw = at::empty({MW, D}, fopts);
  auto grh = at::empty({MH, D}, fopts);
  hipMemsetAsync(grw.data_ptr(), 0, grw.numel() * 4, cur_stream());
  hipMemsetAsync(grh.data_ptr(), 0, grh.numel() * 4, cur_stream());
  const int64_t rows = (int64_t)B * L;
  const int rpb = 128;
  for (int which = 0; which < 2; ++which) {
    MhsaRelGradParams rp;
    rp.dr = which == 0 ? drw.data_ptr<float>() : drh.data_ptr<float>();
    rp.q = (const __hip_bfloat16*)q.data_ptr();
    rp.out = which == 0 ? grw.data_ptr<float>() : grh.data_ptr<float>();
    rp.rows = rows;
    rp.M = which == 0 ? MW : MH;
    rp.D = D;
    rp.rows_per_block = rpb;
    rp.heads = (int)heads;
    rp.L = L;
    rp.qpix = qpix;
    rp.scale = (float)scale;
    hipLaunchKernelGGL(mhsa_rel_grad_kernel,
                       dim3((int)ceil_div(rows, rpb)), dim3(256), 0,
                       cur_stream(), rp);
  }
  return {dqk_out, dv_out, grw, grh};
}

// Geometry from vt [B, D, L] (the one dense operand). heads=1, pix=D is the
// dense [B, L, D] layout; heads>1 reads q/k from the qkv convs'
// channels_last output in place and writes o as channels_last
// [N, heads*D, H, W]. q scale is folded into the logits.
at::Tensor mhsa_fwd(at::Tensor q, at::Tensor k, at::Tensor vt, at::Tensor rw,
                    at::Tensor rh, int64_t H, int64_t W,
                    c10::optional<at::Tensor> pout, int64_t heads,
                    int64_t qpix, int64_t kpix, double scale) {
  CHECK_GPU(q);
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "mhsa: bf16 only");
  const int B = vt.size(0), D = vt.size(1), L = vt.size(2);
  TORCH_CHECK(D % 32 == 0 && D <= 128, "mhsa: D must be <=128, %32==0");
  TORCH_CHECK(L == H * W, "L != H*W");
  const int N = B / (int)heads;
  at::Tensor o;
  int64_t opix;
  if (heads > 1) {
    o = at::empty({N, heads * D, H, W},
                  q.options().memory_format(at::MemoryFormat::ChannelsLast));
    opix = heads * D;
  } else {
    o = at::empty({B, (int64_t)L, (int64_t)D}, q.options());
    opix = D;
  }
  MhsaParams p;
  p.q = (const __hip_bfloat16*)q.data_ptr();
  p.k = (const __hip_bfloat16*)k.data_ptr();
  p.vt = (const __hip_bfloat16*)vt.data_ptr();
  p.rw = rw.data_ptr<float>();
  p.rh = rh.data_ptr<float>();
  p.o = (__hip_bfloat16*)o.data_ptr();
  p.pout = pout.has_value() ? (__hip_bfloat16*)pout->data_ptr() : nullptr;
  p.B = B; p.L = L; p.D = D; p.H = H; p.W = W;
  p.heads = (int)heads;
  p.qpix = qpix; p.kpix = kpix; p.vpix = 0; p.opix = opix;
  p.scale = (float)scale;
  p.ltiles16 = (L + 15) / 16;
  p.lpad = ((L + 31) / 32) * 32;  // multiple of 32 for the P fragment chunks
  TORCH_CHECK(p.ltiles16 <= 16, "mhsa: L must be <= 256");
  const int smem_bytes = QT * p.lpad * 2;  // P tile only (logits in regs)
  dim3 grid(B, (L + QT - 1) / QT);
  if (p.ltiles16 <= 4)
    hipLaunchKernelGGL(mhsa_fwd_kernel<4>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);
  else if (p.ltiles16 <= 8)
    hipLaunchKernelGGL(mhsa_fwd_kernel<8>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);
  else if (p.ltiles16 <= 13)
    hipLaunchKernelGGL(mhsa_fwd_kernel<13>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);
  else
    hipLaunchKernelGGL(mhsa_fwd_kernel<16>, grid, dim3(256), smem_bytes,
                       cur_stream(), p);
  return o;
}
