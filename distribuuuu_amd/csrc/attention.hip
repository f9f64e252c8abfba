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

struct MhsaParams {
  const __hip_bfloat16* q;   // [B, L, D]
  const __hip_bfloat16* k;   // [B, L, D]
  const __hip_bfloat16* vt;  // [B, D, L]
  const float* rw;           // [B, L, 2W-1]
  const float* rh;           // [B, L, 2H-1]
  __hip_bfloat16* o;         // [B, L, D]
  __hip_bfloat16* pout;      // optional [B, L, L] softmax probs (backward)
  int B, L, D, H, W;
  int ltiles16;  // ceil(L/16)
  int lpad;      // LDS row width for S (multiple of 16 + pad)
};

__global__ __launch_bounds__(256) void mhsa_fwd_kernel(MhsaParams p) {
  const int b = blockIdx.x;
  const int q0 = blockIdx.y * QT;
  extern __shared__ __align__(16) char smem[];
  // S: [QT][lpad] fp32 ; P: [QT][lpad] bf16 (after S)
  float* S = reinterpret_cast<float*>(smem);
  __hip_bfloat16* P =
      reinterpret_cast<__hip_bfloat16*>(smem + QT * p.lpad * 4);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // wave -> 16 q-rows
  const int il = lane & 15, kq = lane >> 4;
  const int qrow_base = q0 + wid * 16;

  // ---- load this wave's q fragments (A operand, reused across k tiles) ----
  // A[i = lane%16][kk = (lane/16)*8 + j], 4 kk-chunks of 32 over D=128
  const int nd = p.D / 32;
  bf16x8a qfrag[4];
  {
    const int qr = qrow_base + il;
    const bool ok = qr < p.L;
    const __hip_bfloat16* qp = p.q + ((int64_t)b * p.L + (ok ? qr : 0)) * p.D;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (c < nd && ok)
        qfrag[c] = *reinterpret_cast<const bf16x8a*>(qp + c * 32 + kq * 8);
      else
        qfrag[c] = zero8();
    }
  }

  // ---- S = q@k^T + rel, one 16-wide k tile at a time ----------------------
  for (int nt = 0; nt < p.ltiles16; ++nt) {
    f32x4a acc = {0.f, 0.f, 0.f, 0.f};
    const int kr = nt * 16 + il;
    const bool kok = kr < p.L;
    const __hip_bfloat16* kp = p.k + ((int64_t)b * p.L + (kok ? kr : 0)) * p.D;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (c >= nd) break;
      bf16x8a kfrag = kok ? *reinterpret_cast<const bf16x8a*>(kp + c * 32 + kq * 8)
                          : zero8();
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[c], kfrag, acc, 0,
                                                    0, 0);
    }
    // D layout: col kpos = nt*16 + lane%16, row qrow = (lane/16)*4 + rr
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int qr_local = kq * 4 + rr;
      const int qr = qrow_base + qr_local;
      const int kpos = nt * 16 + il;
      float v = acc[rr];
      if (qr < p.L && kpos < p.L) {
        const int dy = (kpos % p.W) - (qr % p.W) + p.W - 1;
        const int dx = (kpos / p.W) - (qr / p.W) + p.H - 1;
        v += p.rw[((int64_t)b * p.L + qr) * (2 * p.W - 1) + dy] +
             p.rh[((int64_t)b * p.L + qr) * (2 * p.H - 1) + dx];
      } else {
        v = -INFINITY;
      }
      S[(wid * 16 + qr_local) * p.lpad + kpos] = v;
    }
  }
  __builtin_amdgcn_wave_barrier();

  // ---- row softmax (wave handles its own 16 rows; S rows are private) -----
  const int lrounds = (p.lpad + 63) / 64;
  for (int r = 0; r < 16; ++r) {
    float* row = &S[(wid * 16 + r) * p.lpad];
    float m = -INFINITY;
    for (int j = lane; j < p.L; j += 64) m = fmaxf(m, row[j]);
    m = wave_reduce_max(m);
    m = __shfl(m, 0);
    float s = 0.f;
    for (int j = lane; j < p.L; j += 64) s += __expf(row[j] - m);
    s = wave_reduce_sum(s);
    s = __shfl(s, 0);
    const float inv = 1.f / s;
    __hip_bfloat16* prow = &P[(wid * 16 + r) * p.lpad];
    const int qr = qrow_base + r;
    __hip_bfloat16* gp = (p.pout != nullptr && qr < p.L)
                             ? p.pout + ((int64_t)b * p.L + qr) * p.L
                             : nullptr;
    for (int jr = 0; jr < lrounds; ++jr) {
      const int j = jr * 64 + lane;
      if (j < p.lpad) {
        const __hip_bfloat16 pv = from_f32<__hip_bfloat16>(
            j < p.L ? __expf(row[j] - m) * inv : 0.f);
        prow[j] = pv;
        if (gp && j < p.L) gp[j] = pv;
      }
    }
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
        p.o[((int64_t)b * p.L + qr) * p.D + dt * 16 + il] =
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
  const __hip_bfloat16* dO;  // [B, L, D]
  const __hip_bfloat16* P;   // [B, L, L]
  const __hip_bfloat16* v;   // [B, L, D] (rows; B-operand of dP)
  const __hip_bfloat16* kt;  // [B, D, L] (B-operand of dQ)
  const __hip_bfloat16* rw;  // rel_w [2W-1, D]
  const __hip_bfloat16* rh;  // rel_h [2H-1, D]
  __hip_bfloat16* dq;        // [B, L, D]
  __hip_bfloat16* ds;        // [B, L, L]
  float* drw;                // [B, L, 2W-1]
  float* drh;                // [B, L, 2H-1]
  int B, L, D, H, W;
  int ltiles16, lpad;
};

__global__ __launch_bounds__(256) void mhsa_bwd_q_kernel(MhsaBwdQParams p) {
  const int b = blockIdx.x;
  const int q0 = blockIdx.y * QT;
  extern __shared__ __align__(16) char smem[];
  float* S = reinterpret_cast<float*>(smem);                  // dP then free
  __hip_bfloat16* DS =
      reinterpret_cast<__hip_bfloat16*>(smem + QT * p.lpad * 4);
  float* RWs = reinterpret_cast<float*>(smem + QT * p.lpad * 6);  // [QT][32]
  float* RHs = RWs + QT * 32;                                     // [QT][32]

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
        p.dO + ((int64_t)b * p.L + (ok ? qr : 0)) * p.D;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      dofrag[c] = (c < nd && ok)
                      ? *reinterpret_cast<const bf16x8a*>(dop + c * 32 + kq * 8)
                      : zero8();
  }

  // ---- dP = dO @ v^T into the fp32 slab -----------------------------------
  for (int nt = 0; nt < p.ltiles16; ++nt) {
    f32x4a acc = {0.f, 0.f, 0.f, 0.f};
    const int kr = nt * 16 + il;
    const bool kok = kr < p.L;
    const __hip_bfloat16* vp = p.v + ((int64_t)b * p.L + (kok ? kr : 0)) * p.D;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      if (c >= nd) break;
      bf16x8a vfrag =
          kok ? *reinterpret_cast<const bf16x8a*>(vp + c * 32 + kq * 8)
              : zero8();
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dofrag[c], vfrag, acc, 0,
                                                    0, 0);
    }
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int qr_local = kq * 4 + rr;
      const int kpos = nt * 16 + il;
      S[(wid * 16 + qr_local) * p.lpad + kpos] = acc[rr];
    }
  }
  __builtin_amdgcn_wave_barrier();

  // ---- dS = P (dP - rowdot); write LDS bf16 + global ----------------------
  for (int r = 0; r < 16; ++r) {
    const int qr = qrow_base + r;
    float* row = &S[(wid * 16 + r) * p.lpad];
    const __hip_bfloat16* prow =
        p.P + ((int64_t)b * p.L + (qr < p.L ? qr : 0)) * p.L;
    float dot = 0.f;
    for (int j = lane; j < p.L; j += 64) dot += row[j] * to_f32(prow[j]);
    dot = wave_reduce_sum(dot);
    dot = __shfl(dot, 0);
    __hip_bfloat16* dsrow = &DS[(wid * 16 + r) * p.lpad];
    __hip_bfloat16* gds =
        (qr < p.L) ? p.ds + ((int64_t)b * p.L + qr) * p.L : nullptr;
    for (int j = lane; j < p.lpad; j += 64) {
      const float dsv =
          (j < p.L) ? to_f32(prow[j]) * (row[j] - dot) : 0.f;
      const __hip_bfloat16 dsb = from_f32<__hip_bfloat16>(dsv);
      dsrow[j] = dsb;
      if (gds && j < p.L) gds[j] = dsb;
    }
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
      RWs[(wid * 16 + r) * 32 + lane] = s;
      if (qr < p.L) p.drw[((int64_t)b * p.L + qr) * (2 * p.W - 1) + lane] = s;
    }
    if (lane < 2 * p.H - 1) {
      const int hj = lane + hi - (p.H - 1);
      float s = 0.f;
      if (qr < p.L && hj >= 0 && hj < p.H)
        for (int wj = 0; wj < p.W; ++wj) s += to_f32(dsrow[hj * p.W + wj]);
      RHs[(wid * 16 + r) * 32 + lane] = s;
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
        const float* rws = &RWs[(wid * 16 + rloc) * 32];
        const float* rhs = &RHs[(wid * 16 + rloc) * 32];
        for (int m = 0; m < 2 * p.W - 1; ++m)
          out += rws[m] * to_f32(p.rw[m * p.D + d]);
        for (int m = 0; m < 2 * p.H - 1; ++m)
          out += rhs[m] * to_f32(p.rh[m * p.D + d]);
        p.dq[((int64_t)b * p.L + qr) * p.D + d] =
            from_f32<__hip_bfloat16>(out);
      }
    }
  }
}

// ---- dK = dS^T q, dV = P^T dO ---------------------------------------------
struct MhsaBwdKVParams {
  const __hip_bfloat16* ds;  // [B, L, L]
  const __hip_bfloat16* q;   // [B, L, D]
  const __hip_bfloat16* P;   // [B, L, L]
  const __hip_bfloat16* dO;  // [B, L, D]
  __hip_bfloat16* dk;        // [B, L, D]
  __hip_bfloat16* dv;        // [B, L, D]
  int B, L, D;
};

__global__ __launch_bounds__(256) void mhsa_bwd_kv_kernel(MhsaBwdKVParams p) {
  const int b = blockIdx.x;
  const int k0 = blockIdx.y * 64;
  // LDS: q16/do16 [16][D], ds16/p16 [16][64]
  __shared__ __hip_bfloat16 q16[16][128];
  __shared__ __hip_bfloat16 do16[16][128];
  __shared__ __hip_bfloat16 ds16[16][64];
  __shared__ __hip_bfloat16 p16[16][64];
  const int tid = threadIdx.x;
  const int kj = k0 + (tid >> 2);          // this thread's k row
  const int d0 = (tid & 3) * 32;           // 32 d-columns
  float acck[32] = {}, accv[32] = {};
  const int nchunk = (p.L + 15) / 16;
  for (int ch = 0; ch < nchunk; ++ch) {
    // cooperative staging: thread t loads 8 elems of q/do rows (t>>4, col
    // (t&15)*8) and 8 elems of ds/p rows (t>>2 of 64... use first 128 thr)
    {
      const int r = tid >> 4, c8 = (tid & 15) << 3;
      const int qi = ch * 16 + r;
      const bool ok = qi < p.L && c8 < p.D;
      const int64_t rowoff = ((int64_t)b * p.L + (qi < p.L ? qi : 0)) * p.D;
      *reinterpret_cast<bf16x8a*>(&q16[r][c8]) =
          ok ? *reinterpret_cast<const bf16x8a*>(p.q + rowoff + c8) : zero8();
      *reinterpret_cast<bf16x8a*>(&do16[r][c8]) =
          ok ? *reinterpret_cast<const bf16x8a*>(p.dO + rowoff + c8) : zero8();
    }
    if (tid < 128) {
      const int r = tid >> 3, c8 = (tid & 7) << 3;
      const int qi = ch * 16 + r;
      const int64_t soff = ((int64_t)b * p.L + (qi < p.L ? qi : 0)) * p.L + k0;
      bf16x8a dsv = zero8(), pv = zero8();
      if (qi < p.L) {
        if (k0 + c8 + 8 <= p.L) {
          dsv = *reinterpret_cast<const bf16x8a*>(p.ds + soff + c8);
          pv = *reinterpret_cast<const bf16x8a*>(p.P + soff + c8);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            dsv[j] = (k0 + c8 + j < p.L) ? (__bf16)p.ds[soff + c8 + j]
                                         : __bf16(0.f);
            pv[j] = (k0 + c8 + j < p.L) ? (__bf16)p.P[soff + c8 + j]
                                        : __bf16(0.f);
          }
        }
      }
      *reinterpret_cast<bf16x8a*>(&ds16[r][c8]) = dsv;
      *reinterpret_cast<bf16x8a*>(&p16[r][c8]) = pv;
    }
    __syncthreads();
    const int jmax = min(16, p.L - ch * 16);
    for (int j = 0; j < jmax; ++j) {
      const float sv = to_f32(ds16[j][kj - k0]);
      const float pvv = to_f32(p16[j][kj - k0]);
      // vectorized LDS reads of the 32-wide d slice
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        bf16x8a qv = *reinterpret_cast<const bf16x8a*>(&q16[j][d0 + c * 8]);
        bf16x8a dov = *reinterpret_cast<const bf16x8a*>(&do16[j][d0 + c * 8]);
#pragma unroll
        for (int u = 0; u < 8; ++u) {
          acck[c * 8 + u] += sv * to_f32((__hip_bfloat16)qv[u]);
          accv[c * 8 + u] += pvv * to_f32((__hip_bfloat16)dov[u]);
        }
      }
    }
    __syncthreads();
  }
  if (kj < p.L && d0 < p.D) {
    __hip_bfloat16* dkp = p.dk + ((int64_t)b * p.L + kj) * p.D + d0;
    __hip_bfloat16* dvp = p.dv + ((int64_t)b * p.L + kj) * p.D + d0;
#pragma unroll
    for (int u = 0; u < 32; ++u) {
      dkp[u] = from_f32<__hip_bfloat16>(acck[u]);
      dvp[u] = from_f32<__hip_bfloat16>(accv[u]);
    }
  }
}

// ---- grw[m, d] = sum_rows dRW[row, m] * q[row, d] -------------------------
struct MhsaRelGradParams {
  const float* dr;          // [rows, M] (dRW or dRH)
  const __hip_bfloat16* q;  // [rows, D]
  float* out;               // [M, D]
  int64_t rows;
  int M, D, rows_per_block;
};

__global__ __launch_bounds__(256) void mhsa_rel_grad_kernel(
    MhsaRelGradParams p) {
  __shared__ float drow[32];
  __shared__ __hip_bfloat16 qrow[128];
  const int tid = threadIdx.x;
  const int d = tid & 127;
  const int half = tid >> 7;               // m in [half*14, ...)
  const int m0 = half * ((p.M + 1) / 2);
  const int mn = min(p.M - m0, (p.M + 1) / 2);
  float acc[16] = {};
  const int64_t r0 = (int64_t)blockIdx.x * p.rows_per_block;
  const int64_t r1 = min(r0 + p.rows_per_block, p.rows);
  for (int64_t r = r0; r < r1; ++r) {
    if (tid < p.M) drow[tid] = p.dr[r * p.M + tid];
    if (tid >= 128 && tid - 128 < p.D / 8) {
      const int c8 = (tid - 128) * 8;
      *reinterpret_cast<bf16x8a*>(&qrow[c8]) =
          *reinterpret_cast<const bf16x8a*>(p.q + r * p.D + c8);
    }
    __syncthreads();
    if (d < p.D) {
      const float qv = to_f32(qrow[d]);
      for (int m = 0; m < mn; ++m) acc[m] += drow[m0 + m] * qv;
    }
    __syncthreads();
  }
  if (d < p.D)
    for (int m = 0; m < mn; ++m)
      atomicAdd(&p.out[(int64_t)(m0 + m) * p.D + d], acc[m]);
}

}  // namespace

// Backward: dO + saved (P, q, k, v) -> dq (incl. rel-pos fold), dk, dv,
// grw, grh. kt/vt are the transposed operands ([B, D, L]).
std::vector<at::Tensor> mhsa_bwd(at::Tensor dO, at::Tensor P, at::Tensor q,
                                 at::Tensor kt, at::Tensor v,
                                 at::Tensor rel_w, at::Tensor rel_h,
                                 int64_t H, int64_t W) {
  CHECK_GPU(dO);
  const int B = q.size(0), L = q.size(1), D = q.size(2);
  const int MW = 2 * W - 1, MH = 2 * H - 1;
  TORCH_CHECK(MW <= 32 && MH <= 32, "mhsa_bwd: rel table > 32");
  auto dq = at::empty_like(q);
  auto ds = at::empty_like(P);
  auto dk = at::empty_like(q);
  auto dv = at::empty_like(q);
  auto fopts = q.options().dtype(at::kFloat);
  auto drw = at::empty({(int64_t)B * L, MW}, fopts);
  auto drh = at::empty({(int64_t)B * L, MH}, fopts);
  const int lpad = ((L + 31) / 32) * 32;
  MhsaBwdQParams p;
  p.dO = (const __hip_bfloat16*)dO.data_ptr();
  p.P = (const __hip_bfloat16*)P.data_ptr();
  p.v = (const __hip_bfloat16*)v.data_ptr();
  p.kt = (const __hip_bfloat16*)kt.data_ptr();
  p.rw = (const __hip_bfloat16*)rel_w.data_ptr();
  p.rh = (const __hip_bfloat16*)rel_h.data_ptr();
  p.dq = (__hip_bfloat16*)dq.data_ptr();
  p.ds = (__hip_bfloat16*)ds.data_ptr();
  p.drw = drw.data_ptr<float>();
  p.drh = drh.data_ptr<float>();
  p.B = B; p.L = L; p.D = D; p.H = H; p.W = W;
  p.ltiles16 = (L + 15) / 16;
  p.lpad = lpad;
  const int smem_bytes = QT * lpad * 6 + QT * 32 * 8;
  static int bwd_smem_set = 0;
  if (smem_bytes > 65536 && !bwd_smem_set) {
    hipFuncSetAttribute((const void*)mhsa_bwd_q_kernel,
                        hipFuncAttributeMaxDynamicSharedMemorySize, 163840);
    bwd_smem_set = 1;
  }
  dim3 grid(B, (L + QT - 1) / QT);
  hipLaunchKernelGGL(mhsa_bwd_q_kernel, grid, dim3(256), smem_bytes,
                     cur_stream(), p);

  MhsaBwdKVParams kv;
  kv.ds = (const __hip_bfloat16*)ds.data_ptr();
  kv.q = (const __hip_bfloat16*)q.data_ptr();
  kv.P = (const __hip_bfloat16*)P.data_ptr();
  kv.dO = (const __hip_bfloat16*)dO.data_ptr();
  kv.dk = (__hip_bfloat16*)dk.data_ptr();
  kv.dv = (__hip_bfloat16*)dv.data_ptr();
  kv.B = B; kv.L = L; kv.D = D;
  dim3 gkv(B, (L + 63) / 64);
  hipLaunchKernelGGL(mhsa_bwd_kv_kernel, gkv, dim3(256), 0, cur_stream(), kv);

  auto grw = at::zeros({MW, D}, fopts);
  auto grh = at::zeros({MH, D}, fopts);
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
    hipLaunchKernelGGL(mhsa_rel_grad_kernel,
                       dim3((int)ceil_div(rows, rpb)), dim3(256), 0,
                       cur_stream(), rp);
  }
  return {dq, dk, dv, grw, grh};
}

// q pre-scaled; returns O [B, L, D]
at::Tensor mhsa_fwd(at::Tensor q, at::Tensor k, at::Tensor vt, at::Tensor rw,
                    at::Tensor rh, int64_t H, int64_t W,
                    c10::optional<at::Tensor> pout) {
  CHECK_GPU(q);
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "mhsa: bf16 only");
  const int B = q.size(0), L = q.size(1), D = q.size(2);
  TORCH_CHECK(D % 32 == 0 && D <= 128, "mhsa: D must be <=128, %32==0");
  TORCH_CHECK(L == H * W, "L != H*W");
  auto o = at::empty_like(q);
  MhsaParams p;
  p.q = (const __hip_bfloat16*)q.data_ptr();
  p.k = (const __hip_bfloat16*)k.data_ptr();
  p.vt = (const __hip_bfloat16*)vt.data_ptr();
  p.rw = rw.data_ptr<float>();
  p.rh = rh.data_ptr<float>();
  p.o = (__hip_bfloat16*)o.data_ptr();
  p.pout = pout.has_value() ? (__hip_bfloat16*)pout->data_ptr() : nullptr;
  p.B = B; p.L = L; p.D = D; p.H = H; p.W = W;
  p.ltiles16 = (L + 15) / 16;
  p.lpad = ((L + 31) / 32) * 32;  // multiple of 32 for the P fragment chunks
  const int smem_bytes = QT * p.lpad * 4 + QT * p.lpad * 2;
  static int max_smem_set = 0;
  if (smem_bytes > 65536 && !max_smem_set) {
    hipFuncSetAttribute((const void*)mhsa_fwd_kernel,
                        hipFuncAttributeMaxDynamicSharedMemorySize, 163840);
    max_smem_set = 1;
  }
  dim3 grid(B, (L + QT - 1) / QT);
  hipLaunchKernelGGL(mhsa_fwd_kernel, grid, dim3(256), smem_bytes,
                     cur_stream(), p);
  return o;
}
