#include "hip/hip_runtime.h"
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
#include "common_hip.h"

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

}  // namespace

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
