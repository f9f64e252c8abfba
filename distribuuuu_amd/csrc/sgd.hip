// Fused multi-tensor SGD-momentum step.
// ONE kernel launch updates every parameter: a device-side chunk table
// (built once by the Python optimizer and cached across steps — no per-step
// host work, hipGraph-capturable) drives 256-thread blocks, each owning one
// 16K-element chunk. Momentum and master weights are fp32; bf16 params get
// w = master update then a bf16 store (fp32-master numerics).
//
// Table layout: int64 [nchunks, 6]:
//   [grad_ptr, param_ptr, mom_ptr, master_ptr, offset, count | flags<<32]
//   flags bit0: param bf16, bit1: grad bf16
#include "common.h"

namespace {

__global__ void sgd_kernel(const long long* __restrict__ recs, float lr,
                           float mu, float damp, float wd, int nesterov) {
  const long long* rec = recs + (int64_t)blockIdx.x * 6;
  const void* grad = (const void*)rec[0];
  void* param = (void*)rec[1];
  float* mom = (float*)rec[2];
  float* master = (float*)rec[3];
  const int64_t off = rec[4];
  const int count = (int)(rec[5] & 0xffffffffll);
  const int flags = (int)(rec[5] >> 32);
  const bool p_bf16 = flags & 1, g_bf16 = flags & 2;
  for (int i = threadIdx.x; i < count; i += blockDim.x) {
    const int64_t j = off + i;
    float g = g_bf16 ? to_f32(((const __hip_bfloat16*)grad)[j])
                     : ((const float*)grad)[j];
    float w = master[j];
    if (wd != 0.f) g += wd * w;
    float m = mom[j] * mu + (1.f - damp) * g;
    mom[j] = m;
    const float upd = nesterov ? g + mu * m : m;
    w -= lr * upd;
    master[j] = w;
    if (p_bf16)
      ((__hip_bfloat16*)param)[j] = from_f32<__hip_bfloat16>(w);
    else if ((void*)master != param)
      ((float*)param)[j] = w;
  }
}

}  // namespace

void sgd_step(at::Tensor table, double lr, double momentum, double dampening,
              double weight_decay, bool nesterov) {
  CHECK_GPU(table);
  TORCH_CHECK(table.scalar_type() == at::kLong && table.size(1) == 6,
              "sgd table must be int64 [n, 6]");
  const int n = table.size(0);
  hipLaunchKernelGGL(sgd_kernel, dim3(n), dim3(256), 0, cur_stream(),
                     (const long long*)table.data_ptr(), (float)lr,
                     (float)momentum, (float)dampening, (float)weight_decay,
                     nesterov ? 1 : 0);
}
