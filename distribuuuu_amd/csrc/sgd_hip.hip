#include "hip/hip_runtime.h"
// Fused multi-tensor SGD-momentum step.
// ONE kernel launch updates every parameter: a device-side chunk table
// (param/grad/momentum/master pointers + extent) drives 256-thread blocks,
// each owning one 16K-element chunk. Momentum and master weights are fp32;
// bf16 params get w = master update then a bf16 store (fp32-master numerics).
#include "common_hip.h"

namespace {

constexpr int CHUNK = 16384;

struct ChunkRec {
  const void* grad;
  void* param;
  float* mom;
  float* master;
  int64_t offset;
  int count;
  int flags;  // bit0: param bf16, bit1: grad bf16
};

__global__ void sgd_kernel(const ChunkRec* __restrict__ recs, float lr,
                           float mu, float damp, float wd, int nesterov) {
  const ChunkRec rec = recs[blockIdx.x];
  const int64_t off = rec.offset;
  const bool p_bf16 = rec.flags & 1, g_bf16 = rec.flags & 2;
  for (int i = threadIdx.x; i < rec.count; i += blockDim.x) {
    const int64_t j = off + i;
    float g = g_bf16 ? to_f32(((const __hip_bfloat16*)rec.grad)[j])
                     : ((const float*)rec.grad)[j];
    float w = rec.master[j];
    if (wd != 0.f) g += wd * w;
    float m = rec.mom[j] * mu + (1.f - damp) * g;
    rec.mom[j] = m;
    const float upd = nesterov ? g + mu * m : m;
    w -= lr * upd;
    rec.master[j] = w;
    if (p_bf16)
      ((__hip_bfloat16*)rec.param)[j] = from_f32<__hip_bfloat16>(w);
    else if ((void*)rec.master != rec.param)
      ((float*)rec.param)[j] = w;
  }
}

}  // namespace

void sgd_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
              std::vector<at::Tensor> moms, std::vector<at::Tensor> masters,
              double lr, double momentum, double dampening,
              double weight_decay, bool nesterov) {
  TORCH_CHECK(!params.empty(), "empty param list");
  std::vector<ChunkRec> recs;
  recs.reserve(256);
  for (size_t t = 0; t < params.size(); ++t) {
    auto& p = params[t];
    const int64_t n = p.numel();
    int flags = 0;
    if (p.scalar_type() == at::kBFloat16) flags |= 1;
    if (grads[t].scalar_type() == at::kBFloat16) flags |= 2;
    for (int64_t off = 0; off < n; off += CHUNK) {
      recs.push_back(ChunkRec{grads[t].data_ptr(), p.data_ptr(),
                              moms[t].data_ptr<float>(),
                              masters[t].data_ptr<float>(), off,
                              (int)std::min<int64_t>(CHUNK, n - off), flags});
    }
  }
  auto table = at::empty({(int64_t)(recs.size() * sizeof(ChunkRec))},
                         at::TensorOptions().dtype(at::kByte));
  std::memcpy(table.data_ptr(), recs.data(), recs.size() * sizeof(ChunkRec));
  auto dtable = table.to(params[0].device(), /*non_blocking=*/true);
  hipLaunchKernelGGL(sgd_kernel, dim3((int)recs.size()), dim3(256), 0,
                     cur_stream(), (const ChunkRec*)dtable.data_ptr(),
                     (float)lr, (float)momentum, (float)dampening,
                     (float)weight_decay, nesterov ? 1 : 0);
}
