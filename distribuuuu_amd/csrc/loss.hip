// Cross-entropy loss and top-k accuracy (K15).
// ce_fwd: one block per row; block-reduce max, sum-exp; mean loss via one
// fp32 atomic per row. ce_bwd: gx = gl * (softmax - onehot) / N.
#include "common.h"

namespace {

__global__ void ce_fwd_kernel(const float* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              float* __restrict__ loss,
                              float* __restrict__ lse, int N, int K) {
  __shared__ float lds[16];
  const int n = blockIdx.x;
  const float* row = logits + (int64_t)n * K;
  float m = -INFINITY;
  for (int j = threadIdx.x; j < K; j += blockDim.x) m = fmaxf(m, row[j]);
  m = block_reduce_max<true>(m, lds);
  __syncthreads();
  float s = 0.f;
  for (int j = threadIdx.x; j < K; j += blockDim.x) s += __expf(row[j] - m);
  s = block_reduce_sum<true>(s, lds);
  const float l = m + __logf(s);
  if (threadIdx.x == 0) {
    lse[n] = l;
    atomicAdd(loss, (l - row[target[n]]) / N);
  }
}

__global__ void ce_bwd_kernel(const float* __restrict__ logits,
                              const int64_t* __restrict__ target,
                              const float* __restrict__ lse,
                              const float* __restrict__ gl,
                              float* __restrict__ gx, int64_t total, int K) {
  const float g = gl[0];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t n = i / K;
    const int j = i % K;
    float p = __expf(logits[i] - lse[n]);
    float t = (j == (int)target[n]) ? 1.f : 0.f;
    gx[i] = g * (p - t) / (total / K);
  }
}

// top-k accuracy: per row, count how many of the top-maxk logits beat the
// target's logit (rank); correct@k when rank < k. One wave per row.
__global__ void topk_acc_kernel(const float* __restrict__ logits,
                                const int64_t* __restrict__ target,
                                int* __restrict__ correct1,
                                int* __restrict__ correctk, int N, int K,
                                int topk) {
  const int n = blockIdx.x * (blockDim.x / WAVE) + threadIdx.x / WAVE;
  if (n >= N) return;
  const int lane = threadIdx.x & (WAVE - 1);
  const float* row = logits + (int64_t)n * K;
  const float tv = row[target[n]];
  const int tj = (int)target[n];
  int rank = 0;  // number of entries strictly better than target
  for (int j = lane; j < K; j += WAVE) {
    float v = row[j];
    if (v > tv || (v == tv && j < tj)) ++rank;
  }
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) rank += __shfl_down(rank, off);
  if (lane == 0) {
    if (rank == 0) atomicAdd(correct1, 1);
    if (rank < topk) atomicAdd(correctk, 1);
  }
}

}  // namespace

std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor target) {
  CHECK_GPU(logits);
  TORCH_CHECK(logits.scalar_type() == at::kFloat, "ce_fwd expects fp32 logits");
  const int N = logits.size(0), K = logits.size(1);
  auto loss = at::zeros({}, logits.options());
  auto lse = at::empty({N}, logits.options());
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(N), dim3(256), 0, cur_stream(),
                     logits.data_ptr<float>(), target.data_ptr<int64_t>(),
                     loss.data_ptr<float>(), lse.data_ptr<float>(), N, K);
  return {loss, lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor target, at::Tensor lse,
                  at::Tensor gl) {
  CHECK_GPU(logits);
  const int N = logits.size(0), K = logits.size(1);
  auto gx = at::empty_like(logits);
  int64_t total = (int64_t)N * K;
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid_1d(total, 256)), dim3(256), 0,
                     cur_stream(), logits.data_ptr<float>(),
                     target.data_ptr<int64_t>(), lse.data_ptr<float>(),
                     gl.data_ptr<float>(), gx.data_ptr<float>(), total, K);
  return gx;
}

std::vector<at::Tensor> topk_acc(at::Tensor logits, at::Tensor target,
                                 int64_t topk) {
  CHECK_GPU(logits);
  auto lf = logits.scalar_type() == at::kFloat ? logits : logits.to(at::kFloat);
  const int N = lf.size(0), K = lf.size(1);
  auto c1 = at::zeros({1}, lf.options().dtype(at::kInt));
  auto ck = at::zeros({1}, lf.options().dtype(at::kInt));
  const int waves_per_block = 4;
  int grid = (int)ceil_div(N, waves_per_block);
  hipLaunchKernelGGL(topk_acc_kernel, dim3(grid), dim3(waves_per_block * WAVE),
                     0, cur_stream(), lf.data_ptr<float>(),
                     target.data_ptr<int64_t>(), c1.data_ptr<int>(),
                     ck.data_ptr<int>(), N, K, (int)topk);
  return {c1, ck};
}
