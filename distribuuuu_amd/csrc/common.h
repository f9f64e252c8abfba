// Common helpers for the gfx950 (CDNA4) kernel set.
// Wave size is 64 on CDNA4 — hard-coded per the platform guide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <algorithm>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------------------
// dtype conversion helpers
// ---------------------------------------------------------------------------
DEV_INLINE float to_f32(float v) { return v; }
DEV_INLINE float to_f32(__hip_bfloat16 v) { return __bfloat162float(v); }

template <typename T>
DEV_INLINE T from_f32(float v);
template <>
DEV_INLINE float from_f32<float>(float v) { return v; }
template <>
DEV_INLINE __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}

// Packed vector of V elements of T, 16-byte aligned for dwordx4 loads.
template <typename T, int V>
struct alignas(16) Pack {
  T v[V];
};

// elements per 16-byte pack
template <typename T>
constexpr int pack_width() { return 16 / sizeof(T); }

// ---------------------------------------------------------------------------
// wave / block reductions
// ---------------------------------------------------------------------------
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) v += __shfl_down(v, off);
  return v;  // valid in lane 0
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off));
  return v;
}

// Block reduction via LDS; blockDim.x threads, returns result in thread 0
// (all threads get the value when bcast=true). LDS buffer: 16 floats.
template <bool BCAST>
DEV_INLINE float block_reduce_sum(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  float r = (threadIdx.x < nw) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) r = wave_reduce_sum(r);
  if (BCAST) {
    if (threadIdx.x == 0) lds[0] = r;
    __syncthreads();
    r = lds[0];
  }
  return r;
}

template <bool BCAST>
DEV_INLINE float block_reduce_max(float v, float* lds) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  float r = (threadIdx.x < nw) ? lds[threadIdx.x] : -INFINITY;
  if (wid == 0) r = wave_reduce_max(r);
  if (BCAST) {
    if (threadIdx.x == 0) lds[0] = r;
    __syncthreads();
    r = lds[0];
  }
  return r;
}

// ---------------------------------------------------------------------------
// BN-conv fusion (F1): per-wave epilogue partial store
// ---------------------------------------------------------------------------
// Every conv epilogue restages its 64x64 wave tile through a private fp32
// slab (16 rows x 68-stride) and stores 16-channel stripes per lane. When the
// conv's output feeds a BatchNorm, the same epilogue also accumulates
// per-channel sum / sum-of-squares of the bf16-ROUNDED stored values
// (bit-identical to what a separate bn_sums pass over y would read) in
// accs/accq[16] per lane, then calls this helper: transpose-reduce over the
// 16 er-lanes through the slab and store ONE fp32 partial row slice
// part[slot_base + col] (+C for sumsq), col = lane, masked to col < kmax.
// Deterministic: fixed (block, wave) -> slot mapping, no atomics.
DEV_INLINE void bn_partial_wave_store(float* slab, const float* accs,
                                      const float* accq,
                                      float* __restrict__ part,
                                      int64_t slot_base, int C, int lane,
                                      int kmax) {
  const int er = lane >> 2, ec = (lane & 3) << 4;
  __builtin_amdgcn_wave_barrier();
#pragma unroll
  for (int j = 0; j < 16; ++j) slab[er * 68 + ec + j] = accs[j];
  __builtin_amdgcn_wave_barrier();
  float s = 0.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) s += slab[r * 68 + lane];
  __builtin_amdgcn_wave_barrier();
#pragma unroll
  for (int j = 0; j < 16; ++j) slab[er * 68 + ec + j] = accq[j];
  __builtin_amdgcn_wave_barrier();
  float q = 0.f;
#pragma unroll
  for (int r = 0; r < 16; ++r) q += slab[r * 68 + lane];
  if (lane < kmax) {
    part[slot_base + lane] = s;
    part[slot_base + C + lane] = q;
  }
}

// ---------------------------------------------------------------------------
// launch helpers
// ---------------------------------------------------------------------------
inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// memory-bound grid cap (Guideline 11): ~2048 blocks, grid-stride the rest
inline int grid_1d(int64_t work, int block) {
  return (int)std::min<int64_t>(ceil_div(work, block), 2048);
}

#define CHECK_GPU(x) TORCH_CHECK((x).is_cuda(), #x " must be on GPU")

inline hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

// NHWC (channels_last) accessor checks
inline void check_nhwc(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.dim() == 4, name, " must be 4-D");
  TORCH_CHECK(t.is_contiguous(at::MemoryFormat::ChannelsLast), name,
              " must be channels_last (NHWC)");
}

#define DISPATCH_FLOAT_AND_BF16(TYPE, NAME, ...)                     \
  [&] {                                                              \
    switch (TYPE) {                                                  \
      case at::ScalarType::Float: {                                  \
        using scalar_t = float;                                      \
        return __VA_ARGS__();                                        \
      }                                                              \
      case at::ScalarType::BFloat16: {                               \
        using scalar_t = __hip_bfloat16;                             \
        return __VA_ARGS__();                                        \
      }                                                              \
      default:                                                       \
        TORCH_CHECK(false, NAME, ": unsupported dtype ", TYPE);      \
    }                                                                \
  }()
