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
// BN-conv fusion (F1): per-wave epilogue partial accumulation
// ---------------------------------------------------------------------------
// Every conv epilogue restages its 64x64 wave tile through a private fp32
// slab (16 rows x 68-stride), one 16-row stripe at a time. When the conv's
// output feeds a BatchNorm, each lane ALSO reads its own column (col = lane)
// of the live stripe and accumulates sum / sum-of-squares of the
// bf16-rounded value — bit-identical to what a separate bn_sums pass over
// the stored y would compute. Only 2 accumulator VGPRs per lane (a 16-wide
// per-lane register accumulator cost conv_gemm_smallc a full wave of
// occupancy: 168 -> 208 VGPRs).
//
// Call between the two wave barriers of a stripe (slab contents valid);
// rows_valid = clamp(M - stripe_row_base, 0, 16) masks M-tail garbage rows.
DEV_INLINE void bn_partial_col_accum(const float* slab, float& s, float& q,
                                     int rows_valid, int lane) {
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    if (r < rows_valid) {
      const float v =
          to_f32(from_f32<__hip_bfloat16>(slab[r * 68 + lane]));
      s += v;
      q += v * v;
    }
  }
}

// Final per-wave store: one fp32 partial row slice part[slot_base + col]
// (+C for sumsq), col = lane, masked to col < kmax. Deterministic: fixed
// (block, wave) -> slot mapping, no atomics.
DEV_INLINE void bn_partial_store(float* __restrict__ part, int64_t slot_base,
                                 int C, int lane, int kmax, float s, float q) {
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
