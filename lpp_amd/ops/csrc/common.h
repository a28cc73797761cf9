// Common helpers for the gfx950 (CDNA4) kernels.
//
// Conventions (per the CDNA HIP guide):
//  - wave64: every warp-style idiom uses 64 lanes.
//  - bf16 loads/stores vectorised as 16-byte packets (8 x bf16) — hipcc does
//    not auto-vectorise scalar bf16 (guide G13).
//  - memory-bound grids capped at ~2048 blocks with grid-stride loops
//    (guide G11).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPStream.h>

#define LPP_CHECK_HIP(expr)                                              \
  do {                                                                   \
    hipError_t _e = (expr);                                              \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e)); \
  } while (0)

namespace lpp {

constexpr int kWave = 64;

static inline hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ---- scalar conversions ----------------------------------------------------
template <typename T>
__device__ __forceinline__ float to_f32(T v);
template <>
__device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <>
__device__ __forceinline__ float to_f32<__half>(__half v) { return __half2float(v); }

template <typename T>
__device__ __forceinline__ T from_f32(float v);
template <>
__device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <>
__device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <>
__device__ __forceinline__ __half from_f32<__half>(float v) { return __float2half(v); }

// ---- vector packet: N elements moved as one aligned load/store -------------
template <typename T, int N>
struct alignas(sizeof(T) * N) Pack {
  T v[N];
};

// ---- wave + block reductions ----------------------------------------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, kWave);
  return v;  // valid in lane 0 of the wave
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_down(v, off, kWave));
  return v;
}

// Block reduction (block size <= 1024 i.e. <= 16 waves). Every thread gets
// the result (broadcast through LDS slot 0).
template <int BLOCK>
__device__ __forceinline__ float block_reduce_sum(float v, float* lds /*[BLOCK/kWave]*/) {
  constexpr int NW = BLOCK / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  v = wave_reduce_sum(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll
  for (int i = 0; i < NW; ++i) total += lds[i];
  __syncthreads();
  return total;
}

template <int BLOCK>
__device__ __forceinline__ float block_reduce_max(float v, float* lds) {
  constexpr int NW = BLOCK / kWave;
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  v = wave_reduce_max(v);
  if (lane == 0) lds[wid] = v;
  __syncthreads();
  float m = -INFINITY;
#pragma unroll
  for (int i = 0; i < NW; ++i) m = fmaxf(m, lds[i]);
  __syncthreads();
  return m;
}

// ---- dtype dispatch ---------------------------------------------------------
// Maps torch scalar types onto HIP-native element types.
#define LPP_DISPATCH_FLOAT(TYPE, NAME, ...)                                 \
  [&] {                                                                     \
    switch (TYPE) {                                                         \
      case at::ScalarType::Float: {                                         \
        using scalar_t = float;                                             \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::ScalarType::BFloat16: {                                      \
        using scalar_t = __hip_bfloat16;                                    \
        return __VA_ARGS__();                                               \
      }                                                                     \
      case at::ScalarType::Half: {                                          \
        using scalar_t = __half;                                            \
        return __VA_ARGS__();                                               \
      }                                                                     \
      default:                                                              \
        TORCH_CHECK(false, NAME, ": unsupported dtype ", TYPE);             \
    }                                                                       \
  }()

static inline int grid_for(int64_t work_items, int block, int cap = 2048) {
  int64_t g = (work_items + block - 1) / block;
  return (int)std::min<int64_t>(g, cap);
}

}  // namespace lpp
