// apex_amd common device helpers — hand-written HIP for gfx950 (CDNA4).
// Wave width is 64 on CDNA4; all reductions here are wave64 shuffles
// (the reference's 32-wide warp idioms do not apply).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#define WAVE_SIZE 64

#define HIP_CHECK(expr)                                                        \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    TORCH_CHECK(_e == hipSuccess, "HIP error: ", hipGetErrorString(_e));       \
  } while (0)

static inline hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ---------- dtype dispatch ----------
#define APEX_DISPATCH_FLOAT_HALF_BF(TYPE, NAME, ...)                           \
  switch (TYPE) {                                                              \
    case at::ScalarType::Float: {                                              \
      using scalar_t = float;                                                  \
      __VA_ARGS__;                                                             \
      break;                                                                   \
    }                                                                          \
    case at::ScalarType::Half: {                                               \
      using scalar_t = __half;                                                 \
      __VA_ARGS__;                                                             \
      break;                                                                   \
    }                                                                          \
    case at::ScalarType::BFloat16: {                                           \
      using scalar_t = __hip_bfloat16;                                         \
      __VA_ARGS__;                                                             \
      break;                                                                   \
    }                                                                          \
    default:                                                                   \
      TORCH_CHECK(false, NAME, ": unsupported dtype");                         \
  }

// ---------- scalar conversion ----------
template <typename T>
__device__ __forceinline__ float to_float(T v);
template <>
__device__ __forceinline__ float to_float<float>(float v) { return v; }
template <>
__device__ __forceinline__ float to_float<__half>(__half v) { return __half2float(v); }
template <>
__device__ __forceinline__ float to_float<__hip_bfloat16>(__hip_bfloat16 v) { return __bfloat162float(v); }

template <typename T>
__device__ __forceinline__ T from_float(float v);
template <>
__device__ __forceinline__ float from_float<float>(float v) { return v; }
template <>
__device__ __forceinline__ __half from_float<__half>(float v) { return __float2half(v); }
template <>
__device__ __forceinline__ __hip_bfloat16 from_float<__hip_bfloat16>(float v) { return __float2bfloat16(v); }

// ---------- wave64 + block reductions ----------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) v += __shfl_xor(v, off);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE_SIZE / 2; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off));
  return v;
}

// Block reduction over up to 16 waves via LDS. `smem` needs blockDim.x/64
// floats. Result valid on every thread.
__device__ __forceinline__ float block_reduce_sum(float v, float* smem) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  v = wave_reduce_sum(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += smem[i];
  __syncthreads();
  return total;
}

__device__ __forceinline__ float block_reduce_max(float v, float* smem) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x / WAVE_SIZE;
  const int nwaves = blockDim.x / WAVE_SIZE;
  v = wave_reduce_max(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  float total = -INFINITY;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total = fmaxf(total, smem[i]);
  __syncthreads();
  return total;
}
