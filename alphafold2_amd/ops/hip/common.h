// Common device helpers for the gfx950 (CDNA4) kernels.
// Wave width is 64 on CDNA — hard-coded per the CDNA HIP guide.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

// ---- dtype conversion helpers -------------------------------------------

template <typename T> struct AccT { using type = float; };

__device__ __forceinline__ float to_f32(float v) { return v; }
__device__ __forceinline__ float to_f32(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
__device__ __forceinline__ float to_f32(__half v) { return __half2float(v); }

template <typename T> __device__ __forceinline__ T from_f32(float v);
template <> __device__ __forceinline__ float from_f32<float>(float v) {
  return v;
}
template <> __device__ __forceinline__ __hip_bfloat16
from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <> __device__ __forceinline__ __half from_f32<__half>(float v) {
  return __float2half(v);
}

// ---- wave + block reductions --------------------------------------------

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    v += __shfl_down(v, off, WAVE);
  }
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    v = fmaxf(v, __shfl_down(v, off, WAVE));
  }
  return v;
}

// Block reduction for blockDim.x threads (multiple of 64), result valid on
// every thread.  `scratch` must hold blockDim.x / 64 floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int nwaves = blockDim.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wave] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  __syncthreads();
  return total;
}


// ---- explicit wide vector access ----------------------------------------
// the compiler does not reliably merge unrolled scalar bf16 accesses
// (guide §5 common-mistake #2); these force float4/float2 moves.

template <typename T, int VEC>
__device__ __forceinline__ void vload(const T* __restrict__ p, T* dst) {
  constexpr int BYTES = sizeof(T) * VEC;
  if constexpr (BYTES == 16) {
    *reinterpret_cast<float4*>(dst) = *reinterpret_cast<const float4*>(p);
  } else if constexpr (BYTES == 8) {
    *reinterpret_cast<float2*>(dst) = *reinterpret_cast<const float2*>(p);
  } else {
#pragma unroll
    for (int k = 0; k < VEC; ++k) dst[k] = p[k];
  }
}

template <typename T, int VEC>
__device__ __forceinline__ void vstore(T* __restrict__ p, const T* src) {
  constexpr int BYTES = sizeof(T) * VEC;
  if constexpr (BYTES == 16) {
    *reinterpret_cast<float4*>(p) = *reinterpret_cast<const float4*>(src);
  } else if constexpr (BYTES == 8) {
    *reinterpret_cast<float2*>(p) = *reinterpret_cast<const float2*>(src);
  } else {
#pragma unroll
    for (int k = 0; k < VEC; ++k) p[k] = src[k];
  }
}

// exact gelu (erf form) matching torch.nn.functional.gelu default
__device__ __forceinline__ float gelu_f(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f));
}

__device__ __forceinline__ float gelu_grad_f(float x) {
  // d/dx [x * Phi(x)] = Phi(x) + x * phi(x)
  const float kInvSqrt2 = 0.70710678118654752440f;
  const float kInvSqrt2Pi = 0.39894228040143267794f;
  float cdf = 0.5f * (1.0f + erff(x * kInvSqrt2));
  float pdf = kInvSqrt2Pi * expf(-0.5f * x * x);
  return cdf + x * pdf;
}
