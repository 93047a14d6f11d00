// Fused sigmoid gating: out = x * sigmoid(g) — gfx950.
//
// Used by attention output gating (reference alphafold2.py:184-185) and
// the three triangle-multiplicative gates (:306-311).  One kernel
// instead of sigmoid + mul (and their backward chains); vectorized
// 8-wide bf16 loads (guide G13).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

__device__ __forceinline__ float sigmoid_f(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

template <typename T, int VEC>
__global__ void gatemul_fwd_kernel(const T* __restrict__ x,
                                   const T* __restrict__ g,
                                   T* __restrict__ y, long total) {
  const long stride = (long)gridDim.x * blockDim.x * VEC;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
       base < total; base += stride) {
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      y[base + k] = from_f32<T>(to_f32(x[base + k]) *
                                sigmoid_f(to_f32(g[base + k])));
    }
  }
}

template <typename T, int VEC>
__global__ void gatemul_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ x,
                                   const T* __restrict__ g,
                                   T* __restrict__ dx, T* __restrict__ dg,
                                   long total) {
  const long stride = (long)gridDim.x * blockDim.x * VEC;
  for (long base = ((long)blockIdx.x * blockDim.x + threadIdx.x) * VEC;
       base < total; base += stride) {
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float go = to_f32(dy[base + k]);
      float xv = to_f32(x[base + k]);
      float s = sigmoid_f(to_f32(g[base + k]));
      dx[base + k] = from_f32<T>(go * s);
      dg[base + k] = from_f32<T>(go * xv * s * (1.f - s));
    }
  }
}

}  // namespace

at::Tensor gatemul_fwd(at::Tensor x, at::Tensor g) {
  TORCH_CHECK(x.is_contiguous() && g.is_contiguous(),
              "gatemul_fwd: inputs must be contiguous");
  TORCH_CHECK(x.sizes() == g.sizes(), "gatemul_fwd: shape mismatch");
  auto y = at::empty_like(x);
  const long total = x.numel();
  const int block = 256;
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH(T, VEC)                                                   \
  do {                                                                   \
    long grid = (total / VEC + block - 1) / block;                       \
    if (grid > 2048) grid = 2048;                                        \
    if (grid < 1) grid = 1;                                              \
    hipLaunchKernelGGL((gatemul_fwd_kernel<T, VEC>), dim3(grid),         \
                       dim3(block), 0, stream,                           \
                       reinterpret_cast<const T*>(x.data_ptr()),         \
                       reinterpret_cast<const T*>(g.data_ptr()),         \
                       reinterpret_cast<T*>(y.data_ptr()), total);       \
  } while (0)

  const bool vec8 = (total % 8) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8); else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if ((total % 4) == 0) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "gatemul_fwd: unsupported dtype");
  }
#undef LAUNCH
  return y;
}

std::vector<at::Tensor> gatemul_bwd(at::Tensor dy, at::Tensor x,
                                    at::Tensor g) {
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous() && g.is_contiguous());
  auto dx = at::empty_like(x);
  auto dg = at::empty_like(g);
  const long total = x.numel();
  const int block = 256;
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH(T, VEC)                                                   \
  do {                                                                   \
    long grid = (total / VEC + block - 1) / block;                       \
    if (grid > 2048) grid = 2048;                                        \
    if (grid < 1) grid = 1;                                              \
    hipLaunchKernelGGL((gatemul_bwd_kernel<T, VEC>), dim3(grid),         \
                       dim3(block), 0, stream,                           \
                       reinterpret_cast<const T*>(dy.data_ptr()),        \
                       reinterpret_cast<const T*>(x.data_ptr()),         \
                       reinterpret_cast<const T*>(g.data_ptr()),         \
                       reinterpret_cast<T*>(dx.data_ptr()),              \
                       reinterpret_cast<T*>(dg.data_ptr()), total);      \
  } while (0)

  const bool vec8 = (total % 8) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8); else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if ((total % 4) == 0) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "gatemul_bwd: unsupported dtype");
  }
#undef LAUNCH
  return {dx, dg};
}
