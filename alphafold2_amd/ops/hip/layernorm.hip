// Fused LayerNorm over the last dim — gfx950.
//
// The Evoformer pre-norms every block input (reference alphafold2.py:
// FeedForward :84, AxialAttention :224, TriangleMultiplicative :297,
// OuterMean :338) over tensors up to (b, n, n, D) — a pure memory-bound
// op.  One workgroup per row, fp32 accumulation, vectorized 8-wide bf16
// loads on the fast path (CDNA guide G13: hipcc does not auto-vectorize
// bf16 loads).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T, int VEC>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const float* __restrict__ w,
                                     const float* __restrict__ b,
                                     T* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out,
                                     int rows, int D, float eps) {
  __shared__ float scratch[16];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * D;
    T* yr = y + (long)row * D;

    float sum = 0.f, sumsq = 0.f;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
      float v[VEC];
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        v[k] = to_f32(xr[i + k]);
        sum += v[k];
        sumsq += v[k] * v[k];
      }
    }
    sum = block_reduce_sum(sum, scratch);
    sumsq = block_reduce_sum(sumsq, scratch);

    const float mean = sum / D;
    const float var = sumsq / D - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }

    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float xhat = (to_f32(xr[i + k]) - mean) * rstd;
        yr[i + k] = from_f32<T>(xhat * w[i + k] + b[i + k]);
      }
    }
    __syncthreads();
  }
}

// dx = rstd * (dy*w - mean_j(dy*w) - xhat * mean_j(dy*w*xhat))
// dw/db: each thread owns fixed columns across its block's rows — it
// accumulates locally in registers and issues ONE atomicAdd per column
// at the end (grid-many adds per column total, distinct-address mostly).
template <typename T, int VEC, int MAXCHUNK>
__global__ void layernorm_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ w,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     T* __restrict__ dx,
                                     float* __restrict__ dw,
                                     float* __restrict__ db,
                                     int rows, int D) {
  __shared__ float scratch[16];
  float dw_loc[MAXCHUNK * VEC];
  float db_loc[MAXCHUNK * VEC];
#pragma unroll
  for (int i = 0; i < MAXCHUNK * VEC; ++i) dw_loc[i] = db_loc[i] = 0.f;

  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + (long)row * D;
    const T* xr = x + (long)row * D;
    T* dxr = dx + (long)row * D;
    const float m = mean[row], rs = rstd[row];

    float c1 = 0.f, c2 = 0.f;
    int chunk = 0;
    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC, ++chunk) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float g = to_f32(dyr[i + k]);
        float xhat = (to_f32(xr[i + k]) - m) * rs;
        float gw = g * w[i + k];
        c1 += gw;
        c2 += gw * xhat;
        if (chunk < MAXCHUNK) {
          dw_loc[chunk * VEC + k] += g * xhat;
          db_loc[chunk * VEC + k] += g;
        } else {  // huge-D fallback: direct atomics
          atomicAdd(&dw[i + k], g * xhat);
          atomicAdd(&db[i + k], g);
        }
      }
    }
    c1 = block_reduce_sum(c1, scratch) / D;
    c2 = block_reduce_sum(c2, scratch) / D;

    for (int i = threadIdx.x * VEC; i < D; i += blockDim.x * VEC) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float g = to_f32(dyr[i + k]);
        float xhat = (to_f32(xr[i + k]) - m) * rs;
        dxr[i + k] = from_f32<T>(rs * (g * w[i + k] - c1 - xhat * c2));
      }
    }
    __syncthreads();
  }

  int chunk = 0;
  for (int i = threadIdx.x * VEC; i < D && chunk < MAXCHUNK;
       i += blockDim.x * VEC, ++chunk) {
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      atomicAdd(&dw[i + k], dw_loc[chunk * VEC + k]);
      atomicAdd(&db[i + k], db_loc[chunk * VEC + k]);
    }
  }
}

int pick_grid(int rows) {
  // memory-bound: cap the grid and grid-stride (guide §6 G11)
  const int cap = 256 * 8;
  return rows < cap ? rows : cap;
}

}  // namespace

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w,
                                      at::Tensor b, double eps) {
  TORCH_CHECK(x.is_contiguous(), "layernorm_fwd: x must be contiguous");
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto y = at::empty_like(x);
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto wf = w.to(at::kFloat).contiguous();
  auto bf = b.to(at::kFloat).contiguous();

  const int block = 256;
  const int grid = pick_grid(rows);
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH(T, VEC)                                                     \
  hipLaunchKernelGGL((layernorm_fwd_kernel<T, VEC>), dim3(grid),           \
                     dim3(block), 0, stream,                               \
                     reinterpret_cast<const T*>(x.data_ptr()),             \
                     wf.data_ptr<float>(), bf.data_ptr<float>(),           \
                     reinterpret_cast<T*>(y.data_ptr()),                   \
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),       \
                     (int)rows, D, (float)eps)

  const bool vec8 = (D % 8) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8); else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if (vec8) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "layernorm_fwd: unsupported dtype");
  }
#undef LAUNCH
  return {y, mean, rstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd) {
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous(),
              "layernorm_bwd: inputs must be contiguous");
  const int D = x.size(-1);
  const long rows = x.numel() / D;
  auto dx = at::empty_like(x);
  auto wf = w.to(at::kFloat).contiguous();

  const int block = 256;
  const int grid = pick_grid(rows);
  auto dw = at::zeros({D}, x.options().dtype(at::kFloat));
  auto db = at::zeros({D}, x.options().dtype(at::kFloat));
  auto stream = at::cuda::getCurrentHIPStream();

  // MAXCHUNK covers D <= 256*VEC*MAXCHUNK in registers (D<=2048 for
  // bf16); beyond that the kernel falls back to per-row atomics.
#define LAUNCH(T, VEC)                                                     \
  hipLaunchKernelGGL((layernorm_bwd_kernel<T, VEC, 1>), dim3(grid),        \
                     dim3(block), 0, stream,                               \
                     reinterpret_cast<const T*>(dy.data_ptr()),            \
                     reinterpret_cast<const T*>(x.data_ptr()),             \
                     wf.data_ptr<float>(), mean.data_ptr<float>(),         \
                     rstd.data_ptr<float>(),                               \
                     reinterpret_cast<T*>(dx.data_ptr()),                  \
                     dw.data_ptr<float>(), db.data_ptr<float>(),           \
                     (int)rows, D)

  const bool vec8 = (D % 8) == 0 && D <= 2048;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8); else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if ((D % 4) == 0 && D <= 1024) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "layernorm_bwd: unsupported dtype");
  }
#undef LAUNCH
  return {dx, dw.to(w.scalar_type()), db.to(w.scalar_type())};
}
