// Fused LayerNorm over the last dim — gfx950.
//
// The Evoformer pre-norms every block input (reference alphafold2.py:
// FeedForward :84, AxialAttention :224, TriangleMultiplicative :297,
// OuterMean :338) over tensors up to (b, n, n, D) — a pure memory-bound
// op.  Parallelization: a GROUP of lanes (16/32/64, sized to D) owns one
// row; 256-thread blocks process 256/GROUP rows concurrently, so all
// lanes stay busy at small D (D=256 would leave 7/8 idle with a
// block-per-row scheme).  Row statistics reduce with shfl_xor inside the
// group — no LDS.  fp32 accumulation, vectorized 8-wide bf16 loads
// (CDNA guide G13).  dw/db accumulate in registers per thread and issue
// one atomicAdd per column per block at the end.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <int GROUP>
__device__ __forceinline__ float group_sum(float v) {
#pragma unroll
  for (int off = GROUP / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, GROUP);
  return v;
}

template <typename T, int VEC, int GROUP>
__global__ void layernorm_fwd_kernel(const T* __restrict__ x,
                                     const float* __restrict__ w,
                                     const float* __restrict__ b,
                                     T* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out,
                                     long rows, int D, float eps) {
  const int RPB = blockDim.x / GROUP;
  const int lane = threadIdx.x % GROUP;
  const int grp = threadIdx.x / GROUP;

  for (long row = (long)blockIdx.x * RPB + grp; row < rows;
       row += (long)gridDim.x * RPB) {
    const T* xr = x + row * D;
    T* yr = y + row * D;

    float sum = 0.f, sumsq = 0.f;
    for (int i = lane * VEC; i < D; i += GROUP * VEC) {
      T xv[VEC];
      vload<T, VEC>(xr + i, xv);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float v = to_f32(xv[k]);
        sum += v;
        sumsq += v * v;
      }
    }
    sum = group_sum<GROUP>(sum);
    sumsq = group_sum<GROUP>(sumsq);

    const float mean = sum / D;
    const float var = sumsq / D - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (lane == 0) {
      mean_out[row] = mean;
      rstd_out[row] = rstd;
    }

    for (int i = lane * VEC; i < D; i += GROUP * VEC) {
      T xv[VEC], yv[VEC];
      vload<T, VEC>(xr + i, xv);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float xhat = (to_f32(xv[k]) - mean) * rstd;
        yv[k] = from_f32<T>(xhat * w[i + k] + b[i + k]);
      }
      vstore<T, VEC>(yr + i, yv);
    }
  }
}

// dx = rstd * (dy*w - mean_j(dy*w) - xhat * mean_j(dy*w*xhat))
template <typename T, int VEC, int GROUP, int MAXCHUNK>
__global__ void layernorm_bwd_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ w,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     T* __restrict__ dx,
                                     float* __restrict__ dw,
                                     float* __restrict__ db,
                                     long rows, int D) {
  const int RPB = blockDim.x / GROUP;
  const int lane = threadIdx.x % GROUP;
  const int grp = threadIdx.x / GROUP;

  // STATIC indexing only — a runtime-indexed local array is allocated
  // in scratch memory (guide common-mistake #20: 5x slowdowns)
  float dw_loc[MAXCHUNK][VEC];
  float db_loc[MAXCHUNK][VEC];
#pragma unroll
  for (int c = 0; c < MAXCHUNK; ++c)
#pragma unroll
    for (int k = 0; k < VEC; ++k) dw_loc[c][k] = db_loc[c][k] = 0.f;

  for (long row = (long)blockIdx.x * RPB + grp; row < rows;
       row += (long)gridDim.x * RPB) {
    const T* dyr = dy + row * D;
    const T* xr = x + row * D;
    T* dxr = dx + row * D;
    const float m = mean[row], rs = rstd[row];

    float c1 = 0.f, c2 = 0.f;
#pragma unroll
    for (int chunk = 0; chunk < MAXCHUNK; ++chunk) {
      const int i = (chunk * GROUP + lane) * VEC;
      if (i < D) {
        T gv[VEC], xv[VEC];
        vload<T, VEC>(dyr + i, gv);
        vload<T, VEC>(xr + i, xv);
#pragma unroll
        for (int k = 0; k < VEC; ++k) {
          float g = to_f32(gv[k]);
          float xhat = (to_f32(xv[k]) - m) * rs;
          float gw = g * w[i + k];
          c1 += gw;
          c2 += gw * xhat;
          dw_loc[chunk][k] += g * xhat;
          db_loc[chunk][k] += g;
        }
      }
    }
    // huge-D tail beyond the register-held chunks: direct atomics
    for (int i = (MAXCHUNK * GROUP + lane) * VEC; i < D; i += GROUP * VEC) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float g = to_f32(dyr[i + k]);
        float xhat = (to_f32(xr[i + k]) - m) * rs;
        float gw = g * w[i + k];
        c1 += gw;
        c2 += gw * xhat;
        atomicAdd(&dw[i + k], g * xhat);
        atomicAdd(&db[i + k], g);
      }
    }
    c1 = group_sum<GROUP>(c1) / D;
    c2 = group_sum<GROUP>(c2) / D;

    for (int i = lane * VEC; i < D; i += GROUP * VEC) {
      T gv[VEC], xv[VEC], ov[VEC];
      vload<T, VEC>(dyr + i, gv);
      vload<T, VEC>(xr + i, xv);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float g = to_f32(gv[k]);
        float xhat = (to_f32(xv[k]) - m) * rs;
        ov[k] = from_f32<T>(rs * (g * w[i + k] - c1 - xhat * c2));
      }
      vstore<T, VEC>(dxr + i, ov);
    }
  }

  // dw/db: write one PARTIAL row per block (no global same-address
  // atomics — 2048 serialized RMW chains per column measured 3.2 ms on
  // HW vs 21 us without; tools/ln_bench.hip).  A small tree-reduce
  // kernel folds the (grid, D) partials afterwards.
  __shared__ float red[256 * VEC > 4096 ? 1 : 256 * VEC];  // dw then db
  float* dwp = dw + (long)blockIdx.x * D;  // dw points at partials here
  float* dbp = db + (long)blockIdx.x * D;
  const bool lds_combine = (256 * VEC <= 4096) && (D <= GROUP * VEC);
  if (lds_combine) {
    // each thread owns exactly VEC columns (chunk 0); combine the RPB
    // groups through LDS, then one coalesced partial-row store
#pragma unroll
    for (int k = 0; k < VEC; ++k) red[threadIdx.x * VEC + k] = dw_loc[0][k];
    __syncthreads();
    if (grp == 0) {
      for (int g2 = 1; g2 < RPB; ++g2)
#pragma unroll
        for (int k = 0; k < VEC; ++k)
          dw_loc[0][k] += red[(g2 * GROUP + lane) * VEC + k];
      const int i = lane * VEC;
      if (i < D)
#pragma unroll
        for (int k = 0; k < VEC; ++k) dwp[i + k] = dw_loc[0][k];
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < VEC; ++k) red[threadIdx.x * VEC + k] = db_loc[0][k];
    __syncthreads();
    if (grp == 0) {
      for (int g2 = 1; g2 < RPB; ++g2)
#pragma unroll
        for (int k = 0; k < VEC; ++k)
          db_loc[0][k] += red[(g2 * GROUP + lane) * VEC + k];
      const int i = lane * VEC;
      if (i < D)
#pragma unroll
        for (int k = 0; k < VEC; ++k) dbp[i + k] = db_loc[0][k];
    }
  } else {
    // rare large-D path: per-block atomics into the partial row
    // (contention = RPB groups only)
    if (threadIdx.x < GROUP) {
      for (int i = threadIdx.x; i < D; i += GROUP) dwp[i] = dbp[i] = 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int chunk = 0; chunk < MAXCHUNK; ++chunk) {
      const int i = (chunk * GROUP + lane) * VEC;
      if (i < D) {
#pragma unroll
        for (int k = 0; k < VEC; ++k) {
          atomicAdd(&dwp[i + k], dw_loc[chunk][k]);
          atomicAdd(&dbp[i + k], db_loc[chunk][k]);
        }
      }
    }
  }
}

// fold (nrows, D) partial matrices into (D,): 64 blocks each sum a
// strided row subset, then 64-deep atomic chains (negligible)
__global__ void fold_partials_kernel(const float* __restrict__ part,
                                     float* __restrict__ out,
                                     int nrows, int D) {
  for (int i = threadIdx.x; i < D; i += blockDim.x) {
    float acc = 0.f;
    for (int r = blockIdx.x; r < nrows; r += gridDim.x)
      acc += part[(long)r * D + i];
    atomicAdd(&out[i], acc);
  }
}

int pick_group(int D, int VEC) {
  int per_row = (D + VEC - 1) / VEC;
  if (per_row <= 16) return 16;
  if (per_row <= 32) return 32;
  return 64;
}

long pick_grid(long rows, int rpb) {
  long blocks = (rows + rpb - 1) / rpb;
  const long cap = 256 * 8;
  return blocks < cap ? blocks : cap;
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
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH_G(T, VEC, GROUP)                                            \
  hipLaunchKernelGGL((layernorm_fwd_kernel<T, VEC, GROUP>),                \
                     dim3(pick_grid(rows, block / GROUP)), dim3(block), 0, \
                     stream, reinterpret_cast<const T*>(x.data_ptr()),     \
                     wf.data_ptr<float>(), bf.data_ptr<float>(),           \
                     reinterpret_cast<T*>(y.data_ptr()),                   \
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),       \
                     rows, D, (float)eps)
#define LAUNCH(T, VEC)                                                     \
  do {                                                                     \
    int g = pick_group(D, VEC);                                            \
    if (g == 16) LAUNCH_G(T, VEC, 16);                                     \
    else if (g == 32) LAUNCH_G(T, VEC, 32);                                \
    else LAUNCH_G(T, VEC, 64);                                             \
  } while (0)

  const bool vec8 = (D % 8) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8); else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if ((D % 4) == 0) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "layernorm_fwd: unsupported dtype");
  }
#undef LAUNCH
#undef LAUNCH_G
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
  auto stream = at::cuda::getCurrentHIPStream();

  // per-block partial rows for dw/db (the kernel writes row blockIdx)
  int max_grid = 0;
  {
    for (int g : {16, 32, 64}) {
      int gr = (int)pick_grid(rows, block / g);
      if (gr > max_grid) max_grid = gr;
    }
  }
  auto dw = at::empty({max_grid, D}, x.options().dtype(at::kFloat));
  auto db = at::empty({max_grid, D}, x.options().dtype(at::kFloat));

  // MAXCHUNK sized to the actual D so accumulators stay in registers
  // (MAXCHUNK*VEC floats x2 per thread; 16 would spill)
#define LAUNCH_GC(T, VEC, GROUP, MC)                                       \
  hipLaunchKernelGGL((layernorm_bwd_kernel<T, VEC, GROUP, MC>),            \
                     dim3(pick_grid(rows, block / GROUP)), dim3(block), 0, \
                     stream, reinterpret_cast<const T*>(dy.data_ptr()),    \
                     reinterpret_cast<const T*>(x.data_ptr()),             \
                     wf.data_ptr<float>(), mean.data_ptr<float>(),         \
                     rstd.data_ptr<float>(),                               \
                     reinterpret_cast<T*>(dx.data_ptr()),                  \
                     dw.data_ptr<float>(), db.data_ptr<float>(), rows, D)
#define LAUNCH_G(T, VEC, GROUP)                                            \
  do {                                                                     \
    int chunks = (D + GROUP * VEC - 1) / (GROUP * VEC);                    \
    if (chunks <= 1) LAUNCH_GC(T, VEC, GROUP, 1);                          \
    else if (chunks <= 4) LAUNCH_GC(T, VEC, GROUP, 4);                     \
    else LAUNCH_GC(T, VEC, GROUP, 16);                                     \
  } while (0)
#define LAUNCH(T, VEC)                                                     \
  do {                                                                     \
    int g = pick_group(D, VEC);                                            \
    if (g == 16) LAUNCH_G(T, VEC, 16);                                     \
    else if (g == 32) LAUNCH_G(T, VEC, 32);                                \
    else LAUNCH_G(T, VEC, 64);                                             \
  } while (0)

  const bool vec8 = (D % 8) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8); else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if ((D % 4) == 0) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "layernorm_bwd: unsupported dtype");
  }
#undef LAUNCH
#undef LAUNCH_G
#undef LAUNCH_GC

  // fold the partial rows that were actually written
  int vec_used = 1;
  if (x.scalar_type() == at::kFloat) vec_used = ((D % 4) == 0) ? 4 : 1;
  else vec_used = ((D % 8) == 0) ? 8 : 1;
  const int g_used = pick_group(D, vec_used);
  const int rows_written = (int)pick_grid(rows, block / g_used);
  auto dw_out = at::zeros({D}, x.options().dtype(at::kFloat));
  auto db_out = at::zeros({D}, x.options().dtype(at::kFloat));
  hipLaunchKernelGGL(fold_partials_kernel, dim3(64), dim3(256), 0, stream,
                     dw.data_ptr<float>(), dw_out.data_ptr<float>(),
                     rows_written, D);
  hipLaunchKernelGGL(fold_partials_kernel, dim3(64), dim3(256), 0, stream,
                     db.data_ptr<float>(), db_out.data_ptr<float>(),
                     rows_written, D);
  return {dx, dw_out.to(w.scalar_type()), db_out.to(w.scalar_type())};
}
