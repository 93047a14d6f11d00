// Fused GEGLU activation — gfx950.
//
// FeedForward transition (reference alphafold2.py:69-94): the Linear
// produces (..., 2H); GEGLU keeps a = x[..., :H], gates = x[..., H:],
// out = a * gelu(gates).  Fusing the chunk + gelu + multiply saves two
// full tensor read/writes on a memory-bound op (K6 epilogue of
// SURVEY.md §2.17).  Vectorized 8-wide loads (guide G13).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

// row-outer loop with multi-row packing: when H/VEC < blockDim the
// block covers several rows per iteration (shift-based row split for
// power-of-2 H — a runtime `idx / H` in the hot loop serializes on the
// integer-div unit).  row_shift = log2(H / VEC); 0 disables packing.
template <typename T, int VEC>
__global__ void geglu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                 long rows, int H, int row_shift) {
  const int sub = row_shift ? (threadIdx.x >> row_shift) : 0;
  const int rows_per_iter = row_shift ? (blockDim.x >> row_shift) : 1;
  const int col0 = row_shift
      ? (threadIdx.x & ((1 << row_shift) - 1)) * VEC : threadIdx.x * VEC;
  const int cstep = row_shift ? H : blockDim.x * VEC;
  for (long row = (long)blockIdx.x * rows_per_iter + sub; row < rows;
       row += (long)gridDim.x * rows_per_iter) {
    const T* xr = x + row * (2L * H);
    T* yr = y + row * (long)H;
    for (int col = col0; col < H; col += cstep) {
      T av[VEC], gv[VEC], yv[VEC];
      vload<T, VEC>(xr + col, av);
      vload<T, VEC>(xr + H + col, gv);
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        yv[k] = from_f32<T>(to_f32(av[k]) * gelu_f(to_f32(gv[k])));
      vstore<T, VEC>(yr + col, yv);
    }
  }
}

template <typename T, int VEC>
__global__ void geglu_bwd_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x,
                                 T* __restrict__ dx, long rows, int H,
                                 int row_shift) {
  const int sub = row_shift ? (threadIdx.x >> row_shift) : 0;
  const int rows_per_iter = row_shift ? (blockDim.x >> row_shift) : 1;
  const int col0 = row_shift
      ? (threadIdx.x & ((1 << row_shift) - 1)) * VEC : threadIdx.x * VEC;
  const int cstep = row_shift ? H : blockDim.x * VEC;
  for (long row = (long)blockIdx.x * rows_per_iter + sub; row < rows;
       row += (long)gridDim.x * rows_per_iter) {
    const T* dyr = dy + row * (long)H;
    const T* xr = x + row * (2L * H);
    T* dxr = dx + row * (2L * H);
    for (int col = col0; col < H; col += cstep) {
      T av[VEC], gv[VEC], dov[VEC], dav[VEC], dgv[VEC];
      vload<T, VEC>(xr + col, av);
      vload<T, VEC>(xr + H + col, gv);
      vload<T, VEC>(dyr + col, dov);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float go = to_f32(dov[k]);
        float g = to_f32(gv[k]);
        dav[k] = from_f32<T>(go * gelu_f(g));
        dgv[k] = from_f32<T>(go * to_f32(av[k]) * gelu_grad_f(g));
      }
      vstore<T, VEC>(dxr + col, dav);
      vstore<T, VEC>(dxr + H + col, dgv);
    }
  }
}

}  // namespace

at::Tensor geglu_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_contiguous(), "geglu_fwd: x must be contiguous");
  const int H2 = x.size(-1);
  TORCH_CHECK(H2 % 2 == 0, "geglu_fwd: last dim must be even");
  const int H = H2 / 2;
  const long rows = x.numel() / H2;

  auto sizes = x.sizes().vec();
  sizes.back() = H;
  auto y = at::empty(sizes, x.options());

  const int block = 256;
  const long total = rows * (long)H;
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH(T, VEC)                                                   \
  do {                                                                   \
    int colt = H / VEC;                                                  \
    int row_shift = 0;                                                   \
    if (colt < block && colt > 0 && (colt & (colt - 1)) == 0)            \
      row_shift = __builtin_ctz(colt);                                   \
    const int rpi = row_shift ? block >> row_shift : 1;                  \
    long grid = (rows + rpi - 1) / rpi;                                  \
    if (grid > 65536) grid = 65536;                                        \
    if (grid < 1) grid = 1;                                              \
    hipLaunchKernelGGL((geglu_fwd_kernel<T, VEC>), dim3(grid),           \
                       dim3(block), 0, stream,                           \
                       reinterpret_cast<const T*>(x.data_ptr()),         \
                       reinterpret_cast<T*>(y.data_ptr()), rows, H,      \
                       row_shift);                                       \
  } while (0)

  // multi-row packing keeps all lanes busy at any H; prefer 16B lanes
  const bool vec8 = (H % 8) == 0;
  const bool vec4 = (H % 4) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8);
    else if (vec4) LAUNCH(__hip_bfloat16, 4);
    else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if (vec4) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8);
    else if (vec4) LAUNCH(__half, 4);
    else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "geglu_fwd: unsupported dtype");
  }
#undef LAUNCH
  return y;
}

at::Tensor geglu_bwd(at::Tensor dy, at::Tensor x) {
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous(),
              "geglu_bwd: inputs must be contiguous");
  const int H2 = x.size(-1);
  const int H = H2 / 2;
  const long rows = x.numel() / H2;
  auto dx = at::empty_like(x);

  const int block = 256;
  const long total = rows * (long)H;
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH(T, VEC)                                                   \
  do {                                                                   \
    int colt = H / VEC;                                                  \
    int row_shift = 0;                                                   \
    if (colt < block && colt > 0 && (colt & (colt - 1)) == 0)            \
      row_shift = __builtin_ctz(colt);                                   \
    const int rpi = row_shift ? block >> row_shift : 1;                  \
    long grid = (rows + rpi - 1) / rpi;                                  \
    if (grid > 65536) grid = 65536;                                        \
    if (grid < 1) grid = 1;                                              \
    hipLaunchKernelGGL((geglu_bwd_kernel<T, VEC>), dim3(grid),           \
                       dim3(block), 0, stream,                           \
                       reinterpret_cast<const T*>(dy.data_ptr()),        \
                       reinterpret_cast<const T*>(x.data_ptr()),         \
                       reinterpret_cast<T*>(dx.data_ptr()), rows, H,     \
                       row_shift);                                       \
  } while (0)

  // multi-row packing keeps all lanes busy at any H; prefer 16B lanes
  const bool vec8 = (H % 8) == 0;
  const bool vec4 = (H % 4) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (vec8) LAUNCH(__hip_bfloat16, 8);
    else if (vec4) LAUNCH(__hip_bfloat16, 4);
    else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if (vec4) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (vec8) LAUNCH(__half, 8);
    else if (vec4) LAUNCH(__half, 4);
    else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "geglu_bwd: unsupported dtype");
  }
#undef LAUNCH
  return dx;
}
