// Fused sigmoid gating: out = x * sigmoid(g) — gfx950.
//
// Used by attention output gating (reference alphafold2.py:184-185) and
// the three triangle-multiplicative gates (:306-311).  One kernel
// instead of sigmoid + mul; inputs may be row-strided SLICES of a fused
// projection (no contiguous() copies); row-group parallelism keeps all
// lanes on 16-byte accesses at any channel width.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

__device__ __forceinline__ float sigmoid_f(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

// rowmask: optional per-row multiplier (0/1 byte) fused in — saves the
// separate mask-multiply passes in the triangle module
template <typename T, int VEC, int GROUP, bool RM>
__global__ void gatemul_fwd_kernel(const T* __restrict__ x, long xs,
                                   const T* __restrict__ g, long gs,
                                   const unsigned char* __restrict__ rowmask,
                                   T* __restrict__ y,
                                   long rows, int C) {
  const int RPB = blockDim.x / GROUP;
  const int lane = threadIdx.x % GROUP;
  const int grp = threadIdx.x / GROUP;
  for (long row = (long)blockIdx.x * RPB + grp; row < rows;
       row += (long)gridDim.x * RPB) {
    const T* xr = x + row * xs;
    const T* gr = g + row * gs;
    T* yr = y + row * (long)C;
    const float mv = RM ? (float)rowmask[row] : 1.f;
    for (int i = lane * VEC; i < C; i += GROUP * VEC) {
      T xv[VEC], gv[VEC], yv[VEC];
      vload<T, VEC>(xr + i, xv);
      vload<T, VEC>(gr + i, gv);
#pragma unroll
      for (int k = 0; k < VEC; ++k)
        yv[k] = from_f32<T>(to_f32(xv[k]) * mv *
                            sigmoid_f(to_f32(gv[k])));
      vstore<T, VEC>(yr + i, yv);
    }
  }
}

template <typename T, int VEC, int GROUP, bool RM>
__global__ void gatemul_bwd_kernel(const T* __restrict__ dy,
                                   const T* __restrict__ x, long xs,
                                   const T* __restrict__ g, long gs,
                                   const unsigned char* __restrict__ rowmask,
                                   T* __restrict__ dx, long dxs,
                                   T* __restrict__ dg, long dgs,
                                   long rows, int C) {
  const int RPB = blockDim.x / GROUP;
  const int lane = threadIdx.x % GROUP;
  const int grp = threadIdx.x / GROUP;
  for (long row = (long)blockIdx.x * RPB + grp; row < rows;
       row += (long)gridDim.x * RPB) {
    const T* dyr = dy + row * (long)C;
    const T* xr = x + row * xs;
    const T* gr = g + row * gs;
    T* dxr = dx + row * dxs;
    T* dgr = dg + row * dgs;
    const float mv = RM ? (float)rowmask[row] : 1.f;
    for (int i = lane * VEC; i < C; i += GROUP * VEC) {
      T xv[VEC], gv[VEC], dov[VEC], dxv[VEC], dgv[VEC];
      vload<T, VEC>(xr + i, xv);
      vload<T, VEC>(gr + i, gv);
      vload<T, VEC>(dyr + i, dov);
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float go = to_f32(dov[k]) * mv;
        float s = sigmoid_f(to_f32(gv[k]));
        dxv[k] = from_f32<T>(go * s);
        dgv[k] = from_f32<T>(go * to_f32(xv[k]) * s * (1.f - s));
      }
      vstore<T, VEC>(dxr + i, dxv);
      vstore<T, VEC>(dgr + i, dgv);
    }
  }
}

int gm_group(int C, int VEC) {
  int per_row = (C + VEC - 1) / VEC;
  if (per_row <= 16) return 16;
  if (per_row <= 32) return 32;
  return 64;
}

long gm_grid(long rows, int rpb) {
  long blocks = (rows + rpb - 1) / rpb;
  const long cap = 65536;
  return blocks < cap ? blocks : (cap > 0 ? cap : 1);
}

}  // namespace

// x/g: (rows, C) with uniform row strides xs/gs (elements); y contiguous
at::Tensor gatemul_fwd(at::Tensor x, at::Tensor g, long xs, long gs,
                       c10::optional<at::Tensor> rowmask) {
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  auto y = at::empty(x.sizes(), x.options());
  const int block = 256;
  auto stream = at::cuda::getCurrentHIPStream();
  const bool rm = rowmask.has_value();
  const unsigned char* rm_ptr =
      rm ? rowmask->data_ptr<unsigned char>() : nullptr;

#define LAUNCH_G2(T, VEC, GROUP, RM)                                        \
  hipLaunchKernelGGL((gatemul_fwd_kernel<T, VEC, GROUP, RM>),               \
                     dim3(gm_grid(rows, block / GROUP)), dim3(block), 0,    \
                     stream, reinterpret_cast<const T*>(x.data_ptr()), xs,  \
                     reinterpret_cast<const T*>(g.data_ptr()), gs, rm_ptr,  \
                     reinterpret_cast<T*>(y.data_ptr()), rows, C)
#define LAUNCH_G(T, VEC, GROUP)                                             \
  do {                                                                      \
    if (rm) LAUNCH_G2(T, VEC, GROUP, true);                                 \
    else LAUNCH_G2(T, VEC, GROUP, false);                                   \
  } while (0)
#define LAUNCH(T, VEC)                                                      \
  do {                                                                      \
    int gg = gm_group(C, VEC);                                              \
    if (gg == 16) LAUNCH_G(T, VEC, 16);                                     \
    else if (gg == 32) LAUNCH_G(T, VEC, 32);                                \
    else LAUNCH_G(T, VEC, 64);                                              \
  } while (0)

  const bool al8 = (C % 8) == 0 && (xs % 8) == 0 && (gs % 8) == 0;
  const bool al4 = (C % 4) == 0 && (xs % 4) == 0 && (gs % 4) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (al8) LAUNCH(__hip_bfloat16, 8);
    else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if (al4) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (al8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "gatemul_fwd: unsupported dtype");
  }
#undef LAUNCH
#undef LAUNCH_G
#undef LAUNCH_G2
  return y;
}

// dx_out/dg_out (with row strides dxs/dgs) let the backward write its
// gradients straight into slices of a packed buffer — the autograd
// SplitBackward concatenation disappears (it was ~16 ms/step).
std::vector<at::Tensor> gatemul_bwd(at::Tensor dy, at::Tensor x,
                                    at::Tensor g, long xs, long gs,
                                    c10::optional<at::Tensor> rowmask,
                                    c10::optional<at::Tensor> dx_out,
                                    c10::optional<at::Tensor> dg_out,
                                    long dxs, long dgs) {
  TORCH_CHECK(dy.is_contiguous(), "gatemul_bwd: dy must be contiguous");
  const int C = x.size(-1);
  const long rows = x.numel() / C;
  at::Tensor dx, dg;
  if (dx_out.has_value()) {
    dx = *dx_out;
    dg = *dg_out;
  } else {
    dx = at::empty(x.sizes(), x.options());
    dg = at::empty(x.sizes(), x.options());
    dxs = dgs = C;
  }
  const int block = 256;
  auto stream = at::cuda::getCurrentHIPStream();
  const bool rm = rowmask.has_value();
  const unsigned char* rm_ptr =
      rm ? rowmask->data_ptr<unsigned char>() : nullptr;

#define LAUNCH_G2(T, VEC, GROUP, RM)                                        \
  hipLaunchKernelGGL((gatemul_bwd_kernel<T, VEC, GROUP, RM>),               \
                     dim3(gm_grid(rows, block / GROUP)), dim3(block), 0,    \
                     stream, reinterpret_cast<const T*>(dy.data_ptr()),     \
                     reinterpret_cast<const T*>(x.data_ptr()), xs,          \
                     reinterpret_cast<const T*>(g.data_ptr()), gs, rm_ptr,  \
                     reinterpret_cast<T*>(dx.data_ptr()), dxs,              \
                     reinterpret_cast<T*>(dg.data_ptr()), dgs, rows, C)
#define LAUNCH_G(T, VEC, GROUP)                                             \
  do {                                                                      \
    if (rm) LAUNCH_G2(T, VEC, GROUP, true);                                 \
    else LAUNCH_G2(T, VEC, GROUP, false);                                   \
  } while (0)
#define LAUNCH(T, VEC)                                                      \
  do {                                                                      \
    int gg = gm_group(C, VEC);                                              \
    if (gg == 16) LAUNCH_G(T, VEC, 16);                                     \
    else if (gg == 32) LAUNCH_G(T, VEC, 32);                                \
    else LAUNCH_G(T, VEC, 64);                                              \
  } while (0)

  const bool al8 = (C % 8) == 0 && (xs % 8) == 0 && (gs % 8) == 0
      && (dxs % 8) == 0 && (dgs % 8) == 0;
  const bool al4 = (C % 4) == 0 && (xs % 4) == 0 && (gs % 4) == 0
      && (dxs % 4) == 0 && (dgs % 4) == 0;
  if (x.scalar_type() == at::kBFloat16) {
    if (al8) LAUNCH(__hip_bfloat16, 8);
    else LAUNCH(__hip_bfloat16, 1);
  } else if (x.scalar_type() == at::kFloat) {
    if (al4) LAUNCH(float, 4); else LAUNCH(float, 1);
  } else if (x.scalar_type() == at::kHalf) {
    if (al8) LAUNCH(__half, 8); else LAUNCH(__half, 1);
  } else {
    TORCH_CHECK(false, "gatemul_bwd: unsupported dtype");
  }
#undef LAUNCH
#undef LAUNCH_G
#undef LAUNCH_G2
  return {dx, dg};
}
