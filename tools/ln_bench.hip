// A/B ablation of the LayerNorm-backward kernel phases on (65536, 256)
// bf16 — find what costs 431us when the forward takes 24us.
// V0: full kernel    V1: no atomic epilogue    V2: no dw/db accumulation
// V3: dx passes only (no c1/c2 reduce)         V4: pure read+write copy
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/ln_bench.hip -o /tmp/ln_bench
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>

#define GROUP 32
#define VEC 8
typedef __hip_bfloat16 bf16;

__device__ __forceinline__ float gsum(float v) {
#pragma unroll
  for (int off = GROUP / 2; off > 0; off >>= 1) v += __shfl_xor(v, off, GROUP);
  return v;
}

template <int VARIANT>
__global__ void ln_bwd(const bf16* __restrict__ dy, const bf16* __restrict__ x,
                       const float* __restrict__ w,
                       const float* __restrict__ mean,
                       const float* __restrict__ rstd, bf16* __restrict__ dx,
                       float* __restrict__ dw, float* __restrict__ db,
                       long rows, int D) {
  const int RPB = blockDim.x / GROUP;
  const int lane = threadIdx.x % GROUP;
  const int grp = threadIdx.x / GROUP;

  float dw_loc[VEC], db_loc[VEC];
#pragma unroll
  for (int k = 0; k < VEC; ++k) dw_loc[k] = db_loc[k] = 0.f;

  for (long row = (long)blockIdx.x * RPB + grp; row < rows;
       row += (long)gridDim.x * RPB) {
    const bf16* dyr = dy + row * D;
    const bf16* xr = x + row * D;
    bf16* dxr = dx + row * D;
    const float m = mean[row], rs = rstd[row];
    const int i = lane * VEC;

    if (VARIANT == 4) {  // pure copy: dy -> dx
#pragma unroll
      for (int k = 0; k < VEC; ++k) dxr[i + k] = dyr[i + k];
      continue;
    }

    float c1 = 0.f, c2 = 0.f;
    if (VARIANT <= 2) {
#pragma unroll
      for (int k = 0; k < VEC; ++k) {
        float g = __bfloat162float(dyr[i + k]);
        float xhat = (__bfloat162float(xr[i + k]) - m) * rs;
        float gw = g * w[i + k];
        c1 += gw;
        c2 += gw * xhat;
        if (VARIANT <= 1) {
          dw_loc[k] += g * xhat;
          db_loc[k] += g;
        }
      }
      c1 = gsum(c1) / D;
      c2 = gsum(c2) / D;
    }

#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      float g = __bfloat162float(dyr[i + k]);
      float xhat = (__bfloat162float(xr[i + k]) - m) * rs;
      dxr[i + k] = (bf16)(rs * (g * w[i + k] - c1 - xhat * c2));
    }
  }

  if (VARIANT == 0) {
    const int i = lane * VEC;
#pragma unroll
    for (int k = 0; k < VEC; ++k) {
      atomicAdd(&dw[i + k], dw_loc[k]);
      atomicAdd(&db[i + k], db_loc[k]);
    }
  } else {
    // keep the accumulators alive without atomics
    asm volatile("" ::"v"(dw_loc[0]), "v"(db_loc[0]));
  }
}

int main() {
  const long rows = 65536;
  const int D = 256;
  bf16 *dy, *x, *dx;
  float *w, *mean, *rstd, *dw, *db;
  hipMalloc(&dy, rows * D * 2);
  hipMalloc(&x, rows * D * 2);
  hipMalloc(&dx, rows * D * 2);
  hipMalloc(&w, D * 4);
  hipMalloc(&mean, rows * 4);
  hipMalloc(&rstd, rows * 4);
  hipMalloc(&dw, D * 4);
  hipMalloc(&db, D * 4);
  hipMemset(dy, 0x3c, rows * D * 2);
  hipMemset(x, 0x3c, rows * D * 2);
  hipMemset(w, 0, D * 4);
  hipMemset(mean, 0, rows * 4);
  hipMemset(rstd, 0, rows * 4);

  const int block = 256, RPB = block / GROUP;

  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
#define BENCH(V, GRID)                                                       \
  {                                                                          \
    for (int it = 0; it < 3; ++it)                                           \
      hipLaunchKernelGGL((ln_bwd<V>), dim3(GRID), dim3(block), 0, 0, dy, x,  \
                         w, mean, rstd, dx, dw, db, rows, D);                \
    hipDeviceSynchronize();                                                  \
    hipEventRecord(a);                                                       \
    for (int it = 0; it < 20; ++it)                                          \
      hipLaunchKernelGGL((ln_bwd<V>), dim3(GRID), dim3(block), 0, 0, dy, x,  \
                         w, mean, rstd, dx, dw, db, rows, D);                \
    hipEventRecord(b);                                                       \
    hipEventSynchronize(b);                                                  \
    float ms;                                                                \
    hipEventElapsedTime(&ms, a, b);                                          \
    double bytes = (V == 4 ? 2.0 : 5.0) * rows * D * 2.0;                    \
    printf("V%d grid=%d: %.1f us  (%.2f TB/s)\n", V, GRID, ms * 50,          \
           bytes / (ms / 20 * 1e-3) / 1e12);                                 \
  }

  BENCH(0, 2048);
  BENCH(1, 2048);
  BENCH(2, 2048);
  BENCH(3, 2048);
  BENCH(4, 2048);
  BENCH(0, 8192);
  BENCH(3, 8192);
  BENCH(4, 8192);
  BENCH(0, 512);
  return 0;
}
