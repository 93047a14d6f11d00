// Fused cdist + bucketize — gfx950 (K8 of SURVEY.md §2.17).
//
// Used by recycling distance embedding (reference alphafold2.py:734-737)
// and distogram target construction (reference utils.py:45-50).  One
// pass: pairwise distance + binary search over the bin edges, no (b,n,n)
// fp32 distance tensor materialized.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace {

template <typename T>
__global__ void dist_buckets_kernel(const T* __restrict__ coords,
                                    const float* __restrict__ bounds,
                                    long* __restrict__ out,
                                    int b, int n, int nbounds) {
  const long total = (long)b * n * n;
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long bi = idx / ((long)n * n);
    const int rem = idx - bi * n * n;
    const int i = rem / n;
    const int j = rem - i * n;

    const T* ci = coords + (bi * n + i) * 3;
    const T* cj = coords + (bi * n + j) * 3;
    const float dx = to_f32(ci[0]) - to_f32(cj[0]);
    const float dy = to_f32(ci[1]) - to_f32(cj[1]);
    const float dz = to_f32(ci[2]) - to_f32(cj[2]);
    const float d = sqrtf(dx * dx + dy * dy + dz * dz);

    // torch.bucketize (right=False): count of bounds <= d ... precisely
    // index of first bound >= d is the result when right=False? torch
    // semantics: out[i] = number of bounds b_k with b_k < d ... for
    // right=False boundaries[j-1] < v <= boundaries[j] -> j.
    int lo = 0, hi = nbounds;
    while (lo < hi) {
      int mid = (lo + hi) >> 1;
      if (bounds[mid] < d) lo = mid + 1; else hi = mid;
    }
    out[idx] = lo;
  }
}

}  // namespace

at::Tensor dist_buckets(at::Tensor coords, at::Tensor boundaries) {
  TORCH_CHECK(coords.is_contiguous(), "dist_buckets: coords must be contiguous");
  TORCH_CHECK(coords.size(-1) == 3, "dist_buckets: coords must be (..., 3)");
  const int n = coords.size(-2);
  const int b = coords.numel() / (3 * n);
  auto bf = boundaries.to(at::kFloat).contiguous();
  auto out = at::empty({b, n, n}, coords.options().dtype(at::kLong));

  const long total = (long)b * n * n;
  const int block = 256;
  long grid = (total + block - 1) / block;
  if (grid > 2048) grid = 2048;
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH(T)                                                         \
  hipLaunchKernelGGL((dist_buckets_kernel<T>), dim3(grid), dim3(block),   \
                     0, stream,                                           \
                     reinterpret_cast<const T*>(coords.data_ptr()),       \
                     bf.data_ptr<float>(), out.data_ptr<long>(), b, n,    \
                     (int)bf.numel())

  if (coords.scalar_type() == at::kFloat) LAUNCH(float);
  else if (coords.scalar_type() == at::kBFloat16) LAUNCH(__hip_bfloat16);
  else if (coords.scalar_type() == at::kHalf) LAUNCH(__half);
  else TORCH_CHECK(false, "dist_buckets: unsupported dtype");
#undef LAUNCH
  return out;
}
