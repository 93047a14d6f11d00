// Fused pair-representation build — gfx950 (K13 of SURVEY.md §2.17,
// reference alphafold2.py:715-726):
//
//   out[b, i, j, :] = left[b, i, :] + right[b, j, :] + emb[rel[b, i, j], :]
//
// The eager composition materializes the broadcast outer sum AND the
// gathered positional embedding before adding them (3 full (b,n,n,d)
// passes); this kernel writes the sum once.  One block per (b, i) row:
// the left row is staged in LDS, right rows and embedding rows stream
// through L2 (both are tiny and hot).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef __bf16 pb16;

namespace {

template <typename T>
__global__ void pairrep_fwd_kernel(const T* __restrict__ left,
                                   const T* __restrict__ right,
                                   const T* __restrict__ emb,
                                   const long* __restrict__ rel,
                                   T* __restrict__ out,
                                   int n, int d) {
  extern __shared__ char smem[];
  T* lrow = reinterpret_cast<T*>(smem);
  const long bi = blockIdx.x;          // b * n + i
  const long b = bi / n;
  const T* lg = left + bi * (long)d;
  for (int c = threadIdx.x; c < d; c += blockDim.x) lrow[c] = lg[c];
  __syncthreads();

  const T* rb = right + b * (long)n * d;
  const long* relrow = rel + bi * (long)n;
  T* og = out + bi * (long)n * d;
  // thread t owns 8-wide chunk (t % chunks) of row j = t / chunks.
  // When chunks does not divide blockDim the trailing threads duplicate
  // a row (identical values written twice — benign).
  const int chunks = d / 8;
  const int c8 = (threadIdx.x % chunks) * 8;
  const int j0 = threadIdx.x / chunks;
  const int jstep = blockDim.x / chunks;
  for (int j = j0; j < n; j += jstep) {
    const long e = relrow[j];
    T rv[8], ev[8], ov[8];
    vload<T, 8>(rb + (long)j * d + c8, rv);
    vload<T, 8>(emb + e * (long)d + c8, ev);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      ov[k] = from_f32<T>(to_f32(lrow[c8 + k]) + to_f32(rv[k]) +
                          to_f32(ev[k]));
    }
    vstore<T, 8>(og + (long)j * d + c8, ov);
  }
}

}  // namespace

// left/right (b, n, d); emb (V, d); rel (b, n, n) long -> out (b, n, n, d)
at::Tensor pairrep_fwd(at::Tensor left, at::Tensor right, at::Tensor emb,
                       at::Tensor rel) {
  TORCH_CHECK(left.is_contiguous() && right.is_contiguous() &&
              emb.is_contiguous() && rel.is_contiguous(),
              "pairrep_fwd: contiguous inputs required");
  TORCH_CHECK(rel.scalar_type() == at::kLong, "pairrep_fwd: rel must be long");
  const int b = left.size(0), n = left.size(1), d = left.size(2);
  TORCH_CHECK(d % 8 == 0, "pairrep_fwd: d must be a multiple of 8");
  TORCH_CHECK(d <= 2048, "pairrep_fwd: d > 2048 not covered (one block "
              "row owns d/8 chunks across 256 threads)");
  TORCH_CHECK(emb.size(1) == d && right.size(2) == d, "pairrep_fwd: dim");
  auto out = at::empty({b, n, n, d}, left.options());
  const int block = 256;
  const int smem = d * left.element_size();
  auto stream = at::cuda::getCurrentHIPStream();

#define LAUNCH(T)                                                         \
  hipLaunchKernelGGL((pairrep_fwd_kernel<T>), dim3((long)b * n),          \
                     dim3(block), smem, stream,                           \
                     reinterpret_cast<const T*>(left.data_ptr()),         \
                     reinterpret_cast<const T*>(right.data_ptr()),        \
                     reinterpret_cast<const T*>(emb.data_ptr()),          \
                     rel.data_ptr<long>(),                                \
                     reinterpret_cast<T*>(out.data_ptr()), n, d)
  if (left.scalar_type() == at::kBFloat16) LAUNCH(__hip_bfloat16);
  else if (left.scalar_type() == at::kFloat) LAUNCH(float);
  else if (left.scalar_type() == at::kHalf) LAUNCH(__half);
  else TORCH_CHECK(false, "pairrep_fwd: unsupported dtype");
#undef LAUNCH
  return out;
}
