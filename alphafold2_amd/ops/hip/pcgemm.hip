// Per-channel batched GEMM — gfx950 MFMA (K3 + K4 of SURVEY.md §2.17).
//
//   C[b, m, n, d] = sum_k A[b, ., ., d] * B[b, ., ., d]
//
// with the m/k (resp. n/k) axes of A (resp. B) selected by ELEMENT
// STRIDES, so one kernel covers, without any permute copies:
//   * triangle multiplicative mix, outgoing and ingoing
//     (reference alphafold2.py:313: '... i k d, ... j k d -> ... i j d')
//   * outer-product mean accumulation (reference :341-349)
//   * all four backward products of the above (dL, dR by swapping
//     operand roles/strides)
//
// The d axis is innermost/contiguous in every tensor; a block owns an
// 8-channel d-chunk and a 32x32 output tile.  Staging loads 16 bytes
// (8 channels) per (m, k) element — per-lane strided at D*2B, with the
// sibling d-chunks of the same cacheline served by L2/L3 (inputs for a
// whole batch entry fit the 256 MiB Infinity Cache).  LDS holds per-
// channel [32][32] planes with +8 element row padding (2-way conflicts
// only).  Each of 4 waves computes 2 channels with 16x16x32 bf16 MFMA.
// The C tile round-trips through LDS so global writes are 16-byte,
// d-contiguous.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef __bf16 pb16;

namespace {

constexpr int TM = 32, TN = 32, TK = 32, DC = 8;
constexpr int PLANE_ROW = TK + 8;           // padded row, elements
constexpr int PLANE = TM * PLANE_ROW;       // elements per channel plane

__global__ __launch_bounds__(256, 2)
void pcgemm_kernel(const pb16* __restrict__ A, const pb16* __restrict__ B,
                   pb16* __restrict__ C,
                   int M, int N, int K, int D,
                   long a_bs, long a_ms, long a_ks,
                   long b_bs, long b_ns, long b_ks,
                   int ntiles, float alpha) {
  __shared__ pb16 a_lds[DC * PLANE];
  __shared__ pb16 b_lds[DC * PLANE];
  __shared__ pb16 c_lds[TM * TN * DC];

  const int mtile = blockIdx.x / ntiles;
  const int ntile = blockIdx.x - mtile * ntiles;
  const int m0 = mtile * TM, n0 = ntile * TN;
  const int dchunks = D / DC;
  const int batch = blockIdx.y / dchunks;
  const int d0 = (blockIdx.y - batch * dchunks) * DC;

  const pb16* Ab = A + (long)batch * a_bs + d0;
  const pb16* Bb_ = B + (long)batch * b_bs + d0;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  f32x4_t acc[2][2][2];  // [dd][mt][nt]
#pragma unroll
  for (int dd = 0; dd < 2; ++dd)
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt) acc[dd][mt][nt] = f32x4_t{0, 0, 0, 0};

  const int m_rows = min(TM, M - m0);
  const int n_rows = min(TN, N - n0);

  for (int k0 = 0; k0 < K; k0 += TK) {
    const int k_rows = min(TK, K - k0);
    __syncthreads();
    // stage A and B tiles: each thread 4 (row, k) elements per matrix,
    // 16B of 8 channels each, scattered into the 8 LDS planes
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      int idx = threadIdx.x + pass * 256;   // 0..1023
      int r = idx >> 5;                     // row in tile
      int kk = idx & 31;
      bf16x8_t av = {};
      bf16x8_t bv = {};
      if (r < m_rows && kk < k_rows) {
        av = *reinterpret_cast<const bf16x8_t*>(
            Ab + (long)(m0 + r) * a_ms + (long)(k0 + kk) * a_ks);
      }
      if (r < n_rows && kk < k_rows) {
        bv = *reinterpret_cast<const bf16x8_t*>(
            Bb_ + (long)(n0 + r) * b_ns + (long)(k0 + kk) * b_ks);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        a_lds[j * PLANE + r * PLANE_ROW + kk] = av[j];
        b_lds[j * PLANE + r * PLANE_ROW + kk] = bv[j];
      }
    }
    __syncthreads();

    // MFMA: wave handles channels 2*wave and 2*wave+1
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dd = 0; dd < 2; ++dd) {
      const pb16* ap = a_lds + (2 * wave + dd) * PLANE;
      const pb16* bp = b_lds + (2 * wave + dd) * PLANE;
#pragma unroll
      for (int mt = 0; mt < 2; ++mt) {
        // A fragment: row = mt*16 + (lane&15), k = (lane>>4)*8 + j
        bf16x8_t af = *reinterpret_cast<const bf16x8_t*>(
            ap + (mt * 16 + (lane & 15)) * PLANE_ROW + ((lane >> 4) << 3));
#pragma unroll
        for (int nt = 0; nt < 2; ++nt) {
          bf16x8_t bf = *reinterpret_cast<const bf16x8_t*>(
              bp + (nt * 16 + (lane & 15)) * PLANE_ROW + ((lane >> 4) << 3));
          acc[dd][mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bf, acc[dd][mt][nt], 0, 0, 0);
        }
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // C tile -> LDS in [m][n][d] layout, then 16B coalesced global writes
  __syncthreads();
#pragma unroll
  for (int dd = 0; dd < 2; ++dd) {
    const int d = 2 * wave + dd;
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int nt = 0; nt < 2; ++nt)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int row = mt * 16 + (lane >> 4) * 4 + reg;
          const int col = nt * 16 + (lane & 15);
          c_lds[(row * TN + col) * DC + d] =
              (pb16)(acc[dd][mt][nt][reg] * alpha);
        }
  }
  __syncthreads();
  // 32*32 (m,n) cells * 16B = 1024 chunks; 4 per thread
  pb16* Cb = C + (((long)batch * M) * N) * D + d0;
#pragma unroll
  for (int pass = 0; pass < 4; ++pass) {
    int idx = threadIdx.x + pass * 256;
    int r = idx >> 5;
    int cc = idx & 31;
    if (r < m_rows && cc < n_rows) {
      *reinterpret_cast<bf16x8_t*>(
          Cb + ((long)(m0 + r) * N + (n0 + cc)) * D) =
          *reinterpret_cast<const bf16x8_t*>(c_lds + (r * TN + cc) * DC);
    }
  }
}

}  // namespace

// C (Bb, M, N, D) bf16 = alpha * sum_k A*B per channel; strides in elements.
at::Tensor pcgemm(at::Tensor A, at::Tensor B, long Bb, long M, long N,
                  long K, long D,
                  long a_bs, long a_ms, long a_ks,
                  long b_bs, long b_ns, long b_ks, double alpha) {
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 &&
              B.scalar_type() == at::kBFloat16, "pcgemm: bf16 only");
  TORCH_CHECK(D % DC == 0, "pcgemm: D must be a multiple of 8");
  auto C = at::empty({Bb, M, N, D}, A.options());
  const int mtiles = (M + TM - 1) / TM;
  const int ntiles = (N + TN - 1) / TN;
  dim3 grid(mtiles * ntiles, Bb * (D / DC));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(pcgemm_kernel, grid, dim3(256), 0, stream,
                     reinterpret_cast<const pb16*>(A.data_ptr()),
                     reinterpret_cast<const pb16*>(B.data_ptr()),
                     reinterpret_cast<pb16*>(C.data_ptr()),
                     (int)M, (int)N, (int)K, (int)D,
                     a_bs, a_ms, a_ks, b_bs, b_ns, b_ks,
                     ntiles, (float)alpha);
  return C;
}
