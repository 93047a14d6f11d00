// Split-K weight-gradient GEMM — gfx950 MFMA.
//
//   dW[m, n] = sum_k A[k, m] * B[k, n]      (A = dY, B = X activations)
//
// The Evoformer's wgrads reduce over K = b*m*n up to ~300k rows into
// tiny (N1 x N2) outputs — Tensile lands at ~350-475 TF on these
// (both operands "transposed", limited split-K).  This kernel stages
// k-major slabs of both operands transposed into LDS (coalesced 16-byte
// row loads; stores swizzled to kill read conflicts) and splits K
// across blocks with one fp32 atomic drain per block.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef __bf16 pb16;

namespace {

constexpr int WM = 128;        // dW rows per block (A columns)
constexpr int WN = 128;        // dW cols per block (B columns)
constexpr int KS = 64;         // k rows staged per iteration
constexpr int NT = 256;        // 4 waves; wave w owns rows [w*32, w*32+32)

// LDS tiles hold the operands TRANSPOSED: [m_or_n][k], padded rows
constexpr int PR = KS + 8;

__global__ __launch_bounds__(NT, 2)
void wgrad_kernel(const pb16* __restrict__ A, const pb16* __restrict__ B,
                  float* __restrict__ dW,
                  long K, int M, int N, long k_per_block,
                  int mtiles, int ntiles) {
  __shared__ pb16 a_lds[WM * PR];
  __shared__ pb16 b_lds[WN * PR];

  const int tile = blockIdx.x;
  const int kslot = blockIdx.y;
  const int mtile = tile / ntiles;
  const int ntile = tile - mtile * ntiles;
  const int m0 = mtile * WM;
  const int n0 = ntile * WN;
  const long k0 = (long)kslot * k_per_block;
  const long k1 = min(K, k0 + k_per_block);
  const int m_cols = min(WM, M - m0);
  const int n_cols = min(WN, N - n0);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave * 32;

  f32x4_t acc[2][8];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) acc[mt][nt] = f32x4_t{0, 0, 0, 0};

  for (long ks = k0; ks < k1; ks += KS) {
    const int k_rows = min((long)KS, k1 - ks);
    __syncthreads();
    // stage 64 k-rows of A (WM cols) and B (WN cols) transposed:
    // thread loads 8 consecutive columns of one k-row (16-byte,
    // coalesced along the tensor row) and scatters them into 8 LDS
    // rows at column k.  1024 chunks per operand = 4 per thread.
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int idx = threadIdx.x + pass * NT;
      const int kk = idx >> 4;              // k row within the slab
      const int c8 = (idx & 15) << 3;       // first of 8 source columns
      bf16x8_t av = {};
      if (kk < k_rows && c8 < m_cols)
        av = *reinterpret_cast<const bf16x8_t*>(
            A + (ks + kk) * (long)M + m0 + c8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        a_lds[(c8 + j) * PR + kk] = av[j];

      bf16x8_t bv = {};
      if (kk < k_rows && c8 < n_cols)
        bv = *reinterpret_cast<const bf16x8_t*>(
            B + (ks + kk) * (long)N + n0 + c8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        b_lds[(c8 + j) * PR + kk] = bv[j];
    }
    __syncthreads();

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int koff = kk * 32 + ((lane >> 4) << 3);
      bf16x8_t af0 = *reinterpret_cast<const bf16x8_t*>(
          a_lds + (wr + (lane & 15)) * PR + koff);
      bf16x8_t af1 = *reinterpret_cast<const bf16x8_t*>(
          a_lds + (wr + 16 + (lane & 15)) * PR + koff);
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        bf16x8_t bf = *reinterpret_cast<const bf16x8_t*>(
            b_lds + (nt * 16 + (lane & 15)) * PR + koff);
        acc[0][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af0, bf, acc[0][nt], 0, 0, 0);
        acc[1][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af1, bf, acc[1][nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // drain: fp32 atomics (one partial per k-slot); C/D frag col=lane&15,
  // row=(lane>>4)*4+reg
  const int fcol = lane & 15;
  const int frow = (lane >> 4) << 2;
  const bool single = gridDim.y == 1;
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = wr + mt * 16 + frow + reg;
        const int col = nt * 16 + fcol;
        if (row < m_cols && col < n_cols) {
          float* addr = dW + (long)(m0 + row) * N + n0 + col;
          if (single) *addr = acc[mt][nt][reg];
          else atomicAdd(addr, acc[mt][nt][reg]);
        }
      }
}

}  // namespace

// dW (M, N) fp32 = dY^T @ X with dY (K, M), X (K, N) bf16 row-major.
at::Tensor wgrad(at::Tensor dY, at::Tensor X) {
  TORCH_CHECK(dY.scalar_type() == at::kBFloat16 &&
              X.scalar_type() == at::kBFloat16, "wgrad: bf16 only");
  TORCH_CHECK(dY.is_contiguous() && X.is_contiguous(),
              "wgrad: contiguous inputs required");
  TORCH_CHECK(dY.dim() == 2 && X.dim() == 2 && dY.size(0) == X.size(0),
              "wgrad: (K, M) and (K, N) expected");
  const long K = dY.size(0);
  const int M = dY.size(1);
  const int N = X.size(1);
  TORCH_CHECK(M % 8 == 0 && N % 8 == 0, "wgrad: M, N multiples of 8");

  const int mtiles = (M + WM - 1) / WM;
  const int ntiles = (N + WN - 1) / WN;
  // enough k-slots to fill the chip (>= ~2048 blocks), each a multiple
  // of the 64-row stage
  long kslots = 2048 / ((long)mtiles * ntiles);
  kslots = std::max(1L, std::min(kslots, (K + KS - 1) / KS));
  long k_per_block = ((K + kslots - 1) / kslots + KS - 1) / KS * KS;
  kslots = (K + k_per_block - 1) / k_per_block;

  auto dW = kslots > 1
      ? at::zeros({M, N}, dY.options().dtype(at::kFloat))
      : at::empty({M, N}, dY.options().dtype(at::kFloat));
  dim3 grid(mtiles * ntiles, kslots);
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(wgrad_kernel, grid, dim3(NT), 0, stream,
                     reinterpret_cast<const pb16*>(dY.data_ptr()),
                     reinterpret_cast<const pb16*>(X.data_ptr()),
                     dW.data_ptr<float>(), K, M, N, k_per_block,
                     mtiles, ntiles);
  return dW;
}
