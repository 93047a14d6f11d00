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
// 8-channel d-chunk and a 64x64 output tile (8 waves, one channel per
// wave).  Staging loads 16 bytes (8 channels) per (m, k) element —
// per-lane strided at D*2B, with the sibling d-chunks of the same
// cacheline served by L2/L3 (inputs for a batch entry fit the 256 MiB
// Infinity Cache).  LDS holds per-channel [64][32] planes with +8
// element row padding; the C tile reuses the staging pool and drains
// as 16-byte d-contiguous global writes.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef __bf16 pb16;

namespace {

constexpr int TM = 64, TN = 64, TK = 32, DC = 8;
constexpr int NTHREADS = 512;               // 8 waves; one d-channel each
constexpr int PLANE_ROW = TK + 8;           // padded row, elements
constexpr int PLANE = TM * PLANE_ROW;       // elements per channel plane

__global__ __launch_bounds__(NTHREADS, 2)
void pcgemm_kernel(const pb16* __restrict__ A, const pb16* __restrict__ B,
                   pb16* __restrict__ C,
                   int M, int N, int K, int D,
                   long a_bs, long a_ms, long a_ks,
                   long b_bs, long b_ns, long b_ks,
                   int ntiles, int mtiles, int swizzle, float alpha) {
  // a/b staging planes and the C drain buffer share one pool: the
  // C round-trip begins only after the K loop has consumed a/b
  __shared__ pb16 pool[2 * DC * PLANE > TM * TN * DC
                       ? 2 * DC * PLANE : TM * TN * DC];
  pb16* a_lds = pool;
  pb16* b_lds = pool + DC * PLANE;
  pb16* c_lds = pool;

  // logical work id L enumerates (batch, tile, chunk) with the d-chunk
  // fastest; the XCD swizzle places the 8 sibling chunks of a tile
  // (which share 128-byte cachelines of A/B/C) on ONE XCD's L2 —
  // without it each sibling re-fetches the same line into a different
  // XCD L2 (8x HBM/L3 traffic; MfmaUtil measured at 3%).
  // phys = (G%8) + 8*(m + 8*(G/8)) for L = 8G + m  (bijective when the
  // grid is a multiple of 64; identity otherwise).
  long L = (long)blockIdx.y * gridDim.x + blockIdx.x;
  if (swizzle) {
    const long phys = L;
    const long c = phys & 7;
    const long t = phys >> 3;
    const long m_ = t & 7;
    const long Ghi = t >> 3;
    L = ((Ghi << 3) + c) * 8 + m_;
  }
  const int dchunks = D / DC;
  const int chunk = L % dchunks;
  const long Lt = L / dchunks;
  const int tile = Lt % ((long)mtiles * ntiles);
  const int batch = Lt / ((long)mtiles * ntiles);
  const int mtile = tile / ntiles;
  const int ntile = tile - mtile * ntiles;
  const int m0 = mtile * TM, n0 = ntile * TN;
  const int d0 = chunk * DC;

  const pb16* Ab = A + (long)batch * a_bs + d0;
  const pb16* Bb_ = B + (long)batch * b_bs + d0;

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;   // = d channel within the chunk

  f32x4_t acc[4][4];  // [mt][nt] 16x16 fragments of the 64x64 tile
#pragma unroll
  for (int mt = 0; mt < 4; ++mt)
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) acc[mt][nt] = f32x4_t{0, 0, 0, 0};

  const int m_rows = min(TM, M - m0);
  const int n_rows = min(TN, N - n0);

  for (int k0 = 0; k0 < K; k0 += TK) {
    const int k_rows = min(TK, K - k0);
    __syncthreads();
    // stage A and B tiles: 64x32 (row, k) elements each, one 16-byte
    // 8-channel load per element, scattered into the 8 LDS planes
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      int idx = threadIdx.x + pass * NTHREADS;  // 0..2047
      int r = idx >> 5;                         // row in tile
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

    // MFMA: each wave owns one d channel of the full 64x64 tile
    const pb16* ap = a_lds + wave * PLANE;
    const pb16* bp = b_lds + wave * PLANE;
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < 4; ++mt) {
      // A fragment: row = mt*16 + (lane&15), k = (lane>>4)*8 + j
      bf16x8_t af = *reinterpret_cast<const bf16x8_t*>(
          ap + (mt * 16 + (lane & 15)) * PLANE_ROW + ((lane >> 4) << 3));
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        bf16x8_t bf = *reinterpret_cast<const bf16x8_t*>(
            bp + (nt * 16 + (lane & 15)) * PLANE_ROW + ((lane >> 4) << 3));
        acc[mt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af, bf, acc[mt][nt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // C tile -> LDS in [m][n][d] layout, then 16B coalesced global writes
  __syncthreads();
  const int d = wave;
#pragma unroll
  for (int mt = 0; mt < 4; ++mt)
#pragma unroll
    for (int nt = 0; nt < 4; ++nt)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = mt * 16 + (lane >> 4) * 4 + reg;
        const int col = nt * 16 + (lane & 15);
        c_lds[(row * TN + col) * DC + d] =
            (pb16)(acc[mt][nt][reg] * alpha);
      }
  __syncthreads();
  // 64*64 (m,n) cells * 16B = 4096 chunks; 8 per thread
  pb16* Cb = C + (((long)batch * M) * N) * D + d0;
#pragma unroll
  for (int pass = 0; pass < 8; ++pass) {
    int idx = threadIdx.x + pass * NTHREADS;
    int r = idx >> 6;
    int cc = idx & 63;
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
  const long nblocks = (long)mtiles * ntiles * Bb * (D / DC);
  const int swizzle = (D / DC) % 8 == 0 && nblocks % 64 == 0;
  dim3 grid(mtiles * ntiles, Bb * (D / DC));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(pcgemm_kernel, grid, dim3(NTHREADS), 0, stream,
                     reinterpret_cast<const pb16*>(A.data_ptr()),
                     reinterpret_cast<const pb16*>(B.data_ptr()),
                     reinterpret_cast<pb16*>(C.data_ptr()),
                     (int)M, (int)N, (int)K, (int)D,
                     a_bs, a_ms, a_ks, b_bs, b_ns, b_ks,
                     ntiles, mtiles, swizzle, (float)alpha);
  return C;
}
