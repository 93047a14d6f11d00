// Fused tall-M / small-K linear GEMM — gfx950 MFMA (K6 of SURVEY.md
// §2.17, reference alphafold2.py:84-94).
//
// The Evoformer's Linear layers are extremely tall and skinny in K
// (M = b*m*n up to ~300k rows, K = dim = 256, N = 128..2048).  Tensile
// picks deep-K macro-tiles that cannot amortize at K=256 and lands at
// ~250 TF; this kernel is shaped for exactly these GEMMs:
//
//   * A (M, K) row-major and W (N, K) row-major (the nn.Linear weight
//     layout) both feed MFMA fragments as contiguous 8-element K runs —
//     no transpose, no repack.
//   * 128x128 output tile, BK=64, 4 waves, each wave computing a 32x128
//     slab as 2x8 16x16x32 fragments (guide §5 ladder structure).
//   * epilogues fused: bias add, GEGLU pair gating (out = a * gelu(g),
//     halving the hot write), residual add — each a template variant so
//     the elementwise passes disappear from the step.
//
// XCD-aware: logical work ids are swizzled (bijective, guide m204) so
// each XCD's L2 sees a contiguous run of M-tiles (A strips read once
// per XCD; W fits L2 outright).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8_t __attribute__((ext_vector_type(8)));
typedef float f32x4_t __attribute__((ext_vector_type(4)));
typedef __bf16 pb16;

namespace {

constexpr int BM = 128;        // output rows per block
constexpr int BN = 128;        // staged B rows (=output cols, or val+gate)
constexpr int BK = 64;         // K step
constexpr int PR = BK + 8;     // padded LDS row (elements) — 2-way banks
constexpr int NT = 256;        // 4 waves

enum Epi { EPI_NONE = 0, EPI_GEGLU = 1, EPI_RESID = 2 };

// bijective XCD swizzle (guide m204): phys blockIdx -> logical work id
// such that each XCD owns a contiguous logical range.
__device__ __forceinline__ long xcd_logical(long phys, long nwg) {
  if (nwg < 16) return phys;
  const long q = nwg >> 3, r = nwg & 7;
  const long xcd = phys & 7, idx = phys >> 3;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

template <int EPI, bool HAS_BIAS>
__global__ __launch_bounds__(NT, 2)
void linear_gemm_kernel(const pb16* __restrict__ A,
                        const pb16* __restrict__ W,
                        const pb16* __restrict__ bias,
                        const pb16* __restrict__ resid,
                        pb16* __restrict__ out,
                        pb16* __restrict__ inter,   // GEGLU: raw (M, N)
                        int M, int N, int K, int mtiles, int ntiles) {
  // staging pool: A[128][72] + B[128][72] bf16 = 36 KB; the C restage
  // ([128][128] bf16 = 32 KB) reuses it after the K loop
  __shared__ pb16 pool[2 * BM * PR];
  pb16* a_lds = pool;
  pb16* b_lds = pool + BM * PR;
  pb16* c_lds = pool;

  const long nwg = (long)mtiles * ntiles;
  long L = xcd_logical((long)blockIdx.x, nwg);
  const int mtile = L / ntiles;
  const int ntile = L - (long)mtile * ntiles;
  const int m0 = mtile * BM;
  // GEGLU: the block's 64 output cols pair W rows [n0, n0+64) with
  // [N/2 + n0, N/2 + n0 + 64); plain: 128 cols [n0, n0+128)
  const int half = N >> 1;
  const int n0 = EPI == EPI_GEGLU ? ntile * 64 : ntile * BN;
  const int m_rows = min(BM, M - m0);
  const int n_cols = EPI == EPI_GEGLU ? min(64, half - n0)
                                      : min(BN, N - n0);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int wr = wave * 32;          // wave's row base in the tile

  f32x4_t acc[2][8];
#pragma unroll
  for (int mt = 0; mt < 2; ++mt)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) acc[mt][nt] = f32x4_t{0, 0, 0, 0};

  for (int k0 = 0; k0 < K; k0 += BK) {
    const int k_rows = min(BK, K - k0);
    __syncthreads();
    // stage A and B tiles: 128 rows x 64 k each as 8-element chunks;
    // 1024 chunks per tile = 4 per thread
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int idx = threadIdx.x + pass * NT;
      const int row = idx >> 3;
      const int kc = (idx & 7) << 3;
      bf16x8_t av = {};
      if (row < m_rows && kc < k_rows)
        av = *reinterpret_cast<const bf16x8_t*>(
            A + (long)(m0 + row) * K + k0 + kc);
      *reinterpret_cast<bf16x8_t*>(a_lds + row * PR + kc) = av;

      bf16x8_t bv = {};
      int grow;
      bool ok;
      if (EPI == EPI_GEGLU) {
        grow = row < 64 ? n0 + row : half + n0 + (row - 64);
        ok = (row & 63) < n_cols;
      } else {
        grow = n0 + row;
        ok = row < n_cols;
      }
      if (ok && kc < k_rows)
        bv = *reinterpret_cast<const bf16x8_t*>(
            W + (long)grow * K + k0 + kc);
      *reinterpret_cast<bf16x8_t*>(b_lds + row * PR + kc) = bv;
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

  // ---- epilogue -----------------------------------------------------
  // C/D fragment: col = lane&15, row = (lane>>4)*4 + reg
  const int fcol = lane & 15;
  const int frow = (lane >> 4) << 2;

  float bval[8];
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) {
    bval[nt] = 0.f;
    if (HAS_BIAS) {
      int col = nt * 16 + fcol;   // staged-B row index of this column
      int gcol = EPI == EPI_GEGLU
                     ? (col < 64 ? n0 + col : half + n0 + (col - 64))
                     : n0 + col;
      bool ok = EPI == EPI_GEGLU ? (col & 63) < n_cols : col < n_cols;
      if (ok) bval[nt] = to_f32(bias[gcol]);
    }
  }

  __syncthreads();   // staging pool -> C restage

  if (EPI == EPI_GEGLU) {
    // raw pre-activation (val | gate) goes straight to `inter` (needed
    // by backward); the gated product is restaged for coalesced writes
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int nt = 0; nt < 4; ++nt)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int row = wr + mt * 16 + frow + reg;
          const int col = nt * 16 + fcol;
          const float val = acc[mt][nt][reg] + bval[nt];
          const float gat = acc[mt][nt + 4][reg] + bval[nt + 4];
          if (row < m_rows && col < n_cols) {
            inter[(long)(m0 + row) * N + n0 + col] = (pb16)val;
            inter[(long)(m0 + row) * N + half + n0 + col] = (pb16)gat;
          }
          c_lds[(row << 6) + col] = (pb16)(val * gelu_f(gat));
        }
    __syncthreads();
    // drain 128x64 tile: 1024 8-chunks, 4 per thread
#pragma unroll
    for (int pass = 0; pass < 4; ++pass) {
      const int idx = threadIdx.x + pass * NT;
      const int row = idx >> 3;
      const int col = (idx & 7) << 3;
      if (row < m_rows && col < n_cols)
        *reinterpret_cast<bf16x8_t*>(
            out + (long)(m0 + row) * half + n0 + col) =
            *reinterpret_cast<const bf16x8_t*>(c_lds + (row << 6) + col);
    }
  } else {
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int row = wr + mt * 16 + frow + reg;
          const int col = nt * 16 + fcol;
          c_lds[(row << 7) + col] = (pb16)(acc[mt][nt][reg] + bval[nt]);
        }
    __syncthreads();
    // drain 128x128 tile: 2048 8-chunks, 8 per thread
#pragma unroll
    for (int pass = 0; pass < 8; ++pass) {
      const int idx = threadIdx.x + pass * NT;
      const int row = idx >> 4;
      const int col = (idx & 15) << 3;
      if (row < m_rows && col < n_cols) {
        bf16x8_t o = *reinterpret_cast<const bf16x8_t*>(
            c_lds + (row << 7) + col);
        if (EPI == EPI_RESID) {
          bf16x8_t r = *reinterpret_cast<const bf16x8_t*>(
              resid + (long)(m0 + row) * N + n0 + col);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            o[j] = (pb16)(to_f32(o[j]) + to_f32(r[j]));
        }
        *reinterpret_cast<bf16x8_t*>(
            out + (long)(m0 + row) * N + n0 + col) = o;
      }
    }
  }
}

}  // namespace

// out = x @ W^T (+ bias) (+ residual); x (M, K), W (N, K) — bf16.
at::Tensor linear_fwd(at::Tensor x, at::Tensor W,
                      c10::optional<at::Tensor> bias,
                      c10::optional<at::Tensor> resid) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              W.scalar_type() == at::kBFloat16, "linear_fwd: bf16 only");
  TORCH_CHECK(x.is_contiguous() && W.is_contiguous(),
              "linear_fwd: contiguous inputs required");
  const long M = x.numel() / x.size(-1);
  const int K = x.size(-1);
  const int N = W.size(0);
  TORCH_CHECK(W.size(1) == K, "linear_fwd: K mismatch");
  TORCH_CHECK(K % 8 == 0 && N % 8 == 0,
              "linear_fwd: K and N must be multiples of 8");

  auto sizes = x.sizes().vec();
  sizes.back() = N;
  auto out = at::empty(sizes, x.options());
  if (resid.has_value())
    TORCH_CHECK(resid->is_contiguous() && resid->numel() == out.numel(),
                "linear_fwd: residual shape mismatch");

  const int mtiles = (M + BM - 1) / BM;
  const int ntiles = (N + BN - 1) / BN;
  dim3 grid((long)mtiles * ntiles);
  auto stream = at::cuda::getCurrentHIPStream();
  const pb16* bp = bias.has_value()
      ? reinterpret_cast<const pb16*>(bias->data_ptr()) : nullptr;
  const pb16* rp = resid.has_value()
      ? reinterpret_cast<const pb16*>(resid->data_ptr()) : nullptr;

#define LAUNCH(EPI, HB)                                                  \
  hipLaunchKernelGGL((linear_gemm_kernel<EPI, HB>), grid, dim3(NT), 0,   \
                     stream, reinterpret_cast<const pb16*>(x.data_ptr()),\
                     reinterpret_cast<const pb16*>(W.data_ptr()), bp, rp,\
                     reinterpret_cast<pb16*>(out.data_ptr()), nullptr,   \
                     (int)M, N, K, mtiles, ntiles)
  if (resid.has_value()) {
    if (bp) LAUNCH(EPI_RESID, true); else LAUNCH(EPI_RESID, false);
  } else {
    if (bp) LAUNCH(EPI_NONE, true); else LAUNCH(EPI_NONE, false);
  }
#undef LAUNCH
  return out;
}

// GEGLU-fused FF1: returns (out (…, N/2) = a * gelu(g), inter (…, N))
// where inter = x @ W^T + bias is the raw pre-activation (for backward).
std::vector<at::Tensor> ff1_geglu_fwd(at::Tensor x, at::Tensor W,
                                      c10::optional<at::Tensor> bias) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              W.scalar_type() == at::kBFloat16, "ff1_geglu_fwd: bf16 only");
  TORCH_CHECK(x.is_contiguous() && W.is_contiguous(),
              "ff1_geglu_fwd: contiguous inputs required");
  const long M = x.numel() / x.size(-1);
  const int K = x.size(-1);
  const int N = W.size(0);
  TORCH_CHECK(W.size(1) == K && N % 16 == 0, "ff1_geglu_fwd: bad W shape");
  TORCH_CHECK(K % 8 == 0, "ff1_geglu_fwd: K must be a multiple of 8");

  auto osz = x.sizes().vec();
  osz.back() = N / 2;
  auto out = at::empty(osz, x.options());
  auto isz = x.sizes().vec();
  isz.back() = N;
  auto inter = at::empty(isz, x.options());

  const int mtiles = (M + BM - 1) / BM;
  const int ntiles = (N / 2 + 63) / 64;
  dim3 grid((long)mtiles * ntiles);
  auto stream = at::cuda::getCurrentHIPStream();
  const pb16* bp = bias.has_value()
      ? reinterpret_cast<const pb16*>(bias->data_ptr()) : nullptr;

#define LAUNCH(HB)                                                       \
  hipLaunchKernelGGL((linear_gemm_kernel<EPI_GEGLU, HB>), grid,          \
                     dim3(NT), 0, stream,                                \
                     reinterpret_cast<const pb16*>(x.data_ptr()),        \
                     reinterpret_cast<const pb16*>(W.data_ptr()), bp,    \
                     nullptr, reinterpret_cast<pb16*>(out.data_ptr()),   \
                     reinterpret_cast<pb16*>(inter.data_ptr()),          \
                     (int)M, N, K, mtiles, ntiles)
  if (bp) LAUNCH(true); else LAUNCH(false);
#undef LAUNCH
  return {out, inter};
}
