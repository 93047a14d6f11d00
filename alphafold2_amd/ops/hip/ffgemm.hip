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

// STAGE 0: manual reg staging into a +8-padded LDS row (2-way conflicts,
// VALU-heavy).  STAGE 1: 16-byte global_load_lds into a LINEAR [128][64]
// tile with the st_16x32 XOR swizzle applied to BOTH the global source
// mapping and the ds_read address (guide §5: swizzle must be
// both-sides-or-neither; the involution is byte ^= ((byte>>9)&1)<<5).
enum Stage { STAGE_PAD = 0, STAGE_GLDS = 1 };

__device__ __forceinline__ int swz(int byte) {
  return byte ^ (((byte >> 9) & 1) << 5);
}

template <int STAGE>
__device__ __forceinline__ int lds_off(int row, int k) {
  // byte offset of element (row, k) in one staged tile
  if (STAGE == STAGE_PAD) return (row * PR + k) * 2;
  return swz((row << 7) + (k << 1));
}

typedef __attribute__((address_space(3))) void lds_void;
typedef const __attribute__((address_space(1))) void g_void;

__device__ __forceinline__ void gload_lds16(const pb16* gsrc, char* lds,
                                            int lds_byte) {
  __builtin_amdgcn_global_load_lds(
      (g_void*)gsrc, (lds_void*)(lds + lds_byte), 16, 0, 0);
}

// bijective XCD swizzle (guide m204): phys blockIdx -> logical work id
// such that each XCD owns a contiguous logical range.
__device__ __forceinline__ long xcd_logical(long phys, long nwg) {
  if (nwg < 16) return phys;
  const long q = nwg >> 3, r = nwg & 7;
  const long xcd = phys & 7, idx = phys >> 3;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

template <int EPI, bool HAS_BIAS, int STAGE = STAGE_PAD>
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
  char* a_lds_b = reinterpret_cast<char*>(pool);
  char* b_lds_b = a_lds_b + (STAGE == STAGE_PAD ? BM * PR * 2 : BM * BK * 2);
  pb16* a_lds = pool;
  pb16* b_lds = reinterpret_cast<pb16*>(b_lds_b);
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

  // per-residue row mapping for staged B
  auto b_src_row = [&](int row) {
    if (EPI == EPI_GEGLU)
      return row < 64 ? n0 + row : half + n0 + (row - 64);
    return n0 + row;
  };
  auto b_row_ok = [&](int row) {
    if (EPI == EPI_GEGLU) return (row & 63) < n_cols;
    return row < n_cols;
  };

  for (int k0 = 0; k0 < K; k0 += BK) {
    const int k_rows = min(BK, K - k0);
    __syncthreads();
    if (STAGE == STAGE_PAD) {
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
        if (b_row_ok(row) && kc < k_rows)
          bv = *reinterpret_cast<const bf16x8_t*>(
              W + (long)b_src_row(row) * K + k0 + kc);
        *reinterpret_cast<bf16x8_t*>(b_lds + row * PR + kc) = bv;
      }
    } else {
      // async DMA staging: each wave fills 4 x 1024 B chunks per tile;
      // lane writes dest byte chunk*1024 + lane*16, whose swizzled
      // logical home decides the global source.  OOB rows are clamped
      // (garbage values only reach outputs the epilogue guards off);
      // requires K % 64 == 0 (host-checked).
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int d = (wave * 4 + i) * 1024 + lane * 16;
        const int lb = swz(d);
        const int row = lb >> 7;
        const int kb = (lb & 127) >> 1;
        const int ar = min(m0 + row, M - 1);
        gload_lds16(A + (long)ar * K + k0 + kb, a_lds_b, d);
        const int br = min(b_src_row(row), (EPI == EPI_GEGLU ? N : N) - 1);
        gload_lds16(W + (long)br * K + k0 + kb, b_lds_b, d);
      }
    }
    __syncthreads();

    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      const int koff = kk * 32 + ((lane >> 4) << 3);
      bf16x8_t af0 = *reinterpret_cast<const bf16x8_t*>(
          a_lds_b + lds_off<STAGE>(wr + (lane & 15), koff));
      bf16x8_t af1 = *reinterpret_cast<const bf16x8_t*>(
          a_lds_b + lds_off<STAGE>(wr + 16 + (lane & 15), koff));
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        bf16x8_t bf = *reinterpret_cast<const bf16x8_t*>(
            b_lds_b + lds_off<STAGE>(nt * 16 + (lane & 15), koff));
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
    // raw pre-activation (val | gate) goes straight to `inter` when the
    // caller needs it for backward (inference passes nullptr and skips
    // the 2x-width write); the gated product is restaged for coalesced
    // writes
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
          if (inter != nullptr && row < m_rows && col < n_cols) {
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
// stage: -1 auto (async DMA staging when K % 64 == 0), else force 0/1.
at::Tensor linear_fwd(at::Tensor x, at::Tensor W,
                      c10::optional<at::Tensor> bias,
                      c10::optional<at::Tensor> resid, long stage) {
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

  const bool glds = stage == 1 || (stage < 0 && K % 64 == 0);
#define LAUNCH(EPI, HB, ST)                                              \
  hipLaunchKernelGGL((linear_gemm_kernel<EPI, HB, ST>), grid, dim3(NT),  \
                     0, stream,                                          \
                     reinterpret_cast<const pb16*>(x.data_ptr()),        \
                     reinterpret_cast<const pb16*>(W.data_ptr()), bp, rp,\
                     reinterpret_cast<pb16*>(out.data_ptr()), nullptr,   \
                     (int)M, N, K, mtiles, ntiles)
#define LAUNCH_ST(EPI, HB)                                               \
  do { if (glds) LAUNCH(EPI, HB, STAGE_GLDS);                            \
       else LAUNCH(EPI, HB, STAGE_PAD); } while (0)
  if (resid.has_value()) {
    if (bp) LAUNCH_ST(EPI_RESID, true); else LAUNCH_ST(EPI_RESID, false);
  } else {
    if (bp) LAUNCH_ST(EPI_NONE, true); else LAUNCH_ST(EPI_NONE, false);
  }
#undef LAUNCH_ST
#undef LAUNCH
  return out;
}

// GEGLU-fused FF1: returns (out (…, N/2) = a * gelu(g), inter (…, N))
// where inter = x @ W^T + bias is the raw pre-activation (for backward).
std::vector<at::Tensor> ff1_geglu_fwd(at::Tensor x, at::Tensor W,
                                      c10::optional<at::Tensor> bias,
                                      long stage, bool want_inter) {
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
  at::Tensor inter;
  if (want_inter) {
    auto isz = x.sizes().vec();
    isz.back() = N;
    inter = at::empty(isz, x.options());
  }

  const int mtiles = (M + BM - 1) / BM;
  const int ntiles = (N / 2 + 63) / 64;
  dim3 grid((long)mtiles * ntiles);
  auto stream = at::cuda::getCurrentHIPStream();
  const pb16* bp = bias.has_value()
      ? reinterpret_cast<const pb16*>(bias->data_ptr()) : nullptr;

  const bool glds = stage == 1 || (stage < 0 && K % 64 == 0);
#define LAUNCH(HB, ST)                                                   \
  hipLaunchKernelGGL((linear_gemm_kernel<EPI_GEGLU, HB, ST>), grid,      \
                     dim3(NT), 0, stream,                                \
                     reinterpret_cast<const pb16*>(x.data_ptr()),        \
                     reinterpret_cast<const pb16*>(W.data_ptr()), bp,    \
                     nullptr, reinterpret_cast<pb16*>(out.data_ptr()),   \
                     want_inter ? reinterpret_cast<pb16*>(inter.data_ptr())\
                                : nullptr,                               \
                     (int)M, N, K, mtiles, ntiles)
#define LAUNCH_ST(HB)                                                    \
  do { if (glds) LAUNCH(HB, STAGE_GLDS); else LAUNCH(HB, STAGE_PAD); }   \
  while (0)
  if (bp) LAUNCH_ST(true); else LAUNCH_ST(false);
#undef LAUNCH_ST
#undef LAUNCH
  if (want_inter) return {out, inter};
  return {out};
}
