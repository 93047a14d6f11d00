// Fused flash-style attention for the Evoformer — gfx950 (CDNA4).
//
// Covers K1 of SURVEY.md §2.17: softmax(Q K^T * scale + pair_bias + mask) V
// for every attention in the trunk — MSA row attention (batch b*m, len n),
// MSA column attention (batch b*n, len m), both triangle self-attentions
// (batch b*n, len n, pair bias), and template pointwise attention.
//
// MI355X-first design decisions:
//  * MFMA bf16 16x16x32 tiles; fp32 accumulation.  Forward: one
//    workgroup = 8 waves = 128 query rows (FBQ/FNT); backward kernels
//    run 4-wave workgroups over 64-row tiles (BQ/NWAVES).  KV tiled by
//    64 with double-buffered LDS staging (fwd: single barrier/tile).
//  * The pair bias is NOT materialized per folded axis: the kernel takes
//    bias of shape (B / bias_repeat, h, Lq, Lk) and folds the repeat in
//    the index — the eager path would replicate it axial_dim times
//    (reference alphafold2.py:248), ~268 MB per triangle attention at
//    n=256.
//  * LDS tiles are XOR-swizzled (byte ^= (row&7)<<4) — a row-major
//    [64][64] bf16 tile read down a column is a 16-way bank conflict
//    otherwise (guide §6 G4).
//  * Online softmax entirely in registers; the P tile round-trips
//    through a per-wave swizzled LDS scratch to reach MFMA A-fragment
//    layout for P·V.
//  * Backward is FlashAttention-2 style: delta = rowsum(dO*O), a dQ
//    kernel and a dK/dV kernel, each recomputing P from (Q,K,bias,lse).
//    dBias is emitted with fp32 atomics folded over bias_repeat.
//
// MFMA fragment conventions (verified on HW by tools/mfma_probe.hip):
//  * C/D: col = lane&15, row = (lane>>4)*4 + reg   [guide §3, measured]
//  * A/B: the k axis is a pure reduction axis — any lane->k permutation
//    works if A and B agree; we use k = (lane>>4)*8 + j (contiguous 8,
//    16-byte ds_read per fragment).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16_t;

namespace {

constexpr int BQ = 64;     // query rows per workgroup
constexpr int BK = 64;     // kv rows per tile
constexpr int DH = 64;     // head dim (template-fixed)
constexpr int NWAVES = 4;  // waves per workgroup; each owns 16 q rows
constexpr int ROWB = DH * sizeof(bf16_t);  // 128 bytes per tile row
constexpr float NEG_INF = -1e30f;

// XOR swizzle: spread 16B chunks of a column access across banks
__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return row * ROWB + (byte_in_row ^ ((row & 7) << 4));
}

// strided (B, h, L, 64) tensor view: element strides, innermost dim
// contiguous.  Lets the kernels consume the Linear outputs' natural
// (B, L, h*64) layout through a permuted view — no contiguous() copies.
struct TView {
  const bf16_t* p;
  long bs, hs, rs;  // batch, head, row strides (elements)
  __device__ const bf16_t* base(int b, int h) const {
    return p + (long)b * bs + (long)h * hs;
  }
};
struct TViewMut {
  bf16_t* p;
  long bs, hs, rs;
  __device__ bf16_t* base(int b, int h) const {
    return p + (long)b * bs + (long)h * hs;
  }
};

// cooperative stage of a [rows<=64][64] bf16 tile global->LDS (swizzled).
// Each of 256 threads moves 2 16-byte chunks.  OOB rows zero-filled.
// row_stride in elements (bf16); rows are 64-element contiguous.
__device__ __forceinline__ void stage_tile(const bf16_t* __restrict__ g,
                                           long row_stride, int rows,
                                           char* lds) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int idx = tid + pass * 256;        // chunk index 0..511
    int row = idx >> 3;                // 8 chunks per row
    int c16 = (idx & 7) << 4;          // byte offset in row
    float4 val = {0, 0, 0, 0};
    if (row < rows) {
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(g + row * row_stride) + c16);
    }
    *reinterpret_cast<float4*>(lds + swz(row, c16)) = val;
  }
}

// stage a [rows<=64][<=64] bf16 tile with arbitrary row stride into a
// swizzled LDS tile; used for the pair-bias tile in the dK/dV kernel
// (its transposed per-lane reads would otherwise scatter 2B loads).
__device__ __forceinline__ void stage_tile_rowstride(
    const bf16_t* __restrict__ g, long row_stride, int rows, int cols,
    char* lds) {
  const int tid = threadIdx.x;
  if ((row_stride % 8) == 0 && (((uintptr_t)g) & 15) == 0) {
#pragma unroll
    for (int pass = 0; pass < 2; ++pass) {
      int idx = tid + pass * 256;
      int row = idx >> 3;
      int c16 = (idx & 7) << 4;
      float4 val = {0, 0, 0, 0};
      if (row < rows && c16 < cols * (int)sizeof(bf16_t)) {
        val = *reinterpret_cast<const float4*>(
            reinterpret_cast<const char*>(g + row * row_stride) + c16);
      }
      *reinterpret_cast<float4*>(lds + swz(row, c16)) = val;
    }
  } else {  // unaligned fallback: scalar u16 staging
    for (int idx = tid; idx < 64 * 64; idx += 256) {
      int row = idx >> 6;
      int col = idx & 63;
      bf16_t val = (bf16_t)0.f;
      if (row < rows && col < cols) val = g[row * row_stride + col];
      *reinterpret_cast<bf16_t*>(
          lds + swz(row, col * (int)sizeof(bf16_t))) = val;
    }
  }
}

// stage a [rows<=64][64] bf16 tile TRANSPOSED into LDS ([col][row],
// swizzled): the PV B-operand wants V^T rows so its fragments become
// single 16-byte ds_reads instead of 8 scalar strided reads.
__device__ __forceinline__ void stage_tile_t(const bf16_t* __restrict__ g,
                                             long row_stride, int rows,
                                             char* lds) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int idx = tid + pass * 256;
    int row = idx >> 3;                // source row (kv)
    int c8 = (idx & 7) << 3;           // first of 8 source cols (dv)
    float4 val = {0, 0, 0, 0};
    if (row < rows) {
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(g + row * row_stride) + c8 * 2);
    }
    const bf16_t* vv = reinterpret_cast<const bf16_t*>(&val);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<bf16_t*>(
          lds + swz(c8 + j, row * (int)sizeof(bf16_t))) = vv[j];
    }
  }
}

// async-stage split (guide T14): issue the tile's global loads into
// registers early (overlapping prior compute), write LDS after the
// barrier.  The same registers can be stored in BOTH layouts (normal +
// transposed), halving global traffic for double-layout tiles.
struct StageRegs {
  float4 v0, v1;
};

__device__ __forceinline__ void stage_load(const bf16_t* __restrict__ g,
                                           long row_stride, int rows,
                                           StageRegs& r) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int idx = tid + pass * 256;
    int row = idx >> 3;
    int c16 = (idx & 7) << 4;
    float4 val = {0, 0, 0, 0};
    if (row < rows) {
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(g + row * row_stride) + c16);
    }
    (pass ? r.v1 : r.v0) = val;
  }
}

__device__ __forceinline__ void stage_store(const StageRegs& r, char* lds) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int idx = tid + pass * 256;
    int row = idx >> 3;
    int c16 = (idx & 7) << 4;
    *reinterpret_cast<float4*>(lds + swz(row, c16)) = (pass ? r.v1 : r.v0);
  }
}

__device__ __forceinline__ void stage_store_t(const StageRegs& r, char* lds) {
  const int tid = threadIdx.x;
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    int idx = tid + pass * 256;
    int row = idx >> 3;
    int c8 = (idx & 7) << 3;  // first of 8 source cols
    const float4 val = pass ? r.v1 : r.v0;
    const bf16_t* vv = reinterpret_cast<const bf16_t*>(&val);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<bf16_t*>(
          lds + swz(c8 + j, row * (int)sizeof(bf16_t))) = vv[j];
    }
  }
}

// block-width/row-count generic staging (the fwd kernel runs 512
// threads over a 128-row Q tile and 64-row K/V tiles)
template <int NT, int R>
struct StageRegsN {
  float4 v[(R * 8 + NT - 1) / NT];
};

template <int NT, int R>
__device__ __forceinline__ void stage_load_n(const bf16_t* __restrict__ g,
                                             long row_stride, int rows,
                                             StageRegsN<NT, R>& r) {
  constexpr int P = (R * 8 + NT - 1) / NT;
#pragma unroll
  for (int pass = 0; pass < P; ++pass) {
    int idx = threadIdx.x + pass * NT;
    int row = idx >> 3;
    int c16 = (idx & 7) << 4;
    float4 val = {0, 0, 0, 0};
    if (row < rows) {
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(g + row * row_stride) + c16);
    }
    r.v[pass] = val;
  }
}

template <int NT, int R>
__device__ __forceinline__ void stage_store_n(const StageRegsN<NT, R>& r,
                                              char* lds) {
  constexpr int P = (R * 8 + NT - 1) / NT;
#pragma unroll
  for (int pass = 0; pass < P; ++pass) {
    int idx = threadIdx.x + pass * NT;
    int row = idx >> 3;
    int c16 = (idx & 7) << 4;
    *reinterpret_cast<float4*>(lds + swz(row, c16)) = r.v[pass];
  }
}

// column-wise-mapped variant for tiles that are ONLY stored transposed
// (fwd V^T): thread idx -> (row = idx&63, chunk = idx>>6), so within a
// wave the transposed-store byte offset `row*2` spans the full 0..126
// range -> ~2-way LDS bank conflict instead of the 16-way conflict of
// the row-wise mapping (where row*2 spans only 0..14 for fixed j).
// Global-load cost: each 128-byte row cacheline is read in 16B chunks
// by 8 different waves (L2-served) instead of one coalesced pull.
template <int NT>
__device__ __forceinline__ void stage_load_colwise_n(
    const bf16_t* __restrict__ g, long row_stride, int rows,
    StageRegsN<NT, 64>& r) {
  constexpr int P = (64 * 8 + NT - 1) / NT;
#pragma unroll
  for (int pass = 0; pass < P; ++pass) {
    int idx = threadIdx.x + pass * NT;
    int row = idx & 63;
    int c16 = (idx >> 6) << 4;
    float4 val = {0, 0, 0, 0};
    if (row < rows) {
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(g + row * row_stride) + c16);
    }
    r.v[pass] = val;
  }
}

template <int NT>
__device__ __forceinline__ void stage_store_t_colwise_n(
    const StageRegsN<NT, 64>& r, char* lds) {
  constexpr int P = (64 * 8 + NT - 1) / NT;
#pragma unroll
  for (int pass = 0; pass < P; ++pass) {
    int idx = threadIdx.x + pass * NT;
    int row = idx & 63;
    int c8 = (idx >> 6) << 3;
    const float4 val = r.v[pass];
    const bf16_t* vv = reinterpret_cast<const bf16_t*>(&val);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<bf16_t*>(
          lds + swz(c8 + j, row * (int)sizeof(bf16_t))) = vv[j];
    }
  }
}

template <int NT, int R>
__device__ __forceinline__ void stage_store_t_n(const StageRegsN<NT, R>& r,
                                                char* lds) {
  constexpr int P = (R * 8 + NT - 1) / NT;
#pragma unroll
  for (int pass = 0; pass < P; ++pass) {
    int idx = threadIdx.x + pass * NT;
    int row = idx >> 3;
    int c8 = (idx & 7) << 3;
    const float4 val = r.v[pass];
    const bf16_t* vv = reinterpret_cast<const bf16_t*>(&val);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *reinterpret_cast<bf16_t*>(
          lds + swz(c8 + j, row * (int)sizeof(bf16_t))) = vv[j];
    }
  }
}

// read an 8-bf16 A/B fragment (k = (lane>>4)*8 + j) for tile row `row`,
// k-block `kblk` (32 wide) from a swizzled LDS tile
__device__ __forceinline__ bf16x8 frag_row(const char* lds, int row,
                                           int kblk) {
  const int lane = threadIdx.x & 63;
  int byte_in_row = kblk * 64 + ((lane >> 4) << 4);
  return *reinterpret_cast<const bf16x8*>(lds + swz(row, byte_in_row));
}

// B-fragment where the matrix is stored [k][col] (k = tile row):
// lane needs col = lane&15, k = (lane>>4)*8 + j  -> 8 strided u16 reads
__device__ __forceinline__ bf16x8 frag_col(const char* lds, int col,
                                           int kblk) {
  const int lane = threadIdx.x & 63;
  bf16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int k = kblk * 32 + ((lane >> 4) << 3) + j;
    out[j] = *reinterpret_cast<const bf16_t*>(
        lds + swz(k, col * (int)sizeof(bf16_t)));
  }
  return out;
}

// ---------------------------------------------------------------------------
// forward

// S-tile layout per wave: rows 16 (q), cols 64 (kv) as 4 col-blocks of
// f32x4.  Row r of the wave lives in lanes [16*(r/4), 16*(r/4)+16) at
// reg r%4; a full row reduce is 4 col-blocks + shfl_xor over 16 lanes.
__device__ __forceinline__ float rowred_max(const f32x4 s[4], int reg) {
  float v = fmaxf(fmaxf(s[0][reg], s[1][reg]), fmaxf(s[2][reg], s[3][reg]));
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 16));
  return v;
}

__device__ __forceinline__ float rowred_sum(const f32x4 s[4], int reg) {
  float v = s[0][reg] + s[1][reg] + s[2][reg] + s[3][reg];
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, 16);
  return v;
}

constexpr int FBQ = 128;    // fwd query rows per workgroup
constexpr int FNT = 512;    // fwd threads (8 waves x 16 q rows)

template <bool HAS_BIAS, bool HAS_MASK>
// minWavesPerEU=4 caps VGPRs at 128: the <bias,mask> variant was
// 129 VGPRs = 3 waves/SIMD = only ONE 8-wave block resident per CU
__global__ __launch_bounds__(FNT, 4)
void attn_fwd_kernel(TView q, TView k, TView v,
                     const bf16_t* __restrict__ bias,
                     const unsigned char* __restrict__ mask,
                     TViewMut out, float* __restrict__ lse,
                     int Lq, int Lk, int heads, int bias_repeat, int q_repeat,
                     float scale) {
  __shared__ char q_lds[FBQ * ROWB];
  // K / V^T / mask tiles are DOUBLE-buffered so the loop needs a single
  // __syncthreads per KV tile (placed between store and compute): the
  // barrier bounds the wave spread to one iteration, so store(t+1)
  // writes buf[(t+1)&1] while the slowest wave still reads buf[t&1] —
  // never the same buffer.  (~64KB LDS total: 2 blocks/CU preserved.)
  __shared__ char k_lds[2][BK * ROWB];
  __shared__ char vt_lds[2][BK * ROWB];  // V transposed: [dv][kv]
  __shared__ char p_lds[FNT / 64][16 * ROWB];
  __shared__ unsigned char m_lds[2][BK];

  const int qtile = blockIdx.x;
  const int bh = blockIdx.y;            // batch*heads + head
  const int batch = bh / heads;
  const int head = bh - batch * heads;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  const bf16_t* q_g = q.base(batch / q_repeat, head) + (long)qtile * FBQ * q.rs;
  const bf16_t* k_g = k.base(batch, head);
  const bf16_t* v_g = v.base(batch, head);
  const bf16_t* bias_g = nullptr;
  if (HAS_BIAS) {
    const int bias_batch = batch / bias_repeat;
    bias_g = bias + ((long)(bias_batch * heads + head) * Lq
                     + (long)qtile * FBQ) * Lk;
  }

  const int q_rows = min(FBQ, Lq - qtile * FBQ);
  {
    StageRegsN<FNT, FBQ> qr;
    stage_load_n<FNT, FBQ>(q_g, q.rs, q_rows, qr);
    stage_store_n<FNT, FBQ>(qr, q_lds);
  }
  __syncthreads();

  // per-wave Q fragments (rows wave*16 + (lane&15))
  bf16x8 q_frag[2];
  const int qrow = wave * 16 + (lane & 15);
#pragma unroll
  for (int dblk = 0; dblk < 2; ++dblk)
    q_frag[dblk] = frag_row(q_lds, qrow, dblk);

  float m_i[4], l_i[4];
  f32x4 o_acc[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_i[r] = NEG_INF;
    l_i[r] = 0.f;
    o_acc[r] = f32x4{0, 0, 0, 0};
  }

  const int n_kv = (Lk + BK - 1) / BK;
  StageRegsN<FNT, BK> kreg, vreg;
  stage_load_n<FNT, BK>(k_g, k.rs, min(BK, Lk), kreg);
  stage_load_colwise_n<FNT>(v_g, v.rs, min(BK, Lk), vreg);
  for (int t = 0; t < n_kv; ++t) {
    const int kv_rows = min(BK, Lk - t * BK);
    const int buf = t & 1;
    stage_store_n<FNT, BK>(kreg, k_lds[buf]);
    stage_store_t_colwise_n<FNT>(vreg, vt_lds[buf]);
    if (HAS_MASK && threadIdx.x < BK) {
      m_lds[buf][threadIdx.x] = (threadIdx.x < kv_rows)
          ? mask[(long)batch * Lk + t * BK + threadIdx.x] : 0;
    }
    if (t + 1 < n_kv) {
      // next tile's loads fly across the barrier and under compute
      // (the ds_writes above read kreg/vreg at issue, in order)
      const int next_rows = min(BK, Lk - (t + 1) * BK);
      stage_load_n<FNT, BK>(k_g + (long)(t + 1) * BK * k.rs, k.rs,
                            next_rows, kreg);
      stage_load_colwise_n<FNT>(v_g + (long)(t + 1) * BK * v.rs, v.rs,
                                next_rows, vreg);
    }
    __syncthreads();

    // S = Q K^T  (16 q x 64 kv per wave); setprio favors the MFMA
    // cluster when co-resident waves are staging (guide T5)
    f32x4 s[4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 kf = frag_row(k_lds[buf], c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[dblk], kf, acc,
                                                      0, 0, 0);
      }
      s[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // scale + bias + masking (C layout: row=(lane>>4)*4+reg, col=lane&15)
    const bf16_t* brow[4];
    if (HAS_BIAS) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = wave * 16 + (lane >> 4) * 4 + reg;
        brow[reg] = (row < q_rows)
            ? bias_g + (long)row * Lk + t * BK : nullptr;
      }
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int col = c * 16 + (lane & 15);
      const bool col_ok = col < kv_rows &&
          (!HAS_MASK || m_lds[buf][col]);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float val = s[c][reg] * scale;
        if (HAS_BIAS && col_ok && brow[reg] != nullptr)
          val += to_f32(brow[reg][col]);
        s[c][reg] = col_ok ? val : NEG_INF;
      }
    }

    // online softmax
    float alpha[4], m_new[4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      float tile_max = rowred_max(s, reg);
      m_new[reg] = fmaxf(m_i[reg], tile_max);
      alpha[reg] = (m_i[reg] <= NEG_INF) ? 0.f : __expf(m_i[reg] - m_new[reg]);
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        s[c][reg] = (m_new[reg] <= NEG_INF) ? 0.f
            : __expf(s[c][reg] - m_new[reg]);
      }
      l_i[reg] = l_i[reg] * alpha[reg] + rowred_sum(s, reg);
      m_i[reg] = m_new[reg];
#pragma unroll
      for (int c = 0; c < 4; ++c) o_acc[c][reg] *= alpha[reg];
    }

    // P tile -> per-wave swizzled LDS scratch (C layout -> A layout)
    char* pw = p_lds[wave];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int col = c * 16 + (lane & 15);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = (lane >> 4) * 4 + reg;
        *reinterpret_cast<bf16_t*>(pw + swz(row, col * (int)sizeof(bf16_t)))
            = (bf16_t)s[c][reg];
      }
    }
    // wave-internal LDS dependency: ds ops from this wave only
    // LDS-only wait: vmcnt stays open so the prefetched next
    // K/V tile's global loads keep flying under the PV MFMAs
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    bf16x8 p_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk)
      p_frag[kblk] = frag_row(pw, lane & 15, kblk);

    // O += P V  (B-operand = V^T rows: one b128 read per fragment)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = o_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 vf = frag_row(vt_lds[buf], c * 16 + (lane & 15), kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(p_frag[kblk], vf, acc,
                                                      0, 0, 0);
      }
      o_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // epilogue: O /= l, store out + lse
  bf16_t* out_g = out.base(batch, head) + (long)qtile * FBQ * out.rs;
  float* lse_g = lse + (long)bh * Lq + (long)qtile * FBQ;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    const float linv = l_i[reg] > 0.f ? 1.f / l_i[reg] : 0.f;
    if (row < q_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        out_g[(long)row * out.rs + c * 16 + (lane & 15)] =
            (bf16_t)(o_acc[c][reg] * linv);
      }
      if ((lane & 15) == 0) {
        lse_g[row] = (l_i[reg] > 0.f) ? m_i[reg] + logf(l_i[reg]) : NEG_INF;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward: delta = rowsum(dO * O)

__global__ void attn_delta_kernel(TView dout, TView out,
                                  float* __restrict__ delta,
                                  int heads, int Lq, long rows) {
  // 8 rows per 64-lane wave: lane l covers row (l>>3), 8 channels
  // starting at (l&7)*8 — 16-byte loads, 1 KiB per wave transaction
  const long row = ((long)blockIdx.x * blockDim.x + threadIdx.x) >> 3;
  if (row >= rows) return;
  const int sub = threadIdx.x & 7;
  const int l = row % Lq;
  const long bh = row / Lq;
  const int b = bh / heads, h = bh - (bh / heads) * heads;
  const bf16_t* dp = dout.base(b, h) + (long)l * dout.rs + sub * 8;
  const bf16_t* op = out.base(b, h) + (long)l * out.rs + sub * 8;
  bf16x8 dv8 = *reinterpret_cast<const bf16x8*>(dp);
  bf16x8 ov8 = *reinterpret_cast<const bf16x8*>(op);
  float v = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) v += to_f32(dv8[j]) * to_f32(ov8[j]);
#pragma unroll
  for (int off = 4; off > 0; off >>= 1) v += __shfl_xor(v, off, 8);
  if (sub == 0) delta[row] = v;
}

// ---------------------------------------------------------------------------
// backward dQ: loop kv tiles; dS = P*(dP - delta)*scale; dQ += dS K

template <bool HAS_BIAS, bool HAS_MASK>
// minWavesPerEU=3 caps VGPRs at 170 (the <bias,mask> variant was 178
// -> 2 waves/SIMD); 3 waves = 3 resident 4-wave blocks per CU
__global__ __launch_bounds__(256, 3)
void attn_bwd_dq_kernel(TView q, TView k, TView v,
                        const bf16_t* __restrict__ bias,
                        const unsigned char* __restrict__ mask,
                        TView dout,
                        const float* __restrict__ lse,
                        const float* __restrict__ delta,
                        TViewMut dq,
                        int Lq, int Lk, int heads, int bias_repeat, int q_repeat,
                        float scale) {
  __shared__ char q_lds[BQ * ROWB];
  __shared__ char do_lds[BQ * ROWB];
  __shared__ char k_lds[BK * ROWB];
  __shared__ char kt_lds[BK * ROWB];  // K transposed: [d][kv]
  __shared__ char v_lds[BK * ROWB];
  __shared__ char s_lds[NWAVES][16 * ROWB];
  __shared__ unsigned char m_lds[BK];

  const int qtile = blockIdx.x;
  const int bh = blockIdx.y;
  const int batch = bh / heads;
  const int head = bh - batch * heads;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  const bf16_t* q_g = q.base(batch / q_repeat, head) + (long)qtile * BQ * q.rs;
  const bf16_t* do_g = dout.base(batch, head) + (long)qtile * BQ * dout.rs;
  const bf16_t* k_g = k.base(batch, head);
  const bf16_t* v_g = v.base(batch, head);
  const float* lse_g = lse + (long)bh * Lq + (long)qtile * BQ;
  const float* delta_g = delta + (long)bh * Lq + (long)qtile * BQ;
  const bf16_t* bias_g = nullptr;
  if (HAS_BIAS) {
    const int bias_batch = batch / bias_repeat;
    bias_g = bias + ((long)(bias_batch * heads + head) * Lq
                     + (long)qtile * BQ) * Lk;
  }

  const int q_rows = min(BQ, Lq - qtile * BQ);
  stage_tile(q_g, q.rs, q_rows, q_lds);
  stage_tile(do_g, dout.rs, q_rows, do_lds);
  __syncthreads();

  const int qrow = wave * 16 + (lane & 15);
  bf16x8 q_frag[2], do_frag[2];
#pragma unroll
  for (int dblk = 0; dblk < 2; ++dblk) {
    q_frag[dblk] = frag_row(q_lds, qrow, dblk);
    do_frag[dblk] = frag_row(do_lds, qrow, dblk);
  }

  // per-row lse/delta (C layout rows)
  float lse_r[4], delta_r[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    lse_r[reg] = (row < q_rows) ? lse_g[row] : NEG_INF;
    delta_r[reg] = (row < q_rows) ? delta_g[row] : 0.f;
  }

  f32x4 dq_acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) dq_acc[c] = f32x4{0, 0, 0, 0};

  const int n_kv = (Lk + BK - 1) / BK;
  StageRegs kreg, vreg;
  stage_load(k_g, k.rs, min(BK, Lk), kreg);
  stage_load(v_g, v.rs, min(BK, Lk), vreg);
  for (int t = 0; t < n_kv; ++t) {
    const int kv_rows = min(BK, Lk - t * BK);
    __syncthreads();
    stage_store(kreg, k_lds);
    stage_store_t(kreg, kt_lds);  // one load feeds both layouts
    stage_store(vreg, v_lds);
    if (HAS_MASK && threadIdx.x < BK) {
      m_lds[threadIdx.x] = (threadIdx.x < kv_rows)
          ? mask[(long)batch * Lk + t * BK + threadIdx.x] : 0;
    }
    __syncthreads();
    if (t + 1 < n_kv) {
      const int next_rows = min(BK, Lk - (t + 1) * BK);
      stage_load(k_g + (long)(t + 1) * BK * k.rs, k.rs, next_rows, kreg);
      stage_load(v_g + (long)(t + 1) * BK * v.rs, v.rs, next_rows, vreg);
    }

    // recompute S then P = exp(S*scale + bias - lse)
    f32x4 p[4], dp[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 kf = frag_row(k_lds, c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[dblk], kf, acc,
                                                      0, 0, 0);
      }
      p[c] = acc;
    }
    const bf16_t* brow[4];
    if (HAS_BIAS) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = wave * 16 + (lane >> 4) * 4 + reg;
        brow[reg] = (row < q_rows)
            ? bias_g + (long)row * Lk + t * BK : nullptr;
      }
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int col = c * 16 + (lane & 15);
      const bool col_ok = col < kv_rows && (!HAS_MASK || m_lds[col]);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float val = p[c][reg] * scale;
        if (HAS_BIAS && col_ok && brow[reg] != nullptr)
          val += to_f32(brow[reg][col]);
        p[c][reg] = (col_ok && lse_r[reg] > NEG_INF)
            ? __expf(val - lse_r[reg]) : 0.f;
      }
    }

    // dP = dO V^T : A = dO (k=dv), B col=kv row of V (k=dv contiguous)
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 vf = frag_row(v_lds, c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(do_frag[dblk], vf, acc,
                                                      0, 0, 0);
      }
      dp[c] = acc;
    }

    // dS = P * (dP - delta) * scale  -> bf16 via LDS scratch
    char* sw = s_lds[wave];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int col = c * 16 + (lane & 15);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = (lane >> 4) * 4 + reg;
        float ds = p[c][reg] * (dp[c][reg] - delta_r[reg]) * scale;
        *reinterpret_cast<bf16_t*>(sw + swz(row, col * (int)sizeof(bf16_t)))
            = (bf16_t)ds;
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    bf16x8 ds_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk)
      ds_frag[kblk] = frag_row(sw, lane & 15, kblk);

    // dQ += dS K : B col = d, k = kv -> b128 rows of the K^T tile
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = dq_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 kf = frag_row(kt_lds, c * 16 + (lane & 15), kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ds_frag[kblk], kf, acc,
                                                      0, 0, 0);
      }
      dq_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);
  }

  bf16_t* dq_g = dq.base(batch, head) + (long)qtile * BQ * dq.rs;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    if (row < q_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        dq_g[(long)row * dq.rs + c * 16 + (lane & 15)] =
            (bf16_t)dq_acc[c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// backward dK/dV (+ dBias): one block per kv tile, loop q tiles.
// Computes transposed products so kv is the row axis:
//   S^T = K Q^T,  P^T,  dV += P^T dO,  dP^T = V dO^T,
//   dS^T = P^T (dP^T - delta) scale,  dK += dS^T Q.

template <bool HAS_BIAS, bool HAS_MASK, bool NEED_DBIAS>
__global__ __launch_bounds__(256, 2)
void attn_bwd_dkv_kernel(TView q, TView k, TView v,
                         const bf16_t* __restrict__ bias,
                         const unsigned char* __restrict__ mask,
                         TView dout,
                         const float* __restrict__ lse,
                         const float* __restrict__ delta,
                         TViewMut dk, TViewMut dv,
                         float* __restrict__ dbias,
                         int dbias_chunks, long dbias_stride,
                         int Lq, int Lk, int heads, int bias_repeat, int q_repeat,
                         float scale) {
  __shared__ char k_lds[BK * ROWB];
  __shared__ char v_lds[BK * ROWB];
  __shared__ char q_lds[BQ * ROWB];
  __shared__ char qt_lds[BQ * ROWB];   // Q transposed: [d][q]
  __shared__ char do_lds[BQ * ROWB];
  __shared__ char dot_lds[BQ * ROWB];  // dO transposed: [dv][q]
  __shared__ char s_lds[NWAVES][16 * ROWB];
  __shared__ char b_lds[BQ * ROWB];  // bias tile [q][kv], swizzled
  __shared__ float lse_lds[BQ];
  __shared__ float delta_lds[BQ];
  __shared__ unsigned char m_lds[BK];

  const int ktile = blockIdx.x;
  const int bh = blockIdx.y;
  const int batch = bh / heads;
  const int head = bh - batch * heads;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  const bf16_t* k_g = k.base(batch, head) + (long)ktile * BK * k.rs;
  const bf16_t* v_g = v.base(batch, head) + (long)ktile * BK * v.rs;
  const bf16_t* q_g = q.base(batch / q_repeat, head);
  const bf16_t* do_g = dout.base(batch, head);
  const float* lse_g = lse + (long)bh * Lq;
  const float* delta_g = delta + (long)bh * Lq;
  const bf16_t* bias_g = nullptr;
  float* dbias_g = nullptr;
  const int bias_batch = batch / bias_repeat;
  if (HAS_BIAS)
    bias_g = bias + (long)(bias_batch * heads + head) * Lq * Lk;
  if (NEED_DBIAS) {
    // fold into one of `dbias_chunks` scratch copies: the repeat group
    // splits across chunks, cutting same-address atomic chain depth
    const int chunk = (batch % bias_repeat) % dbias_chunks;
    dbias_g = dbias + (long)chunk * dbias_stride
        + (long)(bias_batch * heads + head) * Lq * Lk;
  }

  const int kv_rows = min(BK, Lk - ktile * BK);
  stage_tile(k_g, k.rs, kv_rows, k_lds);
  stage_tile(v_g, v.rs, kv_rows, v_lds);
  if (HAS_MASK && threadIdx.x < BK) {
    m_lds[threadIdx.x] = (threadIdx.x < kv_rows)
        ? mask[(long)batch * Lk + ktile * BK + threadIdx.x] : 0;
  }
  __syncthreads();

  const int krow = wave * 16 + (lane & 15);
  bf16x8 k_frag[2], v_frag[2];
#pragma unroll
  for (int dblk = 0; dblk < 2; ++dblk) {
    k_frag[dblk] = frag_row(k_lds, krow, dblk);
    v_frag[dblk] = frag_row(v_lds, krow, dblk);
  }
  const bool krow_ok = krow < kv_rows && (!HAS_MASK || m_lds[krow]);
  // C-layout kv rows for this wave
  bool krow_ok_c[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    krow_ok_c[reg] = row < kv_rows && (!HAS_MASK || m_lds[row]);
  }
  (void)krow_ok;

  f32x4 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    dk_acc[c] = f32x4{0, 0, 0, 0};
    dv_acc[c] = f32x4{0, 0, 0, 0};
  }

  const int n_q = (Lq + BQ - 1) / BQ;
  StageRegs qreg, doreg;
  stage_load(q_g, q.rs, min(BQ, Lq), qreg);
  stage_load(do_g, dout.rs, min(BQ, Lq), doreg);
  for (int t = 0; t < n_q; ++t) {
    const int q_rows = min(BQ, Lq - t * BQ);
    __syncthreads();
    stage_store(qreg, q_lds);
    stage_store_t(qreg, qt_lds);    // one load feeds both layouts
    stage_store(doreg, do_lds);
    stage_store_t(doreg, dot_lds);
    if (HAS_BIAS) {
      stage_tile_rowstride(bias_g + (long)t * BQ * Lk + (long)ktile * BK,
                           Lk, q_rows, min(BK, Lk - ktile * BK), b_lds);
    }
    if (threadIdx.x < BQ) {
      const int qq = t * BQ + threadIdx.x;
      lse_lds[threadIdx.x] = (threadIdx.x < q_rows) ? lse_g[qq] : NEG_INF;
      delta_lds[threadIdx.x] = (threadIdx.x < q_rows) ? delta_g[qq] : 0.f;
    }
    __syncthreads();
    if (t + 1 < n_q) {
      const int next_rows = min(BQ, Lq - (t + 1) * BQ);
      stage_load(q_g + (long)(t + 1) * BQ * q.rs, q.rs, next_rows, qreg);
      stage_load(do_g + (long)(t + 1) * BQ * dout.rs, dout.rs, next_rows,
                 doreg);
    }

    // S^T = K Q^T : A = K (k = d), B col = q row of Q (k = d contiguous)
    f32x4 pt[4], dpt[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 qf = frag_row(q_lds, c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(k_frag[dblk], qf, acc,
                                                      0, 0, 0);
      }
      pt[c] = acc;
    }
    // P^T = exp(S^T*scale + bias^T - lse[q])
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int qcol = c * 16 + (lane & 15);       // q index in tile
      const float l = lse_lds[qcol];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kvrow = wave * 16 + (lane >> 4) * 4 + reg;
        float val = pt[c][reg] * scale;
        if (HAS_BIAS && krow_ok_c[reg] && qcol < q_rows)
          val += to_f32(*reinterpret_cast<const bf16_t*>(
              b_lds + swz(qcol, kvrow * (int)sizeof(bf16_t))));
        pt[c][reg] = (krow_ok_c[reg] && qcol < q_rows && l > NEG_INF)
            ? __expf(val - l) : 0.f;
      }
    }

    // dV += P^T dO : A = P^T (k = q, via LDS), B col = dv, k = q
    char* sw = s_lds[wave];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int col = c * 16 + (lane & 15);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = (lane >> 4) * 4 + reg;
        *reinterpret_cast<bf16_t*>(sw + swz(row, col * (int)sizeof(bf16_t)))
            = (bf16_t)pt[c][reg];
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    bf16x8 pt_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk)
      pt_frag[kblk] = frag_row(sw, lane & 15, kblk);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = dv_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 dof = frag_row(dot_lds, c * 16 + (lane & 15), kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pt_frag[kblk], dof, acc,
                                                      0, 0, 0);
      }
      dv_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // dP^T = V dO^T : A = V (k = dv), B col = q, k = dv (contiguous)
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 dof = frag_row(do_lds, c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(v_frag[dblk], dof, acc,
                                                      0, 0, 0);
      }
      dpt[c] = acc;
    }

    // dS^T = P^T (dP^T - delta[q]) scale ; emit dBias; stage for dK
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int qcol = c * 16 + (lane & 15);
      const float dl = delta_lds[qcol];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        // dL/d(S*scale + bias): stored UNscaled (it IS the bias
        // gradient); the 1/sqrt(d) scale is applied to dK at the
        // epilogue instead
        float ds_total = pt[c][reg] * (dpt[c][reg] - dl);
        const int row = (lane >> 4) * 4 + reg;
        *reinterpret_cast<bf16_t*>(sw + swz(row, qcol * (int)sizeof(bf16_t)))
            = (bf16_t)ds_total;
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    bf16x8 dst_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk)
      dst_frag[kblk] = frag_row(sw, lane & 15, kblk);

    if (NEED_DBIAS) {
      // cooperative transposed drain of the 64x64 dS tile from the four
      // per-wave scratches into dbias.  Lanes map to CONTIGUOUS kv
      // addresses so each store/atomic instruction coalesces; when
      // bias_repeat == 1 this block is the sole writer of its (q, kv)
      // range and plain stores replace atomics.
      __syncthreads();
      const int kvr = lane;            // 0..63, contiguous per wave
      for (int j = 0; j < 16; ++j) {
        const int qq = j * NWAVES + wave;
        if (qq < q_rows && kvr < kv_rows) {
          float dsv = to_f32(*reinterpret_cast<const bf16_t*>(
              s_lds[kvr >> 4] + swz(kvr & 15, qq * (int)sizeof(bf16_t))));
          float* addr = dbias_g + (long)(t * BQ + qq) * Lk
              + (long)ktile * BK + kvr;
          if (bias_repeat == 1) *addr = dsv; else atomicAdd(addr, dsv);
        }
      }
    }

    // dK += dS^T Q : B col = d, k = q -> b128 rows of the Q^T tile
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = dk_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 qf = frag_row(qt_lds, c * 16 + (lane & 15), kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dst_frag[kblk], qf, acc,
                                                      0, 0, 0);
      }
      dk_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);
  }

  bf16_t* dk_g = dk.base(batch, head) + (long)ktile * BK * dk.rs;
  bf16_t* dv_g = dv.base(batch, head) + (long)ktile * BK * dv.rs;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    if (row < kv_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        dk_g[(long)row * dk.rs + c * 16 + (lane & 15)] =
            (bf16_t)(dk_acc[c][reg] * scale);
        dv_g[(long)row * dv.rs + c * 16 + (lane & 15)] =
            (bf16_t)dv_acc[c][reg];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// split backward (r02): the combined dK/dV kernel sits at 196-249 VGPRs
// = 2 waves/SIMD.  Splitting recomputes S^T/P^T once more (5 -> 7
// GEMM-equivalents) but the dV pass drops to ~4 waves/SIMD and the dK
// pass to 3 — at single-digit MfmaUtil the residency win dominates.

template <bool HAS_BIAS, bool HAS_MASK>
__global__ __launch_bounds__(256, 4)
void attn_bwd_dv_kernel(TView q, TView k,
                        const bf16_t* __restrict__ bias,
                        const unsigned char* __restrict__ mask,
                        TView dout,
                        const float* __restrict__ lse,
                        TViewMut dv,
                        int Lq, int Lk, int heads, int bias_repeat, int q_repeat,
                        float scale) {
  __shared__ char k_lds[BK * ROWB];
  __shared__ char q_lds[BQ * ROWB];
  __shared__ char dot_lds[BQ * ROWB];  // dO transposed: [dv][q]
  // bias tile [q][kv] and the P^T scratch ALIAS: the bias is fully
  // consumed (P^T formed in regs) before the scratch write, and the
  // barrier below separates the cross-wave read/write windows.  Saves
  // 8 KB -> 4 workgroups/CU.
  __shared__ char sb_lds[BQ * ROWB];
  char* b_lds = sb_lds;
  char (*s_lds)[16 * ROWB] =
      reinterpret_cast<char (*)[16 * ROWB]>(sb_lds);
  __shared__ float lse_lds[BQ];
  __shared__ unsigned char m_lds[BK];

  const int ktile = blockIdx.x;
  const int bh = blockIdx.y;
  const int batch = bh / heads;
  const int head = bh - batch * heads;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  const bf16_t* k_g = k.base(batch, head) + (long)ktile * BK * k.rs;
  const bf16_t* q_g = q.base(batch / q_repeat, head);
  const bf16_t* do_g = dout.base(batch, head);
  const float* lse_g = lse + (long)bh * Lq;
  const bf16_t* bias_g = HAS_BIAS
      ? bias + (long)((batch / bias_repeat) * heads + head) * Lq * Lk
      : nullptr;

  const int kv_rows = min(BK, Lk - ktile * BK);
  stage_tile(k_g, k.rs, kv_rows, k_lds);
  if (HAS_MASK && threadIdx.x < BK) {
    m_lds[threadIdx.x] = (threadIdx.x < kv_rows)
        ? mask[(long)batch * Lk + ktile * BK + threadIdx.x] : 0;
  }
  __syncthreads();

  bf16x8 k_frag[2];
#pragma unroll
  for (int dblk = 0; dblk < 2; ++dblk)
    k_frag[dblk] = frag_row(k_lds, wave * 16 + (lane & 15), dblk);
  bool krow_ok_c[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    krow_ok_c[reg] = row < kv_rows && (!HAS_MASK || m_lds[row]);
  }

  f32x4 dv_acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) dv_acc[c] = f32x4{0, 0, 0, 0};

  const int n_q = (Lq + BQ - 1) / BQ;
  // direct (prefetch-free) staging: this kernel trades the async-stage
  // register cost for residency — occupancy is its lever
  for (int t = 0; t < n_q; ++t) {
    const int q_rows = min(BQ, Lq - t * BQ);
    __syncthreads();
    stage_tile(q_g + (long)t * BQ * q.rs, q.rs, q_rows, q_lds);
    stage_tile_t(do_g + (long)t * BQ * dout.rs, dout.rs, q_rows, dot_lds);
    if (HAS_BIAS) {
      stage_tile_rowstride(bias_g + (long)t * BQ * Lk + (long)ktile * BK,
                           Lk, q_rows, min(BK, Lk - ktile * BK), b_lds);
    }
    if (threadIdx.x < BQ) {
      const int qq = t * BQ + threadIdx.x;
      lse_lds[threadIdx.x] = (threadIdx.x < q_rows) ? lse_g[qq] : NEG_INF;
    }
    __syncthreads();

    // S^T = K Q^T ; P^T = exp(S^T*scale + bias^T - lse[q])
    f32x4 pt[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 qf = frag_row(q_lds, c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(k_frag[dblk], qf, acc,
                                                      0, 0, 0);
      }
      pt[c] = acc;
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int qcol = c * 16 + (lane & 15);
      const float l = lse_lds[qcol];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kvrow = wave * 16 + (lane >> 4) * 4 + reg;
        float val = pt[c][reg] * scale;
        if (HAS_BIAS && krow_ok_c[reg] && qcol < q_rows)
          val += to_f32(*reinterpret_cast<const bf16_t*>(
              b_lds + swz(qcol, kvrow * (int)sizeof(bf16_t))));
        pt[c][reg] = (krow_ok_c[reg] && qcol < q_rows && l > NEG_INF)
            ? __expf(val - l) : 0.f;
      }
    }

    // dV += P^T dO : A = P^T (k = q, via per-wave LDS scratch).
    // The scratch aliases the bias tile: wait until EVERY wave has
    // finished its bias reads before overwriting.
    if (HAS_BIAS) __syncthreads();
    char* sw = s_lds[wave];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int col = c * 16 + (lane & 15);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int row = (lane >> 4) * 4 + reg;
        *reinterpret_cast<bf16_t*>(sw + swz(row, col * (int)sizeof(bf16_t)))
            = (bf16_t)pt[c][reg];
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    bf16x8 pt_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk)
      pt_frag[kblk] = frag_row(sw, lane & 15, kblk);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = dv_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 dof = frag_row(dot_lds, c * 16 + (lane & 15), kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pt_frag[kblk], dof, acc,
                                                      0, 0, 0);
      }
      dv_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);
  }

  bf16_t* dv_g = dv.base(batch, head) + (long)ktile * BK * dv.rs;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    if (row < kv_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c)
        dv_g[(long)row * dv.rs + c * 16 + (lane & 15)] =
            (bf16_t)dv_acc[c][reg];
    }
  }
}

template <bool HAS_BIAS, bool HAS_MASK, bool NEED_DBIAS>
__global__ __launch_bounds__(256, 3)
void attn_bwd_dk_kernel(TView q, TView k, TView v,
                        const bf16_t* __restrict__ bias,
                        const unsigned char* __restrict__ mask,
                        TView dout,
                        const float* __restrict__ lse,
                        const float* __restrict__ delta,
                        TViewMut dk,
                        float* __restrict__ dbias,
                        int dbias_chunks, long dbias_stride,
                        int Lq, int Lk, int heads, int bias_repeat, int q_repeat,
                        float scale) {
  __shared__ char k_lds[BK * ROWB];
  __shared__ char v_lds[BK * ROWB];
  __shared__ char q_lds[BQ * ROWB];
  __shared__ char qt_lds[BQ * ROWB];   // Q transposed: [d][q]
  __shared__ char do_lds[BQ * ROWB];
  // bias tile and dS scratch alias (see dv kernel) — 48 KB total LDS
  __shared__ char sb_lds[BQ * ROWB];
  char* b_lds = sb_lds;
  char (*s_lds)[16 * ROWB] =
      reinterpret_cast<char (*)[16 * ROWB]>(sb_lds);
  __shared__ float lse_lds[BQ];
  __shared__ float delta_lds[BQ];
  __shared__ unsigned char m_lds[BK];

  const int ktile = blockIdx.x;
  const int bh = blockIdx.y;
  const int batch = bh / heads;
  const int head = bh - batch * heads;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;

  const bf16_t* k_g = k.base(batch, head) + (long)ktile * BK * k.rs;
  const bf16_t* v_g = v.base(batch, head) + (long)ktile * BK * v.rs;
  const bf16_t* q_g = q.base(batch / q_repeat, head);
  const bf16_t* do_g = dout.base(batch, head);
  const float* lse_g = lse + (long)bh * Lq;
  const float* delta_g = delta + (long)bh * Lq;
  const int bias_batch = batch / bias_repeat;
  const bf16_t* bias_g = HAS_BIAS
      ? bias + (long)(bias_batch * heads + head) * Lq * Lk : nullptr;
  float* dbias_g = nullptr;
  if (NEED_DBIAS) {
    const int chunk = (batch % bias_repeat) % dbias_chunks;
    dbias_g = dbias + (long)chunk * dbias_stride
        + (long)(bias_batch * heads + head) * Lq * Lk;
  }

  const int kv_rows = min(BK, Lk - ktile * BK);
  stage_tile(k_g, k.rs, kv_rows, k_lds);
  stage_tile(v_g, v.rs, kv_rows, v_lds);
  if (HAS_MASK && threadIdx.x < BK) {
    m_lds[threadIdx.x] = (threadIdx.x < kv_rows)
        ? mask[(long)batch * Lk + ktile * BK + threadIdx.x] : 0;
  }
  __syncthreads();

  const int krow = wave * 16 + (lane & 15);
  bf16x8 k_frag[2], v_frag[2];
#pragma unroll
  for (int dblk = 0; dblk < 2; ++dblk) {
    k_frag[dblk] = frag_row(k_lds, krow, dblk);
    v_frag[dblk] = frag_row(v_lds, krow, dblk);
  }
  bool krow_ok_c[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    krow_ok_c[reg] = row < kv_rows && (!HAS_MASK || m_lds[row]);
  }

  f32x4 dk_acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) dk_acc[c] = f32x4{0, 0, 0, 0};

  const int n_q = (Lq + BQ - 1) / BQ;
  // prefetch-free staging (register diet for residency); the dual
  // q layouts still come from one StageRegs load
  StageRegs qreg;
  for (int t = 0; t < n_q; ++t) {
    const int q_rows = min(BQ, Lq - t * BQ);
    stage_load(q_g + (long)t * BQ * q.rs, q.rs, q_rows, qreg);
    __syncthreads();
    stage_store(qreg, q_lds);
    stage_store_t(qreg, qt_lds);
    stage_tile(do_g + (long)t * BQ * dout.rs, dout.rs, q_rows, do_lds);
    if (HAS_BIAS) {
      stage_tile_rowstride(bias_g + (long)t * BQ * Lk + (long)ktile * BK,
                           Lk, q_rows, min(BK, Lk - ktile * BK), b_lds);
    }
    if (threadIdx.x < BQ) {
      const int qq = t * BQ + threadIdx.x;
      lse_lds[threadIdx.x] = (threadIdx.x < q_rows) ? lse_g[qq] : NEG_INF;
      delta_lds[threadIdx.x] = (threadIdx.x < q_rows) ? delta_g[qq] : 0.f;
    }
    __syncthreads();

    // S^T = K Q^T ; P^T
    f32x4 pt[4], dpt[4];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 qf = frag_row(q_lds, c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(k_frag[dblk], qf, acc,
                                                      0, 0, 0);
      }
      pt[c] = acc;
    }
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int qcol = c * 16 + (lane & 15);
      const float l = lse_lds[qcol];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kvrow = wave * 16 + (lane >> 4) * 4 + reg;
        float val = pt[c][reg] * scale;
        if (HAS_BIAS && krow_ok_c[reg] && qcol < q_rows)
          val += to_f32(*reinterpret_cast<const bf16_t*>(
              b_lds + swz(qcol, kvrow * (int)sizeof(bf16_t))));
        pt[c][reg] = (krow_ok_c[reg] && qcol < q_rows && l > NEG_INF)
            ? __expf(val - l) : 0.f;
      }
    }

    // dP^T = V dO^T
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 dof = frag_row(do_lds, c * 16 + (lane & 15), dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(v_frag[dblk], dof, acc,
                                                      0, 0, 0);
      }
      dpt[c] = acc;
    }

    // dS^T = P^T (dP^T - delta[q]) ; stage for dK (+ dBias drain).
    // the scratch aliases the bias tile — barrier off the bias reads
    if (HAS_BIAS) __syncthreads();
    char* sw = s_lds[wave];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int qcol = c * 16 + (lane & 15);
      const float dl = delta_lds[qcol];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float ds_total = pt[c][reg] * (dpt[c][reg] - dl);
        const int row = (lane >> 4) * 4 + reg;
        *reinterpret_cast<bf16_t*>(sw + swz(row, qcol * (int)sizeof(bf16_t)))
            = (bf16_t)ds_total;
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    bf16x8 dst_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk)
      dst_frag[kblk] = frag_row(sw, lane & 15, kblk);

    if (NEED_DBIAS) {
      __syncthreads();
      const int kvr = lane;
      // unrolling this drain costs ~30 VGPRs of address computation —
      // keep it rolled (occupancy is this kernel's lever)
#pragma clang loop unroll(disable)
      for (int j = 0; j < 16; ++j) {
        const int qq = j * NWAVES + wave;
        if (qq < q_rows && kvr < kv_rows) {
          float dsv = to_f32(*reinterpret_cast<const bf16_t*>(
              s_lds[kvr >> 4] + swz(kvr & 15, qq * (int)sizeof(bf16_t))));
          float* addr = dbias_g + (long)(t * BQ + qq) * Lk
              + (long)ktile * BK + kvr;
          if (bias_repeat == 1) *addr = dsv; else atomicAdd(addr, dsv);
        }
      }
    }

    // dK += dS^T Q
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = dk_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 qf = frag_row(qt_lds, c * 16 + (lane & 15), kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dst_frag[kblk], qf, acc,
                                                      0, 0, 0);
      }
      dk_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);
  }

  bf16_t* dk_g = dk.base(batch, head) + (long)ktile * BK * dk.rs;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + (lane >> 4) * 4 + reg;
    if (row < kv_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c)
        dk_g[(long)row * dk.rs + c * 16 + (lane & 15)] =
            (bf16_t)(dk_acc[c][reg] * scale);
    }
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// host launchers

namespace {

TView make_view(const at::Tensor& t) {
  TORCH_CHECK(t.dim() == 4 && t.size(3) == DH && t.stride(3) == 1,
              "attention tensors must be 4-D (B, h, L, 64), last dim dense");
  TORCH_CHECK(t.stride(2) % 8 == 0 && t.stride(1) % 8 == 0,
              "attention tensor strides must be 16-byte aligned");
  return TView{reinterpret_cast<const bf16_t*>(t.data_ptr()),
               t.stride(0), t.stride(1), t.stride(2)};
}

TViewMut make_view_mut(at::Tensor& t) {
  TView v = make_view(t);
  return TViewMut{const_cast<bf16_t*>(v.p), v.bs, v.hs, v.rs};
}

}  // namespace

std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                 c10::optional<at::Tensor> bias,
                                 c10::optional<at::Tensor> mask,
                                 long bias_repeat, double scale,
                                 long q_repeat) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16, "attn_fwd: bf16 only");
  const int B = k.size(0), H = q.size(1), Lq = q.size(2), Lk = k.size(2);
  TORCH_CHECK(q.size(0) * q_repeat == B,
              "q batch must be B / q_repeat");

  // out in (B, Lq, H, DH) memory (the layout downstream Linears want),
  // exposed as a (B, H, Lq, DH) strided view
  auto out_bnhd = at::empty({B, Lq, H, DH}, q.options());
  auto out = out_bnhd.permute({0, 2, 1, 3});
  auto lse = at::empty({B, H, Lq}, q.options().dtype(at::kFloat));

  const bool has_bias = bias.has_value();
  const bool has_mask = mask.has_value();
  at::Tensor bias_c, mask_u8;
  if (has_bias) bias_c = bias->contiguous();
  if (has_mask) {
    // the kernel indexes mask[batch * Lk + j]: a broadcast (1, Lk) mask
    // at B > 1 would read out of bounds (advisory r01)
    TORCH_CHECK(mask->dim() == 2 && mask->size(0) == B &&
                mask->size(1) == Lk,
                "attn_fwd: mask must be (B, Lk); expand broadcast masks "
                "host-side");
    mask_u8 = mask->to(at::kByte).contiguous();
  }

  dim3 grid((Lq + FBQ - 1) / FBQ, B * H);
  auto stream = at::cuda::getCurrentHIPStream();
  TView qv = make_view(q), kv = make_view(k), vv = make_view(v);
  TViewMut ov = make_view_mut(out);

#define DISPATCH(HB, HM)                                                      \
  hipLaunchKernelGGL((attn_fwd_kernel<HB, HM>), grid, dim3(FNT), 0, stream,   \
      qv, kv, vv,                                                             \
      has_bias ? reinterpret_cast<const bf16_t*>(bias_c.data_ptr()) : nullptr,\
      has_mask ? mask_u8.data_ptr<unsigned char>() : nullptr,                 \
      ov, lse.data_ptr<float>(),                                              \
      Lq, Lk, H, (int)bias_repeat, (int)q_repeat, (float)scale)

  if (has_bias && has_mask) DISPATCH(true, true);
  else if (has_bias) DISPATCH(true, false);
  else if (has_mask) DISPATCH(false, true);
  else DISPATCH(false, false);
#undef DISPATCH
  return {out, lse};
}

std::vector<at::Tensor> attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                 at::Tensor v, at::Tensor out, at::Tensor lse,
                                 c10::optional<at::Tensor> bias,
                                 c10::optional<at::Tensor> mask,
                                 long bias_repeat, double scale,
                                 bool need_dbias, long q_repeat,
                                 c10::optional<at::Tensor> dq_out,
                                 c10::optional<at::Tensor> dk_out,
                                 c10::optional<at::Tensor> dv_out) {
  const int B = k.size(0), H = q.size(1), Lq = q.size(2), Lk = k.size(2);
  TORCH_CHECK(q.size(0) * q_repeat == B,
              "q batch must be B / q_repeat");
  if (dout.stride(3) != 1) dout = dout.contiguous();

  auto delta = at::empty({B, H, Lq}, lse.options());
  // gradients written through strided views — either caller-provided
  // slices of a packed buffer (no concatenation in autograd) or fresh
  // (B, L, H, DH) memory viewed (B, H, L, DH)
  at::Tensor dq, dk, dv;
  if (dq_out.has_value()) {
    dq = *dq_out; dk = *dk_out; dv = *dv_out;
  } else {
    dq = at::empty({B, Lq, H, DH}, q.options()).permute({0, 2, 1, 3});
    dk = at::empty({B, Lk, H, DH}, q.options()).permute({0, 2, 1, 3});
    dv = at::empty({B, Lk, H, DH}, q.options()).permute({0, 2, 1, 3});
  }
  at::Tensor dbias;
  // chunks > 1 shortens same-address atomic chains but the extra
  // zeros+sum traffic measured NET-NEGATIVE (tri bwd 4.0 -> 4.3 ms at
  // chunks=8, tools/attn_bench.py) — keep a single copy
  int dbias_chunks = 1;
  if (need_dbias) {
    dbias = at::zeros({dbias_chunks, B / bias_repeat, H, Lq, Lk},
                      q.options().dtype(at::kFloat));
  }
  const bool has_bias = bias.has_value();
  const bool has_mask = mask.has_value();
  at::Tensor bias_c, mask_u8;
  if (has_bias) bias_c = bias->contiguous();
  if (has_mask) {
    TORCH_CHECK(mask->dim() == 2 && mask->size(0) == B &&
                mask->size(1) == Lk,
                "attn_bwd: mask must be (B, Lk); expand broadcast masks "
                "host-side");
    mask_u8 = mask->to(at::kByte).contiguous();
  }

  auto stream = at::cuda::getCurrentHIPStream();
  TView qv = make_view(q), kvv = make_view(k), vv = make_view(v);
  TView dov = make_view(dout), outv = make_view(out);
  TViewMut dqv = make_view_mut(dq), dkv = make_view_mut(dk),
           dvv = make_view_mut(dv);

  {
    const long rows = (long)B * H * Lq;
    const int block = 256;
    const long grid = (rows * 8 + block - 1) / block;
    hipLaunchKernelGGL(attn_delta_kernel, dim3(grid), dim3(block), 0, stream,
                       dov, outv, delta.data_ptr<float>(), H, Lq, rows);
  }

  dim3 grid_q((Lq + BQ - 1) / BQ, B * H);
  dim3 grid_k((Lk + BK - 1) / BK, B * H);

#define DISPATCH_DQ(HB, HM)                                                   \
  hipLaunchKernelGGL((attn_bwd_dq_kernel<HB, HM>), grid_q, dim3(256), 0,      \
      stream, qv, kvv, vv,                                                    \
      has_bias ? reinterpret_cast<const bf16_t*>(bias_c.data_ptr()) : nullptr,\
      has_mask ? mask_u8.data_ptr<unsigned char>() : nullptr,                 \
      dov, lse.data_ptr<float>(), delta.data_ptr<float>(), dqv,               \
      Lq, Lk, H, (int)bias_repeat, (int)q_repeat, (float)scale)

  if (has_bias && has_mask) DISPATCH_DQ(true, true);
  else if (has_bias) DISPATCH_DQ(true, false);
  else if (has_mask) DISPATCH_DQ(false, true);
  else DISPATCH_DQ(false, false);
#undef DISPATCH_DQ

#define DISPATCH_DKV(HB, HM, DB)                                              \
  hipLaunchKernelGGL((attn_bwd_dkv_kernel<HB, HM, DB>), grid_k, dim3(256), 0, \
      stream, qv, kvv, vv,                                                    \
      has_bias ? reinterpret_cast<const bf16_t*>(bias_c.data_ptr()) : nullptr,\
      has_mask ? mask_u8.data_ptr<unsigned char>() : nullptr,                 \
      dov, lse.data_ptr<float>(), delta.data_ptr<float>(), dkv, dvv,          \
      DB ? dbias.data_ptr<float>() : nullptr,                                 \
      dbias_chunks, need_dbias ? dbias.stride(0) : 0L,                        \
      Lq, Lk, H, (int)bias_repeat, (int)q_repeat, (float)scale)

#define DISPATCH_DV(HB, HM)                                                   \
  hipLaunchKernelGGL((attn_bwd_dv_kernel<HB, HM>), grid_k, dim3(256), 0,      \
      stream, qv, kvv,                                                        \
      has_bias ? reinterpret_cast<const bf16_t*>(bias_c.data_ptr()) : nullptr,\
      has_mask ? mask_u8.data_ptr<unsigned char>() : nullptr,                 \
      dov, lse.data_ptr<float>(), dvv,                                        \
      Lq, Lk, H, (int)bias_repeat, (int)q_repeat, (float)scale)

#define DISPATCH_DK(HB, HM, DB)                                               \
  hipLaunchKernelGGL((attn_bwd_dk_kernel<HB, HM, DB>), grid_k, dim3(256), 0,  \
      stream, qv, kvv, vv,                                                    \
      has_bias ? reinterpret_cast<const bf16_t*>(bias_c.data_ptr()) : nullptr,\
      has_mask ? mask_u8.data_ptr<unsigned char>() : nullptr,                 \
      dov, lse.data_ptr<float>(), delta.data_ptr<float>(), dkv,               \
      DB ? dbias.data_ptr<float>() : nullptr,                                 \
      dbias_chunks, need_dbias ? dbias.stride(0) : 0L,                        \
      Lq, Lk, H, (int)bias_repeat, (int)q_repeat, (float)scale)

  // split dV/dK passes (4/3 waves per SIMD, but prefetch-free and
  // +40% flops): measured SLOWER end-to-end than the combined kernel
  // (748 vs 699 ms/step, r02) — combined stays default;
  // AF2AMD_SPLIT_DKV=1 re-enables for tuning.
  static const bool split_dkv = [] {
    const char* e = getenv("AF2AMD_SPLIT_DKV");
    return e != nullptr && e[0] == '1';
  }();

  if (split_dkv) {
    if (has_bias && has_mask) DISPATCH_DV(true, true);
    else if (has_bias) DISPATCH_DV(true, false);
    else if (has_mask) DISPATCH_DV(false, true);
    else DISPATCH_DV(false, false);
    if (need_dbias) {
      if (has_mask) DISPATCH_DK(true, true, true);
      else DISPATCH_DK(true, false, true);
    } else if (has_bias && has_mask) DISPATCH_DK(true, true, false);
    else if (has_bias) DISPATCH_DK(true, false, false);
    else if (has_mask) DISPATCH_DK(false, true, false);
    else DISPATCH_DK(false, false, false);
  } else {
    if (need_dbias) {
      if (has_mask) DISPATCH_DKV(true, true, true);
      else DISPATCH_DKV(true, false, true);
    } else if (has_bias && has_mask) DISPATCH_DKV(true, true, false);
    else if (has_bias) DISPATCH_DKV(true, false, false);
    else if (has_mask) DISPATCH_DKV(false, true, false);
    else DISPATCH_DKV(false, false, false);
  }
#undef DISPATCH_DKV
#undef DISPATCH_DV
#undef DISPATCH_DK

  std::vector<at::Tensor> ret = {dq, dk, dv};
  if (need_dbias) ret.push_back(dbias.sum(0));
  return ret;
}
