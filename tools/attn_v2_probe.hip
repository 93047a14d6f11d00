// Prototype: swapped-QK^T / in-register-softmax attention forward
// (docs/ROADMAP.md Appendix B) — NOT YET HW-VALIDATED, not wired into
// the production library.  Round-2 first GPU call:
//
//   hipcc --offload-arch=gfx950 -O3 tools/attn_v2_probe.hip -o /tmp/v2 \
//     && /tmp/v2
//
// prints PASS/FAIL vs a CPU fp32 reference for {bias,mask} x shapes,
// plus TF rates at the production triangle-attention shape to compare
// against tools/attn_bench.py (end of round 1: fwd ~190 TF, dq
// 1.37 ms, dkv 4.0 ms per call at batch 5).
//
// Contains all THREE kernels of the v2 design (every dataflow
// machine-verified on CPU by tests/test_v2_layout.py):
//   attn_fwd_v2   94 VGPRs / 48KB LDS   (prod: 128 / 64KB)
//   attn_dq_v2   126 VGPRs / 80KB LDS   (prod: 178, 12 waves/CU -> 16)
//   attn_dkv_v2  122 VGPRs / 64KB LDS   (prod: 249, 8 waves/CU -> 16)
// Known prototype gaps vs production semantics: no bias_repeat fold
// (pass pre-expanded bias), no dbias accumulation in dkv (the
// production drain logic ports unchanged — dS is available per-lane at
// the marked spot), and dkv's per-lane bias loads should be LDS-staged
// in the production port.
//
// Key derivation (verified on paper against the HW-probed fragment
// layouts in tools/mfma_probe.hip — C/D: col=lane&15,
// row=(lane>>4)*4+reg; A/B: k=(lane>>4)*8+j):
//
//  * The SWAP IS FREE: today's q_frag (row=lane&15 over the wave's 16 q,
//    k-slice (lane>>4)*8+j) is exactly a valid B-operand, and today's
//    K fragment is exactly a valid A-operand, so
//        s = mfma(kf, qf)        (operands swapped, fragments unchanged)
//    yields S[kv=(lane>>4)*4+reg + 16c][q=lane&15]: each lane holds 16
//    kv-scores FOR ONE q-row.
//  * Row softmax = in-lane reduce over 16 + shfl_xor(16) + shfl_xor(32)
//    (the 4 lanes holding the same q sit 16 apart) — replaces the
//    4-step shfl ladder per reg of the production kernel, and the
//    state (m, l) ends up replicated across those 4 lanes.
//  * P -> PV A-fragment needs kv chunks {2g, 2g+1} (+8 per kblk) in
//    lane-group g; lane-group g' holds chunk c at c&3 == g'.  So the
//    redistribution is 8 cvt_pk pack pairs + 8 __shfl per KV tile —
//    the production P->LDS round-trip and its lgkmcnt wait disappear.
//  * PV's B-operand (V^T rows) and the C/D epilogue layout are
//    unchanged from production; the per-q softmax state just has to be
//    shfl'd from lane q when rescaling O rows (q=(lane>>4)*4+reg).
#include <hip/hip_runtime.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define HIP_CHECK(x)                                                   \
  do {                                                                 \
    hipError_t e_ = (x);                                               \
    if (e_ != hipSuccess) {                                            \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e_),\
              __FILE__, __LINE__);                                     \
      exit(1);                                                         \
    }                                                                  \
  } while (0)

constexpr int DH = 64;
constexpr int BK = 64;
constexpr int FBQ = 128;   // q rows per block (8 waves x 16)
constexpr int FNT = 512;
constexpr int ROWB = DH * sizeof(bf16_t);
constexpr float NEG_INF = -1e30f;

__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return row * ROWB + (byte_in_row ^ ((row & 7) << 4));
}

__device__ __forceinline__ bf16x8 frag_row(const char* lds, int row,
                                           int kblk) {
  const int lane = threadIdx.x & 63;
  int byte_in_row = kblk * 64 + ((lane >> 4) << 4);
  return *reinterpret_cast<const bf16x8*>(lds + swz(row, byte_in_row));
}

// ---- staging (same scheme as the production kernel) ----------------------
template <int R>
struct StageRegs {
  float4 v[(R * 8 + FNT - 1) / FNT];
};

template <int R>
__device__ __forceinline__ void stage_load(const bf16_t* __restrict__ g,
                                           long rs, int rows,
                                           StageRegs<R>& r) {
  constexpr int P = (R * 8 + FNT - 1) / FNT;
#pragma unroll
  for (int pass = 0; pass < P; ++pass) {
    int idx = threadIdx.x + pass * FNT;
    int row = idx >> 3, c16 = (idx & 7) << 4;
    float4 val = {0, 0, 0, 0};
    if (row < rows)
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(g + row * rs) + c16);
    r.v[pass] = val;
  }
}

template <int R>
__device__ __forceinline__ void stage_store(const StageRegs<R>& r,
                                            char* lds) {
  constexpr int P = (R * 8 + FNT - 1) / FNT;
#pragma unroll
  for (int pass = 0; pass < P; ++pass) {
    int idx = threadIdx.x + pass * FNT;
    *reinterpret_cast<float4*>(lds + swz(idx >> 3, (idx & 7) << 4)) =
        r.v[pass];
  }
}

__device__ __forceinline__ void stage_load_colwise(
    const bf16_t* __restrict__ g, long rs, int rows, StageRegs<BK>& r) {
#pragma unroll
  for (int pass = 0; pass < (BK * 8 + FNT - 1) / FNT; ++pass) {
    int idx = threadIdx.x + pass * FNT;
    int row = idx & 63, c16 = (idx >> 6) << 4;
    float4 val = {0, 0, 0, 0};
    if (row < rows)
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(g + row * rs) + c16);
    r.v[pass] = val;
  }
}

__device__ __forceinline__ void stage_store_t_colwise(
    const StageRegs<BK>& r, char* lds) {
#pragma unroll
  for (int pass = 0; pass < (BK * 8 + FNT - 1) / FNT; ++pass) {
    int idx = threadIdx.x + pass * FNT;
    int row = idx & 63, c8 = (idx >> 6) << 3;
    const float4 val = r.v[pass];
    const bf16_t* vv = reinterpret_cast<const bf16_t*>(&val);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      *reinterpret_cast<bf16_t*>(
          lds + swz(c8 + j, row * (int)sizeof(bf16_t))) = vv[j];
  }
}

// pack two f32 into a u32 of 2 bf16 (round-to-nearest via cvt)
__device__ __forceinline__ unsigned pack_bf16(float a, float b) {
  union {
    bf16_t h[2];
    unsigned u;
  } cv;
  cv.h[0] = (bf16_t)a;
  cv.h[1] = (bf16_t)b;
  return cv.u;
}

// ---- v2 forward kernel ---------------------------------------------------
template <bool HAS_BIAS, bool HAS_MASK>
__global__ __launch_bounds__(FNT, 4)
void attn_fwd_v2(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                 const bf16_t* __restrict__ v,
                 const bf16_t* __restrict__ bias,
                 const unsigned char* __restrict__ mask,
                 bf16_t* __restrict__ out, float* __restrict__ lse,
                 int B, int Lq, int Lk, float scale) {
  __shared__ char q_lds[FBQ * ROWB];
  __shared__ char k_lds[2][BK * ROWB];
  __shared__ char vt_lds[2][BK * ROWB];
  __shared__ unsigned char m_lds[2][BK];

  const int qtile = blockIdx.x;
  const int batch = blockIdx.y;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int g = lane >> 4;              // lane group 0..3
  const int myq = lane & 15;            // this lane's q row (within wave)

  const bf16_t* q_g = q + ((long)batch * Lq + (long)qtile * FBQ) * DH;
  const bf16_t* k_g = k + (long)batch * Lk * DH;
  const bf16_t* v_g = v + (long)batch * Lk * DH;
  const bf16_t* bias_g =
      HAS_BIAS ? bias + ((long)batch * Lq + (long)qtile * FBQ) * Lk
               : nullptr;

  const int q_rows = min(FBQ, Lq - qtile * FBQ);
  {
    StageRegs<FBQ> qr;
    stage_load<FBQ>(q_g, DH, q_rows, qr);
    stage_store<FBQ>(qr, q_lds);
  }
  __syncthreads();

  bf16x8 q_frag[2];
#pragma unroll
  for (int dblk = 0; dblk < 2; ++dblk)
    q_frag[dblk] = frag_row(q_lds, wave * 16 + myq, dblk);

  // per-lane softmax state for q row `myq` (replicated across the 4
  // lane groups after each cross-lane reduce)
  float m_i = NEG_INF, l_i = 0.f;
  f32x4 o_acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) o_acc[c] = f32x4{0, 0, 0, 0};

  const bf16_t* brow =
      HAS_BIAS ? bias_g + (long)(wave * 16 + myq) * Lk : nullptr;
  const bool q_ok = (wave * 16 + myq) < q_rows;

  const int n_kv = (Lk + BK - 1) / BK;
  StageRegs<BK> kreg, vreg;
  stage_load<BK>(k_g, DH, min(BK, Lk), kreg);
  stage_load_colwise(v_g, DH, min(BK, Lk), vreg);

  for (int t = 0; t < n_kv; ++t) {
    const int kv_rows = min(BK, Lk - t * BK);
    const int buf = t & 1;
    stage_store<BK>(kreg, k_lds[buf]);
    stage_store_t_colwise(vreg, vt_lds[buf]);
    if (HAS_MASK && threadIdx.x < BK)
      m_lds[buf][threadIdx.x] = (threadIdx.x < kv_rows)
          ? mask[(long)batch * Lk + t * BK + threadIdx.x] : 0;
    if (t + 1 < n_kv) {
      const int nr = min(BK, Lk - (t + 1) * BK);
      stage_load<BK>(k_g + (long)(t + 1) * BK * DH, DH, nr, kreg);
      stage_load_colwise(v_g + (long)(t + 1) * BK * DH, DH, nr, vreg);
    }
    __syncthreads();

    // S = K Q^T (swapped): s[c][reg] = S[kv = c*16 + g*4 + reg][myq]
    f32x4 s[4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 kf = frag_row(k_lds[buf], c * 16 + myq, dblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, q_frag[dblk],
                                                      acc, 0, 0, 0);
      }
      s[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // scale + bias + key mask (per lane: fixed q row, 16 kv values)
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kv = c * 16 + g * 4 + reg;
        const bool ok = kv < kv_rows && (!HAS_MASK || m_lds[buf][kv]);
        float val = s[c][reg] * scale;
        if (HAS_BIAS && ok && q_ok) val += (float)brow[t * BK + kv];
        s[c][reg] = ok ? val : NEG_INF;
      }
    }

    // online softmax: in-lane over 16, cross-lane over the 4 groups
    float tmax = NEG_INF;
#pragma unroll
    for (int c = 0; c < 4; ++c)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) tmax = fmaxf(tmax, s[c][reg]);
    tmax = fmaxf(tmax, __shfl_xor(tmax, 16, 64));
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));

    const float m_new = fmaxf(m_i, tmax);
    const float alpha = (m_i <= NEG_INF) ? 0.f : __expf(m_i - m_new);
    float tsum = 0.f;
#pragma unroll
    for (int c = 0; c < 4; ++c)
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const float p = (m_new <= NEG_INF) ? 0.f : __expf(s[c][reg] - m_new);
        s[c][reg] = p;
        tsum += p;
      }
    tsum += __shfl_xor(tsum, 16, 64);
    tsum += __shfl_xor(tsum, 32, 64);
    l_i = l_i * alpha + tsum;
    m_i = m_new;

    // rescale O rows: o_acc row q' = g*4 + reg needs alpha of lane q'
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const float a_row = __shfl(alpha, (g << 2) + reg, 64);
#pragma unroll
      for (int c = 0; c < 4; ++c) o_acc[c][reg] *= a_row;
    }

    // P -> A-fragments via cvt_pk + shfl (no LDS round-trip).
    // Target lane (q=myq, group g) needs kv = g*8 + j (+32*kblk);
    // chunk c_src = kv>>4 in {2kblk, 2kblk+1}, source group (kv>>2)&3.
    bf16x8 a_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk) {
      // every lane packs its two candidate chunks for this kblk
      const int ca = 2 * kblk, cb = 2 * kblk + 1;
      const unsigned pa0 = pack_bf16(s[ca][0], s[ca][1]);
      const unsigned pa1 = pack_bf16(s[ca][2], s[ca][3]);
      const unsigned pb0 = pack_bf16(s[cb][0], s[cb][1]);
      const unsigned pb1 = pack_bf16(s[cb][2], s[cb][3]);
      // kv0 = g*8 + 32*kblk; low 4 values live in chunk kv0>>2..,
      // c_src = kv0>>4: g<2 -> ca, g>=2 -> cb; groups (2g)&3, (2g+1)&3
      const int src_lo = myq + 16 * ((2 * g) & 3);
      const int src_hi = myq + 16 * ((2 * g + 1) & 3);
      unsigned lo0a = __shfl(pa0, src_lo, 64), lo1a = __shfl(pa1, src_lo, 64);
      unsigned hi0a = __shfl(pa0, src_hi, 64), hi1a = __shfl(pa1, src_hi, 64);
      unsigned lo0b = __shfl(pb0, src_lo, 64), lo1b = __shfl(pb1, src_lo, 64);
      unsigned hi0b = __shfl(pb0, src_hi, 64), hi1b = __shfl(pb1, src_hi, 64);
      union {
        unsigned u[4];
        bf16x8 f;
      } af;
      const bool use_b = g >= 2;
      af.u[0] = use_b ? lo0b : lo0a;
      af.u[1] = use_b ? lo1b : lo1a;
      af.u[2] = use_b ? hi0b : hi0a;
      af.u[3] = use_b ? hi1b : hi1a;
      a_frag[kblk] = af.f;
    }

    // O += P V (B = V^T rows, unchanged from production)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = o_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 vf = frag_row(vt_lds[buf], c * 16 + myq, kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag[kblk], vf,
                                                      acc, 0, 0, 0);
      }
      o_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // epilogue: O rows q' = g*4+reg; fetch that row's l from lane q'
  bf16_t* out_g = out + ((long)batch * Lq + (long)qtile * FBQ) * DH;
  float* lse_g = lse + (long)batch * Lq + (long)qtile * FBQ;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + g * 4 + reg;
    const float l_row = __shfl(l_i, (g << 2) + reg, 64);
    const float linv = l_row > 0.f ? 1.f / l_row : 0.f;
    if (row < q_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c)
        out_g[(long)row * DH + c * 16 + myq] =
            (bf16_t)(o_acc[c][reg] * linv);
    }
  }
  if (g == 0 && (wave * 16 + myq) < q_rows)
    lse_g[wave * 16 + myq] =
        (l_i > 0.f) ? m_i + logf(l_i) : NEG_INF;
}

// simple transposed store (row-wise mapping; fine for a prototype)
__device__ __forceinline__ void stage_store_t(const StageRegs<BK>& r,
                                              char* lds) {
#pragma unroll
  for (int pass = 0; pass < (BK * 8 + FNT - 1) / FNT; ++pass) {
    int idx = threadIdx.x + pass * FNT;
    int row = idx >> 3, c8 = (idx & 7) << 3;
    const float4 val = r.v[pass];
    const bf16_t* vv = reinterpret_cast<const bf16_t*>(&val);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      *reinterpret_cast<bf16_t*>(
          lds + swz(c8 + j, row * (int)sizeof(bf16_t))) = vv[j];
  }
}

// ---- v2 backward-dQ kernel (dataflow CPU-verified by
// tests/test_v2_layout.py::test_v2_dq_dataflow) ----------------------------
// Swapped orientation: per lane q=lane&15, kv spread over the groups.
// P is recomputed from lse, dS = P*(dP - delta) — both lse and delta
// are SINGLE per-lane scalars (the production kernel tracks 4 per-reg
// values), there are NO cross-lane reductions at all, and the dS
// C->A relayout is the same cvt_pk+shfl exchange as the forward
// (replacing the production s_lds round-trip).
template <bool HAS_BIAS>
__global__ __launch_bounds__(FNT, 4)
void attn_dq_v2(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                const bf16_t* __restrict__ v,
                const bf16_t* __restrict__ bias,
                const bf16_t* __restrict__ dout,
                const float* __restrict__ lse,
                const float* __restrict__ delta,
                bf16_t* __restrict__ dq_out,
                int B, int Lq, int Lk, float scale) {
  __shared__ char q_lds[FBQ * ROWB];
  __shared__ char do_lds[FBQ * ROWB];
  __shared__ char k_lds[2][BK * ROWB];
  __shared__ char kt_lds[2][BK * ROWB];
  __shared__ char v_lds[2][BK * ROWB];

  const int qtile = blockIdx.x;
  const int batch = blockIdx.y;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int g = lane >> 4;
  const int myq = lane & 15;

  const bf16_t* q_g = q + ((long)batch * Lq + (long)qtile * FBQ) * DH;
  const bf16_t* do_g = dout + ((long)batch * Lq + (long)qtile * FBQ) * DH;
  const bf16_t* k_g = k + (long)batch * Lk * DH;
  const bf16_t* v_g = v + (long)batch * Lk * DH;
  const float* lse_g = lse + (long)batch * Lq + (long)qtile * FBQ;
  const float* dl_g = delta + (long)batch * Lq + (long)qtile * FBQ;
  const bf16_t* bias_g =
      HAS_BIAS ? bias + ((long)batch * Lq + (long)qtile * FBQ) * Lk
               : nullptr;

  const int q_rows = min(FBQ, Lq - qtile * FBQ);
  {
    StageRegs<FBQ> qr, dor;
    stage_load<FBQ>(q_g, DH, q_rows, qr);
    stage_load<FBQ>(do_g, DH, q_rows, dor);
    stage_store<FBQ>(qr, q_lds);
    stage_store<FBQ>(dor, do_lds);
  }
  __syncthreads();

  bf16x8 q_frag[2], do_frag[2];
#pragma unroll
  for (int dblk = 0; dblk < 2; ++dblk) {
    q_frag[dblk] = frag_row(q_lds, wave * 16 + myq, dblk);
    do_frag[dblk] = frag_row(do_lds, wave * 16 + myq, dblk);
  }

  const bool q_ok = (wave * 16 + myq) < q_rows;
  const float lse_l = q_ok ? lse_g[wave * 16 + myq] : NEG_INF;
  const float dl_l = q_ok ? dl_g[wave * 16 + myq] : 0.f;
  const bf16_t* brow =
      HAS_BIAS ? bias_g + (long)(wave * 16 + myq) * Lk : nullptr;

  f32x4 dq_acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) dq_acc[c] = f32x4{0, 0, 0, 0};

  const int n_kv = (Lk + BK - 1) / BK;
  StageRegs<BK> kreg, vreg;
  stage_load<BK>(k_g, DH, min(BK, Lk), kreg);
  stage_load<BK>(v_g, DH, min(BK, Lk), vreg);

  for (int t = 0; t < n_kv; ++t) {
    const int kv_rows = min(BK, Lk - t * BK);
    const int buf = t & 1;
    stage_store<BK>(kreg, k_lds[buf]);
    stage_store_t(kreg, kt_lds[buf]);   // one load, both layouts
    stage_store<BK>(vreg, v_lds[buf]);
    if (t + 1 < n_kv) {
      const int nr = min(BK, Lk - (t + 1) * BK);
      stage_load<BK>(k_g + (long)(t + 1) * BK * DH, DH, nr, kreg);
      stage_load<BK>(v_g + (long)(t + 1) * BK * DH, DH, nr, vreg);
    }
    __syncthreads();

    // S^T and dP^T in swapped orientation
    f32x4 s[4], dp[4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 sa = {0, 0, 0, 0}, da = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        bf16x8 kf = frag_row(k_lds[buf], c * 16 + myq, dblk);
        bf16x8 vf = frag_row(v_lds[buf], c * 16 + myq, dblk);
        sa = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf, q_frag[dblk], sa,
                                                     0, 0, 0);
        da = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf, do_frag[dblk], da,
                                                     0, 0, 0);
      }
      s[c] = sa;
      dp[c] = da;
    }
    __builtin_amdgcn_s_setprio(0);

    // dS = P * (dP - delta), all per-lane
#pragma unroll
    for (int c = 0; c < 4; ++c) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kv = c * 16 + g * 4 + reg;
        float val = s[c][reg] * scale;
        if (HAS_BIAS && q_ok && kv < kv_rows) val += (float)brow[t * BK + kv];
        const float p = (kv < kv_rows && lse_l > NEG_INF)
            ? __expf(val - lse_l) : 0.f;
        s[c][reg] = p * (dp[c][reg] - dl_l);
      }
    }

    // dS C->A exchange (identical pattern to the forward P exchange)
    bf16x8 ds_frag[2];
#pragma unroll
    for (int kblk = 0; kblk < 2; ++kblk) {
      const int ca = 2 * kblk, cb = 2 * kblk + 1;
      const unsigned pa0 = pack_bf16(s[ca][0], s[ca][1]);
      const unsigned pa1 = pack_bf16(s[ca][2], s[ca][3]);
      const unsigned pb0 = pack_bf16(s[cb][0], s[cb][1]);
      const unsigned pb1 = pack_bf16(s[cb][2], s[cb][3]);
      const int src_lo = myq + 16 * ((2 * g) & 3);
      const int src_hi = myq + 16 * ((2 * g + 1) & 3);
      unsigned lo0a = __shfl(pa0, src_lo, 64), lo1a = __shfl(pa1, src_lo, 64);
      unsigned hi0a = __shfl(pa0, src_hi, 64), hi1a = __shfl(pa1, src_hi, 64);
      unsigned lo0b = __shfl(pb0, src_lo, 64), lo1b = __shfl(pb1, src_lo, 64);
      unsigned hi0b = __shfl(pb0, src_hi, 64), hi1b = __shfl(pb1, src_hi, 64);
      union {
        unsigned u[4];
        bf16x8 f;
      } af;
      const bool use_b = g >= 2;
      af.u[0] = use_b ? lo0b : lo0a;
      af.u[1] = use_b ? lo1b : lo1a;
      af.u[2] = use_b ? hi0b : hi0a;
      af.u[3] = use_b ? hi1b : hi1a;
      ds_frag[kblk] = af.f;
    }

    // dQ += dS K  (B = K^T rows)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      f32x4 acc = dq_acc[c];
#pragma unroll
      for (int kblk = 0; kblk < 2; ++kblk) {
        bf16x8 ktf = frag_row(kt_lds[buf], c * 16 + myq, kblk);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ds_frag[kblk], ktf,
                                                      acc, 0, 0, 0);
      }
      dq_acc[c] = acc;
    }
    __builtin_amdgcn_s_setprio(0);
  }

  bf16_t* dq_g = dq_out + ((long)batch * Lq + (long)qtile * FBQ) * DH;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + g * 4 + reg;
    if (row < q_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c)
        dq_g[(long)row * DH + c * 16 + myq] =
            (bf16_t)(dq_acc[c][reg] * scale);
    }
  }
}

// ---- CPU reference -------------------------------------------------------
static void cpu_ref(const std::vector<float>& q, const std::vector<float>& k,
                    const std::vector<float>& v, const std::vector<float>& b,
                    const std::vector<unsigned char>& msk, bool has_bias,
                    bool has_mask, int B, int Lq, int Lk, float scale,
                    std::vector<float>& out) {
  std::vector<float> srow(Lk);
  for (int bb = 0; bb < B; ++bb)
    for (int i = 0; i < Lq; ++i) {
      float mx = -1e30f;
      for (int j = 0; j < Lk; ++j) {
        bool ok = !has_mask || msk[bb * Lk + j];
        float s = 0.f;
        for (int d = 0; d < DH; ++d)
          s += q[((long)bb * Lq + i) * DH + d] *
               k[((long)bb * Lk + j) * DH + d];
        s *= scale;
        if (has_bias && ok) s += b[((long)bb * Lq + i) * Lk + j];
        srow[j] = ok ? s : -1e30f;
        mx = fmaxf(mx, srow[j]);
      }
      float l = 0.f;
      for (int j = 0; j < Lk; ++j) {
        srow[j] = (mx <= -1e30f) ? 0.f : expf(srow[j] - mx);
        l += srow[j];
      }
      const float linv = l > 0.f ? 1.f / l : 0.f;
      for (int d = 0; d < DH; ++d) {
        float acc = 0.f;
        for (int j = 0; j < Lk; ++j)
          acc += srow[j] * v[((long)bb * Lk + j) * DH + d];
        out[((long)bb * Lq + i) * DH + d] = acc * linv;
      }
    }
}

template <typename T>
static T* to_dev(const std::vector<T>& h) {
  T* d;
  HIP_CHECK(hipMalloc(&d, h.size() * sizeof(T)));
  HIP_CHECK(hipMemcpy(d, h.data(), h.size() * sizeof(T),
                      hipMemcpyHostToDevice));
  return d;
}

// replicate the first Bu batches across the full batch axis so big
// timing shapes don't need a full-size CPU reference (the kernel output
// for batch b is then checked against reference batch b % Bu)
template <typename T>
static void tile_batches(std::vector<T>& v, long per, int Bu, int B) {
  for (int b = Bu; b < B; ++b)
    memcpy(v.data() + (long)b * per, v.data() + (long)(b % Bu) * per,
           per * sizeof(T));
}

static int run_case(int B, int Lq, int Lk, bool has_bias, bool has_mask,
                    bool timing) {
  const float scale = 1.f / sqrtf((float)DH);
  const int Bu = timing ? (B < 4 ? B : 4) : B;  // unique batches
  srand(12345);
  auto rnd = [&]() { return (rand() / (float)RAND_MAX - 0.5f) * 2.f; };

  std::vector<float> qf((long)B * Lq * DH), kf((long)B * Lk * DH),
      vf((long)B * Lk * DH), bf(has_bias ? (long)B * Lq * Lk : 1);
  std::vector<unsigned char> mk(has_mask ? (long)B * Lk : 1, 1);
  for (long i = 0; i < (long)Bu * Lq * DH; ++i) qf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lk * DH; ++i) kf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lk * DH; ++i) vf[i] = rnd();
  if (has_bias)
    for (long i = 0; i < (long)Bu * Lq * Lk; ++i) bf[i] = rnd();
  if (has_mask)
    for (long i = 0; i < (long)Bu * Lk; ++i)
      mk[i] = (i % Lk == 0) ? 1 : (rnd() > -0.6f);  // keep >=1 key valid
  tile_batches(qf, (long)Lq * DH, Bu, B);
  tile_batches(kf, (long)Lk * DH, Bu, B);
  tile_batches(vf, (long)Lk * DH, Bu, B);
  if (has_bias) tile_batches(bf, (long)Lq * Lk, Bu, B);
  if (has_mask) tile_batches(mk, (long)Lk, Bu, B);

  auto to_bf = [](const std::vector<float>& s) {
    std::vector<bf16_t> o(s.size());
    for (size_t i = 0; i < s.size(); ++i) o[i] = (bf16_t)s[i];
    return o;
  };
  // quantize inputs to bf16 before the CPU reference so only the
  // compute path differs
  auto qb = to_bf(qf), kb = to_bf(kf), vb = to_bf(vf), bb = to_bf(bf);
  for (size_t i = 0; i < qf.size(); ++i) qf[i] = (float)qb[i];
  for (size_t i = 0; i < kf.size(); ++i) kf[i] = (float)kb[i];
  for (size_t i = 0; i < vf.size(); ++i) vf[i] = (float)vb[i];
  for (size_t i = 0; i < bf.size(); ++i) bf[i] = (float)bb[i];

  bf16_t *dq = to_dev(qb), *dk = to_dev(kb), *dv = to_dev(vb),
         *db = to_dev(bb);
  unsigned char* dm = to_dev(mk);
  bf16_t* dout;
  float* dlse;
  HIP_CHECK(hipMalloc(&dout, (long)B * Lq * DH * sizeof(bf16_t)));
  HIP_CHECK(hipMalloc(&dlse, (long)B * Lq * sizeof(float)));

  dim3 grid((Lq + FBQ - 1) / FBQ, B), block(FNT);
#define LAUNCH(HB, HM)                                                  \
  hipLaunchKernelGGL((attn_fwd_v2<HB, HM>), grid, block, 0, 0, dq, dk,  \
                     dv, db, dm, dout, dlse, B, Lq, Lk, scale)
  if (has_bias && has_mask) LAUNCH(true, true);
  else if (has_bias) LAUNCH(true, false);
  else if (has_mask) LAUNCH(false, true);
  else LAUNCH(false, false);
  HIP_CHECK(hipDeviceSynchronize());

  std::vector<bf16_t> outb((long)B * Lq * DH);
  HIP_CHECK(hipMemcpy(outb.data(), dout, outb.size() * sizeof(bf16_t),
                      hipMemcpyDeviceToHost));
  std::vector<float> ref((long)Bu * Lq * DH);
  cpu_ref(qf, kf, vf, bf, mk, has_bias, has_mask, Bu, Lq, Lk, scale, ref);
  float err = 0.f;
  const long per_out = (long)Lq * DH;
  for (int b = 0; b < B; ++b)
    for (long r = 0; r < per_out; ++r)
      err = fmaxf(err, fabsf((float)outb[(long)b * per_out + r] -
                             ref[(long)(b % Bu) * per_out + r]));
  const bool pass = err < 3e-2f;
  printf("B=%d Lq=%d Lk=%d bias=%d mask=%d  max_err=%.4f  %s\n", B, Lq,
         Lk, has_bias, has_mask, err, pass ? "PASS" : "FAIL");

  if (timing && pass) {
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    for (int i = 0; i < 5; ++i) LAUNCH(true, false);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipEventRecord(e0));
    const int iters = 50;
    for (int i = 0; i < iters; ++i) LAUNCH(true, false);
    HIP_CHECK(hipEventRecord(e1));
    HIP_CHECK(hipEventSynchronize(e1));
    float ms;
    HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
    ms /= iters;
    const double fl = 4.0 * B * (double)Lq * Lk * DH;  // QK^T + PV
    printf("  timing: %.3f ms  %.1f TF  (production fwd ~190 TF at this "
           "shape)\n", ms, fl / (ms * 1e-3) / 1e12);
  }
#undef LAUNCH
  hipFree(dq); hipFree(dk); hipFree(dv); hipFree(db); hipFree(dm);
  hipFree(dout); hipFree(dlse);
  return pass ? 0 : 1;
}

// ---- v2 backward-dK/dV kernel (dataflow CPU-verified by
// tests/test_v2_layout.py::test_v2_dkv_dataflow) ---------------------------
// STANDARD orientation: block owns BKV=128 kv rows (8 waves x 16), q
// streams through in 32-row chunks.  Per lane: kv = wave*16 + (lane&15)
// fixed; S/dP arrive in C layout [q][kv-col]; P and dS are formed
// per-lane from lse/delta, then the SAME cvt_pk+shfl exchange used by
// the forward turns their column layout into A-fragments (row=kv,
// k=q32) for the dV/dK MFMAs — replacing the production kernel's two
// LDS round-trips.  dBias is omitted in the prototype (the production
// drain logic ports unchanged: dS is available per-lane right here).
// B-fragment over a 32-long q k-axis from a [dh][q] transposed tile
__device__ __forceinline__ bf16x8 fragq_row32(const char* lds, int dhrow) {
  const int lane = threadIdx.x & 63;
  return *reinterpret_cast<const bf16x8*>(
      lds + dhrow * 64 + ((lane >> 4) << 4));
}

constexpr int BKV = 128;   // kv rows per block
constexpr int QC = 32;     // q rows per iteration (mfma k-axis)

template <bool HAS_BIAS, bool HAS_MASK>
__global__ __launch_bounds__(FNT, 4)
void attn_dkv_v2(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                 const bf16_t* __restrict__ v,
                 const bf16_t* __restrict__ bias,
                 const unsigned char* __restrict__ mask,
                 const bf16_t* __restrict__ dout,
                 const float* __restrict__ lse,
                 const float* __restrict__ delta,
                 bf16_t* __restrict__ dk_out, bf16_t* __restrict__ dv_out,
                 int B, int Lq, int Lk, float scale) {
  __shared__ char k_lds[BKV * ROWB];      // block's kv tile, row-major
  __shared__ char v_lds[BKV * ROWB];
  __shared__ char q_lds[2][QC * ROWB];    // q chunk, row-major (S A-op)
  __shared__ char do_lds[2][QC * ROWB];   // dO chunk, row-major (dP A-op)
  __shared__ char qt_lds[2][DH * (QC * 2)];   // Q^T  [dh][q] (dK B-op)
  __shared__ char dot_lds[2][DH * (QC * 2)];  // dO^T [dh][q] (dV B-op)

  const int kvtile = blockIdx.x;
  const int batch = blockIdx.y;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int g = lane >> 4;
  const int mykv = lane & 15;                // kv col within the wave tile
  const int kv_row = wave * 16 + mykv;       // within the block tile

  const bf16_t* k_g = k + ((long)batch * Lk + (long)kvtile * BKV) * DH;
  const bf16_t* v_g = v + ((long)batch * Lk + (long)kvtile * BKV) * DH;
  const bf16_t* q_g = q + (long)batch * Lq * DH;
  const bf16_t* do_g = dout + (long)batch * Lq * DH;
  const float* lse_g = lse + (long)batch * Lq;
  const float* dl_g = delta + (long)batch * Lq;
  const bf16_t* bias_g = HAS_BIAS
      ? bias + (long)batch * Lq * Lk + (long)kvtile * BKV : nullptr;

  const int kv_rows = min(BKV, Lk - kvtile * BKV);
  const bool kv_ok = kv_row < kv_rows &&
      (!HAS_MASK ||
       mask[(long)batch * Lk + kvtile * BKV + kv_row]);

  {  // stage the block's K/V tiles once (128 rows each)
    StageRegs<BKV> kr, vr;
    stage_load<BKV>(k_g, DH, kv_rows, kr);
    stage_load<BKV>(v_g, DH, kv_rows, vr);
    stage_store<BKV>(kr, k_lds);
    stage_store<BKV>(vr, v_lds);
  }
  __syncthreads();

  f32x4 dk_acc[4], dv_acc[4];
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    dk_acc[c] = f32x4{0, 0, 0, 0};
    dv_acc[c] = f32x4{0, 0, 0, 0};
  }

  // stage a 32-row q/dO chunk in BOTH layouts: threads 0..255 move the
  // q tile, 256..511 the dO tile (32 rows x 8 16B chunks each).  The
  // row-major copy is swizzled (frag_row applies swz on reads); the
  // transposed [dh][q] copy has 64B rows — a b128 read of 16 rows hits
  // the 8-slot LDS floor with or without a swizzle, so it stays plain.
  auto stage_chunk = [&](const bf16_t* gq, const bf16_t* gdo, int rows,
                         int buf) {
    const int half = threadIdx.x >> 8;
    const int idx = threadIdx.x & 255;
    const int row = idx >> 3;            // 0..31
    const int c16 = (idx & 7) << 4;      // 16B chunk in the 128B row
    const bf16_t* gsrc = half ? gdo : gq;
    char* rm = half ? do_lds[buf] : q_lds[buf];
    char* tr = half ? dot_lds[buf] : qt_lds[buf];
    float4 val = {0, 0, 0, 0};
    if (row < rows)
      val = *reinterpret_cast<const float4*>(
          reinterpret_cast<const char*>(gsrc + row * DH) + c16);
    *reinterpret_cast<float4*>(rm + swz(row, c16)) = val;
    const bf16_t* vv = reinterpret_cast<const bf16_t*>(&val);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int dh = (c16 >> 1) + j;
      *reinterpret_cast<bf16_t*>(
          tr + dh * (QC * 2) + row * (int)sizeof(bf16_t)) = vv[j];
    }
  };

  const int n_qc = (Lq + QC - 1) / QC;
  stage_chunk(q_g, do_g, min(QC, Lq), 0);
  __syncthreads();

  for (int qc = 0; qc < n_qc; ++qc) {
    const int buf = qc & 1;
    const int q_base = qc * QC;
    const int q_rows = min(QC, Lq - q_base);
    if (qc + 1 < n_qc) {
      stage_chunk(q_g + (long)(qc + 1) * QC * DH,
                  do_g + (long)(qc + 1) * QC * DH,
                  min(QC, Lq - (qc + 1) * QC), 1 - buf);
    }
    // NOTE prototype keeps 1 barrier before compute; the prefetch above
    // writes the OTHER buffer so this is the same single-barrier scheme
    __syncthreads();

    // S and dP chunks in C layout: per lane [q=16qt+g*4+reg][kv=mykv]
    f32x4 s_cols[2], dp_cols[2];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
      f32x4 sa = {0, 0, 0, 0}, da = {0, 0, 0, 0};
#pragma unroll
      for (int dblk = 0; dblk < 2; ++dblk) {
        // A = Q / dO rows (16 q), B = K / V rows as columns (16 kv)
        bf16x8 qa = frag_row(q_lds[buf], qt * 16 + mykv, dblk);
        bf16x8 doa = frag_row(do_lds[buf], qt * 16 + mykv, dblk);
        bf16x8 kb = frag_row(k_lds, wave * 16 + mykv, dblk);
        bf16x8 vb = frag_row(v_lds, wave * 16 + mykv, dblk);
        sa = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qa, kb, sa, 0, 0, 0);
        da = __builtin_amdgcn_mfma_f32_16x16x32_bf16(doa, vb, da, 0, 0, 0);
      }
      s_cols[qt] = sa;
      dp_cols[qt] = da;
    }
    __builtin_amdgcn_s_setprio(0);

    // per-lane P and dS columns
    f32x4 p_cols[2], ds_cols[2];
#pragma unroll
    for (int qt = 0; qt < 2; ++qt) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int qi = q_base + qt * 16 + g * 4 + reg;
        const bool q_in = qi < Lq && (qt * 16 + g * 4 + reg) < q_rows;
        float val = s_cols[qt][reg] * scale;
        if (HAS_BIAS && q_in && kv_ok)
          val += (float)bias_g[(long)qi * Lk + kv_row];
        const float lse_l = q_in ? lse_g[qi] : NEG_INF;
        const float p = (kv_ok && q_in && lse_l > NEG_INF)
            ? __expf(val - lse_l) : 0.f;
        p_cols[qt][reg] = p;
        ds_cols[qt][reg] = p * (dp_cols[qt][reg] - (q_in ? dl_g[qi] : 0.f));
      }
    }

    // exchange columns -> A-fragments (row = kv, k = the 32 q rows)
    auto exchange = [&](const f32x4* cols) -> bf16x8 {
      const unsigned p00 = pack_bf16(cols[0][0], cols[0][1]);
      const unsigned p01 = pack_bf16(cols[0][2], cols[0][3]);
      const unsigned p10 = pack_bf16(cols[1][0], cols[1][1]);
      const unsigned p11 = pack_bf16(cols[1][2], cols[1][3]);
      const int src_lo = mykv + 16 * ((2 * g) & 3);
      const int src_hi = mykv + 16 * ((2 * g + 1) & 3);
      unsigned lo00 = __shfl(p00, src_lo, 64), lo01 = __shfl(p01, src_lo, 64);
      unsigned hi00 = __shfl(p00, src_hi, 64), hi01 = __shfl(p01, src_hi, 64);
      unsigned lo10 = __shfl(p10, src_lo, 64), lo11 = __shfl(p11, src_lo, 64);
      unsigned hi10 = __shfl(p10, src_hi, 64), hi11 = __shfl(p11, src_hi, 64);
      union {
        unsigned u[4];
        bf16x8 f;
      } af;
      const bool use1 = g >= 2;   // q chunks {2g,2g+1}: qtile = chunk>=4
      af.u[0] = use1 ? lo10 : lo00;
      af.u[1] = use1 ? lo11 : lo01;
      af.u[2] = use1 ? hi10 : hi00;
      af.u[3] = use1 ? hi11 : hi01;
      return af.f;
    };
    bf16x8 p_frag = exchange(p_cols);
    bf16x8 ds_frag = exchange(ds_cols);

    // dV += P^T dO ; dK += dS^T Q   (B = transposed chunks, k = q)
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 dob = fragq_row32(dot_lds[buf], c * 16 + mykv);
      bf16x8 qb = fragq_row32(qt_lds[buf], c * 16 + mykv);
      dv_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          p_frag, dob, dv_acc[c], 0, 0, 0);
      dk_acc[c] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          ds_frag, qb, dk_acc[c], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
  }

  // epilogue: rows kv' = wave*16 + g*4 + reg
  bf16_t* dk_g = dk_out + ((long)batch * Lk + (long)kvtile * BKV) * DH;
  bf16_t* dv_g = dv_out + ((long)batch * Lk + (long)kvtile * BKV) * DH;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = wave * 16 + g * 4 + reg;
    if (row < kv_rows) {
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        dk_g[(long)row * DH + c * 16 + mykv] =
            (bf16_t)(dk_acc[c][reg] * scale);
        dv_g[(long)row * DH + c * 16 + mykv] = (bf16_t)dv_acc[c][reg];
      }
    }
  }
}

static int run_dq_case(int B, int Lq, int Lk, bool has_bias,
                       bool timing) {
  const float scale = 1.f / sqrtf((float)DH);
  const int Bu = timing ? (B < 4 ? B : 4) : B;  // unique batches
  srand(777);
  auto rnd = [&]() { return (rand() / (float)RAND_MAX - 0.5f) * 2.f; };

  std::vector<float> qf((long)B * Lq * DH), kf((long)B * Lk * DH),
      vf((long)B * Lk * DH), dof((long)B * Lq * DH),
      bf(has_bias ? (long)B * Lq * Lk : 1);
  for (long i = 0; i < (long)Bu * Lq * DH; ++i) qf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lk * DH; ++i) kf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lk * DH; ++i) vf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lq * DH; ++i) dof[i] = rnd();
  if (has_bias)
    for (long i = 0; i < (long)Bu * Lq * Lk; ++i) bf[i] = rnd();
  tile_batches(qf, (long)Lq * DH, Bu, B);
  tile_batches(kf, (long)Lk * DH, Bu, B);
  tile_batches(vf, (long)Lk * DH, Bu, B);
  tile_batches(dof, (long)Lq * DH, Bu, B);
  if (has_bias) tile_batches(bf, (long)Lq * Lk, Bu, B);

  auto to_bf = [](std::vector<float>& s) {
    std::vector<bf16_t> o(s.size());
    for (size_t i = 0; i < s.size(); ++i) {
      o[i] = (bf16_t)s[i];
      s[i] = (float)o[i];
    }
    return o;
  };
  auto qb = to_bf(qf), kb = to_bf(kf), vb = to_bf(vf), dob = to_bf(dof),
       bb = to_bf(bf);

  // CPU: lse, delta, and reference dQ (fp32 over bf16-quantized inputs)
  // — unique batches only; lse/delta are tiled up for the kernel input
  std::vector<float> lse((long)B * Lq), delta((long)B * Lq),
      dq_ref((long)Bu * Lq * DH, 0.f);
  std::vector<float> srow(Lk), prow(Lk);
  for (int b = 0; b < Bu; ++b)
    for (int i = 0; i < Lq; ++i) {
      float mx = -1e30f;
      for (int j = 0; j < Lk; ++j) {
        float s = 0.f;
        for (int d = 0; d < DH; ++d)
          s += qf[((long)b * Lq + i) * DH + d] *
               kf[((long)b * Lk + j) * DH + d];
        s *= scale;
        if (has_bias) s += bf[((long)b * Lq + i) * Lk + j];
        srow[j] = s;
        mx = fmaxf(mx, s);
      }
      float l = 0.f;
      for (int j = 0; j < Lk; ++j) {
        prow[j] = expf(srow[j] - mx);
        l += prow[j];
      }
      lse[(long)b * Lq + i] = mx + logf(l);
      // O row and delta
      float dl = 0.f;
      for (int d = 0; d < DH; ++d) {
        float o = 0.f;
        for (int j = 0; j < Lk; ++j)
          o += prow[j] / l * vf[((long)b * Lk + j) * DH + d];
        dl += o * dof[((long)b * Lq + i) * DH + d];
      }
      delta[(long)b * Lq + i] = dl;
      for (int j = 0; j < Lk; ++j) {
        float dp = 0.f;
        for (int d = 0; d < DH; ++d)
          dp += dof[((long)b * Lq + i) * DH + d] *
                vf[((long)b * Lk + j) * DH + d];
        const float ds = prow[j] / l * (dp - dl) * scale;
        for (int d = 0; d < DH; ++d)
          dq_ref[((long)b * Lq + i) * DH + d] +=
              ds * kf[((long)b * Lk + j) * DH + d];
      }
    }

  tile_batches(lse, (long)Lq, Bu, B);
  tile_batches(delta, (long)Lq, Bu, B);
  bf16_t *dq_ = to_dev(qb), *dk_ = to_dev(kb), *dv_ = to_dev(vb),
         *ddo = to_dev(dob), *db_ = to_dev(bb);
  float *dlse = to_dev(lse), *ddelta = to_dev(delta);
  bf16_t* dout_;
  HIP_CHECK(hipMalloc(&dout_, (long)B * Lq * DH * sizeof(bf16_t)));

  dim3 grid((Lq + FBQ - 1) / FBQ, B), block(FNT);
#define LAUNCH_DQ(HB)                                                     hipLaunchKernelGGL((attn_dq_v2<HB>), grid, block, 0, 0, dq_, dk_,                          dv_, db_, ddo, dlse, ddelta, dout_, B, Lq, Lk,                          scale)
  if (has_bias) LAUNCH_DQ(true); else LAUNCH_DQ(false);
  HIP_CHECK(hipDeviceSynchronize());

  std::vector<bf16_t> outb((long)B * Lq * DH);
  HIP_CHECK(hipMemcpy(outb.data(), dout_, outb.size() * sizeof(bf16_t),
                      hipMemcpyDeviceToHost));
  float err = 0.f, ref_max = 0.f;
  const long per_dq = (long)Lq * DH;
  for (int b = 0; b < B; ++b)
    for (long r = 0; r < per_dq; ++r) {
      const float rv = dq_ref[(long)(b % Bu) * per_dq + r];
      err = fmaxf(err, fabsf((float)outb[(long)b * per_dq + r] - rv));
      ref_max = fmaxf(ref_max, fabsf(rv));
    }
  // dS goes through bf16 before the dQ GEMM; tolerance scales with |dq|
  const bool pass = err < 6e-2f * fmaxf(1.f, ref_max);
  printf("dq: B=%d Lq=%d Lk=%d bias=%d  max_err=%.4f (ref_max %.2f)  %s\n",
         B, Lq, Lk, has_bias, err, ref_max, pass ? "PASS" : "FAIL");

  if (timing && pass) {
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    for (int i = 0; i < 5; ++i) LAUNCH_DQ(true);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipEventRecord(e0));
    const int iters = 50;
    for (int i = 0; i < iters; ++i) LAUNCH_DQ(true);
    HIP_CHECK(hipEventRecord(e1));
    HIP_CHECK(hipEventSynchronize(e1));
    float ms;
    HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
    ms /= iters;
    const double fl = 6.0 * B * (double)Lq * Lk * DH;  // S + dP + dQ
    printf("  dq timing: %.3f ms  %.1f TF  (production dq ~1.37 ms at "
           "this shape)\n", ms, fl / (ms * 1e-3) / 1e12);
  }
#undef LAUNCH_DQ
  hipFree(dq_); hipFree(dk_); hipFree(dv_); hipFree(ddo); hipFree(db_);
  hipFree(dlse); hipFree(ddelta); hipFree(dout_);
  return pass ? 0 : 1;
}

static int run_dkv_case(int B, int Lq, int Lk, bool has_bias,
                        bool has_mask, bool timing) {
  const float scale = 1.f / sqrtf((float)DH);
  const int Bu = timing ? (B < 4 ? B : 4) : B;  // unique batches
  srand(4242);
  auto rnd = [&]() { return (rand() / (float)RAND_MAX - 0.5f) * 2.f; };

  std::vector<float> qf((long)B * Lq * DH), kf((long)B * Lk * DH),
      vf((long)B * Lk * DH), dof((long)B * Lq * DH),
      bf(has_bias ? (long)B * Lq * Lk : 1);
  std::vector<unsigned char> mk(has_mask ? (long)B * Lk : 1, 1);
  for (long i = 0; i < (long)Bu * Lq * DH; ++i) qf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lk * DH; ++i) kf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lk * DH; ++i) vf[i] = rnd();
  for (long i = 0; i < (long)Bu * Lq * DH; ++i) dof[i] = rnd();
  if (has_bias)
    for (long i = 0; i < (long)Bu * Lq * Lk; ++i) bf[i] = rnd();
  if (has_mask)
    for (long i = 0; i < (long)Bu * Lk; ++i)
      mk[i] = (i % Lk == 0) ? 1 : (rnd() > -0.6f);
  tile_batches(qf, (long)Lq * DH, Bu, B);
  tile_batches(kf, (long)Lk * DH, Bu, B);
  tile_batches(vf, (long)Lk * DH, Bu, B);
  tile_batches(dof, (long)Lq * DH, Bu, B);
  if (has_bias) tile_batches(bf, (long)Lq * Lk, Bu, B);
  if (has_mask) tile_batches(mk, (long)Lk, Bu, B);

  auto to_bf = [](std::vector<float>& s) {
    std::vector<bf16_t> o(s.size());
    for (size_t i = 0; i < s.size(); ++i) {
      o[i] = (bf16_t)s[i];
      s[i] = (float)o[i];
    }
    return o;
  };
  auto qb = to_bf(qf), kb = to_bf(kf), vb = to_bf(vf), dob = to_bf(dof),
       bb = to_bf(bf);

  // CPU: lse/delta + reference dK/dV — unique batches only
  std::vector<float> lse((long)B * Lq), delta((long)B * Lq),
      dk_ref((long)Bu * Lk * DH, 0.f), dv_ref((long)Bu * Lk * DH, 0.f);
  std::vector<float> srow(Lk), prow(Lk);
  for (int b = 0; b < Bu; ++b)
    for (int i = 0; i < Lq; ++i) {
      float mx = -1e30f;
      for (int j = 0; j < Lk; ++j) {
        bool ok = !has_mask || mk[(long)b * Lk + j];
        float s = 0.f;
        for (int d = 0; d < DH; ++d)
          s += qf[((long)b * Lq + i) * DH + d] *
               kf[((long)b * Lk + j) * DH + d];
        s *= scale;
        if (has_bias && ok) s += bf[((long)b * Lq + i) * Lk + j];
        srow[j] = ok ? s : -1e30f;
        mx = fmaxf(mx, srow[j]);
      }
      float l = 0.f;
      for (int j = 0; j < Lk; ++j) {
        prow[j] = (mx <= -1e30f) ? 0.f : expf(srow[j] - mx);
        l += prow[j];
      }
      lse[(long)b * Lq + i] = (l > 0.f) ? mx + logf(l) : -1e30f;
      float dl = 0.f;
      for (int d = 0; d < DH; ++d) {
        float o = 0.f;
        for (int j = 0; j < Lk; ++j)
          o += (l > 0 ? prow[j] / l : 0.f) *
               vf[((long)b * Lk + j) * DH + d];
        dl += o * dof[((long)b * Lq + i) * DH + d];
      }
      delta[(long)b * Lq + i] = dl;
      for (int j = 0; j < Lk; ++j) {
        const float p = l > 0 ? prow[j] / l : 0.f;
        float dp = 0.f;
        for (int d = 0; d < DH; ++d)
          dp += dof[((long)b * Lq + i) * DH + d] *
                vf[((long)b * Lk + j) * DH + d];
        const float ds = p * (dp - dl);
        for (int d = 0; d < DH; ++d) {
          dv_ref[((long)b * Lk + j) * DH + d] +=
              p * dof[((long)b * Lq + i) * DH + d];
          dk_ref[((long)b * Lk + j) * DH + d] +=
              scale * ds * qf[((long)b * Lq + i) * DH + d];
        }
      }
    }

  tile_batches(lse, (long)Lq, Bu, B);
  tile_batches(delta, (long)Lq, Bu, B);
  bf16_t *dq_ = to_dev(qb), *dk_ = to_dev(kb), *dv_ = to_dev(vb),
         *ddo = to_dev(dob), *db_ = to_dev(bb);
  unsigned char* dm = to_dev(mk);
  float *dlse = to_dev(lse), *ddelta = to_dev(delta);
  bf16_t *dko, *dvo;
  HIP_CHECK(hipMalloc(&dko, (long)B * Lk * DH * sizeof(bf16_t)));
  HIP_CHECK(hipMalloc(&dvo, (long)B * Lk * DH * sizeof(bf16_t)));

  dim3 grid((Lk + BKV - 1) / BKV, B), block(FNT);
#define LAUNCH_DKV(HB, HM)                                                hipLaunchKernelGGL((attn_dkv_v2<HB, HM>), grid, block, 0, 0, dq_,                          dk_, dv_, db_, dm, ddo, dlse, ddelta, dko, dvo,                         B, Lq, Lk, scale)
  if (has_bias && has_mask) LAUNCH_DKV(true, true);
  else if (has_bias) LAUNCH_DKV(true, false);
  else if (has_mask) LAUNCH_DKV(false, true);
  else LAUNCH_DKV(false, false);
  HIP_CHECK(hipDeviceSynchronize());

  std::vector<bf16_t> dkb((long)B * Lk * DH), dvb((long)B * Lk * DH);
  HIP_CHECK(hipMemcpy(dkb.data(), dko, dkb.size() * sizeof(bf16_t),
                      hipMemcpyDeviceToHost));
  HIP_CHECK(hipMemcpy(dvb.data(), dvo, dvb.size() * sizeof(bf16_t),
                      hipMemcpyDeviceToHost));
  float err_k = 0.f, err_v = 0.f, mx_k = 0.f, mx_v = 0.f;
  const long per_kv = (long)Lk * DH;
  for (int b = 0; b < B; ++b)
    for (long r = 0; r < per_kv; ++r) {
      const long ri = (long)(b % Bu) * per_kv + r, oi = (long)b * per_kv + r;
      err_k = fmaxf(err_k, fabsf((float)dkb[oi] - dk_ref[ri]));
      err_v = fmaxf(err_v, fabsf((float)dvb[oi] - dv_ref[ri]));
      mx_k = fmaxf(mx_k, fabsf(dk_ref[ri]));
      mx_v = fmaxf(mx_v, fabsf(dv_ref[ri]));
    }
  const bool pass = err_k < 6e-2f * fmaxf(1.f, mx_k) &&
                    err_v < 6e-2f * fmaxf(1.f, mx_v);
  printf("dkv: B=%d Lq=%d Lk=%d bias=%d mask=%d  dK_err=%.4f dV_err=%.4f"
         "  %s\n", B, Lq, Lk, has_bias, has_mask, err_k, err_v,
         pass ? "PASS" : "FAIL");

  if (timing && pass) {
    hipEvent_t e0, e1;
    HIP_CHECK(hipEventCreate(&e0));
    HIP_CHECK(hipEventCreate(&e1));
    for (int i = 0; i < 5; ++i) LAUNCH_DKV(true, false);
    HIP_CHECK(hipDeviceSynchronize());
    HIP_CHECK(hipEventRecord(e0));
    const int iters = 50;
    for (int i = 0; i < iters; ++i) LAUNCH_DKV(true, false);
    HIP_CHECK(hipEventRecord(e1));
    HIP_CHECK(hipEventSynchronize(e1));
    float ms;
    HIP_CHECK(hipEventElapsedTime(&ms, e0, e1));
    ms /= iters;
    const double fl = 8.0 * B * (double)Lq * Lk * DH;  // S+dP+dV+dK
    printf("  dkv timing: %.3f ms  %.1f TF  (production dkv ~2.6 ms "
           "at this shape, incl. dbias)\n", ms,
           fl / (ms * 1e-3) / 1e12);
  }
#undef LAUNCH_DKV
  hipFree(dq_); hipFree(dk_); hipFree(dv_); hipFree(ddo); hipFree(db_);
  hipFree(dm); hipFree(dlse); hipFree(ddelta); hipFree(dko); hipFree(dvo);
  return pass ? 0 : 1;
}

int main() {
  setvbuf(stdout, nullptr, _IONBF, 0);  // survive a timeout kill
  int rc = 0;
  rc |= run_case(4, 128, 128, false, false, false);
  rc |= run_case(4, 128, 128, true, false, false);
  rc |= run_case(4, 128, 128, false, true, false);
  rc |= run_case(3, 100, 72, true, true, false);
  rc |= run_case(2, 64, 257, true, true, false);
  // production triangle-attention shape (b=5, n=256, h=8 folded into
  // the batch dim: 5*256*8 = 10240)
  rc |= run_case(10240, 256, 256, true, false, true);
  // backward-dQ prototype
  rc |= run_dq_case(4, 128, 128, false, false);
  rc |= run_dq_case(3, 100, 72, true, false);
  rc |= run_dq_case(10240, 256, 256, true, true);
  // backward-dK/dV prototype
  rc |= run_dkv_case(4, 128, 128, false, false, false);
  rc |= run_dkv_case(3, 100, 72, true, true, false);
  rc |= run_dkv_case(2, 257, 64, true, false, false);
  rc |= run_dkv_case(10240, 256, 256, true, false, true);
  printf(rc == 0 ? "ALL PASS\n" : "FAILURES PRESENT\n");
  return rc;
}
