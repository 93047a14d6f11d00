"""Lane-exact simulation of the swapped-QK^T attention dataflow
(tools/attn_v2_probe.hip / docs/ROADMAP.md Appendix B).

Simulates one wave's register file through the full v2 forward — MFMA
fragment layouts (HW-verified by tools/mfma_probe.hip), the swapped
operand order, the per-lane softmax with cross-lane reduction, the
cvt_pk+shfl P redistribution, PV, and the epilogue state shuffles — and
checks the result against plain softmax(QK^T*s + bias)V.  A bug in any
index derivation fails this test on CPU, before any GPU time is spent.

Layout conventions under test (A/B: k=(lane>>4)*8+j; C/D: col=lane&15,
row=(lane>>4)*4+reg):
  * swapped QK^T: s[c][reg] = S[kv = c*16 + (lane>>4)*4 + reg][q=lane&15]
  * P->A exchange: lane group g needs kv chunks {2g, 2g+1} (+8/kblk),
    held by groups (2g)&3 / (2g+1)&3 at chunk index c = kv>>4
  * O rows are q=(lane>>4)*4+reg, softmax state lives at lane q
"""
import numpy as np

NL = 64  # lanes per wave
DH = 64
NQ = 16  # q rows per wave
NKV = 64  # kv per tile


def mfma_16x16x32(a_regs, b_regs):
    """Simulate one mfma_f32_16x16x32: per-lane fragments -> per-lane D.

    a_regs/b_regs: (NL, 8) lane registers.
    A[row][k]: row = lane&15, k = (lane>>4)*8 + j
    B[k][col]: col = lane&15, k = (lane>>4)*8 + j
    D[row][col]: col = lane&15, row = (lane>>4)*4 + reg  -> (NL, 4)
    """
    A = np.zeros((16, 32))
    B = np.zeros((32, 16))
    for lane in range(NL):
        for j in range(8):
            A[lane & 15, (lane >> 4) * 8 + j] = a_regs[lane, j]
            B[(lane >> 4) * 8 + j, lane & 15] = b_regs[lane, j]
    D = A @ B
    d_regs = np.zeros((NL, 4))
    for lane in range(NL):
        for reg in range(4):
            d_regs[lane, reg] = D[(lane >> 4) * 4 + reg, lane & 15]
    return d_regs


def k_fragment(K, ktile, dblk):
    """A-operand fragment of K rows [ktile*16, +16), dh slice dblk*32."""
    regs = np.zeros((NL, 8))
    for lane in range(NL):
        for j in range(8):
            regs[lane, j] = K[ktile * 16 + (lane & 15),
                              dblk * 32 + (lane >> 4) * 8 + j]
    return regs


def q_fragment(Q, dblk):
    """B-operand fragment of the wave's 16 q rows (col=lane&15)."""
    regs = np.zeros((NL, 8))
    for lane in range(NL):
        for j in range(8):
            regs[lane, j] = Q[lane & 15, dblk * 32 + (lane >> 4) * 8 + j]
    return regs


def vt_fragment(V, ctile, kblk):
    """B-operand fragment of V^T rows = dh [ctile*16,+16), k = kv slice."""
    regs = np.zeros((NL, 8))
    for lane in range(NL):
        for j in range(8):
            regs[lane, j] = V[kblk * 32 + (lane >> 4) * 8 + j,
                              ctile * 16 + (lane & 15)]
    return regs


def shfl(vals, src_lane_per_lane):
    return np.array([vals[src_lane_per_lane[l]] for l in range(NL)])


def shfl_xor(vals, mask):
    return np.array([vals[l ^ mask] for l in range(NL)])


def test_v2_dataflow_matches_attention():
    rng = np.random.default_rng(0)
    Q = rng.standard_normal((NQ, DH))
    K = rng.standard_normal((NKV, DH))
    V = rng.standard_normal((NKV, DH))
    bias = rng.standard_normal((NQ, NKV))
    scale = DH ** -0.5

    # ---- reference ----
    S_ref = Q @ K.T * scale + bias
    P_ref = np.exp(S_ref - S_ref.max(axis=1, keepdims=True))
    O_ref = (P_ref / P_ref.sum(axis=1, keepdims=True)) @ V

    # ---- simulated v2 wave program ----
    qf = [q_fragment(Q, dblk) for dblk in range(2)]

    # swapped QK^T: s[c] = mfma(A=K-frag, B=Q-frag)
    s = []
    for c in range(4):
        acc = np.zeros((NL, 4))
        for dblk in range(2):
            acc += mfma_16x16x32(k_fragment(K, c, dblk), qf[dblk])
        s.append(acc)
    s = np.stack(s, axis=0)  # (4, NL, 4)

    lanes = np.arange(NL)
    g = lanes >> 4
    myq = lanes & 15

    # scale + bias: lane holds S[kv=c*16+g*4+reg][myq]
    for c in range(4):
        for reg in range(4):
            kv = c * 16 + g * 4 + reg
            s[c, :, reg] = s[c, :, reg] * scale + bias[myq, kv]
            # cross-check the claimed S layout itself
            assert np.allclose(s[c, :, reg], S_ref[myq, kv], atol=1e-9)

    # per-lane softmax (single tile: no online rescale needed, but the
    # cross-lane reduction pattern is what's under test)
    tmax = s.max(axis=(0, 2))
    tmax = np.maximum(tmax, shfl_xor(tmax, 16))
    tmax = np.maximum(tmax, shfl_xor(tmax, 32))
    p = np.exp(s - tmax[None, :, None])
    tsum = p.sum(axis=(0, 2))
    tsum = tsum + shfl_xor(tsum, 16)
    tsum = tsum + shfl_xor(tsum, 32)
    # state now replicated across the 4 groups of each q
    assert np.allclose(tmax, S_ref[myq].max(axis=1))
    assert np.allclose(tsum, P_ref[myq].sum(axis=1))

    # P -> A-fragment exchange (cvt_pk pairs modeled as value pairs)
    a_frag = np.zeros((2, NL, 8))
    for kblk in range(2):
        ca, cb = 2 * kblk, 2 * kblk + 1
        packs = {  # what each lane would pack for this kblk
            ('a', 0): p[ca, :, 0:2], ('a', 1): p[ca, :, 2:4],
            ('b', 0): p[cb, :, 0:2], ('b', 1): p[cb, :, 2:4],
        }
        src_lo = myq + 16 * ((2 * g) & 3)
        src_hi = myq + 16 * ((2 * g + 1) & 3)
        for which, src in (('lo', src_lo), ('hi', src_hi)):
            base = 0 if which == 'lo' else 4
            for half in range(2):
                got_a = shfl(packs[('a', half)], src)
                got_b = shfl(packs[('b', half)], src)
                use_b = (g >= 2)[:, None]
                a_frag[kblk, :, base + 2 * half: base + 2 * half + 2] = \
                    np.where(use_b, got_b, got_a)
    # a_frag[kblk] must be the A-operand of P: A[row=myq][k=g*8+j+32kblk]
    for kblk in range(2):
        for j in range(8):
            kv = g * 8 + j + 32 * kblk
            assert np.allclose(a_frag[kblk, :, j], P_ref[myq, kv] /
                               np.exp(S_ref[myq].max(axis=1) - tmax),
                               atol=1e-9)

    # PV: O[q][dh] accumulated over 2 kblks per dh tile
    o = np.zeros((4, NL, 4))
    for c in range(4):
        for kblk in range(2):
            o[c] += mfma_16x16x32(a_frag[kblk], vt_fragment(V, c, kblk))

    # epilogue: O rows q' = g*4+reg need l of lane q'
    out = np.zeros((NQ, DH))
    for reg in range(4):
        l_row = shfl(tsum, (g << 2) + reg)
        for c in range(4):
            row = g * 4 + reg
            out[row, c * 16 + myq] = o[c, :, reg] / l_row
    assert np.allclose(out, O_ref, atol=1e-9), \
        np.abs(out - O_ref).max()
