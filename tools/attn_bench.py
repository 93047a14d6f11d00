"""Kernel-level timing of the fused attention + pcgemm ops across the
trunk's real shape classes (stable within-run numbers, independent of
full-bench box noise).  Run on an MI355X:

    PYTHONPATH=/root/repo python tools/attn_bench.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from alphafold2_amd.ops.dispatch import _load_ext
from alphafold2_amd.ops.hip_autograd import _pcg


def time_fn(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def attn_shapes(b=5, n=256, m=128, h=8, d=64):
    # (label, B, Lq, Lk, bias_repeat(None=no bias))
    return [
        ("msa_row", b * m, n, n, m),
        ("msa_col", b * n, m, m, None),
        ("tri_out", b * n, n, n, n),
        ("tri_in", b * n, n, n, n),
    ]


def main():
    ext = _load_ext()
    assert ext is not None
    h, d = 8, 64
    print(f"{'shape':8s} {'fwd_ms':>8s} {'fwd_TF':>7s} {'bwd_ms':>8s} {'bwd_TF':>7s}")
    for label, B, Lq, Lk, brep in attn_shapes():
        mk = lambda *s: torch.randn(*s, device='cuda', dtype=torch.bfloat16)
        q, k, v = mk(B, h, Lq, d), mk(B, h, Lk, d), mk(B, h, Lk, d)
        bias = mk(B // brep, h, Lq, Lk) if brep else None
        scale = d ** -0.5
        out, lse = ext.attn_fwd(q, k, v, bias, None, brep or 1, scale)
        dout = mk(B, h, Lq, d)

        fwd = lambda: ext.attn_fwd(q, k, v, bias, None, brep or 1, scale)
        bwd = lambda: ext.attn_bwd(dout, q, k, v, out, lse, bias, None,
                                   brep or 1, scale, bias is not None)
        t_f = time_fn(fwd)
        t_b = time_fn(bwd, iters=15)
        gf = 4e-12 * B * h * Lq * Lk * d
        gb = 10e-12 * B * h * Lq * Lk * d  # 5 matmul-equivalents
        print(f"{label:8s} {t_f:8.3f} {gf / (t_f / 1e3):7.1f} "
              f"{t_b:8.3f} {gb / (t_b / 1e3):7.1f}")

    # pcgemm: trimul fwd shape at b=5, n=256, d=256
    b, n, D = 5, 256, 256
    L = torch.randn(b, n, n, D, device='cuda', dtype=torch.bfloat16)
    R = torch.randn(b, n, n, D, device='cuda', dtype=torch.bfloat16)
    f = lambda: _pcg(L, R, n, n, n, 1, 2, 1, 2)
    t = time_fn(f, iters=20)
    tf = (2e-12 * b * n * n * n * D) / (t / 1e3)
    print(f"{'pcgemm':8s} {t:8.3f} {tf:7.1f}")
    # einsum comparison
    g = lambda: torch.einsum('bikd,bjkd->bijd', L, R)
    t2 = time_fn(g, iters=20)
    tf2 = (2e-12 * b * n * n * n * D) / (t2 / 1e3)
    print(f"{'einsum':8s} {t2:8.3f} {tf2:7.1f}")


if __name__ == '__main__':
    main()
