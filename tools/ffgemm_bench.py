"""Within-run A/B of the fused linear GEMM (ffgemm.hip) vs hipBLASLt on
the production FF/projection shapes (flagship config dim=256, crop 256,
msa 128, batch 5).  Run on a GPU box:

    python tools/ffgemm_bench.py
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from alphafold2_amd.ops.dispatch import _load_ext


def time_fn(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(iters):
        fn()
    t1.record()
    t1.synchronize()
    return t0.elapsed_time(t1) / iters


def main():
    ext = _load_ext()
    assert ext is not None
    torch.manual_seed(0)
    dev = 'cuda'
    dt = torch.bfloat16

    # (name, M, K, N, epi)
    shapes = [
        ('msa_ff1_geglu', 5 * 128 * 256, 256, 4096, 'geglu'),
        ('pair_ff1_geglu', 5 * 256 * 256, 256, 4096, 'geglu'),
        ('msa_ff2_resid', 5 * 128 * 256, 1024, 256, 'resid'),
        ('pair_ff2_resid', 5 * 256 * 256, 1024, 256, 'resid'),
        ('attn_qkvg', 5 * 256 * 256, 256, 2048, 'none'),
        ('attn_out', 5 * 256 * 256, 512, 256, 'none'),
        ('trimul_proj', 5 * 256 * 256, 256, 640, 'none'),
        ('dgrad_ff1', 5 * 128 * 256, 2048, 256, 'none'),
    ]
    for name, M, K, N, epi in shapes:
        x = torch.randn(M, K, device=dev, dtype=dt)
        w = torch.randn(N, K, device=dev, dtype=dt) * 0.1
        b = torch.randn(N, device=dev, dtype=dt)
        r = torch.randn(M, N, device=dev, dtype=dt) if epi == 'resid' \
            else None
        fl = 2.0 * M * K * N

        t_blas = time_fn(lambda: torch.nn.functional.linear(x, w, b))
        if epi == 'geglu':
            t_pad = time_fn(lambda: ext.ff1_geglu_fwd(x, w, b, 0))
            t_glds = time_fn(lambda: ext.ff1_geglu_fwd(x, w, b, 1))
            # reference composition cost: linear + geglu
            t_ref2 = time_fn(lambda: ext.geglu_fwd(
                torch.nn.functional.linear(x, w, b)))
            extra = f'  blas+geglu={t_ref2:.3f}ms'
        elif epi == 'resid':
            t_pad = time_fn(lambda: ext.linear_fwd(x, w, b, r, 0))
            t_glds = time_fn(lambda: ext.linear_fwd(x, w, b, r, 1))
            t_ref2 = time_fn(
                lambda: torch.nn.functional.linear(x, w, b) + r)
            extra = f'  blas+add={t_ref2:.3f}ms'
        else:
            t_pad = time_fn(lambda: ext.linear_fwd(x, w, b, None, 0))
            t_glds = time_fn(lambda: ext.linear_fwd(x, w, b, None, 1))
            extra = ''
        print(f'{name:16s} M={M:7d} K={K:4d} N={N:4d}  '
              f'blas={t_blas:.3f}ms ({fl / t_blas / 1e9:.0f} TF)  '
              f'pad={t_pad:.3f}ms ({fl / t_pad / 1e9:.0f} TF)  '
              f'glds={t_glds:.3f}ms ({fl / t_glds / 1e9:.0f} TF)'
              f'{extra}', flush=True)

    print('--- wgrad (dW = dY^T @ X, split-K) ---', flush=True)
    for name, K, M, N in [
        ('ff1_wgrad_msa', 5 * 128 * 256, 2048, 256),
        ('ff1_wgrad_pair', 5 * 256 * 256, 2048, 256),
        ('ff2_wgrad_pair', 5 * 256 * 256, 256, 1024),
        ('qkvg_wgrad', 5 * 256 * 256, 2048, 256),
        ('out_wgrad', 5 * 256 * 256, 256, 512),
        ('trimul_out_wgrad', 5 * 256 * 256, 256, 256),
        ('gating_wgrad', 5 * 256 * 256, 512, 256),
    ]:
        dy = torch.randn(K, M, device=dev, dtype=dt) * 0.1
        x = torch.randn(K, N, device=dev, dtype=dt) * 0.1
        fl = 2.0 * K * M * N
        t_blas = time_fn(lambda: dy.t() @ x)
        t_swap = time_fn(lambda: (x.t() @ dy).t().contiguous())
        t_swapv = time_fn(lambda: x.t() @ dy)   # consumer-transposed view
        t_mine = time_fn(lambda: ext.wgrad(dy, x))
        print(f'{name:16s} K={K:7d} M={M:4d} N={N:4d}  '
              f'blas={t_blas:.3f}ms ({fl / t_blas / 1e9:.0f} TF)  '
              f'swap={t_swap:.3f}ms  swapv={t_swapv:.3f}ms '
              f'({fl / t_swapv / 1e9:.0f} TF)  '
              f'mine={t_mine:.3f}ms ({fl / t_mine / 1e9:.0f} TF)',
              flush=True)


    print('--- geglu (fused activation kernels) ---', flush=True)
    for name, rows, H in [('geglu_fwd_ff1', 5 * 256 * 256, 2048),
                          ('geglu_bwd_ff1', 5 * 256 * 256, 2048),
                          ('geglu_bwd_h1024', 5 * 256 * 256, 1024)]:
        xg = torch.randn(rows, 2 * H, device=dev, dtype=dt)
        dyg = torch.randn(rows, H, device=dev, dtype=dt)
        gb = (rows * H * 3 + rows * H) * 2 / 1e9  # ~GB moved
        if 'fwd' in name:
            t = time_fn(lambda: ext.geglu_fwd(xg))
        else:
            t = time_fn(lambda: ext.geglu_bwd(dyg, xg))
        print(f'{name:16s} rows={rows:7d} H={H:5d}  {t:.3f}ms '
              f'(~{gb / t:.1f} TB/s)', flush=True)


    # transcendental-cost probes: same element counts, known traffic
    xp = torch.randn(327680, 2048, device=dev, dtype=dt)
    t_gelu = time_fn(lambda: torch.nn.functional.gelu(xp))
    t_gelut = time_fn(lambda: torch.nn.functional.gelu(xp, approximate='tanh'))
    t_sig = time_fn(lambda: torch.sigmoid(xp))
    t_add = time_fn(lambda: xp + xp)
    gbp = 327680 * 2048 * 2 * 2 / 1e9
    print(f'probe: torch gelu(erf)={t_gelu:.3f}ms ({gbp/t_gelu:.1f} TB/s)  '
          f'gelu(tanh)={t_gelut:.3f}ms  sigmoid={t_sig:.3f}ms  '
          f'add={t_add:.3f}ms', flush=True)


if __name__ == '__main__':
    main()
