"""Localize the fused-attention bias-path mismatch."""
import torch

from alphafold2_amd.ops import eager
from alphafold2_amd.ops.hip_autograd import hip_attention_core


def err(a, b):
    return (a.float() - b.float()).abs().max().item()


def run(B=1, h=1, Lq=64, Lk=64, bias_mode='none', mask=None, seed=0):
    torch.manual_seed(seed)
    mk = lambda *s: torch.randn(*s, device='cuda', dtype=torch.bfloat16)
    q, k, v = mk(B, h, Lq, 64), mk(B, h, Lk, 64), mk(B, h, Lk, 64)
    bias = None
    if bias_mode == 'randn':
        bias = mk(B, h, Lq, Lk)
    elif bias_mode == 'zeros':
        bias = torch.zeros(B, h, Lq, Lk, device='cuda', dtype=torch.bfloat16)
    elif bias_mode == 'colramp':
        bias = torch.arange(Lk, device='cuda', dtype=torch.float32)[None, None, None, :] \
            .expand(B, h, Lq, Lk).bfloat16().contiguous() * 0.1
    elif bias_mode == 'rowramp':
        bias = torch.arange(Lq, device='cuda', dtype=torch.float32)[None, None, :, None] \
            .expand(B, h, Lq, Lk).bfloat16().contiguous() * 0.1

    out = hip_attention_core(q, k, v, bias=bias, context_mask=mask)
    qmask = torch.ones(B, Lq, device='cuda').bool()
    ref = eager.attention_core(q.float(), k.float(), v.float(),
                               bias=bias.float() if bias is not None else None,
                               mask=qmask if mask is not None else None,
                               context_mask=mask)
    e = err(out, ref)
    print(f"B{B} h{h} Lq{Lq} Lk{Lk} bias={bias_mode} mask={mask is not None}: "
          f"err={e:.4f}")
    if e > 3e-2:
        em = (out.float() - ref.float()).abs().amax(dim=-1)[0, 0]  # (Lq,)
        bad_rows = (em > 3e-2).nonzero().view(-1).tolist()
        print("   bad q rows:", bad_rows[:20], "..." if len(bad_rows) > 20 else "")
        # per-d error of first bad row
        if bad_rows:
            r = bad_rows[0]
            ed = (out.float() - ref.float()).abs()[0, 0, r]
            print(f"   row {r} err by d (first 16): "
                  f"{[round(x, 3) for x in ed[:16].tolist()]}")
    return e


if __name__ == '__main__':
    run(bias_mode='none')
    run(bias_mode='zeros')
    run(bias_mode='colramp')
    run(bias_mode='rowramp')
    run(bias_mode='randn')
    run(B=2, h=2, Lq=128, Lk=128, bias_mode='randn')
    m = torch.ones(1, 64, device='cuda').bool()
    m[:, 40:] = False
    run(mask=m)
    run(bias_mode='randn', mask=m)
