"""Hot-op tests: eager semantics on CPU, and (gpu-marked) HIP-kernel
parity against the fp32 eager reference."""
import pytest
import torch

from alphafold2_amd.ops import eager


def test_attention_core_matches_naive():
    torch.manual_seed(0)
    B, h, i, j, d = 3, 2, 5, 7, 8
    q = torch.randn(B, h, i, d)
    k = torch.randn(B, h, j, d)
    v = torch.randn(B, h, j, d)
    out = eager.attention_core(q, k, v)
    # naive reference
    dots = (q * d ** -0.5) @ k.transpose(-1, -2)
    ref = dots.softmax(-1) @ v
    assert torch.allclose(out, ref, atol=1e-6)


def test_attention_core_masking():
    torch.manual_seed(0)
    B, h, n, d = 2, 2, 6, 4
    q, k, v = (torch.randn(B, h, n, d) for _ in range(3))
    mask = torch.ones(B, n).bool()
    mask[:, -2:] = False
    out = eager.attention_core(q, k, v, mask=mask)
    # masked keys must not contribute: perturbing them changes nothing
    v2 = v.clone()
    v2[:, :, -2:] = 100.
    out2 = eager.attention_core(q, k, v2, mask=mask)
    assert torch.allclose(out[:, :, :4], out2[:, :, :4], atol=1e-6)


def test_attention_core_bias():
    torch.manual_seed(0)
    B, h, n, d = 2, 2, 6, 4
    q, k, v = (torch.randn(B, h, n, d) for _ in range(3))
    bias = torch.zeros(B, h, n, n)
    bias[..., 0] = 1e9  # force all attention onto key 0
    out = eager.attention_core(q, k, v, bias=bias)
    assert torch.allclose(out, v[:, :, 0:1].expand_as(out), atol=1e-4)


def test_attention_tie_dim():
    torch.manual_seed(0)
    b, r, h, n, d = 2, 3, 2, 5, 4
    q = torch.randn(b * r, h, n, d)
    k = torch.randn(b * r, h, n, d)
    v = torch.randn(b * r, h, n, d)
    out = eager.attention_core(q, k, v, tie_dim=r)
    # manual: averaged q over rows
    qm = (q * d ** -0.5).reshape(b, r, h, n, d).mean(dim=1, keepdim=True)
    kg = k.reshape(b, r, h, n, d)
    dots = torch.einsum('bxhid,brhjd->brhij', qm, kg)
    ref = dots.softmax(-1)
    ref = torch.einsum('brhij,brhjd->brhid',
                       ref, v.reshape(b, r, h, n, d)).reshape(b * r, h, n, d)
    assert torch.allclose(out, ref, atol=1e-6)


def test_geglu():
    x = torch.randn(4, 10)
    out = eager.geglu(x)
    a, g = x.chunk(2, -1)
    assert torch.allclose(out, a * torch.nn.functional.gelu(g))


def test_outer_product_mean_unmasked():
    torch.manual_seed(0)
    b, m, n, d = 2, 4, 6, 8
    left = torch.randn(b, m, n, d)
    right = torch.randn(b, m, n, d)
    out = eager.outer_product_mean(left, right)
    ref = (left[:, :, :, None, :] * right[:, :, None, :, :]).mean(dim=1)
    assert torch.allclose(out, ref, atol=1e-5)


def test_outer_product_mean_masked_reference_numerics():
    """Masked branch must reproduce the reference normalization exactly
    (sum_m / (m * (count + eps)) — reference alphafold2.py:341-349)."""
    torch.manual_seed(0)
    b, m, n, d = 2, 4, 6, 8
    eps = 1e-5
    left = torch.randn(b, m, n, d)
    right = torch.randn(b, m, n, d)
    mask = torch.rand(b, m, n) > 0.3
    out = eager.outer_product_mean(left, right, mask=mask, eps=eps)

    outer = left[:, :, :, None, :] * right[:, :, None, :, :]
    pair_mask = (mask[:, :, :, None] & mask[:, :, None, :])[..., None]
    outer = outer.masked_fill(~pair_mask, 0.)
    ref = outer.mean(dim=1) / (pair_mask.float().sum(dim=1) + eps)
    assert torch.allclose(out, ref, atol=1e-5)


def test_triangle_mix():
    torch.manual_seed(0)
    b, n, d = 2, 5, 4
    left = torch.randn(b, n, n, d)
    right = torch.randn(b, n, n, d)
    out_g = eager.triangle_mix(left, right, 'outgoing')
    out_i = eager.triangle_mix(left, right, 'ingoing')
    ref_g = torch.einsum('bikd,bjkd->bijd', left, right)
    ref_i = torch.einsum('bkjd,bkid->bijd', left, right)
    assert torch.allclose(out_g, ref_g, atol=1e-5)
    assert torch.allclose(out_i, ref_i, atol=1e-5)


def test_pair_outer_sum():
    xl = torch.randn(2, 5, 8)
    xr = torch.randn(2, 5, 8)
    out = eager.pair_outer_sum(xl, xr)
    assert out.shape == (2, 5, 5, 8)
    assert torch.allclose(out[0, 1, 3], xl[0, 1] + xr[0, 3])


def test_distance_buckets():
    coords = torch.randn(2, 10, 3) * 4
    boundaries = torch.linspace(2, 20, 32)[:-1]
    out = eager.distance_buckets(coords, boundaries)
    ref = torch.bucketize(torch.cdist(coords, coords), boundaries)
    assert (out == ref).all()


# ---------------------------------------------------------------------------
# HIP kernel parity (MI355X only)


@pytest.mark.gpu
def test_hip_extension_loads():
    from alphafold2_amd.ops.dispatch import hip_ops_available
    assert hip_ops_available(), \
        "gfx950 extension must be importable on a GPU box"


@pytest.mark.gpu
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_hip_attention_parity(dtype):
    from alphafold2_amd.ops import dispatch
    if not dispatch.using_hip(torch.zeros(1, device='cuda'), 'attn_fwd'):
        pytest.skip('attn kernel not present in extension')
    torch.manual_seed(0)
    B, h, n, d = 4, 8, 128, 64
    q = torch.randn(B, h, n, d, device='cuda', dtype=dtype)
    k = torch.randn(B, h, n, d, device='cuda', dtype=dtype)
    v = torch.randn(B, h, n, d, device='cuda', dtype=dtype)
    bias = torch.randn(B, h, n, n, device='cuda', dtype=dtype)
    out = dispatch.attention_core(q, k, v, bias=bias)
    ref = eager.attention_core(q.float(), k.float(), v.float(),
                               bias=bias.float())
    tol = 2e-2 if dtype == torch.bfloat16 else 1e-4
    assert (out.float() - ref).abs().max().item() < tol


def test_attention_dropout_eager():
    """attn_dropout routes through the eager path and actually drops."""
    torch.manual_seed(0)
    from alphafold2_amd.models.evoformer import Attention
    m = Attention(dim=32, heads=2, dim_head=16, dropout=0.5).train()
    torch.nn.init.normal_(m.to_out.weight)  # zero-init would hide dropout
    x = torch.randn(2, 8, 32)
    o1 = m(x)
    o2 = m(x)
    assert not torch.allclose(o1, o2), 'dropout must be stochastic'
    m.eval()
    with torch.no_grad():
        o3 = m(x)
        o4 = m(x)
    assert torch.allclose(o3, o4), 'eval must be deterministic'


def test_pair_rep_build_eager_matches_composition():
    import torch
    from alphafold2_amd import ops
    torch.manual_seed(0)
    b, n, d, V = 2, 12, 16, 9
    left = torch.randn(b, n, d)
    right = torch.randn(b, n, d)
    emb = torch.randn(V, d)
    rel = torch.randint(0, V, (1, n, n))
    out = ops.pair_rep_build(left, right, emb, rel)
    ref = left[:, :, None, :] + right[:, None, :, :] \
        + torch.nn.functional.embedding(rel, emb)
    assert torch.allclose(out, ref, atol=1e-6)


def test_tri_proj_gates_eager_path():
    import torch
    from alphafold2_amd import ops
    torch.manual_seed(1)
    b, n, h = 1, 8, 16
    fused = torch.randn(b, n, n, 5 * h)
    mask = torch.rand(b, n, n) > 0.3
    left, right, og = ops.tri_proj_gates(fused, h, row_mask=mask)
    l0, r0, lg, rg, og0 = fused.split([h] * 5, dim=-1)
    m = mask.unsqueeze(-1).float()
    assert torch.allclose(left, l0 * torch.sigmoid(lg) * m, atol=1e-6)
    assert torch.allclose(right, r0 * torch.sigmoid(rg) * m, atol=1e-6)
    assert torch.equal(og, og0)


def test_fused_linear_eager_residual():
    import torch
    from alphafold2_amd import ops
    torch.manual_seed(2)
    x = torch.randn(6, 16)
    w = torch.randn(8, 16)
    b = torch.randn(8)
    r = torch.randn(6, 8)
    out = ops.fused_linear(x, w, b, residual=r)
    assert torch.allclose(out, x @ w.t() + b + r, atol=1e-5)


def test_ff1_geglu_eager_matches_reference_math():
    import torch
    from alphafold2_amd import ops
    torch.manual_seed(3)
    x = torch.randn(5, 16)
    w = torch.randn(64, 16)
    b = torch.randn(64)
    out = ops.ff1_geglu(x, w, b)
    i = x @ w.t() + b
    a, g = i.chunk(2, dim=-1)
    assert torch.allclose(out, a * torch.nn.functional.gelu(g), atol=1e-5)
