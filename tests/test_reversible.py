"""Reversible-trunk correctness: the custom backward (input
reconstruction) must produce the same gradients as plain autograd on the
identical computation."""
import copy

import torch

from alphafold2_amd.models.reversible import make_reversible_evoformer


def _run_naive(net, x, m, mask, msa_mask):
    """Same math as ReversibleEvoformer.forward but under plain autograd."""
    x1, x2, m1, m2 = x, x.clone(), m, m.clone()
    for block in net.blocks:
        m1 = m1 + block.msa_attn.net(m2, mask=msa_mask, pairwise_repr=x2)
        m2 = m2 + block.msa_ff.net(m1)
        x1 = x1 + block.pair_attn.net(x2, mask=mask, msa_repr=m2,
                                      msa_mask=msa_mask)
        x2 = x2 + block.pair_ff.net(x1)
    return 0.5 * (x1 + x2), 0.5 * (m1 + m2)


def test_reversible_matches_autograd():
    torch.manual_seed(0)
    dim, depth, n, mrows, b = 16, 2, 6, 3, 2
    net = make_reversible_evoformer(dim=dim, depth=depth, seq_len=n,
                                    heads=2, dim_head=8).double()
    net.train()

    x = torch.randn(b, n, n, dim, dtype=torch.float64, requires_grad=True)
    m = torch.randn(b, mrows, n, dim, dtype=torch.float64, requires_grad=True)
    mask = torch.ones(b, n, n).bool()
    msa_mask = torch.ones(b, mrows, n).bool()

    net2 = copy.deepcopy(net)
    x2 = x.detach().clone().requires_grad_(True)
    m2 = m.detach().clone().requires_grad_(True)

    xo, mo = net(x, m, mask=mask, msa_mask=msa_mask)
    loss = xo.pow(2).sum() + mo.pow(2).sum()
    loss.backward()

    xo2, mo2 = _run_naive(net2, x2, m2, mask, msa_mask)
    loss2 = xo2.pow(2).sum() + mo2.pow(2).sum()
    loss2.backward()

    assert torch.allclose(xo, xo2, atol=1e-9)
    assert torch.allclose(x.grad, x2.grad, atol=1e-7), \
        (x.grad - x2.grad).abs().max()
    assert torch.allclose(m.grad, m2.grad, atol=1e-7)

    for p1, p2 in zip(net.parameters(), net2.parameters()):
        assert (p1.grad is None) == (p2.grad is None)
        if p1.grad is not None:
            assert torch.allclose(p1.grad, p2.grad, atol=1e-7), \
                (p1.grad - p2.grad).abs().max()


def test_reversible_memory_constant_graph():
    """The reversible forward must not store per-block activations: the
    output's grad graph holds only the custom Function node."""
    torch.manual_seed(0)
    net = make_reversible_evoformer(dim=8, depth=3, seq_len=4, heads=1,
                                    dim_head=4)
    net.train()
    x = torch.randn(1, 4, 4, 8, requires_grad=True)
    m = torch.randn(1, 2, 4, 8, requires_grad=True)
    xo, mo = net(x, m)
    assert xo.grad_fn is not None
    (xo.sum() + mo.sum()).backward()
    assert x.grad is not None and m.grad is not None


def test_reversible_dim384_config():
    """BASELINE configs[2] shape class (dim=384, reversible): forward +
    backward at a reduced depth/length on CPU."""
    from alphafold2_amd import Alphafold2
    torch.manual_seed(0)
    model = Alphafold2(dim=384, depth=1, heads=6, dim_head=64,
                       reversible=True)
    model.train()
    seq = torch.randint(0, 21, (1, 8))
    msa = torch.randint(0, 21, (1, 2, 8))
    ret = model(seq, msa)
    (ret.distance.pow(2).mean() + ret.msa_mlm_loss).backward()
    grads = [p.grad for p in model.net.parameters() if p.grad is not None]
    assert len(grads) > 0
    assert all(torch.isfinite(g).all() for g in grads)
