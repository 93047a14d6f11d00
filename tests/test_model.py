"""Model-level tests mirroring the reference suite
(/root/reference/tests/test_attention.py) plus state-dict layout checks."""
import torch

from alphafold2_amd import Alphafold2


def tiny_model(**kwargs):
    defaults = dict(dim=32, depth=2, heads=2, dim_head=32)
    defaults.update(kwargs)
    return Alphafold2(**defaults)


def test_main():
    model = tiny_model()
    seq = torch.randint(0, 21, (2, 64))
    msa = torch.randint(0, 21, (2, 5, 64))
    mask = torch.ones_like(seq).bool()
    msa_mask = torch.ones_like(msa).bool()
    ret = model(seq, msa, mask=mask, msa_mask=msa_mask)
    assert ret.distance.shape == (2, 64, 64, 37)


def test_no_msa():
    model = tiny_model()
    seq = torch.randint(0, 21, (2, 64))
    mask = torch.ones_like(seq).bool()
    ret = model(seq, mask=mask)
    assert ret.distance.shape == (2, 64, 64, 37)


def test_anglegrams():
    model = tiny_model(predict_angles=True)
    seq = torch.randint(0, 21, (2, 32))
    msa = torch.randint(0, 21, (2, 5, 32))
    mask = torch.ones_like(seq).bool()
    msa_mask = torch.ones_like(msa).bool()
    ret = model(seq, msa, mask=mask, msa_mask=msa_mask)
    assert ret.theta_logits.shape == (2, 32, 32, 25)
    assert ret.phi_logits.shape == (2, 32, 32, 13)
    assert ret.omega_logits.shape == (2, 32, 32, 25)


def test_symmetrize_omega():
    """symmetrize_omega routes the omega head through the symmetrized
    pair rep: logits must be transpose-symmetric."""
    model = tiny_model(predict_angles=True, symmetrize_omega=True).eval()
    seq = torch.randint(0, 21, (1, 16))
    msa = torch.randint(0, 21, (1, 3, 16))
    with torch.no_grad():
        ret = model(seq, msa)
    assert torch.allclose(ret.omega_logits,
                          ret.omega_logits.transpose(1, 2), atol=1e-5)
    # theta stays on the raw (asymmetric) pair rep
    assert not torch.allclose(ret.theta_logits,
                              ret.theta_logits.transpose(1, 2), atol=1e-3)


def test_templates():
    model = tiny_model(templates_dim=32, templates_angles_feats_dim=32)
    seq = torch.randint(0, 21, (2, 16))
    mask = torch.ones_like(seq).bool()
    msa = torch.randint(0, 21, (2, 5, 16))
    msa_mask = torch.ones_like(msa).bool()
    templates_feats = torch.randn(2, 3, 16, 16, 32)
    templates_angles = torch.randn(2, 3, 16, 32)
    templates_mask = torch.ones(2, 3, 16).bool()
    ret = model(seq, msa, mask=mask, msa_mask=msa_mask,
                templates_feats=templates_feats,
                templates_angles=templates_angles,
                templates_mask=templates_mask)
    assert ret.distance.shape == (2, 16, 16, 37)


def test_extra_msa():
    model = tiny_model(dim=64, predict_coords=True)
    seq = torch.randint(0, 21, (2, 4))
    mask = torch.ones_like(seq).bool()
    msa = torch.randint(0, 21, (2, 5, 4))
    msa_mask = torch.ones_like(msa).bool()
    extra_msa = torch.randint(0, 21, (2, 5, 4))
    extra_msa_mask = torch.ones_like(extra_msa).bool()
    coords = model(seq, msa, mask=mask, msa_mask=msa_mask,
                   extra_msa=extra_msa, extra_msa_mask=extra_msa_mask)
    assert coords.shape == (2, 4, 3)


def test_embeddings():
    model = tiny_model()
    seq = torch.randint(0, 21, (2, 16))
    mask = torch.ones_like(seq).bool()
    embedds = torch.randn(2, 1, 16, 1280)

    ret = model(seq, mask=mask, embedds=embedds, msa_mask=None)
    assert ret.distance.shape == (2, 16, 16, 37)

    embedds_mask = torch.ones_like(embedds[..., -1]).bool()
    ret = model(seq, mask=mask, embedds=embedds, msa_mask=embedds_mask)
    assert ret.distance.shape == (2, 16, 16, 37)


def test_coords():
    model = tiny_model(predict_coords=True, structure_module_depth=1,
                       structure_module_heads=1, structure_module_dim_head=1)
    seq = torch.randint(0, 21, (2, 16))
    mask = torch.ones_like(seq).bool()
    msa = torch.randint(0, 21, (2, 5, 16))
    msa_mask = torch.ones_like(msa).bool()
    coords = model(seq, msa, mask=mask, msa_mask=msa_mask)
    assert coords.shape == (2, 16, 3), 'must output coordinates'


def test_coords_backwards():
    model = tiny_model(dim=64, predict_coords=True,
                       structure_module_depth=1, structure_module_heads=1,
                       structure_module_dim_head=1)
    seq = torch.randint(0, 21, (2, 16))
    mask = torch.ones_like(seq).bool()
    msa = torch.randint(0, 21, (2, 5, 16))
    msa_mask = torch.ones_like(msa).bool()
    coords = model(seq, msa, mask=mask, msa_mask=msa_mask)
    coords.sum().backward()
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert len(grads) > 0, 'gradients must flow back through the structure module'


def test_confidence():
    model = tiny_model(dim=64, depth=1, predict_coords=True)
    seq = torch.randint(0, 21, (2, 16))
    mask = torch.ones_like(seq).bool()
    msa = torch.randint(0, 21, (2, 5, 16))
    msa_mask = torch.ones_like(msa).bool()
    coords, confidences = model(seq, msa, mask=mask, msa_mask=msa_mask,
                                return_confidence=True)
    assert coords.shape[:-1] == confidences.shape[:-1]


def test_recycling():
    model = tiny_model(dim=64, predict_coords=True)
    seq = torch.randint(0, 21, (2, 4))
    mask = torch.ones_like(seq).bool()
    msa = torch.randint(0, 21, (2, 5, 4))
    msa_mask = torch.ones_like(msa).bool()
    extra_msa = torch.randint(0, 21, (2, 5, 4))
    extra_msa_mask = torch.ones_like(extra_msa).bool()

    coords, ret = model(seq, msa, mask=mask, msa_mask=msa_mask,
                        extra_msa=extra_msa, extra_msa_mask=extra_msa_mask,
                        return_aux_logits=True, return_recyclables=True)
    coords, ret = model(seq, msa, mask=mask, msa_mask=msa_mask,
                        extra_msa=extra_msa, extra_msa_mask=extra_msa_mask,
                        recyclables=ret.recyclables,
                        return_aux_logits=True, return_recyclables=True)
    assert coords.shape == (2, 4, 3)


def test_mlm_loss_during_training():
    model = tiny_model()
    model.train()
    seq = torch.randint(0, 21, (2, 32))
    msa = torch.randint(0, 21, (2, 5, 32))
    mask = torch.ones_like(seq).bool()
    msa_mask = torch.ones_like(msa).bool()
    ret = model(seq, msa, mask=mask, msa_mask=msa_mask)
    assert ret.msa_mlm_loss is not None
    assert ret.msa_mlm_loss.requires_grad


def test_state_dict_layout():
    """Checkpoint layout parity with the reference model
    (reference alphafold2.py:469-628 module attribute map)."""
    model = tiny_model(predict_angles=True, predict_coords=True)
    keys = set(model.state_dict().keys())

    expected = [
        'token_emb.weight',
        'to_pairwise_repr.weight',
        'pos_emb.weight',
        'to_template_embed.weight',
        'template_angle_mlp.0.weight',
        'template_angle_mlp.2.weight',
        'to_prob_theta.weight',
        'to_prob_phi.weight',
        'to_prob_omega.weight',
        'embedd_project.weight',
        'mlm.to_logits.weight',
        'to_distogram_logits.0.weight',
        'to_distogram_logits.1.weight',
        'msa_to_single_repr_dim.weight',
        'trunk_to_pairwise_repr_dim.weight',
        'ipa_block.attn.to_scalar_q.weight',
        'ipa_block.attn.point_weights',
        'ipa_block.attn.to_out.weight',
        'ipa_block.ff.0.weight',
        'ipa_block.ff.2.weight',
        'ipa_block.ff.4.weight',
        'to_quaternion_update.weight',
        'to_points.weight',
        'lddt_linear.weight',
        'recycling_msa_norm.weight',
        'recycling_pairwise_norm.weight',
        'recycling_distance_embed.weight',
        # trunk block 0 internals
        'net.layers.0.layer.0.outer_mean.left_proj.weight',
        'net.layers.0.layer.0.triangle_multiply_outgoing.left_proj.weight',
        'net.layers.0.layer.0.triangle_attention_outgoing.attn.to_q.weight',
        'net.layers.0.layer.0.triangle_attention_outgoing.edges_to_attn_bias.0.weight',
        'net.layers.0.layer.1.net.0.weight',
        'net.layers.0.layer.1.net.3.weight',
        'net.layers.0.layer.2.row_attn.attn.gating.weight',
        'net.layers.0.layer.3.norm.weight',
        'extra_msa_evoformer.layers.0.layer.0.outer_mean.norm.weight',
        'template_pairwise_embedder.triangle_multiply_ingoing.out_gate.weight',
        'template_pointwise_attn.to_kv.weight',
    ]
    missing = [k for k in expected if k not in keys]
    assert not missing, f'missing reference-layout keys: {missing}'


def test_reversible_trunk():
    model = tiny_model(dim=32, depth=2, reversible=True)
    model.train()
    seq = torch.randint(0, 21, (2, 16))
    msa = torch.randint(0, 21, (2, 3, 16))
    mask = torch.ones_like(seq).bool()
    msa_mask = torch.ones_like(msa).bool()
    ret = model(seq, msa, mask=mask, msa_mask=msa_mask)
    loss = ret.distance.sum() + ret.msa_mlm_loss
    loss.backward()
    grads = [p.grad for p in model.net.parameters() if p.grad is not None]
    assert len(grads) > 0


def test_egnn_structure_module():
    model = tiny_model(predict_coords=True, structure_module_type='egnn',
                       structure_module_depth=2)
    seq = torch.randint(0, 21, (2, 12))
    mask = torch.ones_like(seq).bool()
    msa = torch.randint(0, 21, (2, 3, 12))
    msa_mask = torch.ones_like(msa).bool()
    coords = model(seq, msa, mask=mask, msa_mask=msa_mask)
    assert coords.shape == (2, 12, 3)
    coords.sum().backward()


def test_se3_structure_module_equivariance():
    """Rotating/translating the recycled input coords must rotate the
    output coords identically (SE(3) equivariance of the refiner)."""
    from alphafold2_amd.models.equivariant import EquivariantStructureModule
    torch.manual_seed(0)
    mod = EquivariantStructureModule(dim=16, depth=2, kind='se3').double()
    h = torch.randn(1, 10, 16, dtype=torch.float64)
    pair = torch.randn(1, 10, 10, 16, dtype=torch.float64)
    x0 = torch.randn(1, 10, 3, dtype=torch.float64)

    theta = torch.tensor(0.6, dtype=torch.float64)
    R = torch.tensor([[torch.cos(theta), -torch.sin(theta), 0],
                      [torch.sin(theta), torch.cos(theta), 0],
                      [0., 0., 1.]], dtype=torch.float64)
    t = torch.tensor([1., -2., 3.], dtype=torch.float64)

    _, out1 = mod(h, pair, coords=x0)
    _, out2 = mod(h, pair, coords=x0 @ R.T + t)
    assert torch.allclose(out2, out1 @ R.T + t, atol=1e-8), \
        (out2 - (out1 @ R.T + t)).abs().max()


def test_egnn_equivariance():
    from alphafold2_amd.models.equivariant import EquivariantStructureModule
    torch.manual_seed(0)
    mod = EquivariantStructureModule(dim=16, depth=2, kind='egnn').double()
    h = torch.randn(1, 8, 16, dtype=torch.float64)
    pair = torch.randn(1, 8, 8, 16, dtype=torch.float64)
    x0 = torch.randn(1, 8, 3, dtype=torch.float64)
    theta = torch.tensor(1.1, dtype=torch.float64)
    R = torch.tensor([[1., 0., 0.],
                      [0., torch.cos(theta), -torch.sin(theta)],
                      [0., torch.sin(theta), torch.cos(theta)]],
                     dtype=torch.float64)
    t = torch.tensor([-1., 0.5, 2.], dtype=torch.float64)
    _, out1 = mod(h, pair, coords=x0)
    _, out2 = mod(h, pair, coords=x0 @ R.T + t)
    assert torch.allclose(out2, out1 @ R.T + t, atol=1e-8)


def test_refinement_iters_alias():
    m = tiny_model(predict_coords=True, structure_module_refinement_iters=2)
    assert m.structure_module_depth == 2


def test_quaternion_properties():
    from alphafold2_amd.models.quaternion import (
        quaternion_multiply, quaternion_to_matrix, matrix_to_quaternion)
    torch.manual_seed(0)
    q = torch.randn(32, 4, dtype=torch.float64)
    q = q / q.norm(dim=-1, keepdim=True)
    R = quaternion_to_matrix(q)
    # orthonormal, det +1
    eye = torch.eye(3, dtype=torch.float64).expand(32, 3, 3)
    assert torch.allclose(R @ R.transpose(-1, -2), eye, atol=1e-10)
    assert torch.allclose(torch.linalg.det(R), torch.ones(32, dtype=torch.float64), atol=1e-10)
    # multiply <-> matrix product consistency
    q2 = torch.randn(32, 4, dtype=torch.float64)
    q2 = q2 / q2.norm(dim=-1, keepdim=True)
    R12 = quaternion_to_matrix(quaternion_multiply(q, q2))
    # row-vector convention: rotate(v, q1*q2) == rotate(rotate(v, q2), q1)
    v = torch.randn(32, 3, dtype=torch.float64)
    a = torch.einsum('bc,bcr->br', v, R12)
    b = torch.einsum('bc,bcr->br', torch.einsum('bc,bcr->br', v, quaternion_to_matrix(q2)), quaternion_to_matrix(q))
    ok1 = torch.allclose(a, b, atol=1e-9)
    b2 = torch.einsum('bc,bcr->br', torch.einsum('bc,bcr->br', v, quaternion_to_matrix(q)), quaternion_to_matrix(q2))
    ok2 = torch.allclose(a, b2, atol=1e-9)
    assert ok1 or ok2
    # matrix -> quaternion roundtrip (up to sign, standardized)
    q3 = matrix_to_quaternion(R)
    assert torch.allclose(quaternion_to_matrix(q3), R, atol=1e-6)


def test_module_alias_imports():
    import alphafold2_amd.embeds
    import alphafold2_amd.reversible
    import alphafold2_amd.rotary
    assert hasattr(alphafold2_amd.embeds, 'ESMEmbedWrapper')
    assert hasattr(alphafold2_amd.reversible, 'ReversibleEvoformer')
    assert hasattr(alphafold2_amd.rotary, 'FixedPositionalEmbedding')


def test_overfit_single_batch_cpu():
    """End-to-end training sanity: the model must overfit one synthetic
    batch (loss decreases substantially) — exercises every backward."""
    from alphafold2_amd.data import synthetic_batch
    from alphafold2_amd.utils import get_bucketed_distance_matrix
    torch.manual_seed(0)
    model = Alphafold2(dim=32, depth=2, heads=2, dim_head=16,
                       checkpoint_blocks=False).train()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    b = synthetic_batch(1, 24, 4, seed=0)
    tgt = get_bucketed_distance_matrix(b['coords'], b['mask'])

    losses = []
    for _ in range(25):
        opt.zero_grad()
        ret = model(b['seq'], b['msa'], mask=b['mask'], msa_mask=b['msa_mask'])
        loss = torch.nn.functional.cross_entropy(
            ret.distance.permute(0, 3, 1, 2), tgt, ignore_index=-100)
        loss.backward()
        opt.step()
        losses.append(float(loss.detach()))
    assert losses[-1] < losses[0] * 0.7, losses


def test_seq_index_custom_numbering():
    model = tiny_model().eval()
    seq = torch.randint(0, 21, (1, 16))
    msa = torch.randint(0, 21, (1, 2, 16))
    with torch.no_grad():
        r1 = model(seq, msa)
        # chain-broken numbering (gap of 100 between residues 8 and 9)
        idx = torch.cat([torch.arange(8), torch.arange(108, 116)])
        r2 = model(seq, msa, seq_index=idx)
    assert not torch.allclose(r1.distance, r2.distance)


def test_disable_token_embed():
    model = tiny_model(disable_token_embed=True).eval()
    seq = torch.randint(0, 21, (1, 12))
    msa = torch.randint(0, 21, (1, 2, 12))
    seq_embed = torch.randn(1, 12, 32)
    msa_embed = torch.randn(1, 2, 12, 32)
    with torch.no_grad():
        ret = model(seq, msa, seq_embed=seq_embed, msa_embed=msa_embed)
    assert ret.distance.shape == (1, 12, 12, 37)
    import pytest
    with pytest.raises(AssertionError):
        model(seq, msa)  # embeddings are mandatory in this mode


def test_return_trunk_skips_structure():
    model = tiny_model(predict_coords=True, structure_module_depth=1)
    model.eval()
    seq = torch.randint(0, 21, (1, 8))
    msa = torch.randint(0, 21, (1, 2, 8))
    with torch.no_grad():
        ret = model(seq, msa, return_trunk=True)
    assert ret.distance is not None  # trunk output, no coords


def test_rotary_embeddings_functional():
    from alphafold2_amd.models.rotary import (
        AxialRotaryEmbedding, DepthWiseConv1d, FixedPositionalEmbedding,
        apply_rotary_pos_emb)
    fp = FixedPositionalEmbedding(dim=16)
    sin, cos = fp(10, torch.device('cpu'))
    assert sin.shape == (10, 16)
    q = torch.randn(2, 10, 16)
    k = torch.randn(2, 10, 16)
    q2, k2 = apply_rotary_pos_emb(q, k, (sin, cos))
    assert q2.shape == q.shape
    # rotary preserves norms per position
    assert torch.allclose(q2.norm(dim=-1), q.norm(dim=-1), atol=1e-5)

    ax = AxialRotaryEmbedding(dim=16)
    sin2, cos2 = ax(6, torch.device('cpu'))
    assert sin2.shape[0] == 36

    conv = DepthWiseConv1d(8, 16, kernel_size=3, padding=1)
    y = conv(torch.randn(2, 8, 12))
    assert y.shape == (2, 16, 12)


def test_selective_ff_checkpointing_gradient_exact():
    """checkpoint_blocks='ff' (recompute only the FF transitions) must
    give bit-identical gradients to no checkpointing."""
    torch.manual_seed(0)
    ref = tiny_model(checkpoint_blocks=False)
    torch.manual_seed(0)
    sel = tiny_model(checkpoint_blocks='ff')
    sel.load_state_dict(ref.state_dict())

    seq = torch.randint(0, 21, (1, 16))
    msa = torch.randint(0, 21, (1, 3, 16))

    for model in (ref, sel):
        model.train()
        torch.manual_seed(7)  # identical MLM corruption
        ret = model(seq, msa)
        (ret.distance.pow(2).mean() + ret.msa_mlm_loss).backward()

    for (n1, p1), (n2, p2) in zip(ref.named_parameters(),
                                  sel.named_parameters()):
        assert n1 == n2
        if p1.grad is None:
            assert p2.grad is None
            continue
        assert torch.equal(p1.grad, p2.grad), n1


def test_forward_kwarg_combinations():
    """Combinatorial smoke over optional-input interactions: every
    combination of {msa|embedds} x templates x angles x coords x mask
    must run (the reference crashes on several of these)."""
    import itertools
    model = tiny_model(predict_angles=True, predict_coords=True,
                       templates_dim=16, templates_angles_feats_dim=16,
                       structure_module_depth=1).eval()
    n, t = 12, 2
    seq = torch.randint(0, 21, (1, n))
    msa = torch.randint(0, 21, (1, 3, n))
    embedds = torch.randn(1, 1, n, 1280)
    tf = torch.randn(1, t, n, n, 16)
    ta = torch.randn(1, t, n, 16)
    tm = torch.ones(1, t, n).bool()

    for (use_embedds, use_templates, use_angles_feats, use_mask,
         return_trunk) in itertools.product([False, True], repeat=5):
        kwargs = {}
        if use_embedds:
            args = (seq, None)
            kwargs['embedds'] = embedds
        else:
            args = (seq, msa)
        if use_mask:
            kwargs['mask'] = torch.ones(1, n).bool()
        if use_templates:
            kwargs['templates_feats'] = tf
            kwargs['templates_mask'] = tm
            if use_angles_feats:
                kwargs['templates_angles'] = ta
        elif use_angles_feats:
            continue  # angles require templates
        kwargs['return_trunk'] = return_trunk
        with torch.no_grad():
            out = model(*args, **kwargs)
        if return_trunk:
            assert out.distance.shape == (1, n, n, 37)
        else:
            assert out.shape == (1, n, 3)
