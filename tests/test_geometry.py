"""Geometry-toolbox tests mirroring the reference suite
(/root/reference/tests/test_utils.py) plus numerical-correctness checks
the reference lacked."""
import numpy as np
import torch

from alphafold2_amd.utils import (
    GDT, Kabsch, MDScaling, RMSD, TMscore, center_distogram_torch,
    distmat_loss_torch, get_bucketed_distance_matrix, lddt_ca_torch,
    mat_input_to_masked, scn_atom_embedd, scn_backbone_mask, scn_cloud_mask,
    sidechain_container,
)


def test_mat_to_masked():
    x = torch.ones(19, 3)
    x_mask = torch.randn(19) > -0.3
    edges_mat = torch.randn(19, 19) < 1
    edges = torch.nonzero(edges_mat, as_tuple=False).t()

    mat_input_to_masked(x, x_mask, edges=edges)
    mat_input_to_masked(x, x_mask, edges_mat=edges_mat)

    x_ = torch.stack([x] * 2, dim=0)
    x_mask_ = torch.stack([x_mask] * 2, dim=0)
    edges_mat_ = torch.stack([edges_mat] * 2, dim=0)
    mat_input_to_masked(x_, x_mask_, edges_mat=edges_mat_)


def test_bucketed_distance_matrix():
    coords = torch.randn(2, 16, 3) * 5
    mask = torch.ones(2, 16).bool()
    mask[0, -3:] = False
    buckets = get_bucketed_distance_matrix(coords, mask)
    assert buckets.shape == (2, 16, 16)
    assert (buckets[0, -3:, :] == -100).all()
    valid = buckets[buckets != -100]
    assert valid.min() >= 0 and valid.max() <= 36
    # diagonal distance 0 -> bucket 0
    assert (buckets[1].diagonal() == 0).all()


def test_center_distogram_median():
    distogram = torch.randn(1, 64, 64, 37)
    distances, weights = center_distogram_torch(distogram, center='median')
    assert distances.shape == (1, 64, 64)
    assert weights.shape == (1, 64, 64)


def test_masks():
    seqs = torch.randint(20, size=(2, 50))
    cloud_masks = scn_cloud_mask(seqs, boolean=True)
    assert cloud_masks.shape == (2, 50, 14)
    # backbone always occupied for non-pad residues
    assert cloud_masks[..., :4].all()
    N_mask, CA_mask, C_mask = scn_backbone_mask(seqs, boolean=True)
    assert N_mask.sum() == 2 * 50
    atom_tokens = scn_atom_embedd(seqs)
    assert atom_tokens.shape == (2, 50, 14)


def test_mds_and_mirrors():
    distogram = torch.randn(2, 32 * 3, 32 * 3, 37)
    distances, weights = center_distogram_torch(distogram)
    paddings = [7, 0]
    for i, pad in enumerate(paddings):
        if pad > 0:
            weights[i, -pad:, -pad:] = 0.

    masker = torch.arange(distogram.shape[1]) % 3
    N_mask = (masker == 0).bool()
    CA_mask = (masker == 1).bool()
    coords_3d, _ = MDScaling(distances, weights=weights, iters=5,
                             fix_mirror=True, N_mask=N_mask, CA_mask=CA_mask,
                             C_mask=None)
    assert list(coords_3d.shape) == [2, 3, 32 * 3]


def test_mds_recovers_structure():
    """MDS on an exact distance matrix must reproduce the geometry."""
    torch.manual_seed(0)
    pts = torch.randn(1, 24, 3)
    dist = torch.cdist(pts, pts)
    coords, _ = MDScaling(dist, iters=50, fix_mirror=False)
    d2 = torch.cdist(coords.transpose(-1, -2), coords.transpose(-1, -2))
    assert (d2 - dist).abs().mean() < 0.15


def test_sidechain_container():
    seqs = torch.tensor([[0] * 137, [3] * 137]).long()
    bb = torch.randn(2, 137 * 4, 3)
    atom_mask = torch.tensor([1] * 4 + [0] * (14 - 4))
    proto_3d = sidechain_container(seqs, bb, atom_mask=atom_mask)
    assert list(proto_3d.shape) == [2, 137, 14, 3]
    assert torch.isfinite(proto_3d).all()


def test_sidechain_container_differentiable():
    seqs = torch.tensor([[4] * 8]).long()
    bb = torch.randn(1, 8 * 4, 3, requires_grad=True)
    atom_mask = torch.tensor([1] * 4 + [0] * 10)
    out = sidechain_container(seqs, bb, atom_mask=atom_mask)
    out.sum().backward()
    assert bb.grad is not None
    assert torch.isfinite(bb.grad).all()


def test_sidechain_geometry_sane():
    """Built CB must sit ~1.52 Å from CA."""
    seqs = torch.tensor([[0] * 4]).long()  # poly-alanine
    # idealized straight backbone
    n = torch.tensor([0., 0., 0.])
    ca = torch.tensor([1.46, 0., 0.])
    c = torch.tensor([2.0, 1.42, 0.])
    o = torch.tensor([1.6, 2.5, 0.])
    res = torch.stack([n, ca, c, o])
    bb = torch.cat([res + i * torch.tensor([3.8, 0., 0.]) for i in range(4)])
    bb = bb.unsqueeze(0)
    atom_mask = torch.tensor([1] * 4 + [0] * 10)
    out = sidechain_container(seqs, bb, atom_mask=atom_mask)
    cb = out[0, :, 4]
    ca_all = out[0, :, 1]
    d = (cb - ca_all).norm(dim=-1)
    assert torch.allclose(d, torch.full_like(d, 1.52), atol=0.05)


def test_distmat_loss():
    a = torch.randn(2, 137, 14, 3)
    b = torch.randn(2, 137, 14, 3)
    loss = distmat_loss_torch(a, b, p=2, q=2)
    assert torch.isfinite(loss)
    assert distmat_loss_torch(a, a, p=2, q=2) == 0


def test_lddt():
    a = torch.randn(2, 137, 14, 3)
    b = torch.randn(2, 137, 14, 3)
    cloud_mask = torch.ones(a.shape[:-1]).bool()
    lddt_result = lddt_ca_torch(a, b, cloud_mask)
    assert list(lddt_result.shape) == [2, 137]
    # identical structures must score a perfect 1
    perfect = lddt_ca_torch(a, a, cloud_mask)
    assert torch.allclose(perfect, torch.ones_like(perfect), atol=1e-5)


def test_kabsch():
    a = torch.randn(3, 8)
    b = torch.randn(3, 8)
    a_, b_ = Kabsch(a, b)
    assert a.shape == a_.shape


def test_kabsch_recovers_rotation():
    """Aligning a rotated copy must give (near) zero RMSD."""
    torch.manual_seed(1)
    a = torch.randn(3, 32).double()
    theta = torch.tensor(0.7)
    R = torch.tensor([[torch.cos(theta), -torch.sin(theta), 0.],
                      [torch.sin(theta), torch.cos(theta), 0.],
                      [0., 0., 1.]]).double()
    b = R @ a + torch.tensor([[1.], [2.], [3.]]).double()
    a_, b_ = Kabsch(a, b)
    assert RMSD(a_, b_).item() < 1e-5


def test_tmscore():
    a = torch.randn(2, 3, 8)
    b = torch.randn(2, 3, 8)
    out = TMscore(a, b)
    assert out.shape == (2,)
    # self comparison = 1
    assert torch.allclose(TMscore(a, a), torch.ones(2))


def test_gdt():
    a = torch.randn(1, 3, 8)
    b = torch.randn(1, 3, 8)
    GDT(a, b, weights=1)
    assert torch.allclose(GDT(a, a, weights=1), torch.ones(1))


def test_numpy_backend_agreement():
    a = torch.randn(2, 3, 16)
    b = torch.randn(2, 3, 16)
    t = TMscore(a, b)
    n = TMscore(a.numpy(), b.numpy())
    assert np.allclose(t.numpy(), n, atol=1e-5)
    t = RMSD(a, b)
    n = RMSD(a.numpy(), b.numpy())
    assert np.allclose(t.numpy(), n, atol=1e-5)


def test_prot_covalent_bond_and_adjacency():
    from alphafold2_amd.utils import prot_covalent_bond, nth_deg_adjacency
    seqs = torch.tensor([[0, 5, 7]])  # A, G, I
    mask_mat, attr_mat = prot_covalent_bond(seqs, adj_degree=1)
    assert mask_mat.any()
    # peptide bond between residue CA chain: symmetric adjacency
    assert torch.equal(attr_mat[0], attr_mat[0].t())
    adj = torch.zeros(5, 5)
    adj[0, 1] = adj[1, 0] = adj[1, 2] = adj[2, 1] = 1
    new_adj, attr = nth_deg_adjacency(adj, n=2)
    assert attr[0, 2] == 2  # two hops


def test_coords2pdb_writer(tmp_path):
    from alphafold2_amd.utils import coords2pdb, scn_cloud_mask
    seq = torch.tensor([0, 5, 3])  # A G E (alphabetical vocab)
    cloud = scn_cloud_mask(seq[None])[0]
    n_atoms = int(cloud.sum())
    coords = torch.randn(n_atoms, 3)
    path = coords2pdb(seq, coords, cloud, prefix=str(tmp_path) + '/')
    text = open(path).read()
    assert text.count('ATOM') == n_atoms
    assert 'ALA' in text and 'GLY' in text and 'GLU' in text


def test_mds_numpy_backend():
    import numpy as np
    from alphafold2_amd.utils import MDScaling
    pts = np.random.randn(20, 3)
    dist = np.linalg.norm(pts[:, None] - pts[None], axis=-1)
    coords, _ = MDScaling(dist, iters=30, fix_mirror=False)
    assert coords.shape == (1, 3, 20)


def test_distmat_loss_options():
    from alphafold2_amd.utils import distmat_loss_torch
    a = torch.randn(10, 3)
    b = torch.randn(10, 3)
    l1 = distmat_loss_torch(a, b, p=2, q=1)         # MAE-style
    l2 = distmat_loss_torch(a, b, clamp=(0, 5))
    l3 = distmat_loss_torch(a, b, custom=lambda x, y: (x - y).abs())
    assert all(torch.isfinite(t) for t in (l1, l2, l3))


def test_prot_covalent_bond_batch_independence():
    """Each batch item's hop attributes must equal its single-item
    computation (regression: nth_deg applied once, not per item)."""
    from alphafold2_amd.utils import prot_covalent_bond
    seqs = torch.tensor([[0, 5, 7], [3, 3, 3]])
    _, attr_batch = prot_covalent_bond(seqs, adj_degree=2)
    for i in range(2):
        _, attr_single = prot_covalent_bond(seqs[i:i + 1], adj_degree=2)
        assert torch.equal(attr_batch[i], attr_single[0])


# ---------------------------------------------------------------------------
# property-based invariants (hypothesis)

try:
    from hypothesis import given, settings, strategies as st
    HAS_HYP = True
except ImportError:  # pragma: no cover
    HAS_HYP = False

if HAS_HYP:
    @settings(max_examples=25, deadline=None, derandomize=True)
    @given(st.integers(min_value=4, max_value=64), st.integers(0, 10**6))
    def test_kabsch_rigid_motion_invariant(n, seed):
        """Kabsch must recover ANY rigid motion: RMSD(after) ~ 0."""
        from alphafold2_amd.models.quaternion import quaternion_to_matrix
        g = torch.Generator().manual_seed(seed)
        a = torch.randn(3, n, generator=g).double()
        q = torch.randn(4, generator=g).double()
        q = q / q.norm()
        R = quaternion_to_matrix(q[None])[0]
        t = torch.randn(3, 1, generator=g).double() * 10
        b = R @ a + t
        a_, b_ = Kabsch(a, b)
        assert RMSD(a_, b_).item() < 1e-6

    @settings(max_examples=25, deadline=None, derandomize=True)
    @given(st.integers(min_value=5, max_value=40), st.integers(0, 10**6))
    def test_tmscore_bounds_and_self(n, seed):
        g = torch.Generator().manual_seed(seed)
        a = torch.randn(1, 3, n, generator=g)
        b = torch.randn(1, 3, n, generator=g)
        tm = TMscore(a, b)
        assert 0.0 <= tm.item() <= 1.0
        assert torch.allclose(TMscore(a, a), torch.ones(1))

    @settings(max_examples=15, deadline=None, derandomize=True)
    @given(st.integers(min_value=6, max_value=24), st.integers(0, 10**6))
    def test_lddt_bounds(n, seed):
        g = torch.Generator().manual_seed(seed)
        a = torch.randn(1, n, 14, 3, generator=g) * 5
        b = a + torch.randn(1, n, 14, 3, generator=g) * 0.1
        cloud = torch.ones(1, n, 14).bool()
        val = lddt_ca_torch(a, b, cloud)
        assert (val >= 0).all() and (val <= 1).all()

    @settings(max_examples=15, deadline=None, derandomize=True)
    @given(st.integers(min_value=2, max_value=8), st.integers(0, 10**6))
    def test_distogram_bucket_bounds(b, seed):
        from alphafold2_amd.utils import get_bucketed_distance_matrix
        g = torch.Generator().manual_seed(seed)
        coords = torch.randn(b, 12, 3, generator=g) * 8
        mask = torch.ones(b, 12).bool()
        buckets = get_bucketed_distance_matrix(coords, mask)
        assert buckets.min() >= 0 and buckets.max() <= 36
        # symmetry
        assert torch.equal(buckets, buckets.transpose(1, 2))


def test_sidechain_ring_closure():
    """Every chemically true side-chain bond — including the aromatic
    ring-closure bonds of F/Y/H/W and proline's CD-N — must come out of
    the NeRF builder at a plausible bond length (regression: the W
    indole benzene ring was built from wrong parents)."""
    from alphafold2_amd.vocab import SC_ATOM_NAMES
    TRUE_BONDS = {
        'F': [('CB','CG'),('CG','CD1'),('CG','CD2'),('CD1','CE1'),
              ('CD2','CE2'),('CE1','CZ'),('CE2','CZ')],
        'Y': [('CB','CG'),('CG','CD1'),('CG','CD2'),('CD1','CE1'),
              ('CD2','CE2'),('CE1','CZ'),('CE2','CZ'),('CZ','OH')],
        'H': [('CB','CG'),('CG','ND1'),('CG','CD2'),('ND1','CE1'),
              ('CD2','NE2'),('CE1','NE2')],
        'W': [('CB','CG'),('CG','CD1'),('CG','CD2'),('CD1','NE1'),
              ('NE1','CE2'),('CD2','CE2'),('CD2','CE3'),('CE2','CZ2'),
              ('CE3','CZ3'),('CZ2','CH2'),('CZ3','CH2')],
        'P': [('CB','CG'),('CG','CD'),('CD','N')],
    }
    n = torch.tensor([0., 0., 0.])
    ca = torch.tensor([1.46, 0., 0.])
    c = torch.tensor([2.0, 1.42, 0.])
    o = torch.tensor([1.6, 2.5, 0.])
    res = torch.stack([n, ca, c, o])
    from alphafold2_amd.vocab import VOCAB
    for aa, bonds in TRUE_BONDS.items():
        names = ['N', 'CA', 'C', 'O'] + SC_ATOM_NAMES[aa]
        out = sidechain_container(
            torch.tensor([[VOCAB._char2int[aa]]]), res[None],
            atom_mask=torch.tensor([1] * 4 + [0] * 10))
        coords = out[0, 0]
        for a, b in bonds:
            d = (coords[names.index(a)] - coords[names.index(b)]).norm()
            assert 1.2 < d.item() < 1.95, (aa, a, b, d.item())


def test_sidechain_no_intra_residue_clashes():
    # no two atoms of any built residue may overlap (< 1.15 A)
    from alphafold2_amd.vocab import VOCAB, SC_ATOM_NAMES
    n = torch.tensor([0., 0., 0.])
    ca = torch.tensor([1.46, 0., 0.])
    c = torch.tensor([2.0, 1.42, 0.])
    o = torch.tensor([1.6, 2.5, 0.])
    res = torch.stack([n, ca, c, o])
    for aa_id in range(20):
        aa = VOCAB._int2char[aa_id]
        nat = 4 + len(SC_ATOM_NAMES[aa])
        out = sidechain_container(torch.tensor([[aa_id]]), res[None],
                                  atom_mask=torch.tensor([1] * 4 + [0] * 10))
        coords = out[0, 0, :nat]
        d = torch.cdist(coords, coords) + torch.eye(nat) * 99
        assert d.min().item() > 1.15, (aa, d.min().item())


def test_mirror_fix_enforces_protein_chirality():
    # MDS reconstruction is chirality-blind; fix_mirror must leave the
    # output with a negative-phi MAJORITY (the biological convention)
    # whichever mirror image the distances came from.
    import math
    from alphafold2_amd.geometry.mds import calc_phis_torch
    L = 24
    pts = []
    for i in range(L * 3):
        t = i * (2 * math.pi / 10.8)
        pts.append([2.3 * math.cos(t), 2.3 * math.sin(t), 0.5 * i])
    coords = torch.tensor(pts)              # (N, 3) helix
    for flip in (1.0, -1.0):                # both mirror images
        c = coords.clone()
        c[:, 2] *= flip
        dist = torch.cdist(c[None], c[None])[0]
        masker = torch.arange(L * 3) % 3
        N_mask, CA_mask = (masker == 0), (masker == 1)
        out, _ = MDScaling(dist, iters=60, fix_mirror=True,
                           N_mask=N_mask, CA_mask=CA_mask, C_mask=None)
        frac = calc_phis_torch(out, N_mask, CA_mask, prop=True)[0]
        assert frac >= 0.5, (flip, float(frac))
