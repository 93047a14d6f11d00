"""Geometry-stack validation on physically realistic protein structure
(parity with the reference's real-PDB notebook checks,
reference notebooks/structure_utils_tests.ipynb).

No network and no copied PDB files: the fixture is an ideal alpha-helix
built by NeRF from textbook internal coordinates (bond lengths N-CA
1.458 / CA-C 1.525 / C-N 1.329 A; angles 111/117.2/121.7 deg;
phi=-57, psi=-47, omega=180).  That gives chemically correct backbone
geometry — negative phis, ~3.8 A CA spacing, helical distogram — which
is what the real-PDB tests exercise.
"""
import math

import pytest
import torch

from alphafold2_amd.utils import (
    GDT, Kabsch, MDScaling, RMSD, TMscore, lddt_ca_torch,
)
from alphafold2_amd.geometry.mds import calc_phis_torch, get_dihedral_torch


def _place_atom(a, b, c, bond, angle, torsion):
    """NeRF: position of atom D given A-B-C and (|CD|, angle BCD,
    torsion ABCD)."""
    bc = c - b
    bc = bc / bc.norm()
    n = torch.cross(b - a, bc, dim=-1)
    n = n / n.norm()
    m = torch.cross(n, bc, dim=-1)
    ang = math.pi - angle            # internal -> placement angle
    d_local = torch.tensor([
        bond * math.cos(ang),
        bond * math.sin(ang) * math.cos(torsion),
        bond * math.sin(ang) * math.sin(torsion)])
    rot = torch.stack([bc, m, n], dim=-1)
    return c + rot @ d_local


def ideal_helix(n_res=24):
    """Backbone (N, CA, C) coords of an ideal right-handed alpha-helix.
    Returns (n_res, 3, 3) float tensor."""
    d2r = math.pi / 180.0
    phi, psi, omega = -57 * d2r, -47 * d2r, 180 * d2r
    b_nca, b_cac, b_cn = 1.458, 1.525, 1.329
    a_ncac, a_cacn, a_cnca = 111 * d2r, 117.2 * d2r, 121.7 * d2r

    # seed residue
    atoms = [torch.tensor([0.0, 0.0, 0.0]),           # N
             torch.tensor([b_nca, 0.0, 0.0])]         # CA
    atoms.append(_place_atom(
        atoms[0] + torch.tensor([0., 1., 0.]), atoms[0], atoms[1],
        b_cac, a_ncac, 0.5))                          # C (arbitrary seed dir)
    for _ in range(1, n_res):
        n_prev, ca_prev, c_prev = atoms[-3], atoms[-2], atoms[-1]
        n_new = _place_atom(n_prev, ca_prev, c_prev, b_cn, a_cacn, psi)
        ca_new = _place_atom(ca_prev, c_prev, n_new, b_nca, a_cnca, omega)
        c_new = _place_atom(c_prev, n_new, ca_new, b_cac, a_ncac, phi)
        atoms.extend([n_new, ca_new, c_new])
    return torch.stack(atoms).reshape(n_res, 3, 3)


@pytest.fixture(scope='module')
def helix():
    return ideal_helix(24)


def test_helix_is_chemically_sane(helix):
    ca = helix[:, 1]
    d = (ca[1:] - ca[:-1]).norm(dim=-1)
    # consecutive CA-CA distance in an alpha-helix ~= 3.8 A
    assert (d - 3.8).abs().max() < 0.15, d
    # helical pitch: CA(i)-CA(i+4) hydrogen-bond partner ~6.2 A
    d4 = (ca[4:] - ca[:-4]).norm(dim=-1)
    assert 5.0 < d4.mean() < 7.0

    # backbone phi dihedrals equal the -57 deg used to build it
    n_at, ca_at, c_at = helix[:, 0], helix[:, 1], helix[:, 2]
    phis = get_dihedral_torch(c_at[:-1], n_at[1:], ca_at[1:], c_at[1:])
    assert (phis - math.radians(-57)).abs().max() < 1e-3


def test_phi_ratio_detects_chirality(helix):
    flat = helix.reshape(-1, 3).t()[None]          # (1, 3, N_atoms)
    n_mask = torch.zeros(flat.shape[-1], dtype=torch.bool)
    ca_mask = torch.zeros_like(n_mask)
    n_mask[0::3] = True
    ca_mask[1::3] = True
    frac_neg = calc_phis_torch(flat, n_mask, ca_mask, prop=True)[0]
    assert frac_neg == 1.0                         # all phis negative
    # mirror image: all phis flip sign
    mirrored = flat.clone()
    mirrored[:, 2] = -mirrored[:, 2]
    assert calc_phis_torch(mirrored, n_mask, ca_mask, prop=True)[0] == 0.0


def test_mds_reconstructs_helix_with_mirror_fix(helix):
    flat = helix.reshape(-1, 3)                    # (72, 3)
    dist = torch.cdist(flat[None], flat[None])[0]
    n_mask = torch.zeros(flat.shape[0], dtype=torch.bool)
    ca_mask = torch.zeros_like(n_mask)
    n_mask[0::3] = True
    ca_mask[1::3] = True
    coords, _ = MDScaling(dist, iters=100, tol=1e-7, fix_mirror=True,
                          N_mask=n_mask, CA_mask=ca_mask)
    # rigid-align and compare: reconstruction error well under a bond
    a, b = Kabsch(coords[0], flat.t())
    rmsd = RMSD(a, b)[0]
    assert rmsd < 0.5, rmsd
    # chirality correct after the mirror fix: phis negative
    frac = calc_phis_torch(coords, n_mask, ca_mask, prop=True)[0]
    assert frac > 0.5


def test_lddt_on_helix(helix):
    scn = torch.zeros(1, helix.shape[0], 14, 3)
    scn[0, :, :3] = helix                          # N, CA, C slots
    cloud = torch.zeros(1, helix.shape[0], 14, dtype=torch.bool)
    cloud[:, :, :3] = True
    perfect = lddt_ca_torch(scn, scn, cloud.float())
    assert torch.allclose(perfect, torch.ones_like(perfect))

    # increasing noise monotonically degrades the score
    g = torch.Generator().manual_seed(0)
    prev = 1.0
    for sigma in (0.2, 1.0, 3.0):
        noisy = scn + torch.randn(scn.shape, generator=g) * sigma
        score = lddt_ca_torch(scn, noisy, cloud.float()).mean().item()
        assert score < prev + 1e-6
        prev = score
    assert prev < 0.5


def test_alignment_metrics_on_perturbed_helix(helix):
    ca = helix[:, 1].t()                           # (3, L)
    g = torch.Generator().manual_seed(1)
    noise = torch.randn(ca.shape, generator=g) * 0.3
    # rigid motion + small noise
    theta = math.radians(30)
    rot = torch.tensor([[math.cos(theta), -math.sin(theta), 0],
                        [math.sin(theta), math.cos(theta), 0],
                        [0, 0, 1.0]])
    moved = rot @ (ca + noise) + torch.tensor([[5.0], [3.0], [-2.0]])

    a, b = Kabsch(moved, ca)
    r = RMSD(a, b)[0]
    assert r < 0.45                                # noise floor, not 5 A
    assert TMscore(a[None], b[None])[0] > 0.7  # short-chain d0 is harsh
    assert GDT(a[None], b[None], mode='HA')[0] > 0.8
    # GDT_TS is more permissive than GDT_HA by construction
    assert GDT(a[None], b[None], mode='TS')[0] >= \
        GDT(a[None], b[None], mode='HA')[0]


def test_distogram_mds_pipeline_on_helix():
    """distogram -> central estimate -> MDS -> align: the reference's
    end-to-end distogram fallback on real geometry.
    Short helix so every pairwise distance sits inside the 2-20 A
    distogram range (beyond-range pairs are unrecoverable by design)."""
    from alphafold2_amd.utils import center_distogram_torch
    from alphafold2_amd.constants import DISTOGRAM_BUCKETS
    ca = ideal_helix(12)[:, 1]
    L = ca.shape[0]
    dist = torch.cdist(ca[None], ca[None])
    # build a sharp synthetic distogram peaked at the true bin
    bins = torch.linspace(2, 20, DISTOGRAM_BUCKETS - 1)
    idx = torch.bucketize(dist, bins).clamp(max=DISTOGRAM_BUCKETS - 1)
    logits = torch.full((1, L, L, DISTOGRAM_BUCKETS), -8.0)
    logits.scatter_(-1, idx[..., None], 8.0)
    distogram = logits.softmax(dim=-1)
    est, weights = center_distogram_torch(distogram)[:2]
    coords, _ = MDScaling(est[0], weights=weights, iters=200, tol=1e-7,
                          fix_mirror=False)
    a, b = Kabsch(coords[0], ca.t())
    assert RMSD(a, b)[0] < 1.0
