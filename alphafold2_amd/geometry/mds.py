"""Multidimensional scaling (distogram -> 3D) + chirality fix.

Capability parity: reference utils.py:766-879 (mds twins), 881-993
(dihedrals / phi-based mirror selection), 1162-1201 + 1254-1279
(mdscaling wrappers).  The torch path is fully batched (batched
svd_lowrank init + Guttman transform via bmm) and runs on the MI355X.
"""
import numpy as np
import torch

from .backend import set_backend_kwarg, invoke_torch_or_numpy, expand_dims_to


def mds_torch(pre_dist_mat, weights=None, iters=10, tol=1e-5, eigen=False,
              verbose=0):
    """Distance matrix (b, N, N) -> coords (b, 3, N) + stress history.

    Eigen-style init from the Gram matrix (svd_lowrank), then iterative
    Guttman-transform refinement (SMACOF).
    """
    device = pre_dist_mat.device
    pre_dist_mat = expand_dims_to(pre_dist_mat, length=3 - len(pre_dist_mat.shape))
    batch, N, _ = pre_dist_mat.shape
    diag_idxs = torch.arange(N, device=device)
    his = [torch.full((batch,), float('inf'), device=device)]

    # classical-MDS style init: Gram matrix from squared distances
    D = pre_dist_mat ** 2
    M = 0.5 * (D[:, :1, :] + D[:, :, :1] - D)
    # per-item svd_lowrank (batched svd_lowrank is slower in practice)
    svds = [torch.svd_lowrank(mi) for mi in M]
    u = torch.stack([s[0] for s in svds], dim=0)
    s = torch.stack([s[1] for s in svds], dim=0)
    best_3d_coords = torch.bmm(u, torch.diag_embed(s).abs().sqrt())[..., :3]

    if weights is None and eigen:
        return best_3d_coords.transpose(-1, -2), torch.zeros_like(torch.stack(his, dim=0))
    elif eigen and verbose:
        print("Can't use eigen flag if weights are active. Fallback to iterative")

    if weights is None:
        weights = torch.ones_like(pre_dist_mat)

    for i in range(iters):
        best_3d_coords = best_3d_coords.contiguous()
        dist_mat = torch.cdist(best_3d_coords, best_3d_coords, p=2).clone()

        stress = (weights * (dist_mat - pre_dist_mat) ** 2).sum(dim=(-1, -2)) * 0.5
        dist_mat[dist_mat <= 0] += 1e-7
        ratio = weights * (pre_dist_mat / dist_mat)
        B = -ratio
        B[:, diag_idxs, diag_idxs] += ratio.sum(dim=-1)

        coords = (1. / N) * torch.matmul(B, best_3d_coords)
        dis = torch.norm(coords, dim=(-1, -2))

        if verbose >= 2:
            print(f'it: {i}, stress {stress}')
        if (his[-1] - stress / dis).mean() <= tol:
            if verbose:
                print(f'breaking at iteration {i} with stress {stress / dis}')
            break

        best_3d_coords = coords
        his.append(stress / dis)

    return best_3d_coords.transpose(-1, -2), torch.stack(his, dim=0)


def mds_numpy(pre_dist_mat, weights=None, iters=10, tol=1e-5, eigen=False,
              verbose=0):
    if weights is None:
        weights = np.ones_like(pre_dist_mat)
    pre_dist_mat = expand_dims_to(pre_dist_mat, length=3 - len(pre_dist_mat.shape))
    batch, N, _ = pre_dist_mat.shape
    # per-batch history (a scalar seed breaks numpy>=2 stacking)
    his = [np.full(batch, np.inf)]
    best_stress = np.inf * np.ones(batch)
    best_3d_coords = 2 * np.random.rand(batch, 3, N) - 1
    for i in range(iters):
        dist_mat = np.linalg.norm(
            best_3d_coords[:, :, :, None] - best_3d_coords[:, :, None, :], axis=-3)
        stress = ((weights * (dist_mat - pre_dist_mat)) ** 2).sum(axis=(-1, -2)) * 0.5
        dist_mat[dist_mat == 0] = 1e-7
        ratio = weights * (pre_dist_mat / dist_mat)
        B = -ratio
        B[:, np.arange(N), np.arange(N)] += ratio.sum(axis=-1)
        coords = (1. / N) * np.matmul(best_3d_coords, B)
        dis = np.linalg.norm(coords, axis=(-1, -2))
        if verbose >= 2:
            print(f'it: {i}, stress {stress}')
        if (best_stress - stress / dis).mean() <= tol:
            if verbose:
                print(f'breaking at iteration {i} with stress {stress / dis}')
            break
        best_3d_coords = coords
        best_stress = stress / dis
        his.append(best_stress)
    return best_3d_coords, np.array(his)


# ---------------------------------------------------------------------------
# dihedrals


def get_dihedral_torch(c1, c2, c3, c4):
    """Dihedral angle (radians) for four points, batched over leading dims."""
    u1 = c2 - c1
    u2 = c3 - c2
    u3 = c4 - c3
    return torch.atan2(
        ((torch.norm(u2, dim=-1, keepdim=True) * u1) * torch.cross(u2, u3, dim=-1)).sum(dim=-1),
        (torch.cross(u1, u2, dim=-1) * torch.cross(u2, u3, dim=-1)).sum(dim=-1))


def get_dihedral_numpy(c1, c2, c3, c4):
    u1 = c2 - c1
    u2 = c3 - c2
    u3 = c4 - c3
    return np.arctan2(
        ((np.linalg.norm(u2, axis=-1, keepdims=True) * u1) * np.cross(u2, u3, axis=-1)).sum(axis=-1),
        (np.cross(u1, u2, axis=-1) * np.cross(u2, u3, axis=-1)).sum(axis=-1))


def calc_phis_torch(pred_coords, N_mask, CA_mask, C_mask=None, prop=True,
                    verbose=0):
    """Proportion of negative backbone phi dihedrals per structure.

    Used to pick the correct mirror image after MDS.  pred_coords is
    (batch, 3, N_atoms) with per-atom boolean masks for N/CA/C positions.
    """
    pred_coords_ = pred_coords.detach().transpose(-1, -2).cpu()
    N_mask = expand_dims_to(N_mask, 2 - len(N_mask.shape))
    CA_mask = expand_dims_to(CA_mask, 2 - len(CA_mask.shape))
    if C_mask is not None:
        C_mask = expand_dims_to(C_mask, 2 - len(C_mask.shape))
    else:
        C_mask = torch.logical_not(torch.logical_or(N_mask, CA_mask))

    n_terms = pred_coords_[:, N_mask[0].squeeze()]
    c_alphas = pred_coords_[:, CA_mask[0].squeeze()]
    c_terms = pred_coords_[:, C_mask[0].squeeze()]

    phis = [get_dihedral_torch(c_terms[i, :-1], n_terms[i, 1:],
                               c_alphas[i, 1:], c_terms[i, 1:])
            for i in range(pred_coords.shape[0])]
    if prop:
        return torch.stack([(x < 0).float().mean() for x in phis], dim=0)
    return phis


def calc_phis_numpy(pred_coords, N_mask, CA_mask, C_mask=None, prop=True,
                    verbose=0):
    pred_coords_ = np.transpose(pred_coords, (0, 2, 1))
    n_terms = pred_coords_[:, N_mask.squeeze()]
    c_alphas = pred_coords_[:, CA_mask.squeeze()]
    if C_mask is not None:
        c_terms = pred_coords_[:, C_mask]
    else:
        c_terms = pred_coords_[:, (np.ones_like(N_mask) - N_mask - CA_mask).squeeze().astype(bool)]
    phis = [get_dihedral_numpy(c_terms[i, :-1], n_terms[i, 1:],
                               c_alphas[i, 1:], c_terms[i, 1:])
            for i in range(pred_coords.shape[0])]
    if prop:
        return np.array([(x < 0).mean() for x in phis])
    return phis


# ---------------------------------------------------------------------------
# protein-aware MDS wrappers


def mdscaling_torch(pre_dist_mat, weights=None, iters=10, tol=1e-5,
                    fix_mirror=True, N_mask=None, CA_mask=None, C_mask=None,
                    eigen=False, verbose=0):
    preds, stresses = mds_torch(pre_dist_mat, weights=weights, iters=iters,
                                tol=tol, eigen=eigen, verbose=verbose)
    if not fix_mirror:
        return preds, stresses
    phi_ratios = calc_phis_torch(preds, N_mask, CA_mask, C_mask, prop=True)
    to_correct = torch.nonzero(phi_ratios < 0.5).view(-1)
    # flip Z of structures whose phi distribution says "wrong mirror"
    preds[to_correct, -1] = -preds[to_correct, -1]
    if verbose == 2:
        print("Corrected mirror idxs:", to_correct)
    return preds, stresses


def mdscaling_numpy(pre_dist_mat, weights=None, iters=10, tol=1e-5,
                    fix_mirror=True, N_mask=None, CA_mask=None, C_mask=None,
                    verbose=0):
    preds, stresses = mds_numpy(pre_dist_mat, weights=weights, iters=iters,
                                tol=tol, verbose=verbose)
    if not fix_mirror:
        return preds, stresses
    phi_ratios = calc_phis_numpy(preds, N_mask, CA_mask, C_mask, prop=True)
    for i in range(len(preds)):
        if phi_ratios[i] < 0.5:
            preds[i, -1] = -preds[i, -1]
            if verbose == 2:
                print("Corrected mirror in struct no.", i)
    return preds, stresses


@set_backend_kwarg
@invoke_torch_or_numpy(mdscaling_torch, mdscaling_numpy)
def MDScaling(pre_dist_mat, **kwargs):
    """Distance matrix (N, N) (or batch) -> 3D coords (3, N) via MDS with
    optional phi-based mirror correction.  See mds_torch for details."""
    pre_dist_mat = expand_dims_to(pre_dist_mat, 3 - len(pre_dist_mat.shape))
    return pre_dist_mat, kwargs
