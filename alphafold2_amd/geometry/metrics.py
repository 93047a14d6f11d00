"""Structure alignment + quality metrics (Kabsch, RMSD, GDT, TM-score, lDDT).

Capability parity: reference utils.py:999-1052 (Kabsch twins),
1057-1247 (losses/metrics), 1281-1344 (public wrappers).  Implementations
are fresh, vectorized where the reference loops (lddt is fully batched
here), and keep the dual torch/numpy backend discipline.
"""
import numpy as np
import torch

from .backend import (
    set_backend_kwarg, expand_arg_dims, invoke_torch_or_numpy,
)

# ---------------------------------------------------------------------------
# Kabsch alignment


def kabsch_torch(X, Y, cpu=True):
    """Optimal rigid alignment of X onto Y; both (D, N).

    Returns (X_aligned, Y_centered).  The rotation is computed on a
    detached covariance (gradients flow through the centering only, as in
    the reference); SVD runs on CPU by default — the 3x3 SVD is tiny and
    ROCm SVD launch latency dominates on-device.
    """
    device = X.device
    X_ = X - X.mean(dim=-1, keepdim=True)
    Y_ = Y - Y.mean(dim=-1, keepdim=True)
    C = torch.matmul(X_, Y_.t()).detach()
    if cpu:
        C = C.cpu()
    V, S, W = torch.linalg.svd(C)
    # right-handedness fix
    d = (torch.det(V) * torch.det(W)) < 0.0
    if d:
        S = S.clone()
        V = V.clone()
        S[-1] = -S[-1]
        V[:, -1] = -V[:, -1]
    U = torch.matmul(V, W).to(device)
    X_ = torch.matmul(X_.t(), U).t()
    return X_, Y_


def kabsch_numpy(X, Y):
    X_ = X - X.mean(axis=-1, keepdims=True)
    Y_ = Y - Y.mean(axis=-1, keepdims=True)
    C = np.dot(X_, Y_.transpose())
    V, S, W = np.linalg.svd(C)
    d = (np.linalg.det(V) * np.linalg.det(W)) < 0.0
    if d:
        S[-1] = -S[-1]
        V[:, -1] = -V[:, -1]
    U = np.dot(V, W)
    X_ = np.dot(X_.T, U).T
    return X_, Y_


# ---------------------------------------------------------------------------
# losses / metrics


def distmat_loss_torch(X=None, Y=None, X_mat=None, Y_mat=None, p=2, q=2,
                       custom=None, distmat_mask=None, clamp=None):
    """Distance-matrix loss between predicted and true structures.

    Accepts either coordinates (N, d) or precomputed distance matrices
    (N, N); `q` scales the loss power (2=MSE, 1=MAE).
    """
    assert (X is not None or X_mat is not None) and \
           (Y is not None or Y_mat is not None), \
        "true and predicted coords or dist mats must be provided"
    if X_mat is None:
        X = X.squeeze()
        if clamp is not None:
            X = torch.clamp(X, *clamp)
        X_mat = torch.cdist(X, X, p=p)
    if Y_mat is None:
        Y = Y.squeeze()
        if clamp is not None:
            Y = torch.clamp(Y, *clamp)
        Y_mat = torch.cdist(Y, Y, p=p)
    if distmat_mask is None:
        distmat_mask = torch.ones_like(Y_mat).bool()

    if custom is not None:
        return custom(X_mat.squeeze(), Y_mat.squeeze()).mean()
    loss = (X_mat - Y_mat) ** 2
    if q != 2:
        loss = loss ** (q / 2)
    return loss[distmat_mask].mean()


def rmsd_torch(X, Y):
    """X, Y: (B, D, N) -> (B,)"""
    return torch.sqrt(torch.mean((X - Y) ** 2, dim=(-1, -2)))


def rmsd_numpy(X, Y):
    return np.sqrt(np.mean((X - Y) ** 2, axis=(-1, -2)))


def gdt_torch(X, Y, cutoffs, weights=None):
    """Global distance test. X, Y: (B, D, N) -> (B,)"""
    device = X.device
    if weights is None:
        weights = torch.ones(1, len(cutoffs), device=device)
    else:
        weights = torch.tensor([weights], device=device)
    dist = ((X - Y) ** 2).sum(dim=1).sqrt()  # (B, N)
    # fraction of residues within each cutoff, all cutoffs at once
    cut = torch.tensor(cutoffs, device=device).view(1, -1, 1)
    GDT = (dist.unsqueeze(1) <= cut).float().mean(dim=-1)  # (B, K)
    return (GDT * weights).mean(-1)


def gdt_numpy(X, Y, cutoffs, weights=None):
    if weights is None:
        weights = np.ones((1, len(cutoffs)))
    else:
        weights = np.array([weights])
    dist = np.sqrt(((X - Y) ** 2).sum(axis=1))
    cut = np.array(cutoffs).reshape(1, -1, 1)
    GDT = (dist[:, None, :] <= cut).mean(axis=-1)
    return (GDT * weights).mean(-1)


def tmscore_torch(X, Y):
    """Template-modeling score. X, Y: (B, D, N) -> (B,)"""
    L = max(15, X.shape[-1])
    d0 = 1.24 * (L - 15) ** (1 / 3) - 1.8
    dist = ((X - Y) ** 2).sum(dim=1).sqrt()
    return (1 / (1 + (dist / d0) ** 2)).mean(dim=-1)


def tmscore_numpy(X, Y):
    L = max(15, X.shape[-1])
    d0 = 1.24 * np.cbrt(L - 15) - 1.8
    dist = np.sqrt(((X - Y) ** 2).sum(axis=1))
    return (1 / (1 + (dist / d0) ** 2)).mean(axis=-1)


def lddt_ca_torch(true_coords, pred_coords, cloud_mask, r_0=15.):
    """Per-residue lDDT over C-alpha atoms, fully batched.

    Inputs in scn format: coords (b, l, c, d), cloud_mask (b, l, c).
    Output: (b, l) scores in [0, 1].  Thresholds 0.5/1/2/4 Å within an
    inclusion radius r_0 of the reference structure (diagonal excluded).
    Replaces the reference's per-batch python loop (utils.py:1204-1247)
    with one vectorized pass — GPU-friendly (eval hot path).
    """
    device = true_coords.device
    thresholds = torch.tensor([0.5, 1., 2., 4.], device=device,
                              dtype=true_coords.dtype)
    b, l = true_coords.shape[:2]

    ca_mask = cloud_mask[..., 1].bool()            # (b, l) residue exists
    ca_true = true_coords[:, :, 1, :]              # (b, l, 3)
    ca_pred = pred_coords[:, :, 1, :]

    dist_true = torch.cdist(ca_true, ca_true, p=2)     # (b, l, l)
    dist_pred = torch.cdist(ca_pred, ca_pred, p=2)

    pair_mask = ca_mask[:, :, None] & ca_mask[:, None, :]
    eye = torch.eye(l, device=device, dtype=torch.bool).unsqueeze(0)
    included = (dist_true < r_0) & pair_mask & ~eye    # (b, l, l)

    delta = (dist_pred - dist_true).abs()
    # count of thresholds the deviation stays under (0..4)
    under = (delta.unsqueeze(-1) < thresholds).sum(dim=-1).to(true_coords.dtype)
    score_sum = (under * included).sum(dim=-1)         # (b, l)
    denom = 4. * included.sum(dim=-1)
    out = torch.zeros(b, l, device=device, dtype=true_coords.dtype)
    has_pairs = denom > 0
    out[has_pairs] = score_sum[has_pairs] / denom[has_pairs]
    out = out * ca_mask.to(out.dtype)
    return out


# ---------------------------------------------------------------------------
# public wrappers (backend-dispatching, parity with reference utils.py:1281+)


@expand_arg_dims(dim_len=2)
@set_backend_kwarg
@invoke_torch_or_numpy(kabsch_torch, kabsch_numpy)
def Kabsch(A, B):
    """Kabsch-align A (3, N) onto B (3, N); returns the aligned pair."""
    return A, B


@expand_arg_dims()
@set_backend_kwarg
@invoke_torch_or_numpy(rmsd_torch, rmsd_numpy)
def RMSD(A, B):
    """RMSD between A and B, (B, 3, N) or (3, N) -> (B,)."""
    return A, B


@expand_arg_dims()
@set_backend_kwarg
@invoke_torch_or_numpy(gdt_torch, gdt_numpy)
def GDT(A, B, *, mode="TS", cutoffs=None, weights=None):
    """GDT_TS (cutoffs 1/2/4/8) or GDT_HA (0.5/1/2/4); higher is better."""
    cutoffs = [0.5, 1, 2, 4] if mode in ("HA", "ha") else [1, 2, 4, 8]
    return A, B, cutoffs, {'weights': weights}


@expand_arg_dims()
@set_backend_kwarg
@invoke_torch_or_numpy(tmscore_torch, tmscore_numpy)
def TMscore(A, B):
    """TM-score between A and B, (B, 3, N) or (3, N) -> (B,)."""
    return A, B
