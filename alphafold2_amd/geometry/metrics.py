"""Structure alignment + quality metrics (Kabsch, RMSD, GDT, TM-score, lDDT).

Capability parity: reference utils.py:999-1052 (Kabsch twins),
1057-1247 (losses/metrics), 1281-1344 (public wrappers).  Implementations
are fresh, vectorized where the reference loops (lddt is fully batched
here), and keep the dual torch/numpy backend discipline.
"""
import numpy as np
import torch

from .backend import dual_backend, named_wrapper

# ---------------------------------------------------------------------------
# Kabsch alignment


def kabsch_torch(X, Y, cpu=True):
    """Optimal rigid alignment of X onto Y; (D, N) or batched (B, D, N).

    Returns (X_aligned, Y_centered) with the input rank preserved.  The
    rotation is computed on a detached covariance (gradients flow
    through the centering only); the 3x3 SVDs run batched in ONE host
    call by default — one sync for the whole batch instead of a CPU
    round-trip per structure (LAPACK on a (B,3,3) stack beats rocSOLVER
    launch latency at these sizes).
    """
    batched = X.dim() == 3
    Xb = X if batched else X[None]
    Yb = Y if batched else Y[None]
    Xc = Xb - Xb.mean(dim=-1, keepdim=True)
    Yc = Yb - Yb.mean(dim=-1, keepdim=True)
    cov = torch.matmul(Xc, Yc.transpose(-1, -2)).detach()  # (B, D, D)
    dev = cov.cpu() if cpu else cov
    V, _, W = torch.linalg.svd(dev.float())
    # right-handedness: flip the last left-singular vector where the
    # proposed rotation would be a reflection (vectorized over batch)
    flip = (torch.det(V) * torch.det(W)) < 0.0
    V = torch.where(flip[:, None, None],
                    torch.cat([V[..., :-1], -V[..., -1:]], dim=-1), V)
    R = torch.matmul(V, W).to(device=Xb.device, dtype=Xb.dtype)  # (B, D, D)
    X_out = torch.matmul(R.transpose(-1, -2), Xc)
    if not batched:
        return X_out[0], Yc[0]
    return X_out, Yc


def kabsch_numpy(X, Y):
    batched = X.ndim == 3
    Xb = X if batched else X[None]
    Yb = Y if batched else Y[None]
    Xc = Xb - Xb.mean(axis=-1, keepdims=True)
    Yc = Yb - Yb.mean(axis=-1, keepdims=True)
    cov = np.matmul(Xc, np.swapaxes(Yc, -1, -2))
    V, _, W = np.linalg.svd(cov)
    flip = (np.linalg.det(V) * np.linalg.det(W)) < 0.0
    V[flip, :, -1] = -V[flip, :, -1]
    R = np.matmul(V, W)
    X_out = np.matmul(np.swapaxes(R, -1, -2), Xc)
    if not batched:
        return X_out[0], Yc[0]
    return X_out, Yc


# ---------------------------------------------------------------------------
# losses / metrics


def _as_distmat(coords, mat, p, clamp):
    """Coordinates (N, d) -> pairwise distance matrix, unless a matrix
    was supplied directly."""
    if mat is not None:
        return mat
    pts = coords.squeeze()
    if clamp is not None:
        pts = pts.clamp(*clamp)
    return torch.cdist(pts, pts, p=p)


def distmat_loss_torch(X=None, Y=None, X_mat=None, Y_mat=None, p=2, q=2,
                       custom=None, distmat_mask=None, clamp=None):
    """Distance-matrix loss between predicted and true structures.

    Accepts either coordinates (N, d) or precomputed distance matrices
    (N, N); `q` scales the loss power (2=MSE, 1=MAE).
    """
    assert (X is not None or X_mat is not None) and \
           (Y is not None or Y_mat is not None), \
        "true and predicted coords or dist mats must be provided"
    pred = _as_distmat(X, X_mat, p, clamp)
    true = _as_distmat(Y, Y_mat, p, clamp)
    if custom is not None:
        return custom(pred.squeeze(), true.squeeze()).mean()
    per_pair = (pred - true).square()
    if q != 2:
        per_pair = per_pair.pow(q / 2)
    if distmat_mask is not None:
        per_pair = per_pair[distmat_mask]
    return per_pair.mean()


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

Kabsch = named_wrapper(
    dual_backend(kabsch_torch, kabsch_numpy, pair_ndim=2),
    'Kabsch', "Kabsch-align A (3, N) onto B (3, N); returns the aligned pair.")

RMSD = named_wrapper(
    dual_backend(rmsd_torch, rmsd_numpy, pair_ndim=3),
    'RMSD', "RMSD between A and B, (B, 3, N) or (3, N) -> (B,).")

TMscore = named_wrapper(
    dual_backend(tmscore_torch, tmscore_numpy, pair_ndim=3),
    'TMscore', "TM-score between A and B, (B, 3, N) or (3, N) -> (B,).")


def _gdt_prepare(A, B, mode="TS", cutoffs=None, weights=None):
    if cutoffs is None:
        cutoffs = [0.5, 1, 2, 4] if mode in ("HA", "ha") else [1, 2, 4, 8]
    return (A, B, cutoffs), {'weights': weights}


GDT = named_wrapper(
    dual_backend(gdt_torch, gdt_numpy, pair_ndim=3, prepare=_gdt_prepare),
    'GDT', "GDT_TS (cutoffs 1/2/4/8) or GDT_HA (0.5/1/2/4); higher is better.")
