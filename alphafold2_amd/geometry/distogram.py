"""Distogram binning and distogram -> distance-matrix centering.

Capability parity: reference utils.py:41-50 (get_bucketed_distance_matrix)
and utils.py:718-761 (center_distogram_torch).  The cdist+bucketize hot
path has a fused HIP kernel on MI355X (ops.distance_buckets); this module
is the always-available eager path and the CPU reference.
"""
import numpy as np
import torch

from .. import constants

DISTANCE_THRESHOLDS = torch.linspace(
    constants.DISTOGRAM_MIN_DIST, constants.DISTOGRAM_MAX_DIST,
    steps=constants.DISTOGRAM_BUCKETS)


def get_bucketed_distance_matrix(coords, mask,
                                 num_buckets=constants.DISTOGRAM_BUCKETS,
                                 ignore_index=-100):
    """coords (b, n, 3), mask (b, n) -> (b, n, n) long bucket targets.

    Pairwise distances binned into `num_buckets` edges spanning 2..20 Å;
    pairs touching a masked residue are set to `ignore_index`.
    """
    distances = torch.cdist(coords, coords, p=2)
    boundaries = torch.linspace(constants.DISTOGRAM_MIN_DIST,
                                constants.DISTOGRAM_MAX_DIST,
                                steps=num_buckets, device=coords.device)
    buckets = torch.bucketize(distances, boundaries[:-1])
    pair_mask = mask[..., None] & mask[..., None, :]
    buckets.masked_fill_(~pair_mask, ignore_index)
    return buckets


def center_distogram_torch(distogram, bins=DISTANCE_THRESHOLDS, min_t=1.,
                           center="mean", wide="std"):
    """Distogram (b, N, N, B) -> (central (b,N,N), weights (b,N,N)).

    Central estimate (mean or median over bin midpoints) and a 0-1 weight
    map derived from the dispersion, with the diagonal and the
    above-last-threshold class zeroed.  Mirrors reference
    utils.py:718-761 semantics.
    """
    shape, device = distogram.shape, distogram.device
    # bin centers (midpoint below each threshold); clamp the two ends
    n_bins = (bins - 0.5 * (bins[2] - bins[1])).to(device)
    n_bins[0] = 1.5
    n_bins[-1] = 1.33 * bins[-1]  # catch-all class above the last threshold
    max_bin_allowed = torch.tensor(n_bins.shape[0] - 1, device=device).long()

    magnitudes = distogram.sum(dim=-1)
    if center == "median":
        cum_dist = torch.cumsum(distogram, dim=-1)
        medium = 0.5 * cum_dist[..., -1:]
        central = torch.searchsorted(cum_dist, medium).squeeze()
        central = n_bins[torch.min(central, max_bin_allowed)]
    else:  # mean
        central = (distogram * n_bins).sum(dim=-1) / magnitudes

    # mask out the catch-all last class
    mask = (central <= bins[-2].item()).float()

    diag_idxs = np.arange(shape[-2])
    from .backend import expand_dims_to
    central = expand_dims_to(central, 3 - len(central.shape))
    central[:, diag_idxs, diag_idxs] *= 0.

    if wide == "var":
        dispersion = (distogram * (n_bins - central.unsqueeze(-1)) ** 2).sum(dim=-1) / magnitudes
    elif wide == "std":
        dispersion = ((distogram * (n_bins - central.unsqueeze(-1)) ** 2).sum(dim=-1) / magnitudes).sqrt()
    else:
        dispersion = torch.zeros_like(central, device=device)

    # lower dispersion -> weight closer to 1; nan-safe; zero diagonal
    weights = mask / (1 + dispersion)
    weights[weights != weights] *= 0.
    weights[:, diag_idxs, diag_idxs] *= 0.
    return central, weights
