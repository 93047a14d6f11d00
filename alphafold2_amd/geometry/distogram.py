"""Distogram binning and distogram -> distance-matrix centering.

Capability parity: reference utils.py:41-50 (get_bucketed_distance_matrix)
and utils.py:718-761 (center_distogram_torch).  The cdist+bucketize hot
path has a fused HIP kernel on MI355X (ops.distance_buckets); this module
is the always-available eager path and the CPU reference.
"""
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


def _bin_centers(bins, device):
    """Representative distance per distogram class.

    Interior classes use the midpoint below their threshold.  The two
    ends follow the reference's calibration (utils.py:718-761, kept for
    numerics parity with models trained against it): the first class
    maps to 1.5 A and the catch-all above-last class to 1.33x the last
    threshold.
    """
    width = bins[2] - bins[1]
    centers = (bins - 0.5 * width).to(device)
    centers[0] = 1.5
    centers[-1] = 1.33 * bins[-1]
    return centers


def center_distogram_torch(distogram, bins=DISTANCE_THRESHOLDS, min_t=1.,
                           center="mean", wide="std"):
    """Distogram (b, N, N, B) -> (central (b,N,N), weights (b,N,N)).

    Central estimate (probability-weighted mean, or median class) plus a
    0-1 confidence weight from the dispersion; the diagonal and pairs
    landing in the catch-all (beyond-range) class get weight 0.
    """
    from .backend import as_batched
    device = distogram.device
    centers = _bin_centers(bins, device)
    total = distogram.sum(dim=-1)

    if center == "median":
        cdf = torch.cumsum(distogram, dim=-1)
        median_class = torch.searchsorted(cdf, 0.5 * cdf[..., -1:]) \
            .squeeze(-1).clamp(max=centers.shape[0] - 1)
        central = centers[median_class]
    else:
        central = (distogram * centers).sum(dim=-1) / total

    central = as_batched(central, 3)
    eye = torch.arange(central.shape[-1], device=device)
    central[:, eye, eye] = 0.

    in_range = (central <= bins[-2].item()).to(central.dtype)
    if wide in ("var", "std"):
        spread = (distogram * (centers - central.unsqueeze(-1)).square()) \
            .sum(dim=-1) / total
        if wide == "std":
            spread = spread.sqrt()
    else:
        spread = torch.zeros_like(central)

    weights = in_range / (1 + spread)
    weights = torch.nan_to_num(weights, nan=0.0)
    weights[:, eye, eye] = 0.
    return central, weights
