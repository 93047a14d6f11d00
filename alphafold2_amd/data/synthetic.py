"""Synthetic protein-batch generator.

There is no network in this environment (no sidechainnet / trRosetta
downloads), so training and benchmarking run on synthetic data shaped
exactly like the real pipeline's output: random sequences/MSAs and a
random — but geometrically self-consistent — backbone whose pairwise
distances feed the distogram targets.
"""
import torch

from .. import constants


def _random_chain(b, n, device, step=3.8, generator=None):
    """Random self-avoiding-ish 3D walk with ~3.8 Å CA-CA steps."""
    steps = torch.randn(b, n, 3, device=device, generator=generator)
    steps = steps / steps.norm(dim=-1, keepdim=True) * step
    # smooth the walk so it looks chain-like rather than a hairball
    steps = 0.6 * steps + 0.4 * steps.roll(1, dims=1)
    coords = steps.cumsum(dim=1)
    return coords - coords.mean(dim=1, keepdim=True)


def synthetic_batch(batch_size=1, seq_len=256, msa_depth=128, device='cpu',
                    seed=None, num_buckets=constants.DISTOGRAM_BUCKETS):
    """Returns a dict with seq, msa, masks, CA coords and distogram
    targets, matching the shapes of the real data pipeline."""
    generator = None
    if seed is not None:
        generator = torch.Generator(device=device).manual_seed(seed)
    seq = torch.randint(0, constants.NUM_AMINO_ACIDS,
                        (batch_size, seq_len), device=device,
                        generator=generator)
    msa = torch.randint(0, constants.NUM_AMINO_ACIDS,
                        (batch_size, msa_depth, seq_len), device=device,
                        generator=generator)
    mask = torch.ones_like(seq).bool()
    msa_mask = torch.ones_like(msa).bool()
    coords = _random_chain(batch_size, seq_len, device, generator=generator)

    distances = torch.cdist(coords, coords, p=2)
    boundaries = torch.linspace(constants.DISTOGRAM_MIN_DIST,
                                constants.DISTOGRAM_MAX_DIST,
                                steps=num_buckets, device=device)
    discretized = torch.bucketize(distances, boundaries[:-1])

    return {
        'seq': seq,
        'msa': msa,
        'mask': mask,
        'msa_mask': msa_mask,
        'coords': coords,
        'distogram_target': discretized,
    }


class SyntheticProteinDataset(torch.utils.data.Dataset):
    """Map-style dataset of synthetic batches (one protein per item)."""

    def __init__(self, length=1024, seq_len=256, msa_depth=128, seed=0):
        self.length = length
        self.seq_len = seq_len
        self.msa_depth = msa_depth
        self.seed = seed

    def __len__(self):
        return self.length

    def __getitem__(self, idx):
        item = synthetic_batch(1, self.seq_len, self.msa_depth,
                               seed=self.seed + idx)
        return {k: v[0] for k, v in item.items()}
