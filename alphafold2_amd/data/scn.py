"""Offline sidechainnet-format loader.

The reference trains from the `sidechainnet` pip package
(reference train_pre.py:37-43: `scn.load(casp_version=12, thinning=30,
with_pytorch='dataloaders')`, batches exposing one-hot `.seqs`, flat
`.crds`, `.msks`).  This module reads the same RAW data layout from a
local pickle — no network, no sidechainnet dependency — and serves
batches with the same field semantics plus this framework's boolean
masks.

Expected pickle structure (sidechainnet's published format):

    {split: {'seq': [str], 'crd': [ndarray (L*14, 3)],
             'msk': [str of '+'/'-'], 'ang': [ndarray (L, 12)]  # opt
             'ids': [str]}}
"""
import pickle

import numpy as np
import torch
from torch.utils.data import DataLoader, Dataset

from .. import constants
from ..vocab import VOCAB

C = constants.NUM_COORDS_PER_RES


class SCNDataset(Dataset):
    """One split of a sidechainnet-format pickle as model-ready items."""

    def __init__(self, entries, crop_len=256):
        self.crop_len = crop_len
        self.seqs = entries['seq']
        self.crds = entries['crd']
        self.msks = entries.get('msk', None)
        self.angs = entries.get('ang', None)
        self.ids = entries.get('ids', [str(i) for i in range(len(self.seqs))])

    def __len__(self):
        return len(self.seqs)

    def __getitem__(self, idx):
        seq_str = self.seqs[idx]
        L = len(seq_str)
        seq = torch.tensor([VOCAB._char2int.get(c, 20) for c in seq_str],
                           dtype=torch.long)
        crd = torch.as_tensor(
            np.asarray(self.crds[idx], dtype=np.float32)).reshape(L, C, 3)
        if self.msks is not None:
            mask = torch.tensor([c == '+' for c in self.msks[idx]],
                                dtype=torch.bool)
        else:
            mask = torch.ones(L, dtype=torch.bool)
        ang = None
        if self.angs is not None:
            ang = torch.as_tensor(
                np.asarray(self.angs[idx], dtype=np.float32))

        if L > self.crop_len:
            start = torch.randint(0, L - self.crop_len + 1, (1,)).item()
            sl = slice(start, start + self.crop_len)
            seq, crd, mask = seq[sl], crd[sl], mask[sl]
            if ang is not None:
                ang = ang[sl]
        item = {'seq': seq, 'coords': crd, 'mask': mask,
                'id': self.ids[idx]}
        if ang is not None:
            item['angles'] = ang
        return item


def collate_scn(items):
    """Pad a list of SCN items to a (b, L, ...) batch with masks."""
    L = max(it['seq'].shape[0] for it in items)
    b = len(items)
    seq = torch.full((b, L), 20, dtype=torch.long)
    coords = torch.zeros(b, L, C, 3)
    mask = torch.zeros(b, L, dtype=torch.bool)
    has_ang = all('angles' in it for it in items)
    ang = torch.zeros(b, L, items[0]['angles'].shape[-1]) if has_ang else None
    for i, it in enumerate(items):
        n = it['seq'].shape[0]
        seq[i, :n] = it['seq']
        coords[i, :n] = it['coords']
        mask[i, :n] = it['mask']
        if has_ang:
            ang[i, :n] = it['angles']
    out = {'seq': seq, 'coords': coords, 'mask': mask,
           'ids': [it['id'] for it in items],
           # Ca-only view matching the reference's distogram target use
           'ca_coords': coords[:, :, 1]}
    if has_ang:
        out['angles'] = ang
    return out


def load(path, batch_size=8, crop_len=256, num_workers=0, splits=None,
         seed=0):
    """Sidechainnet-style entry point: pickle path -> dict of DataLoaders
    (mirrors `scn.load(..., with_pytorch='dataloaders')`)."""
    with open(path, 'rb') as f:
        raw = pickle.load(f)
    gens = {}
    for split, entries in raw.items():
        if splits is not None and split not in splits:
            continue
        if not isinstance(entries, dict) or 'seq' not in entries:
            continue  # metadata keys (date, settings, ...)
        ds = SCNDataset(entries, crop_len=crop_len)
        g = torch.Generator().manual_seed(seed)
        gens[split] = DataLoader(
            ds, batch_size=batch_size, collate_fn=collate_scn,
            shuffle=split.startswith('train'), generator=g,
            num_workers=num_workers)
    return gens
