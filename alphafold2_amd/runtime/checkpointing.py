"""Checkpoint / resume.

The reference has no persistence at all (no torch.save
anywhere).  This module provides standard state_dict checkpointing with
the reference-compatible parameter layout, plus optimizer / scheduler /
RNG / step state for exact resume, written atomically.
"""
import os
import tempfile

import torch


def save_checkpoint(path, model, optimizer=None, scheduler=None, step=None,
                    extra=None):
    """Atomically write a training checkpoint (rank 0 only under DDP)."""
    from ..parallel import get_rank
    if get_rank() != 0:
        return None
    state = {
        'model': model.state_dict(),
        'optimizer': optimizer.state_dict() if optimizer is not None else None,
        'scheduler': scheduler.state_dict() if scheduler is not None else None,
        'step': step,
        'torch_rng': torch.get_rng_state(),
        'cuda_rng': (torch.cuda.get_rng_state_all()
                     if torch.cuda.is_available() else None),
        'extra': extra,
    }
    d = os.path.dirname(os.path.abspath(path)) or '.'
    os.makedirs(d, exist_ok=True)
    fd, tmp = tempfile.mkstemp(dir=d, suffix='.tmp')
    try:
        with os.fdopen(fd, 'wb') as f:
            torch.save(state, f)
        os.replace(tmp, path)
    except BaseException:
        if os.path.exists(tmp):
            os.unlink(tmp)
        raise
    return path


def load_checkpoint(path, model, optimizer=None, scheduler=None,
                    map_location='cpu', strict=True, restore_rng=True):
    """Load a checkpoint; returns the stored step (or None)."""
    state = torch.load(path, map_location=map_location, weights_only=False)
    model.load_state_dict(state['model'], strict=strict)
    if optimizer is not None and state.get('optimizer') is not None:
        optimizer.load_state_dict(state['optimizer'])
    if scheduler is not None and state.get('scheduler') is not None:
        scheduler.load_state_dict(state['scheduler'])
    if restore_rng and state.get('torch_rng') is not None:
        torch.set_rng_state(state['torch_rng'].cpu().to(torch.uint8))
        if torch.cuda.is_available() and state.get('cuda_rng') is not None:
            try:
                torch.cuda.set_rng_state_all(
                    [s.cpu().to(torch.uint8) for s in state['cuda_rng']])
            except RuntimeError:
                pass  # different device count than at save time
    return state.get('step')


# ---------------------------------------------------------------------------
# standard <-> reversible trunk key mapping

# the two trunk engines hold identical sub-modules under different paths:
#   standard:   net.layers.{i}.layer.{0,1,2,3}.<rest>
#               (0=pairwise attn block, 1=pair FF, 2=msa attn block,
#                3=msa FF — models/evoformer.py EvoformerBlock)
#   reversible: net.blocks.{i}.{pair_attn,pair_ff,msa_attn,msa_ff}.net.<rest>
#               (models/reversible.py ReversibleEvoformerBlock)
_REV_SLOT = {'0': 'pair_attn', '1': 'pair_ff', '2': 'msa_attn',
             '3': 'msa_ff'}
_STD_SLOT = {v: k for k, v in _REV_SLOT.items()}


def convert_trunk_state_dict(state_dict, to):
    """Rewrite trunk keys between the standard and reversible layouts.

    `to` is 'reversible' or 'standard'.  Non-trunk keys pass through
    unchanged, so a checkpoint pretrained with one trunk engine can be
    resumed with the other (the 288 GB sizing lever — VERDICT r01)."""
    import re
    assert to in ('reversible', 'standard')
    out = {}
    if to == 'reversible':
        pat = re.compile(r'^net\.layers\.(\d+)\.layer\.([0-3])\.(.+)$')
        for k, v in state_dict.items():
            m = pat.match(k)
            if m:
                k = (f'net.blocks.{m.group(1)}.{_REV_SLOT[m.group(2)]}'
                     f'.net.{m.group(3)}')
            out[k] = v
    else:
        pat = re.compile(
            r'^net\.blocks\.(\d+)\.(pair_attn|pair_ff|msa_attn|msa_ff)'
            r'\.net\.(.+)$')
        for k, v in state_dict.items():
            m = pat.match(k)
            if m:
                k = (f'net.layers.{m.group(1)}.layer.'
                     f'{_STD_SLOT[m.group(2)]}.{m.group(3)}')
            out[k] = v
    return out
