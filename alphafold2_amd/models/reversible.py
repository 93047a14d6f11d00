"""Reversible Evoformer execution — O(1) activation memory across depth.

Capability parity: reference reversible.py:26-347 (the advertised
reversible-trunk capability, README.md:40).  Re-designed for the current
two-stream Evoformer: the pair rep (x) and MSA rep (m) are each
channel-duplicated into (x1, x2) / (m1, m2) and every residual update
reads only the *other* half, so the backward pass reconstructs inputs
from outputs instead of storing them:

    m1 += MsaAttn(m2 | pair = x2)      m2 += MsaFF(m1)
    x1 += PairBlock(x2 | msa = m2)     x2 += PairFF(x1)

Dropout / MLM randomness is replayed exactly via CPU+device RNG state
capture (Deterministic wrapper), which on ROCm records the HIP
philox state through the standard torch.cuda generator API.

This is what lets crop_len / MSA depth scale into the MI355X's 288 GB
HBM3E: activation memory stays flat in depth, at the price of one extra
forward recompute during backward (same trade as per-block
checkpointing, but with no stored block inputs at all).
"""
import torch
import torch.nn as nn
from torch.autograd.function import Function
from torch.utils.checkpoint import get_device_states, set_device_states


class Deterministic(nn.Module):
    """Record RNG state on forward; replay it for the recompute."""

    def __init__(self, net):
        super().__init__()
        self.net = net
        self.cpu_state = None
        self.cuda_in_fwd = None
        self.gpu_devices = None
        self.gpu_states = None

    def record_rng(self, *args):
        self.cpu_state = torch.get_rng_state()
        if torch.cuda.is_initialized():
            self.cuda_in_fwd = True
            self.gpu_devices, self.gpu_states = get_device_states(*args)

    def forward(self, *args, record_rng=False, set_rng=False, **kwargs):
        if record_rng:
            self.record_rng(*args)
        if not set_rng:
            return self.net(*args, **kwargs)
        rng_devices = []
        if self.cuda_in_fwd:
            rng_devices = self.gpu_devices
        with torch.random.fork_rng(devices=rng_devices, enabled=True):
            torch.set_rng_state(self.cpu_state)
            if self.cuda_in_fwd:
                set_device_states(self.gpu_devices, self.gpu_states)
            return self.net(*args, **kwargs)


class ReversibleEvoformerBlock(nn.Module):
    """One Evoformer block in reversible form over split channels."""

    def __init__(self, pair_block, pair_ff, msa_block, msa_ff):
        super().__init__()
        self.msa_attn = Deterministic(msa_block)
        self.msa_ff = Deterministic(msa_ff)
        self.pair_attn = Deterministic(pair_block)
        self.pair_ff = Deterministic(pair_ff)

    # ---- forward (no grad; outputs only) ----

    def forward(self, x1, x2, m1, m2, mask=None, msa_mask=None,
                record_rng=False):
        m1 = m1 + self.msa_attn(m2, mask=msa_mask, pairwise_repr=x2,
                                record_rng=record_rng)
        m2 = m2 + self.msa_ff(m1, record_rng=record_rng)
        x1 = x1 + self.pair_attn(x2, mask=mask, msa_repr=m2,
                                 msa_mask=msa_mask, record_rng=record_rng)
        x2 = x2 + self.pair_ff(x1, record_rng=record_rng)
        return x1, x2, m1, m2

    # ---- backward: reconstruct inputs and accumulate grads ----

    def backward_pass(self, y, dy, mask=None, msa_mask=None):
        x1, x2, m1, m2 = y
        dx1, dx2, dm1, dm2 = dy

        # invert x2 += pair_ff(x1)
        with torch.enable_grad():
            x1.requires_grad = True
            fx1 = self.pair_ff(x1, set_rng=True)
            fx1.backward(dx2)
        with torch.no_grad():
            x2 = x2 - fx1
            dx1 = dx1 + x1.grad
            x1.grad = None
            x1 = x1.detach()

        # invert x1 += pair_attn(x2, m2)
        with torch.enable_grad():
            x2.requires_grad = True
            m2.requires_grad = True
            fx2 = self.pair_attn(x2, mask=mask, msa_repr=m2,
                                 msa_mask=msa_mask, set_rng=True)
            fx2.backward(dx1)
        with torch.no_grad():
            x1 = x1 - fx2
            dx2 = dx2 + x2.grad
            dm2 = dm2 + m2.grad
            x2.grad = None
            m2.grad = None
            x2 = x2.detach()
            m2 = m2.detach()

        # invert m2 += msa_ff(m1)
        with torch.enable_grad():
            m1.requires_grad = True
            fm1 = self.msa_ff(m1, set_rng=True)
            fm1.backward(dm2)
        with torch.no_grad():
            m2 = m2 - fm1
            dm1 = dm1 + m1.grad
            m1.grad = None
            m1 = m1.detach()

        # invert m1 += msa_attn(m2, x2)
        with torch.enable_grad():
            m2.requires_grad = True
            x2.requires_grad = True
            fm2 = self.msa_attn(m2, mask=msa_mask, pairwise_repr=x2,
                                set_rng=True)
            fm2.backward(dm1)
        with torch.no_grad():
            m1 = m1 - fm2
            dm2 = dm2 + m2.grad
            dx2 = dx2 + x2.grad
            m2.grad = None
            x2.grad = None
            m2 = m2.detach()
            x2 = x2.detach()

        return (x1, x2, m1, m2), (dx1, dx2, dm1, dm2)


class _ReversibleFunction(Function):
    @staticmethod
    def forward(ctx, x1, x2, m1, m2, blocks, mask, msa_mask):
        ctx.blocks = blocks
        ctx.mask = mask
        ctx.msa_mask = msa_mask
        # the reconstruction passes in backward must recompute under the
        # same autocast regime as this forward, or dtypes diverge
        ctx.amp_enabled = torch.is_autocast_enabled()
        ctx.amp_dtype = torch.get_autocast_dtype('cuda') \
            if ctx.amp_enabled else None
        with torch.no_grad():
            for block in blocks:
                x1, x2, m1, m2 = block(
                    x1, x2, m1, m2, mask=mask, msa_mask=msa_mask,
                    record_rng=block.training)
        ctx.y = (x1.detach(), x2.detach(), m1.detach(), m2.detach())
        return x1, x2, m1, m2

    @staticmethod
    def backward(ctx, dx1, dx2, dm1, dm2):
        import contextlib
        amp = torch.autocast('cuda', dtype=ctx.amp_dtype) \
            if ctx.amp_enabled else contextlib.nullcontext()
        y = ctx.y
        dy = (dx1, dx2, dm1, dm2)
        with amp:
            for block in reversed(ctx.blocks):
                y, dy = block.backward_pass(y, dy, mask=ctx.mask,
                                            msa_mask=ctx.msa_mask)
        return (*dy, None, None, None)


class ReversibleEvoformer(nn.Module):
    """Drop-in trunk executor: duplicate channels in, run reversible
    blocks, mean-reduce the halves out."""

    def __init__(self, blocks):
        super().__init__()
        self.blocks = nn.ModuleList(blocks)

    def forward(self, x, m, mask=None, msa_mask=None):
        if not (self.training and torch.is_grad_enabled()):
            # inference: plain execution, no reversibility machinery
            x1, x2, m1, m2 = x, x.clone(), m, m.clone()
            for block in self.blocks:
                x1, x2, m1, m2 = block(x1, x2, m1, m2, mask=mask,
                                       msa_mask=msa_mask)
            return 0.5 * (x1 + x2), 0.5 * (m1 + m2)

        x1, x2, m1, m2 = _ReversibleFunction.apply(
            x, x.clone(), m, m.clone(), list(self.blocks), mask, msa_mask)
        return 0.5 * (x1 + x2), 0.5 * (m1 + m2)


def make_reversible_evoformer(dim, depth, seq_len, heads, dim_head,
                              attn_dropout=0., ff_dropout=0.,
                              global_column_attn=False):
    """Build a ReversibleEvoformer with the same sub-modules as the
    standard Evoformer block (state-dict keys differ by design: the
    reversible trunk is a distinct execution engine)."""
    from .evoformer import (FeedForward, MsaAttentionBlock,
                            PairwiseAttentionBlock)
    blocks = []
    for _ in range(depth):
        pair_block = PairwiseAttentionBlock(
            dim=dim, seq_len=seq_len, heads=heads, dim_head=dim_head,
            dropout=attn_dropout, global_column_attn=global_column_attn)
        pair_ff = FeedForward(dim=dim, dropout=ff_dropout)
        msa_block = MsaAttentionBlock(
            dim=dim, seq_len=seq_len, heads=heads, dim_head=dim_head,
            dropout=attn_dropout)
        msa_ff = FeedForward(dim=dim, dropout=ff_dropout)
        blocks.append(ReversibleEvoformerBlock(
            pair_block, pair_ff, msa_block, msa_ff))
    return ReversibleEvoformer(blocks)
