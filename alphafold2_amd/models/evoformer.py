"""Evoformer trunk: gated attention, axial (row/col + triangle) attention,
triangle multiplicative update, outer-product mean, GEGLU transitions.

Capability parity: reference alphafold2.py:69-467 — same module/parameter
layout (state-dict compatible), same math.  Every hot loop is routed
through `alphafold2_amd.ops`, where the gfx950 HIP kernels live
(K1-K6/K13 of SURVEY.md §2.17); the nn.Linear projections hit hipBLASLt
through PyTorch-ROCm.
"""
import torch
import torch.nn.functional as F
from torch import nn
from torch.utils.checkpoint import checkpoint

from .. import ops
from ..ops.fused_modules import FusedLayerNorm


def exists(val):
    return val is not None


def default(val, d):
    if exists(val):
        return val
    return d() if callable(d) else d


def cast_tuple(val, depth=1):
    return val if isinstance(val, tuple) else (val,) * depth


def init_zero_(layer):
    nn.init.constant_(layer.weight, 0.)
    if exists(layer.bias):
        nn.init.constant_(layer.bias, 0.)


class Always(nn.Module):
    def __init__(self, val):
        super().__init__()
        self.val = val

    def forward(self, x):
        return self.val


# ---------------------------------------------------------------------------
# feed forward (GEGLU transition)


class GEGLU(nn.Module):
    def forward(self, x):
        return ops.geglu(x)


class FeedForward(nn.Module):
    """Pre-norm GEGLU MLP: LN -> Linear(d, 8d) -> GEGLU -> Linear(4d, d).

    Zero-init on the output projection so blocks start as identity.
    """

    def __init__(self, dim, mult=4, dropout=0.):
        super().__init__()
        self.norm = FusedLayerNorm(dim)
        self.net = nn.Sequential(
            nn.Linear(dim, dim * mult * 2),
            GEGLU(),
            nn.Dropout(dropout),
            nn.Linear(dim * mult, dim),
        )
        init_zero_(self.net[-1])

    def forward(self, x, residual=None, **kwargs):
        """`residual`, when given, is added inside the output GEMM's
        epilogue (saves a full elementwise pass over the tensor)."""
        x = self.norm(x)
        h = ops.ff1_geglu(x, self.net[0].weight, self.net[0].bias)
        h = self.net[2](h)
        return ops.fused_linear(h, self.net[3].weight, self.net[3].bias,
                                residual=residual)


# ---------------------------------------------------------------------------
# gated attention core


class Attention(nn.Module):
    """Multi-head attention with sigmoid output gating, optional additive
    pair bias, cross-attention context, and tied-query ("global column")
    mode.  The softmax core runs through ops.attention_core (fused HIP
    flash kernel on gfx950)."""

    def __init__(self, dim, seq_len=None, heads=8, dim_head=64, dropout=0.,
                 gating=True):
        super().__init__()
        inner_dim = dim_head * heads
        self.seq_len = seq_len
        self.heads = heads
        self.dim_head = dim_head

        self.to_q = nn.Linear(dim, inner_dim, bias=False)
        self.to_kv = nn.Linear(dim, inner_dim * 2, bias=False)
        self.to_out = nn.Linear(inner_dim, dim)

        self.gating = nn.Linear(dim, inner_dim)
        nn.init.constant_(self.gating.weight, 0.)
        nn.init.constant_(self.gating.bias, 1.)

        self.dropout_p = dropout
        self.dropout = nn.Dropout(dropout)
        init_zero_(self.to_out)

    def forward(self, x, mask=None, attn_bias=None, context=None,
                context_mask=None, tie_dim=None, attn_bias_repeat=1):
        h = self.heads
        has_context = exists(context)
        context = default(context, x)

        if not has_context:
            # self-attention: one fused GEMM for q, k, v and the gate
            # (weights stay as separate parameters — reference layout —
            # concatenated per call; the cat is ~1 MB vs 4 GEMM launches)
            w = torch.cat([self.to_q.weight, self.to_kv.weight,
                           self.gating.weight], dim=0)
            bias_cat = torch.cat([
                torch.zeros(self.to_q.weight.shape[0] * 3,
                            device=x.device, dtype=self.gating.bias.dtype),
                self.gating.bias])
            inner = self.to_q.weight.shape[0]
            fused = F.linear(x, w, bias_cat)

            if tie_dim is None and not (self.dropout_p > 0 and self.training):
                # packed fast path: the fused kernel consumes q/k/v as
                # strided slices and writes one packed grad (no
                # split-backward concatenation)
                out = ops.attention_core_packed(
                    fused, h, inner, bias=attn_bias, mask=mask,
                    bias_repeat=attn_bias_repeat)
                if out is not None:
                    out = out.transpose(-2, -3).reshape(*x.shape[:-1], -1)
                    gates = fused.narrow(-1, 3 * inner, inner)
                    out = ops.softclamp_gate(out, gates)
                    return ops.fused_linear(out, self.to_out.weight,
                                            self.to_out.bias)

            q, k, v, gates = fused.split([inner, inner, inner, inner], dim=-1)
        else:
            q = self.to_q(x)
            k, v = self.to_kv(context).chunk(2, dim=-1)
            gates = None

        def split_heads(t):
            return t.reshape(*t.shape[:-1], h, -1).transpose(-2, -3)

        q, k, v = map(split_heads, (q, k, v))  # (B, h, n, d)

        if exists(mask) and has_context and not exists(context_mask):
            context_mask = torch.ones(
                1, k.shape[-2], device=k.device).bool()

        out = ops.attention_core(
            q, k, v, bias=attn_bias, mask=mask, context_mask=context_mask,
            tie_dim=tie_dim, bias_repeat=attn_bias_repeat,
            dropout=self.dropout_p, training=self.training)

        out = out.transpose(-2, -3).reshape(*x.shape[:-1], -1)

        # sigmoid output gating (init to identity)
        if gates is None:
            gates = self.gating(x)
        out = ops.softclamp_gate(out, gates)
        return ops.fused_linear(out, self.to_out.weight, self.to_out.bias)


class AxialAttention(nn.Module):
    """Row- or column-attention on a (b, h, w, d) grid by folding the
    orthogonal axis into batch.  Used for MSA row/col attention and for
    both triangle self-attentions (rows = around-start-node, cols =
    around-end-node), with the pair rep projected to a per-head bias."""

    def __init__(self, dim, heads, row_attn=True, col_attn=True,
                 accept_edges=False, global_query_attn=False, **kwargs):
        super().__init__()
        assert not (not row_attn and not col_attn), \
            'row or column attention must be turned on'
        self.row_attn = row_attn
        self.col_attn = col_attn
        self.global_query_attn = global_query_attn

        self.norm = FusedLayerNorm(dim)
        self.attn = Attention(dim=dim, heads=heads, **kwargs)

        self.edges_to_attn_bias = nn.Sequential(
            nn.Linear(dim, heads, bias=False)) if accept_edges else None

    def forward(self, x, edges=None, mask=None):
        assert self.row_attn ^ self.col_attn, \
            'has to be either row or column attention, but not both'
        b, h, w, d = x.shape

        x = self.norm(x)

        if self.col_attn:
            # fold width into batch, attend along height
            axial_dim = w
            inp = x.permute(0, 2, 1, 3).reshape(b * w, h, d)
            m = mask.permute(0, 2, 1).reshape(b * w, h) if exists(mask) else None
        else:
            # fold height into batch, attend along width
            axial_dim = h
            inp = x.reshape(b * h, w, d)
            m = mask.reshape(b * h, w) if exists(mask) else None

        # pair-rep bias: kept at (b, heads, i, j); the fold over the
        # axial dim is a broadcast handled inside the attention core
        # (the eager reference materializes (b*axial, h, i, j) —
        # reference alphafold2.py:248)
        attn_bias = None
        if exists(self.edges_to_attn_bias) and exists(edges):
            bias = self.edges_to_attn_bias[0](edges)         # (b, i, j, heads)
            attn_bias = bias.permute(0, 3, 1, 2).contiguous()

        tie_dim = axial_dim if self.global_query_attn else None

        out = self.attn(inp, mask=m, attn_bias=attn_bias, tie_dim=tie_dim,
                        attn_bias_repeat=axial_dim if exists(attn_bias) else 1)

        if self.col_attn:
            out = out.reshape(b, w, h, d).permute(0, 2, 1, 3)
        else:
            out = out.reshape(b, h, w, d)
        return out


# ---------------------------------------------------------------------------
# triangle multiplicative update


class TriangleMultiplicativeModule(nn.Module):
    """AF2 triangle multiplicative update (outgoing/ingoing): per-channel
    n x n GEMM over the shared k axis with three sigmoid gates
    (identity-init) and a post-norm.  Mixing einsum runs through
    ops.triangle_mix (K3 fused kernel target)."""

    def __init__(self, *, dim, hidden_dim=None, mix='ingoing'):
        super().__init__()
        assert mix in {'ingoing', 'outgoing'}, \
            'mix must be either ingoing or outgoing'
        hidden_dim = default(hidden_dim, dim)
        self.mix = mix
        self.norm = FusedLayerNorm(dim)

        self.left_proj = nn.Linear(dim, hidden_dim)
        self.right_proj = nn.Linear(dim, hidden_dim)

        self.left_gate = nn.Linear(dim, hidden_dim)
        self.right_gate = nn.Linear(dim, hidden_dim)
        self.out_gate = nn.Linear(dim, hidden_dim)

        for gate in (self.left_gate, self.right_gate, self.out_gate):
            nn.init.constant_(gate.weight, 0.)
            nn.init.constant_(gate.bias, 1.)

        self.to_out_norm = FusedLayerNorm(hidden_dim)
        self.to_out = nn.Linear(hidden_dim, dim)

    def forward(self, x, mask=None):
        assert x.shape[1] == x.shape[2], 'feature map must be symmetrical'

        x = self.norm(x)

        # one fused GEMM for left/right projections and all three gates
        w = torch.cat([self.left_proj.weight, self.right_proj.weight,
                       self.left_gate.weight, self.right_gate.weight,
                       self.out_gate.weight], dim=0)
        bias_cat = torch.cat([self.left_proj.bias, self.right_proj.bias,
                              self.left_gate.bias, self.right_gate.bias,
                              self.out_gate.bias])
        hdim = self.left_proj.weight.shape[0]
        fused = F.linear(x, w, bias_cat)
        # packed gated projections: the pair mask folds into the gate
        # kernel, and the backward writes both gatemul gradients into
        # one packed buffer (no SplitBackward concatenation)
        left, right, og = ops.tri_proj_gates(fused, hdim, row_mask=mask)

        out = ops.triangle_mix(left, right, self.mix)

        out = self.to_out_norm(out)
        out = ops.softclamp_gate(out, og)
        return ops.fused_linear(out, self.to_out.weight, self.to_out.bias)


# ---------------------------------------------------------------------------
# outer-product mean (MSA -> pair)


class OuterMean(nn.Module):
    def __init__(self, dim, hidden_dim=None, eps=1e-5):
        super().__init__()
        self.eps = eps
        self.norm = FusedLayerNorm(dim)
        hidden_dim = default(hidden_dim, dim)
        self.left_proj = nn.Linear(dim, hidden_dim)
        self.right_proj = nn.Linear(dim, hidden_dim)
        self.proj_out = nn.Linear(hidden_dim, dim)

    def forward(self, x, mask=None):
        x = self.norm(x)
        w = torch.cat([self.left_proj.weight, self.right_proj.weight], dim=0)
        bias_cat = torch.cat([self.left_proj.bias, self.right_proj.bias])
        hdim = self.left_proj.weight.shape[0]
        left, right = F.linear(x, w, bias_cat).split([hdim, hdim], dim=-1)
        outer = ops.outer_product_mean(left, right, mask=mask, eps=self.eps)
        return ops.fused_linear(outer, self.proj_out.weight,
                                self.proj_out.bias)


# ---------------------------------------------------------------------------
# evoformer blocks


class PairwiseAttentionBlock(nn.Module):
    def __init__(self, dim, seq_len, heads, dim_head, dropout=0.,
                 global_column_attn=False):
        super().__init__()
        self.outer_mean = OuterMean(dim)

        self.triangle_attention_outgoing = AxialAttention(
            dim=dim, heads=heads, dim_head=dim_head,
            row_attn=True, col_attn=False, accept_edges=True)
        self.triangle_attention_ingoing = AxialAttention(
            dim=dim, heads=heads, dim_head=dim_head,
            row_attn=False, col_attn=True, accept_edges=True,
            global_query_attn=global_column_attn)
        self.triangle_multiply_outgoing = TriangleMultiplicativeModule(
            dim=dim, mix='outgoing')
        self.triangle_multiply_ingoing = TriangleMultiplicativeModule(
            dim=dim, mix='ingoing')

    def forward(self, x, mask=None, msa_repr=None, msa_mask=None):
        if exists(msa_repr):
            x = x + self.outer_mean(msa_repr, mask=msa_mask)

        x = self.triangle_multiply_outgoing(x, mask=mask) + x
        x = self.triangle_multiply_ingoing(x, mask=mask) + x
        x = self.triangle_attention_outgoing(x, edges=x, mask=mask) + x
        x = self.triangle_attention_ingoing(x, edges=x, mask=mask) + x
        return x


class MsaAttentionBlock(nn.Module):
    def __init__(self, dim, seq_len, heads, dim_head, dropout=0.):
        super().__init__()
        self.row_attn = AxialAttention(
            dim=dim, heads=heads, dim_head=dim_head,
            row_attn=True, col_attn=False, accept_edges=True)
        self.col_attn = AxialAttention(
            dim=dim, heads=heads, dim_head=dim_head,
            row_attn=False, col_attn=True)

    def forward(self, x, mask=None, pairwise_repr=None):
        x = self.row_attn(x, mask=mask, edges=pairwise_repr) + x
        x = self.col_attn(x, mask=mask) + x
        return x


class EvoformerBlock(nn.Module):
    def __init__(self, *, dim, seq_len, heads, dim_head, attn_dropout,
                 ff_dropout, global_column_attn=False, checkpoint_ffs=False):
        super().__init__()
        self.layer = nn.ModuleList([
            PairwiseAttentionBlock(dim=dim, seq_len=seq_len, heads=heads,
                                   dim_head=dim_head, dropout=attn_dropout,
                                   global_column_attn=global_column_attn),
            FeedForward(dim=dim, dropout=ff_dropout),
            MsaAttentionBlock(dim=dim, seq_len=seq_len, heads=heads,
                              dim_head=dim_head, dropout=attn_dropout),
            FeedForward(dim=dim, dropout=ff_dropout),
        ])
        # selective checkpointing: recompute only the FF transitions —
        # their 8x-dim GEGLU hiddens dominate activation memory, and the
        # recompute is just 2 GEMMs (vs a whole block for
        # checkpoint_blocks=True).  ~17x cheaper recompute per byte
        # saved than block checkpointing for a mid-size memory win.
        self.checkpoint_ffs = checkpoint_ffs
        self.ff_dropout_p = ff_dropout

    def forward(self, inputs):
        x, m, mask, msa_mask = inputs
        attn, ff, msa_attn, msa_ff = self.layer

        ckpt_ff = self.checkpoint_ffs and self.training \
            and torch.is_grad_enabled()

        def run_ff(f, t):
            # residual folded into the FF output GEMM's epilogue
            if ckpt_ff:
                return checkpoint(lambda u: f(u, residual=u), t,
                                  use_reentrant=False,
                                  preserve_rng_state=self.ff_dropout_p > 0)
            return f(t, residual=t)

        # MSA attention and transition
        m = msa_attn(m, mask=msa_mask, pairwise_repr=x)
        m = run_ff(msa_ff, m)

        # pairwise attention and transition
        x = attn(x, mask=mask, msa_repr=m, msa_mask=msa_mask)
        x = run_ff(ff, x)

        return x, m, mask, msa_mask


class Evoformer(nn.Module):
    """Stack of EvoformerBlocks with per-block activation checkpointing
    as the default memory strategy during training (the reference's
    approach — alphafold2.py:466); `checkpoint_blocks=False` disables it
    (e.g. when the reversible trunk handles memory instead).

    RNG state is only stashed per checkpoint when dropout is active —
    the stash is a host sync that would break hipGraph step capture."""

    def __init__(self, *, depth, checkpoint_blocks=True, **kwargs):
        super().__init__()
        # checkpoint_blocks: True = per-block (reference behavior),
        # False = none, 'ff' = selective FF-only checkpointing
        self.layers = nn.ModuleList(
            [EvoformerBlock(checkpoint_ffs=(checkpoint_blocks == 'ff'),
                            **kwargs) for _ in range(depth)])
        self.checkpoint_blocks = checkpoint_blocks is True
        self.preserve_rng_state = (kwargs.get('attn_dropout', 0.) > 0
                                   or kwargs.get('ff_dropout', 0.) > 0)

    def forward(self, x, m, mask=None, msa_mask=None):
        inp = (x, m, mask, msa_mask)
        use_ckpt = self.checkpoint_blocks and self.training \
            and torch.is_grad_enabled()
        for layer in self.layers:
            if use_ckpt:
                inp = checkpoint(layer, inp, use_reentrant=False,
                                 preserve_rng_state=self.preserve_rng_state)
            else:
                inp = layer(inp)
        x, m, *_ = inp
        return x, m
