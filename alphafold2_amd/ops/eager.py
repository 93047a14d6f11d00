"""Eager (pure PyTorch) implementations of every hot op.

These are the CPU path, the parity references for the HIP kernels, and
the fallback for ops whose kernel has not landed yet.  Written in plain
fp32-safe torch so numerics tests can compare the gfx950 kernels against
them at fp32.
"""
import torch
import torch.nn.functional as F


def attention_core(q, k, v, bias=None, mask=None, context_mask=None,
                   tie_dim=None, dropout=0., training=False):
    """softmax(q k^T * scale + bias) v with optional key masking.

    * q, k, v: (B, h, n, d) — B may fold extra axes (MSA rows, pair rows)
    * bias:    broadcastable to (B, h, i, j) — pair bias
    * mask:    (B, i) bool query-side mask
    * context_mask: (B, j) bool key-side mask (defaults to `mask`)
    * tie_dim: r — queries are averaged over groups of r consecutive
      batch entries (MSAColumnGlobalAttention); k/v stay per-entry.
    Returns (B, h, n, d).
    """
    scale = q.shape[-1] ** -0.5
    q = q * scale

    if tie_dim is not None:
        Bh = q.shape[0]
        b = Bh // tie_dim
        q = q.reshape(b, tie_dim, *q.shape[1:]).mean(dim=1, keepdim=True)
        k_g = k.reshape(b, tie_dim, *k.shape[1:])
        dots = torch.einsum('b x h i d, b r h j d -> b r h i j', q, k_g)
        dots = dots.reshape(Bh, *dots.shape[2:])
    else:
        dots = torch.einsum('b h i d, b h j d -> b h i j', q, k)

    if bias is not None:
        dots = dots + bias

    if mask is not None:
        i, j = dots.shape[-2], dots.shape[-1]
        cmask = context_mask if context_mask is not None else mask
        mask_value = -torch.finfo(dots.dtype).max
        full = mask[:, None, :, None] * cmask[:, None, None, :]
        dots = dots.masked_fill(~full, mask_value)

    attn = dots.softmax(dim=-1)
    if dropout > 0. and training:
        attn = F.dropout(attn, p=dropout, training=True)
    return torch.einsum('b h i j, b h j d -> b h i d', attn, v)


def geglu(x):
    """GEGLU activation: split the last dim, gate with gelu."""
    a, gates = x.chunk(2, dim=-1)
    return a * F.gelu(gates)


def outer_product_mean(left, right, mask=None, eps=1e-5):
    """MSA -> pair outer-product mean.

    * left, right: (b, m, n, d) projected MSA representations
    * mask: (b, m, n) bool MSA occupancy
    Returns (b, n, n, d): mean over MSA rows of left_i (x) right_j.

    NOTE reference-numerics parity (reference alphafold2.py:341-349): in
    the masked branch the reference divides the row-mean *again* by
    (count + eps), i.e. result = sum_m / (m * (count + eps)).  We keep
    that exact normalization so weights transfer 1:1.
    """
    m = left.shape[1]
    if mask is not None:
        fmask = mask.to(left.dtype)
        left = left * fmask[..., None]
        right = right * fmask[..., None]
        outer_sum = torch.einsum('b m i d, b m j d -> b i j d', left, right)
        count = torch.einsum('b m i, b m j -> b i j', fmask, fmask)
        return outer_sum / m / (count[..., None] + eps)
    outer_sum = torch.einsum('b m i d, b m j d -> b i j d', left, right)
    return outer_sum / m


def triangle_mix(left, right, mix):
    """Triangle multiplicative mixing over the shared k axis.

    * left, right: (b, n, n, d)
    * mix: 'outgoing' -> out[i,j] = sum_k left[i,k] * right[j,k]
           'ingoing'  -> out[i,j] = sum_k left[k,j] * right[k,i]
    Returns (b, n, n, d).  Per-channel n x n GEMM over k.
    """
    if mix == 'outgoing':
        return torch.einsum('b i k d, b j k d -> b i j d', left, right)
    elif mix == 'ingoing':
        return torch.einsum('b k j d, b k i d -> b i j d', left, right)
    raise ValueError(f"mix must be 'ingoing' or 'outgoing', got {mix!r}")


def pair_outer_sum(x_left, x_right):
    """(b, n, d) + (b, n, d) -> (b, n, n, d) broadcast outer sum."""
    return x_left[:, :, None, :] + x_right[:, None, :, :]


def distance_buckets(coords, boundaries):
    """cdist + bucketize: (b, n, 3) -> (b, n, n) long bucket indices."""
    distances = torch.cdist(coords, coords, p=2)
    return torch.bucketize(distances, boundaries)


def layer_norm(x, weight, bias, eps=1e-5):
    return F.layer_norm(x, x.shape[-1:], weight, bias, eps)


def softclamp_gate(x, gates):
    """Sigmoid output gating: x * sigmoid(gates)."""
    return x * gates.sigmoid()
