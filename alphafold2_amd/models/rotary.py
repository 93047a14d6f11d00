"""Optional positional-encoding variants: fixed sinusoidal, rotary, and
2D axial rotary for the pair grid.

Capability parity: reference rotary.py:9-67 (kept as optional variants;
the main model uses the learned relative-position embedding).  The
reference's DepthWiseConv1d had a broken `default` import — fixed here.
"""
import torch
from torch import nn


def rotate_every_two(x):
    x1 = x[..., ::2]
    x2 = x[..., 1::2]
    x = torch.stack((-x2, x1), dim=-1)
    return x.flatten(start_dim=-2)


def apply_rotary_pos_emb(q, k, sinu_pos):
    sin, cos = sinu_pos
    q = (q * cos) + (rotate_every_two(q) * sin)
    k = (k * cos) + (rotate_every_two(k) * sin)
    return q, k


class DepthWiseConv1d(nn.Module):
    def __init__(self, dim_in, dim_out, kernel_size, groups=None,
                 padding=0, stride=1, bias=True):
        super().__init__()
        groups = groups if groups is not None else dim_in
        self.net = nn.Sequential(
            nn.Conv1d(dim_in, dim_in, kernel_size=kernel_size,
                      padding=padding, groups=groups, stride=stride,
                      bias=bias),
            nn.Conv1d(dim_in, dim_out, 1, bias=bias),
        )

    def forward(self, x):
        return self.net(x)


class FixedPositionalEmbedding(nn.Module):
    def __init__(self, dim):
        super().__init__()
        inv_freq = 1. / (10000 ** (torch.arange(0, dim, 2).float() / dim))
        self.register_buffer('inv_freq', inv_freq)

    def forward(self, n, device):
        seq = torch.arange(n, device=device).type_as(self.inv_freq)
        freqs = torch.einsum('i, j -> i j', seq, self.inv_freq)
        freqs = torch.repeat_interleave(freqs, 2, dim=-1)
        return [freqs.sin(), freqs.cos()]


class AxialRotaryEmbedding(nn.Module):
    """2D (i, j) rotary embedding for attention over the pair grid."""

    def __init__(self, dim, max_freq=10):
        super().__init__()
        self.dim = dim
        inv_freq = 1. / (10000 ** (torch.arange(0, dim // 2, 2).float()
                                   / (dim // 2)))
        self.register_buffer('inv_freq', inv_freq)

    def forward(self, n, device):
        seq = torch.arange(n, device=device).type_as(self.inv_freq)
        freqs = torch.einsum('i, j -> i j', seq, self.inv_freq)
        freqs = torch.repeat_interleave(freqs, 2, dim=-1)

        freqs_i = freqs[:, None, :].expand(n, n, -1)
        freqs_j = freqs[None, :, :].expand(n, n, -1)
        furthest = torch.cat((freqs_i, freqs_j), dim=-1).reshape(n * n, -1)
        return [furthest.sin(), furthest.cos()]
