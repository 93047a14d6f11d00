"""Equivariant structure-module alternatives: EGNN and SE(3)-style
attention refinement.

BASELINE.json's workload configs include structure_module_type 'se3' and
'egnn' (historical capability of the reference line: earlier versions of
lucidrains/alphafold2 wired SE3-Transformer / EGNN / En-transformer
structure modules before the IPA rewrite).  These are native
implementations — pairwise-distance featurisation + equivariant message
passing / attention over backbone atoms, fp32 (equivariance):

* EGNN  (Satorras et al. 2021 style): h_i updated from messages
  m_ij = phi_e(h_i, h_j, |x_i - x_j|^2, e_ij); coordinates updated along
  displacement vectors with phi_x(m_ij) weights — E(n)-equivariant.
* SE3Refiner: same displacement-vector coordinate update but messages
  are attention-weighted (per-head softmax over j with pair bias), i.e.
  an SE(3)-equivariant transformer layer specialised to the single
  backbone point per residue.

Both consume (single_repr, pairwise_repr, coords, mask) and return
updated (single_repr, coords), and plug into Alphafold2's refinement
loop behind `structure_module_type`.
"""
import torch
from torch import nn


class EGNNLayer(nn.Module):
    def __init__(self, dim, edge_dim=0, m_dim=64, eps=1e-8,
                 clamp_coors=50.0):
        super().__init__()
        self.eps = eps
        self.clamp_coors = clamp_coors
        edge_input_dim = dim * 2 + edge_dim + 1

        self.edge_mlp = nn.Sequential(
            nn.Linear(edge_input_dim, m_dim),
            nn.SiLU(),
            nn.Linear(m_dim, m_dim),
            nn.SiLU(),
        )
        self.node_mlp = nn.Sequential(
            nn.Linear(dim + m_dim, dim * 2),
            nn.SiLU(),
            nn.Linear(dim * 2, dim),
        )
        self.coors_mlp = nn.Sequential(
            nn.Linear(m_dim, m_dim),
            nn.SiLU(),
            nn.Linear(m_dim, 1),
        )
        nn.init.zeros_(self.coors_mlp[-1].weight)
        nn.init.zeros_(self.coors_mlp[-1].bias)

    def forward(self, h, x, edges=None, mask=None):
        b, n, d = h.shape
        rel = x[:, :, None, :] - x[:, None, :, :]          # (b, n, n, 3)
        dist2 = rel.pow(2).sum(dim=-1, keepdim=True)       # (b, n, n, 1)

        hi = h[:, :, None, :].expand(b, n, n, d)
        hj = h[:, None, :, :].expand(b, n, n, d)
        feats = [hi, hj, dist2]
        if edges is not None:
            feats.append(edges)
        m_ij = self.edge_mlp(torch.cat(feats, dim=-1))     # (b, n, n, m)

        if mask is not None:
            pair_mask = (mask[:, :, None] & mask[:, None, :]).unsqueeze(-1)
            m_ij = m_ij * pair_mask

        # E(n)-equivariant coordinate update along normalized displacements
        coor_w = self.coors_mlp(m_ij)                      # (b, n, n, 1)
        rel_norm = rel / (dist2 + self.eps).sqrt()
        dx = (rel_norm * coor_w).sum(dim=2)
        if self.clamp_coors:
            dx = dx.clamp(-self.clamp_coors, self.clamp_coors)
        x = x + dx

        m_i = m_ij.sum(dim=2)                              # (b, n, m)
        h = h + self.node_mlp(torch.cat((h, m_i), dim=-1))
        return h, x


class SE3RefinerLayer(nn.Module):
    """Attention-weighted equivariant update: per-head softmax over j
    with pair bias and distance features; value messages update h, and
    attention-weighted displacement vectors update x (rotation/
    translation equivariant by construction)."""

    def __init__(self, dim, heads=4, dim_head=32, edge_dim=None, eps=1e-8):
        super().__init__()
        inner = heads * dim_head
        self.heads = heads
        self.scale = dim_head ** -0.5
        self.eps = eps
        edge_dim = edge_dim if edge_dim is not None else dim

        self.norm = nn.LayerNorm(dim)
        self.to_q = nn.Linear(dim, inner, bias=False)
        self.to_k = nn.Linear(dim, inner, bias=False)
        self.to_v = nn.Linear(dim, inner, bias=False)
        self.edge_bias = nn.Linear(edge_dim + 1, heads)
        self.to_out = nn.Linear(inner, dim)
        self.coors_head = nn.Linear(dim, heads)
        nn.init.zeros_(self.coors_head.weight)
        nn.init.zeros_(self.coors_head.bias)
        self.ff = nn.Sequential(
            nn.Linear(dim, dim * 2), nn.SiLU(), nn.Linear(dim * 2, dim))

    def forward(self, h, x, edges=None, mask=None):
        b, n, _ = h.shape
        hh = self.norm(h)

        def split(t):
            return t.reshape(b, n, self.heads, -1).permute(0, 2, 1, 3)

        q, k, v = split(self.to_q(hh)), split(self.to_k(hh)), split(self.to_v(hh))

        rel = x[:, :, None, :] - x[:, None, :, :]
        dist2 = rel.pow(2).sum(dim=-1, keepdim=True)

        bias_in = torch.cat([edges, dist2], dim=-1) if edges is not None \
            else torch.cat([torch.zeros(b, n, n, self.edge_bias.in_features - 1,
                                        device=h.device, dtype=h.dtype),
                            dist2], dim=-1)
        bias = self.edge_bias(bias_in).permute(0, 3, 1, 2)   # (b, H, n, n)

        dots = (q * self.scale) @ k.transpose(-1, -2) + bias
        if mask is not None:
            pm = mask[:, None, None, :]
            dots = dots.masked_fill(~pm, -torch.finfo(dots.dtype).max)
        attn = dots.softmax(dim=-1)                          # (b, H, n, n)

        out = (attn @ v).permute(0, 2, 1, 3).reshape(b, n, -1)
        h = h + self.to_out(out)

        # equivariant coordinate update: attention-weighted displacements
        rel_norm = rel / (dist2 + self.eps).sqrt()
        cw = self.coors_head(hh)                             # (b, n, H)
        dx = torch.einsum('b h i j, b i j c -> b i h c',
                          attn, rel_norm)                    # (b, n, H, 3)
        x = x + (dx * cw.unsqueeze(-1)).sum(dim=2)

        h = h + self.ff(self.norm(h))
        return h, x


class EquivariantStructureModule(nn.Module):
    """Iterative refinement head used when structure_module_type is
    'egnn' or 'se3': starts coordinates at the origin (or recycled
    coords) and applies `depth` equivariant layers."""

    def __init__(self, dim, depth=4, kind='egnn', heads=4, dim_head=32):
        super().__init__()
        assert kind in ('egnn', 'se3')
        self.kind = kind
        if kind == 'egnn':
            self.layers = nn.ModuleList(
                [EGNNLayer(dim, edge_dim=dim) for _ in range(depth)])
        else:
            self.layers = nn.ModuleList(
                [SE3RefinerLayer(dim, heads=heads, dim_head=dim_head,
                                 edge_dim=dim) for _ in range(depth)])
        self.to_points = nn.Linear(dim, 3)

    def forward(self, single_repr, pairwise_repr, mask=None, coords=None):
        h = single_repr
        if coords is None:
            # symmetry-broken init from the representation itself
            x = self.to_points(h)
        else:
            x = coords
        for layer in self.layers:
            h, x = layer(h, x, edges=pairwise_repr, mask=mask)
        return h, x
