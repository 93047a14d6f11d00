"""Graph utilities: padded batch -> flat graph, adjacency powers, covalent
bonds.  Capability parity: reference utils.py:497-650."""
import torch

from .. import constants
from ..vocab import VOCAB


def mat_input_to_masked(x, x_mask=None, edges_mat=None, edges=None,
                        edge_mask=None, edge_attr_mat=None, edge_attr=None):
    """Strip padding from a (batched) node/edge representation.

    * x: ((b), N, D) node features; x_mask ((b), N) keeps
    * one of edges (2, E) or edges_mat ((b), N, N) must be given
    Returns (x, edge_index, edge_attr, batch) in flat PyG-style form.
    """
    if len(x.shape) == 3:
        batch_dim = x.shape[1]
        x = x.reshape(-1, *x.shape[2:])
        if x_mask is not None:
            x_mask = x_mask.reshape(-1, *x_mask.shape[2:])
        else:
            x_mask = torch.ones_like(x[..., 0]).bool()
        if edges_mat is not None and edges is None:
            edges = torch.nonzero(edges_mat, as_tuple=False).t()
            edges = edges[1:] + edges[:1] * batch_dim
        batch = (torch.arange(x.shape[0], device=x.device) // batch_dim)[x_mask]
    else:
        if edges_mat is not None and edges is None:
            edges = torch.nonzero(edges_mat, as_tuple=False).t()
        batch = torch.zeros(x.shape[0], device=x.device)

    if edge_attr_mat is not None and edge_attr is None:
        edge_attr = edge_attr_mat[edges_mat.bool()]
    if edge_mask is None:
        edge_mask = torch.ones_like(edges[-1]).bool()

    x = x[x_mask]
    max_num = edges.max().item() + 1
    wrapper = torch.zeros(max_num, max_num, device=x.device)
    wrapper[edges[0][edge_mask], edges[1][edge_mask]] = 1
    wrapper = wrapper[x_mask, :][:, x_mask]
    edge_index = torch.nonzero(wrapper, as_tuple=False).t()
    edge_attr = edge_attr[edge_mask] if edge_attr is not None else None
    return x, edge_index, edge_attr, batch


def nth_deg_adjacency(adj_mat, n=1, sparse=False):
    """n-th degree adjacency: (new_adj_mat, attr_mat) where attr encodes
    the hop count at which each pair first becomes connected."""
    adj_mat = adj_mat.float()
    attr_mat = torch.zeros_like(adj_mat)
    new_adj_mat = adj_mat.clone()
    for i in range(n):
        if i == 0:
            attr_mat += adj_mat
            continue
        new_adj_mat = (new_adj_mat @ adj_mat).bool().float()
        attr_mat.masked_fill_(
            (new_adj_mat - attr_mat.bool().float()).bool(), i + 1)
    return new_adj_mat, attr_mat


def prot_covalent_bond(seqs, adj_degree=1, cloud_mask=None, mat=True,
                       sparse=False):
    """Covalent-bond adjacency of a protein in the 14-atom scn layout.

    * seqs: (b, n) long residue ids
    Returns (edge_idxs, edge_types) or boolean/attr matrices if mat=True.
    """
    device = seqs.device
    C = constants.NUM_COORDS_PER_RES
    adj_mat = torch.zeros(seqs.shape[0], seqs.shape[1] * C, seqs.shape[1] * C)
    seq_list = seqs.cpu().tolist()
    attr_mat = None
    for s, seq in enumerate(seq_list):
        next_idx = 0
        for i, idx in enumerate(seq):
            aa_bonds = constants.AA_DATA[VOCAB._int2char[idx]]['bonds']
            if len(aa_bonds) == 0:
                break  # padding: end of chain
            # last atom index of this residue's bond graph
            next_aa = max(aa_bonds, key=lambda x: max(x))[-1]
            # intra-residue bonds plus the C -> next-N peptide bond
            bonds = next_idx + torch.tensor(aa_bonds + [[2, next_aa]]).t()
            next_idx += next_aa
            if i == seqs.shape[1] - 1:
                bonds = bonds[:, :-1]
            adj_mat[s, bonds[0], bonds[1]] = 1
        adj_mat[s] = adj_mat[s] + adj_mat[s].t()
    # power the adjacency ONCE over the whole batch (applying it inside
    # the per-item loop would re-power earlier items' adjacency at every
    # later item — wrong hop attributes for batch > 1, adj_degree >= 2)
    adj_mat, attr_mat = nth_deg_adjacency(adj_mat, n=adj_degree,
                                          sparse=sparse)
    if mat:
        return attr_mat.bool().to(device), attr_mat.to(device)
    edge_idxs = attr_mat[0].nonzero().t().long()
    edge_types = attr_mat[0, edge_idxs[0], edge_idxs[1]]
    return edge_idxs.to(device), edge_types.to(device)
