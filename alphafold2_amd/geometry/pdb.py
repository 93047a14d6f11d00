"""PDB / MSA file I/O.  Capability parity: reference utils.py:152-252.

Heavy external deps (mdtraj, Bio) are imported lazily and these helpers
degrade gracefully when they are absent (this image has no network, so
`download_pdb` is a capability slot rather than a hot path).
"""
import itertools
import os
import string

import numpy as np
import torch

from ..vocab import VOCAB


def download_pdb(name, route):
    """Fetch a PDB entry from RCSB (requires network). Returns `route`."""
    os.system(f"curl https://files.rcsb.org/download/{name}.pdb > {route}")
    return route


def clean_pdb(name, route=None, chain_num=None):
    """Keep only the selected chain(s) of a PDB file (requires mdtraj)."""
    import mdtraj
    destin = route if route is not None else name
    raw_prot = mdtraj.load_pdb(name)
    idxs = []
    for chain in raw_prot.topology.chains:
        if chain_num is not None and chain_num != chain.index:
            continue
        chain_idxs = raw_prot.topology.select(f"chainid == {str(chain.index)}")
        idxs.extend(chain_idxs.tolist())
    idxs = sorted(idxs)
    prot = mdtraj.Trajectory(xyz=raw_prot.xyz[:, idxs],
                             topology=raw_prot.topology.subset(idxs))
    prot.save(destin)
    return destin


def custom2pdb(coords, proteinnet_id, route):
    """Write coords into a scaffold downloaded for `proteinnet_id`."""
    import mdtraj
    if isinstance(coords, torch.Tensor):
        coords = coords.detach().cpu().numpy()
    if coords.shape[1] == 3:
        coords = coords.T
    coords = np.expand_dims(coords, axis=0)
    pdb_name, chain_num = proteinnet_id.split("#")[-1].split("_")[:-1]
    pdb_destin = "/".join(route.split("/")[:-1]) + "/" + pdb_name + ".pdb"
    download_pdb(pdb_name, pdb_destin)
    clean_pdb(pdb_destin, chain_num=chain_num)
    scaffold = mdtraj.load_pdb(pdb_destin)
    scaffold.xyz = coords
    scaffold.save(route)
    return pdb_destin, route


def coords2pdb(seq, coords, cloud_mask, prefix="", name="af2_struct.pdb"):
    """Minimal PDB writer for scn-format coordinates (no external deps).

    * seq: (L,) ints in vocab convention
    * coords: (3, N) atom coords for the atoms selected by cloud_mask
    * cloud_mask: (L, 14) occupancy
    """
    from ..vocab import atom_names_for, ONE_TO_THREE_LETTER_MAP
    scaffold = torch.zeros((*cloud_mask.shape, 3))
    scaffold[cloud_mask] = coords.t().cpu().float() if coords.shape[0] == 3 \
        else coords.cpu().float()
    lines, serial = [], 1
    for li in range(cloud_mask.shape[0]):
        aa = VOCAB._int2char[int(seq[li])]
        if aa == '_':
            continue
        res3 = ONE_TO_THREE_LETTER_MAP.get(aa, 'UNK')
        names = atom_names_for(aa)
        for ci in range(cloud_mask.shape[1]):
            if not cloud_mask[li, ci]:
                continue
            x, y, z = scaffold[li, ci].tolist()
            atom = names[ci] if ci < len(names) else 'X'
            lines.append(
                f"ATOM  {serial:5d} {atom:<4s}{res3:>3s} A{li + 1:4d}    "
                f"{x:8.3f}{y:8.3f}{z:8.3f}  1.00  0.00")
            serial += 1
    lines.append("END")
    with open(prefix + name, "w") as f:
        f.write("\n".join(lines) + "\n")
    return prefix + name


# adapted behavior from the ESM a3m-reading convention


def remove_insertions(sequence: str) -> str:
    """Drop lowercase/insertion characters from an aligned sequence."""
    deletekeys = dict.fromkeys(string.ascii_lowercase)
    deletekeys["."] = None
    deletekeys["*"] = None
    translation = str.maketrans(deletekeys)
    return sequence.translate(translation)


def read_msa(filename: str, nseq: int):
    """First `nseq` sequences of an a3m/fasta MSA as (desc, seq) pairs.

    Uses Bio.SeqIO when available, else a small built-in fasta parser.
    """
    try:
        from Bio import SeqIO
        return [(record.description, remove_insertions(str(record.seq)))
                for record in itertools.islice(SeqIO.parse(filename, "fasta"), nseq)]
    except ImportError:
        out, desc, seq = [], None, []
        with open(filename) as f:
            for line in f:
                line = line.rstrip()
                if line.startswith(">"):
                    if desc is not None:
                        out.append((desc, remove_insertions("".join(seq))))
                        if len(out) >= nseq:
                            return out
                    desc, seq = line[1:], []
                else:
                    seq.append(line)
            if desc is not None and len(out) < nseq:
                out.append((desc, remove_insertions("".join(seq))))
        return out
