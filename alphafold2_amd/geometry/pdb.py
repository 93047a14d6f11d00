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
    import urllib.request
    urllib.request.urlretrieve(
        f"https://files.rcsb.org/download/{name}.pdb", route)
    return route


def clean_pdb(name, route=None, chain_num=None):
    """Keep only the selected chain (or all chains, re-saved) of a PDB
    file (requires mdtraj)."""
    import mdtraj
    traj = mdtraj.load_pdb(name)
    query = "all" if chain_num is None else f"chainid {chain_num}"
    out = traj.atom_slice(traj.topology.select(query))
    destination = route if route is not None else name
    out.save(destination)
    return destination


def custom2pdb(coords, proteinnet_id, route):
    """Write coords into a scaffold fetched for a proteinnet-style id
    ("<set>#<pdb>_<chain>_<extra>"); requires network + mdtraj."""
    import mdtraj
    if isinstance(coords, torch.Tensor):
        coords = coords.detach().cpu().numpy()
    if coords.shape[1] == 3:  # (3, N) -> (N, 3)
        coords = coords.T
    pdb_name, chain_num = proteinnet_id.split("#")[-1].split("_")[:-1]
    scaffold_path = os.path.join(os.path.dirname(route), pdb_name + ".pdb")
    download_pdb(pdb_name, scaffold_path)
    clean_pdb(scaffold_path, chain_num=chain_num)
    scaffold = mdtraj.load_pdb(scaffold_path)
    scaffold.xyz = coords[None]
    scaffold.save(route)
    return scaffold_path, route


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


# a3m convention: lowercase = insertion columns, './*' = gap/stop markers
_A3M_DROP = str.maketrans('', '', string.ascii_lowercase + '.*')


def remove_insertions(sequence: str) -> str:
    """Drop insertion characters from an a3m-aligned sequence."""
    return sequence.translate(_A3M_DROP)


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
