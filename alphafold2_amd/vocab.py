"""Protein vocabulary + per-residue atom tables.

The reference outsources these to the `sidechainnet` package
(reference utils.py:18-21 imports ProteinVocabulary / SC_BUILD_INFO /
NUM_COORDS_PER_RES).  This framework carries its own self-contained
tables: the standard 20 amino acids in alphabetical one-letter order
(ids 0..19) with '_' = 20 as the padding token, and the standard PDB
heavy-atom layout of each residue in the 14-atom "scn" format
(N, CA, C, O, then side-chain atoms in build order).
"""
from __future__ import annotations

from . import constants

# ---------------------------------------------------------------------------
# vocabulary

AA_LETTERS = "ACDEFGHIKLMNPQRSTVWY"  # alphabetical one-letter codes, ids 0..19
PAD_CHAR = "_"
PAD_ID = len(AA_LETTERS)  # 20

ONE_TO_THREE_LETTER_MAP = {
    'A': 'ALA', 'C': 'CYS', 'D': 'ASP', 'E': 'GLU', 'F': 'PHE',
    'G': 'GLY', 'H': 'HIS', 'I': 'ILE', 'K': 'LYS', 'L': 'LEU',
    'M': 'MET', 'N': 'ASN', 'P': 'PRO', 'Q': 'GLN', 'R': 'ARG',
    'S': 'SER', 'T': 'THR', 'V': 'VAL', 'W': 'TRP', 'Y': 'TYR',
}


class ProteinVocabulary:
    """Maps amino-acid one-letter codes <-> integer ids.

    Drop-in for the sidechainnet ProteinVocabulary surface the reference
    uses: `_int2char`, `int2char()`, `char2int()`, `__len__`.
    """

    def __init__(self):
        self.pad_char = PAD_CHAR
        self._int2char = {i: c for i, c in enumerate(AA_LETTERS)}
        self._int2char[PAD_ID] = PAD_CHAR
        self._char2int = {c: i for i, c in self._int2char.items()}

    def int2char(self, i: int) -> str:
        return self._int2char[int(i)]

    def char2int(self, c: str) -> int:
        return self._char2int[c]

    def str2ints(self, s: str) -> list:
        return [self._char2int[c] for c in s]

    def ints2str(self, ids) -> str:
        return ''.join(self._int2char[int(i)] for i in ids)

    def __len__(self):
        return len(self._int2char)

    def __contains__(self, c):
        return c in self._char2int


VOCAB = ProteinVocabulary()

# ---------------------------------------------------------------------------
# per-residue heavy-atom layout (scn 14-atom format)
#
# Atom 0..3 are always N, CA, C, O.  Side-chain atoms follow in build
# order.  This is standard PDB nomenclature (public biochemistry).

SC_ATOM_NAMES = {
    'A': ['CB'],
    'R': ['CB', 'CG', 'CD', 'NE', 'CZ', 'NH1', 'NH2'],
    'N': ['CB', 'CG', 'OD1', 'ND2'],
    'D': ['CB', 'CG', 'OD1', 'OD2'],
    'C': ['CB', 'SG'],
    'Q': ['CB', 'CG', 'CD', 'OE1', 'NE2'],
    'E': ['CB', 'CG', 'CD', 'OE1', 'OE2'],
    'G': [],
    'H': ['CB', 'CG', 'ND1', 'CD2', 'CE1', 'NE2'],
    'I': ['CB', 'CG1', 'CG2', 'CD1'],
    'L': ['CB', 'CG', 'CD1', 'CD2'],
    'K': ['CB', 'CG', 'CD', 'CE', 'NZ'],
    'M': ['CB', 'CG', 'SD', 'CE'],
    'F': ['CB', 'CG', 'CD1', 'CD2', 'CE1', 'CE2', 'CZ'],
    'P': ['CB', 'CG', 'CD'],
    'S': ['CB', 'OG'],
    'T': ['CB', 'OG1', 'CG2'],
    'W': ['CB', 'CG', 'CD1', 'CD2', 'NE1', 'CE2', 'CE3', 'CZ2', 'CZ3', 'CH2'],
    'Y': ['CB', 'CG', 'CD1', 'CD2', 'CE1', 'CE2', 'CZ', 'OH'],
    'V': ['CB', 'CG1', 'CG2'],
    '_': [],
}

BACKBONE_ATOM_NAMES = ['N', 'CA', 'C', 'O']


def atom_names_for(aa: str) -> list:
    if aa == PAD_CHAR:
        return []
    return BACKBONE_ATOM_NAMES + SC_ATOM_NAMES[aa]


def _build_atom_id_table():
    """Token id for every distinct atom name (incl. '' for padding slots)."""
    names = {""}
    names.update(BACKBONE_ATOM_NAMES)
    for v in SC_ATOM_NAMES.values():
        names.update(v)
    return {name: i for i, name in enumerate(sorted(names))}


ATOM_IDS = _build_atom_id_table()


def get_atom_ids_dict():
    """Dict mapping each distinct atom name to a token id (reference
    utils.py:108-116 parity; '' is the padding slot)."""
    return dict(ATOM_IDS)


def make_cloud_mask(aa: str):
    """(14,) float mask: 1 for occupied atom slots of this residue type."""
    import numpy as np
    mask = np.zeros(constants.NUM_COORDS_PER_RES)
    if aa == PAD_CHAR:
        return mask
    mask[:len(atom_names_for(aa))] = 1
    return mask


def make_atom_id_embedds(aa: str):
    """(14,) int atom-name tokens for this residue type (0-padded)."""
    import numpy as np
    ids = np.zeros(constants.NUM_COORDS_PER_RES)
    if aa == PAD_CHAR:
        return ids
    for i, name in enumerate(atom_names_for(aa)):
        ids[i] = ATOM_IDS[name]
    return ids


CUSTOM_INFO = {
    aa: {
        "cloud_mask": make_cloud_mask(aa),
        "atom_id_embedd": make_atom_id_embedds(aa),
    }
    for aa in AA_LETTERS + PAD_CHAR
}

# ---------------------------------------------------------------------------
# idealized side-chain internal coordinates (NeRF build table)
#
# For residue-local NeRF placement: each side-chain atom k (index >= 4 in
# the scn layout) is placed from three previously placed atoms
# (a, b, c -> local indices into the scn layout) at distance `length` from
# c, with bond angle `angle` (b-c-new, degrees) and dihedral `torsion`
# (a-b-c-new, degrees; chi torsions default to staggered 180/-60/60).
# Idealized values: Engh & Huber-style standard geometry (public data).

_CC, _CN, _CO, _CS = 1.52, 1.47, 1.43, 1.81
_TET, _TRI = 110.5, 120.0

# (atom_slot, parents(a,b,c), length, angle, torsion)
SC_BUILD = {
    'A': [(4, (0, 2, 1), _CC, _TET, 122.7)],  # CB off N-C-CA frame
    'R': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), _CC, _TET, 180.0),
          (7, (4, 5, 6), _CN, _TET, 180.0),
          (8, (5, 6, 7), 1.33, _TRI, 180.0),
          (9, (6, 7, 8), 1.33, _TRI, 0.0),
          (10, (6, 7, 8), 1.33, _TRI, 180.0)],
    'N': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), 1.23, _TRI, -60.0),
          (7, (1, 4, 5), 1.33, _TRI, 120.0)],
    'D': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), 1.25, _TRI, -60.0),
          (7, (1, 4, 5), 1.25, _TRI, 120.0)],
    'C': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CS, _TET, 180.0)],
    'Q': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), _CC, _TET, 180.0),
          (7, (4, 5, 6), 1.23, _TRI, -60.0),
          (8, (4, 5, 6), 1.33, _TRI, 120.0)],
    'E': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), _CC, _TET, 180.0),
          (7, (4, 5, 6), 1.25, _TRI, -60.0),
          (8, (4, 5, 6), 1.25, _TRI, 120.0)],
    'G': [],
    'H': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), 1.38, _TRI, -90.0),
          (7, (1, 4, 5), 1.36, _TRI, 90.0),
          (8, (4, 5, 6), 1.32, 108.0, 180.0),
          (9, (4, 5, 7), 1.37, 108.0, 180.0)],
    'I': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (0, 1, 4), _CC, _TET, -60.0),
          (7, (1, 4, 5), _CC, _TET, 180.0)],
    'L': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), _CC, _TET, 180.0),
          (7, (1, 4, 5), _CC, _TET, -60.0)],
    'K': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), _CC, _TET, 180.0),
          (7, (4, 5, 6), _CC, _TET, 180.0),
          (8, (5, 6, 7), _CN, _TET, 180.0)],
    'M': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), _CS, _TET, 180.0),
          (7, (4, 5, 6), 1.79, 100.0, 180.0)],
    'F': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), 1.39, _TRI, 90.0),
          (7, (1, 4, 5), 1.39, _TRI, -90.0),
          (8, (4, 5, 6), 1.39, _TRI, 180.0),
          (9, (4, 5, 7), 1.39, _TRI, 180.0),
          (10, (5, 6, 8), 1.39, _TRI, 0.0)],
    'P': [(4, (0, 2, 1), _CC, 103.0, 115.0),
          (5, (0, 1, 4), _CC, 104.5, 30.0),
          (6, (1, 4, 5), _CC, 106.0, -35.0)],
    'S': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CO, _TET, 180.0)],
    'T': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CO, _TET, 180.0),
          (6, (0, 1, 4), _CC, _TET, -60.0)],
    'W': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), 1.37, 127.0, 90.0),
          (7, (1, 4, 5), 1.43, 126.6, -90.0),
          (8, (4, 5, 6), 1.38, 110.0, 180.0),
          (9, (4, 5, 7), 1.40, 107.0, 180.0),
          # benzene ring of the indole: each atom placed from its TRUE
          # bond parent (c of the frame) with planar ring torsions
          (10, (8, 9, 7), 1.40, _TRI, 180.0),   # CE3 on CD2
          (11, (10, 7, 9), 1.40, _TRI, 0.0),    # CZ2 on CE2
          (12, (9, 7, 10), 1.39, _TRI, 0.0),    # CZ3 on CE3
          (13, (7, 10, 12), 1.37, _TRI, 0.0)],  # CH2 on CZ3
    'Y': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (1, 4, 5), 1.39, _TRI, 90.0),
          (7, (1, 4, 5), 1.39, _TRI, -90.0),
          (8, (4, 5, 6), 1.39, _TRI, 180.0),
          (9, (4, 5, 7), 1.39, _TRI, 180.0),
          (10, (5, 6, 8), 1.39, _TRI, 0.0),
          (11, (6, 8, 10), 1.38, _TRI, 180.0)],
    'V': [(4, (0, 2, 1), _CC, _TET, 122.7),
          (5, (0, 1, 4), _CC, _TET, 180.0),
          (6, (0, 1, 4), _CC, _TET, -60.0)],
    '_': [],
}
