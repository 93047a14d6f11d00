"""Flat re-export surface mirroring the reference's
`alphafold2_pytorch.utils` module (reference utils.py), so code written
against the reference (`from alphafold2_pytorch.utils import *` style)
ports with only the package name changed.
"""
from .vocab import (  # noqa: F401
    VOCAB, ProteinVocabulary, ATOM_IDS, CUSTOM_INFO, get_atom_ids_dict,
    make_cloud_mask, make_atom_id_embedds, ONE_TO_THREE_LETTER_MAP,
)
from .geometry import (  # noqa: F401
    exists, as_batched, expand_dims_to, dual_backend, resolve_backend,
    torch_default_dtype,
    DISTANCE_THRESHOLDS, get_bucketed_distance_matrix, center_distogram_torch,
    kabsch_torch, kabsch_numpy, rmsd_torch, rmsd_numpy,
    gdt_torch, gdt_numpy, tmscore_torch, tmscore_numpy,
    distmat_loss_torch, lddt_ca_torch,
    Kabsch, RMSD, GDT, TMscore,
    mds_torch, mds_numpy, mdscaling_torch, mdscaling_numpy,
    get_dihedral_torch, get_dihedral_numpy,
    calc_phis_torch, calc_phis_numpy, MDScaling,
    scn_cloud_mask, scn_backbone_mask, scn_atom_embedd, sidechain_container,
    mat_input_to_masked, nth_deg_adjacency, prot_covalent_bond,
    download_pdb, clean_pdb, custom2pdb, coords2pdb,
    remove_insertions, read_msa,
)
from .embedd_utils import (  # noqa: F401
    ids_to_embed_input, ids_to_prottran_input,
    get_prottran_embedd, get_msa_embedd, get_esm_embedd, get_t5_embedd,
    get_all_protein_ids,
)
