from .backend import (
    exists, as_batched, expand_dims_to, dual_backend, resolve_backend,
    torch_default_dtype,
)
from .distogram import (
    DISTANCE_THRESHOLDS, get_bucketed_distance_matrix, center_distogram_torch,
)
from .metrics import (
    kabsch_torch, kabsch_numpy, rmsd_torch, rmsd_numpy,
    gdt_torch, gdt_numpy, tmscore_torch, tmscore_numpy,
    distmat_loss_torch, lddt_ca_torch,
    Kabsch, RMSD, GDT, TMscore,
)
from .mds import (
    mds_torch, mds_numpy, mdscaling_torch, mdscaling_numpy,
    get_dihedral_torch, get_dihedral_numpy,
    calc_phis_torch, calc_phis_numpy, MDScaling,
)
from .sidechain import (
    scn_cloud_mask, scn_backbone_mask, scn_atom_embedd, sidechain_container,
)
from .graphs import (
    mat_input_to_masked, nth_deg_adjacency, prot_covalent_bond,
)
from .pdb import (
    download_pdb, clean_pdb, custom2pdb, coords2pdb,
    remove_insertions, read_msa,
)
