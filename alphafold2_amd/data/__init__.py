from .synthetic import SyntheticProteinDataset, synthetic_batch
from .trrosetta import TrRosettaDataset, TrRosettaDataModule, collate_batch
from .scn import SCNDataset, collate_scn
from . import scn
