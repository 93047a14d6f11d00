from .synthetic import SyntheticProteinDataset, synthetic_batch
from .trrosetta import TrRosettaDataset, TrRosettaDataModule, collate_batch
