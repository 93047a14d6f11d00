from .synthetic import SyntheticProteinDataset, synthetic_batch
