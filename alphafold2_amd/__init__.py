"""alphafold2_amd — MI355X-native AlphaFold2-style protein structure
framework.

Same public surface as lucidrains/alphafold2 (`Alphafold2`, `Evoformer`)
re-designed for CDNA4: PyTorch-ROCm front end, hand-written gfx950 HIP
kernels behind `alphafold2_amd.ops`, RCCL-over-xGMI data parallelism in
`alphafold2_amd.parallel`.
"""
from alphafold2_amd.models.alphafold2 import Alphafold2, Recyclables, ReturnValues
from alphafold2_amd.models.evoformer import Evoformer

__version__ = "0.1.0"

__all__ = ["Alphafold2", "Evoformer", "Recyclables", "ReturnValues"]
