"""Top-level alias matching the reference's module layout
(`alphafold2_pytorch.reversible` -> `alphafold2_amd.reversible`)."""
from .models.reversible import (  # noqa: F401
    Deterministic, ReversibleEvoformer, ReversibleEvoformerBlock,
    make_reversible_evoformer,
)
