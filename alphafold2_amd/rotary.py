"""Top-level alias matching the reference's module layout
(`alphafold2_pytorch.rotary` -> `alphafold2_amd.rotary`)."""
from .models.rotary import (  # noqa: F401
    AxialRotaryEmbedding, DepthWiseConv1d, FixedPositionalEmbedding,
    apply_rotary_pos_emb, rotate_every_two,
)
