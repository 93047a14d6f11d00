"""Top-level alias matching the reference's module layout
(`alphafold2_pytorch.embeds` -> `alphafold2_amd.embeds`)."""
from .models.embeds import (  # noqa: F401
    ESMEmbedWrapper, FakeEmbedder, MSAEmbedWrapper, ProtTranEmbedWrapper,
)
