"""Top-level alias matching the reference's module layout
(`alphafold2_pytorch.embeds` -> `alphafold2_amd.embeds`), including the
names the reference module re-exposed from its own imports."""
from .models.embeds import (  # noqa: F401
    ESMEmbedWrapper, FakeEmbedder, MSAEmbedWrapper, ProtTranEmbedWrapper,
)
from .constants import (  # noqa: F401
    ESM_EMBED_DIM, ESM_MODEL_PATH, MSA_EMBED_DIM, MSA_MODEL_PATH,
    PROTTRAN_EMBED_DIM,
)
from .embedd_utils import (  # noqa: F401
    get_esm_embedd, get_msa_embedd, get_prottran_embedd,
)
from .geometry.backend import exists  # noqa: F401
