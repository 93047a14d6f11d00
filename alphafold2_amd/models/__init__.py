from .alphafold2 import Alphafold2, Recyclables, ReturnValues
from .evoformer import (
    Attention, AxialAttention, Evoformer, EvoformerBlock, FeedForward,
    GEGLU, MsaAttentionBlock, OuterMean, PairwiseAttentionBlock,
    TriangleMultiplicativeModule,
)
from .ipa import IPABlock, InvariantPointAttention
from .quaternion import (
    quaternion_multiply, quaternion_to_matrix, matrix_to_quaternion,
)
