"""Hot-op functional surface for the MI355X build.

Every numerically heavy op in the model goes through this package so the
hand-written CDNA4 HIP kernels can slot in behind a stable interface:

* `attention_core`   — softmax(QK^T * s + bias + mask) @ V (K1/K2 of
                        SURVEY.md §2.17): MSA row/col, triangle, template
                        pointwise attention all reduce to this.
* `geglu`            — chunk + gelu-gate (K6 epilogue)
* `outer_product_mean` — MSA -> pair communication (K4)
* `triangle_mix`     — triangle multiplicative einsum (K3)
* `pair_outer_sum`   — pair-rep build broadcast-add (K13)
* `distance_buckets` — fused cdist+bucketize (K8)
* `masked_layer_norm`/`layer_norm` — LN used by every block

Dispatch: the HIP extension (`alphafold2_amd._hip_ops`, built in-tree for
gfx950) is used whenever the tensors live on a ROCm device and the
extension is importable; otherwise the eager composition runs.  Set
AF2AMD_FORCE_EAGER=1 to pin the eager path (parity tests).  On a GPU box
a missing extension raises rather than silently falling back — set
AF2AMD_ALLOW_EAGER_GPU=1 to override.
"""
from .dispatch import (  # noqa: F401
    hip_ops_available, using_hip,
    attention_core, attention_core_packed, geglu, outer_product_mean, triangle_mix,
    pair_outer_sum, pair_rep_build, distance_buckets, layer_norm, softclamp_gate,
    tri_proj_gates,
    fused_linear, ff1_geglu,
)
