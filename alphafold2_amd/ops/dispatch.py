"""Backend dispatch between the gfx950 HIP extension and the eager path.

Policy:
* CPU tensors -> eager, always.
* ROCm tensors + extension loaded -> HIP kernel when one exists for the
  op (per-op hasattr check so kernels can land incrementally), else the
  eager composition (which still runs on-device through rocBLAS/hipBLASLt).
* ROCm tensors + extension NOT importable -> hard error ("native code not
  loaded") unless AF2AMD_ALLOW_EAGER_GPU=1.
* AF2AMD_FORCE_EAGER=1 pins eager everywhere (kernel parity tests).
"""
import os

import torch

from . import eager

_FORCE_EAGER = os.environ.get("AF2AMD_FORCE_EAGER", "0") == "1"
_ALLOW_EAGER_GPU = os.environ.get("AF2AMD_ALLOW_EAGER_GPU", "0") == "1"

_EXT = None
_EXT_TRIED = False
_EXT_ERR = None


def _load_ext():
    global _EXT, _EXT_TRIED, _EXT_ERR
    if _EXT_TRIED:
        return _EXT
    _EXT_TRIED = True
    try:
        from alphafold2_amd import _hip_ops  # built in-tree for gfx950
        _EXT = _hip_ops
    except ImportError as e:  # extension not built (CPU-only dev box)
        _EXT = None
        _EXT_ERR = e
    return _EXT


def hip_ops_available():
    return _load_ext() is not None


def _want_hip(t: torch.Tensor) -> bool:
    if _FORCE_EAGER or not t.is_cuda:
        return False
    ext = _load_ext()
    if ext is None:
        if _ALLOW_EAGER_GPU:
            return False
        raise RuntimeError(
            "alphafold2_amd._hip_ops is not importable on a GPU device "
            f"(build with `python setup.py build_ext --inplace`): {_EXT_ERR}")
    return True


def using_hip(t: torch.Tensor, opname: str) -> bool:
    """True when `opname` should run through the HIP extension."""
    return _want_hip(t) and hasattr(_load_ext(), opname)


_warned = set()


def warn_gpu_fallback(opname: str, reason: str):
    """Log (once per op+reason) when a GPU tensor takes the eager path
    even though the extension is loaded — silent fallbacks hide
    'MI355X-native in name only' configs."""
    key = (opname, reason)
    if key in _warned or _FORCE_EAGER:
        return
    _warned.add(key)
    import warnings
    warnings.warn(
        f"alphafold2_amd: op '{opname}' is running the EAGER composition "
        f"on GPU ({reason}); the fused gfx950 kernel does not cover this "
        "configuration", RuntimeWarning, stacklevel=3)


# ---------------------------------------------------------------------------
# op entry points


def attention_core(q, k, v, bias=None, mask=None, context_mask=None,
                   tie_dim=None, bias_repeat=1, dropout=0., training=False):
    """bias (when given) has shape (B // bias_repeat, h, Lq, Lk); the
    repeat fold is resolved inside the fused kernel (never materialized),
    and expanded explicitly only on the eager path.  Attention-prob
    dropout (reference alphafold2.py:172) routes to the eager path."""
    drop_active = dropout > 0. and training
    # head dims < 64 run on the 64-wide tile via zero padding (see
    # hip_attention_core); only dim_head > 64 or non-multiple-of-8
    # layouts fall back to eager
    fusable = (
        not drop_active
        and q.dtype == torch.bfloat16
        and q.shape[-1] <= 64 and q.shape[-1] % 8 == 0
        and (bias is None or bias.dtype == torch.bfloat16)
        and (tie_dim is None or q.shape[0] % tie_dim == 0)
        and using_hip(q, 'attn_fwd')
    )
    if fusable:
        from .hip_autograd import hip_attention_core
        return hip_attention_core(q, k, v, bias=bias, mask=mask,
                                  context_mask=context_mask,
                                  tie_dim=tie_dim, bias_repeat=bias_repeat)
    if q.is_cuda and hip_ops_available() and not _FORCE_EAGER:
        reason = ('attention-prob dropout active' if drop_active
                  else f'dim_head={q.shape[-1]} (fused kernel covers '
                       'multiples of 8 up to 64)'
                  if not (q.shape[-1] <= 64 and q.shape[-1] % 8 == 0)
                  else f'dtype={q.dtype} (fused kernel is bf16)')
        warn_gpu_fallback('attention_core', reason)
    if bias is not None and bias_repeat != 1:
        bias = bias.repeat_interleave(bias_repeat, dim=0)
    return eager.attention_core(q, k, v, bias=bias, mask=mask,
                                context_mask=context_mask, tie_dim=tie_dim,
                                dropout=dropout, training=training)


def attention_core_packed(packed, heads, inner, bias=None, mask=None,
                          context_mask=None, bias_repeat=1):
    """Self-attention consuming a packed [q|k|v|...] projection (B, L, W).
    HIP-only fast path; returns None if not fusable (caller falls back
    to the split path)."""
    key_mask = context_mask if context_mask is not None else mask
    fusable = (
        packed.dtype == torch.bfloat16
        and (inner // heads) == 64
        and (bias is None or bias.dtype == torch.bfloat16)
        and packed.is_cuda
        and using_hip(packed, 'attn_fwd')
    )
    if not fusable:
        return None
    from .hip_autograd import hip_attention_packed
    return hip_attention_packed(packed, heads, inner, bias=bias,
                                mask=key_mask, bias_repeat=bias_repeat)


def geglu(x):
    if using_hip(x, 'geglu_fwd'):
        from .hip_autograd import hip_geglu
        return hip_geglu(x)
    return eager.geglu(x)


def _bf16_ok(*tensors):
    """True when every tensor is bf16 already, or will be autocast to
    bf16 inside the custom_fwd wrapper (fp32 under cuda bf16 autocast)."""
    amp = (torch.is_autocast_enabled('cuda')
           and torch.get_autocast_dtype('cuda') == torch.bfloat16)
    for t in tensors:
        if t is None:
            continue
        if t.dtype == torch.bfloat16:
            continue
        if amp and t.dtype in (torch.float32, torch.float16):
            continue
        return False
    return True


def fused_linear(x, weight, bias=None, residual=None):
    """out = x @ weight.T (+ bias) (+ residual).

    Measured (profiles/r02_ffgemm_ab.log): hipBLASLt beats the custom
    128x128 GEMM on plain/residual FORWARDS, so _LinearFn runs Tensile
    forward by default (AF2AMD_CUSTOM_LINEAR=1 forces the custom GEMM
    for A/B work); the routing still pays in backward, where the
    split-K wgrad kernel takes the small-output/huge-K gradient class
    Tensile runs at ~98 TF."""
    ok = (x.shape[-1] % 8 == 0 and weight.shape[0] % 8 == 0
          and _bf16_ok(x, weight, residual)
          and using_hip(x, 'linear_fwd'))
    if ok:
        from .hip_autograd import hip_linear
        return hip_linear(x, weight, bias, residual)
    out = torch.nn.functional.linear(x, weight, bias)
    if residual is not None:
        out = out + residual
    return out


def ff1_geglu(x, weight, bias=None):
    """GEGLU(x @ weight.T + bias) with the GEMM + chunk + gelu + mul
    fused into one kernel (K6 of SURVEY.md §2.17)."""
    ok = (x.shape[-1] % 8 == 0 and weight.shape[0] % 16 == 0
          and _bf16_ok(x, weight)
          and using_hip(x, 'ff1_geglu_fwd'))
    if ok:
        if not torch.is_grad_enabled():
            # inference: skip materializing the 2x-width pre-activation
            ext = _load_ext()
            x16, w16, b16 = (t.to(torch.bfloat16).contiguous()
                             if t is not None else None
                             for t in (x, weight, bias))
            return ext.ff1_geglu_fwd(x16, w16, b16, -1, False)[0]
        from .hip_autograd import hip_ff1_geglu
        return hip_ff1_geglu(x, weight, bias)
    return geglu(torch.nn.functional.linear(x, weight, bias))


def outer_product_mean(left, right, mask=None, eps=1e-5):
    if left.dtype == torch.bfloat16 and using_hip(left, 'pcgemm'):
        from .hip_autograd import hip_outer_product_mean
        return hip_outer_product_mean(left, right, mask=mask, eps=eps)
    return eager.outer_product_mean(left, right, mask=mask, eps=eps)


def triangle_mix(left, right, mix):
    if left.dtype == torch.bfloat16 and using_hip(left, 'pcgemm'):
        from .hip_autograd import hip_triangle_mix
        return hip_triangle_mix(left, right, mix)
    return eager.triangle_mix(left, right, mix)


def pair_outer_sum(x_left, x_right):
    return eager.pair_outer_sum(x_left, x_right)


def pair_rep_build(x_left, x_right, emb_weight, rel):
    """K13: out[b,i,j] = left[i] + right[j] + emb[rel[i,j]] in one pass
    (the eager composition materializes the outer sum AND the gathered
    embedding before adding)."""
    ok = (x_left.shape[-1] % 8 == 0 and x_left.shape[-1] <= 2048
          and _bf16_ok(x_left, x_right, emb_weight)
          and using_hip(x_left, 'pairrep_fwd'))
    if ok:
        from .hip_autograd import hip_pair_rep
        rel = rel.expand(x_left.shape[0], -1, -1).contiguous().long()
        return hip_pair_rep(x_left, x_right, emb_weight, rel)
    return eager.pair_outer_sum(x_left, x_right) \
        + torch.nn.functional.embedding(rel, emb_weight)


def distance_buckets(coords, boundaries):
    if using_hip(coords, 'dist_buckets'):
        ext = _load_ext()
        return ext.dist_buckets(coords.contiguous(), boundaries.contiguous())
    return eager.distance_buckets(coords, boundaries)


def layer_norm(x, weight, bias, eps=1e-5):
    if using_hip(x, 'layernorm_fwd'):
        from .hip_autograd import hip_layer_norm
        return hip_layer_norm(x, weight, bias, eps)
    return eager.layer_norm(x, weight, bias, eps)


def tri_proj_gates(fused, hdim, row_mask=None):
    """TriangleMultiplicative's [left|right|lg|rg|og] projection -> the
    two gated operands + the out-gate slice, as ONE autograd unit whose
    backward writes into a packed buffer (kills the 5-way SplitBackward
    concatenation)."""
    ok = (fused.dtype == torch.bfloat16 and fused.shape[-1] == 5 * hdim
          and hdim % 8 == 0 and using_hip(fused, 'gatemul_bwd'))
    if ok:
        from .hip_autograd import hip_tri_proj_gates
        return hip_tri_proj_gates(fused, hdim, row_mask)
    left, right, lg, rg, og = fused.split([hdim] * 5, dim=-1)
    left = eager.softclamp_gate(left, lg)
    right = eager.softclamp_gate(right, rg)
    if row_mask is not None:
        m = row_mask.unsqueeze(-1).to(left.dtype)
        left = left * m
        right = right * m
    return left, right, og


def softclamp_gate(x, gates, row_mask=None):
    """out = x * sigmoid(gates) (* row_mask broadcast per row).
    row_mask: optional bool with shape == x.shape[:-1]."""
    if x.shape == gates.shape and using_hip(x, 'gatemul_fwd'):
        from .hip_autograd import hip_gatemul
        return hip_gatemul(x, gates, row_mask)
    out = eager.softclamp_gate(x, gates)
    if row_mask is not None:
        out = out * row_mask.unsqueeze(-1).to(out.dtype)
    return out
