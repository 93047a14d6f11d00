"""nn.Module shims that keep standard parameter layouts (state-dict
compatible with nn.LayerNorm etc.) while routing compute through the
fused gfx950 kernels."""
from torch import nn

from . import dispatch


class FusedLayerNorm(nn.LayerNorm):
    """Drop-in nn.LayerNorm: same parameters, fused HIP kernel on ROCm
    devices (fp32 accumulation regardless of input dtype)."""

    def forward(self, x):
        if x.is_cuda and self.elementwise_affine \
                and dispatch.using_hip(x, 'layernorm_fwd'):
            from .hip_autograd import hip_layer_norm
            return hip_layer_norm(x, self.weight, self.bias, self.eps)
        return super().forward(x)
