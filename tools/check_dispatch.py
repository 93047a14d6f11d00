"""Verify which backend each hot op dispatches to under the bench's
autocast regime (run on a GPU box)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from alphafold2_amd.models.evoformer import FeedForward
from alphafold2_amd.ops import dispatch, hip_autograd

calls = {'ff1': 0, 'lin': 0}
_orig_ff1 = hip_autograd.hip_ff1_geglu
_orig_lin = hip_autograd.hip_linear


def ff1(*a, **k):
    calls['ff1'] += 1
    return _orig_ff1(*a, **k)


def lin(*a, **k):
    calls['lin'] += 1
    return _orig_lin(*a, **k)


hip_autograd.hip_ff1_geglu = ff1
hip_autograd.hip_linear = lin

m = FeedForward(dim=256).cuda().train()
x = torch.randn(8, 64, 256, device='cuda', requires_grad=True)
with torch.autocast('cuda', dtype=torch.bfloat16):
    out = m(x, residual=x)
out.float().pow(2).mean().backward()
torch.cuda.synchronize()
print('ff1 fused calls:', calls['ff1'], ' linear fused calls:', calls['lin'])
print('out dtype:', out.dtype, 'grad ok:', x.grad is not None)
assert calls['ff1'] == 1, 'ff1_geglu fused path NOT active in autocast!'
print('DISPATCH OK')
