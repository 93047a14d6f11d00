"""Print per-assert max errors for the failing LN/GEGLU parity shapes."""
import torch

from alphafold2_amd.ops.hip_autograd import hip_geglu, hip_layer_norm


def ln_case(shape, dtype):
    torch.manual_seed(0)
    D = shape[-1]
    x = torch.randn(*shape, device='cuda', dtype=dtype)
    w = torch.randn(D, device='cuda') * 0.5 + 1
    b = torch.randn(D, device='cuda') * 0.1
    x1 = x.clone().requires_grad_(True)
    w1 = w.clone().requires_grad_(True)
    b1 = b.clone().requires_grad_(True)
    x2 = x.float().clone().requires_grad_(True)
    w2 = w.clone().requires_grad_(True)
    b2 = b.clone().requires_grad_(True)

    y1 = hip_layer_norm(x1, w1, b1, 1e-5)
    y2 = torch.nn.functional.layer_norm(x2, (D,), w2, b2, 1e-5)
    g = torch.randn_like(y2)
    y1.backward(g.to(dtype))
    y2.backward(g)
    print(f"LN {shape} {dtype}: y={(y1.float()-y2).abs().max().item():.3e} "
          f"dx={(x1.grad.float()-x2.grad).abs().max().item():.3e} "
          f"dw={(w1.grad-w2.grad).abs().max().item():.3e} "
          f"db={(b1.grad-b2.grad).abs().max().item():.3e}")


def geglu_case(dtype):
    torch.manual_seed(0)
    x = torch.randn(64, 128, 512, device='cuda', dtype=dtype)
    x1 = x.clone().requires_grad_(True)
    x2 = x.float().clone().requires_grad_(True)
    y1 = hip_geglu(x1)
    a, gt = x2.chunk(2, dim=-1)
    y2 = a * torch.nn.functional.gelu(gt)
    g = torch.randn_like(y2)
    y1.backward(g.to(dtype))
    y2.backward(g)
    print(f"GEGLU {dtype}: y={(y1.float()-y2).abs().max().item():.3e} "
          f"dx={(x1.grad.float()-x2.grad).abs().max().item():.3e}")
    # also vs eager bf16 (same-precision comparison)
    x3 = x.clone().requires_grad_(True)
    a3, g3 = x3.chunk(2, dim=-1)
    y3 = a3 * torch.nn.functional.gelu(g3)
    print(f"GEGLU {dtype} vs eager same-dtype: "
          f"y={(y1-y3).abs().max().float().item():.3e}")


if __name__ == '__main__':
    for shape in [(128, 256), (64, 64, 384), (4096, 384), (64, 384), (7, 33)]:
        for dtype in (torch.float32, torch.bfloat16):
            ln_case(shape, dtype)
    for dtype in (torch.float32, torch.bfloat16):
        geglu_case(dtype)
