"""Quaternion algebra for the structure module.

Replaces the reference's pytorch3d dependency
(reference alphafold2.py:20: quaternion_multiply, quaternion_to_matrix)
with small native torch implementations (wxyz convention, standardized
to non-negative real part as pytorch3d does).  fp32 throughout — these
feed the equivariant structure refinement.
"""
import torch


def standardize_quaternion(q):
    """Flip sign so the real part is non-negative."""
    return torch.where(q[..., 0:1] < 0, -q, q)


def quaternion_raw_multiply(a, b):
    """Hamilton product of quaternions (..., 4) in (w, x, y, z) order."""
    aw, ax, ay, az = a.unbind(-1)
    bw, bx, by, bz = b.unbind(-1)
    ow = aw * bw - ax * bx - ay * by - az * bz
    ox = aw * bx + ax * bw + ay * bz - az * by
    oy = aw * by - ax * bz + ay * bw + az * bx
    oz = aw * bz + ax * by - ay * bx + az * bw
    return torch.stack((ow, ox, oy, oz), dim=-1)


def quaternion_multiply(a, b):
    """Hamilton product, standardized to non-negative real part."""
    return standardize_quaternion(quaternion_raw_multiply(a, b))


def quaternion_to_matrix(q):
    """Quaternions (..., 4) wxyz -> rotation matrices (..., 3, 3)."""
    r, i, j, k = q.unbind(-1)
    two_s = 2.0 / (q * q).sum(-1)
    o = torch.stack((
        1 - two_s * (j * j + k * k),
        two_s * (i * j - k * r),
        two_s * (i * k + j * r),
        two_s * (i * j + k * r),
        1 - two_s * (i * i + k * k),
        two_s * (j * k - i * r),
        two_s * (i * k - j * r),
        two_s * (j * k + i * r),
        1 - two_s * (i * i + j * j),
    ), dim=-1)
    return o.reshape(*q.shape[:-1], 3, 3)


def matrix_to_quaternion(M):
    """Rotation matrices (..., 3, 3) -> quaternions (..., 4) wxyz."""
    m00, m01, m02 = M[..., 0, 0], M[..., 0, 1], M[..., 0, 2]
    m10, m11, m12 = M[..., 1, 0], M[..., 1, 1], M[..., 1, 2]
    m20, m21, m22 = M[..., 2, 0], M[..., 2, 1], M[..., 2, 2]
    tr = m00 + m11 + m22
    # numerically stable branch select
    qw = 0.5 * torch.sqrt(torch.clamp(1 + tr, min=1e-12))
    qx = 0.5 * torch.sqrt(torch.clamp(1 + m00 - m11 - m22, min=1e-12))
    qy = 0.5 * torch.sqrt(torch.clamp(1 - m00 + m11 - m22, min=1e-12))
    qz = 0.5 * torch.sqrt(torch.clamp(1 - m00 - m11 + m22, min=1e-12))
    qx = qx * torch.sign(torch.where((m21 - m12).abs() > 0, m21 - m12,
                                     torch.ones_like(m21)))
    qy = qy * torch.sign(torch.where((m02 - m20).abs() > 0, m02 - m20,
                                     torch.ones_like(m02)))
    qz = qz * torch.sign(torch.where((m10 - m01).abs() > 0, m10 - m01,
                                     torch.ones_like(m10)))
    q = torch.stack((qw, qx, qy, qz), dim=-1)
    return standardize_quaternion(q / q.norm(dim=-1, keepdim=True))
