"""Real spherical harmonics evaluated directly from cartesian directions.

Replaces the reference's angle-based Legendre recursion pipeline
(se3_dynamics/.../from_se3cnn/representations.py + the
``get_spherical_from_cartesian_torch`` conversion): closed-form real
tesseral harmonics for l <= 4 as polynomials of the unit vector — no
trig, no recursion, GPU-friendly. Any consistent per-degree normalization
yields an equivariant basis (the Wigner-D matrices in basis.py are
derived from THESE functions, so conventions cancel); we use the
orthonormal real SH.
"""

from __future__ import annotations

import math

import torch


def real_spherical_harmonics(l: int, xyz: torch.Tensor) -> torch.Tensor:
    """Y_l(x) for unit vectors xyz [..., 3] -> [..., 2l+1], m = -l..l."""
    x, y, z = xyz[..., 0], xyz[..., 1], xyz[..., 2]
    pi = math.pi
    if l == 0:
        return torch.full_like(x, 0.5 / math.sqrt(pi)).unsqueeze(-1)
    if l == 1:
        c = math.sqrt(3.0 / (4 * pi))
        return torch.stack([c * y, c * z, c * x], dim=-1)
    if l == 2:
        c = [0.5 * math.sqrt(15 / pi), 0.5 * math.sqrt(15 / pi),
             0.25 * math.sqrt(5 / pi), 0.5 * math.sqrt(15 / pi),
             0.25 * math.sqrt(15 / pi)]
        return torch.stack([
            c[0] * x * y,
            c[1] * y * z,
            c[2] * (3 * z * z - 1),
            c[3] * x * z,
            c[4] * (x * x - y * y),
        ], dim=-1)
    if l == 3:
        return torch.stack([
            0.25 * math.sqrt(35 / (2 * pi)) * y * (3 * x * x - y * y),
            0.5 * math.sqrt(105 / pi) * x * y * z,
            0.25 * math.sqrt(21 / (2 * pi)) * y * (5 * z * z - 1),
            0.25 * math.sqrt(7 / pi) * z * (5 * z * z - 3),
            0.25 * math.sqrt(21 / (2 * pi)) * x * (5 * z * z - 1),
            0.25 * math.sqrt(105 / pi) * (x * x - y * y) * z,
            0.25 * math.sqrt(35 / (2 * pi)) * x * (x * x - 3 * y * y),
        ], dim=-1)
    if l == 4:
        x2, y2, z2 = x * x, y * y, z * z
        return torch.stack([
            0.75 * math.sqrt(35 / pi) * x * y * (x2 - y2),
            0.75 * math.sqrt(35 / (2 * pi)) * y * z * (3 * x2 - y2),
            0.75 * math.sqrt(5 / pi) * x * y * (7 * z2 - 1),
            0.75 * math.sqrt(5 / (2 * pi)) * y * z * (7 * z2 - 3),
            (3.0 / 16) * math.sqrt(1 / pi) * (35 * z2 * z2 - 30 * z2 + 3),
            0.75 * math.sqrt(5 / (2 * pi)) * x * z * (7 * z2 - 3),
            (3.0 / 8) * math.sqrt(5 / pi) * (x2 - y2) * (7 * z2 - 1),
            0.75 * math.sqrt(35 / (2 * pi)) * x * z * (x2 - 3 * y2),
            (3.0 / 16) * math.sqrt(35 / pi) * (x2 * (x2 - 3 * y2)
                                               - y2 * (3 * x2 - y2)),
        ], dim=-1)
    raise NotImplementedError(f"real SH implemented for l<=4 (got {l})")


def precompute_sh(directions: torch.Tensor, max_j: int) -> dict:
    """Y_J for J = 0..max_j on normalized directions [M, 3] -> {J: [M, 2J+1]}.

    (Reference precompute_sh, utils_steerable.py:273-295.)"""
    n = directions / directions.norm(dim=-1, keepdim=True).clamp(min=1e-12)
    return {j: real_spherical_harmonics(j, n) for j in range(max_j + 1)}
