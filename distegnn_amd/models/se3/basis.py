"""Equivariant kernel basis construction without lie_learn.

The reference obtains real Wigner-D matrices from ``lie_learn`` and solves
a Sylvester null-space problem for the basis change matrices Q_J
(from_se3cnn/utils_steerable.py:35-69, disk-cached). We re-own both:

* ``wigner_d(l, R)`` — the real degree-l rotation matrix, recovered
  NUMERICALLY from our own spherical harmonics by least squares:
  Y_l(R x) = D_l(R) Y_l(x) over a fixed sample of directions. Exact to
  fp64 round-off because Y_l are degree-l polynomials and the sample is
  overdetermined. Using the same Y in get_basis makes every convention
  cancel (verified end-to-end by the rotation-equivariance tests).
* ``basis_transformation_Q_J(J, d_in, d_out)`` — null space of the
  Sylvester operators kron(D_out x D_in, I) - kron(I, D_J^T) over a fixed
  set of rotations (the reference's exact construction), via SVD.
  Results are cached in-process (lru_cache) in fp64.
"""

from __future__ import annotations

from functools import lru_cache

import torch

from .sh import precompute_sh, real_spherical_harmonics


def _rot_z(a):
    c, s = torch.cos(a), torch.sin(a)
    return torch.tensor([[c, -s, 0.0], [s, c, 0.0], [0.0, 0.0, 1.0]],
                        dtype=torch.float64)


def _rot_y(a):
    c, s = torch.cos(a), torch.sin(a)
    return torch.tensor([[c, 0.0, s], [0.0, 1.0, 0.0], [-s, 0.0, c]],
                        dtype=torch.float64)


def rot(alpha, beta, gamma):
    """ZYZ Euler rotation (reference SO3.rot convention)."""
    a = torch.as_tensor(alpha, dtype=torch.float64)
    b = torch.as_tensor(beta, dtype=torch.float64)
    c = torch.as_tensor(gamma, dtype=torch.float64)
    return _rot_z(a) @ _rot_y(b) @ _rot_z(c)


@lru_cache(maxsize=None)
def _sample_dirs(count: int = 64):
    g = torch.Generator().manual_seed(1234)
    v = torch.randn(count, 3, generator=g, dtype=torch.float64)
    return v / v.norm(dim=-1, keepdim=True)


def wigner_d(l: int, rotation: torch.Tensor) -> torch.Tensor:
    """Real Wigner-D: Y_l(R x) = D Y_l(x). rotation: [3,3] fp64."""
    if l == 0:
        return torch.ones(1, 1, dtype=torch.float64)
    dirs = _sample_dirs()
    a = real_spherical_harmonics(l, dirs)                      # [S, 2l+1]
    b = real_spherical_harmonics(l, dirs @ rotation.T)         # [S, 2l+1]
    sol = torch.linalg.lstsq(a, b).solution                    # a @ sol = b
    return sol.T


def irr_repr(order: int, alpha, beta, gamma) -> torch.Tensor:
    """Real irreducible representation at ZYZ Euler angles (reference
    SO3.irr_repr parity surface)."""
    return wigner_d(order, rot(alpha, beta, gamma))


def _kron(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    return torch.kron(a.contiguous(), b.contiguous())


_RANDOM_ANGLES = [
    [4.41301023, 5.56684102, 4.59384642],
    [4.93325116, 6.12697327, 4.14574096],
    [0.53878964, 4.09050444, 5.36539036],
    [2.16017393, 3.48835314, 5.55174441],
    [2.52385107, 0.2908958, 3.90040975],
]


@lru_cache(maxsize=None)
def basis_transformation_Q_J(J: int, order_in: int,
                             order_out: int) -> torch.Tensor:
    """Q_J [ (2*order_out+1)*(2*order_in+1), 2J+1 ] s.t. for all R:
    (D_out(R) x D_in(R)) Q_J = Q_J D_J(R)."""
    mats = []
    for a, b, c in _RANDOM_ANGLES:
        r_tensor = _kron(irr_repr(order_out, a, b, c),
                         irr_repr(order_in, a, b, c))
        r_j = irr_repr(J, a, b, c)
        mats.append(_kron(r_tensor, torch.eye(r_j.size(0),
                                              dtype=torch.float64))
                    - _kron(torch.eye(r_tensor.size(0), dtype=torch.float64),
                            r_j.T))
    stacked = torch.cat(mats, dim=0)
    # null space via SVD: right singular vectors with (near-)zero singular
    # values, plus any rows beyond rank(s)
    _, s, vh = torch.linalg.svd(stacked, full_matrices=True)
    rows = [int(i) for i in torch.nonzero(s < 1e-8).flatten()]
    rows += list(range(s.numel(), vh.size(0)))
    null = vh[rows]
    assert null.size(0) == 1, f"expected unique solution, got {null.size(0)}"
    q_j = null[0].view((2 * order_out + 1) * (2 * order_in + 1), 2 * J + 1)
    # verify on fresh rotations
    for a, b, c in torch.rand(3, 3, dtype=torch.float64) * 6.0:
        lhs = _kron(irr_repr(order_out, a, b, c),
                    irr_repr(order_in, a, b, c)) @ q_j
        rhs = q_j @ irr_repr(J, a, b, c)
        assert torch.allclose(lhs, rhs, atol=1e-6)
    return q_j


def get_basis(Y: dict, max_degree: int) -> dict:
    """Equivariant weight basis (reference modules.py:18-49).

    Returns dict['{d_in},{d_out}'] of shape
    [M, 1, 2*d_out+1, 1, 2*d_in+1, 2*min(d_in,d_out)+1]."""
    device = Y[0].device
    dtype = Y[0].dtype
    with torch.no_grad():
        basis = {}
        for d_in in range(max_degree + 1):
            for d_out in range(max_degree + 1):
                k_js = []
                for j in range(abs(d_in - d_out), d_in + d_out + 1):
                    q_j = basis_transformation_Q_J(j, d_in, d_out)
                    q_j = q_j.to(dtype).T.to(device)       # [2J+1, mo*mi]
                    k_js.append(Y[j] @ q_j)                # [M, mo*mi]
                size = (-1, 1, 2 * d_out + 1, 1, 2 * d_in + 1,
                        2 * min(d_in, d_out) + 1)
                basis[f"{d_in},{d_out}"] = torch.stack(k_js, -1).view(*size)
        return basis


def get_basis_and_r(G, max_degree: int):
    """Basis + internodal distances for a forward pass (reference
    modules.py:52-76). G is an EdgeGraph with edata['d'] [M,3]."""
    d = G.edata["d"]
    Y = precompute_sh(d, 2 * max_degree)
    basis = get_basis(Y, max_degree)
    r = d.pow(2).sum(-1, keepdim=True).sqrt()
    return basis, r
