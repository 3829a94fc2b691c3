from .fibers import Fiber
from .sh import precompute_sh, real_spherical_harmonics
from .basis import basis_transformation_Q_J, get_basis, get_basis_and_r, wigner_d
from .graph import EdgeGraph
from . import modules

__all__ = [
    "Fiber", "precompute_sh", "real_spherical_harmonics",
    "basis_transformation_Q_J", "get_basis", "get_basis_and_r", "wigner_d",
    "EdgeGraph", "modules",
]
