"""Random SE(3) transforms for equivariance tests and test-time augmentation.

Parity with reference utils/rotate.py:6-57 (rotx/roty/rotz, random_rotate,
random_rotate_y); used by the equivariance property tests and the protein
pipeline's test-split augmentation (reference process_dataset.py:162-174).
"""

import numpy as np


def rotx(theta: float) -> np.ndarray:
    c, s = np.cos(theta), np.sin(theta)
    return np.array([[1, 0, 0], [0, c, -s], [0, s, c]], dtype=np.float64)


def roty(theta: float) -> np.ndarray:
    c, s = np.cos(theta), np.sin(theta)
    return np.array([[c, 0, s], [0, 1, 0], [-s, 0, c]], dtype=np.float64)


def rotz(theta: float) -> np.ndarray:
    c, s = np.cos(theta), np.sin(theta)
    return np.array([[c, -s, 0], [s, c, 0], [0, 0, 1]], dtype=np.float64)


def random_rotate(rng: np.random.Generator | None = None) -> np.ndarray:
    """A uniformly-distributed-angle (not Haar-uniform) composite rotation,
    matching the reference's rotx@roty@rotz construction."""
    rng = rng or np.random.default_rng()
    a, b, c = rng.uniform(0, 2 * np.pi, size=3)
    return rotx(a) @ roty(b) @ rotz(c)


def random_rotate_y(rng: np.random.Generator | None = None) -> np.ndarray:
    rng = rng or np.random.default_rng()
    return roty(rng.uniform(0, 2 * np.pi))
