"""Rigid composite objects for the N-body generator: Stick and Hinge.

Re-owned physics of the reference simulator's composite bodies
(reference dataset_generation/nbody/physical_objects.py, system.py):

* ``Stick`` — two unit-mass balls joined by a rigid rod: state is the
  center of mass (xc, vc) plus an angular velocity wc; forces integrate
  into linear acceleration of the COM and angular acceleration via the
  torque / moment-of-inertia ratio; positions rotate about the COM by
  Rodrigues' formula each step. Rod length and the equality of the
  rod-parallel velocity components are exact invariants (checked).
* ``Hinge`` — three balls: a pivot (ball 0) with two rigid beams to
  balls 1 and 2, each beam carrying its own angular velocity. The pivot
  acceleration solves the 3x3 constraint system
  ``A a0 = F_total - w1 x v01 - w2 x v02 - (I - e1 e1^T) f1 - (I - e2 e2^T) f2``
  with ``A = I + e1 e1^T + e2 e2^T`` (beam-parallel force balance), then
  each beam rotates about the pivot. Beam lengths and the beam-parallel
  velocity matching are invariants.

The force field (clamped pairwise Coulomb q_i q_j / r^2) comes from the
caller (generate_dataset.CompositeSystem). All math is numpy float64;
``check`` raises on invariant drift beyond 1e-6 like the reference.
"""

from __future__ import annotations

import numpy as np

EPS = 1e-6


def rotation_matrix(theta: float, axis: np.ndarray) -> np.ndarray:
    """Rodrigues rotation by ``theta`` around unit vector ``axis``."""
    x, y, z = axis
    K = np.array([[0.0, -z, y], [z, 0.0, -x], [-y, x, 0.0]])
    return (np.eye(3) + np.sin(theta) * K
            + (1.0 - np.cos(theta)) * (K @ K))


def _project(v: np.ndarray, d: np.ndarray) -> np.ndarray:
    return (v @ d) / (d @ d) * d


class Isolated:
    """Single free ball — forward-Euler per-object update (the composite
    system integrates per object, matching the reference System)."""

    n_balls = 1
    type = "Isolated"

    def __init__(self, node_idx):
        self.node_idx = list(node_idx)

    def initialize(self, X, V):
        return X, V

    def update(self, X, V, F, dt):
        i = self.node_idx[0]
        V[i] = V[i] + F[i] * dt
        X[i] = X[i] + V[i] * dt
        return X, V

    def check(self, X, V):
        return True


class Stick:
    n_balls = 2
    type = "Stick"

    def __init__(self, node_idx):
        self.node_idx = list(node_idx)
        self.xc = self.vc = self.wc = None
        self.length = None

    def initialize(self, X, V):
        i, j = self.node_idx
        x0, x1 = X[i], X[j]
        v0, v1 = V[i], V[j]
        d = x1 - x0
        # make the rod-parallel velocity components equal (rigid rod)
        p0, p1 = _project(v0, d), _project(v1, d)
        mean_p = 0.5 * (p0 + p1)
        v0, v1 = v0 - p0 + mean_p, v1 - p1 + mean_p
        self.xc = 0.5 * (x0 + x1)
        self.vc = 0.5 * (v0 + v1)
        r0 = x0 - self.xc
        self.wc = np.cross(r0, v0 - self.vc) / (r0 @ r0)
        self.length = float(np.linalg.norm(d))
        V[i], V[j] = v0, v1
        return X, V

    def update(self, X, V, F, dt):
        i, j = self.node_idx
        r0, r1 = X[i] - self.xc, X[j] - self.xc
        f0, f1 = F[i], F[j]
        # COM translation (unit masses)
        self.vc = self.vc + 0.5 * (f0 + f1) * dt
        self.xc = self.xc + self.vc * dt
        # angular: beta = torque / inertia
        inertia = r0 @ r0 + r1 @ r1
        torque = np.cross(r0, f0) + np.cross(r1, f1)
        self.wc = self.wc + torque / inertia * dt
        wn = float(np.linalg.norm(self.wc))
        R = rotation_matrix(wn * dt, self.wc / wn) if wn > 0 else np.eye(3)
        r0, r1 = R @ r0, R @ r1
        X[i], X[j] = self.xc + r0, self.xc + r1
        V[i] = self.vc + np.cross(self.wc, r0)
        V[j] = self.vc + np.cross(self.wc, r1)
        return X, V

    def check(self, X, V):
        i, j = self.node_idx
        d = X[j] - X[i]
        assert abs(float(np.linalg.norm(d)) - self.length) < EPS
        assert np.abs(_project(V[i], d) - _project(V[j], d)).sum() < EPS


class Hinge:
    n_balls = 3
    type = "Hinge"

    def __init__(self, node_idx):
        self.node_idx = list(node_idx)
        self.w1 = self.w2 = None
        self.length1 = self.length2 = None

    def initialize(self, X, V):
        i, j, k = self.node_idx
        x0, x1, x2 = X[i], X[j], X[k]
        v0, v1, v2 = V[i], V[j], V[k]
        d1, d2 = x1 - x0, x2 - x0
        # beam-parallel velocity of each endpoint matches the pivot's
        v1 = _project(v0, d1) + (v1 - _project(v1, d1))
        v2 = _project(v0, d2) + (v2 - _project(v2, d2))
        self.w1 = np.cross(d1, v1 - v0) / (d1 @ d1)
        self.w2 = np.cross(d2, v2 - v0) / (d2 @ d2)
        self.length1 = float(np.linalg.norm(d1))
        self.length2 = float(np.linalg.norm(d2))
        V[j], V[k] = v1, v2
        return X, V

    def update(self, X, V, F, dt):
        i, j, k = self.node_idx
        x0 = X[i]
        r1, r2 = X[j] - x0, X[k] - x0
        v0 = V[i]
        v01, v02 = V[j] - v0, V[k] - v0
        f0, f1, f2 = F[i], F[j], F[k]
        e1 = (r1 / np.linalg.norm(r1)).reshape(3, 1)
        e2 = (r2 / np.linalg.norm(r2)).reshape(3, 1)
        P1, P2 = e1 @ e1.T, e2 @ e2.T
        A = np.eye(3) + P1 + P2
        rhs = (f0 + f1 + f2
               - np.cross(self.w1, v01) - np.cross(self.w2, v02)
               - (np.eye(3) - P1) @ f1 - (np.eye(3) - P2) @ f2)
        a0 = np.linalg.solve(A, rhs)
        v0 = v0 + a0 * dt
        x0 = x0 + v0 * dt
        self.w1 = self.w1 + np.cross(r1, f1 - a0) / (r1 @ r1) * dt
        self.w2 = self.w2 + np.cross(r2, f2 - a0) / (r2 @ r2) * dt
        for beam, (w, r, idx) in enumerate(
                ((self.w1, r1, j), (self.w2, r2, k))):
            wn = float(np.linalg.norm(w))
            R = rotation_matrix(wn * dt, w / wn) if wn > 0 else np.eye(3)
            rr = R @ r
            X[idx] = x0 + rr
            V[idx] = v0 + np.cross(w, rr)
        X[i], V[i] = x0, v0
        return X, V

    def check(self, X, V):
        i, j, k = self.node_idx
        d1, d2 = X[j] - X[i], X[k] - X[i]
        assert abs(float(np.linalg.norm(d1)) - self.length1) < EPS
        assert abs(float(np.linalg.norm(d2)) - self.length2) < EPS
        assert np.abs(_project(V[i], d1) - _project(V[j], d1)).sum() < EPS
        assert np.abs(_project(V[i], d2) - _project(V[k], d2)).sum() < EPS
