"""Charged-particle N-body trajectory generator (offline).

Re-owned equivalent of the reference's dataset_generation/nbody pipeline
(generate_dataset.py + system.py): isolated charged particles initialized
in Gaussian clusters, Coulomb-like pairwise forces (q_i q_j / r^2 along
r-hat, force-clamped), leapfrog integration, trajectories subsampled every
``--sample-freq`` steps. Output file naming matches the training
pipeline's reader (datasets preprocessing expects
``loc_{train,valid,test}_charged{N}_{n_stick}_{n_hinge}_{clusters}{suffix}.npy``
etc.).

Usage (the reference's headline run):
    python generate_dataset.py --num-train 5000 --seed 43 \
        --n_isolated 100 --clusters 10 --path ../../data/n_body_system

Stick/Hinge composite objects (rigid rods / hinged beams, reference
physical_objects.py + system.py) are provided by rigid.py: any of
``--n_stick/--n_hinge`` > 0 switches to the per-object CompositeSystem
integrator (forward-Euler per body, rigid-constraint preserving) while
isolated-only runs keep the vectorized leapfrog path.
"""

import argparse
import os

import numpy as np

try:
    from joblib import Parallel, delayed

    HAVE_JOBLIB = True
except ImportError:  # pragma: no cover
    HAVE_JOBLIB = False


class ChargedSystem:
    def __init__(self, n_particles, clusters=1, delta_t=0.001, loc_std=1.0,
                 vel_norm=0.5, interaction_strength=1.0, box_size=None,
                 rng=None):
        self.rng = rng or np.random.default_rng()
        self.n = n_particles
        self.delta_t = delta_t
        self.max_f = 0.1 / delta_t
        self.box_size = box_size
        self.loc_std = loc_std * (float(self.n) / 5.0) ** (1 / 3) + 0.1
        self.strength = interaction_strength

        self.charges = self.rng.choice([1.0, -1.0], size=(self.n, 1))
        self.qq = self.charges @ self.charges.T

        if clusters == 1:
            centers = np.zeros((1, 3))
        elif clusters == 3:
            centers = self.rng.uniform(-10 * clusters, 10 * clusters,
                                       size=(clusters, 3))
        else:
            centers = self.rng.uniform(-3 * clusters, 3 * clusters,
                                       size=(clusters, 3))
        which = self.rng.integers(0, len(centers), size=self.n)
        self.x = (self.rng.standard_normal((self.n, 3)) * self.loc_std
                  + centers[which])
        v = self.rng.standard_normal((self.n, 3))
        self.v = v / np.linalg.norm(v, axis=1, keepdims=True) * vel_norm

    def forces(self, x):
        d = x[:, None, :] - x[None, :, :]                  # [n, n, 3]
        r2 = (d ** 2).sum(-1)
        np.fill_diagonal(r2, 1.0)
        inv_r3 = r2 ** -1.5
        np.fill_diagonal(inv_r3, 0.0)
        f = (self.strength * self.qq * inv_r3)[:, :, None] * d
        f = f.sum(axis=1)
        return np.clip(f, -self.max_f, self.max_f)

    def trajectory(self, length, sample_freq):
        t_out = length // sample_freq
        loc = np.zeros((t_out, self.n, 3))
        vel = np.zeros((t_out, self.n, 3))
        x, v = self.x, self.v
        # leapfrog
        f = self.forces(x)
        v = v + 0.5 * self.delta_t * f
        k = 0
        for step in range(length):
            if step % sample_freq == 0:
                loc[k], vel[k] = x, v
                k += 1
            x = x + self.delta_t * v
            if self.box_size is not None:
                over = np.abs(x) > self.box_size
                v[over] *= -1
                x = np.clip(x, -self.box_size, self.box_size)
            f = self.forces(x)
            v = v + self.delta_t * f
        return loc, vel, self.charges


class CompositeSystem(ChargedSystem):
    """Mixed Isolated/Stick/Hinge system (reference system.py semantics):
    the shared clamped-Coulomb force field drives per-object updates that
    preserve each body's rigid constraints (rigid.py). Ball count is
    n_isolated + 2*n_stick + 3*n_hinge; object membership is drawn
    randomly without replacement like the reference (:66-90)."""

    def __init__(self, n_isolated, n_stick, n_hinge, **kw):
        try:
            from . import rigid
        except ImportError:       # script-style / loose-module execution
            import importlib.util

            _spec = importlib.util.spec_from_file_location(
                "nbody_rigid",
                os.path.join(os.path.dirname(os.path.abspath(__file__)),
                             "rigid.py"))
            rigid = importlib.util.module_from_spec(_spec)
            _spec.loader.exec_module(rigid)
        n = n_isolated + 2 * n_stick + 3 * n_hinge
        super().__init__(n, **kw)
        rest = list(self.rng.permutation(n))
        self.objects = []
        for _ in range(n_isolated):
            self.objects.append(rigid.Isolated([rest.pop()]))
        for _ in range(n_stick):
            self.objects.append(rigid.Stick([rest.pop(), rest.pop()]))
        for _ in range(n_hinge):
            self.objects.append(
                rigid.Hinge([rest.pop(), rest.pop(), rest.pop()]))
        for obj in self.objects:
            self.x, self.v = obj.initialize(self.x, self.v)

    def trajectory(self, length, sample_freq):
        t_out = length // sample_freq
        loc = np.zeros((t_out, self.n, 3))
        vel = np.zeros((t_out, self.n, 3))
        k = 0
        for step in range(length):
            if step % sample_freq == 0:
                loc[k], vel[k] = self.x, self.v
                k += 1
            f = self.forces(self.x)
            for obj in self.objects:
                self.x, self.v = obj.update(self.x, self.v, f,
                                            self.delta_t)
        return loc, vel, self.charges

    def check(self):
        for obj in self.objects:
            obj.check(self.x, self.v)


def simulate_one(seed, args):
    rng = np.random.default_rng(seed)
    if args.n_stick or args.n_hinge:
        sys_ = CompositeSystem(args.n_isolated, args.n_stick, args.n_hinge,
                               clusters=args.clusters,
                               box_size=args.box_size, rng=rng)
    else:
        sys_ = ChargedSystem(args.n_isolated, clusters=args.clusters,
                             box_size=args.box_size, rng=rng)
    return sys_.trajectory(args.length, args.sample_freq)


def generate(args, partition, count, base_seed):
    seeds = [base_seed + i for i in range(count)]
    if HAVE_JOBLIB and args.n_workers > 1:
        results = Parallel(n_jobs=args.n_workers)(
            delayed(simulate_one)(s, args) for s in seeds)
    else:
        results = [simulate_one(s, args) for s in seeds]
    loc = np.stack([r[0] for r in results])    # [S, T, n, 3]
    vel = np.stack([r[1] for r in results])
    charges = np.stack([r[2] for r in results])  # [S, n, 1]
    suffix = (f"_charged{args.n_isolated}_{args.n_stick}_{args.n_hinge}_"
              f"{args.clusters}{args.suffix}")
    os.makedirs(args.path, exist_ok=True)
    np.save(os.path.join(args.path, f"loc_{partition}{suffix}.npy"), loc)
    np.save(os.path.join(args.path, f"vel_{partition}{suffix}.npy"), vel)
    np.save(os.path.join(args.path, f"charges_{partition}{suffix}.npy"),
            charges)
    print(f"{partition}: {loc.shape} -> {args.path}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--simulation", type=str, default="charged")
    ap.add_argument("--path", type=str, default="data")
    ap.add_argument("--num-train", type=int, default=10000)
    ap.add_argument("--num-valid", type=int, default=2000)
    ap.add_argument("--num-test", type=int, default=2000)
    ap.add_argument("--length", type=int, default=5000)
    ap.add_argument("--sample-freq", type=int, default=100)
    ap.add_argument("--n_isolated", type=int, default=5)
    ap.add_argument("--n_stick", type=int, default=0)
    ap.add_argument("--n_hinge", type=int, default=0)
    ap.add_argument("--clusters", type=int, default=1)
    ap.add_argument("--seed", type=int, default=42)
    ap.add_argument("--suffix", type=str, default="")
    ap.add_argument("--n_workers", type=int, default=1)
    ap.add_argument("--box_size", type=float, default=None)
    args = ap.parse_args()
    np.random.seed(args.seed)
    generate(args, "train", args.num_train, args.seed)
    generate(args, "valid", args.num_valid, args.seed + 10_000_000)
    generate(args, "test", args.num_test, args.seed + 20_000_000)


if __name__ == "__main__":
    main()
