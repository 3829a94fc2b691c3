"""Iterate generated fluid records as (frame_t, frame_t+window) samples.

Generation-side counterpart of the training reader
(distegnn_amd/data/readers/fluid113k.py) — reference
dataset_generation/Fluid113K/dataset_reader_physics.py exposes the same
functionality as a tensorpack DataFlow over the msgpack.zst chunks
(PhysicsSimDataFlow: windowed frame pairs, optional random rotation,
shuffling). This is a dependency-free generator version: no dataflow /
tensorpack, same record schema, same window semantics.
"""

from __future__ import annotations

import glob
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__)))))

from distegnn_amd.data.readers.fluid113k import _read_chunk  # noqa: E402


def _random_rotation_matrix(rng: np.random.Generator) -> np.ndarray:
    """Uniform random 3D rotation (QR of a gaussian matrix, det fixed)."""
    q, r = np.linalg.qr(rng.normal(size=(3, 3)))
    q *= np.sign(np.diag(r))
    if np.linalg.det(q) < 0:
        q[:, 0] = -q[:, 0]
    return q


def read_sim_frames(prefix: str):
    """All frames of one simulation written by create_physics_records.py.

    ``prefix`` is the path prefix of its chunk files
    (``{prefix}_{chunk:02d}.msgpack[.zst]``, chunks in order)."""
    frames = []
    paths = sorted(glob.glob(prefix + "_*.msgpack*"))
    if not paths:
        raise FileNotFoundError(f"no chunks match {prefix}_*.msgpack*")
    for p in paths:
        frames.extend(_read_chunk(p))
    return frames


def iter_sim_samples(files, window: int = 2, random_rotation: bool = False,
                     shuffle: bool = False, seed: int = 0):
    """Yield dicts of windowed frames across simulations.

    files: list of chunk-path prefixes (one per simulation).
    Each yielded sample has keys ``pos0..pos{window-1}``,
    ``vel0..vel{window-1}`` plus the first frame's static fields —
    the reference DataFlow's schema (dataset_reader_physics.py:34-77).
    """
    if window < 1:
        raise ValueError(f"window must be >= 1, got {window}")
    rng = np.random.default_rng(seed)
    order = np.arange(len(files))
    if shuffle:
        rng.shuffle(order)
    for fi in order:
        frames = read_sim_frames(files[fi])
        starts = np.arange(len(frames) - window + 1)
        if shuffle:
            rng.shuffle(starts)
        for s in starts:
            rot = (_random_rotation_matrix(rng) if random_rotation
                   else None)
            sample = {"sim": int(fi), "frame0": int(s)}
            first = frames[s]
            for k, v in first.items():
                if k not in ("pos", "vel"):
                    sample[k] = v
            for w in range(window):
                pos = np.asarray(frames[s + w]["pos"], dtype=np.float32)
                vel = np.asarray(frames[s + w]["vel"], dtype=np.float32)
                if rot is not None:
                    pos = pos @ rot.T
                    vel = vel @ rot.T
                sample[f"pos{w}"] = pos
                sample[f"vel{w}"] = vel
            yield sample


if __name__ == "__main__":
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--input", required=True,
                    help="directory of *_NN.msgpack[.zst] record chunks")
    ap.add_argument("--window", type=int, default=2)
    args = ap.parse_args()
    prefixes = sorted({p.rsplit("_", 1)[0]
                       for p in glob.glob(os.path.join(args.input,
                                                       "*.msgpack*"))})
    n = 0
    for sample in iter_sim_samples(prefixes, window=args.window):
        n += 1
    print(f"{len(prefixes)} sims, {n} window-{args.window} samples")
