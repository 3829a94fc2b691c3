"""Convert simulated SPlisHSPlasH scene directories into the training
record format: 16 chunk files ``sim_NNNN_CC.msgpack.zst`` per simulation.

Re-owned counterpart of the reference's create_physics_records.py
(:14-110): per scene, the partio frame outputs are id-sorted, fluids
concatenated per frame, per-particle viscosity/mass columns attached
(mass scaled by (2 * PARTICLE_RADIUS)^3 like the reference :87), frames
split into 16 chunk lists and packed. The array→record core
(``frames_to_chunks``) is pure python and unit-tested; bgeo reading needs
partio (guarded) and compression uses the shared msgpack_numpy-compatible
writer in distegnn_amd.data.readers.fluid113k (zstd optional —
uncompressed .msgpack files are accepted by the training reader).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
from glob import glob

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)),
                                "..", ".."))

from physics_data_helper import (fluid_bgeo_files, fluid_ids_in_dir,
                                 numpy_from_bgeo)

PARTICLE_RADIUS = 0.025


def frames_to_chunks(frames, num_chunks: int = 16):
    """frames: list of dicts with ``pos``/``vel`` [N,3] and per-particle
    ``viscosity``/``m`` — returns ``num_chunks`` frame lists (np.array_split
    semantics, matching the reference's sublists)."""
    idx_lists = np.array_split(np.arange(len(frames)), num_chunks)
    return [[frames[i] for i in idx] for idx in idx_lists]


def scene_frames(scene_dir: str):
    """Assemble per-frame dicts from a simulated scene directory."""
    with open(os.path.join(scene_dir, "scene.json")) as f:
        scene = json.load(f)
    partio_dir = os.path.join(scene_dir, "partio")
    fluid_ids = fluid_ids_in_dir(partio_dir)
    if not fluid_ids:
        raise FileNotFoundError(f"no ParticleData bgeo files under "
                                f"{partio_dir} — did the simulator run?")
    per_fluid = {k: fluid_bgeo_files(partio_dir, k) for k in fluid_ids}
    n_frames = {k: len(v) for k, v in per_fluid.items()}
    if len(set(n_frames.values())) != 1:
        raise ValueError(f"inconsistent frame counts per fluid: {n_frames}")
    frames = []
    for fi in range(next(iter(n_frames.values()))):
        pos, vel, visc, mass = [], [], [], []
        for fid in fluid_ids:
            p, v = numpy_from_bgeo(per_fluid[fid][fi])
            pos.append(p)
            vel.append(v)
            visc.append(np.full(p.shape[0], scene[fid]["viscosity"],
                                dtype=np.float32))
            mass.append(np.full(p.shape[0], scene[fid]["density0"],
                                dtype=np.float32))
        frames.append({
            "frame_id": np.int64(fi),
            "pos": np.concatenate(pos).astype(np.float32),
            "vel": np.concatenate(vel).astype(np.float32),
            "viscosity": np.concatenate(visc),
            "m": (np.concatenate(mass)
                  * (2 * PARTICLE_RADIUS) ** 3).astype(np.float32),
        })
    return frames


def write_records(frames, out_prefix: str, num_chunks: int = 16,
                  compress: bool = True):
    """Write ``{out_prefix}_CC.msgpack[.zst]`` chunk files readable by
    distegnn_amd.data.readers.fluid113k.read_fluid_sim."""
    from distegnn_amd.data.readers.fluid113k import write_chunk

    suffix = ".msgpack.zst" if compress else ".msgpack"
    paths = []
    for ci, chunk in enumerate(frames_to_chunks(frames, num_chunks)):
        path = f"{out_prefix}_{ci:02d}{suffix}"
        write_chunk(path, chunk)
        paths.append(path)
    return paths


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--input", required=True,
                    help="directory of simulated sim_NNNN scene dirs")
    ap.add_argument("--output", required=True)
    ap.add_argument("--splits", type=int, default=16)
    ap.add_argument("--no-compress", action="store_true",
                    help="write plain .msgpack (no zstandard installed)")
    args = ap.parse_args()
    os.makedirs(args.output, exist_ok=True)
    for scene_dir in sorted(glob(os.path.join(args.input, "*"))):
        if not os.path.isdir(scene_dir):
            continue
        name = os.path.basename(scene_dir)
        frames = scene_frames(scene_dir)
        paths = write_records(frames, os.path.join(args.output, name),
                              args.splits, compress=not args.no_compress)
        print(f"{name}: {len(frames)} frames -> {len(paths)} chunks")


if __name__ == "__main__":
    main()
