"""Randomized SPlisHSPlasH scene construction for the Fluid113K dataset.

Re-owned counterpart of the reference's create_physics_scenes.py: random
fluid bodies sampled inside a bounding box, dropped with random velocities,
emitted as a SPlisHSPlasH scene.json (the Configuration/Simulation/
RigidBodies/FluidModels blocks use the reference's parameter values —
that is the compatibility surface the simulator consumes) plus the bgeo
initial states. The geometric core is dependency-free and unit-tested:

* ``load_obj`` / ``sample_obj_surface`` / ``sample_box_volume`` — pure
  python OBJ parsing and triangle-area-weighted surface sampling with
  normals (open3d's Poisson-disk sampler is used instead when importable).
* ``rasterize_points`` — occupancy voxelization of a particle cloud.
* ``find_fluid_positions`` — valid drop positions via an FFT
  cross-correlation of the fluid mask with the free-space mask (the
  reference scans every offset in a python triple loop; the correlation
  computes the same fits-entirely test for all offsets at once).

Actually RUNNING a simulation needs the SPlisHSPlasH binaries: set
``SPLISHSPLASH_BIN=/path/to/DynamicBoundarySimulator`` (VolumeSampling is
expected next to it). Without it, ``--scene-only`` still writes scene
dirs; volume particle sampling falls back to box-filling for box meshes.
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
from copy import deepcopy

import numpy as np

PARTICLE_RADIUS = 0.025
MAX_FLUID_START_VELOCITY_XZ = 4.0
MAX_FLUID_START_VELOCITY_Y = 1.0

# SPlisHSPlasH parameter blocks (values = reference
# create_physics_scenes.py:36-90; the simulator's input contract)
DEFAULT_CONFIGURATION = {
    "pause": False, "stopAt": 4.0, "particleRadius": 0.025,
    "numberOfStepsPerRenderUpdate": 1, "density0": 1000,
    "simulationMethod": 4, "gravitation": [0, -9.81, 0], "cflMethod": 0,
    "cflFactor": 1, "cflMaxTimeStepSize": 0.005, "maxIterations": 100,
    "maxError": 0.01, "maxIterationsV": 100, "maxErrorV": 0.1,
    "stiffness": 50000, "exponent": 7, "velocityUpdateMethod": 0,
    "enableDivergenceSolver": True, "enablePartioExport": True,
    "enableRigidBodyExport": True, "particleFPS": 50.0,
    "partioAttributes": "density;velocity",
}
DEFAULT_SIMULATION = {"contactTolerance": 0.0125}
DEFAULT_FLUID = {
    "surfaceTension": 0.2, "surfaceTensionMethod": 0, "viscosity": 0.01,
    "viscosityMethod": 3, "viscoMaxIter": 200, "viscoMaxError": 0.05,
}
DEFAULT_RIGIDBODY = {
    "translation": [0, 0, 0], "rotationAxis": [0, 1, 0],
    "rotationAngle": 0, "scale": [1.0, 1.0, 1.0],
    "color": [0.1, 0.4, 0.6, 1.0], "isDynamic": False, "isWall": True,
    "restitution": 0.6, "friction": 0.0, "collisionObjectType": 5,
    "collisionObjectScale": [1.0, 1.0, 1.0], "invertSDF": True,
}
DEFAULT_FLUIDMODEL = {"translation": [0.0, 0.0, 0.0],
                      "scale": [1.0, 1.0, 1.0]}


# ---------------------------------------------------------------------------
# geometry (pure python/numpy)

def load_obj(path: str):
    """Vertices [V,3] + triangle index array [T,3] from a wavefront OBJ
    (triangulates polygon faces as fans)."""
    verts, tris = [], []
    with open(path) as f:
        for line in f:
            parts = line.split()
            if not parts:
                continue
            if parts[0] == "v":
                verts.append([float(x) for x in parts[1:4]])
            elif parts[0] == "f":
                idx = [int(p.split("/")[0]) - 1 for p in parts[1:]]
                for i in range(1, len(idx) - 1):
                    tris.append([idx[0], idx[i], idx[i + 1]])
    return np.asarray(verts, dtype=np.float64), np.asarray(tris,
                                                           dtype=np.int64)


def sample_obj_surface(path: str, radius: float = PARTICLE_RADIUS,
                       rng: np.random.Generator = None):
    """Surface particles + outward-flipped normals for a boundary mesh.

    With open3d: Poisson-disk sampling (reference :135-145). Without:
    triangle-area-weighted uniform sampling at the same target density
    (1.9 * area / (pi r^2)) with face normals."""
    try:
        import open3d as o3d  # type: ignore

        mesh = o3d.io.read_triangle_mesh(path)
        num = int(1.9 * mesh.get_surface_area() / (np.pi * radius ** 2))
        pcd = mesh.sample_points_poisson_disk(num, use_triangle_normal=True)
        return (np.asarray(pcd.points, dtype=np.float32),
                -np.asarray(pcd.normals, dtype=np.float32))
    except ImportError:
        pass
    rng = rng or np.random.default_rng(0)
    v, t = load_obj(path)
    a, b, c = v[t[:, 0]], v[t[:, 1]], v[t[:, 2]]
    cross = np.cross(b - a, c - a)
    area2 = np.linalg.norm(cross, axis=1)
    total_area = area2.sum() / 2
    normals = cross / np.clip(area2, 1e-12, None)[:, None]
    num = int(1.9 * total_area / (np.pi * radius ** 2))
    probs = area2 / area2.sum()
    pick = rng.choice(len(t), size=num, p=probs)
    r1 = np.sqrt(rng.random(num))
    r2 = rng.random(num)
    pts = ((1 - r1)[:, None] * a[pick] + (r1 * (1 - r2))[:, None] * b[pick]
           + (r1 * r2)[:, None] * c[pick])
    # flip normals inward->outward convention like the reference (-n)
    return pts.astype(np.float32), (-normals[pick]).astype(np.float32)


def sample_box_volume(vmin, vmax, radius: float = PARTICLE_RADIUS):
    """Regular grid fill of an axis-aligned box at 2r spacing — the
    VolumeSampling fallback for box-shaped fluid/bound meshes."""
    axes = [np.arange(lo + radius, hi - radius + 1e-9, 2 * radius)
            for lo, hi in zip(vmin, vmax)]
    g = np.meshgrid(*axes, indexing="ij")
    return np.stack([x.reshape(-1) for x in g], axis=-1).astype(np.float32)


def obj_volume_to_particles(path: str, scale: float = 1.0,
                            radius: float = PARTICLE_RADIUS):
    """Volume-filled particles for a mesh: SPlisHSPlasH's VolumeSampling
    binary when configured, else grid fill of the mesh's bounding box
    (exact for the shipped Box/Fluid box meshes)."""
    vol_bin = _volume_sampling_bin()
    if vol_bin is not None:
        import tempfile

        from physics_data_helper import numpy_from_bgeo

        with tempfile.TemporaryDirectory() as td:
            out = os.path.join(td, "out.bgeo")
            subprocess.run([vol_bin, "-i", path, "-o", out,
                            "-r", str(radius), "-s", str(scale)],
                           check=True)
            return numpy_from_bgeo(out)[0]
    v, _ = load_obj(path)
    v = v * scale
    return sample_box_volume(v.min(0), v.max(0), radius)


def rasterize_points(points: np.ndarray, voxel_size: float,
                     particle_radius: float):
    """Occupancy voxelization: each particle marks the 8 voxels its
    +-radius corners land in (reference :148-180). Returns (grid origin
    index, voxel size, bool occupancy array)."""
    if not voxel_size > 2 * particle_radius:
        raise ValueError("voxel_size must exceed the particle diameter")
    pmin = (points - particle_radius).min(axis=0)
    pmax = (points + particle_radius).max(axis=0)
    amin = np.floor_divide(pmin, voxel_size).astype(np.int32)
    amax = np.floor_divide(pmax, voxel_size).astype(np.int32) + 1
    arr = np.zeros(tuple(amax - amin), dtype=bool)
    for sz in (-1, 1):
        for sy in (-1, 1):
            for sx in (-1, 1):
                off = np.array([sz, sy, sx]) * particle_radius
                idx = np.floor_divide(points + off,
                                      voxel_size).astype(np.int32) - amin
                arr[idx[:, 0], idx[:, 1], idx[:, 2]] = True
    return amin, voxel_size, arr


def find_fluid_positions(box_rast, fluid_rast):
    """All grid offsets where the fluid mask fits entirely inside the free
    space, restricted (like the reference :182-210) to offsets resting on
    the floor or with no valid offset below them in the same column.

    The fits-everywhere test for every offset is one FFT cross-correlation
    (free-space ⋆ fluid mask == fluid voxel count) instead of the
    reference's python loop over all offsets."""
    from scipy.signal import fftconvolve

    free = box_rast[2].astype(np.float64)
    fl = fluid_rast[2].astype(np.float64)
    fits = fftconvolve(free, fl[::-1, ::-1, ::-1], mode="valid")
    ok = fits > fl.sum() - 0.5
    if not ok.any():
        raise RuntimeError("fluid does not fit anywhere in the free space")
    # keep lowest-in-column candidates (reference's y-stacking rule)
    valid = np.zeros_like(ok)
    first_y = np.argmax(ok, axis=1)
    zi, xi = np.nonzero(ok.any(axis=1))
    valid[zi, first_y[zi, xi], xi] = True
    return valid


def place_fluid(box_rast, fluid_rast, rng: np.random.Generator):
    """Pick one valid position, carve the fluid out of the free space, and
    return the world-space offset for the fluid particles."""
    valid = find_fluid_positions(box_rast, fluid_rast)
    cand = np.stack(np.nonzero(valid), axis=-1)
    sel = cand[rng.integers(0, cand.shape[0])]
    p2 = sel + np.array(fluid_rast[2].shape)
    view = box_rast[2][sel[0]:p2[0], sel[1]:p2[1], sel[2]:p2[2]]
    box_rast[2][sel[0]:p2[0], sel[1]:p2[1], sel[2]:p2[2]] = \
        view & ~fluid_rast[2]
    world = (sel + box_rast[0]).astype(np.float64) * box_rast[1]
    return world - fluid_rast[0] * fluid_rast[1]


def random_rotation(rng: np.random.Generator) -> np.ndarray:
    """Uniform random rotation (Arvo's method, as the reference :92-119)."""
    theta = rng.random() * 2 * np.pi
    phi = rng.random() * 2 * np.pi
    z = rng.random()
    r = np.sqrt(z)
    V = np.array([np.sin(phi) * r, np.cos(phi) * r, np.sqrt(2.0 - z)])
    ct, st = np.cos(theta), np.sin(theta)
    Rz = np.array([[ct, st, 0], [-st, ct, 0], [0, 0, 1]])
    return (np.outer(V, V) - np.eye(3)) @ Rz


# ---------------------------------------------------------------------------
# scene assembly

def build_scene(seed: int, models_dir: str, out_dir: str,
                num_objects: int = 0, target_particles: int = 113140,
                default_viscosity: bool = False,
                write_bgeo: bool = True) -> dict:
    """Build one randomized scene directory (scene.json [+ box/fluid bgeo
    when partio is importable]). Returns the scene summary."""
    rng = np.random.default_rng(seed)
    from glob import glob as _glob

    boxes = sorted(_glob(os.path.join(models_dir, "Box*.obj")))
    fluids = sorted(_glob(os.path.join(models_dir, "Fluid*.obj")))
    if not boxes or not fluids:
        raise FileNotFoundError(f"no Box*/Fluid* obj models in {models_dir}")
    n_obj = num_objects or int(rng.choice([1, 2, 3]))

    box_obj = boxes[rng.integers(0, len(boxes))]
    bb_pts, bb_normals = sample_obj_surface(box_obj, rng=rng)
    bb_vol = obj_volume_to_particles(box_obj)
    box_rast = rasterize_points(np.concatenate([bb_vol, bb_pts]),
                                2.01 * PARTICLE_RADIUS, PARTICLE_RADIUS)
    from scipy.ndimage import binary_erosion

    box_rast = (box_rast[0], box_rast[1],
                binary_erosion(box_rast[2], structure=np.ones((3, 3, 3)),
                               iterations=3))

    objects = []
    for _ in range(n_obj):
        for _attempt in range(10):
            try:
                fobj = fluids[rng.integers(0, len(fluids))]
                pts = obj_volume_to_particles(
                    fobj, scale=float(rng.uniform(0.9, 1.0)))
                pts = pts @ random_rotation(rng)
                frast = rasterize_points(pts, 2.01 * PARTICLE_RADIUS,
                                         PARTICLE_RADIUS)
                pts = pts + place_fluid(box_rast, frast, rng)
                vel = np.zeros_like(pts)
                vel[:, 0] = rng.uniform(-MAX_FLUID_START_VELOCITY_XZ,
                                        MAX_FLUID_START_VELOCITY_XZ)
                vel[:, 2] = rng.uniform(-MAX_FLUID_START_VELOCITY_XZ,
                                        MAX_FLUID_START_VELOCITY_XZ)
                vel[:, 1] = rng.uniform(-MAX_FLUID_START_VELOCITY_Y,
                                        MAX_FLUID_START_VELOCITY_Y)
                objects.append({
                    "positions": pts, "velocities": vel,
                    "density": (1000.0 if default_viscosity
                                else float(rng.uniform(500, 2000))),
                    "viscosity": (0.01 if default_viscosity else
                                  float(rng.exponential(1 / 20) + 0.01)),
                })
                break
            except RuntimeError:
                continue

    sim_dir = os.path.join(out_dir, f"sim_{seed:04d}")
    os.makedirs(sim_dir, exist_ok=True)
    scene = {"Configuration": dict(DEFAULT_CONFIGURATION),
             "Simulation": dict(DEFAULT_SIMULATION),
             "RigidBodies": [], "FluidModels": []}
    rb = deepcopy(DEFAULT_RIGIDBODY)
    rb["id"] = 1
    rb["geometryFile"] = "box.obj"
    rb["resolutionSDF"] = [64, 64, 64]
    scene["RigidBodies"].append(rb)
    import shutil

    shutil.copyfile(box_obj, os.path.join(sim_dir, "box.obj"))
    if write_bgeo:
        try:
            from physics_data_helper import write_bgeo_from_numpy

            write_bgeo_from_numpy(os.path.join(sim_dir, "box.bgeo"),
                                  bb_pts, bb_normals)
        except ImportError:
            np.savez(os.path.join(sim_dir, "box.npz"), pos=bb_pts,
                     normals=bb_normals)
    for i, obj in enumerate(objects):
        fid = f"fluid{i}"
        fl = deepcopy(DEFAULT_FLUID)
        fl["viscosity"] = obj["viscosity"]
        fl["density0"] = obj["density"]
        scene[fid] = fl
        fm = deepcopy(DEFAULT_FLUIDMODEL)
        fm["id"] = fid
        fm["particleFile"] = f"{fid}.bgeo"
        scene["FluidModels"].append(fm)
        if write_bgeo:
            try:
                from physics_data_helper import write_bgeo_from_numpy

                write_bgeo_from_numpy(os.path.join(sim_dir, f"{fid}.bgeo"),
                                      obj["positions"], obj["velocities"])
            except ImportError:
                np.savez(os.path.join(sim_dir, f"{fid}.npz"),
                         pos=obj["positions"], vel=obj["velocities"])
    with open(os.path.join(sim_dir, "scene.json"), "w") as f:
        json.dump(scene, f, indent=4)
    total = sum(o["positions"].shape[0] for o in objects)
    return {"sim_dir": sim_dir, "num_fluid_particles": int(total),
            "num_objects": len(objects)}


def _simulator_bin():
    return os.environ.get("SPLISHSPLASH_BIN")


def _volume_sampling_bin():
    sim = _simulator_bin()
    if sim is None:
        return None
    cand = os.path.join(os.path.dirname(sim), "VolumeSampling")
    return cand if os.path.isfile(cand) else None


def run_simulator(scene_json: str, output_dir: str):
    sim = _simulator_bin()
    if sim is None:
        raise RuntimeError(
            "set SPLISHSPLASH_BIN=/path/to/DynamicBoundarySimulator to run "
            "simulations (scene dirs were still written)")
    subprocess.run([sim, "--no-cache", "--no-gui", "--no-initial-pause",
                    "--output-dir", output_dir, scene_json], check=True)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--output", required=True)
    ap.add_argument("--seed-start", type=int, default=1)
    ap.add_argument("--num-scenes", type=int, default=1)
    ap.add_argument("--num-objects", type=int, default=0)
    ap.add_argument("--models", type=str,
                    default=os.path.join(os.path.dirname(
                        os.path.abspath(__file__)), "models"))
    ap.add_argument("--scene-only", action="store_true",
                    help="write scene dirs without running the simulator")
    args = ap.parse_args()
    for seed in range(args.seed_start, args.seed_start + args.num_scenes):
        info = build_scene(seed, args.models, args.output,
                           num_objects=args.num_objects)
        print(json.dumps(info))
        if not args.scene_only:
            run_simulator(os.path.join(info["sim_dir"], "scene.json"),
                          info["sim_dir"])


if __name__ == "__main__":
    main()
