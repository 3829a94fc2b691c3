"""Simulation I/O helpers for the Fluid113K (SPlisHSPlasH) pipeline.

Re-owned counterpart of the reference's physics_data_helper.py: bgeo frame
discovery is pure python (unit-tested); bgeo array I/O needs the partio
package (ships with SPlisHSPlasH) and is import-guarded — this offline
image has neither, so everything above the partio boundary stays testable.
"""

from __future__ import annotations

import os
import re
from glob import glob

import numpy as np

_BGEO_RE = re.compile(r".*ParticleData_(.+)_(\d+)\.bgeo$")


def fluid_frame_id(path: str) -> int:
    m = _BGEO_RE.match(path)
    if m is None:
        raise ValueError(f"not a ParticleData bgeo path: {path}")
    return int(m.group(2))


def fluid_ids_in_dir(partio_dir: str):
    """Sorted fluid ids present in a partio output directory."""
    ids = set()
    for p in glob(os.path.join(partio_dir, "ParticleData*.bgeo")):
        m = _BGEO_RE.match(p)
        if m:
            ids.add(m.group(1))
    return sorted(ids)


def fluid_bgeo_files(partio_dir: str, fluid_id: str):
    """Frame-ordered bgeo files of one fluid."""
    files = glob(os.path.join(partio_dir, f"ParticleData_{fluid_id}_*.bgeo"))
    return sorted(files, key=fluid_frame_id)


def _require_partio():
    try:
        import partio  # type: ignore

        return partio
    except ImportError as exc:
        raise ImportError(
            "reading/writing .bgeo requires the partio python module "
            "(built with SPlisHSPlasH); it is not installed in this image"
        ) from exc


def numpy_from_bgeo(path: str):
    """(positions [N,3], velocities [N,3] or None), id-sorted when the
    file carries particle ids (stable across frames)."""
    partio = _require_partio()
    p = partio.read(path)
    pos_attr = p.attributeInfo("position")
    vel_attr = p.attributeInfo("velocity")
    id_attr = p.attributeInfo("trackid") or p.attributeInfo("id")
    n = p.numParticles()
    pos = np.array([p.get(pos_attr, i) for i in range(n)], dtype=np.float64)
    vel = None
    if vel_attr is not None:
        vel = np.array([p.get(vel_attr, i) for i in range(n)],
                       dtype=np.float64)
    if id_attr is not None:
        ids = np.array([p.get(id_attr, i)[0] for i in range(n)],
                       dtype=np.int64)
        order = np.argsort(ids)
        pos = pos[order]
        vel = vel[order] if vel is not None else None
    return (pos, vel) if vel is not None else (pos, None)


def write_bgeo_from_numpy(path: str, pos: np.ndarray, vel: np.ndarray):
    partio = _require_partio()
    pos = np.asarray(pos)
    vel = np.asarray(vel)
    if pos.shape != vel.shape or pos.shape[1] != 3:
        raise ValueError(f"bad shapes {pos.shape} / {vel.shape}")
    p = partio.create()
    pa = p.addAttribute("position", partio.VECTOR, 3)
    va = p.addAttribute("velocity", partio.VECTOR, 3)
    for i in range(pos.shape[0]):
        idx = p.addParticle()
        p.set(pa, idx, pos[i].astype(float))
        p.set(va, idx, vel[i].astype(float))
    partio.write(path, p)
