"""Offline dataset preprocessing with reference cache-key parity.

Dispatchers (reference datasets/process_dataset.py:32-58):
* ``process_dataset_edge_cutoff(data_config)`` → [train, valid, test] .pt
  paths (single-device mode, optional distance-percentile edge cutoff).
* ``process_dataset_distribute(rank, world_size, data_config)`` → this
  rank's [train, valid, test] partition files. Rank 0 writes every rank's
  file; other ranks return their paths immediately and wait at the caller's
  barrier (reference :317-318,462-463). The filesystem is the rank-0 →
  rank-i transport; the cache key bakes in the partition count
  (``..._{ws}_{outer:.3f}_{inner:.3f}_{max_samples}_{dt}__{rank}-{ws}.pt``).

Real-data readers are implemented where this image has the codec (nbody
.npy via numpy); Water-3D needs h5py and Fluid113K needs zstandard — absent
here, those readers raise with a clear message unless
``data_config.synthetic`` is set, in which case samples come from
data/synthetic.py at the same scales and are cached under the same key
scheme with a ``synthetic-`` dataset prefix.
"""

from __future__ import annotations

import os
import random
import zlib
from typing import List

import numpy as np
import torch

from ..ops import reference as ref_ops
from .graph import Data
from .partition import SPLITTERS
from . import synthetic as synth


def _is_synthetic(data_config) -> bool:
    return bool(data_config.get("synthetic", False))


def _num_samples(data_config, partition: str) -> int:
    scale = {"train": 1.0, "valid": 0.2, "test": 0.2}[partition]
    n = data_config.get("synthetic_samples", 8)
    return max(1, int(round(n * scale)))


def cutoff_edge(edge_index: torch.Tensor, pos: torch.Tensor,
                cutoff_rate: float) -> torch.Tensor:
    """Drop the longest ``cutoff_rate`` fraction of edges (reference :300-305)."""
    if cutoff_rate <= 0:
        return edge_index
    d = (pos[edge_index[0]] - pos[edge_index[1]]).norm(p=2, dim=1)
    keep = torch.sort(d).indices[: int(edge_index.size(1) * (1 - cutoff_rate))]
    return edge_index[:, keep]


# --------------------------------------------------------------------------
# cutoff (single-device) mode
# --------------------------------------------------------------------------

def process_dataset_edge_cutoff(data_config) -> List[str]:
    name = data_config.dataset_name
    if name == "nbody_100":
        return _process_nbody_cutoff(data_config)
    if name in ("protein", "Water-3D"):
        return _process_cutoff_generic(data_config)
    raise NotImplementedError(f"cutoff mode for dataset {name}")


def _cutoff_cache_names(data_config, extra: str) -> List[str]:
    name = data_config.dataset_name
    tag = "synthetic-" if _is_synthetic(data_config) else ""
    processed = os.path.join(data_config.data_dir, name, "processed")
    os.makedirs(processed, exist_ok=True)
    out = []
    for partition in ("train", "valid", "test"):
        fn = (f"{tag}{name}_{partition}_{data_config.radius}_"
              f"{data_config.cutoff_rate:.3f}_{data_config.max_samples}_"
              f"{extra}.pt")
        out.append(os.path.join(processed, fn))
    return out


def _process_nbody_cutoff(data_config) -> List[str]:
    """N-body charged-particle dataset (reference :61-125)."""
    paths = _cutoff_cache_names(
        data_config, f"{data_config.frame_0}_{data_config.frame_T}")
    for partition, path in zip(("train", "valid", "test"), paths):
        if os.path.exists(path):
            print(f"{path} exists!")
            continue
        if _is_synthetic(data_config):
            data = synth.make_cutoff_dataset(
                "nbody_100", _num_samples(data_config, partition),
                seed=zlib.crc32(partition.encode()) % (2 ** 31),
                radius=data_config.radius,
                cutoff_rate=data_config.cutoff_rate)
            torch.save(data, path)
            continue
        suffix = f"{partition}_charged100_0_0_1"
        base = os.path.join(data_config.data_dir, data_config.dataset_name)
        loc = torch.tensor(np.load(os.path.join(base, f"loc_{suffix}.npy"))).float()
        vel = torch.tensor(np.load(os.path.join(base, f"vel_{suffix}.npy"))).float()
        charges = torch.tensor(
            np.load(os.path.join(base, f"charges_{suffix}.npy"))).float()
        ms = data_config.max_samples
        loc, vel, charges = loc[:ms], vel[:ms], charges[:ms]
        loc_0, loc_t = loc[:, data_config.frame_0], loc[:, data_config.frame_T]
        vel_0 = vel[:, data_config.frame_0]
        data = []
        for k in range(charges.size(0)):
            p, v, q, t = loc_0[k], vel_0[k], charges[k], loc_t[k]
            ei = ref_ops.radius_graph(p, data_config.radius)
            ei = cutoff_edge(ei, p, data_config.cutoff_rate)
            ea = (p[ei[0]] - p[ei[1]]).norm(p=2, dim=1).unsqueeze(-1).repeat(1, 2)
            speed = v.pow(2).sum(1, keepdim=True).sqrt()
            x = torch.cat([speed, q / q.max()], dim=1)
            data.append(Data(
                x=x, pos=p, vel=v, attr=q, target=t,
                loc_mean=p.mean(0, keepdim=True), edge_index=ei, edge_attr=ea,
                special_nodes=torch.ones(p.size(0), dtype=torch.bool)))
        torch.save(data, path)
        print(f"{path} processed!")
    return paths


def _process_cutoff_generic(data_config) -> List[str]:
    """Water-3D / protein cutoff-mode datasets.

    Real readers need h5py (Water-3D, reference :225-297) or MDAnalysis
    (protein, reference :128-222) — not in this image; synthetic mode
    reproduces the scales."""
    paths = _cutoff_cache_names(data_config, f"{data_config.delta_t}")
    for partition, path in zip(("train", "valid", "test"), paths):
        if os.path.exists(path):
            print(f"{path} exists!")
            continue
        if not _is_synthetic(data_config):
            raise NotImplementedError(
                f"real-data reader for {data_config.dataset_name} requires "
                f"h5py/MDAnalysis (absent in this image); set data.synthetic "
                f"to use the synthetic generator at published scales")
        data = synth.make_cutoff_dataset(
            data_config.dataset_name, _num_samples(data_config, partition),
            seed=zlib.crc32(partition.encode()) % (2 ** 31), radius=data_config.radius,
            cutoff_rate=data_config.cutoff_rate)
        torch.save(data, path)
        print(f"{path} processed!")
    return paths


# --------------------------------------------------------------------------
# distribute (DistEGNN) mode
# --------------------------------------------------------------------------

def process_dataset_distribute(rank: int, world_size: int, data_config
                               ) -> List[str]:
    name = data_config.dataset_name
    if name not in ("Fluid113K", "Water-3D"):
        raise NotImplementedError(f"distribute mode for dataset {name}")
    tag = "synthetic-" if _is_synthetic(data_config) else ""
    processed = os.path.join(data_config.data_dir, name, "processed")

    def fname(partition, r):
        return os.path.join(
            processed,
            f"{tag}{name}_{data_config.split_mode}_{partition}_{world_size}_"
            f"{data_config.outer_radius:.3f}_{data_config.inner_radius:.3f}_"
            f"{data_config.max_samples}_{data_config.delta_t}"
            f"__{r}-{world_size}.pt")

    mine = [fname(p, rank) for p in ("train", "valid", "test")]
    if rank != 0:
        return mine  # rank 0 writes; caller barriers (reference :317-318)

    os.makedirs(processed, exist_ok=True)
    for partition in ("train", "valid", "test"):
        if all(os.path.exists(fname(partition, r)) for r in range(world_size)):
            print(f"{fname(partition, 0)} (and peers) exist!")
            continue
        if not _is_synthetic(data_config):
            raise NotImplementedError(
                f"real-data reader for {name} requires "
                f"{'zstandard' if name == 'Fluid113K' else 'h5py'} (absent "
                f"in this image); set data.synthetic for synthetic data at "
                f"published scales")
        per_rank = synth.make_distributed_dataset(
            name, _num_samples(data_config, partition), world_size,
            split_mode=data_config.split_mode,
            seed=zlib.crc32(partition.encode()) % (2 ** 31),
            outer_radius=data_config.outer_radius,
            inner_radius=data_config.inner_radius,
            n_override=data_config.get("synthetic_nodes", None))
        for i in range(world_size - 1):
            assert len(per_rank[i]) == len(per_rank[i + 1])
        for i in range(world_size):
            torch.save(per_rank[i], fname(partition, i))
            print(f"{fname(partition, i)} processed!")
    return mine
