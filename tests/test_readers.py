"""Real-data reader tests with synthesized fixture files (VERDICT round-1
gap #1): a 3-trajectory Water-3D HDF5 (read via hdf5lite — no h5py in this
image), a 16-chunk Fluid113K msgpack sim, and the protein test-split
rot/trans augmentation — all driven end-to-end through the preprocessing
dispatchers with reference cache keys.
"""

import os
import random

import numpy as np
import pytest
import torch

from distegnn_amd.data import preprocess
from distegnn_amd.data.readers import hdf5lite
from distegnn_amd.data.readers.fluid113k import (_decode_numpy, encode_numpy,
                                                 read_fluid_sim, write_chunk)
from distegnn_amd.utils import AttrDict


# ---------------------------------------------------------------------------
# hdf5lite

def _traj_tree(n_traj=3, T=30, N=40, seed=0):
    rng = np.random.default_rng(seed)
    return {
        f"traj{i:04d}": {
            "particle_type": np.full(N, 5, dtype=np.int64),
            "position": rng.random((T, N, 3), dtype=np.float32) * 0.4,
        } for i in range(n_traj)
    }


def test_hdf5lite_roundtrip(tmp_path):
    tree = _traj_tree()
    path = str(tmp_path / "fix.h5")
    hdf5lite.write(path, tree)
    with hdf5lite.File(path) as f:
        assert sorted(f.keys()) == sorted(tree.keys())
        for k, grp in tree.items():
            for ds, arr in grp.items():
                got = np.array(f[k][ds])
                assert got.dtype == arr.dtype
                assert np.array_equal(got, arr)


def test_hdf5lite_h5py_crosscheck(tmp_path):
    h5py = pytest.importorskip("h5py")
    tree = _traj_tree(n_traj=2)
    ours = str(tmp_path / "ours.h5")
    hdf5lite.write(ours, tree)
    # h5py reads our file
    with h5py.File(ours, "r") as f:
        for k, grp in tree.items():
            for ds, arr in grp.items():
                assert np.array_equal(np.array(f[k][ds]), arr)
    # we read h5py's file
    theirs = str(tmp_path / "theirs.h5")
    with h5py.File(theirs, "w") as f:
        for k, grp in tree.items():
            g = f.create_group(k)
            for ds, arr in grp.items():
                g.create_dataset(ds, data=arr)
    with hdf5lite.File(theirs) as f:
        for k, grp in tree.items():
            for ds, arr in grp.items():
                assert np.array_equal(np.array(f[k][ds]), arr)


# ---------------------------------------------------------------------------
# Water-3D end-to-end through the dispatcher

def _water_cfg(tmp_path, **kw):
    base = {
        "data_dir": str(tmp_path / "data"), "dataset_name": "Water-3D",
        "max_samples": 8, "batch_size": 2, "delta_t": 5, "radius": 0.12,
        "cutoff_rate": 0.0, "accelerate_mode": "cutoff_edges",
        "synthetic": False,
    }
    base.update(kw)
    return AttrDict(base)


def _write_water_fixture(cfg, n_traj=3, T=30, N=40):
    d = os.path.join(cfg.data_dir, cfg.dataset_name)
    os.makedirs(d, exist_ok=True)
    tree = _traj_tree(n_traj, T, N)
    for part in ("train", "valid", "test"):
        hdf5lite.write(os.path.join(d, f"{part}.h5"), tree)
    return tree


def test_water3d_cutoff_real_reader(tmp_path):
    cfg = _water_cfg(tmp_path)
    _write_water_fixture(cfg)
    random.seed(0)
    paths = preprocess.process_dataset_edge_cutoff(cfg)
    for p in paths:
        assert os.path.exists(p)
        # reference cache-key template (no synthetic- tag for real data)
        assert "Water-3D" in os.path.basename(p)
        assert "synthetic" not in os.path.basename(p)
        data = torch.load(p, weights_only=False)
        assert 0 < len(data) <= cfg.max_samples
        s = data[0]
        assert s.pos.shape == (40, 3) and s.target.shape == (40, 3)
        assert s.x.shape == (40, 2)
        # x = [|v|, type/max type]; fixture types are all 5 -> column of 1
        assert torch.allclose(s.x[:, 1], torch.ones(40))
        assert torch.allclose(
            s.x[:, 0], s.vel.pow(2).sum(-1).sqrt(), atol=1e-6)
        # edge_attr = distance repeated x2, all within radius
        d = (s.pos[s.edge_index[0]] - s.pos[s.edge_index[1]]).norm(dim=1)
        assert torch.allclose(s.edge_attr[:, 0], d, atol=1e-6)
        assert torch.allclose(s.edge_attr[:, 0], s.edge_attr[:, 1])
        assert (d <= cfg.radius + 1e-6).all()
        assert (s.edge_index[0] != s.edge_index[1]).all()


def test_water3d_distribute_real_reader(tmp_path):
    cfg = _water_cfg(tmp_path, accelerate_mode="distribute",
                     outer_radius=0.12, inner_radius=0.12,
                     split_mode="random", max_samples=4)
    _write_water_fixture(cfg)
    random.seed(0)
    ws = 2
    paths0 = preprocess.process_dataset_distribute(0, ws, cfg)
    paths1 = preprocess.process_dataset_distribute(1, ws, cfg)
    for p0, p1 in zip(paths0, paths1):
        d0 = torch.load(p0, weights_only=False)
        d1 = torch.load(p1, weights_only=False)
        assert len(d0) == len(d1) > 0
        for a, b in zip(d0, d1):
            assert a.pos.size(0) + b.pos.size(0) == 40
            assert torch.allclose(a.loc_mean, b.loc_mean)  # global mean


# ---------------------------------------------------------------------------
# Fluid113K

def test_msgpack_numpy_codec_roundtrip():
    import msgpack

    arr = np.arange(12, dtype=np.float32).reshape(3, 4)
    packed = msgpack.packb({"pos": arr, "m": np.float64(0.125)},
                           default=encode_numpy, use_bin_type=True)
    out = msgpack.unpackb(packed, raw=False, object_hook=_decode_numpy,
                          strict_map_key=False)
    assert np.array_equal(out["pos"], arr)
    assert out["m"] == 0.125


def _write_fluid_fixture(base_dir, idx=1, n_chunks=16, frames_per_chunk=2,
                         N=50, seed=0):
    os.makedirs(base_dir, exist_ok=True)
    rng = np.random.default_rng(seed)
    visc = np.full(N, 0.01, dtype=np.float32)
    mass = np.full(N, 0.125, dtype=np.float32)
    t = 0
    for c in range(n_chunks):
        frames = []
        for _ in range(frames_per_chunk):
            frames.append({
                "pos": rng.random((N, 3), dtype=np.float32) * 0.5,
                "vel": rng.standard_normal((N, 3)).astype(np.float32) * 0.01,
                "viscosity": visc, "m": mass, "frame_id": t,
            })
            t += 1
    # uncompressed .msgpack chunks (no zstandard in this image; the
    # reader prefers .zst and falls back to .msgpack)
            pass
        write_chunk(os.path.join(base_dir,
                                 f"sim_{idx:04d}_{c:02d}.msgpack"), frames)
    return t


def test_read_fluid_sim(tmp_path):
    total = _write_fluid_fixture(str(tmp_path), idx=1)
    pos, vel, visc, mass = read_fluid_sim(str(tmp_path), 1)
    assert pos.shape == (total, 50, 3)
    assert vel.shape == (total, 50, 3)
    assert visc.shape == (50,) and mass.shape == (50,)
    assert torch.allclose(mass, torch.full((50,), 0.125))


def test_fluid113k_distribute_real_reader(tmp_path):
    cfg = AttrDict({
        "data_dir": str(tmp_path / "data"), "dataset_name": "Fluid113K",
        "max_samples": 3, "delta_t": 5, "outer_radius": 0.2,
        "inner_radius": 0.2, "split_mode": "random",
        "accelerate_mode": "distribute", "synthetic": False,
    })
    base = os.path.join(cfg.data_dir, cfg.dataset_name)
    # valid/test splits start at sim 101/121 — write sims for all splits
    for idx in list(range(1, 4)) + [101, 121]:
        _write_fluid_fixture(base, idx=idx)
    random.seed(0)
    ws = 2
    paths = preprocess.process_dataset_distribute(0, ws, cfg)
    for p in paths:
        data = torch.load(p, weights_only=False)
        assert 0 < len(data) <= cfg.max_samples
        s = data[0]
        # fluid features: x = [viscosity, mass, |v|]
        assert s.x.shape[1] == 3
        assert torch.allclose(s.x[:, 1], torch.full((s.x.size(0),), 0.125))
        assert s.attr.shape[1] == 2


# ---------------------------------------------------------------------------
# protein test-split augmentation (synthetic path: no MDAnalysis needed)

def _protein_cfg(tmp_path, **kw):
    base = {
        "data_dir": str(tmp_path / "data"), "dataset_name": "protein",
        "max_samples": 500, "delta_t": 3, "radius": 10.0,
        "cutoff_rate": 0.0, "accelerate_mode": "cutoff_edges",
        "synthetic": True, "synthetic_samples": 4, "backbone": True,
        "test_rot": False, "test_trans": False,
    }
    base.update(kw)
    return AttrDict(base)


def _load_test_split(cfg):
    paths = preprocess.process_dataset_edge_cutoff(cfg)
    return torch.load(paths[2], weights_only=False)


def test_protein_test_rot_trans_augmentation(tmp_path):
    torch.manual_seed(0)
    np.random.seed(0)
    plain = _load_test_split(_protein_cfg(tmp_path / "a"))
    torch.manual_seed(0)
    np.random.seed(0)
    aug = _load_test_split(_protein_cfg(tmp_path / "b", test_rot=True,
                                        test_trans=True))
    assert len(plain) == len(aug)
    for p, a in zip(plain, aug):
        # rigid motion: coordinates moved ...
        assert not torch.allclose(p.pos, a.pos)
        # ... but distances (edge structure, speeds) preserved; neighbor
        # ORDER may differ (kD-tree traversal), so compare canonical sets
        n = p.pos.size(0)
        key_p = (p.edge_index[0] * n + p.edge_index[1]).sort().values
        key_a = (a.edge_index[0] * n + a.edge_index[1]).sort().values
        assert torch.equal(key_p, key_a)
        assert torch.allclose(p.edge_attr.sort(0).values,
                              a.edge_attr.sort(0).values, atol=1e-4)
        assert torch.allclose(p.x[:, 0], a.x[:, 0], atol=1e-4)
        # the same rigid motion maps pos AND target (equivariance of the
        # ground truth): relative displacement norms match
        assert torch.allclose((p.target - p.pos).norm(dim=1),
                              (a.target - a.pos).norm(dim=1), atol=1e-4)

    # train split must NOT be augmented
    cfg_b = _protein_cfg(tmp_path / "b", test_rot=True, test_trans=True)
    paths = preprocess.process_dataset_edge_cutoff(cfg_b)
    cfg_a = _protein_cfg(tmp_path / "a")
    paths_a = preprocess.process_dataset_edge_cutoff(cfg_a)
    tr_b = torch.load(paths[0], weights_only=False)
    tr_a = torch.load(paths_a[0], weights_only=False)
    for s_a, s_b in zip(tr_a, tr_b):
        assert torch.allclose(s_a.pos, s_b.pos)
