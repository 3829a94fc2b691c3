import torch

from distegnn_amd.data.graph import Batch, Data, collate, sort_edges_by_row
from distegnn_amd.data.partition import (SPLITTERS, graph_partition,
                                         split_large_graph_random)
from distegnn_amd.data.synthetic import (WORKLOADS, make_cloud_sample,
                                         make_cutoff_dataset,
                                         make_distributed_dataset)
from distegnn_amd.ops import reference as R


def test_sort_edges():
    ei = torch.tensor([[3, 1, 2, 1], [0, 2, 1, 0]])
    ea = torch.arange(4).float().unsqueeze(-1)
    ei2, ea2 = sort_edges_by_row(ei, ea)
    assert ei2[0].tolist() == [1, 1, 2, 3]
    # stable: original order kept within row groups
    assert ea2.squeeze(-1).tolist() == [1.0, 3.0, 2.0, 0.0]


def test_collate_offsets_and_csr(small_batch):
    b = small_batch
    assert b.num_graphs == 3
    assert b.ptr.tolist() == [0, 100, 200, 300]
    assert b.batch.shape == (300,)
    assert b.counts.tolist() == [100.0, 100.0, 100.0]
    # rowptr consistent with edge rows
    row = b.edge_index[0]
    deg = torch.bincount(row, minlength=300)
    assert torch.equal(b.rowptr[1:] - b.rowptr[:-1], deg)
    assert torch.all(row[1:] >= row[:-1])
    # edges stay within their graph
    src_g = b.batch[b.edge_index[0]]
    dst_g = b.batch[b.edge_index[1]]
    assert torch.equal(src_g, dst_g)


def test_random_split_balance_and_loc_mean():
    g = torch.Generator().manual_seed(0)
    n = 503
    pos = torch.rand(n, 3, generator=g)
    x = torch.rand(n, 2, generator=g)
    parts = split_large_graph_random(pos, x, pos.clone(), pos.clone(),
                                     x[:, :1], 0.2, 4, generator=g)
    sizes = [p.num_nodes for p in parts]
    assert sum(sizes) == n
    assert max(sizes) - min(sizes) <= 3
    for p in parts:
        assert torch.allclose(p.loc_mean, pos.mean(0, keepdim=True))
        assert p.edge_attr.shape == (p.num_edges, 2)


def test_graph_partition_balanced_and_local():
    g = torch.Generator().manual_seed(1)
    pos = torch.rand(400, 3, generator=g)
    ei = R.radius_graph(pos, 0.25)
    labels = graph_partition(ei, 400, 4, pos=pos)
    sizes = torch.bincount(labels, minlength=4)
    assert sizes.min() >= 90 and sizes.max() <= 110
    # locality: cut fraction well below the random-partition expectation (75%)
    cut = (labels[ei[0]] != labels[ei[1]]).float().mean()
    assert cut < 0.5


def test_kmeans_split_mode():
    data = make_distributed_dataset("Water-3D", 1, 2, split_mode="kmeans",
                                    seed=0, n_override=400)
    assert len(data) == 2 and len(data[0]) == 1
    assert data[0][0].num_nodes + data[1][0].num_nodes == 400


def test_synthetic_density():
    """Synthetic clouds reproduce the published average degree ballpark."""
    rng = torch.Generator().manual_seed(0)
    s = make_cloud_sample("Water-3D", rng)
    ei = R.radius_graph(s["pos"], s["radius"])
    avg_deg = ei.size(1) / s["pos"].size(0)
    assert 8 <= avg_deg <= 18  # published ~12.2


def test_synthetic_fields_fluid():
    rng = torch.Generator().manual_seed(0)
    s = make_cloud_sample("Fluid113K", rng, n_override=1000)
    assert s["x"].shape == (1000, 3)      # [visc, mass, |v|]
    assert s["attr"].shape == (1000, 2)   # [visc, mass]


def test_nbody_generator_smoke(tmp_path):
    """The offline N-body generator produces reader-compatible .npy files."""
    import os
    import subprocess
    import sys

    import numpy as np

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = os.path.join(root, "dataset_generation", "nbody",
                          "generate_dataset.py")
    out = tmp_path / "nb"
    subprocess.run(
        [sys.executable, script, "--path", str(out), "--num-train", "2",
         "--num-valid", "1", "--num-test", "1", "--length", "200",
         "--sample-freq", "50", "--n_isolated", "12", "--clusters", "2",
         "--seed", "7"],
        check=True, timeout=300)
    files = sorted(p.name for p in out.iterdir())
    assert any(f.startswith("loc_train") for f in files), files
    loc = np.load(next(out.glob("loc_train*.npy")))
    vel = np.load(next(out.glob("vel_train*.npy")))
    q = np.load(next(out.glob("charges_train*.npy")))
    assert loc.shape[0] == 2 and loc.shape[2] == 12 and loc.shape[-1] == 3
    assert vel.shape == loc.shape
    assert q.shape[0] == 2 and q.shape[1] == 12
    assert np.isfinite(loc).all() and np.isfinite(vel).all()


def test_batch_csr_invariants_fuzz():
    """Property test of the CSR contract every HIP kernel relies on:
    row-sorted edges, exact rowptr/colptr, col_perm sorting, chunk-table
    coverage — across random graph counts/sizes/densities."""
    from distegnn_amd.data.graph import Data, collate

    g = torch.Generator().manual_seed(123)
    for trial in range(20):
        n_graphs = int(torch.randint(1, 6, (1,), generator=g))
        datas = []
        for _ in range(n_graphs):
            n = int(torch.randint(2, 90, (1,), generator=g))
            m = int(torch.randint(0, 4 * n, (1,), generator=g))
            ei = torch.randint(0, n, (2, m), generator=g)
            keep = ei[0] != ei[1]
            ei = ei[:, keep]
            m = ei.size(1)
            datas.append(Data(x=torch.rand(n, 2, generator=g),
                              pos=torch.rand(n, 3, generator=g),
                              vel=torch.rand(n, 3, generator=g),
                              target=torch.rand(n, 3, generator=g),
                              edge_index=ei,
                              edge_attr=torch.rand(m, 2, generator=g),
                              loc_mean=torch.rand(1, 3, generator=g)))
        b = collate(datas)
        n, m = b.num_nodes, b.num_edges
        row, col = b.edge_index[0], b.edge_index[1]
        # row-sorted
        assert (row[1:] >= row[:-1]).all()
        # rowptr is the exact CSR of row
        assert b.rowptr.numel() == n + 1
        assert b.rowptr[0] == 0 and b.rowptr[-1] == m
        for seg in range(n):
            s, e = int(b.rowptr[seg]), int(b.rowptr[seg + 1])
            assert (row[s:e] == seg).all()
        # col_perm sorts col; colptr is the CSR of the sorted col
        sc = col[b.col_perm]
        assert (sc[1:] >= sc[:-1]).all()
        assert b.colptr[0] == 0 and b.colptr[-1] == m
        for seg in range(n):
            s, e = int(b.colptr[seg]), int(b.colptr[seg + 1])
            assert (sc[s:e] == seg).all()
        # edges stay within their graph block
        for gi in range(b.num_graphs):
            s, e = int(b.ptr[gi]), int(b.ptr[gi + 1])
            mask = (row >= s) & (row < e)
            assert ((col[mask] >= s) & (col[mask] < e)).all()
        # chunk tables (when built) tile [ptr[i], ptr[i+1]) exactly
        if b.pool_chunk_begin is not None:
            cb, ce = b.pool_chunk_begin, b.pool_chunk_end
            scp = b.pool_seg_chunk_ptr
            assert scp[0] == 0 and scp[-1] == cb.numel()
            for gi in range(b.num_graphs):
                ks = int(scp[gi])
                ke = int(scp[gi + 1])
                assert int(cb[ks]) == int(b.ptr[gi])
                assert int(ce[ke - 1]) == int(b.ptr[gi + 1])
                for k in range(ks + 1, ke):
                    assert int(cb[k]) == int(ce[k - 1])


def test_batch_chunk_tables_large_segment():
    """A >4096-node graph triggers the chunk tables; validate coverage."""
    from distegnn_amd.data.graph import Data, collate

    g = torch.Generator().manual_seed(5)
    n = 5000
    ei = torch.randint(0, n, (2, 100), generator=g)
    ei = ei[:, ei[0] != ei[1]]
    b = collate([Data(x=torch.rand(n, 2, generator=g),
                      pos=torch.rand(n, 3, generator=g),
                      vel=torch.rand(n, 3, generator=g),
                      target=torch.rand(n, 3, generator=g),
                      edge_index=ei,
                      edge_attr=torch.rand(ei.size(1), 2, generator=g),
                      loc_mean=torch.rand(1, 3, generator=g))])
    assert b.pool_chunk_begin is not None
    cb, ce, scp = (b.pool_chunk_begin, b.pool_chunk_end,
                   b.pool_seg_chunk_ptr)
    assert scp.tolist() == [0, cb.numel()]
    assert int(cb[0]) == 0 and int(ce[-1]) == n
    assert (cb[1:] == ce[:-1]).all()
    assert ((ce - cb) <= 256).all() and ((ce - cb) > 0).all()


def test_cutoff_edge_drops_longest():
    """cutoff_edge keeps the shortest (1-rate) fraction (reference
    process_dataset.py:300-305)."""
    from distegnn_amd.data.preprocess import cutoff_edge

    g = torch.Generator().manual_seed(3)
    pos = torch.rand(50, 3, generator=g)
    ei = torch.randint(0, 50, (2, 200), generator=g)
    ei = ei[:, ei[0] != ei[1]]
    m = ei.size(1)
    out = cutoff_edge(ei, pos, 0.25)
    assert out.size(1) == int(m * 0.75)
    d_all = (pos[ei[0]] - pos[ei[1]]).norm(dim=1)
    d_kept = (pos[out[0]] - pos[out[1]]).norm(dim=1)
    thresh = torch.sort(d_all).values[int(m * 0.75) - 1]
    assert (d_kept <= thresh + 1e-6).all()
    # rate 0 is the identity
    assert torch.equal(cutoff_edge(ei, pos, 0.0), ei)


def test_nbody_composite_objects_invariants():
    """Stick/Hinge rigid bodies (dataset_generation/nbody/rigid.py):
    rod/beam lengths and beam-parallel velocity matching are exact
    invariants under the per-object integrator."""
    import os
    import sys

    import numpy as np

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(root, "dataset_generation", "nbody"))
    try:
        import generate_dataset as gd
    finally:
        sys.path.pop(0)

    rng = np.random.default_rng(3)
    sys_ = gd.CompositeSystem(4, 3, 2, clusters=1, rng=rng)
    assert sys_.n == 4 + 3 * 2 + 2 * 3
    sys_.check()                 # invariants hold right after init
    loc, vel, q = sys_.trajectory(300, 50)
    sys_.check()                 # ... and after 300 integration steps
    assert loc.shape == (6, sys_.n, 3)
    assert np.isfinite(loc).all() and np.isfinite(vel).all()
    # ball membership covers every index exactly once
    covered = sorted(i for o in sys_.objects for i in o.node_idx)
    assert covered == list(range(sys_.n))


def test_nbody_generator_composite_smoke(tmp_path):
    import os
    import subprocess
    import sys

    import numpy as np

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    script = os.path.join(root, "dataset_generation", "nbody",
                          "generate_dataset.py")
    out = tmp_path / "nb"
    subprocess.run(
        [sys.executable, script, "--path", str(out), "--num-train", "1",
         "--num-valid", "1", "--num-test", "1", "--length", "100",
         "--sample-freq", "50", "--n_isolated", "4", "--n_stick", "2",
         "--n_hinge", "1", "--clusters", "1", "--seed", "7"],
        check=True, timeout=300)
    loc = np.load(next(out.glob("loc_train*charged4_2_1_1*.npy")))
    assert loc.shape[2] == 4 + 2 * 2 + 1 * 3
    assert np.isfinite(loc).all()
