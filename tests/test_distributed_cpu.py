"""Multi-process (gloo, world_size=2) tests of the DistEGNN distributed math.

Key invariant: a 2-rank forward on the two partitions of a
graph equals a single-process forward on the merged graph (union of
partition nodes and partition-internal edges) EXACTLY — the virtual-node
weighted all-reduce is the only cross-rank coupling and it reconstructs the
global per-graph means. This is the strongest available correctness test of
the fused collective path and runs on CPU here (RCCL path is identical
torch.distributed code on the GPU box).
"""

import os
import tempfile

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from distegnn_amd.data.graph import Data, collate
from distegnn_amd.data.partition import split_large_graph_random
from distegnn_amd.data.synthetic import make_cloud_sample
from distegnn_amd.models import FastEGNN
from distegnn_amd.utils import fix_seed


def _make_partitions(num_samples=2, n=160, ws=2, seed=0):
    """Per-rank partition lists + the merged single-graph equivalents."""
    rng = torch.Generator().manual_seed(seed)
    per_rank = [[] for _ in range(ws)]
    merged = []
    for _ in range(num_samples):
        s = make_cloud_sample("Water-3D", rng, n_override=n)
        parts = split_large_graph_random(
            s["pos"], s["x"], s["target"], s["vel"], s["attr"], 0.08, ws,
            generator=rng)
        for i, p in enumerate(parts):
            per_rank[i].append(p)
        offs, fields = 0, {k: [] for k in
                           ("x", "pos", "vel", "attr", "target")}
        eis, eas = [], []
        for p in parts:
            for k in fields:
                fields[k].append(getattr(p, k))
            eis.append(p.edge_index + offs)
            eas.append(p.edge_attr)
            offs += p.num_nodes
        merged.append(Data(
            **{k: torch.cat(v) for k, v in fields.items()},
            loc_mean=parts[0].loc_mean,
            edge_index=torch.cat(eis, dim=1), edge_attr=torch.cat(eas)))
    return per_rank, merged


def _model(ws):
    fix_seed(7)
    return FastEGNN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                    hidden_nf=32, virtual_channels=3, world_size=ws,
                    n_layers=2, normalize=False).double()


def _forward(model, batch, counts_global=None):
    return model(batch.x.double(), batch.pos.double(), batch.vel.double(),
                 batch.loc_mean.double(), batch.edge_index, batch.batch,
                 edge_attr=batch.edge_attr.double(), rowptr=batch.rowptr,
                 ptr=batch.ptr, counts=batch.counts.double(),
                 counts_global=counts_global)


def _rank_worker(rank, ws, init_file, result_dir):
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=ws)
    try:
        per_rank, _ = _make_partitions()
        model = _model(ws)
        batch = collate(per_rank[rank])
        counts = batch.counts.double()
        cg = counts.clone()
        dist.all_reduce(cg)
        loc, vloc = _forward(model, batch, counts_global=cg)
        # include backward so the mirrored all-reduce path is exercised;
        # loss touches node outputs only so Sum_r L_r == merged-graph loss
        loc.pow(2).sum().backward()
        # last-layer node/virtual-feat MLPs get no grad from a loc-only loss
        # (the reference needed DDP find_unused_parameters=True for this)
        grads = {n: (p.grad.clone() if p.grad is not None
                     else torch.zeros_like(p))
                 for n, p in model.named_parameters()}
        torch.save({"loc": loc.detach(), "vloc": vloc.detach(),
                    "grads": grads},
                   os.path.join(result_dir, f"rank{rank}.pt"))
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_two_rank_forward_equals_merged_graph(tmp_path):
    ws = 2
    init_file = tmp_path / "pg_init"
    mp.spawn(_rank_worker, args=(ws, str(init_file), str(tmp_path)),
             nprocs=ws, join=True)

    per_rank, merged = _make_partitions()
    outs = [torch.load(tmp_path / f"rank{r}.pt", weights_only=False)
            for r in range(ws)]

    # single-process reference on the merged graphs
    model = _model(1)
    mb = collate(merged)
    loc_m, vloc_m = _forward(model, mb)
    loc_m.pow(2).sum().backward()

    # node predictions: merged graph is [rank0 graph0, rank1 graph0, ...] per
    # sample; reconstruct the rank blocks
    sizes = [[p.num_nodes for p in per_rank[r]] for r in range(ws)]
    ofs = 0
    blocks = {0: [], 1: []}
    for si in range(len(merged)):
        for r in range(ws):
            blocks[r].append(loc_m[ofs:ofs + sizes[r][si]])
            ofs += sizes[r][si]
    for r in range(ws):
        want = torch.cat(blocks[r])
        assert torch.allclose(outs[r]["loc"], want, atol=1e-9), \
            f"rank {r} node predictions diverge from merged-graph reference"

    # virtual node positions are replicated and must match the merged run
    for r in range(ws):
        assert torch.allclose(outs[r]["vloc"], vloc_m, atol=1e-9)

    # gradient check: the sum of per-rank local grads (what GradBucket's
    # all_reduce(SUM) produces) equals the merged-graph gradient exactly
    for n, p in model.named_parameters():
        g = sum(outs[r]["grads"][n] for r in range(ws))
        want = p.grad if p.grad is not None else torch.zeros_like(p)
        assert torch.allclose(g, want, atol=1e-8), \
            f"summed distributed grad diverges for {n}"


def _war_worker(rank, init_file, result_dir):
    ws = 2
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=ws)
    try:
        from distegnn_amd.parallel import comm

        torch.manual_seed(10 + rank)
        a = torch.randn(3, 4, requires_grad=True)
        b = torch.randn(3, 2, 5, requires_grad=True)
        counts = torch.tensor([2.0, 3.0, 5.0]) * (rank + 1)
        cg = counts.clone()
        dist.all_reduce(cg)
        ra, rb = comm.fused_weighted_average_reduce([a, b], counts, cg)
        (ra.sum() + rb.sum()).backward()
        torch.save({"ra": ra.detach(), "rb": rb.detach(),
                    "a": a.detach(), "b": b.detach(),
                    "counts": counts, "ga": a.grad, "gb": b.grad},
                   os.path.join(result_dir, f"w{rank}.pt"))
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_fused_weighted_average_reduce_math(tmp_path):
    """fused_weighted_average_reduce == explicit weighted mean."""
    ws = 2
    init_file = tmp_path / "pg2_init"
    mp.spawn(_war_worker, args=(str(init_file), str(tmp_path)), nprocs=ws,
             join=True)
    o = [torch.load(tmp_path / f"w{r}.pt", weights_only=False)
         for r in range(ws)]
    wsum = o[0]["counts"] + o[1]["counts"]
    want_a = (o[0]["a"] * o[0]["counts"][:, None]
              + o[1]["a"] * o[1]["counts"][:, None]) / wsum[:, None]
    assert torch.allclose(o[0]["ra"], want_a, atol=1e-6)
    assert torch.allclose(o[1]["ra"], want_a, atol=1e-6)
    want_b = (o[0]["b"] * o[0]["counts"][:, None, None]
              + o[1]["b"] * o[1]["counts"][:, None, None]) / wsum[:, None, None]
    assert torch.allclose(o[0]["rb"], want_b, atol=1e-6)
    # backward: g_in = w/W * allreduce(g_out); g_out = ones on both ranks
    want_ga = (o[0]["counts"][:, None] / wsum[:, None]) * 2.0
    assert torch.allclose(o[0]["ga"], want_ga.expand_as(o[0]["ga"]), atol=1e-6)


def _bench_worker(rank, init_file, result_dir):
    """Mirror bench.py's distributed step on CPU/gloo: identical per-rank
    synthetic partitions, counts_global reduce, model forward with fused
    collectives, GradBucket sync."""
    ws = 2
    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=ws)
    try:
        import bench as bench_mod
        from distegnn_amd.parallel.comm import GradBucket
        from distegnn_amd.parallel import comm as C

        torch.manual_seed(0)
        batches = bench_mod.build_rank_batches(rank, ws, 2, 400, 0.08,
                                               "random", seed=7)
        fix_seed(11)
        model = FastEGNN(node_feat_nf=3, node_attr_nf=2, edge_attr_nf=2,
                         hidden_nf=32, virtual_channels=3, world_size=ws,
                         n_layers=2)
        bucket = GradBucket(model)
        bucket.broadcast_parameters()
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
        losses = []
        for step in range(3):
            data = batches[step % 2]
            data.counts_global = C.global_counts(data.counts)
            total = data.counts_global.sum()
            loc, vloc = model(
                data.x, data.pos, data.vel, data.loc_mean, data.edge_index,
                data.batch, edge_attr=data.edge_attr, node_attr=data.attr,
                rowptr=data.rowptr, ptr=data.ptr, counts=data.counts,
                counts_global=data.counts_global)
            loss = (float(data.num_nodes) / total) * torch.nn.functional \
                .mse_loss(loc, data.target)
            loss.backward()
            bucket.sync()
            opt.step()
            opt.zero_grad(set_to_none=False)
            logged = loss.detach().clone()
            dist.all_reduce(logged)
            losses.append(logged.item())
        # replicated state must stay in lockstep
        flat = torch.cat([p.data.reshape(-1) for p in model.parameters()])
        torch.save({"losses": losses, "params": flat},
                   os.path.join(result_dir, f"bench{rank}.pt"))
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_bench_distributed_step_cpu(tmp_path):
    """The bench.py multi-rank path (identical partition generation,
    counts_global, fused virtual reduces, flat grad bucket) stays in
    lockstep across ranks and produces finite decreasing-ish losses."""
    import sys

    sys.path.insert(0, os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    init_file = tmp_path / "pg3_init"
    mp.spawn(_bench_worker, args=(str(init_file), str(tmp_path)), nprocs=2,
             join=True)
    o0 = torch.load(tmp_path / "bench0.pt", weights_only=False)
    o1 = torch.load(tmp_path / "bench1.pt", weights_only=False)
    assert o0["losses"] == o1["losses"]
    assert all(torch.isfinite(torch.tensor(o0["losses"])))
    assert torch.equal(o0["params"], o1["params"]), \
        "rank parameters diverged after optimizer steps"


def _main_worker(rank, init_file, cfg_path, log_root, ws=2):
    import json
    import os

    os.environ.update({
        "WORLD_SIZE": str(ws), "RANK": str(rank), "LOCAL_RANK": str(rank),
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "0",
        "TORCH_DIST_INIT": "",
    })
    # file:// rendezvous avoids port races under spawn
    import torch.distributed as dist

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=ws)
    import main as entry

    entry.main(["--config_path", cfg_path])
    # rank 0 wrote the log; both ranks trained in lockstep without divergence
    if rank == 0:
        exps = [p for p in os.listdir(log_root)]
        assert exps, "no experiment dir written"
        log = json.load(open(os.path.join(log_root, exps[0], "log",
                                          "log.json")))
        assert log[1]["loss_train"], "no training happened"
        for v in log[1]["loss_train"]:
            assert v == v and v < 1e3, ("diverged", v)


def test_main_two_rank_distribute_mode(tmp_path):
    """Full main.py on 2 gloo ranks in DistEGNN distribute mode (metis
    split): the exact topology of the driver's multi-GPU run, minus RCCL."""
    import yaml

    from tests.test_trainer_cpu import tiny_config

    cfg = tiny_config(tmp_path)
    cfg["data"].update({
        "dataset_name": "Water-3D", "accelerate_mode": "distribute",
        "outer_radius": 0.12, "inner_radius": 0.12, "split_mode": "metis",
        "batch_size": 1, "synthetic_samples": 8, "delta_t": 20,
        "max_samples": 100,
    })
    cfg["data"].pop("frame_0", None)
    cfg["data"].pop("frame_T", None)
    cfg["train"]["epochs"] = 2
    cfg["train"]["accumulation_steps"] = 2
    cfg_path = tmp_path / "cfg_dist.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    init_file = tmp_path / "pg_init"
    mp.spawn(_main_worker,
             args=(str(init_file), str(cfg_path),
                   str(tmp_path / "logs")),
             nprocs=2, join=True)


def test_capture_comm_fallback_flag():
    """The fallback context toggles the module flag and restores on exit
    (used so post-capture eager fallbacks keep the capture communicator)."""
    import distegnn_amd.parallel.comm as C

    assert C._FORCE_CAPTURE_COMM is False
    with C.capture_comm_fallback():
        assert C._FORCE_CAPTURE_COMM is True
        with C.capture_comm_fallback():
            assert C._FORCE_CAPTURE_COMM is True
        assert C._FORCE_CAPTURE_COMM is True
    assert C._FORCE_CAPTURE_COMM is False
    # off-GPU there is never a capture group
    assert C._capture_group() is None


def test_graphedstep_fallback_ctx_used():
    """GraphedStep routes disabled-shape fallbacks through fallback_ctx."""
    from distegnn_amd.runtime.graphs import GraphedStep

    calls = []

    class Ctx:
        def __enter__(self):
            calls.append("enter")

        def __exit__(self, *a):
            calls.append("exit")
            return False

    g = GraphedStep(lambda b: ("out",), [], enabled=True, fallback_ctx=Ctx)
    assert g.enabled is False  # no CUDA here -> plain path, ctx unused
    assert g(object()) == ("out",)
    assert calls == []
    # the internal fallback path itself must wrap with the ctx
    assert g._fallback(object()) == ("out",)
    assert calls == ["enter", "exit"]


def _rf_worker(rank, init_file, result_dir):
    from distegnn_amd.models.fastrf import FastRF

    dist.init_process_group("gloo", init_method=f"file://{init_file}",
                            rank=rank, world_size=2)
    try:
        per_rank, _ = _make_partitions()
        fix_seed(11)
        model = FastRF(edge_attr_nf=2, hidden_nf=32, virtual_channels=3,
                       world_size=2, n_layers=2).double()
        batch = collate(per_rank[rank])
        counts = batch.counts.double()
        cg = counts.clone()
        dist.all_reduce(cg)
        loc, vloc = model(batch.pos.double(), batch.vel.double(),
                          batch.loc_mean.double(), batch.edge_index,
                          batch.batch, edge_attr=batch.edge_attr.double(),
                          rowptr=batch.rowptr, ptr=batch.ptr, counts=counts,
                          counts_global=cg)
        loc.pow(2).sum().backward()
        grads = {n: (p.grad.clone() if p.grad is not None
                     else torch.zeros_like(p))
                 for n, p in model.named_parameters()}
        torch.save({"loc": loc.detach(), "vloc": vloc.detach(),
                    "grads": grads},
                   os.path.join(result_dir, f"rf{rank}.pt"))
        dist.barrier()
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_two_rank_fastrf_equals_merged_graph(tmp_path):
    """FastRF's single virtual-exchange site: 2-rank partitioned forward
    equals the merged-graph forward; summed local grads equal merged grads."""
    from distegnn_amd.models.fastrf import FastRF

    init_file = tmp_path / "pg_init_rf"
    mp.spawn(_rf_worker, args=(str(init_file), str(tmp_path)), nprocs=2,
             join=True)
    r0 = torch.load(tmp_path / "rf0.pt", weights_only=False)
    r1 = torch.load(tmp_path / "rf1.pt", weights_only=False)

    per_rank, merged = _make_partitions()
    fix_seed(11)
    model = FastRF(edge_attr_nf=2, hidden_nf=32, virtual_channels=3,
                   world_size=1, n_layers=2).double()
    mb = collate(merged)
    loc, vloc = model(mb.pos.double(), mb.vel.double(),
                      mb.loc_mean.double(), mb.edge_index, mb.batch,
                      edge_attr=mb.edge_attr.double(), rowptr=mb.rowptr,
                      ptr=mb.ptr, counts=mb.counts.double())
    loc.pow(2).sum().backward()

    # Reconstruct each rank's node block from the merged output (merged =
    # [rank0 g0, rank1 g0, rank0 g1, rank1 g1]); FastRF's gram uses the
    # LOCAL per-rank coord mean by reference design (FastRF.py:165-168 —
    # unlike FastEGNN there is no coord_mean reduce site), so partitioned
    # and merged runs agree only up to that local-vs-global mean effect:
    # loose tolerance, exactness is covered by the FastEGNN variant.
    outs = {0: r0, 1: r1}
    sizes = [[p_.num_nodes for p_ in per_rank[r]] for r in range(2)]
    ofs = 0
    blocks = {0: [], 1: []}
    for si in range(len(merged)):
        for r in range(2):
            blocks[r].append(loc.detach()[ofs:ofs + sizes[r][si]])
            ofs += sizes[r][si]
    for r in range(2):
        want = torch.cat(blocks[r])
        assert torch.allclose(outs[r]["loc"], want, atol=1e-3), r
    assert torch.allclose(r0["vloc"], vloc.detach(), atol=1e-3)
    for n, p in model.named_parameters():
        g = p.grad if p.grad is not None else torch.zeros_like(p)
        assert torch.allclose(r0["grads"][n] + r1["grads"][n], g,
                              atol=1e-2, rtol=1e-2), n


@pytest.mark.timeout(600)
def test_main_four_rank_distribute_mode(tmp_path):
    """Full main.py on FOUR gloo ranks in DistEGNN distribute mode with
    eval epochs — the driver's 8-GPU topology at half width, minus RCCL:
    4-way partition, lockstep samplers, flat grad sync, weighted
    virtual-node reduces, the eval/checkpoint block and the early-stop
    collective all run at world_size > 2."""
    import yaml

    from tests.test_trainer_cpu import tiny_config

    cfg = tiny_config(tmp_path)
    cfg["data"].update({
        "dataset_name": "Water-3D", "accelerate_mode": "distribute",
        "outer_radius": 0.12, "inner_radius": 0.12, "split_mode": "metis",
        "batch_size": 1, "synthetic_samples": 6, "delta_t": 20,
        "max_samples": 100,
    })
    cfg["data"].pop("frame_0", None)
    cfg["data"].pop("frame_T", None)
    cfg["train"]["epochs"] = 2
    cfg["train"]["accumulation_steps"] = 2
    cfg["log"]["test_interval"] = 1
    cfg_path = tmp_path / "cfg_dist4.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    init_file = tmp_path / "pg_init4"
    mp.spawn(_main_worker,
             args=(str(init_file), str(cfg_path), str(tmp_path / "logs"),
                   4),
             nprocs=4, join=True)
