"""Flagship benchmark: LargeFluid-113K DistEGNN training step on MI355X.

Driver contract: ``python bench.py --gpus N --steps K --warmup W`` runs the
headline training workload (BASELINE.json: train step time (ms) for
LargeFluid-113K DistEGNN) on N GPUs of one node; for N>1 it is launched by
``python -m torch.distributed.run --nproc-per-node N ... bench.py`` with one
rank per GPU over RCCL. Rank 0 prints ONE JSON line.

Workload: synthetic 113,140-node fluid cloud at the published density
(~1.7M edges at r=0.075), random-init FastEGNN (H=64, L=4, C=5),
partitioned across ranks (split_mode random by default, --split-mode metis
supported), grad-accumulation 4 as in config/largefluid_distegnn.yaml.
STRONG scaling: the 113K-node graph is fixed; more GPUs = smaller
partitions per rank + the virtual-node all-reduce.

A timed step = H2D of the batch + forward + coord-MSE + MMD + backward
(+ gradient all-reduce, clip and Adam step every 4th step). Compute dtype
bf16 (autocast) for MLP GEMMs; coordinates/reductions fp32.
"""

import argparse
import json
import os
import time

import torch

from distegnn_amd.data.graph import collate
from distegnn_amd.data.synthetic import make_cloud_sample, make_cutoff_dataset
from distegnn_amd.data.partition import SPLITTERS
from distegnn_amd.models import FastEGNN
from distegnn_amd.parallel import comm
from distegnn_amd.parallel.comm import GradBucket
from distegnn_amd.runtime.losses import mmd_loss
from distegnn_amd.utils import fix_seed


def build_rank_batches(rank, world_size, num_batches, n_nodes, radius,
                       split_mode, seed):
    """Identical clouds on every rank (fixed seed); each rank keeps its
    partition — mirrors the reference's offline preprocessing."""
    batches = []
    rng = torch.Generator().manual_seed(seed)
    for _ in range(num_batches):
        s = make_cloud_sample("Fluid113K", rng, n_override=n_nodes)
        if world_size == 1:
            parts = SPLITTERS["random"](
                pos=s["pos"], x=s["x"], target=s["target"], vel=s["vel"],
                attr=s["attr"], radius=radius, world_size=1, device="cpu",
                generator=rng)
        elif split_mode == "random":
            parts = SPLITTERS["random"](
                pos=s["pos"], x=s["x"], target=s["target"], vel=s["vel"],
                attr=s["attr"], radius=radius, world_size=world_size,
                device="cpu", generator=rng)
        else:
            parts = SPLITTERS[split_mode](
                pos=s["pos"], x=s["x"], target=s["target"], vel=s["vel"],
                attr=s["attr"], outer_radius=radius, inner_radius=radius,
                world_size=world_size, device="cpu")
        batches.append(collate([parts[rank]]))
    return batches


def make_step_core(model, accum, mmd_sigma, mmd_samples, autocast_dtype):
    """forward + loss + backward on device-resident batch tensors.

    hipGraph-capturable: no host syncs, no collectives (counts_global is
    reduced eagerly by the caller before the graphed region)."""

    def step_core(data):
        # captured collectives (this counts reduce + the in-forward virtual
        # exchanges) route to the dedicated capture communicator via
        # comm._capture_group(); eager collectives (teardown barrier, timing
        # MAX-reduce) stay on the default group — NCCL forbids mixing
        # captured and eager work on one communicator
        counts_global = comm.global_counts(data.counts)
        total_node_cnt = counts_global.sum()
        chunks = None
        if getattr(data, "pool_chunk_begin", None) is not None:
            chunks = (data.pool_chunk_begin, data.pool_chunk_end,
                      data.pool_seg_chunk_ptr)
        ctx = (torch.autocast("cuda", dtype=autocast_dtype)
               if autocast_dtype is not None else torch.enable_grad())
        with ctx:
            loc_pred, vloc = model(
                data.x, data.pos, data.vel, data.loc_mean, data.edge_index,
                data.batch, edge_attr=data.edge_attr,
                node_attr=(data.attr if model.node_attr_nf else None),
                rowptr=data.rowptr, ptr=data.ptr, counts=data.counts,
                counts_global=counts_global, pool_chunks=chunks,
                colptr=data.colptr, col_perm=data.col_perm)
        loss = torch.nn.functional.mse_loss(loc_pred.float(), data.target)
        weight = float(data.num_nodes) / total_node_cnt
        loss = weight * loss
        mse_log = loss.detach()
        lm = mmd_loss(vloc.permute(0, 2, 1).float(), data.target, data.batch,
                      data.ptr, data.counts, mmd_sigma, mmd_samples,
                      sample_idx=getattr(data, "mmd_idx", None),
                      sample_valid=getattr(data, "mmd_valid", None))
        loss = loss + 0.01 * weight * lm
        (loss / accum).backward()
        return (mse_log,)

    return step_core


def train_step(graphed, batch, optimizer, grad_bucket, step, accum,
               world_size, device, clip=True, mmd_cfg=None):
    def _pre():
        # eager work between replays (H2D copies included) runs on the side
        # stream: its allocations must not alias captured-graph pool blocks
        data = batch.to(device)
        if mmd_cfg is not None:
            # fresh randomness drawn OUTSIDE the captured region
            from distegnn_amd.runtime.losses import draw_sample_indices

            data.mmd_idx, data.mmd_valid = draw_sample_indices(
                data.batch, data.ptr, data.counts, mmd_cfg)
        return data

    data = graphed.run_eager(_pre)
    (mse_log,) = graphed(data)
    if (step + 1) % accum == 0:
        def _opt():
            if grad_bucket is not None:
                # captured-graph replay: eager RCCL between step replays
                # corrupts captured state on this stack
                if graphed.enabled:
                    grad_bucket.graph_sync()
                else:
                    grad_bucket.sync()
            if clip:  # reference clip rule: FastEGNN + (ws>1 or LargeFluid)
                torch.nn.utils.clip_grad_norm_(model_params(graphed),
                                               max_norm=0.3)
            optimizer.step()
            optimizer.zero_grad(set_to_none=False)
            # captured replays read version-cached weight transforms;
            # refresh them after the weights changed (ops/prep.py)
            from distegnn_amd import ops

            ops.refresh_weight_prep()

        # side stream: eager allocs must not alias graph-pool blocks
        graphed.run_eager(_opt)
    return mse_log


def model_params(graphed):
    return graphed.params


# BASELINE.md workloads (reference dataset scales; see data/synthetic.py)
WORKLOADS = {
    # name: (dataset, nodes/graph, graphs/batch, radius, feat, attr, C,
    #        accum, mmd_sigma, mmd_samples, normalize)
    "largefluid": ("Fluid113K", 113140, 1, 0.075, 3, 2, 5, 4, 3.0, 50, False),
    "water3d": ("Water-3D", 7806, 15, 0.035, 2, 0, 3, 1, 1.5, 3, False),
    "protein": ("protein", 855, 5, 10.0, 2, 0, 3, 1, 1.0, 3, False),
    "nbody": ("nbody_100", 100, 250, -1.0, 2, 0, 3, 1, 1.5, 3, True),
}


def build_cutoff_batches(workload, num_batches, graphs_per_batch, seed):
    """Single-device ("cutoff_edges") batches at published scales."""
    dataset, nodes, _, radius, *_ = WORKLOADS[workload]
    samples = make_cutoff_dataset(dataset, num_batches * graphs_per_batch,
                                  seed=seed)
    return [collate(samples[i * graphs_per_batch:(i + 1) * graphs_per_batch])
            for i in range(num_batches)]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=16)
    ap.add_argument("--warmup", type=int, default=8)
    ap.add_argument("--workload", type=str, default="largefluid",
                    choices=list(WORKLOADS))
    ap.add_argument("--nodes", type=int, default=None)
    ap.add_argument("--radius", type=float, default=None)
    ap.add_argument("--split-mode", type=str, default="random",
                    choices=["random", "metis", "kmeans"])
    ap.add_argument("--dtype", type=str, default="bf16",
                    choices=["bf16", "fp32"])
    ap.add_argument("--virtual-channels", type=int, default=None)
    ap.add_argument("--num-batches", type=int, default=2,
                    help="distinct synthetic samples to cycle through")
    ap.add_argument("--integrity", type=str, default="auto",
                    choices=["auto", "on", "off"],
                    help="self-validate capture/replay before the measured "
                         "phase: run a short trajectory eager and graphed "
                         "from identical state, compare losses, and disable "
                         "capture coherently on all ranks on divergence "
                         "(auto = on for multi-rank graphed runs)")
    ap.add_argument("--graphs", type=str, default="auto",
                    choices=["auto", "on", "off"],
                    help="hipGraph-capture the fwd+loss+bwd step (auto = "
                         "on). The in-forward RCCL virtual-node all-reduces "
                         "are captured inside the graph; every eager op "
                         "between replays (optimizer, H2D, barriers, timing "
                         "collectives) runs on a side stream — default-"
                         "stream allocations between replays alias graph-"
                         "pool blocks on this stack (see "
                         "runtime/graphs.py)")
    args = ap.parse_args()

    rank, world_size = comm.init_distributed()
    if world_size != args.gpus and int(os.environ.get("WORLD_SIZE", "1")) > 1:
        args.gpus = world_size
    world_size = max(world_size, 1)
    # DISTEGNN_FORCE_DIST=1 under torchrun -nproc 1: run the FULL collective
    # path (RCCL all-reduces, GradBucket) on a 1-rank group — identity math,
    # but exercises RCCL capture inside hipGraphs on a single GPU.
    force_dist = (os.environ.get("DISTEGNN_FORCE_DIST") == "1"
                  and world_size == 1 and comm.is_distributed())
    ws_eff = 2 if force_dist else world_size
    assert torch.cuda.is_available(), "bench.py requires a GPU"
    device = torch.device(f"cuda:{rank}")
    torch.cuda.set_device(device)

    (dataset, wl_nodes, graphs_per_batch, wl_radius, feat_nf, attr_nf, wl_c,
     accum, mmd_sigma, mmd_samples, normalize) = WORKLOADS[args.workload]
    nodes = args.nodes if args.nodes is not None else wl_nodes
    radius = args.radius if args.radius is not None else wl_radius
    vch = (args.virtual_channels if args.virtual_channels is not None
           else wl_c)

    fix_seed(43)
    t_data = time.perf_counter()
    if args.workload == "largefluid":
        batches = build_rank_batches(rank, world_size, args.num_batches,
                                     nodes, radius, args.split_mode,
                                     seed=43)
    else:
        assert world_size == 1, \
            "cutoff-mode workloads are single-device (reference main.py:173)"
        batches = build_cutoff_batches(args.workload, args.num_batches,
                                       graphs_per_batch, seed=43)
    if rank == 0:
        print(f"# data built in {time.perf_counter() - t_data:.1f}s: "
              f"{batches[0].num_nodes} nodes/rank, "
              f"{batches[0].num_edges} edges/rank", flush=True)

    model = FastEGNN(node_feat_nf=feat_nf, node_attr_nf=attr_nf,
                     edge_attr_nf=2, hidden_nf=64, virtual_channels=vch,
                     world_size=ws_eff, n_layers=4,
                     normalize=normalize).to(device)
    grad_bucket = None
    if ws_eff > 1:
        grad_bucket = GradBucket(model)
        grad_bucket.broadcast_parameters()
    optimizer = torch.optim.Adam(model.parameters(), lr=5e-4,
                                 weight_decay=1e-12)
    autocast_dtype = torch.bfloat16 if args.dtype == "bf16" else None
    model.train()

    from distegnn_amd.runtime.graphs import GraphedStep

    step_core = make_step_core(model, accum, mmd_sigma, mmd_samples,
                               autocast_dtype)
    use_graphs = args.graphs != "off"
    graphed = GraphedStep(step_core, model.parameters(),
                          warmup_occurrences=2,
                          enabled=use_graphs, verbose=True,
                          fallback_ctx=comm.capture_comm_fallback)

    clip = args.workload == "largefluid" or world_size > 1
    num_sample = mmd_samples * vch

    if ws_eff > 1 and grad_bucket is not None and use_graphs:
        # build the captured grad-sync graph BEFORE any step capture (its
        # watchdog drain is a default-group barrier — must not land inside
        # the replay window)
        grad_bucket.prebuild_graph_sync()

    run_gate = (args.integrity == "on"
                or (args.integrity == "auto" and use_graphs and ws_eff > 1))
    if run_gate and graphed.enabled:
        from distegnn_amd.runtime.integrity import run_capture_integrity_gate

        def _gate_step(k):
            return train_step(graphed, batches[k % len(batches)], optimizer,
                              grad_bucket, k, accum, ws_eff, device, clip,
                              mmd_cfg=num_sample)

        n_check = 4 * len(batches)   # warmup_occurrences(2) + 2 replays/key
        run_capture_integrity_gate(graphed, _gate_step, n_check,
                                   list(model.parameters()), optimizer,
                                   rank=rank)
        # the gate's final D2H device-syncs, garbling prebuilt graph execs
        # (GraphedStep.invalidate): the gate dropped its step captures;
        # rebuild the captured grad sync too. Warmup below recaptures.
        if graphed.enabled and ws_eff > 1 and grad_bucket is not None:
            grad_bucket.rebuild_graph_sync()

    mse = None
    # a shape's graph is captured at its (warmup_occurrences+1)-th
    # occurrence; a capture takes SECONDS at large node counts, so the
    # warmup phase must cover capture + at least one replay for EVERY
    # batch shape no matter how small --warmup is (extra steps stay
    # untimed — only the K steps below are measured)
    n_warm = args.warmup
    if graphed.enabled:
        n_warm = max(n_warm, (graphed.warmup + 2) * len(batches))
    for w in range(n_warm):
        mse = train_step(graphed, batches[w % len(batches)], optimizer,
                         grad_bucket, w, accum, ws_eff, device, clip,
                         mmd_cfg=num_sample)
    # Dist+graphs mode: NO barrier and NO device-wide synchronize until
    # every replay is done — a ProcessGroupNCCL barrier OR a bare
    # torch.cuda.synchronize() between replays of an RCCL-containing graph
    # corrupts captured state on this stack (both bisected to the exact
    # call; tools/fd_debug.py --barrier / --midsync). Ranks stay in
    # lockstep through the captured per-step collectives; timing uses
    # device events and the MAX-reduce below absorbs residual skew.
    # graphed.enabled (not use_graphs): the integrity gate may have disabled
    # capture coherently — the eager path then uses barriers safely
    event_timing = graphed.enabled and ws_eff > 1
    if event_timing:
        ev0 = torch.cuda.Event(enable_timing=True)
        ev1 = torch.cuda.Event(enable_timing=True)
        ev0.record()
    else:
        if ws_eff > 1:
            comm.barrier()
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for k in range(args.steps):
        mse = train_step(graphed, batches[k % len(batches)], optimizer,
                         grad_bucket, n_warm + k, accum,
                         ws_eff, device, clip, mmd_cfg=num_sample)
    if event_timing:
        ev1.record()
        torch.cuda.synchronize()   # safe: no replay runs after this point
        final_mse = float(mse.item()) if mse is not None else None
        elapsed = torch.tensor(ev0.elapsed_time(ev1) / 1000.0,
                               device=device)
    else:
        if ws_eff > 1:
            comm.barrier()
        torch.cuda.synchronize()
        final_mse = float(mse.item()) if mse is not None else None
        elapsed = torch.tensor(time.perf_counter() - t0, device=device)
    if ws_eff > 1:
        torch.distributed.all_reduce(elapsed,
                                     op=torch.distributed.ReduceOp.MAX)
    ms_per_step = elapsed.item() * 1000.0 / args.steps

    # destroy BEFORE printing: RCCL writes a version banner to stdout at
    # teardown, and the driver expects the JSON line to come last. Read the
    # loss BEFORE teardown: destroying the process group invalidates
    # device state backing the graph outputs on this stack.
    comm.barrier()
    comm.destroy()
    if rank == 0:
        out = {
            "metric": "train_step_time_ms",
            "value": ms_per_step,
            "unit": "ms",
            "n_gpus": world_size,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": False,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": args.dtype,
            "data": "synthetic",
            "config": {
                "model": ("FastEGNN-DistEGNN" if args.workload ==
                          "largefluid" else "FastEGNN"),
                "dataset": dataset,
                "global_batch": graphs_per_batch,
                "nodes": nodes,
                "radius": radius,
                "split_mode": args.split_mode,
                "hidden_nf": 64,
                "n_layers": 4,
                "virtual_channels": vch,
                "accumulation_steps": accum,
                "parallelism": f"graph-partition dp{world_size}",
                "hip_graphs": bool(graphed.enabled),
                "coord_mse": final_mse,
            },
        }
        print(json.dumps(out), flush=True)


if __name__ == "__main__":
    main()
