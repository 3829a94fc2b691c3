"""Per-model-family training-step timings (forward+loss+backward+Adam).

Covers every `get_model` family on the Water-3D-scale synthetic workload —
breadth companion to the headline bench.py (which owns the BASELINE
contract). Run on a GPU box:

    python benchmarks/zoo.py [--steps 10] [--models FastEGNN,FastRF,...]
"""

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def build_model(name, device):
    from distegnn_amd.models import FastEGNN
    from distegnn_amd.models.fastrf import FastRF
    from distegnn_amd.models.fastschnet import FastSchNet
    from distegnn_amd.models.fasttfn import FastTFN
    from distegnn_amd.models.baselines import (EGNN, RF_vel,
                                                Linear_dynamics, EGHN)
    from distegnn_amd.models.schnet import SchNet
    from distegnn_amd.models.tfn import OurDynamics

    common = dict(hidden_nf=64, virtual_channels=3, n_layers=4)
    if name == "FastEGNN":
        m = FastEGNN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                     world_size=1, **common)
    elif name == "FastRF":
        m = FastRF(edge_attr_nf=2, world_size=1, hidden_nf=64,
                   virtual_channels=3, n_layers=4)
    elif name == "FastSchNet":
        m = FastSchNet(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                       cutoff=0.035, **common)
    elif name == "FastTFN":
        m = FastTFN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                    **common)
    elif name == "EGNN":
        m = EGNN(n_layers=4, in_node_nf=2, in_edge_nf=2, hidden_nf=64,
                 with_v=True)
    elif name == "RF_vel":
        m = RF_vel(hidden_nf=64, edge_attr_nf=2, n_layers=4)
    elif name == "SchNet":
        m = SchNet(hidden_channels=64, num_filters=64, num_interactions=4)
    elif name == "TFN":
        m = OurDynamics(nf=32, n_layers=4, model="tfn", num_degrees=2,
                        div=1)
    elif name == "EGHN":
        m = EGHN(in_node_nf=2, in_edge_nf=2, hidden_nf=64, n_cluster=4,
                 layer_per_block=2, layer_pooling=2)
    elif name == "Linear":
        m = Linear_dynamics()
    else:
        raise ValueError(name)
    return m.to(device)


def run_step(name, model, batch, device):
    from distegnn_amd.runtime.trainer import model_forward

    if name == "EGHN":
        n_node = batch.num_nodes // batch.num_graphs
        with torch.autocast("cuda", dtype=torch.bfloat16,
                            enabled=device.type == "cuda"):
            pred, _, _ = model(batch.pos, batch.x, batch.edge_index,
                               batch.edge_attr, batch.edge_index,
                               batch.edge_attr, n_node, v=batch.vel)
        return torch.nn.functional.mse_loss(pred.float(), batch.target)
    key = {"RF_vel": "RF", "Linear": "Linear"}.get(name, name)
    with torch.autocast("cuda", dtype=torch.bfloat16,
                        enabled=device.type == "cuda"):
        pred, _ = model_forward(model, key, batch, device)
    return torch.nn.functional.mse_loss(pred.float(), batch.target)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--nodes", type=int, default=7806)
    ap.add_argument("--graphs-per-batch", type=int, default=4)
    ap.add_argument("--models", type=str,
                    default="FastEGNN,FastRF,FastSchNet,FastTFN,EGNN,"
                            "RF_vel,SchNet,TFN,EGHN,Linear")
    args = ap.parse_args()
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")

    from distegnn_amd.data.graph import collate
    from distegnn_amd.data.synthetic import make_cutoff_dataset
    from distegnn_amd.utils import fix_seed

    fix_seed(0)
    batch = collate(make_cutoff_dataset(
        "Water-3D", args.graphs_per_batch, seed=0,
        n_override=args.nodes // args.graphs_per_batch)).to(device)
    print(f"# {batch.num_nodes} nodes, {batch.num_edges} edges, "
          f"{batch.num_graphs} graphs, {device}")
    for name in args.models.split(","):
        try:
            fix_seed(0)
            model = build_model(name, device)
            opt = torch.optim.Adam(model.parameters(), lr=5e-4)
            for _ in range(args.warmup):
                opt.zero_grad(set_to_none=False)
                run_step(name, model, batch, device).backward()
                opt.step()
            if device.type == "cuda":
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(args.steps):
                opt.zero_grad(set_to_none=False)
                run_step(name, model, batch, device).backward()
                opt.step()
            if device.type == "cuda":
                torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) * 1000 / args.steps
            print(f"{name:12s} {dt:9.2f} ms/step")
        except Exception as exc:
            print(f"{name:12s} FAILED: {exc}")


if __name__ == "__main__":
    main()
