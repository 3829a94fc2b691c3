"""Quantify the graph partitioners: edge-cut fraction + balance.

VERDICT round-1 gap #6: the in-tree balanced recursive bisection replaces
libmetis (reference datasets/distribute_graphs.py:151-185, which wraps
torch_sparse/pyg_lib METIS) but its cut quality was never measured. This
tool builds the synthetic LargeFluid-113K cloud (the headline workload),
partitions it with every splitter backend at world_size 2/4/8, and reports:

* cut fraction — share of outer-radius-graph edges whose endpoints land in
  different partitions. DistEGNN DROPS cross-partition real edges
  (SURVEY.md §2.2), so the cut fraction directly bounds the information
  the distributed model loses vs the merged graph.
* balance — max partition size / ideal size (1.0 = perfectly balanced;
  METIS's default tolerance is 1.03).

Run: ``python tools/partition_quality.py [--nodes N] [--json out.json]``.
Results are committed in profiles/partition_quality.json and discussed in
docs/PARTITIONING.md.
"""

import argparse
import json
import sys
import time
import os

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from distegnn_amd.data.partition import (graph_partition, kmeans_clustering,
                                         spectral_clustering)
from distegnn_amd.ops import reference as ref_ops


def cut_stats(labels: torch.Tensor, edge_index: torch.Tensor, ws: int):
    cut = (labels[edge_index[0]] != labels[edge_index[1]]).float().mean()
    sizes = torch.bincount(labels, minlength=ws).float()
    ideal = labels.numel() / ws
    return float(cut), float(sizes.max() / ideal), [int(s) for s in sizes]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=113140)
    ap.add_argument("--radius", type=float, default=0.075)
    ap.add_argument("--seed", type=int, default=43)
    ap.add_argument("--json", type=str, default=None)
    ap.add_argument("--skip-spectral", action="store_true",
                    help="spectral is O(N^2) memory in sklearn's rbf "
                         "affinity — skip for the full 113K cloud")
    args = ap.parse_args()

    from distegnn_amd.data.synthetic import make_cloud_sample

    rng = torch.Generator().manual_seed(args.seed)
    s = make_cloud_sample("Fluid113K", rng, n_override=args.nodes)
    pos = s["pos"]
    t0 = time.perf_counter()
    edge_index = ref_ops.radius_graph(pos, args.radius)
    print(f"# cloud: {args.nodes} nodes, {edge_index.size(1)} edges "
          f"(radius graph in {time.perf_counter() - t0:.1f}s)")

    results = {"nodes": args.nodes, "edges": int(edge_index.size(1)),
               "radius": args.radius, "seed": args.seed, "splitters": {}}
    for ws in (2, 4, 8):
        for name in ("random", "metis", "kmeans", "spectral"):
            if name == "spectral" and (args.skip_spectral
                                       or args.nodes > 20000):
                continue
            t0 = time.perf_counter()
            if name == "random":
                g = torch.Generator().manual_seed(args.seed)
                labels = torch.zeros(args.nodes, dtype=torch.long)
                perm = torch.randperm(args.nodes, generator=g)
                chunk = args.nodes // ws
                for i in range(ws):
                    end = args.nodes if i == ws - 1 else (i + 1) * chunk
                    labels[perm[i * chunk:end]] = i
            elif name == "metis":
                labels = graph_partition(edge_index, args.nodes, ws, pos=pos)
            elif name == "kmeans":
                labels = kmeans_clustering(pos, ws)
            else:
                labels = spectral_clustering(pos, ws)
            dt = time.perf_counter() - t0
            cut, bal, sizes = cut_stats(labels, edge_index, ws)
            results["splitters"].setdefault(name, {})[str(ws)] = {
                "cut_fraction": round(cut, 5), "balance": round(bal, 4),
                "sizes": sizes, "seconds": round(dt, 2)}
            print(f"ws={ws} {name:9s} cut={cut * 100:6.2f}%  "
                  f"balance={bal:.3f}  ({dt:.1f}s)")

    if args.json:
        with open(args.json, "w") as f:
            json.dump(results, f, indent=2)
        print(f"# wrote {args.json}")


if __name__ == "__main__":
    main()
