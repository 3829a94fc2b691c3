"""Synthetic workload generators matching the reference's dataset shapes.

There is no network access in this environment, so benchmarks and tests run
on synthetic radius-graph data with random-init weights. Each
generator reproduces the corresponding dataset's published scale and field
semantics (dataset_generation/README.md:7-13 of the reference):

| workload    | N       | edges    | radius | node_feat           | attr        |
|-------------|---------|----------|--------|---------------------|-------------|
| nbody_100   | 100     | 9,900    | inf    | [|v|, q/max q]      | charges [1] |
| protein     | 855     | ~55K     | 10 A   | [|v|, q/max q]      | charges [1] |
| Water-3D    | ~7,806  | ~95K     | 0.035  | [|v|, type/max]     | type [1]    |
| Fluid113K   | ~113,140| ~1.7M    | 0.075  | [visc, mass, |v|]   | [visc,mass] |

Node features follow reference process_dataset.py (:104-107 nbody,
:189-192 protein, :268-271 water, :505-506 fluid). Positions are uniform in
a box sized to reproduce the published average degree at the given radius;
velocities are Gaussian; the target is an integrated position at t+dt with
noise (random-weights training makes the exact dynamics irrelevant to the
benchmark)."""

from __future__ import annotations

import math
from typing import List, Optional

import numpy as np
import torch

from ..ops import reference as ref_ops
from .graph import Data
from .partition import SPLITTERS


WORKLOADS = {
    # name: (num_nodes, radius, avg_degree, feat_kind)
    "nbody_100": (100, -1.0, 99.0, "charge"),
    "protein": (855, 10.0, 64.5, "charge"),
    "Water-3D": (7806, 0.035, 12.2, "type"),
    "Fluid113K": (113140, 0.075, 15.1, "fluid"),
}


def _box_side(n: int, radius: float, avg_degree: float) -> float:
    """Box side so that uniform density gives the target average degree."""
    if radius <= 0:
        return 1.0
    density = avg_degree / (4.0 / 3.0 * math.pi * radius ** 3)
    return (n / density) ** (1.0 / 3.0)


def make_cloud_sample(name: str, rng: torch.Generator,
                      n_override: Optional[int] = None) -> dict:
    """One full-graph raw sample (pos/vel/target/x/attr) for a workload."""
    n, radius, avg_deg, kind = WORKLOADS[name]
    if n_override is not None:
        n = n_override
    side = _box_side(n, radius, avg_deg)
    pos = torch.rand(n, 3, generator=rng) * side
    vel_scale = max(radius, 0.05) * 0.2 if radius > 0 else 0.1
    vel = torch.randn(n, 3, generator=rng) * vel_scale
    target = pos + vel * 5.0 + torch.randn(n, 3, generator=rng) * vel_scale

    speed = vel.pow(2).sum(-1, keepdim=True).sqrt()
    if kind == "charge":
        charges = torch.randint(0, 2, (n, 1), generator=rng).float() * 2 - 1
        x = torch.cat([speed, charges / charges.max().clamp(min=1e-12)], dim=1)
        attr = charges
    elif kind == "type":
        t = torch.randint(1, 4, (n, 1), generator=rng).float()
        x = torch.cat([speed, t / t.max()], dim=1)
        attr = t
    else:  # fluid: attr = [viscosity, mass]
        visc = torch.full((n, 1), 0.01) + torch.rand(n, 1, generator=rng) * 0.001
        mass = torch.full((n, 1), 0.125)
        attr = torch.cat([visc, mass], dim=1)
        x = torch.cat([attr, speed], dim=1)
    return dict(pos=pos, vel=vel, target=target, x=x, attr=attr,
                radius=radius)


def make_cutoff_dataset(name: str, num_samples: int, seed: int = 0,
                        radius: Optional[float] = None,
                        cutoff_rate: float = 0.0,
                        n_override: Optional[int] = None) -> List[Data]:
    """Single-device ("cutoff_edges" mode) synthetic dataset."""
    rng = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(num_samples):
        s = make_cloud_sample(name, rng, n_override=n_override)
        r = s["radius"] if radius is None else radius
        ei = ref_ops.radius_graph(s["pos"], r)
        if cutoff_rate > 0 and ei.size(1) > 0:
            d = (s["pos"][ei[0]] - s["pos"][ei[1]]).pow(2).sum(-1)
            keep = torch.argsort(d)[: int(ei.size(1) * (1 - cutoff_rate))]
            ei = ei[:, keep]
        ea = (s["pos"][ei[0]] - s["pos"][ei[1]]).pow(2).sum(-1).sqrt() \
            .unsqueeze(-1).repeat(1, 2)
        out.append(Data(
            x=s["x"], pos=s["pos"], vel=s["vel"], attr=s["attr"],
            target=s["target"], loc_mean=s["pos"].mean(0, keepdim=True),
            edge_index=ei, edge_attr=ea,
            special_nodes=torch.ones(s["pos"].size(0), dtype=torch.bool)))
    return out


def make_distributed_dataset(name: str, num_samples: int, world_size: int,
                             split_mode: str = "random", seed: int = 0,
                             outer_radius: Optional[float] = None,
                             inner_radius: Optional[float] = None,
                             n_override: Optional[int] = None
                             ) -> List[List[Data]]:
    """Per-rank partitioned synthetic dataset ("distribute" mode).

    Returns data[rank][sample] like the reference's per-rank .pt files."""
    rng = torch.Generator().manual_seed(seed)
    _, radius, _, _ = WORKLOADS[name]
    outer = outer_radius if outer_radius is not None else radius
    inner = inner_radius if inner_radius is not None else radius
    split = SPLITTERS[split_mode]
    data = [[] for _ in range(world_size)]
    for _ in range(num_samples):
        s = make_cloud_sample(name, rng, n_override=n_override)
        if split_mode == "random":
            parts = split(pos=s["pos"], x=s["x"], target=s["target"],
                          vel=s["vel"], attr=s["attr"], radius=inner,
                          world_size=world_size, device="cpu", generator=rng)
        else:
            parts = split(pos=s["pos"], x=s["x"], target=s["target"],
                          vel=s["vel"], attr=s["attr"], outer_radius=outer,
                          inner_radius=inner, world_size=world_size,
                          device="cpu")
        for i, p in enumerate(parts):
            data[i].append(p)
    return data
