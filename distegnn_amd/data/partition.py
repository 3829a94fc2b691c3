"""Graph partitioners for DistEGNN spatial parallelism.

Parity with reference datasets/distribute_graphs.py:17-223: one large
geometric graph is split into ``world_size`` node partitions OFFLINE; each
partition rebuilds its internal radius graph at ``inner_radius`` and stores
the GLOBAL ``loc_mean``; cross-partition edges are dropped (the virtual-node
all-reduce is the only cross-rank coupling at train time).

Split modes (reference :17,54,90,118):
* random   — random permutation into equal chunks
* metis    — k-way partition of the outer-radius graph. The reference binds
  libmetis through torch_sparse/pyg_lib; neither exists in this image, so we
  own a multilevel-free recursive bisection partitioner (geometric-seeded
  BFS growth with balance constraint) in ``_graph_bisection``. Same contract
  (balanced parts, locality-aware), deterministic under the given seed.
* spectral — sklearn SpectralClustering on positions (RBF affinity, sigma =
  median pairwise distance of <=2000 samples) (reference :201-223)
* kmeans   — sklearn KMeans on positions (reference :188-198)

All partitioners are CPU-only (they run in offline preprocessing, reference
asserts device=='cpu' :18,55,91,119).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np
import torch

from ..ops import reference as ref_ops
from .graph import Data


def _edge_attr_from_pos(pos: torch.Tensor, edge_index: torch.Tensor) -> torch.Tensor:
    """Edge feature = pairwise distance, repeated x2 (reference :44,80)."""
    d = (pos[edge_index[0]] - pos[edge_index[1]]).pow(2).sum(-1).sqrt()
    return d.unsqueeze(-1).repeat(1, 2)


def _make_partition_data(pos, x, target, vel, attr, idx, inner_radius,
                         loc_mean, special_nodes=None) -> Data:
    pos_i = pos[idx]
    ei = ref_ops.radius_graph(pos_i, inner_radius)
    fields = dict(
        x=x[idx], pos=pos_i, vel=vel[idx], attr=attr[idx], target=target[idx],
        loc_mean=loc_mean, edge_index=ei, edge_attr=_edge_attr_from_pos(pos_i, ei),
    )
    if special_nodes is not None:
        fields["special_nodes"] = special_nodes[idx]
    return Data(**fields)


def split_large_graph_random(pos, x, target, vel, attr, radius, world_size,
                             device="cpu", special_nodes=None,
                             generator: Optional[torch.Generator] = None
                             ) -> List[Data]:
    """Random equal-chunk split (reference :17-51)."""
    assert device == "cpu"
    n = pos.size(0)
    if special_nodes is None:
        special_nodes = torch.ones(n, dtype=torch.bool)
    indices = torch.randperm(n, generator=generator)
    sizes = [n // world_size] * (world_size - 1)
    sizes.append(n - sum(sizes))
    chunks = torch.split(indices, sizes)
    loc_mean = pos.mean(dim=0, keepdim=True)
    return [
        _make_partition_data(pos, x, target, vel, attr, c, radius, loc_mean,
                             special_nodes)
        for c in chunks
    ]


def split_large_graph_metis(pos, x, target, vel, attr, outer_radius,
                            inner_radius, world_size, device="cpu",
                            special_nodes=None) -> List[Data]:
    """Graph-topology partition of the outer-radius graph (reference :54-87).

    Uses our own balanced recursive graph bisection (libmetis is not in this
    image; contract and output format match the reference's METIS path)."""
    assert device == "cpu"
    n = pos.size(0)
    if special_nodes is None:
        special_nodes = torch.ones(n, dtype=torch.bool)
    edge_index = ref_ops.radius_graph(pos, outer_radius)
    cluster = graph_partition(edge_index, n, world_size, pos=pos)
    loc_mean = pos.mean(dim=0, keepdim=True)
    return [
        _make_partition_data(pos, x, target, vel, attr,
                             (cluster == i).nonzero(as_tuple=True)[0],
                             inner_radius, loc_mean, special_nodes)
        for i in range(world_size)
    ]


def split_large_graph_spectral(pos, x, target, vel, attr, outer_radius,
                               inner_radius, world_size, device="cpu"
                               ) -> List[Data]:
    assert device == "cpu"
    cluster = spectral_clustering(pos, world_size)
    loc_mean = pos.mean(dim=0, keepdim=True)
    return [
        _make_partition_data(pos, x, target, vel, attr,
                             (cluster == i).nonzero(as_tuple=True)[0],
                             inner_radius, loc_mean)
        for i in range(world_size)
    ]


def split_large_graph_kmeans(pos, x, target, vel, attr, outer_radius,
                             inner_radius, world_size, device="cpu"
                             ) -> List[Data]:
    assert device == "cpu"
    cluster = kmeans_clustering(pos, world_size)
    loc_mean = pos.mean(dim=0, keepdim=True)
    return [
        _make_partition_data(pos, x, target, vel, attr,
                             (cluster == i).nonzero(as_tuple=True)[0],
                             inner_radius, loc_mean)
        for i in range(world_size)
    ]


SPLITTERS = {
    "random": split_large_graph_random,
    "metis": split_large_graph_metis,
    "spectral": split_large_graph_spectral,
    "kmeans": split_large_graph_kmeans,
}


# --------------------------------------------------------------------------
# clustering back-ends
# --------------------------------------------------------------------------

def kmeans_clustering(pos: torch.Tensor, num_parts: int,
                      random_state: int = 0) -> torch.Tensor:
    from sklearn.cluster import KMeans

    labels = KMeans(n_clusters=num_parts, random_state=random_state,
                    n_init="auto").fit_predict(
        pos.detach().cpu().numpy().astype(np.float32))
    return torch.from_numpy(labels).to(torch.long)


def spectral_clustering(pos: torch.Tensor, num_parts: int,
                        sigma: Optional[float] = None,
                        random_state: int = 0) -> torch.Tensor:
    """RBF spectral clustering on positions (reference :201-223).

    sigma defaults to the median pairwise distance over <=2000 samples."""
    from sklearn.cluster import SpectralClustering

    xx = pos.detach().cpu().numpy().astype(np.float32)
    n = xx.shape[0]
    if sigma is None:
        m = min(n, 2000)
        idx = np.random.RandomState(0).choice(n, size=m, replace=False)
        d = np.linalg.norm(xx[idx, None, :] - xx[None, idx, :], axis=2)
        sigma = float(np.median(d[d > 0]) + 1e-12)
    gamma = 1.0 / (2.0 * sigma * sigma)
    sc = SpectralClustering(n_clusters=num_parts, affinity="rbf", gamma=gamma,
                            assign_labels="kmeans",
                            random_state=random_state, eigen_solver="arpack")
    labels = sc.fit_predict(xx)
    return torch.from_numpy(labels).to(torch.long)


def graph_partition(edge_index: torch.Tensor, num_nodes: int, num_parts: int,
                    pos: Optional[torch.Tensor] = None,
                    seed: int = 0) -> torch.Tensor:
    """Balanced k-way node partition of a graph (METIS-mode back-end).

    With ``pos`` (the DistEGNN case — radius graphs ARE geometric):
    recursive coordinate bisection, median split along each subcloud's
    principal axis. Cuts are near-planar, so for a radius graph the cut
    fraction approaches the geometric optimum (measured on the 113K
    synthetic cloud: 2.0%/5.3%/8.7% at ws=2/4/8 vs 3.8%/7.8%/11.2% for
    kmeans and ~50-87% for random — profiles/partition_quality.json),
    partitions are EXACTLY balanced (better than METIS's 1.03 tolerance)
    and the result is deterministic, so the preprocessing cache key stays
    meaningful (SURVEY.md §7 hard-part 7).

    Without ``pos``: bisection grows one half by BFS from node 0 —
    topology-only fallback with the same balance guarantee.
    """
    order = np.arange(num_nodes)
    labels = np.zeros(num_nodes, dtype=np.int64)
    row = edge_index[0].cpu().numpy()
    col = edge_index[1].cpu().numpy()
    # CSR adjacency once
    perm = np.argsort(row, kind="stable")
    row_s, col_s = row[perm], col[perm]
    rowptr = np.zeros(num_nodes + 1, dtype=np.int64)
    np.add.at(rowptr, row_s + 1, 1)
    rowptr = np.cumsum(rowptr)
    p = pos.detach().cpu().numpy() if pos is not None else None

    def bisect(nodes: np.ndarray, parts: int, base_label: int):
        if parts == 1:
            labels[nodes] = base_label
            return
        left_parts = parts // 2
        target_left = int(round(len(nodes) * left_parts / parts))
        if p is not None and len(nodes) > 1:
            # Recursive coordinate bisection: median split along the best
            # of {x, y, z, principal axis} — chosen by the ACTUAL edge cut
            # each candidate plane produces on this subgraph (an isotropic
            # cloud's principal axis degenerates to a diagonal, whose cut
            # plane is up to sqrt(3)x larger than an axis-aligned one).
            # Near-planar cuts, exact balance, deterministic.
            sub = p[nodes]
            centered = sub - sub.mean(0)
            cov = centered.T @ centered
            v = np.ones(cov.shape[0])
            for _ in range(16):
                v = cov @ v
                nv = np.linalg.norm(v)
                if nv == 0:
                    break
                v /= nv
            inset = np.zeros(num_nodes, dtype=bool)
            inset[nodes] = True
            sub_e = inset[row_s] & inset[col_s]
            er, ec = row_s[sub_e], col_s[sub_e]
            best = None
            cands = [centered[:, d] for d in range(centered.shape[1])]
            cands.append(centered @ v)
            side = np.zeros(num_nodes, dtype=bool)
            for proj in cands:
                order_ax = np.argsort(proj, kind="stable")
                side[nodes] = False
                side[nodes[order_ax[:target_left]]] = True
                cut = int(np.count_nonzero(side[er] != side[ec]))
                if best is None or cut < best[0]:
                    best = (cut, order_ax)
            order_ax = best[1]
            left = nodes[order_ax[:target_left]]
            right = nodes[order_ax[target_left:]]
            bisect(left, left_parts, base_label)
            bisect(right, parts - left_parts, base_label + left_parts)
            return
        inset = np.zeros(num_nodes, dtype=bool)
        inset[nodes] = True
        seed_node = nodes[0]
        chosen = np.zeros(num_nodes, dtype=bool)
        frontier = [seed_node]
        chosen[seed_node] = True
        count = 1
        qi = 0
        while count < target_left:
            if qi >= len(frontier):
                # disconnected remainder: pick an unchosen in-set node
                rest = nodes[~chosen[nodes]]
                if len(rest) == 0:
                    break
                nxt = rest[0]
                frontier.append(nxt)
                chosen[nxt] = True
                count += 1
                continue
            u = frontier[qi]
            qi += 1
            for e in range(rowptr[u], rowptr[u + 1]):
                w = col_s[e]
                if inset[w] and not chosen[w]:
                    chosen[w] = True
                    frontier.append(w)
                    count += 1
                    if count >= target_left:
                        break
        left = nodes[chosen[nodes]]
        right = nodes[~chosen[nodes]]
        bisect(left, left_parts, base_label)
        bisect(right, parts - left_parts, base_label + left_parts)

    bisect(order, num_parts, 0)
    return torch.from_numpy(labels)
