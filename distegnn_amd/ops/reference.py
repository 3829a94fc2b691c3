"""Eager fp32 reference implementations of the framework's core ops.

These are the semantics contract for the HIP/CDNA4 kernels in ``csrc/``:
every GPU kernel has a numerics test comparing it against these functions in
fp32 (tests/test_ops_*.py). They also serve as the CPU execution path
(preprocessing, CPU-only tests, plumbing runs).

Op inventory (reference-repo counterparts cited per function):
* segment_sum / segment_mean  — torch ``scatter_add_`` helpers
  (reference models/FastEGNN.py:322-337, models/basic.py:50-66)
* graph_sum_pool / graph_mean_pool — PyG ``global_mean_pool``
  (reference models/FastEGNN.py:193,222,258)
* radius_graph — PyG/torch_cluster ``radius_graph``
  (reference datasets/distribute_graphs.py:43,65,79)
"""

from __future__ import annotations

from typing import Optional

import torch


def segment_sum(data: torch.Tensor, row: torch.Tensor, num_segments: int,
                rowptr: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Sum rows of ``data`` [M, F] into segments given by ``row`` [M] → [N, F]."""
    out = data.new_zeros((num_segments,) + data.shape[1:])
    return out.index_add(0, row, data)


def segment_mean(data: torch.Tensor, row: torch.Tensor, num_segments: int,
                 rowptr: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Mean-aggregate rows of ``data`` into ``num_segments`` segments.

    Empty segments yield 0 (the reference clamps the count to min 1,
    models/FastEGNN.py:337)."""
    s = segment_sum(data, row, num_segments)
    cnt = torch.bincount(row, minlength=num_segments).clamp_(min=1)
    return s / cnt.to(s.dtype).view(-1, *([1] * (data.dim() - 1)))


def graph_sum_pool(x: torch.Tensor, batch: torch.Tensor, num_graphs: int) -> torch.Tensor:
    out = x.new_zeros((num_graphs,) + x.shape[1:])
    return out.index_add(0, batch, x)


def graph_mean_pool(x: torch.Tensor, batch: torch.Tensor, num_graphs: int,
                    counts: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Per-graph mean of node rows (PyG ``global_mean_pool`` equivalent).

    ``counts`` [B] may be precomputed (Batch.counts) to avoid a bincount.
    """
    s = graph_sum_pool(x, batch, num_graphs)
    if counts is None:
        counts = torch.bincount(batch, minlength=num_graphs).to(x.dtype)
    return s / counts.clamp(min=1).to(s.dtype).view(-1, *([1] * (x.dim() - 1)))


def radius_graph(pos: torch.Tensor, r: float, loop: bool = False) -> torch.Tensor:
    """All directed pairs (i, j), i != j, with ||pos_i - pos_j|| <= r.

    Returns ``edge_index`` [2, M] sorted by row (edge_index[0]). CPU path
    uses a scipy cKDTree (offline preprocessing parity with the reference's
    CPU radius_graph, distribute_graphs.py:43); the GPU path is the HIP
    cell-list kernel (csrc/radius.hip) behind ops dispatch.

    ``r < 0`` means the full graph (reference nbody config ``radius: -1``).
    """
    n = pos.size(0)
    device = pos.device
    if r is None or r < 0:
        idx = torch.arange(n, device=device)
        row = idx.repeat_interleave(n)
        col = idx.repeat(n)
        if not loop:
            keep = row != col
            row, col = row[keep], col[keep]
        return torch.stack([row, col], dim=0)

    from scipy.spatial import cKDTree

    p = pos.detach().cpu().numpy()
    tree = cKDTree(p)
    pairs = tree.query_pairs(r, output_type="ndarray")  # i < j, dist <= r
    if pairs.size == 0:
        return torch.zeros(2, 0, dtype=torch.long, device=device)
    import numpy as np

    row = np.concatenate([pairs[:, 0], pairs[:, 1]])
    col = np.concatenate([pairs[:, 1], pairs[:, 0]])
    order = np.argsort(row, kind="stable")
    ei = torch.from_numpy(np.stack([row[order], col[order]]).astype("int64"))
    return ei.to(device)
