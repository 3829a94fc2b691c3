"""Graph sample / mini-batch containers (replaces torch_geometric.data).

The reference stores each sample as a PyG ``Data`` with fields
``x, pos, vel, attr, target, loc_mean, edge_index, edge_attr[, special_nodes]``
(reference datasets/distribute_graphs.py:46-49) and batches them with PyG's
``DataLoader(follow_batch=['edge_index'])`` (reference main.py:184-190).

We own a minimal, MI355X-first equivalent:

* ``Data`` — a plain field container (CPU tensors).
* ``Batch`` — disjoint-union collation. Edge lists are kept **sorted by
  destination row** and a CSR ``rowptr`` plus per-graph ``ptr`` are built at
  collate time on the host, so every device-side aggregation (segment mean,
  graph pooling) runs as a deterministic CSR segmented reduction — no
  atomics, no ``.item()`` host syncs in the hot loop (the reference's
  ``batch_total`` python loop, FastEGNN.py:196,226,260, is replaced by the
  precomputed ``counts`` tensor).
"""

from __future__ import annotations

from typing import Iterable, List, Optional, Sequence

import torch

# Node-indexed fields concatenated along dim 0 at collate time.
_NODE_FIELDS = ("x", "pos", "vel", "attr", "target", "special_nodes")
_EDGE_FIELDS = ("edge_attr",)


def sort_edges_by_row(edge_index: torch.Tensor, edge_attr: Optional[torch.Tensor]):
    """Sort an edge list by destination row (edge_index[0]), stably.

    Aggregations in the models scatter into ``row = edge_index[0]``
    (reference models/FastEGNN.py:166-173,203-206); keeping edges row-sorted
    lets the HIP kernels use CSR segments instead of atomics.
    """
    if edge_index.numel() == 0:
        return edge_index, edge_attr
    order = torch.argsort(edge_index[0], stable=True)
    edge_index = edge_index[:, order]
    if edge_attr is not None:
        edge_attr = edge_attr[order]
    return edge_index, edge_attr


def build_rowptr(row: torch.Tensor, num_nodes: int) -> torch.Tensor:
    """CSR row pointer from a (sorted) row vector."""
    return torch.cat(
        [row.new_zeros(1), torch.cumsum(torch.bincount(row, minlength=num_nodes), 0)]
    )


class Data:
    """One graph sample. All tensors live on CPU until batched + moved."""

    def __init__(self, x=None, pos=None, vel=None, attr=None, target=None,
                 loc_mean=None, edge_index=None, edge_attr=None,
                 special_nodes=None, **extra):
        self.x = x
        self.pos = pos
        self.vel = vel
        self.attr = attr
        self.target = target
        self.loc_mean = loc_mean  # [1, 3] global mean of the FULL graph
        if edge_index is not None:
            edge_index, edge_attr = sort_edges_by_row(edge_index, edge_attr)
        self.edge_index = edge_index
        self.edge_attr = edge_attr
        self.special_nodes = special_nodes
        for k, v in extra.items():
            setattr(self, k, v)

    @property
    def num_nodes(self) -> int:
        return 0 if self.pos is None else self.pos.size(0)

    @property
    def num_edges(self) -> int:
        return 0 if self.edge_index is None else self.edge_index.size(1)

    def fields(self) -> dict:
        return {k: v for k, v in self.__dict__.items() if v is not None}

    def __repr__(self):
        parts = [f"{k}={tuple(v.shape)}" for k, v in self.fields().items()
                 if torch.is_tensor(v)]
        return f"Data({', '.join(parts)})"


class Batch:
    """Disjoint union of ``Data`` graphs with CSR metadata.

    Attributes
    ----------
    x, pos, vel, attr, target : concatenated node tensors
    edge_index : [2, M] with per-graph node offsets applied; globally sorted
        by row (graphs are concatenated in id order, each sorted).
    edge_attr  : [M, e]
    batch      : [N] graph id per node
    ptr        : [B+1] node offsets per graph
    rowptr     : [N+1] CSR pointer of edge destination rows
    loc_mean   : [B, 3]
    counts     : [B] float node counts per graph (device-side, replaces the
        reference's per-graph ``.item()`` loops)
    """

    def __init__(self, data_list: Sequence[Data]):
        b = len(data_list)
        node_offsets = [0]
        for d in data_list:
            node_offsets.append(node_offsets[-1] + d.num_nodes)
        self.num_graphs = b
        self.ptr = torch.tensor(node_offsets, dtype=torch.long)

        for f in _NODE_FIELDS:
            vals = [getattr(d, f) for d in data_list]
            if all(v is not None for v in vals):
                setattr(self, f, torch.cat(vals, dim=0))
            else:
                setattr(self, f, None)

        eis = []
        for i, d in enumerate(data_list):
            if d.edge_index is None:
                continue
            eis.append(d.edge_index + node_offsets[i])
        self.edge_index = (torch.cat(eis, dim=1) if eis
                           else torch.zeros(2, 0, dtype=torch.long))
        eattrs = [d.edge_attr for d in data_list if d.edge_attr is not None]
        self.edge_attr = torch.cat(eattrs, dim=0) if eattrs else None

        self.batch = torch.repeat_interleave(
            torch.arange(b, dtype=torch.long),
            torch.tensor([d.num_nodes for d in data_list], dtype=torch.long),
        )
        self.loc_mean = torch.cat(
            [d.loc_mean for d in data_list], dim=0
        ) if data_list[0].loc_mean is not None else None

        n = int(self.ptr[-1])
        row = self.edge_index[0]
        self.rowptr = torch.cat(
            [row.new_zeros(1), torch.cumsum(torch.bincount(row, minlength=n), 0)]
        )
        self.counts = (self.ptr[1:] - self.ptr[:-1]).to(torch.float32)

        # Column-side CSR for gather backward: sorting permutation of the
        # edge destinations (edge_index[1]) + its pointer. Lets the HIP path
        # run scatter-free segment sums in the backward of h[col] gathers
        # (ops._GatherRowsFn).
        col = self.edge_index[1]
        self.col_perm = torch.argsort(col, stable=True)
        self.colptr = torch.cat(
            [col.new_zeros(1), torch.cumsum(torch.bincount(col, minlength=n), 0)]
        )

        # Precomputed chunk tables for the HIP two-stage graph pooling when
        # per-graph node blocks are huge (DistEGNN: one 100K+ node graph per
        # rank). Built HERE on the host (ptr is host-known) so the training
        # hot loop never syncs. See ops/csrc/segment_reduce.hip.
        max_seg = int((self.ptr[1:] - self.ptr[:-1]).max()) if b > 0 else 0
        if max_seg > 4096:
            # 256-row chunks: a 113K-node partition yields ~440 chunks ->
            # enough blocks to fill 256 CUs (2048-row chunks left the chip
            # 4/5 idle — measured 416-600 us per pool, 10 ms/step)
            chunk = 256
            cb, ce, scp = [], [], [0]
            for i in range(b):
                s, e = int(self.ptr[i]), int(self.ptr[i + 1])
                for k in range(s, e, chunk):
                    cb.append(k)
                    ce.append(min(k + chunk, e))
                scp.append(len(cb))
            self.pool_chunk_begin = torch.tensor(cb, dtype=torch.long)
            self.pool_chunk_end = torch.tensor(ce, dtype=torch.long)
            self.pool_seg_chunk_ptr = torch.tensor(scp, dtype=torch.long)
        else:
            self.pool_chunk_begin = None
            self.pool_chunk_end = None
            self.pool_seg_chunk_ptr = None

    @property
    def num_nodes(self) -> int:
        return int(self.ptr[-1])

    @property
    def num_edges(self) -> int:
        return self.edge_index.size(1)

    def to(self, device, non_blocking: bool = True) -> "Batch":
        for k, v in self.__dict__.items():
            if torch.is_tensor(v):
                setattr(self, k, v.to(device, non_blocking=non_blocking))
        return self

    def pin_memory(self) -> "Batch":
        for k, v in self.__dict__.items():
            if torch.is_tensor(v):
                setattr(self, k, v.pin_memory())
        return self


def collate(data_list: Sequence[Data]) -> Batch:
    return Batch(data_list)
