"""Minimal edge-list graph container (replaces the DGL graphs the reference
builds per forward, dynamics.py:84-93 / FastTFN.py:131)."""

from __future__ import annotations

import torch


class EdgeGraph:
    """edge_index [2, M] (src = edge_index[0], dst = edge_index[1]),
    node/edge feature dicts, optional CSR metadata for our segment ops."""

    def __init__(self, edge_index: torch.Tensor, num_nodes: int):
        self.edge_index = edge_index
        self.num_nodes = num_nodes
        self.ndata = {}
        self.edata = {}

    @property
    def src(self):
        return self.edge_index[0]

    @property
    def dst(self):
        return self.edge_index[1]

    @property
    def num_edges(self):
        return self.edge_index.size(1)

    def dst_csr(self):
        """(dstptr [N+1], perm [M]) with dst[perm] sorted — cached; feeds
        the fused CSR edge_softmax kernel."""
        if getattr(self, "_dst_csr", None) is None:
            perm = torch.argsort(self.dst, stable=True)
            ptr = torch.zeros(self.num_nodes + 1, dtype=torch.long,
                              device=self.dst.device)
            ptr.scatter_add_(0, self.dst.index_select(0, perm) + 1,
                             torch.ones_like(perm))
            self._dst_csr = (ptr.cumsum(0), perm)
        return self._dst_csr
