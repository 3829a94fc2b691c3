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
