"""TFN / SE(3)-Transformer models and the dynamics wrapper.

Parity with reference models/se3_dynamics/models.py (TFN :15-75, OursTFN
:78-139, SE3Transformer :142-204, OurSE3Transformer :207-295) and
dynamics.py (OurDynamics :10-105, connect_fully :152-171), rebuilt on our
DGL-free SE(3) stack (models/se3/)."""

from __future__ import annotations

import numpy as np
import torch
from torch import nn

from .se3.basis import get_basis_and_r
from .se3.fibers import Fiber
from .se3.graph import EdgeGraph
from .se3.modules import (GAvgPooling, GConvSE3, GMaxPooling, GNormSE3,
                          GSE3Res)


class TFN(nn.Module):
    """SE(3)-equivariant tensor-field network (reference models.py:15-75)."""

    def __init__(self, num_layers, atom_feature_size, num_channels,
                 num_nlayers=1, num_degrees=4, edge_dim=4, **kwargs):
        super().__init__()
        self.num_layers = num_layers
        self.num_nlayers = num_nlayers
        self.num_channels = num_channels
        self.num_degrees = num_degrees
        self.num_channels_out = num_channels * num_degrees
        self.edge_dim = edge_dim
        self.fibers = {"in": Fiber(1, atom_feature_size),
                       "mid": Fiber(num_degrees, num_channels),
                       "out": Fiber(1, self.num_channels_out)}
        block0 = []
        fin = self.fibers["in"]
        for _ in range(num_layers - 1):
            block0.append(GConvSE3(fin, self.fibers["mid"],
                                   self_interaction=True,
                                   edge_dim=edge_dim))
            block0.append(GNormSE3(self.fibers["mid"],
                                   num_layers=num_nlayers))
            fin = self.fibers["mid"]
        block0.append(GConvSE3(self.fibers["mid"], self.fibers["out"],
                               self_interaction=True, edge_dim=edge_dim))
        self.block0 = nn.ModuleList(block0)
        self.block1 = nn.ModuleList([GMaxPooling()])
        self.block2 = nn.ModuleList([
            nn.Linear(self.num_channels_out, self.num_channels_out),
            nn.ReLU(inplace=True),
            nn.Linear(self.num_channels_out, 1)])

    def forward(self, G):
        basis, r = get_basis_and_r(G, self.num_degrees - 1)
        h = {"0": G.ndata["f"]}
        for layer in self.block0:
            h = layer(h, G=G, r=r, basis=basis)
        h = h["0"][..., -1]
        for layer in self.block1:
            h = layer(G, h)
        for layer in self.block2:
            h = layer(h)
        return h


class OursTFN(nn.Module):
    """TFN variant with vector in/out fibers (reference models.py:78-139)."""

    def __init__(self, num_layers, num_channels, num_nlayers=1,
                 num_degrees=4, act_fn=None, edge_dim=4, out_types={1: 1},
                 in_types={0: 1, 1: 1}, **kwargs):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.ReLU()
        self.num_layers = num_layers
        self.num_degrees = num_degrees
        self.edge_dim = edge_dim
        self.fibers = {"in": Fiber(dictionary=in_types),
                       "mid": Fiber(num_degrees, num_channels),
                       "out": Fiber(dictionary=out_types)}
        block0 = []
        fin = self.fibers["in"]
        for _ in range(num_layers - 1):
            block0.append(GConvSE3(fin, self.fibers["mid"],
                                   self_interaction=True, edge_dim=edge_dim,
                                   act_fn=act_fn))
            block0.append(GNormSE3(self.fibers["mid"],
                                   num_layers=num_nlayers, act_fn=act_fn))
            fin = self.fibers["mid"]
        block0.append(GConvSE3(self.fibers["mid"], self.fibers["out"],
                               self_interaction=True, edge_dim=edge_dim,
                               act_fn=act_fn))
        self.block0 = nn.ModuleList(block0)

    def forward(self, G):
        basis, r = get_basis_and_r(G, self.num_degrees - 1)
        h = {"0": G.ndata["f"], "1": G.ndata["f1"]}
        for layer in self.block0:
            h = layer(h, G=G, r=r, basis=basis)
        return h


class SE3Transformer(nn.Module):
    """SE(3)-Transformer (reference models.py:142-204)."""

    def __init__(self, num_layers, atom_feature_size, num_channels,
                 num_nlayers=1, num_degrees=4, edge_dim=4, div=4,
                 pooling="avg", n_heads=1, **kwargs):
        super().__init__()
        self.num_layers = num_layers
        self.num_degrees = num_degrees
        self.edge_dim = edge_dim
        self.div = div
        self.pooling = pooling
        self.n_heads = n_heads
        self.fibers = {"in": Fiber(1, atom_feature_size),
                       "mid": Fiber(num_degrees, num_channels),
                       "out": Fiber(1, num_degrees * num_channels)}
        gblock = []
        fin = self.fibers["in"]
        for _ in range(num_layers):
            gblock.append(GSE3Res(fin, self.fibers["mid"], edge_dim=edge_dim,
                                  div=div, n_heads=n_heads))
            gblock.append(GNormSE3(self.fibers["mid"]))
            fin = self.fibers["mid"]
        gblock.append(GConvSE3(self.fibers["mid"], self.fibers["out"],
                               self_interaction=True, edge_dim=edge_dim))
        if pooling == "avg":
            gblock.append(GAvgPooling())
        elif pooling == "max":
            gblock.append(GMaxPooling())
        self.Gblock = nn.ModuleList(gblock)
        nf = self.fibers["out"].n_features
        self.FCblock = nn.ModuleList([nn.Linear(nf, nf),
                                      nn.ReLU(inplace=True),
                                      nn.Linear(nf, 1)])

    def forward(self, G):
        basis, r = get_basis_and_r(G, self.num_degrees - 1)
        h = {"0": G.ndata["f"]}
        for layer in self.Gblock:
            if isinstance(layer, (GAvgPooling, GMaxPooling)):
                h = layer(G, h["0"][..., -1] if isinstance(h, dict) else h)
            else:
                h = layer(h, G=G, r=r, basis=basis)
        for layer in self.FCblock:
            h = layer(h)
        return h


class OurSE3Transformer(nn.Module):
    """SE(3)-Transformer with vector fibers (reference models.py:207-295)."""

    def __init__(self, num_layers, num_channels, num_nlayers=1,
                 num_degrees=4, edge_dim=4, div=1, pooling="avg", n_heads=1,
                 act_fn=None, out_types={1: 1}, in_types={0: 1, 1: 1},
                 **kwargs):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.ReLU()
        self.num_layers = num_layers
        self.num_degrees = num_degrees
        self.edge_dim = edge_dim
        self.div = div
        self.n_heads = n_heads
        self.fibers = {"in": Fiber(dictionary=in_types),
                       "mid": Fiber(num_degrees, num_channels),
                       "out": Fiber(dictionary=out_types)}
        gblock = []
        fin = self.fibers["in"]
        for _ in range(num_layers):
            gblock.append(GSE3Res(fin, self.fibers["mid"], edge_dim=edge_dim,
                                  div=div, n_heads=n_heads, act_fn=act_fn,
                                  learnable_skip=False))
            gblock.append(GNormSE3(self.fibers["mid"], act_fn=act_fn))
            fin = self.fibers["mid"]
        gblock.append(GConvSE3(self.fibers["mid"], self.fibers["out"],
                               self_interaction=True, edge_dim=edge_dim,
                               act_fn=act_fn))
        self.Gblock = nn.ModuleList(gblock)
        self.scalar_trick = nn.Parameter(torch.ones(1) * 0.01)

    def forward(self, G):
        basis, r = get_basis_and_r(G, self.num_degrees - 1)
        h = {"0": G.ndata["f"], "1": G.ndata["f1"]}
        for layer in self.Gblock:
            h = layer(h, G=G, r=r, basis=basis)
        for key in h:
            h[key] = h[key] * self.scalar_trick
        return h


def connect_fully(num_atoms):
    """All-pairs edge list (reference dynamics.py:152-171)."""
    src, dst = [], []
    for i in range(num_atoms):
        for j in range(num_atoms):
            if i != j:
                src.append(i)
                dst.append(j)
    w = np.ones(len(src))
    return np.array(src), np.array(dst), w


# Degree-1 fibers live in the real-SH component order (y, z, x): our
# Wigner-D matrices are defined by Y_1(R x) = D_1 Y_1(x), and Y_1 is
# proportional to (y, z, x). Raw xyz vectors are permuted on the way in and
# back on the way out so the feature transformation law matches D_1.
_XYZ_TO_SH = [1, 2, 0]
_SH_TO_XYZ = [2, 0, 1]


class OurDynamics(nn.Module):
    """TFN/SE(3)-Transformer dynamics wrapper (reference dynamics.py:10-105).

    forward(xs, vs, charges, edge_index) -> updated positions [N, 3]."""

    def __init__(self, nf=16, n_layers=3, act_fn=None, model="se3_transformer",
                 num_degrees=4, div=1):
        super().__init__()
        act_fn = act_fn if act_fn is not None else nn.ReLU()
        self._n_dimension = 3
        if model == "se3_transformer":
            self.se3 = OurSE3Transformer(num_layers=n_layers,
                                         num_channels=nf, edge_dim=0,
                                         act_fn=act_fn,
                                         num_degrees=num_degrees, div=div)
        elif model == "tfn":
            self.se3 = OursTFN(num_layers=n_layers, num_channels=nf,
                               edge_dim=0, div=1, act_fn=act_fn,
                               num_degrees=num_degrees)
        else:
            raise Exception("Wrong model")

    def forward(self, xs, vs, charges, edge_index):
        G = EdgeGraph(edge_index, xs.size(0))
        G.ndata["x"] = xs
        G.ndata["vel"] = vs[:, _XYZ_TO_SH].unsqueeze(1)
        G.ndata["f"] = charges.unsqueeze(2)
        G.ndata["f1"] = G.ndata["vel"]
        G.edata["d"] = xs[edge_index[1]] - xs[edge_index[0]]
        out = self.se3(G)["1"].view(xs.size())[:, _SH_TO_XYZ]
        return out + xs
