"""Standalone SE(3)-equivariance check (reference equivariant_test.py parity).

Builds a random 10-node graph, applies a random rotation + translation, and
asserts f(xR + t) == f(x)R + t to atol 1e-4 for FastEGNN. Run directly:

    python equivariant_test.py [--model FastEGNN|FastRF]

The pytest suite covers the same property (and more models, in fp64 at
atol 1e-9) in tests/test_model_fastegnn.py and tests/test_model_zoo.py.
"""

import argparse
import random

import numpy as np
import torch
from torch import nn

from distegnn_amd.models import FastEGNN
from distegnn_amd.models.fastrf import FastRF
from distegnn_amd.utils.rotate import random_rotate


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="FastEGNN",
                    choices=["FastEGNN", "FastRF"])
    ap.add_argument("--seed", type=int, default=None)
    args = ap.parse_args()
    if args.seed is not None:
        random.seed(args.seed)
        torch.manual_seed(args.seed)

    node_cnt, edge_cnt = 10, 20
    data_batch = torch.zeros(node_cnt, dtype=torch.long)
    coordinates = torch.rand(node_cnt, 3) * 10
    velocities = torch.rand(node_cnt, 3) * 10
    node_feat = torch.rand(node_cnt, 1) * 10
    edges = torch.randint(0, node_cnt, (2, edge_cnt))
    edge_attr = torch.rand(edge_cnt, 1) * 10

    rotate_matrix = torch.tensor(random_rotate(), dtype=torch.float)
    translation = torch.randn(3) * 5
    coordinates_r = coordinates @ rotate_matrix + translation
    velocities_r = velocities @ rotate_matrix

    if args.model == "FastEGNN":
        model = FastEGNN(node_feat_nf=1, node_attr_nf=0, edge_attr_nf=1,
                         hidden_nf=64, virtual_channels=3, world_size=1,
                         act_fn=nn.SiLU(), n_layers=4, residual=True,
                         attention=False, normalize=False, tanh=False,
                         gravity=None)

        def fwd(loc, vel, lm):
            out, _ = model(node_feat.detach(), loc.detach(), vel.detach(),
                           lm.detach(), edges, data_batch,
                           edge_attr=edge_attr, node_attr=None)
            return out
    else:
        model = FastRF(edge_attr_nf=1, hidden_nf=64, virtual_channels=3,
                       world_size=1, n_layers=4)

        def fwd(loc, vel, lm):
            out, _ = model(loc.detach(), vel.detach(), lm.detach(), edges,
                           data_batch, edge_attr)
            return out

    loc_mean = coordinates.mean(dim=0, keepdim=True)
    before = fwd(coordinates, velocities, loc_mean)
    loc_mean_r = coordinates_r.mean(dim=0, keepdim=True)
    after = fwd(coordinates_r, velocities_r, loc_mean_r)

    print(f"result_before_rotate: {before}")
    print(f"result_before_rotate @ R + t: {before @ rotate_matrix + translation}")
    print(f"result_after_rotate: {after}")
    assert torch.allclose(before @ rotate_matrix + translation, after,
                          atol=1e-4)
    print("Model is SE(3) Equivariant")


if __name__ == "__main__":
    main()
