"""CPU tests for the non-FastEGNN model families: shapes, equivariance,
trainer dispatch."""

import numpy as np
import pytest
import torch

from distegnn_amd.data.graph import collate
from distegnn_amd.data.synthetic import make_cutoff_dataset
from distegnn_amd.models import get_model
from distegnn_amd.runtime.trainer import model_forward
from distegnn_amd.utils import AttrDict, fix_seed, rotate


def batch(n_graphs=2, seed=0):
    return collate(make_cutoff_dataset("nbody_100", n_graphs, seed=seed))


def cfg(name, **kw):
    base = dict(model_name=name, normalize=False, hidden_nf=16, n_layers=2,
                virtual_channels=2, node_feat_nf=2, node_attr_nf=0,
                edge_attr_nf=2)
    base.update(kw)
    return AttrDict(base)


@pytest.mark.parametrize("name", ["FastRF", "EGNN", "RF", "Linear",
                                  "SchNet", "FastSchNet"])
def test_forward_shapes_via_dispatch(name):
    fix_seed(0)
    b = batch()
    model = get_model(cfg(name), 1, "nbody_100")
    model_name = "RF_vel" if name == "RF" else name
    pred, vloc = model_forward(model, model_name, b, torch.device("cpu"))
    assert pred.shape == (b.num_nodes, 3)
    if name.startswith("Fast"):
        assert vloc.shape == (b.num_graphs, 3, 2)
    assert torch.isfinite(pred).all()


@pytest.mark.parametrize("name", ["FastRF", "EGNN", "RF", "Linear"])
def test_se3_equivariance_zoo(name):
    """f(xR + t) == f(x)R + t for the coordinate-output models."""
    fix_seed(1)
    b = batch(seed=2)
    model = get_model(cfg(name), 1, "nbody_100").double()
    R = torch.tensor(rotate.random_rotate(np.random.default_rng(3)))
    t = torch.randn(3, dtype=torch.float64)

    def run(pos, vel, lm):
        if name == "FastRF":
            out, _ = model(pos, vel, lm, b.edge_index, b.batch,
                           b.edge_attr.double(), rowptr=b.rowptr, ptr=b.ptr,
                           counts=b.counts.double())
            return out
        if name == "EGNN":
            out, _, _ = model(pos, b.x.double(), b.edge_index,
                              b.edge_attr.double(), vel)
            return out
        if name == "RF":
            return model(vel.norm(dim=-1, keepdim=True), pos, b.edge_index,
                         vel, b.edge_attr.double())
        return model(pos, vel)

    o1 = run(b.pos.double(), b.vel.double(), b.loc_mean.double())
    o2 = run(b.pos.double() @ R + t, b.vel.double() @ R,
             b.loc_mean.double() @ R + t)
    assert torch.allclose(o1 @ R + t, o2, atol=1e-8), \
        (o1 @ R + t - o2).abs().max()


def test_fastrf_backward():
    fix_seed(2)
    b = batch(seed=4)
    model = get_model(cfg("FastRF"), 1, "nbody_100")
    pred, vloc = model(b.pos, b.vel, b.loc_mean, b.edge_index, b.batch,
                       b.edge_attr, rowptr=b.rowptr, ptr=b.ptr,
                       counts=b.counts)
    (pred.pow(2).mean() + vloc.pow(2).mean()).backward()
    grads = [p.grad for p in model.parameters() if p.grad is not None]
    assert grads and all(torch.isfinite(g).all() for g in grads)


def test_schnet_coordinate_translation_equivariance():
    """SchNet's coordinate update is translation-equivariant."""
    fix_seed(3)
    b = batch(1, seed=5)
    model = get_model(cfg("SchNet", hidden_nf=16), 1, "nbody_100").double()
    t = torch.randn(3, dtype=torch.float64)
    p1 = model(z=b.x.double(), pos=b.pos.double(),
               edge_index=b.edge_index, batch=b.batch)
    p2 = model(z=b.x.double(), pos=b.pos.double() + t,
               edge_index=b.edge_index, batch=b.batch)
    assert torch.allclose(p1 + t, p2, atol=1e-8)


def test_fastschnet_state_dict_parity_names():
    m = get_model(cfg("FastSchNet", hidden_nf=16), 1, "nbody_100")
    sd = m.state_dict()
    assert sd["virtual_node_feat"].shape == (1, 16, 2)
    assert sd["W"].shape == (1, 2, 3)
    assert "gcl_0.schnet_layer.interactions.0.conv.lin1.weight" in sd
    assert "gcl_0.coord_mlp_r_virtual.0.weight" in sd


@pytest.mark.parametrize("name", ["TFN", "FastTFN"])
def test_tfn_se3_equivariance(name):
    """The re-owned TFN stack (numeric Wigner-D basis, DGL-free) is SE(3)
    equivariant end to end."""
    fix_seed(5)
    b = batch(1, seed=6)
    model = get_model(cfg(name, hidden_nf=8, node_attr_nf=1), 1,
                      "nbody_100").double()
    R = torch.tensor(rotate.random_rotate(np.random.default_rng(7)))
    t = torch.randn(3, dtype=torch.float64)

    def run(pos, vel, lm):
        if name == "TFN":
            return model(pos, vel, b.attr.double(), b.edge_index)
        out, _ = model(b.x.double(), pos, vel, lm, b.edge_index, b.batch,
                       b.attr.double(), edge_attr=b.edge_attr.double(),
                       node_attr=b.attr.double(), rowptr=b.rowptr,
                       ptr=b.ptr, counts=b.counts.double())
        return out

    o1 = run(b.pos.double(), b.vel.double(), b.loc_mean.double())
    o2 = run(b.pos.double() @ R + t, b.vel.double() @ R,
             b.loc_mean.double() @ R + t)
    err = (o1 @ R + t - o2).abs().max()
    assert err < 1e-6, err


def test_tfn_backward():
    fix_seed(6)
    b = batch(1, seed=8)
    model = get_model(cfg("FastTFN", hidden_nf=8, node_attr_nf=1), 1,
                      "nbody_100")
    pred, vloc = model(b.x, b.pos, b.vel, b.loc_mean, b.edge_index, b.batch,
                       b.attr, edge_attr=b.edge_attr, node_attr=b.attr,
                       rowptr=b.rowptr, ptr=b.ptr, counts=b.counts)
    (pred.pow(2).mean() + vloc.pow(2).mean()).backward()
    assert any(p.grad is not None for p in model.parameters())


def test_se3_transformer_forward():
    from distegnn_amd.models.tfn import OurDynamics

    fix_seed(7)
    b = batch(1, seed=9)
    m = OurDynamics(nf=4, n_layers=2, model="se3_transformer",
                    num_degrees=2, div=1)
    out = m(b.pos, b.vel, b.attr, b.edge_index)
    assert out.shape == (b.num_nodes, 3)
    assert torch.isfinite(out).all()


def test_eghn_forward_and_equivariance():
    from distegnn_amd.models.baselines import EGHN

    fix_seed(8)
    b = batch(2, seed=10)
    m = EGHN(in_node_nf=2, in_edge_nf=2, hidden_nf=8, n_cluster=3,
             layer_per_block=1, layer_pooling=1).double()
    n_node = 100

    def run(pos, vel):
        x_out, _, _ = m(pos, b.x.double(), b.edge_index,
                        b.edge_attr.double(), b.edge_index,
                        b.edge_attr.double(), n_node, v=vel)
        return x_out

    o1 = run(b.pos.double(), b.vel.double())
    assert o1.shape == (b.num_nodes, 3)
    R = torch.tensor(rotate.random_rotate(np.random.default_rng(11)))
    o2 = run(b.pos.double() @ R, b.vel.double() @ R)
    # EGHN is O(3)-equivariant around the per-graph mean (translation is
    # removed internally via x_mean); test rotation equivariance
    assert torch.allclose(o1 @ R, o2, atol=1e-7), (o1 @ R - o2).abs().max()
