"""GPU tests for the fused virtual-edge block kernel vs the eager fp32
composition."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires GPU", allow_module_level=True)

from distegnn_amd import ops
from distegnn_amd.data.graph import collate
from distegnn_amd.data.synthetic import make_cutoff_dataset


def dev():
    return torch.device("cuda:0")


def setup(n=2000, b_graphs=2, c=5, seed=0):
    bt = collate(make_cutoff_dataset("Water-3D", b_graphs, seed=seed,
                                     n_override=n)).to(dev())
    g = torch.Generator().manual_seed(seed)

    def t(*shape, s=0.1):
        return (torch.randn(*shape, generator=g) * s).to(dev())

    k = 129 + c
    params = dict(
        w1=t(64, k), b1=t(64), w2=t(64, 64), b2=t(64),
        wxv=t(64, 64), bxv=t(64), wxvv=t(64),
        wX=t(64, 64), bX=t(64), wXv=t(64))
    h = t(bt.num_nodes, 64, s=0.5)
    vcoord = bt.loc_mean.unsqueeze(1).expand(-1, c, 3).contiguous() \
        + t(bt.num_graphs, c, 3, s=0.05)
    vfeat = t(bt.num_graphs, c, 64, s=0.5)
    gram = t(bt.num_graphs, c, c, s=0.3)
    return bt, h, vcoord, vfeat, gram, params


def run_block(bt, h, vcoord, vfeat, gram, params, grad=False,
              force_eager=False):
    if force_eager:
        os.environ["DISTEGNN_DISABLE_FUSED"] = "1"
    else:
        os.environ.pop("DISTEGNN_DISABLE_FUSED", None)
    leaves = {k: v.detach().clone().requires_grad_(grad)
              for k, v in params.items()}
    hh = h.detach().clone().bfloat16().requires_grad_(grad)
    cc = bt.pos.detach().clone().requires_grad_(grad)
    vc = vcoord.detach().clone().requires_grad_(grad)
    vf = vfeat.detach().clone().requires_grad_(grad)
    gr = gram.detach().clone().requires_grad_(grad)
    chunks = None
    if bt.pool_chunk_begin is not None:
        chunks = (bt.pool_chunk_begin, bt.pool_chunk_end,
                  bt.pool_seg_chunk_ptr)
    vmsg, tv, tx = ops.fused_virtual_block(
        hh, cc, vc, vf, gr, bt.batch, bt.ptr, chunks,
        leaves["w1"], leaves["b1"], leaves["w2"], leaves["b2"],
        leaves["wxv"], leaves["bxv"], leaves["wxvv"],
        leaves["wX"], leaves["bX"], leaves["wXv"])
    os.environ.pop("DISTEGNN_DISABLE_FUSED", None)
    if not grad:
        return vmsg.float(), tv, tx
    loss = vmsg.float().pow(2).sum() + tv.pow(2).sum() + tx.pow(2).sum()
    loss.backward()
    grads = {k: v.grad for k, v in leaves.items()}
    grads.update(h=hh.grad.float(), coord=cc.grad, vcoord=vc.grad,
                 vfeat=vf.grad.float(), gram=gr.grad)
    return (vmsg.detach().float(), tv.detach(), tx.detach()), grads


def test_fused_virtual_forward_matches_eager():
    args = setup()
    vm_f, tv_f, tx_f = run_block(*args, grad=False, force_eager=False)
    vm_e, tv_e, tx_e = run_block(*args, grad=False, force_eager=True)
    assert torch.allclose(vm_f, vm_e, atol=0.05, rtol=0.05), \
        (vm_f - vm_e).abs().max()
    for a, b in ((tv_f, tv_e), (tx_f, tx_e)):
        rel = (a - b).norm() / b.norm().clamp(min=1e-9)
        assert rel < 0.05, rel.item()


def test_fused_virtual_backward_matches_eager():
    args = setup(n=1500, c=3, seed=1)
    out_f, g_f = run_block(*args, grad=True, force_eager=False)
    out_e, g_e = run_block(*args, grad=True, force_eager=True)
    for k in g_e:
        a, b = g_f[k], g_e[k]
        denom = b.float().norm().clamp(min=1e-6)
        rel = (a.float() - b.float()).norm() / denom
        assert rel < 0.08, (k, rel.item())


def test_fused_virtual_in_model_matches_cpu():
    """Full FastEGNN forward with the fused virtual block (GPU bf16) vs the
    CPU fp32 reference path."""
    from distegnn_amd.models import FastEGNN
    from distegnn_amd.utils import fix_seed

    fix_seed(0)
    bt_cpu = collate(make_cutoff_dataset("Water-3D", 2, seed=3,
                                         n_override=1200))
    m = FastEGNN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                 hidden_nf=64, virtual_channels=3, world_size=1, n_layers=2)
    loc_c, vloc_c = m(bt_cpu.x, bt_cpu.pos, bt_cpu.vel, bt_cpu.loc_mean,
                      bt_cpu.edge_index, bt_cpu.batch,
                      edge_attr=bt_cpu.edge_attr, rowptr=bt_cpu.rowptr,
                      ptr=bt_cpu.ptr, counts=bt_cpu.counts)
    mg = m.to(dev())
    bt = collate(make_cutoff_dataset("Water-3D", 2, seed=3,
                                     n_override=1200)).to(dev())
    with torch.autocast("cuda", dtype=torch.bfloat16):
        loc_g, vloc_g = mg(bt.x, bt.pos, bt.vel, bt.loc_mean, bt.edge_index,
                           bt.batch, edge_attr=bt.edge_attr,
                           rowptr=bt.rowptr, ptr=bt.ptr, counts=bt.counts,
                           colptr=bt.colptr, col_perm=bt.col_perm)
    rel = (loc_g.cpu() - loc_c).norm() / loc_c.norm()
    assert rel < 0.02, rel.item()
    relv = (vloc_g.cpu() - vloc_c).norm() / vloc_c.norm()
    assert relv < 0.02, relv.item()
