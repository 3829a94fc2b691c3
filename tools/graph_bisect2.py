"""Bisection harness for the hipGraph training corruption (water3d):
compares eager vs graphed loss trajectories with switchable MMD variants
(--mmd-idx static indices, --no-mmd, --manual-cdist, --no-gather,
--detach-mmd, --one-batch, --probe). Kept as evidence/repro for the
side-stream allocator hazard documented in runtime/graphs.py."""
import sys, torch
sys.path.insert(0, "/root/repo")
import bench
from distegnn_amd.models import FastEGNN
from distegnn_amd.runtime.graphs import GraphedStep
from distegnn_amd.utils import fix_seed

def run(graphs_on, steps=24):
    fix_seed(43)
    nb = 1 if "--one-batch" in sys.argv else 2
    batches = bench.build_cutoff_batches("water3d", nb, 15, seed=43)
    dev = torch.device("cuda:0")
    batches = [b.to(dev) for b in batches]
    model = FastEGNN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                     hidden_nf=64, virtual_channels=3, world_size=1,
                     n_layers=4, normalize=False).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=5e-4, weight_decay=1e-12)
    core = bench.make_step_core(model, 1, 1.5, 3, torch.bfloat16)
    g = GraphedStep(core, model.parameters(), warmup_occurrences=2,
                    enabled=graphs_on, verbose=True)
    out = []
    probe = "--probe" in sys.argv and graphs_on
    for k in range(steps):
        b = batches[k % len(batches)]
        mse = bench.train_step(g, b, opt, None, k, 1, 1, dev,
                               clip=False,
                               mmd_cfg=(9 if "--mmd-idx" in sys.argv
                                        else None))
        out.append(round(mse.item(), 7))
        if probe and k >= 12:
            with torch.no_grad(), torch.autocast("cuda",
                                                 dtype=torch.bfloat16):
                lp, _ = model(
                    b.x, b.pos, b.vel, b.loc_mean, b.edge_index, b.batch,
                    edge_attr=b.edge_attr, node_attr=None,
                    rowptr=b.rowptr, ptr=b.ptr, counts=b.counts,
                    counts_global=b.counts, colptr=b.colptr,
                    col_perm=b.col_perm)
            em = torch.nn.functional.mse_loss(lp.float(), b.target).item()
            print(f"  step{k}: graphed_mse={out[-1]:.7f} eager_recompute="
                  f"{em:.7f}")
    return out

import distegnn_amd.runtime.losses as L
import bench as B

if "--no-mmd" in sys.argv:
    def fake_mmd(vloc, *a, **k):
        return vloc.sum() * 0.0
    B.mmd_loss = fake_mmd
if "--manual-cdist" in sys.argv:
    def manual_cdist(x, y, p=2):
        d2 = ((x.unsqueeze(2) - y.unsqueeze(1)) ** 2).sum(-1)
        return d2.clamp_min(1e-30).sqrt()
    L.torch_cdist = manual_cdist
    real_rbf = L.rbf_kernel_sum
    def rbf2(x, y, sigma, mask_x=None):
        d = manual_cdist(x, y)
        k = torch.exp(-d / (2.0 * sigma * sigma))
        if mask_x is not None:
            k = k * mask_x.unsqueeze(-1).to(k.dtype)
        return k.sum()
    L.rbf_kernel_sum = rbf2
    B.mmd_loss = L.mmd_loss
if "--no-gather" in sys.argv:
    real_mmd2 = L.mmd_loss
    def mmd_nogather(vloc, target, batch, ptr, counts, sigma, spc,
                     sample_idx=None, sample_valid=None):
        b, c, _ = vloc.shape
        ns = spc * c
        real = target[: b * ns].reshape(b, ns, -1)
        l_vv = L.rbf_kernel_sum(vloc, vloc, sigma)
        l_rv = L.rbf_kernel_sum(real.detach(), vloc, sigma,
                                mask_x=sample_valid)
        return l_vv / b / c / c - 2.0 * l_rv / b / ns / c
    B.mmd_loss = mmd_nogather
if "--detach-mmd" in sys.argv:
    real_mmd = B.mmd_loss
    def det_mmd(vloc, *a, **k):
        return real_mmd(vloc.detach(), *a, **k).detach() + vloc.sum() * 0.0
    B.mmd_loss = det_mmd
e = run(False)
e2 = run(False)
gr = run(True)
print("eager :", e)
print("eager2:", e2)
print("graphs:", gr)
print("e-vs-e2:", "MATCH" if all(abs(a-b) < 1e-4 + 5e-3*abs(a) for a, b in zip(e, e2)) else "DIVERGE")
print("e-vs-g :", "MATCH" if all(abs(a-b) < 1e-4 + 5e-3*abs(a) for a, b in zip(e, gr)) else "DIVERGE")
