import sys, torch
sys.path.insert(0, "/root/repo")
import bench
from distegnn_amd.models import FastEGNN
from distegnn_amd.runtime.graphs import GraphedStep
from distegnn_amd.utils import fix_seed

def run(graphs_on, steps=24):
    fix_seed(43)
    batches = bench.build_cutoff_batches("water3d", 2, 15, seed=43)
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
    for k in range(steps):
        mse = bench.train_step(g, batches[k % 2], opt, None, k, 1, 1, dev,
                               clip=False)
        out.append(round(mse.item(), 7))
    return out

import distegnn_amd.runtime.losses as L
import bench as B

if "--no-mmd" in sys.argv:
    def fake_mmd(vloc, *a, **k):
        return vloc.sum() * 0.0
    B.mmd_loss = fake_mmd
e = run(False)
gr = run(True)
print("eager :", e)
print("graphs:", gr)
print("MATCH" if all(abs(a-b) < 1e-4 + 5e-3*abs(a) for a,b in zip(e,gr)) else "DIVERGE")
