"""1-rank forced-dist repro harness for the RCCL replay-window hazards:
--barrier injects an eager dist.barrier() between replays (corrupts),
--midsync injects a bare torch.cuda.synchronize() (corrupts),
no flags = the clean captured-collective path. See profiles/README.md."""
import os, sys
sys.path.insert(0, "/root/repo")
os.environ.setdefault("DISTEGNN_FORCE_DIST", "1")
import torch
import bench
from distegnn_amd.models import FastEGNN
from distegnn_amd.parallel import comm
from distegnn_amd.parallel.comm import GradBucket
from distegnn_amd.runtime.graphs import GraphedStep
from distegnn_amd.utils import fix_seed

rank, ws = comm.init_distributed()
dev = torch.device("cuda:0")
torch.cuda.set_device(dev)
force = comm.is_distributed()
ws_eff = 2 if force else 1
fix_seed(43)
batches = bench.build_rank_batches(0, 1, 2, 113140, 0.075, "random", 43)
model = FastEGNN(node_feat_nf=3, node_attr_nf=2, edge_attr_nf=2,
                 hidden_nf=64, virtual_channels=5, world_size=ws_eff,
                 n_layers=4, normalize=False).to(dev)
gb = GradBucket(model) if ws_eff > 1 else None
opt = torch.optim.Adam(model.parameters(), lr=5e-4, weight_decay=1e-12)
core = bench.make_step_core(model, 4, 3.0, 50, torch.bfloat16)
g = GraphedStep(core, model.parameters(), warmup_occurrences=2,
                enabled=("--off" not in sys.argv), verbose=True)
steps = int(sys.argv[sys.argv.index("--steps") + 1])     if "--steps" in sys.argv else 14
nosync = "--nosync" in sys.argv
mse = None
for k in range(steps):
    if "--barrier" in sys.argv and k == 8:
        comm.barrier()
        torch.cuda.synchronize()
        print("mid-run barrier done", flush=True)
    if "--midsync" in sys.argv and k == 8:
        torch.cuda.synchronize()
        print("mid-run synchronize done", flush=True)
    mse = bench.train_step(g, batches[k % 2], opt, gb, k, 4, ws_eff, dev,
                           clip=True, mmd_cfg=250)
    if not nosync:
        print(f"step {k}: mse={mse.item():.8f}", flush=True)
torch.cuda.synchronize()
print(f"final mse={mse.item():.8f}", flush=True)
comm.barrier(); comm.destroy()
