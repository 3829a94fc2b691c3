import torch, sys
sys.path.insert(0, "/root/repo")  # debug tool
from distegnn_amd.data.graph import collate
from distegnn_amd.data.synthetic import make_cutoff_dataset
from distegnn_amd.models import FastEGNN
from distegnn_amd.runtime.graphs import GraphedStep
from distegnn_amd.runtime.losses import mmd_loss
from distegnn_amd.utils import fix_seed

def build():
    fix_seed(0)
    model = FastEGNN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                     hidden_nf=64, virtual_channels=3, world_size=1,
                     n_layers=4).to("cuda:0")
    batches = [collate(make_cutoff_dataset("Water-3D", 15, seed=s,
                                           n_override=1500)).to("cuda:0")
               for s in (3, 5)]
    for b in batches:
        b.counts_global = b.counts
    return model, batches

def step_fn(model, use_mmd):
    def fn(data):
        with torch.autocast("cuda", dtype=torch.bfloat16):
            loc, vloc = model(data.x, data.pos, data.vel, data.loc_mean,
                              data.edge_index, data.batch,
                              edge_attr=data.edge_attr, rowptr=data.rowptr,
                              ptr=data.ptr, counts=data.counts,
                              counts_global=data.counts_global,
                              pool_chunks=(None if data.pool_chunk_begin is
                                           None else (data.pool_chunk_begin,
                                                      data.pool_chunk_end,
                                                      data.pool_seg_chunk_ptr)),
                              colptr=data.colptr, col_perm=data.col_perm)
        loss = torch.nn.functional.mse_loss(loc.float(), data.target)
        mse = loss.detach()
        if use_mmd:
            lm = mmd_loss(vloc.permute(0, 2, 1).float(), data.target,
                          data.batch, data.ptr, data.counts, 1.5, 3)
            loss = loss + 0.01 * lm
        loss.backward()
        return (mse,)
    return fn

def run(enabled, use_mmd, n_steps=10):
    model, batches = build()
    g = GraphedStep(step_fn(model, use_mmd), model.parameters(),
                    warmup_occurrences=2, enabled=enabled)
    opt = torch.optim.Adam(model.parameters(), lr=5e-4)
    out = []
    for k in range(n_steps):
        (mse,) = g(batches[k % 2])
        opt.step(); opt.zero_grad(set_to_none=False)
        out.append(round(mse.item(), 7))
    return out

for use_mmd in (False, True):
    e = run(False, use_mmd)
    gr = run(True, use_mmd)
    tag = "mmd" if use_mmd else "plain"
    ok = all(abs(a-b) < 1e-4 + 5e-3*abs(a) for a, b in zip(e, gr))
    print(tag, "MATCH" if ok else "DIVERGE")
    print("  eager :", e[-4:])
    print("  graphs:", gr[-4:])
