"""Graphed multi-GPU trainer path on a 1-rank RCCL group.

The trainer's captured-collective design (in-graph counts reduce, captured
grad sync, CapturedAllReduce epoch reduces, fp32 eval routed to the capture
communicator, pre-flight integrity gate) cannot run at true world_size > 1
on the 1-GPU CI box — this test exercises the EXACT code path on a 1-rank
RCCL group (every collective an identity), the same validation strategy the
round-1 bench used (profiles/README.md "RCCL capture"). The driver's 8-GPU
scaling run is the real multi-rank execution; the in-trainer integrity gate
makes that run self-checking.
"""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires GPU", allow_module_level=True)

import torch.distributed as dist

from distegnn_amd.data.loader import DatasetWrapper, make_loaders
from distegnn_amd.data.partition import split_large_graph_random
from distegnn_amd.data.synthetic import make_cloud_sample
from distegnn_amd.models import FastEGNN
from distegnn_amd.parallel import comm
from distegnn_amd.parallel.comm import GradBucket
from distegnn_amd.runtime import trainer
from distegnn_amd.utils import AttrDict, fix_seed


def _init_one_rank_group(tmp_path):
    import distegnn_amd.parallel.comm as C

    if not dist.is_initialized():
        dist.init_process_group("nccl", rank=0, world_size=1,
                                init_method=f"file://{tmp_path}/pg_init")
    if C._GRAPH_PG is None:
        C._GRAPH_PG = dist.new_group(backend="nccl")
        warm = torch.ones(1, device="cuda:0")
        dist.all_reduce(warm, group=C._GRAPH_PG)
        torch.cuda.synchronize()


def _partition_samples(n_samples, seed=11):
    rng = torch.Generator().manual_seed(seed)
    out = []
    for _ in range(n_samples):
        s = make_cloud_sample("Water-3D", rng, n_override=2000)
        parts = split_large_graph_random(
            s["pos"], s["x"], s["target"], s["vel"], s["attr"], 0.08, 2,
            generator=rng)
        out.append(parts[0])        # "rank 0"'s partition
    return out


def _config(tmp_path):
    return AttrDict({
        "model": {"model_name": "FastEGNN", "normalize": False,
                  "hidden_nf": 32, "n_layers": 2, "virtual_channels": 3,
                  "node_feat_nf": 2, "node_attr_nf": 0, "edge_attr_nf": 2},
        "data": {"dataset_name": "Water-3D", "world_size": 2,
                 "batch_size": 1},
        "train": {"learning_rate": 5e-4, "weight_decay": 1e-12,
                  "epochs": 2, "early_stop": 100,
                  "mmd": {"sigma": 1.5, "weight": 0.03, "samples": 3},
                  "accumulation_steps": 2, "scheduler": "None",
                  "hip_graphs": "on", "graph_integrity_check": "on"},
        "log": {"log_dir": str(tmp_path / "logs"), "exp_name": "distgraph",
                "test_interval": 1,
                "wandb": {"enable": False}},
        "seed": 43,
    })


@pytest.mark.timeout(600)
def test_graphed_multigpu_trainer_on_one_rank_group(tmp_path, capsys):
    _init_one_rank_group(tmp_path)
    fix_seed(43)
    cfg = _config(tmp_path)
    device = torch.device("cuda:0")
    model = FastEGNN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                     hidden_nf=32, virtual_channels=3, world_size=2,
                     n_layers=2, normalize=False).to(device)
    grad_bucket = GradBucket(model)
    optimizer = torch.optim.Adam(model.parameters(), lr=5e-4,
                                 weight_decay=1e-12)
    train_ds = DatasetWrapper(_partition_samples(6))
    eval_ds = DatasetWrapper(_partition_samples(2, seed=12))
    lt, lv, ltst = make_loaders(train_ds, eval_ds, eval_ds, 1, seed=43)

    try:
        best, hist = trainer.train(
            0, model, "FastEGNN", optimizer, None, lt, lv, ltst,
            cfg.train, cfg.log, cfg, start_epoch=0, device=device,
            grad_bucket=grad_bucket, autocast_dtype=torch.bfloat16,
            progress=False)
    finally:
        out = capsys.readouterr().out
        print(out)

    # the pre-flight gate ran and PASSED (capture survived the eager-vs-
    # replay trajectory comparison on the 1-rank RCCL group)
    assert "[integrity]" in out
    assert "matches eager" in out
    assert "DISABLED" not in out
    assert hist["loss_train"] and all(math.isfinite(v)
                                      for v in hist["loss_train"])
    assert math.isfinite(best["loss_valid"])


@pytest.mark.timeout(600)
def test_graphed_eval_epochs_match_eager(tmp_path):
    """Regression for the eval-epoch replay hazard: eval forwards and
    checkpoint saves device-synchronize between replays, which garbles
    existing hipGraph execs on this stack (a bare torch.cuda.synchronize()
    there NaNs training one epoch later — bisected via
    DISTEGNN_DBG_EVAL_MODE). The trainer must invalidate + recapture after
    every eval epoch; with that, a graphed run with interleaved evals
    follows the eager trajectory."""
    device = torch.device("cuda:0")

    def _run(hip_graphs):
        # trainer.train tears the process group down at the end of every
        # world_size>1 run — re-init per arm (fresh file store each time)
        d = tmp_path / f"pg_{hip_graphs}"
        d.mkdir(exist_ok=True)
        _init_one_rank_group(d)
        fix_seed(43)
        cfg = _config(tmp_path)
        cfg.train.epochs = 6
        cfg.train.hip_graphs = hip_graphs
        # gate off: it consumes loader batches pre-training, which would
        # shift the graphed arm's epoch-1 batch order vs the eager arm
        # (gate coverage lives in the test above)
        cfg.train.graph_integrity_check = "off"
        cfg.log.test_interval = 2          # evals+checkpoints at 2, 4, 6
        cfg.log.exp_name = f"evalpar_{hip_graphs}"
        model = FastEGNN(node_feat_nf=2, node_attr_nf=0, edge_attr_nf=2,
                         hidden_nf=32, virtual_channels=3, world_size=2,
                         n_layers=2, normalize=False).to(device)
        grad_bucket = GradBucket(model)
        optimizer = torch.optim.Adam(model.parameters(), lr=5e-4,
                                     weight_decay=1e-12)
        train_ds = DatasetWrapper(_partition_samples(6))
        eval_ds = DatasetWrapper(_partition_samples(2, seed=12))
        lt, lv, ltst = make_loaders(train_ds, eval_ds, eval_ds, 1, seed=43)
        _, hist = trainer.train(
            0, model, "FastEGNN", optimizer, None, lt, lv, ltst,
            cfg.train, cfg.log, cfg, start_epoch=0, device=device,
            grad_bucket=grad_bucket, autocast_dtype=torch.bfloat16,
            progress=False)
        return hist["loss_train"]

    eager = _run("off")
    graphed = _run("on")
    assert all(math.isfinite(v) for v in graphed), graphed
    assert len(eager) == len(graphed) == 6
    for e, g in zip(eager, graphed):
        assert abs(e - g) <= 5e-3 * max(abs(e), 1e-8), (eager, graphed)
