"""Training entry point — CLI/config parity with the reference main.py.

Run single-GPU:   python main.py --config_path config/nbody_fastegnn.yaml
Run multi-GPU:    python -m torch.distributed.run --nnodes=1 \
                    --nproc-per-node 8 --master-addr 127.0.0.1 main.py \
                    --config_path config/largefluid_distegnn.yaml

Parity surface (reference main.py:95-229): the same YAML schema and the
same CLI overrides (--wandb --lr --seed --model_name --batch_size
--split_mode --early_stop --checkpoint --cutoff_rate --outer_radius
--inner_radius --virtual_channels), the same exp-name templates
(:147-157), checkpoint load behavior (:208-220), and the parameter
consistency check (:40-55).

MI355X differences: RCCL process group from torchrun env (one process per
GPU); our flat GradBucket replaces DDP (identical math, see
parallel/comm.py); data loaders use our own Batch collation with CSR
metadata.
"""

import argparse
import os
import time

import torch
import yaml

from distegnn_amd.utils import AttrDict, fix_seed
from distegnn_amd.data import (DatasetWrapper, make_loaders,
                               process_dataset_distribute,
                               process_dataset_edge_cutoff)
from distegnn_amd.models import get_model
from distegnn_amd.parallel import comm
from distegnn_amd.parallel.comm import GradBucket
from distegnn_amd.runtime import load_state_dict_compat, train


def count_parameters(model):
    return sum(p.numel() for p in model.parameters() if p.requires_grad)


def parse_args(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--config_path", type=str, required=True,
                        help="path to config yaml file")
    parser.add_argument("--wandb", action="store_true")
    parser.add_argument("--lr", type=float, default=None)
    parser.add_argument("--seed", type=int, default=None)
    parser.add_argument("--model_name", type=str, default=None)
    parser.add_argument("--batch_size", type=int, default=None)
    parser.add_argument("--split_mode", type=str, default=None)
    parser.add_argument("--early_stop", type=int, default=None)
    parser.add_argument("--checkpoint", type=str, default=None)
    parser.add_argument("--cutoff_rate", type=float, default=None)
    parser.add_argument("--outer_radius", type=float, default=None)
    parser.add_argument("--inner_radius", type=float, default=None)
    parser.add_argument("--virtual_channels", type=int, default=None)
    parser.add_argument("--dtype", type=str, default=None,
                        choices=["fp32", "bf16"],
                        help="compute dtype for MLP GEMMs (extension)")
    return parser.parse_args(argv)


def apply_overrides(config: AttrDict, args) -> AttrDict:
    if args.wandb:
        config.log.wandb.offline = False
    if args.seed is not None:
        config.seed = args.seed
    if args.lr is not None:
        config.train.learning_rate = args.lr
    if args.model_name is not None:
        config.model.model_name = args.model_name
    if args.batch_size is not None:
        config.data.batch_size = args.batch_size
    if args.split_mode is not None:
        config.data.split_mode = args.split_mode
    if args.early_stop is not None:
        config.train.early_stop = args.early_stop
    if args.checkpoint is not None:
        config.model.checkpoint = args.checkpoint
    if args.cutoff_rate is not None:
        config.data.cutoff_rate = args.cutoff_rate
    if args.outer_radius is not None:
        config.data.outer_radius = args.outer_radius
    if args.inner_radius is not None:
        config.data.inner_radius = args.inner_radius
    if args.virtual_channels is not None:
        config.model.virtual_channels = args.virtual_channels
    return config


def build_exp_name(config, world_size: int) -> str:
    """Experiment-name templates (reference main.py:147-157)."""
    suffix = time.strftime("%Y-%m-%d_%H-%M-%S", time.localtime())
    c, d, m = config, config.data, config.model
    if d.accelerate_mode == "distribute":
        if m.model_name.startswith("Fast"):
            return (f"{d.dataset_name}_{d.split_mode}_{m.model_name}_"
                    f"{d.outer_radius}_{d.inner_radius}_{world_size}_"
                    f"{m.virtual_channels}_{suffix}")
        return (f"{d.dataset_name}_{d.split_mode}_{m.model_name}_"
                f"{d.outer_radius}_{d.inner_radius}_{world_size}_{suffix}")
    if m.model_name.startswith("Fast"):
        return (f"{d.dataset_name}_{m.model_name}_{d.radius}_"
                f"{d.cutoff_rate:.3f}_{m.virtual_channels}_{world_size}_"
                f"{suffix}")
    return (f"{d.dataset_name}_{m.model_name}_{d.radius}_"
            f"{d.cutoff_rate:.3f}_{world_size}_{suffix}")


def main(argv=None):
    args = parse_args(argv)
    with open(args.config_path) as f:
        config = AttrDict(yaml.safe_load(f))
    config = apply_overrides(config, args)

    # One process per GPU under torchrun; WORLD_SIZE=1 otherwise.
    local_rank, world_size = comm.init_distributed()
    config.data.world_size = world_size
    if local_rank == 0:
        print(f"Use {world_size} GPUs!")

    config.log.exp_name = build_exp_name(config, world_size)

    fix_seed(config.seed)

    if config.data.accelerate_mode == "distribute":
        processed = process_dataset_distribute(local_rank, world_size,
                                               config.data)
    elif config.data.accelerate_mode == "cutoff_edges":
        assert world_size == 1
        processed = process_dataset_edge_cutoff(config.data)
    else:
        raise NotImplementedError(
            f"accelerate_mode {config.data.accelerate_mode} not implemented")
    comm.barrier()

    fix_seed(config.seed)

    ds_train, ds_valid, ds_test = (DatasetWrapper(p) for p in processed)
    print(f"Device [{local_rank}]: Data get!")
    loader_train, loader_valid, loader_test = make_loaders(
        ds_train, ds_valid, ds_test, batch_size=config.data.batch_size,
        seed=config.seed, num_workers=config.data.get("num_workers", 4))

    device = (torch.device(f"cuda:{local_rank}")
              if torch.cuda.is_available() else torch.device("cpu"))
    model = get_model(config.model, world_size, config.data.dataset_name)
    model = model.to(device)
    model_name = config.model.model_name

    grad_bucket = None
    if world_size > 1:
        grad_bucket = GradBucket(model)
        grad_bucket.broadcast_parameters()

    optimizer = torch.optim.Adam(model.parameters(),
                                 lr=config.train.learning_rate,
                                 weight_decay=config.train.weight_decay)
    if config.train.scheduler == "cosine":
        scheduler = torch.optim.lr_scheduler.CosineAnnealingLR(
            optimizer,
            T_max=config.train.epochs * len(loader_train)
            // config.train.accumulation_steps,
            eta_min=1e-8)
    else:
        scheduler = None

    if local_rank == 0:
        print(model)
        print(count_parameters(model))

    start_epoch = 0
    if args.checkpoint is not None:
        ckpt = torch.load(args.checkpoint, map_location=device,
                          weights_only=False)
        start_epoch = ckpt["epoch"]
        load_state_dict_compat(model, ckpt["model_state_dict"])
        optimizer.load_state_dict(ckpt["optimizer_state_dict"])
        if scheduler is not None and ckpt.get("scheduler_state_dict"):
            scheduler.load_state_dict(ckpt["scheduler_state_dict"])
        print(f"GPU[{local_rank}]: Checkpoint loaded!")
        comm.barrier()

    if world_size > 1:
        assert comm.check_model_parameters(model), \
            "model parameters inconsistent across ranks"
        if local_rank == 0:
            print(f"Rank {local_rank}: Model parameters consistency check "
                  f"passed!")
        comm.barrier()

    autocast_dtype = torch.bfloat16 if args.dtype == "bf16" else None
    train(local_rank, model, model_name, optimizer, scheduler, loader_train,
          loader_valid, loader_test, config.train, config.log, config,
          start_epoch, device=device, grad_bucket=grad_bucket,
          autocast_dtype=autocast_dtype)


if __name__ == "__main__":
    main()
