"""End-to-end CPU training run: main.py entry with a tiny synthetic config;
checkpoint save/load round-trip."""

import json
import os
import sys

import torch
import yaml

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def tiny_config(tmp_path):
    return {
        "model": {"model_name": "FastEGNN", "normalize": True,
                  "hidden_nf": 16, "n_layers": 2, "virtual_channels": 2,
                  "node_feat_nf": 2, "node_attr_nf": 0, "edge_attr_nf": 2},
        "data": {"data_dir": str(tmp_path / "data"),
                 "dataset_name": "nbody_100", "max_samples": 100,
                 "batch_size": 2, "frame_0": 30, "frame_T": 40,
                 "accelerate_mode": "cutoff_edges", "radius": -1,
                 "cutoff_rate": 0.0, "synthetic": True,
                 "synthetic_samples": 10, "num_workers": 0},
        "train": {"learning_rate": 5e-4, "weight_decay": 1e-12,
                  "epochs": 2, "early_stop": 100, "device": "cpu",
                  "mmd": {"sigma": 1.5, "weight": 0.03, "samples": 3},
                  "accumulation_steps": 1, "warmup_epochs": 0,
                  "scheduler": "None"},
        "log": {"log_dir": str(tmp_path / "logs"), "test_interval": 2,
                "wandb": {"enable": False, "offline": True, "api_key": "",
                          "project": "", "entity": ""}},
        "seed": 43,
    }


def test_end_to_end_train_and_checkpoint(tmp_path, monkeypatch):
    import main as entry

    cfg_path = tmp_path / "cfg.yaml"
    cfg_path.write_text(yaml.safe_dump(tiny_config(tmp_path)))
    monkeypatch.setenv("WORLD_SIZE", "1")
    entry.main(["--config_path", str(cfg_path)])

    # exp dir with checkpoint + log.json exists
    logs = tmp_path / "logs"
    exps = list(logs.iterdir())
    assert len(exps) == 1
    exp = exps[0]
    assert (exp / "state_dict" / "last_model.pth").exists()
    assert (exp / "state_dict" / "best_model.pth").exists()
    log = json.loads((exp / "log" / "log.json").read_text())
    best, hist, cfg = log
    assert hist["loss_train"] and len(hist["loss_train"]) == 2
    assert "time_cost" in best

    ckpt = torch.load(exp / "state_dict" / "last_model.pth",
                      weights_only=False)
    for key in ("epoch", "model_state_dict", "optimizer_state_dict",
                "scheduler_state_dict", "loss_train", "loss_valid",
                "loss_test", "config"):
        assert key in ckpt
    assert ckpt["epoch"] == 2

    # resume from checkpoint
    entry.main(["--config_path", str(cfg_path), "--checkpoint",
                str(exp / "state_dict" / "last_model.pth")])


def test_cli_overrides(tmp_path):
    import main as entry

    cfg_path = tmp_path / "cfg.yaml"
    cfg_path.write_text(yaml.safe_dump(tiny_config(tmp_path)))
    args = entry.parse_args(["--config_path", str(cfg_path), "--lr", "0.01",
                             "--seed", "7", "--virtual_channels", "4",
                             "--batch_size", "3", "--early_stop", "5"])
    import distegnn_amd.utils as U

    cfg = U.AttrDict(yaml.safe_load(cfg_path.read_text()))
    cfg = entry.apply_overrides(cfg, args)
    assert cfg.train.learning_rate == 0.01
    assert cfg.seed == 7
    assert cfg.model.virtual_channels == 4
    assert cfg.data.batch_size == 3
    assert cfg.train.early_stop == 5


def test_exp_name_templates(tmp_path):
    import main as entry
    import distegnn_amd.utils as U

    cfg = U.AttrDict(tiny_config(tmp_path))
    name = entry.build_exp_name(cfg, 1)
    assert name.startswith("nbody_100_FastEGNN_-1_0.000_2_1_")
    cfg.data.accelerate_mode = "distribute"
    cfg.data.split_mode = "metis"
    cfg.data.outer_radius = 0.075
    cfg.data.inner_radius = 0.075
    name = entry.build_exp_name(cfg, 8)
    assert name.startswith("nbody_100_metis_FastEGNN_0.075_0.075_8_2_")


def test_run_eager_cpu_passthrough():
    from distegnn_amd.runtime.graphs import GraphedStep

    g = GraphedStep(lambda b: (b,), [], enabled=True)  # no CUDA -> disabled
    assert g.enabled is False
    assert g.run_eager(lambda: 41 + 1) == 42


def test_wandb_config_without_package(tmp_path, monkeypatch):
    """wandb.enable=True must not crash when the package is missing."""
    import yaml as _yaml

    import main as entry

    monkeypatch.setitem(sys.modules, "wandb", None)  # force ImportError
    cfg = tiny_config(tmp_path)
    cfg["log"]["wandb"]["enable"] = True
    cfg_path = tmp_path / "cfg_wb.yaml"
    cfg_path.write_text(_yaml.safe_dump(cfg))
    monkeypatch.setenv("WORLD_SIZE", "1")
    entry.main(["--config_path", str(cfg_path)])


def test_golden_loss_trajectory(tmp_path, monkeypatch):
    """Deterministic CPU golden run: guards the end-to-end math (model,
    loss weighting, MMD, optimizer wiring) against silent regressions.
    Record mode: DISTEGNN_UPDATE_GOLDEN=1 rewrites the stored values."""
    import numpy as np

    import main as entry

    cfg = tiny_config(tmp_path)
    cfg["train"]["epochs"] = 3
    cfg["log"]["test_interval"] = 3
    cfg_path = tmp_path / "cfg_gold.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    monkeypatch.setenv("WORLD_SIZE", "1")
    entry.main(["--config_path", str(cfg_path)])

    logs = tmp_path / "logs"
    exp = next(logs.iterdir())
    hist = json.loads((exp / "log" / "log.json").read_text())[1]
    got = hist["loss_train"]

    golden_path = os.path.join(os.path.dirname(__file__),
                               "golden_nbody_losses.json")
    if os.environ.get("DISTEGNN_UPDATE_GOLDEN") == "1":
        with open(golden_path, "w") as f:
            json.dump(got, f)
    golden = json.load(open(golden_path))
    assert len(got) == len(golden)
    for a, b in zip(got, golden):
        assert abs(a - b) < 1e-6 + 1e-4 * abs(b), (got, golden)


def test_torch_profile_hook(tmp_path, monkeypatch):
    """DISTEGNN_TORCH_PROFILE=<dir> dumps a chrome trace of one epoch."""
    import main as entry

    cfg = tiny_config(tmp_path)
    cfg["train"]["epochs"] = 2
    cfg_path = tmp_path / "cfg_prof.yaml"
    cfg_path.write_text(yaml.safe_dump(cfg))
    monkeypatch.setenv("WORLD_SIZE", "1")
    monkeypatch.setenv("DISTEGNN_TORCH_PROFILE", str(tmp_path / "prof"))
    entry.main(["--config_path", str(cfg_path)])
    trace = tmp_path / "prof" / "trace_rank0.json"
    assert trace.exists() and trace.stat().st_size > 1000
