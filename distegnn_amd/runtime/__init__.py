from .trainer import (
    train,
    train_single_epoch,
    model_forward,
    state_dict_for_save,
    load_state_dict_compat,
)
from .losses import mmd_loss, rbf_kernel_sum, sample_nodes_per_graph

__all__ = [
    "train", "train_single_epoch", "model_forward", "state_dict_for_save",
    "load_state_dict_compat", "mmd_loss", "rbf_kernel_sum",
    "sample_nodes_per_graph",
]
