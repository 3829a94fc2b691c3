"""Deterministic seeding (parity with reference utils/seed.py:6-14).

The reference seeds python/numpy/torch/cuda and sets cudnn deterministic.
On ROCm the cudnn knobs map to MIOpen; we keep the same torch-level calls.
"""

import os
import random

import numpy as np
import torch


def fix_seed(seed: int = 927) -> None:
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    torch.cuda.manual_seed(seed)
    torch.cuda.manual_seed_all(seed)
    os.environ["PYTHONHASHSEED"] = str(seed)
    torch.backends.cudnn.deterministic = True
    torch.backends.cudnn.benchmark = False
