"""Fiber structures for SE(3)-equivariant feature dicts (reference
se3_dynamics/equivariant_attention/fibers.py:13-152)."""

from __future__ import annotations

import copy
from typing import List, Tuple

import numpy as np
import torch


class Fiber:
    """Degree/multiplicity structure of an SE(3) feature dict."""

    def __init__(self, num_degrees: int = None, num_channels: int = None,
                 structure: List[Tuple[int, int]] = None, dictionary=None):
        if structure:
            self.structure = structure
        elif dictionary:
            self.structure = [(dictionary[o], o)
                              for o in sorted(dictionary.keys())]
        else:
            self.structure = [(num_channels, i) for i in range(num_degrees)]
        self.multiplicities, self.degrees = zip(*self.structure)
        self.max_degree = max(self.degrees)
        self.min_degree = min(self.degrees)
        self.structure_dict = {k: v for v, k in self.structure}
        self.dict = self.structure_dict
        self.n_features = int(np.sum([m * (2 * d + 1)
                                      for m, d in self.structure]))
        self.feature_indices = {}
        idx = 0
        for (m, d) in self.structure:
            length = m * (2 * d + 1)
            self.feature_indices[d] = (idx, idx + length)
            idx += length

    def copy_me(self, multiplicity: int = None):
        s = copy.deepcopy(self.structure)
        if multiplicity is not None:
            s = [(multiplicity, o) for _, o in s]
        return Fiber(structure=s)

    @staticmethod
    def combine(f1: "Fiber", f2: "Fiber") -> "Fiber":
        d = copy.deepcopy(f1.structure_dict)
        for k, m in f2.structure_dict.items():
            d[k] = d.get(k, 0) + m
        return Fiber(structure=[(d[k], k) for k in sorted(d)])

    @staticmethod
    def combine_max(f1: "Fiber", f2: "Fiber") -> "Fiber":
        d = copy.deepcopy(f1.structure_dict)
        for k, m in f2.structure_dict.items():
            if k in d:
                d[k] = max(m, d[k])
        return Fiber(structure=[(d[k], k) for k in sorted(d)])

    def __repr__(self):
        return f"{self.structure}"


def fiber2tensor(F, structure: Fiber, squeeze=False):
    if squeeze:
        parts = [F[f"{i}"].reshape(*F[f"{i}"].shape[:-2], -1)
                 for i in structure.degrees]
        return torch.cat(parts, -1)
    parts = [F[f"{i}"].reshape(*F[f"{i}"].shape[:-2], -1, 1)
             for i in structure.degrees]
    return torch.cat(parts, -2)


def fiber2head(F, h: int, structure: Fiber, squeeze=False):
    if squeeze:
        parts = [F[f"{i}"].reshape(*F[f"{i}"].shape[:-2], h, -1)
                 for i in structure.degrees]
        return torch.cat(parts, -1)
    parts = [F[f"{i}"].reshape(*F[f"{i}"].shape[:-2], h, -1, 1)
             for i in structure.degrees]
    return torch.cat(parts, -2)
