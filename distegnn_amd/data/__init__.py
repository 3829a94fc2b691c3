from .graph import Batch, Data, collate, sort_edges_by_row, build_rowptr
from .loader import DatasetWrapper, make_loaders
from .preprocess import (
    cutoff_edge,
    process_dataset_distribute,
    process_dataset_edge_cutoff,
)
from . import partition, synthetic

__all__ = [
    "Batch", "Data", "collate", "sort_edges_by_row", "build_rowptr",
    "DatasetWrapper", "make_loaders", "cutoff_edge",
    "process_dataset_distribute", "process_dataset_edge_cutoff",
    "partition", "synthetic",
]
