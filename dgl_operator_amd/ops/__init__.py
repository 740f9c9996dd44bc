from . import backend
from .spmm import gspmm, spmm_raw
from .sddmm import sddmm_dot, edge_softmax, edge_softmax_csc, gat_score
from .segment import segment_reduce, mean_nodes
from .sampling import (
    sample_neighbors,
    to_block,
    sample_block_fused,
    NeighborSampler,
    CompactionWorkspace,
)
from .kge import get_score_func, kge_loss, SCORE_FUNCS
from .adagrad import sparse_adagrad_update
from .gather_mm import gather_mm, GatherView

__all__ = [
    "backend",
    "gspmm",
    "spmm_raw",
    "sddmm_dot",
    "edge_softmax",
    "edge_softmax_csc",
    "gat_score",
    "segment_reduce",
    "mean_nodes",
    "sample_neighbors",
    "to_block",
    "sample_block_fused",
    "NeighborSampler",
    "CompactionWorkspace",
    "get_score_func",
    "kge_loss",
    "SCORE_FUNCS",
    "sparse_adagrad_update",
    "gather_mm",
    "GatherView",
]
