"""dgl.nn.functional parity: the reference GAT path imports
``from dgl.nn.functional import edge_softmax`` — same name here, backed by
the HIP segmented-softmax kernels (ops/sddmm.py)."""
from ..ops import edge_softmax, edge_softmax_csc  # noqa: F401

__all__ = ["edge_softmax", "edge_softmax_csc"]
