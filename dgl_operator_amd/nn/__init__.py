from . import functional  # noqa: F401  (dgl.nn.functional path parity)
from .conv import SAGEConv, GraphConv, GATConv

__all__ = ["SAGEConv", "GraphConv", "GATConv", "functional"]
