from .conv import SAGEConv, GraphConv, GATConv

__all__ = ["SAGEConv", "GraphConv", "GATConv"]
