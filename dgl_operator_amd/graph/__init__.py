from .graph import Graph, Block, batch_graphs
from .rmat import rmat_edges, rmat_graph, ogbn_products_shape
from .partition import (
    partition_graph,
    load_partition,
    PartitionSpec,
)

__all__ = [
    "Graph",
    "Block",
    "batch_graphs",
    "rmat_edges",
    "rmat_graph",
    "ogbn_products_shape",
    "partition_graph",
    "load_partition",
    "PartitionSpec",
]
