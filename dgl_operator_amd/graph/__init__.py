from .graph import (
    EID,
    NID,
    Block,
    Graph,
    batch_graphs,
    batch_num_edges,
    to_bidirected,
    unbatch,
)
from .rmat import rmat_edges, rmat_graph, ogbn_products_shape
from .partition import (
    partition_graph,
    load_partition,
    PartitionSpec,
)

__all__ = [
    "Graph",
    "Block",
    "NID",
    "EID",
    "batch_graphs",
    "batch_num_edges",
    "to_bidirected",
    "unbatch",
    "rmat_edges",
    "rmat_graph",
    "ogbn_products_shape",
    "partition_graph",
    "load_partition",
    "PartitionSpec",
]
