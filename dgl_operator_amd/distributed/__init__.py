from . import comm
from .comm import initialize
from .partition_book import PartitionBook
from .dist_graph import DistGraph
from .kvstore import ShardedEmbedding
from .kge import DistKGEModel, KGEdgeSampler
from .dist_tensor import DistTensor, DistNodeDataLoader, node_split

__all__ = [
    "comm",
    "initialize",
    "PartitionBook",
    "DistGraph",
    "ShardedEmbedding",
    "DistKGEModel",
    "KGEdgeSampler",
    "DistTensor",
    "DistNodeDataLoader",
    "node_split",
]
