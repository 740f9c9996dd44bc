from . import comm
from .partition_book import PartitionBook
from .dist_graph import DistGraph
from .kvstore import ShardedEmbedding
from .kge import DistKGEModel, KGEdgeSampler

__all__ = ["comm", "PartitionBook", "DistGraph", "ShardedEmbedding", "DistKGEModel", "KGEdgeSampler"]
