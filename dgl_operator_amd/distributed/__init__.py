from . import comm
from .partition_book import PartitionBook
from .dist_graph import DistGraph
from .kvstore import ShardedEmbedding
from .kge import DistKGEModel, KGEdgeSampler
from .dist_tensor import DistTensor, DistNodeDataLoader, node_split

__all__ = ["comm", "PartitionBook", "DistGraph", "ShardedEmbedding", "DistKGEModel", "KGEdgeSampler"]
