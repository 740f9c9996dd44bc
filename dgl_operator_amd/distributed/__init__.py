from . import comm
from .partition_book import PartitionBook
from .dist_graph import DistGraph

__all__ = ["comm", "PartitionBook", "DistGraph"]
