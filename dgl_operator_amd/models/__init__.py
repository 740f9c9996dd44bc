from .graphsage import GraphSAGE
from .gcn import GCN
from .gat import GATLinkPredictor, DotLinkPredictor
from .kge import KGEModel

__all__ = ["GraphSAGE", "GCN", "GATLinkPredictor", "DotLinkPredictor", "KGEModel"]
