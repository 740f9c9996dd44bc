"""2-layer GCN (node classification / graph classification readout).

Parity with the reference tutorials
(/root/reference/examples/node_classification/code/1_introduction.py:116-126,
examples/graph_classification/code/5_graph_classification.py:155-166).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..graph.graph import Graph
from ..nn import GraphConv
from ..ops import mean_nodes


class GCN(nn.Module):
    def __init__(self, in_feats: int, n_hidden: int, n_classes: int):
        super().__init__()
        self.conv1 = GraphConv(in_feats, n_hidden)
        self.conv2 = GraphConv(n_hidden, n_classes)

    def forward(self, g: Graph, x: torch.Tensor) -> torch.Tensor:
        h = F.relu(self.conv1(g, x))
        return self.conv2(g, h)

    def forward_graph_readout(
        self, g: Graph, x: torch.Tensor, batch_num_nodes: torch.Tensor
    ) -> torch.Tensor:
        """Graph classification: per-graph mean readout (dgl.mean_nodes)."""
        h = F.relu(self.conv1(g, x))
        h = F.relu(self.conv2(g, h))
        return mean_nodes(batch_num_nodes, h)
