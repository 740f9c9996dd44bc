"""GraphSAGE for minibatch (block) and full-graph training.

Architecture parity with the reference's DistSAGE
(/root/reference/examples/GraphSAGE_dist/code/train_dist.py:72-94):
n_layers SAGEConv('mean') with ReLU + dropout between layers.
"""
from __future__ import annotations

from typing import List, Sequence, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..graph.graph import Block, Graph
from ..nn import SAGEConv


class GraphSAGE(nn.Module):
    def __init__(
        self,
        in_feats: int,
        n_hidden: int,
        n_classes: int,
        n_layers: int = 2,
        dropout: float = 0.5,
    ):
        super().__init__()
        self.layers = nn.ModuleList()
        if n_layers == 1:
            self.layers.append(SAGEConv(in_feats, n_classes))
        else:
            self.layers.append(SAGEConv(in_feats, n_hidden))
            for _ in range(n_layers - 2):
                self.layers.append(SAGEConv(n_hidden, n_hidden))
            self.layers.append(SAGEConv(n_hidden, n_classes))
        self.dropout = nn.Dropout(dropout)

    def forward(
        self, blocks: Union[Sequence[Block], Graph], x: torch.Tensor
    ) -> torch.Tensor:
        if isinstance(blocks, Graph):
            blocks = [blocks] * len(self.layers)
        h = x
        for i, (layer, blk) in enumerate(zip(self.layers, blocks)):
            h = layer(blk, h)
            if i != len(self.layers) - 1:
                h = F.relu(h)
                h = self.dropout(h)
        return h
