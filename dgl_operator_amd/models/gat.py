"""Link prediction heads + GAT encoder.

Parity with the reference link-prediction example
(/root/reference/examples/link_predict/code/4_link_predict.py): a GNN encoder
produces node embeddings; the predictor scores candidate edges with
u_dot_v (g-SDDMM); GATConv supplies the SDDMM + edge-softmax attention path
(BASELINE config #5).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..graph.graph import Graph
from ..nn import GATConv
from ..ops import sddmm_dot


class DotLinkPredictor(nn.Module):
    """score[e=(u,v)] = <h_u, h_v>  (fn.u_dot_v of the reference)."""

    def forward(self, g: Graph, h: torch.Tensor) -> torch.Tensor:
        return sddmm_dot(g, h, h)


class GATLinkPredictor(nn.Module):
    """GAT encoder + dot predictor over positive/negative edge graphs."""

    def __init__(self, in_feats: int, n_hidden: int, num_heads: int = 4,
                 n_layers: int = 2):
        super().__init__()
        self.layers = nn.ModuleList()
        dims_in = in_feats
        for i in range(n_layers):
            last = i == n_layers - 1
            heads = 1 if last else num_heads
            self.layers.append(GATConv(dims_in, n_hidden, num_heads=heads))
            dims_in = n_hidden * heads
        self.pred = DotLinkPredictor()

    def encode(self, g: Graph, x: torch.Tensor) -> torch.Tensor:
        h = x
        for i, layer in enumerate(self.layers):
            h = layer(g, h)
            if i != len(self.layers) - 1:
                h = F.elu(h)
        return h

    def forward(self, g: Graph, pos_g: Graph, neg_g: Graph, x: torch.Tensor):
        h = self.encode(g, x)
        return self.pred(pos_g, h), self.pred(neg_g, h)
