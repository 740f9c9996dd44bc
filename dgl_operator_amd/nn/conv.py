"""GNN layers: SAGEConv / GraphConv / GATConv.

Functional equivalents of the dgl.nn layers the reference examples use
(dgl.nn.SAGEConv at /root/reference/examples/GraphSAGE_dist/code/train_dist.py:80-83,
dgl.nn.GraphConv at examples/node_classification/code/1_introduction.py:119-126,
GAT/edge-softmax generalizing examples/link_predict). Aggregation rides the
HIP gspmm/sddmm/edge_softmax ops; the dense projections are plain GEMMs
(hipBLASLt via torch.nn.Linear).

All layers accept either a full Graph (operates on all nodes) or a bipartite
Block (dst nodes are the first ``num_dst`` src rows).
"""
from __future__ import annotations

from typing import Union

import torch
import torch.nn as nn

import torch.nn.functional as F

from ..graph.graph import Block, Graph
from ..ops import gspmm, edge_softmax_csc
from ..ops.sddmm import gat_score
from ..ops.gather_mm import GatherView


def _num_dst(g) -> int:
    return g.num_dst_nodes if isinstance(g, Block) else g.num_nodes


def _dst_feat(g, x: torch.Tensor) -> torch.Tensor:
    return x[: g.num_dst_nodes] if isinstance(g, Block) else x


class SAGEConv(nn.Module):
    """GraphSAGE convolution, 'mean' aggregator:
    out = W_self h_dst + W_neigh mean_{u in N(v)} h_u  (+ bias)."""

    def __init__(self, in_feats: int, out_feats: int, aggregator: str = "mean",
                 bias: bool = True):
        super().__init__()
        assert aggregator in ("mean", "sum", "gcn"), aggregator
        self.aggregator = aggregator
        # 'gcn' folds the self feature into the normalized aggregate (DGL
        # SAGEConv aggregator_type='gcn'): no separate fc_self
        self.fc_self = (None if aggregator == "gcn"
                        else nn.Linear(in_feats, out_feats, bias=False))
        self.fc_neigh = nn.Linear(in_feats, out_feats, bias=bias)
        self.reset_parameters()

    def reset_parameters(self):
        gain = nn.init.calculate_gain("relu")
        if self.fc_self is not None:
            nn.init.xavier_uniform_(self.fc_self.weight, gain=gain)
        nn.init.xavier_uniform_(self.fc_neigh.weight, gain=gain)
        if self.fc_neigh.bias is not None:
            nn.init.zeros_(self.fc_neigh.bias)

    def forward(self, g: Union[Graph, Block], x: torch.Tensor,
                edge_weight: torch.Tensor = None) -> torch.Tensor:
        """Optional per-edge scalar weight gives the reference's
        WeightedSAGEConv (u_mul_e aggregate,
        /root/reference/examples/GraphSAGE/code/3_message_passing.py:263).

        When in_feats > out_feats the projection runs BEFORE aggregation:
        sum/mean are linear, so mean(x W) + b == fc_neigh(mean(x)), and the
        SpMM moves out_feats-wide rows instead of in_feats-wide ones (6x
        less HBM traffic at the bench shape 100 -> 16). Zero-degree rows
        match too: both orders yield exactly the bias."""
        op = "copy_u" if edge_weight is None else "u_mul_e"
        if self.aggregator == "gcn":
            if isinstance(x, GatherView):
                x = x.materialize()
            agg = gspmm(g, op, "sum", x, edge_weight)
            xd = _dst_feat(g, x)
            deg = (
                g.in_degrees() if not isinstance(g, Block)
                else (g.csc_indptr[1:] - g.csc_indptr[:-1])
            ).to(x.dtype).unsqueeze(-1)
            return self.fc_neigh((agg + xd) / (deg + 1))
        if isinstance(x, GatherView):
            fusable = (
                x.feat.is_cuda
                and x.feat.dtype == torch.float32
                and self.fc_neigh.out_features <= 16
                and x.feat.shape[1] <= 204  # gather_mm LDS envelope (64*(Kp+1)+Kp*16 floats <= 64 KiB => Kp <= 204)
            )
            if not fusable:
                x = x.materialize()
        if isinstance(x, GatherView):
            # fused input projections straight from the feature table (MFMA
            # gather_mm): no materialized x, projection before aggregation
            pre = x.project(self.fc_neigh.weight.t())
            h_neigh = gspmm(g, op, self.aggregator, pre, edge_weight)
            if self.fc_neigh.bias is not None:
                h_neigh = h_neigh + self.fc_neigh.bias
            nd = g.num_dst_nodes if isinstance(g, Block) else g.num_nodes
            h_self = x.narrow_rows(nd).project(self.fc_self.weight.t())
            return h_self + h_neigh
        if self.fc_neigh.in_features > self.fc_neigh.out_features:
            pre = F.linear(x, self.fc_neigh.weight)  # bias added after
            h_neigh = gspmm(g, op, self.aggregator, pre, edge_weight)
            if self.fc_neigh.bias is not None:
                h_neigh = h_neigh + self.fc_neigh.bias
        else:
            h_neigh = self.fc_neigh(
                gspmm(g, op, self.aggregator, x, edge_weight)
            )
        return self.fc_self(_dst_feat(g, x)) + h_neigh


class GraphConv(nn.Module):
    """Kipf-Welling GCN layer with symmetric normalization:
    out = D̂^{-1/2} Â D̂^{-1/2} X W (norm='both'); Â should include self loops
    (call g.add_self_loops() like the reference examples do)."""

    def __init__(self, in_feats: int, out_feats: int, norm: str = "both",
                 bias: bool = True, activation=None):
        super().__init__()
        assert norm in ("both", "right", "none")
        self.norm = norm
        self.weight = nn.Parameter(torch.empty(in_feats, out_feats))
        self.bias = nn.Parameter(torch.zeros(out_feats)) if bias else None
        self.activation = activation
        nn.init.xavier_uniform_(self.weight)

    def forward(self, g: Union[Graph, Block], x: torch.Tensor) -> torch.Tensor:
        in_feats, out_feats = self.weight.shape
        if self.norm == "both":
            if isinstance(g, Block):
                rindptr, _, _ = g.csr()
                out_deg = (rindptr[1:] - rindptr[:-1]).clamp(min=1).to(x.dtype)
            else:
                out_deg = g.out_degrees().clamp(min=1).to(x.dtype)
            x = x * out_deg.pow(-0.5).unsqueeze(-1)
        # project first when it shrinks the aggregation width
        if in_feats > out_feats:
            x = x @ self.weight
            h = gspmm(g, "copy_u", "sum", x)
        else:
            h = gspmm(g, "copy_u", "sum", x)
            h = h @ self.weight
        if self.norm in ("both", "right"):
            cindptr, _, _ = g.csc()
            in_deg = (cindptr[1:] - cindptr[:-1]).clamp(min=1).to(h.dtype)
            p = -0.5 if self.norm == "both" else -1.0
            h = h * in_deg.pow(p).unsqueeze(-1)
        if self.bias is not None:
            h = h + self.bias
        if self.activation is not None:
            h = self.activation(h)
        return h


class GATConv(nn.Module):
    """Graph attention layer (multi-head), the SDDMM + edge-softmax op family
    (BASELINE config #5).

    score(e=(u,v)) = LeakyReLU(a_l . z_u + a_r . z_v), alpha = edge_softmax,
    out[v] = sum_e alpha_e z_u,   z = W x  (per head).
    """

    def __init__(self, in_feats: int, out_feats: int, num_heads: int = 1,
                 negative_slope: float = 0.2, bias: bool = True,
                 feat_drop: float = 0.0, attn_drop: float = 0.0,
                 residual: bool = False):
        super().__init__()
        self.num_heads = num_heads
        self.out_feats = out_feats
        self.fc = nn.Linear(in_feats, out_feats * num_heads, bias=False)
        self.attn_l = nn.Parameter(torch.empty(num_heads, out_feats))
        self.attn_r = nn.Parameter(torch.empty(num_heads, out_feats))
        self.bias = nn.Parameter(torch.zeros(num_heads * out_feats)) if bias else None
        self.leaky = nn.LeakyReLU(negative_slope)
        self.feat_drop = nn.Dropout(feat_drop)
        self.attn_drop = nn.Dropout(attn_drop)
        self.res_fc = None
        if residual:
            self.res_fc = (
                nn.Identity() if in_feats == out_feats * num_heads
                else nn.Linear(in_feats, out_feats * num_heads, bias=False)
            )
        self.reset_parameters()

    def reset_parameters(self):
        gain = nn.init.calculate_gain("relu")
        nn.init.xavier_uniform_(self.fc.weight, gain=gain)
        nn.init.xavier_uniform_(self.attn_l, gain=gain)
        nn.init.xavier_uniform_(self.attn_r, gain=gain)

    def forward(self, g: Union[Graph, Block], x: torch.Tensor) -> torch.Tensor:
        H, D = self.num_heads, self.out_feats
        x = self.feat_drop(x)
        z = self.fc(x).view(-1, H, D)  # [N, H, D]
        el = (z * self.attn_l).sum(-1)  # [N, H]
        er = (z * self.attn_r).sum(-1)
        # fused u_add_v + LeakyReLU attention logits (csc order)
        score = gat_score(g, el, er, self.leaky.negative_slope)
        alpha = self.attn_drop(edge_softmax_csc(g, score))
        out = gspmm(g, "u_mul_e", "sum", z, _csc_weight(g, alpha))  # [Nd, H, D]
        out = out.reshape(-1, H * D)
        if self.res_fc is not None:
            out = out + self.res_fc(_dst_feat(g, x))
        if self.bias is not None:
            out = out + self.bias
        return out


def _csc_weight(g, w_csc: torch.Tensor) -> torch.Tensor:
    """gspmm permutes eweight original->csc via eids; hand it a tensor that is
    already in csc order by pre-inverting when the graph carries eids."""
    _, _, eids = g.csc()
    if eids is None:
        return w_csc
    w = torch.empty_like(w_csc)
    w[eids] = w_csc
    return w
