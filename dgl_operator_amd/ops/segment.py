"""Segmented readout over contiguous node ranges — K5 of SURVEY.md §2.4.

``segment_reduce`` computes a per-segment sum/mean over rows of a feature
matrix, where segment i covers rows [offsets[i], offsets[i+1]). This is
``dgl.mean_nodes`` over a batched graph (reference:
/root/reference/examples/graph_classification/code/5_graph_classification.py:166).
"""
from __future__ import annotations

import torch

from . import backend


def _segment_ref(offsets: torch.Tensor, feat: torch.Tensor, mean: bool) -> torch.Tensor:
    n = offsets.numel() - 1
    seg = torch.repeat_interleave(
        torch.arange(n, device=feat.device), offsets[1:] - offsets[:-1]
    )
    out = torch.zeros((n,) + feat.shape[1:], dtype=feat.dtype, device=feat.device)
    out.index_add_(0, seg, feat)
    if mean:
        cnt = (offsets[1:] - offsets[:-1]).clamp(min=1).to(feat.dtype)
        out = out / cnt.view(-1, *([1] * (feat.dim() - 1)))
    return out


class _SegmentReduce(torch.autograd.Function):
    @staticmethod
    def forward(ctx, offsets, feat, mean):
        ctx.save_for_backward(offsets)
        ctx.mean = mean
        if feat.is_cuda:
            ext = backend.ext_for(feat)
            return ext.segment_reduce(offsets, feat.contiguous(), mean)
        return _segment_ref(offsets, feat, mean)

    @staticmethod
    def backward(ctx, grad_out):
        (offsets,) = ctx.saved_tensors
        counts = offsets[1:] - offsets[:-1]
        g = grad_out
        if ctx.mean:
            g = g / counts.clamp(min=1).to(g.dtype).view(-1, *([1] * (g.dim() - 1)))
        grad_feat = torch.repeat_interleave(g, counts, dim=0)
        return None, grad_feat, None


def segment_reduce(offsets: torch.Tensor, feat: torch.Tensor, op: str = "mean"):
    assert op in ("sum", "mean")
    return _SegmentReduce.apply(offsets, feat, op == "mean")


def mean_nodes(batch_num_nodes: torch.Tensor, feat: torch.Tensor) -> torch.Tensor:
    """Per-graph mean of node features over a batched graph."""
    offsets = torch.zeros(
        batch_num_nodes.numel() + 1, dtype=torch.int64, device=feat.device
    )
    offsets[1:] = torch.cumsum(batch_num_nodes.to(feat.device), 0)
    return segment_reduce(offsets, feat, "mean")
