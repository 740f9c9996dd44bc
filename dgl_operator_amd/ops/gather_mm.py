"""Fused gather+GEMM op (fp32 MFMA) and the lazy feature view that feeds it.

``gather_mm(feat, rows, W, b)`` computes feat[rows] @ W + b in ONE kernel on
the gfx950 fp32 matrix cores (csrc/gather_mm.hip) — the GNN input projection
without ever materializing the gathered feature matrix.

``GatherView`` wraps (feature table, row ids) so a layer can consume either
a dense tensor or the lazy view; SAGEConv uses it to run its two input
projections as fused gather-GEMMs. Gradients flow to the weights/bias
(features are frozen inputs in this framework, like the reference's).
"""
from __future__ import annotations

from typing import Optional

import torch

from . import backend


class _GatherMM(torch.autograd.Function):
    @staticmethod
    def forward(ctx, feat, rows, weight, bias):
        ctx.save_for_backward(feat, rows)
        ctx.has_bias = bias is not None
        if (feat.is_cuda and weight.shape[1] <= 16 and feat.shape[1] <= 204
                and feat.dtype == torch.float32 and backend.has_extension()):
            ext = backend.ext_for(feat)
            return ext.gather_mm(feat, rows, weight.contiguous(), bias)
        out = feat[rows] @ weight
        if bias is not None:
            out = out + bias
        return out

    @staticmethod
    def backward(ctx, grad_out):
        feat, rows = ctx.saved_tensors
        grad_out = grad_out.contiguous()
        # rematerialize the gathered block once in backward (features are
        # not trainable, so only W/b need grads)
        if feat.is_cuda and backend.has_extension():
            ext = backend.ext_for(feat)
            x = ext.gather_rows(feat, rows, None, 0)
        else:
            x = feat[rows]
        grad_w = x.t() @ grad_out
        grad_b = grad_out.sum(0) if ctx.has_bias else None
        return None, None, grad_w, grad_b


def gather_mm(
    feat: torch.Tensor,
    rows: torch.Tensor,
    weight: torch.Tensor,
    bias: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """feat[rows] @ weight (+ bias); differentiable through weight/bias only
    (feat must not require grad)."""
    assert not feat.requires_grad, "gather_mm: feature table must be frozen"
    return _GatherMM.apply(feat, rows, weight, bias)


class GatherView:
    """Lazy (feature table, row ids) pair. ``materialize()`` produces the
    dense gathered matrix; fusion-aware layers instead project straight from
    the table with gather_mm."""

    def __init__(self, feat: torch.Tensor, rows: torch.Tensor):
        assert feat.dim() == 2
        self.feat = feat
        self.rows = rows

    @property
    def shape(self):
        return (self.rows.numel(), self.feat.shape[1])

    def __len__(self):
        return self.rows.numel()

    def narrow_rows(self, n: int) -> "GatherView":
        return GatherView(self.feat, self.rows[:n])

    def materialize(self) -> torch.Tensor:
        if self.feat.is_cuda and backend.has_extension():
            ext = backend.ext_for(self.feat)
            return ext.gather_rows(self.feat, self.rows, None, 0)
        return self.feat[self.rows]

    def project(self, weight_t: torch.Tensor,
                bias: Optional[torch.Tensor] = None) -> torch.Tensor:
        """self @ weight_t (+ bias) with weight_t laid out [K, N]."""
        return gather_mm(self.feat, self.rows, weight_t, bias)
