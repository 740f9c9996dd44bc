"""g-SDDMM (per-edge dot product) and segmented edge softmax — K4 of SURVEY.md §2.4.

``sddmm_dot`` computes score[e=(u,v)] = <h_u, h_v> per edge (multi-head aware),
the op behind ``g.apply_edges(fn.u_dot_v)`` in the reference link-prediction
example (/root/reference/examples/link_predict/code/4_link_predict.py:210).

``edge_softmax`` normalizes per-edge scores over each destination's in-edge
segment — the GAT attention normalizer (BASELINE config #5).
"""
from __future__ import annotations

import torch

from . import backend
from .spmm import _edge_dst


def sddmm_dot_raw(
    src: torch.Tensor,
    dst: torch.Tensor,
    feat_u: torch.Tensor,
    feat_v: torch.Tensor,
) -> torch.Tensor:
    """score[e] = sum_d feat_u[src[e], ..., d] * feat_v[dst[e], ..., d].

    feat shapes [N, D] -> out [E]; [N, H, D] -> out [E, H].
    """
    if feat_u.is_cuda:
        ext = backend.ext_for(feat_u)
        return ext.sddmm_dot(src, dst, feat_u.contiguous(), feat_v.contiguous())
    return (feat_u[src] * feat_v[dst]).sum(-1)


class _SDDMMDot(torch.autograd.Function):
    @staticmethod
    def forward(ctx, src, dst, feat_u, feat_v):
        ctx.save_for_backward(src, dst, feat_u, feat_v)
        return sddmm_dot_raw(src, dst, feat_u, feat_v)

    @staticmethod
    def backward(ctx, grad_out):
        src, dst, feat_u, feat_v = ctx.saved_tensors
        grad_u = grad_v = None
        g = grad_out.unsqueeze(-1)
        if ctx.needs_input_grad[2]:
            grad_u = torch.zeros_like(feat_u)
            grad_u.index_add_(0, src, (g * feat_v[dst]).to(feat_u.dtype))
        if ctx.needs_input_grad[3]:
            grad_v = torch.zeros_like(feat_v)
            grad_v.index_add_(0, dst, (g * feat_u[src]).to(feat_v.dtype))
        return None, None, grad_u, grad_v


def sddmm_dot(gstruct, feat_u: torch.Tensor, feat_v: torch.Tensor) -> torch.Tensor:
    """u_dot_v over the edges of gstruct, result in ORIGINAL edge order."""
    indptr, indices, eids = gstruct.csc()
    dst = _edge_dst(indptr)
    out_csc = _SDDMMDot.apply(indices, dst, feat_u, feat_v)
    if eids is None:
        return out_csc
    out = torch.empty_like(out_csc)
    out[eids] = out_csc
    return out


def _edge_softmax_ref(indptr: torch.Tensor, scores: torch.Tensor) -> torch.Tensor:
    """CPU reference: per-dst-segment softmax over csc-ordered scores [E, ...]."""
    dst = _edge_dst(indptr)
    n = indptr.numel() - 1
    shape = (n,) + scores.shape[1:]
    m = torch.full(shape, float("-inf"), dtype=scores.dtype, device=scores.device)
    m = m.index_reduce_(0, dst, scores, "amax", include_self=True)
    ex = torch.exp(scores - m[dst])
    s = torch.zeros(shape, dtype=scores.dtype, device=scores.device)
    s.index_add_(0, dst, ex)
    return ex / s.clamp(min=torch.finfo(scores.dtype).tiny)[dst]


class _EdgeSoftmax(torch.autograd.Function):
    """Segmented softmax over in-edges; scores given in CSC order [E] or [E, H]."""

    @staticmethod
    def forward(ctx, indptr, scores):
        if scores.is_cuda:
            ext = backend.ext_for(scores)
            out = ext.edge_softmax_fwd(indptr, scores.contiguous())
        else:
            out = _edge_softmax_ref(indptr, scores)
        ctx.save_for_backward(indptr, out)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        indptr, out = ctx.saved_tensors
        grad_out = grad_out.contiguous()
        if out.is_cuda:
            ext = backend.ext_for(out)
            return None, ext.edge_softmax_bwd(indptr, out, grad_out)
        # d a_i = a_i * (g_i - sum_j a_j g_j) within each segment
        dst = _edge_dst(indptr)
        n = indptr.numel() - 1
        acc = torch.zeros((n,) + out.shape[1:], dtype=out.dtype, device=out.device)
        acc.index_add_(0, dst, out * grad_out)
        return None, out * (grad_out - acc[dst])


def edge_softmax(gstruct, scores: torch.Tensor) -> torch.Tensor:
    """Softmax of ``scores`` (ORIGINAL edge order) over each dst's in-edges."""
    indptr, _, eids = gstruct.csc()
    s_csc = scores[eids] if eids is not None else scores
    out_csc = _EdgeSoftmax.apply(indptr, s_csc)
    if eids is None:
        return out_csc
    out = torch.empty_like(out_csc)
    out[eids] = out_csc
    return out


def edge_softmax_csc(gstruct, scores_csc: torch.Tensor) -> torch.Tensor:
    """Softmax over in-edge segments with scores ALREADY in CSC order (fast path)."""
    indptr, _, _ = gstruct.csc()
    return _EdgeSoftmax.apply(indptr, scores_csc)


class _GATScore(torch.autograd.Function):
    """Fused LeakyReLU(el[src] + er[dst]) over CSC positions (u_add_v +
    activation, the GAT attention logits). GPU: one kernel each way
    (csrc gat_score_*); CPU: the equivalent torch expression."""

    @staticmethod
    def forward(ctx, src, dst, el, er, slope):
        ctx.save_for_backward(src, dst, el, er)
        ctx.slope = slope
        if el.is_cuda and el.dtype == torch.float32 and backend.has_extension():
            ext = backend.ext_for(el)
            return ext.gat_score_fwd(src, dst, el, er, slope)
        v = el[src] + er[dst]
        return torch.where(v > 0, v, slope * v)

    @staticmethod
    def backward(ctx, grad_out):
        src, dst, el, er = ctx.saved_tensors
        grad_out = grad_out.contiguous()
        if el.is_cuda and el.dtype == torch.float32 and backend.has_extension():
            ext = backend.ext_for(el)
            gel, ger = ext.gat_score_bwd(src, dst, el, er, grad_out, ctx.slope)
            return None, None, gel, ger, None
        v = el[src] + er[dst]
        g = grad_out * torch.where(v > 0, torch.ones_like(v),
                                   torch.full_like(v, ctx.slope))
        gel = torch.zeros_like(el)
        gel.index_add_(0, src, g)
        ger = torch.zeros_like(er)
        ger.index_add_(0, dst, g)
        return None, None, gel, ger, None


def gat_score(gstruct, el: torch.Tensor, er: torch.Tensor,
              negative_slope: float = 0.2) -> torch.Tensor:
    """Per-edge attention logits in CSC order: LeakyReLU(el_u + er_v)."""
    _, indices, _ = gstruct.csc()
    dst = gstruct.csc_dst()
    return _GATScore.apply(indices, dst, el, er, negative_slope)
