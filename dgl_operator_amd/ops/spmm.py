"""Generalized SpMM (message passing aggregate) — K1/K2/K3 of SURVEY.md §2.4.

Computes, over the in-edges of every destination node,
    out[v] = reduce_{e=(u,v)} msg(u, e)
with msg = h[u] (``copy_u``) or h[u] * w[e] (``u_mul_e``, scalar edge weight)
and reduce ∈ {sum, mean}. This is the op behind
``g.update_all(fn.copy_u/u_mul_e, fn.mean/sum)`` in the reference examples
(/root/reference/examples/GraphSAGE/code/3_message_passing.py:113,263) and the
aggregation inside dgl.nn.SAGEConv / GraphConv.

GPU path: hand-written HIP kernels (csrc/gnn_ops.hip) over the CSC
structure; backward is a scatter-atomic walk of the SAME CSC (sampled
blocks; note: float-atomic order makes it run-to-run nondeterministic at
the ulp level) or the cached transposed CSR (full graphs / CPU). CPU path:
pure-PyTorch fp32 reference (index_add_) used for numerics tests.
"""
from __future__ import annotations

from typing import Optional

import torch

from . import backend


def _edge_dst(indptr: torch.Tensor) -> torch.Tensor:
    """Per-CSC-position destination node id."""
    n = indptr.numel() - 1
    return torch.repeat_interleave(
        torch.arange(n, device=indptr.device), indptr[1:] - indptr[:-1]
    )


def _spmm_ref(
    indptr: torch.Tensor,
    indices: torch.Tensor,
    feat: torch.Tensor,
    eweight: Optional[torch.Tensor],
    mean: bool,
) -> torch.Tensor:
    """CPU reference: out[v] = reduce over positions p in [indptr[v], indptr[v+1])."""
    num_rows = indptr.numel() - 1
    msg = feat[indices]
    if eweight is not None:
        msg = msg * eweight.unsqueeze(-1)
    out = torch.zeros(
        (num_rows,) + feat.shape[1:], dtype=feat.dtype, device=feat.device
    )
    out.index_add_(0, _edge_dst(indptr), msg)
    if mean:
        deg = (indptr[1:] - indptr[:-1]).clamp(min=1).to(feat.dtype)
        out = out / deg.view(-1, *([1] * (feat.dim() - 1)))
    return out


def spmm_raw(
    indptr: torch.Tensor,
    indices: torch.Tensor,
    feat: torch.Tensor,
    eweight: Optional[torch.Tensor] = None,
    mean: bool = False,
) -> torch.Tensor:
    """Non-autograd SpMM over an explicit compressed structure."""
    if feat.is_cuda:
        ext = backend.ext_for(feat)
        return ext.spmm(indptr, indices, feat.contiguous(),
                        eweight.contiguous() if eweight is not None else None,
                        mean)
    return _spmm_ref(indptr, indices, feat, eweight, mean)


class _GSpMM(torch.autograd.Function):
    """copy_u/u_mul_e + sum/mean with autograd through feat and eweight.

    eweight, when given, must already be permuted to CSC edge order.
    """

    @staticmethod
    def forward(ctx, gstruct, feat, eweight_csc, mean):
        indptr, indices, _ = gstruct.csc()
        out = spmm_raw(indptr, indices, feat, eweight_csc, mean)
        ctx.gstruct = gstruct
        ctx.mean = mean
        ctx.save_for_backward(
            feat, eweight_csc if eweight_csc is not None else torch.empty(0)
        )
        return out

    @staticmethod
    def backward(ctx, grad_out):
        gstruct, mean = ctx.gstruct, ctx.mean
        feat, eweight = ctx.saved_tensors
        has_w = eweight.numel() > 0
        grad_out = grad_out.contiguous()
        if mean:
            cindptr, _, _ = gstruct.csc()
            deg = (cindptr[1:] - cindptr[:-1]).clamp(min=1).to(grad_out.dtype)
            grad_out = grad_out / deg.view(-1, *([1] * (grad_out.dim() - 1)))
        grad_feat = grad_w = None
        # transpose structure: per-src positions; csr_eids maps csr position ->
        # csc position is NOT direct: both eids map into original edge order.
        cindptr, cindices, ceids = gstruct.csc()
        if ctx.needs_input_grad[1] and grad_out.is_cuda:
            # GPU: scatter-atomic transposed SpMM over the SAME CSC structure
            # (csrc spmm_scatter) — avoids building a per-block CSR (argsort)
            # every backward; weights are already in csc order.
            ext = backend.ext_for(grad_out)
            grad_feat = ext.spmm_scatter(
                cindptr, cindices, grad_out,
                eweight if has_w else None, feat.shape[0],
            )
            if has_w and ctx.needs_input_grad[2]:
                from .sddmm import sddmm_dot_raw

                dst = _edge_dst(cindptr)
                grad_w = sddmm_dot_raw(cindices, dst, feat, grad_out)
            return None, grad_feat, grad_w, None
        rindptr, rindices, reids = gstruct.csr()
        if ctx.needs_input_grad[1]:
            w_r = None
            if has_w:
                # move eweight (csc order) into csr order via original-edge ids
                if ceids is not None:
                    w_orig = torch.empty_like(eweight)
                    w_orig[ceids] = eweight
                else:
                    w_orig = eweight
                w_r = w_orig[reids] if reids is not None else w_orig
            grad_feat = spmm_raw(rindptr, rindices, grad_out, w_r, mean=False)
        if has_w and ctx.needs_input_grad[2]:
            # dL/dw[e=(u,v)] = h[u] . grad_out[v], per csc position
            from .sddmm import sddmm_dot_raw

            dst = _edge_dst(cindptr)
            grad_w = sddmm_dot_raw(cindices, dst, feat, grad_out)
        return None, grad_feat, grad_w, None


def gspmm(
    gstruct,
    op: str,
    reduce: str,
    feat: torch.Tensor,
    eweight: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """Message-passing aggregate over ``gstruct`` (Graph or Block).

    ``eweight`` is given in ORIGINAL edge order (g.edata order) and permuted
    internally to CSC order.
    """
    assert op in ("copy_u", "u_mul_e")
    assert reduce in ("sum", "mean", "max")
    w_csc = None
    if op == "u_mul_e":
        assert eweight is not None, "u_mul_e needs an edge weight"
        _, _, eids = gstruct.csc()
        w_csc = eweight[eids] if eids is not None else eweight
    if reduce == "max":
        # fn.max: eager scatter-amax (autograd routes gradients to the
        # argmax source rows); zero-degree rows get 0 like DGL. Max is not
        # a bandwidth-bound hot path in any reference workload, so no
        # dedicated kernel.
        indptr, indices, _ = gstruct.csc()
        msgs = feat[indices]
        if w_csc is not None:
            w = w_csc.view(w_csc.shape + (1,) * (msgs.dim() - w_csc.dim()))
            msgs = msgs * w
        dst = _edge_dst(indptr)
        num_dst = indptr.numel() - 1
        init = torch.full((num_dst,) + msgs.shape[1:], float("-inf"),
                          dtype=msgs.dtype, device=msgs.device)
        idx = dst.view((-1,) + (1,) * (msgs.dim() - 1)).expand_as(msgs)
        out = torch.scatter_reduce(init, 0, idx, msgs, reduce="amax",
                                   include_self=True)
        deg = (indptr[1:] - indptr[:-1]).view(
            (-1,) + (1,) * (msgs.dim() - 1))
        return torch.where(deg > 0, out, torch.zeros_like(out))
    return _GSpMM.apply(gstruct, feat, w_csc, reduce == "mean")
