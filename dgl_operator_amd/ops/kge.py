"""Knowledge-graph embedding score functions — K9 of SURVEY.md §2.4.

Score functions matching the DGL-KE model zoo the reference runs
(/root/reference/examples/DGL-KE/hotfix/kvserver.py:66-68 lists
TransE/TransE_l1/TransE_l2/TransR/RESCAL/DistMult/ComplEx/RotatE; the
reference's anchor config is ComplEx d=400, γ=143 —
examples/v1alpha1/DGL-KE.yaml:26-27).

Positive scores are per-triple; negative scores follow DGL-KE's CHUNKED
corruption: the batch is split into chunks and each chunk shares one set of
``neg_sample_size`` corrupt entities, which turns neg scoring into small
GEMMs (DistMult/ComplEx -> torch.bmm on rocBLAS/MFMA) or broadcast distance
reductions (TransE/RotatE -> fused HIP kernel on GPU).

All functions return "higher is better" scores (γ − distance for the
translational models).
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn.functional as F

from . import backend

SCORE_FUNCS = {}


class _PDistNeg(torch.autograd.Function):
    """Fused pairwise-distance neg score (HIP kernel, csrc/kge.hip):
    out[C,c,n] = gamma - ||base[C,c,:] - neg[C,n,:]||_p without materializing
    the [C,c,n,D] broadcast."""

    @staticmethod
    def forward(ctx, base, neg, p, gamma):
        ext = backend.ext_for(base)
        base, neg = base.contiguous(), neg.contiguous()
        out = ext.pdist_neg_fwd(base, neg, p, gamma)
        ctx.save_for_backward(base, neg, out)
        ctx.p, ctx.gamma = p, gamma
        return out

    @staticmethod
    def backward(ctx, gout):
        base, neg, out = ctx.saved_tensors
        ext = backend.ext_for(base)
        gb, gn = ext.pdist_neg_bwd(base, neg, out, gout.contiguous(), ctx.p,
                                   ctx.gamma)
        return gb, gn, None, None


class _CPDistNeg(torch.autograd.Function):
    """Fused complex-modulus pairwise distance (RotatE)."""

    @staticmethod
    def forward(ctx, base_r, base_i, neg, gamma):
        ext = backend.ext_for(base_r)
        base_r, base_i, neg = (
            base_r.contiguous(), base_i.contiguous(), neg.contiguous()
        )
        out = ext.cpdist_neg_fwd(base_r, base_i, neg, gamma)
        ctx.save_for_backward(base_r, base_i, neg)
        return out

    @staticmethod
    def backward(ctx, gout):
        base_r, base_i, neg = ctx.saved_tensors
        ext = backend.ext_for(base_r)
        gbr, gbi, gn = ext.cpdist_neg_bwd(base_r, base_i, neg,
                                          gout.contiguous())
        return gbr, gbi, gn, None


def _register(name):
    def deco(cls):
        SCORE_FUNCS[name] = cls()
        return cls

    return deco


class ScoreFunc:
    """head/rel/tail: [B, D] (rel may be [B, Dr]). neg_*: chunked layout
    [num_chunk, chunk, D] vs neg entities [num_chunk, neg, D]."""

    name = "base"

    def edge(self, head, rel, tail) -> torch.Tensor:
        raise NotImplementedError

    def neg(self, head, rel, neg_tail, neg_head: bool = False) -> torch.Tensor:
        """Default: broadcast edge() over the chunk's negatives.

        head/rel: [C, c, D]; neg_tail: [C, n, D] -> scores [C, c, n].
        When ``neg_head``, `head` is the corrupted side: callers pass
        (tail, rel, neg_head) and the score function adjusts direction.
        """
        C, c, D = head.shape
        n = neg_tail.shape[1]
        h = head.unsqueeze(2)  # [C, c, 1, D]
        r = rel.unsqueeze(2)
        t = neg_tail.unsqueeze(1)  # [C, 1, n, D]
        return self._bcast(h, r, t, neg_head)

    def _bcast(self, h, r, t, neg_head):
        raise NotImplementedError


@_register("TransE")
@_register("TransE_l2")
class TransEL2(ScoreFunc):
    name = "TransE_l2"
    gamma_default = 12.0
    p = 2

    def __init__(self, gamma: float = 12.0):
        self.gamma = gamma

    def edge(self, head, rel, tail):
        return self.gamma - torch.norm(head + rel - tail, p=2, dim=-1)

    def _bcast(self, h, r, t, neg_head):
        d = (h + r - t) if not neg_head else (t + r - h)
        return self.gamma - torch.norm(d, p=self.p, dim=-1)

    def neg(self, head, rel, neg_tail, neg_head: bool = False):
        # ||h + r - t_neg|| (neg tail) and ||h_neg - (t - r)|| (neg head,
        # `head` arg carries the tail rows) share one pairwise-distance form
        if head.is_cuda and backend.has_extension():
            base = head - rel if neg_head else head + rel
            return _PDistNeg.apply(base, neg_tail, self.p, self.gamma)
        return super().neg(head, rel, neg_tail, neg_head)


@_register("TransE_l1")
class TransEL1(TransEL2):
    name = "TransE_l1"
    p = 1

    def edge(self, head, rel, tail):
        return self.gamma - torch.norm(head + rel - tail, p=1, dim=-1)


@_register("DistMult")
class DistMult(ScoreFunc):
    name = "DistMult"

    def edge(self, head, rel, tail):
        return (head * rel * tail).sum(-1)

    def neg(self, head, rel, neg_tail, neg_head: bool = False):
        # symmetric in head/tail: (h*r) @ t_neg^T  — GEMM-shaped, rocBLAS/MFMA
        hr = head * rel  # [C, c, D]
        return torch.bmm(hr, neg_tail.transpose(1, 2))  # [C, c, n]


@_register("ComplEx")
class ComplEx(ScoreFunc):
    name = "ComplEx"

    def edge(self, head, rel, tail):
        D = head.shape[-1] // 2
        hr, hi = head[..., :D], head[..., D:]
        rr, ri = rel[..., :D], rel[..., D:]
        tr, ti = tail[..., :D], tail[..., D:]
        return ((hr * rr - hi * ri) * tr + (hr * ri + hi * rr) * ti).sum(-1)

    def neg(self, head, rel, neg_tail, neg_head: bool = False):
        D = head.shape[-1] // 2
        hr, hi = head[..., :D], head[..., D:]
        rr, ri = rel[..., :D], rel[..., D:]
        if neg_head:
            # score(h_neg, r, t) with `head`=tail rows: Re(<conj(t ∘ conj(r)?)…)
            # ComplEx: s(h,r,t)=Re(<h ∘ r, conj(t)>) ⇒ as a function of h it is
            # linear: s = <h, m> with m = Re/Im combination of r,t.
            mr = hr * rr + hi * ri  # here head args are tail embeddings
            mi = hi * rr - hr * ri
        else:
            mr = hr * rr - hi * ri
            mi = hr * ri + hi * rr
        m = torch.cat([mr, mi], dim=-1)  # [C, c, 2D]
        return torch.bmm(m, neg_tail.transpose(1, 2))


@_register("RotatE")
class RotatE(ScoreFunc):
    name = "RotatE"
    gamma_default = 12.0

    def __init__(self, gamma: float = 12.0, emb_init: float = 1.0):
        self.gamma = gamma
        self.emb_init = emb_init

    def _phase(self, rel):
        return rel / (self.emb_init / math.pi)

    def edge(self, head, rel, tail):
        D = head.shape[-1] // 2
        hr, hi = head[..., :D], head[..., D:]
        tr, ti = tail[..., :D], tail[..., D:]
        ph = self._phase(rel)
        rr, ri = torch.cos(ph), torch.sin(ph)
        dr = hr * rr - hi * ri - tr
        di = hr * ri + hi * rr - ti
        return self.gamma - torch.sqrt(dr * dr + di * di).sum(-1)

    def _bcast(self, h, r, t, neg_head):
        D = h.shape[-1] // 2
        ph = self._phase(r)
        rr, ri = torch.cos(ph), torch.sin(ph)
        hr, hi = h[..., :D], h[..., D:]
        tr, ti = t[..., :D], t[..., D:]
        if neg_head:
            # corrupt head: rotate negative heads forward
            dr = tr * rr - ti * ri - hr
            di = tr * ri + ti * rr - hi
        else:
            dr = hr * rr - hi * ri - tr
            di = hr * ri + hi * rr - ti
        return self.gamma - torch.sqrt(dr * dr + di * di).sum(-1)

    def neg(self, head, rel, neg_tail, neg_head: bool = False):
        # one fused pairwise form for both corruption sides: rotation is an
        # isometry, so ||rot(neg) - tail|| = ||neg - rot^-1(tail)|| — corrupt
        # head uses the tail rotated by MINUS the phase as the base
        if head.is_cuda and backend.has_extension():
            D2 = head.shape[-1] // 2
            ph = self._phase(rel)
            rr, ri = torch.cos(ph), torch.sin(ph)
            hr, hi = head[..., :D2], head[..., D2:]
            if neg_head:  # `head` carries tail rows; rotate by -phase
                base_r = hr * rr + hi * ri
                base_i = hi * rr - hr * ri
            else:
                base_r = hr * rr - hi * ri
                base_i = hr * ri + hi * rr
            return _CPDistNeg.apply(base_r, base_i, neg_tail, self.gamma)
        h = head.unsqueeze(2)
        r = rel.unsqueeze(2)
        t = neg_tail.unsqueeze(1)
        return self._bcast(h, r, t, neg_head)


@_register("RESCAL")
class RESCAL(ScoreFunc):
    name = "RESCAL"

    def edge(self, head, rel, tail):
        D = head.shape[-1]
        M = rel.view(*rel.shape[:-1], D, D)
        ht = torch.einsum("...d,...de->...e", head, M)
        return (ht * tail).sum(-1)

    def neg(self, head, rel, neg_tail, neg_head: bool = False):
        D = head.shape[-1]
        M = rel.view(*rel.shape[:-1], D, D)
        if neg_head:
            # head args are tail embeddings; score linear in the corrupt head
            m = torch.einsum("...de,...e->...d", M, head)
        else:
            m = torch.einsum("...d,...de->...e", head, M)
        return torch.bmm(m, neg_tail.transpose(1, 2))


@_register("TransR")
class TransR(ScoreFunc):
    """rel embedding = [r (Dr) | proj matrix (D*Dr)] concatenated."""

    name = "TransR"
    gamma_default = 12.0

    def __init__(self, gamma: float = 12.0):
        self.gamma = gamma

    def _split(self, rel, D):
        Dr = _transr_dr(rel.shape[-1], D)
        r = rel[..., :Dr]
        M = rel[..., Dr:].view(*rel.shape[:-1], D, Dr)
        return r, M

    def edge(self, head, rel, tail):
        D = head.shape[-1]
        r, M = self._split(rel, D)
        hp = torch.einsum("...d,...de->...e", head, M)
        tp = torch.einsum("...d,...de->...e", tail, M)
        return self.gamma - torch.norm(hp + r - tp, p=2, dim=-1)

    def neg(self, head, rel, neg_tail, neg_head: bool = False):
        D = neg_tail.shape[-1]
        r, M = self._split(rel, D)
        hp = torch.einsum("...d,...de->...e", head, M)
        # project the chunk's negatives with every chunk-row's relation matrix
        tp = torch.einsum("Cnd,Ccde->Ccne", neg_tail, M)
        if neg_head:
            d = tp + r.unsqueeze(2) - hp.unsqueeze(2)
        else:
            d = hp.unsqueeze(2) + r.unsqueeze(2) - tp
        return self.gamma - torch.norm(d, p=2, dim=-1)


def _transr_dr(rel_dim: int, d: int) -> int:
    # rel_dim = Dr + D*Dr  =>  Dr = rel_dim / (1 + D)
    dr = rel_dim // (1 + d)
    assert dr * (1 + d) == rel_dim, "TransR relation dim must be Dr*(1+D)"
    return dr


def get_score_func(name: str, gamma: Optional[float] = None, emb_init: float = 1.0):
    fn = SCORE_FUNCS[name]
    if gamma is not None and hasattr(fn, "gamma"):
        import copy

        fn = copy.copy(fn)
        fn.gamma = gamma
        if hasattr(fn, "emb_init"):
            fn.emb_init = emb_init
    return fn


def kge_loss(
    pos_score: torch.Tensor,
    neg_score: torch.Tensor,
    adversarial_temperature: float = 1.0,
    self_adversarial: bool = True,
) -> torch.Tensor:
    """DGL-KE logsigmoid loss with optional self-adversarial negative weighting
    (the -adv flag the reference's dglkerun always passes)."""
    pos_loss = -F.logsigmoid(pos_score).mean()
    if self_adversarial:
        w = torch.softmax(neg_score * adversarial_temperature, dim=-1).detach()
        neg_loss = -(w * F.logsigmoid(-neg_score)).sum(-1).mean()
    else:
        neg_loss = -F.logsigmoid(-neg_score).mean()
    return (pos_loss + neg_loss) / 2
