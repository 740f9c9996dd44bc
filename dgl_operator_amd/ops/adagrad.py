"""Row-sparse Adagrad embedding update — K10 of SURVEY.md §2.4.

Matches the reference's KGEServer push handler semantics
(/root/reference/examples/DGL-KE/hotfix/kvserver.py:41-51):
    state[ids] += mean(grad^2, dim=1)        (duplicate ids accumulate)
    emb[ids]   -= lr * grad / (sqrt(state[ids]) + eps)
with the std read AFTER the full state update. On GPU both phases are HIP
kernels (csrc/adagrad.hip) using atomics so duplicate ids are handled.
"""
from __future__ import annotations

import torch

from . import backend


def sparse_adagrad_update(
    emb: torch.Tensor,
    state: torch.Tensor,
    ids: torch.Tensor,
    grad: torch.Tensor,
    lr: float,
    eps: float = 1e-10,
) -> None:
    """In-place row-sparse Adagrad. emb [N, D], state [N], ids [B], grad [B, D]."""
    assert emb.dim() == 2 and grad.dim() == 2 and state.dim() == 1
    if emb.is_cuda:
        ext = backend.ext_for(emb)
        ext.sparse_adagrad(emb, state, ids, grad.contiguous(), lr, eps)
        return
    state.index_add_(0, ids, (grad * grad).mean(1))
    std = state[ids].sqrt() + eps
    emb.index_add_(0, ids, -lr * grad / std.unsqueeze(1))
