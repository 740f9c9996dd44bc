"""UDF message-passing fallback (K6 of SURVEY.md §2.4).

The reference exercises arbitrary Python message/reduce UDFs
(/root/reference/examples/GraphSAGE/code/3_message_passing.py:300-321):
``update_all(u_mul_e_udf, sum_udf)`` with ``edges.src['h'] * edges.data['w']``
and ``nodes.mailbox['m'].sum(1)``. Builtin pairs go to the fused HIP kernels;
UDFs take this generic gather / degree-bucketed-reduce path: messages are
materialized per edge, then destination nodes with equal in-degree are
batched so the mailbox is a dense [nodes, degree, ...] tensor — DGL's
degree-bucketing strategy. Differentiable through plain torch autograd.
"""
from __future__ import annotations

from typing import Callable, Dict

import torch


class EdgeBatch:
    def __init__(self, src_data, dst_data, edge_data):
        self.src = src_data
        self.dst = dst_data
        self.data = edge_data


class NodeBatch:
    def __init__(self, data, mailbox):
        self.data = data
        self.mailbox = mailbox


def update_all_udf(gstruct, ndata: Dict[str, torch.Tensor],
                   edata: Dict[str, torch.Tensor],
                   message_func: Callable, reduce_func: Callable,
                   num_dst: int, dstdata: Dict[str, torch.Tensor] = None):
    """Returns a dict of reduced node fields over in-edges of every dst.

    Destinations with no in-edges get zero rows (DGL semantics); on a graph
    with NO edges at all the reduce never runs, so no output fields are
    produced (output shapes are only known from the reducer).

    For Blocks ``dstdata`` carries fields stored per-dst-row; dst-side
    lookups (``edges.dst[...]``, ``nodes.data[...]``) read it with
    precedence over ``ndata`` — relying on the dst-first src-row convention
    alone would make dst-only fields invisible to UDFs."""
    indptr, indices, eids = gstruct.csc()
    from .spmm import _edge_dst

    dst = _edge_dst(indptr)
    src_data = {k: v[indices] for k, v in ndata.items()}
    # dst-side view: dst rows are the first num_dst src rows by convention,
    # but fields assigned only on dstdata must win / still be visible
    dst_side = dict(ndata)
    if dstdata:
        dst_side.update(dstdata)
    dst_data = {k: v[dst] for k, v in dst_side.items()}
    if eids is not None:
        edge_data = {k: v[eids] for k, v in edata.items()}
    else:
        edge_data = dict(edata)
    msgs = message_func(EdgeBatch(src_data, dst_data, edge_data))

    deg = indptr[1:] - indptr[:-1]
    out: Dict[str, torch.Tensor] = {}
    for d in torch.unique(deg).tolist():
        if d == 0:
            continue
        rows = (deg == d).nonzero(as_tuple=True)[0]
        # positions of each row's d messages in csc order
        starts = indptr[rows]
        pos = starts.unsqueeze(1) + torch.arange(d, device=starts.device)
        mailbox = {k: v[pos.reshape(-1)].view(rows.numel(), d, *v.shape[1:])
                   for k, v in msgs.items()}
        reduced = reduce_func(
            NodeBatch({k: v[rows] for k, v in dst_side.items()}, mailbox)
        )
        for k, v in reduced.items():
            if k not in out:
                out[k] = torch.zeros(
                    (num_dst,) + tuple(v.shape[1:]), dtype=v.dtype,
                    device=v.device,
                )
            out[k][rows] = v
    return out
