"""Builtin message/reduce function descriptors — the dgl.function ("fn")
API surface the reference examples use
(/root/reference/examples/GraphSAGE/code/3_message_passing.py:113,263,
examples/link_predict/code/4_link_predict.py:210):

    import dgl_operator_amd.fn as fn
    g.update_all(fn.copy_u('h', 'm'), fn.mean('m', 'h_N'))
    g.update_all(fn.u_mul_e('h', 'w', 'm'), fn.sum('m', 'h_N'))
    g.apply_edges(fn.u_dot_v('h', 'h', 'score'))

Builtin pairs map onto the fused HIP gspmm/sddmm kernels; arbitrary Python
UDFs fall back to the degree-bucketing path in ops/udf.py (K6 of
SURVEY.md §2.4).
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class MessageFn:
    op: str  # copy_u | u_mul_e
    src_field: str
    edge_field: str | None
    out_field: str


@dataclass(frozen=True)
class ReduceFn:
    op: str  # sum | mean
    msg_field: str
    out_field: str


@dataclass(frozen=True)
class EdgeFn:
    op: str  # u_dot_v
    lhs_field: str
    rhs_field: str
    out_field: str


def copy_u(u: str, out: str) -> MessageFn:
    return MessageFn("copy_u", u, None, out)


# DGL alias
copy_src = copy_u


def u_mul_e(u: str, e: str, out: str) -> MessageFn:
    return MessageFn("u_mul_e", u, e, out)


def sum(msg: str, out: str) -> ReduceFn:  # noqa: A001 (DGL API name)
    return ReduceFn("sum", msg, out)


def mean(msg: str, out: str) -> ReduceFn:
    return ReduceFn("mean", msg, out)


def max(msg: str, out: str) -> ReduceFn:  # noqa: A001 (DGL API name)
    return ReduceFn("max", msg, out)


def u_dot_v(u: str, v: str, out: str) -> EdgeFn:
    return EdgeFn("u_dot_v", u, v, out)


def u_add_v(u: str, v: str, out: str) -> EdgeFn:
    """Per-edge src+dst field sum — the standard DGL spelling for GAT
    attention logits (el[src] + er[dst]); the fused LeakyReLU variant is
    ops.gat_score."""
    return EdgeFn("u_add_v", u, v, out)


__all__ = [
    "MessageFn",
    "ReduceFn",
    "EdgeFn",
    "copy_u",
    "copy_src",
    "u_mul_e",
    "sum",
    "mean",
    "max",
    "u_dot_v",
    "u_add_v",
]
