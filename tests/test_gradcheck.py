"""Rigorous float64 torch.autograd.gradcheck over the op layer's CPU
reference paths — the ground truth the HIP kernels are tested against."""
import torch
from torch.autograd import gradcheck

from dgl_operator_amd.graph import rmat_graph
from dgl_operator_amd.ops import gspmm, sddmm_dot, edge_softmax, segment_reduce
from dgl_operator_amd.ops.gather_mm import gather_mm


def tiny_graph():
    return rmat_graph(12, 40, seed=5)


def test_gradcheck_gspmm_sum_and_mean():
    g = tiny_graph()
    x = torch.randn(12, 3, dtype=torch.float64, requires_grad=True)
    assert gradcheck(lambda t: gspmm(g, "copy_u", "sum", t), (x,),
                     eps=1e-6, atol=1e-5)
    assert gradcheck(lambda t: gspmm(g, "copy_u", "mean", t), (x,),
                     eps=1e-6, atol=1e-5)


def test_gradcheck_gspmm_weighted():
    g = tiny_graph()
    x = torch.randn(12, 3, dtype=torch.float64, requires_grad=True)
    w = torch.rand(g.num_edges, dtype=torch.float64, requires_grad=True)
    assert gradcheck(lambda t, ww: gspmm(g, "u_mul_e", "mean", t, ww),
                     (x, w), eps=1e-6, atol=1e-5)


def test_gradcheck_sddmm_and_softmax():
    g = tiny_graph()
    h = torch.randn(12, 4, dtype=torch.float64, requires_grad=True)
    assert gradcheck(lambda t: sddmm_dot(g, t, t), (h,), eps=1e-6, atol=1e-5)
    s = torch.randn(g.num_edges, dtype=torch.float64, requires_grad=True)
    assert gradcheck(lambda t: edge_softmax(g, t), (s,), eps=1e-6, atol=1e-5)


def test_gradcheck_segment_reduce():
    x = torch.randn(9, 2, dtype=torch.float64, requires_grad=True)
    offsets = torch.tensor([0, 3, 3, 9])
    assert gradcheck(lambda t: segment_reduce(offsets, t, "mean"), (x,),
                     eps=1e-6, atol=1e-5)


def test_gradcheck_gather_mm():
    feat = torch.randn(10, 6, dtype=torch.float64)
    rows = torch.tensor([0, 3, 3, 9, 1])
    W = torch.randn(6, 4, dtype=torch.float64, requires_grad=True)
    b = torch.randn(4, dtype=torch.float64, requires_grad=True)
    assert gradcheck(lambda w, bb: gather_mm(feat, rows, w, bb), (W, b),
                     eps=1e-6, atol=1e-5)
