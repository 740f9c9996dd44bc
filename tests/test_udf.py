"""K6: DGL-style fn API and the UDF degree-bucketing fallback."""
import torch

import dgl_operator_amd.fn as fn
from dgl_operator_amd.graph import rmat_graph


def make_g():
    g = rmat_graph(40, 300, num_feats=6, seed=1)
    g.ndata["h"] = g.ndata["feat"]
    g.edata["w"] = torch.rand(g.num_edges)
    return g


def test_builtin_update_all_and_apply_edges():
    g = make_g()
    g.update_all(fn.copy_u("h", "m"), fn.mean("m", "h_N"))
    assert g.ndata["h_N"].shape == (40, 6)
    g.apply_edges(fn.u_dot_v("h", "h", "score"))
    src, dst = g.edges()
    ref = (g.ndata["h"][src] * g.ndata["h"][dst]).sum(-1)
    assert torch.allclose(g.edata["score"], ref, atol=1e-5)


def test_udf_matches_builtin_u_mul_e_sum():
    g = make_g()
    g.update_all(fn.u_mul_e("h", "w", "m"), fn.sum("m", "hb"))

    def msg(edges):
        return {"m": edges.src["h"] * edges.data["w"].unsqueeze(-1)}

    def red(nodes):
        return {"hu": nodes.mailbox["m"].sum(1)}

    g.update_all(msg, red)
    assert torch.allclose(g.ndata["hb"], g.ndata["hu"], atol=1e-5)


def test_udf_max_reduce():
    g = make_g()

    def msg(edges):
        return {"m": edges.src["h"]}

    def red(nodes):
        return {"hmax": nodes.mailbox["m"].max(1).values}

    g.update_all(msg, red)
    # verify against explicit python
    indptr, indices, _ = g.csc()
    for v in [0, 3, 17]:
        seg = indices[indptr[v] : indptr[v + 1]]
        if seg.numel():
            assert torch.allclose(
                g.ndata["hmax"][v], g.ndata["h"][seg].max(0).values
            )
        else:
            assert torch.all(g.ndata["hmax"][v] == 0)


def test_udf_autograd_flows():
    g = make_g()
    g.ndata["h"] = g.ndata["h"].clone().requires_grad_(True)

    def msg(edges):
        return {"m": edges.src["h"] * 2.0}

    def red(nodes):
        return {"out": nodes.mailbox["m"].mean(1)}

    g.update_all(msg, red)
    g.ndata["out"].sum().backward()
    assert g.ndata["h"].grad is not None


def test_copy_src_alias():
    assert fn.copy_src("h", "m") == fn.copy_u("h", "m")


def test_block_fn_api():
    from dgl_operator_amd.ops import NeighborSampler

    g = rmat_graph(50, 400, num_feats=6, seed=3)
    indptr, indices, _ = g.csc()
    sampler = NeighborSampler(indptr, indices, [4], num_nodes=50)
    inp, out, (blk,) = sampler.sample_blocks(torch.arange(20))
    blk.srcdata["h"] = g.ndata["feat"][blk.srcdata_nids]
    blk.dstdata["h"] = blk.srcdata["h"][: blk.num_dst_nodes]
    blk.update_all(fn.copy_u("h", "m"), fn.mean("m", "h_N"))
    assert blk.dstdata["h_N"].shape == (20, 6)
    # matches direct gspmm
    from dgl_operator_amd.ops import gspmm

    ref = gspmm(blk, "copy_u", "mean", blk.srcdata["h"])
    assert torch.allclose(blk.dstdata["h_N"], ref, atol=1e-6)
    blk.apply_edges(fn.u_dot_v("h", "h", "score"))
    assert blk.edata["score"].shape == (blk.num_edges,)


def test_u_add_v_builtin():
    """fn.u_add_v (GAT attention-logit spelling) on Graphs and Blocks
    matches the explicit gather-add; Graph edata lands in original edge
    order, Block edata in csc order like the other edge builtins."""
    import dgl_operator_amd.fn as fn
    from dgl_operator_amd.graph import Block, Graph

    g = Graph(torch.tensor([0, 2, 1]), torch.tensor([1, 0, 2]), 3)
    g.ndata["el"] = torch.tensor([1.0, 2.0, 3.0])
    g.ndata["er"] = torch.tensor([10.0, 20.0, 30.0])
    g.apply_edges(fn.u_add_v("el", "er", "e"))
    s, d = g.edges()
    assert torch.equal(g.edata["e"], g.ndata["el"][s] + g.ndata["er"][d])

    blk = Block(torch.tensor([0, 1, 3]), torch.tensor([1, 0, 2]), 4, 2,
                srcdata_nids=torch.tensor([7, 8, 9, 6]))
    blk.srcdata["el"] = torch.tensor([1.0, 2.0, 3.0, 4.0])
    blk.dstdata["er"] = torch.tensor([5.0, 6.0])
    blk.apply_edges(fn.u_add_v("el", "er", "e"))
    # csc order: dst0 edges [1], dst1 edges [0, 2]
    expect = torch.tensor([2.0 + 5.0, 1.0 + 6.0, 3.0 + 6.0])
    assert torch.equal(blk.edata["e"], expect)


def test_fn_max_reducer():
    """fn.max builtin: per-dst max over in-edge messages, zero rows for
    zero-degree dsts (DGL semantics), gradients flow to argmax sources."""
    import dgl_operator_amd.fn as fn
    from dgl_operator_amd.graph import Graph

    g = Graph(torch.tensor([0, 1, 2, 0]), torch.tensor([1, 2, 1, 2]), 4)
    x = torch.tensor([[1.0, 9.0], [5.0, 2.0], [3.0, 3.0], [0.0, 0.0]],
                     requires_grad=True)
    g.ndata["h"] = x
    g.update_all(fn.copy_u("h", "m"), fn.max("m", "out"))
    out = g.ndata["out"]
    # dst1 <- {src0, src2}: max = [3, 9]; dst2 <- {src1, src0}: [5, 9]
    assert torch.equal(out.detach(),
                       torch.tensor([[0.0, 0.0], [3.0, 9.0], [5.0, 9.0],
                                     [0.0, 0.0]]))
    out.sum().backward()
    # argmax sources get gradient 1: src0 col1 twice, src2 col0, src1 col0
    assert torch.equal(x.grad,
                       torch.tensor([[0.0, 2.0], [1.0, 0.0], [1.0, 0.0],
                                     [0.0, 0.0]]))
    # weighted variant
    g2 = Graph(torch.tensor([0, 1]), torch.tensor([2, 2]), 3)
    g2.ndata["h"] = torch.tensor([[2.0], [4.0]])
    g2.edata["w"] = torch.tensor([10.0, 1.0])
    g2.update_all(fn.u_mul_e("h", "w", "m"), fn.max("m", "out"))
    assert torch.equal(g2.ndata["out"],
                       torch.tensor([[0.0], [0.0], [20.0]]))
