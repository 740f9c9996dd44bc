import torch

from dgl_operator_amd.graph import Graph, Block, batch_graphs, rmat_graph


def small_graph():
    #  0 -> 1, 0 -> 2, 1 -> 2, 2 -> 0, 3 -> 2
    src = torch.tensor([0, 0, 1, 2, 3])
    dst = torch.tensor([1, 2, 2, 0, 2])
    return Graph(src, dst, 4)


def test_csc_structure():
    g = small_graph()
    indptr, indices, eids = g.csc()
    assert indptr.tolist() == [0, 1, 2, 5, 5]
    assert sorted(indices[1:2].tolist()) == [0]
    assert sorted(indices[2:5].tolist()) == [0, 1, 3]
    # eids map back to COO edges
    src, dst = g.edges()
    assert torch.equal(src[eids], indices)


def test_csr_structure():
    g = small_graph()
    indptr, indices, eids = g.csr()
    assert indptr.tolist() == [0, 2, 3, 4, 5]
    assert sorted(indices[0:2].tolist()) == [1, 2]
    src, dst = g.edges()
    assert torch.equal(dst[eids], indices)


def test_degrees():
    g = small_graph()
    assert g.in_degrees().tolist() == [1, 1, 3, 0]
    assert g.out_degrees().tolist() == [2, 1, 1, 1]


def test_self_loops_and_reverse():
    g = small_graph().add_self_loops()
    assert g.num_edges == 9
    assert g.in_degrees().tolist() == [2, 2, 4, 1]
    r = small_graph().reverse()
    assert r.in_degrees().tolist() == [2, 1, 1, 1]


def test_batch_graphs():
    g1 = small_graph()
    g2 = Graph(torch.tensor([0]), torch.tensor([1]), 2)
    bg, sizes = batch_graphs([g1, g2])
    assert bg.num_nodes == 6
    assert bg.num_edges == 6
    assert sizes.tolist() == [4, 2]
    src, dst = bg.edges()
    assert src[-1].item() == 4 and dst[-1].item() == 5


def test_rmat_shape_and_determinism():
    g1 = rmat_graph(1000, 5000, num_feats=8, num_classes=3, seed=7)
    g2 = rmat_graph(1000, 5000, num_feats=8, num_classes=3, seed=7)
    assert g1.num_nodes == 1000
    assert g1.num_edges <= 5000  # self loops dropped
    assert g1.num_edges > 4000
    s1, d1 = g1.edges()
    s2, d2 = g2.edges()
    assert torch.equal(s1, s2) and torch.equal(d1, d2)
    assert torch.equal(g1.ndata["feat"], g2.ndata["feat"])
    # power-law-ish: max degree far above mean
    assert g1.in_degrees().max().item() > 5 * g1.in_degrees().float().mean().item()


def test_block_csr_transpose():
    # block: 3 dst, 5 src; dst0 <- {0,3}, dst1 <- {1,4}, dst2 <- {2}
    indptr = torch.tensor([0, 2, 4, 5])
    indices = torch.tensor([0, 3, 1, 4, 2])
    b = Block(indptr, indices, num_src=5, num_dst=3)
    rindptr, rindices, reids = b.csr()
    assert rindptr.tolist() == [0, 1, 2, 3, 4, 5]
    assert rindices.tolist() == [0, 1, 2, 0, 1]


def test_rmat_chunked_generation_deterministic():
    from dgl_operator_amd.graph.rmat import rmat_edges

    s1, d1 = rmat_edges(500, 5000, seed=3, chunk=1 << 10)
    s2, d2 = rmat_edges(500, 5000, seed=3, chunk=1 << 10)
    assert torch.equal(s1, s2) and torch.equal(d1, d2)
    assert s1.numel() > 4000
    assert int(s1.max()) < 500 and int(d1.max()) < 500


def test_block_device_roundtrip():
    indptr = torch.tensor([0, 2, 3])
    indices = torch.tensor([0, 2, 1])
    b = Block(indptr, indices, num_src=3, num_dst=2,
              srcdata_nids=torch.tensor([5, 7, 9]))
    b.edata["w"] = torch.rand(3)
    b2 = b.to("cpu")
    assert torch.equal(b2.csc_indptr, indptr)
    assert torch.equal(b2.srcdata_nids, b.srcdata_nids)
    assert torch.equal(b2.edata["w"], b.edata["w"])
    assert b2.num_src_nodes == 3 and b2.num_dst_nodes == 2


def test_dgl_api_conveniences():
    """DGL-name surface the reference tutorials use: number_of_*,
    local_scope, remove_edges, to_bidirected, unbatch, NID on blocks."""
    import dgl_operator_amd as doa
    from dgl_operator_amd.graph import (
        batch_graphs, batch_num_edges, to_bidirected, unbatch,
    )

    g = Graph(torch.tensor([0, 1, 2]), torch.tensor([1, 2, 0]), 4)
    assert g.number_of_nodes() == 4 and g.number_of_edges() == 3
    # local_scope rolls back assignments
    g.ndata["h"] = torch.ones(4)
    with g.local_scope():
        g.ndata["tmp"] = torch.zeros(4)
        g.edata["w"] = torch.ones(3)
    assert "tmp" not in g.ndata and "w" not in g.edata
    assert "h" in g.ndata
    # remove_edges drops positions + edata rows
    g.edata["w"] = torch.arange(3.0)
    g2 = g.remove_edges(torch.tensor([1]))
    assert g2.num_edges == 2
    assert torch.equal(g2.edata["w"], torch.tensor([0.0, 2.0]))
    # to_bidirected dedups reverses
    gb = to_bidirected(g)
    assert gb.num_edges == 6
    gb2 = to_bidirected(gb)
    assert gb2.num_edges == 6
    # batch/unbatch round trip
    gs = [Graph(torch.tensor([0]), torch.tensor([1]), 2),
          Graph(torch.tensor([0, 1]), torch.tensor([1, 2]), 3)]
    gs[0].ndata["x"] = torch.tensor([1.0, 2.0])
    gs[1].ndata["x"] = torch.tensor([3.0, 4.0, 5.0])
    bg, bnn = batch_graphs(gs)
    bg.ndata["x"] = torch.cat([gs[0].ndata["x"], gs[1].ndata["x"]])
    assert torch.equal(batch_num_edges(bg, bnn), torch.tensor([1, 2]))
    parts = unbatch(bg, bnn)
    assert len(parts) == 2
    assert parts[1].num_nodes == 3 and parts[1].num_edges == 2
    assert torch.equal(parts[1].ndata["x"], gs[1].ndata["x"])
    s, d = parts[1].edges()
    assert torch.equal(s, torch.tensor([0, 1]))
    # Block NID fields
    from dgl_operator_amd.graph import Block, NID

    blk = Block(torch.tensor([0, 1, 2]), torch.tensor([1, 2]), 3, 2,
                srcdata_nids=torch.tensor([10, 20, 30]))
    assert torch.equal(blk.srcdata[NID], torch.tensor([10, 20, 30]))
    assert torch.equal(blk.dstdata[NID], torch.tensor([10, 20]))
    assert doa.NID == "_ID"


def test_num_nodes_both_spellings():
    """DGL spells num_nodes/num_edges as METHODS; this package as
    properties. Both must work on Graphs and Blocks (verbatim-migrated
    reference code calls g.num_edges())."""
    g = Graph(torch.tensor([0, 1]), torch.tensor([1, 2]), 3)
    assert g.num_nodes == 3 and g.num_nodes() == 3
    assert g.num_edges == 2 and g.num_edges() == 2
    assert isinstance(g.num_nodes + 1, int)
    from dgl_operator_amd.graph import Block

    b = Block(torch.tensor([0, 1, 2]), torch.tensor([1, 2]), 3, 2,
              srcdata_nids=torch.tensor([5, 6, 7]))
    assert b.num_src_nodes == 3 and b.num_src_nodes() == 3
    assert b.num_edges() == 2
