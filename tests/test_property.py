"""Property-based invariants (hypothesis) for the structural machinery the
GPU kernels rely on: CSC/CSR transposition, sampler draw bounds, compaction
relabeling, partition-book routing, segment reorder round-trips."""
import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from dgl_operator_amd.distributed.dist_graph import (
    _cumsum0,
    _reorder_segments,
)
from dgl_operator_amd.distributed.partition_book import PartitionBook
from dgl_operator_amd.graph import Graph
from dgl_operator_amd.ops.sampling import CompactionWorkspace, _sample_ref


edges_strategy = st.lists(
    st.tuples(st.integers(0, 19), st.integers(0, 19)), min_size=1, max_size=200
)


@settings(max_examples=30, deadline=None)
@given(edges_strategy)
def test_csc_csr_are_transposes(edges):
    src = torch.tensor([e[0] for e in edges])
    dst = torch.tensor([e[1] for e in edges])
    g = Graph(src, dst, 20)
    ci, cx, ce = g.csc()
    ri, rx, re = g.csr()
    assert cx.numel() == rx.numel() == len(edges)
    # edge multiset is preserved in both layouts
    from collections import Counter

    coo = Counter(zip(src.tolist(), dst.tolist()))
    from_csc = Counter()
    for v in range(20):
        for p in range(int(ci[v]), int(ci[v + 1])):
            from_csc[(int(cx[p]), v)] += 1
    from_csr = Counter()
    for u in range(20):
        for p in range(int(ri[u]), int(ri[u + 1])):
            from_csr[(u, int(rx[p]))] += 1
    assert coo == from_csc == from_csr


@settings(max_examples=30, deadline=None)
@given(edges_strategy, st.integers(1, 8), st.integers(0, 100))
def test_sampler_bounds(edges, fanout, seed):
    src = torch.tensor([e[0] for e in edges])
    dst = torch.tensor([e[1] for e in edges])
    g = Graph(src, dst, 20)
    indptr, indices, _ = g.csc()
    seeds = torch.arange(20)
    nbrs, counts = _sample_ref(indptr, indices, seeds, fanout, False, seed)
    deg = g.in_degrees()
    assert torch.equal(counts, torch.minimum(deg, torch.full_like(deg, fanout)))
    assert nbrs.numel() == int(counts.sum())


@settings(max_examples=30, deadline=None)
@given(
    st.lists(st.integers(0, 49), min_size=1, max_size=40, unique=True),
    st.lists(st.integers(0, 49), min_size=0, max_size=100),
)
def test_compaction_invariants(seed_ids, nbr_ids):
    ws = CompactionWorkspace(50, "cpu")
    seeds = torch.tensor(seed_ids)
    nbrs = torch.tensor(nbr_ids, dtype=torch.int64)
    srcdata, local = ws.relabel(seeds, nbrs)
    assert torch.equal(srcdata[: len(seed_ids)], seeds)
    if nbrs.numel():
        assert torch.equal(srcdata[local], nbrs)
    assert torch.unique(srcdata).numel() == srcdata.numel()
    assert int((ws.table != -1).sum()) == 0  # workspace restored


@settings(max_examples=30, deadline=None)
@given(
    st.lists(st.integers(1, 30), min_size=2, max_size=6),
    st.lists(st.integers(0, 10_000), min_size=1, max_size=50),
)
def test_partition_book_routing(sizes, raw_ids):
    bounds = [0]
    for s in sizes:
        bounds.append(bounds[-1] + s)
    book = PartitionBook(bounds)
    ids = torch.tensor([i % bounds[-1] for i in raw_ids])
    owner = book.owner(ids)
    for gid, p in zip(ids.tolist(), owner.tolist()):
        assert bounds[p] <= gid < bounds[p + 1]
    s, perm, counts = book.partition_by_owner(ids)
    assert int(counts.sum()) == ids.numel()
    assert torch.equal(s, ids[perm])
    # owners of the sorted ids are non-decreasing
    so = book.owner(s)
    assert bool((so[1:] >= so[:-1]).all())


@settings(max_examples=30, deadline=None)
@given(st.lists(st.integers(0, 5), min_size=1, max_size=20), st.integers(0, 99))
def test_reorder_segments_roundtrip(counts_list, seed):
    gen = torch.Generator().manual_seed(seed)
    n = len(counts_list)
    counts_orig = torch.tensor(counts_list)
    perm = torch.randperm(n, generator=gen)  # sorted position s -> orig perm[s]
    counts_sorted = counts_orig[perm]
    # payload in sorted order: segment s holds values tagged by orig index
    segs = []
    for s in range(n):
        o = int(perm[s])
        segs.append(torch.full((int(counts_sorted[s]),), o, dtype=torch.int64))
    payload_sorted = torch.cat(segs) if segs else torch.empty(0, dtype=torch.int64)
    out, counts_back = _reorder_segments(payload_sorted, counts_sorted, perm)
    assert torch.equal(counts_back, counts_orig)
    off = _cumsum0(counts_orig)
    for o in range(n):
        seg = out[int(off[o]) : int(off[o + 1])]
        assert bool((seg == o).all())


@settings(max_examples=25, deadline=None)
@given(edges_strategy, st.integers(0, 50))
def test_builtin_equals_udf_everywhere(edges, seed):
    """fn.u_mul_e+mean == the UDF pair on arbitrary multigraphs."""
    import dgl_operator_amd.fn as fn

    gen = torch.Generator().manual_seed(seed)
    src = torch.tensor([e[0] for e in edges])
    dst = torch.tensor([e[1] for e in edges])
    g = Graph(src, dst, 20)
    g.ndata["h"] = torch.randn(20, 3, generator=gen)
    g.edata["w"] = torch.rand(len(edges), generator=gen)
    g.update_all(fn.u_mul_e("h", "w", "m"), fn.mean("m", "hb"))

    def msg(e):
        return {"m": e.src["h"] * e.data["w"].unsqueeze(-1)}

    def red(n):
        return {"hu": n.mailbox["m"].mean(1)}

    g.update_all(msg, red)
    assert torch.allclose(g.ndata["hb"], g.ndata["hu"], atol=1e-5)


def test_multihead_weight_fn_api():
    import dgl_operator_amd.fn as fn
    from dgl_operator_amd.graph import rmat_graph

    g = rmat_graph(30, 200, seed=1)
    H, D = 2, 4
    g.ndata["z"] = torch.randn(30, H, D)
    g.edata["a"] = torch.rand(g.num_edges, H)
    g.update_all(fn.u_mul_e("z", "a", "m"), fn.sum("m", "out"))
    assert g.ndata["out"].shape == (30, H, D)
    # reference
    from dgl_operator_amd.ops.spmm import _spmm_ref

    indptr, indices, eids = g.csc()
    ref = _spmm_ref(indptr, indices, g.ndata["z"], g.edata["a"][eids], False)
    assert torch.allclose(g.ndata["out"], ref, atol=1e-5)
