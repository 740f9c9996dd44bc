"""GPU numerics tests: every HIP kernel vs the plain-PyTorch fp32 CPU reference."""
import pytest
import torch

pytestmark = pytest.mark.gpu

from dgl_operator_amd.graph import rmat_graph
from dgl_operator_amd.ops import backend
from dgl_operator_amd.ops.spmm import _spmm_ref, spmm_raw
from dgl_operator_amd.ops.sddmm import (
    sddmm_dot_raw,
    _edge_softmax_ref,
    _EdgeSoftmax,
)
from dgl_operator_amd.ops.segment import _segment_ref, segment_reduce
from dgl_operator_amd.ops.sampling import sample_neighbors
from dgl_operator_amd.ops.adagrad import sparse_adagrad_update
from dgl_operator_amd.ops.spmm import _edge_dst


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    assert backend.has_extension(), "HIP extension must be built on GPU boxes"
    return torch.device("cuda:0")


@pytest.fixture(scope="module")
def big_graph():
    # power-law graph large enough to hit long rows and empty rows
    return rmat_graph(50_000, 600_000, seed=11)


@pytest.mark.parametrize("F", [1, 3, 16, 100, 256])
@pytest.mark.parametrize("mean", [False, True])
def test_spmm_copy_u(dev, big_graph, F, mean):
    g = big_graph
    indptr, indices, _ = g.csc()
    x = torch.randn(g.num_nodes, F)
    ref = _spmm_ref(indptr, indices, x, None, mean)
    out = spmm_raw(indptr.to(dev), indices.to(dev), x.to(dev), None, mean)
    assert torch.allclose(out.cpu(), ref, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("F", [16, 100])
def test_spmm_u_mul_e(dev, big_graph, F):
    g = big_graph
    indptr, indices, _ = g.csc()
    x = torch.randn(g.num_nodes, F)
    w = torch.rand(g.num_edges)
    ref = _spmm_ref(indptr, indices, x, w, False)
    out = spmm_raw(indptr.to(dev), indices.to(dev), x.to(dev), w.to(dev), False)
    assert torch.allclose(out.cpu(), ref, atol=1e-4, rtol=1e-4)


def test_spmm_multihead_weight(dev, big_graph):
    g = big_graph
    H, D = 4, 8
    indptr, indices, _ = g.csc()
    x = torch.randn(g.num_nodes, H, D)
    w = torch.rand(g.num_edges, H)
    ref = _spmm_ref(indptr, indices, x, w, False)
    out = spmm_raw(indptr.to(dev), indices.to(dev), x.to(dev), w.to(dev), False)
    assert torch.allclose(out.cpu(), ref, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("shape", [(32,), (4, 8), (1, 100)])
def test_sddmm_dot(dev, big_graph, shape):
    g = big_graph
    src, dst = g.edges()
    n = g.num_nodes
    fu = torch.randn(n, *shape)
    fv = torch.randn(n, *shape)
    ref = (fu[src] * fv[dst]).sum(-1)
    out = sddmm_dot_raw(src.to(dev), dst.to(dev), fu.to(dev), fv.to(dev))
    assert torch.allclose(out.cpu(), ref, atol=1e-4, rtol=1e-4)


@pytest.mark.parametrize("H", [1, 4])
def test_edge_softmax_fwd_bwd(dev, big_graph, H):
    g = big_graph
    indptr, _, _ = g.csc()
    E = g.num_edges
    scores = torch.randn(E, H) if H > 1 else torch.randn(E)
    ref = _edge_softmax_ref(indptr, scores)
    ext = backend.load_extension(required=True)
    out = ext.edge_softmax_fwd(indptr.to(dev), scores.to(dev))
    assert torch.allclose(out.cpu(), ref, atol=1e-5, rtol=1e-4)
    # backward
    grad = torch.randn_like(scores)
    dst = _edge_dst(indptr)
    nn = indptr.numel() - 1
    acc = torch.zeros((nn,) + ref.shape[1:])
    acc.index_add_(0, dst, ref * grad)
    ref_gin = ref * (grad - acc[dst])
    gin = ext.edge_softmax_bwd(indptr.to(dev), out, grad.to(dev))
    assert torch.allclose(gin.cpu(), ref_gin, atol=1e-5, rtol=1e-4)


@pytest.mark.parametrize("F", [1, 7, 64])
def test_segment_reduce(dev, F):
    offsets = torch.tensor([0, 5, 5, 100, 228, 1000])
    x = torch.randn(1000, F)
    for mean in (False, True):
        ref = _segment_ref(offsets, x, mean)
        out = segment_reduce(offsets.to(dev), x.to(dev), "mean" if mean else "sum")
        assert torch.allclose(out.cpu(), ref, atol=1e-4, rtol=1e-4)


def test_sample_neighbors_gpu(dev, big_graph):
    from collections import Counter

    g = big_graph
    indptr, indices, _ = g.csc()
    seeds = torch.randperm(g.num_nodes)[:2000]
    dip, dix, ds = indptr.to(dev), indices.to(dev), seeds.to(dev)
    nbrs, counts = sample_neighbors(dip, dix, ds, fanout=25, seed=7)
    nbrs, counts = nbrs.cpu(), counts.cpu()
    deg = (indptr[1:] - indptr[:-1])[seeds]
    assert torch.equal(counts, torch.minimum(deg, torch.full_like(deg, 25)))
    off = 0
    for i in range(seeds.numel()):
        c = int(counts[i])
        mine = Counter(nbrs[off : off + c].tolist())
        off += c
        s = seeds[i]
        truth = Counter(indices[indptr[s] : indptr[s + 1]].tolist())
        for nid, k in mine.items():
            assert truth[nid] >= k
    assert off == nbrs.numel()
    # determinism
    nbrs2, counts2 = sample_neighbors(dip, dix, ds, fanout=25, seed=7)
    assert torch.equal(nbrs, nbrs2.cpu())
    # different seed -> different draw (statistically)
    nbrs3, _ = sample_neighbors(dip, dix, ds, fanout=25, seed=8)
    assert not torch.equal(nbrs, nbrs3.cpu())


def test_sample_neighbors_replace_gpu(dev, big_graph):
    g = big_graph
    indptr, indices, _ = g.csc()
    seeds = torch.arange(1000)
    nbrs, counts = sample_neighbors(
        indptr.to(dev), indices.to(dev), seeds.to(dev), fanout=10, replace=True,
        seed=3,
    )
    deg = (indptr[1:] - indptr[:-1])[seeds]
    expect = torch.where(deg > 0, torch.full_like(deg, 10), torch.zeros_like(deg))
    assert torch.equal(counts.cpu(), expect)


def test_sparse_adagrad_gpu(dev):
    N, D, B = 5000, 400, 1200
    emb = torch.randn(N, D)
    state = torch.rand(N)
    ids = torch.randint(0, N, (B,))  # duplicates likely
    grad = torch.randn(B, D)
    ref_emb, ref_state = emb.clone(), state.clone()
    sparse_adagrad_update(ref_emb, ref_state, ids, grad, lr=0.1)
    demb, dstate = emb.to(dev), state.to(dev)
    sparse_adagrad_update(demb, dstate, ids.to(dev), grad.to(dev), lr=0.1)
    assert torch.allclose(dstate.cpu(), ref_state, atol=1e-4, rtol=1e-4)
    assert torch.allclose(demb.cpu(), ref_emb, atol=1e-3, rtol=1e-3)


def test_graphsage_end_to_end_gpu(dev):
    """Full minibatch train step on GPU matches CPU reference numerics."""
    import torch.nn.functional as F

    from dgl_operator_amd.models import GraphSAGE
    from dgl_operator_amd.ops import NeighborSampler

    g = rmat_graph(5000, 50_000, num_feats=100, num_classes=47, seed=2)
    torch.manual_seed(0)
    model_cpu = GraphSAGE(100, 16, 47, n_layers=2, dropout=0.0)
    model_gpu = GraphSAGE(100, 16, 47, n_layers=2, dropout=0.0).to(dev)
    model_gpu.load_state_dict(model_cpu.state_dict())

    indptr, indices, _ = g.csc()
    sampler_gpu = NeighborSampler(
        indptr.to(dev), indices.to(dev), [10, 25], num_nodes=g.num_nodes
    )
    seeds = torch.arange(1000)
    inp, out_nodes, blocks = sampler_gpu.sample_blocks(seeds.to(dev))
    x = g.ndata["feat"].to(dev)[inp]
    y = g.ndata["label"].to(dev)[out_nodes]
    loss_gpu = F.cross_entropy(model_gpu(blocks, x), y)
    loss_gpu.backward()

    # replay the SAME blocks on CPU through the reference path
    blocks_cpu = [b.to("cpu") for b in blocks]
    x_cpu, y_cpu = x.cpu(), y.cpu()
    loss_cpu = F.cross_entropy(model_cpu(blocks_cpu, x_cpu), y_cpu)
    loss_cpu.backward()
    assert torch.allclose(loss_gpu.cpu(), loss_cpu, atol=1e-4, rtol=1e-4)
    for (n1, p1), (n2, p2) in zip(
        model_gpu.named_parameters(), model_cpu.named_parameters()
    ):
        assert torch.allclose(p1.grad.cpu(), p2.grad, atol=1e-3, rtol=1e-3), n1


def test_compact_ids_gpu(dev):
    from dgl_operator_amd.ops.sampling import CompactionWorkspace

    N = 10_000
    ws = CompactionWorkspace(N, dev)
    seeds = torch.randperm(N, device=dev)[:500]
    nbrs = torch.randint(0, N, (20_000,), device=dev)
    srcdata, local = ws.relabel(seeds, nbrs)
    # seeds first, in order
    assert torch.equal(srcdata[:500], seeds)
    # translation is consistent: srcdata[local[i]] == nbrs[i]
    assert torch.equal(srcdata[local], nbrs)
    # srcdata ids are unique
    assert torch.unique(srcdata).numel() == srcdata.numel()
    # set of new nodes matches CPU reference
    ref_new = torch.unique(nbrs.cpu()[~torch.isin(nbrs.cpu(), seeds.cpu())])
    got_new = srcdata[500:].cpu().sort().values
    assert torch.equal(got_new, ref_new)
    # table fully reset
    assert int((ws.table != -1).sum()) == 0
    # reusable: second call works
    srcdata2, local2 = ws.relabel(seeds, nbrs)
    assert srcdata2.numel() == srcdata.numel()
    assert torch.equal(srcdata2[local2], nbrs)


@pytest.mark.parametrize("name", ["TransE_l1", "TransE_l2", "RotatE"])
@pytest.mark.parametrize("neg_head", [False, True])
def test_kge_fused_neg_matches_cpu(dev, name, neg_head):
    """Fused HIP pairwise neg-score (fwd+bwd) vs the CPU broadcast reference."""
    from dgl_operator_amd.ops import get_score_func

    C, c, n, D = 3, 8, 16, 32
    Dr = D // 2 if name == "RotatE" else D
    fn = get_score_func(name, gamma=10.0)
    head = torch.randn(C, c, D)
    rel = torch.randn(C, c, Dr)
    negs = torch.randn(C, n, D)

    h_cpu = head.clone().requires_grad_(True)
    r_cpu = rel.clone().requires_grad_(True)
    n_cpu = negs.clone().requires_grad_(True)
    out_cpu = fn.neg(h_cpu, r_cpu, n_cpu, neg_head=neg_head)
    gout = torch.randn_like(out_cpu)
    out_cpu.backward(gout)

    h_gpu = head.to(dev).requires_grad_(True)
    r_gpu = rel.to(dev).requires_grad_(True)
    n_gpu = negs.to(dev).requires_grad_(True)
    out_gpu = fn.neg(h_gpu, r_gpu, n_gpu, neg_head=neg_head)
    out_gpu.backward(gout.to(dev))

    assert torch.allclose(out_gpu.cpu(), out_cpu, atol=1e-4, rtol=1e-4)
    assert torch.allclose(h_gpu.grad.cpu(), h_cpu.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(r_gpu.grad.cpu(), r_cpu.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(n_gpu.grad.cpu(), n_cpu.grad, atol=1e-3, rtol=1e-3)


def test_sample_block_fused_matches_unfused(dev, big_graph):
    """Fused single-sync sample_block == sample_neighbors + to_block."""
    from dgl_operator_amd.ops.sampling import (
        CompactionWorkspace,
        sample_block_fused,
        sample_neighbors,
        to_block,
    )

    g = big_graph
    indptr, indices, _ = g.csc()
    dip, dix = indptr.to(dev), indices.to(dev)
    seeds = torch.randperm(g.num_nodes, device=dev)[:1500]
    ws1 = CompactionWorkspace(g.num_nodes, dev)
    ws2 = CompactionWorkspace(g.num_nodes, dev)

    nbrs, counts = sample_neighbors(dip, dix, seeds, fanout=10, seed=42)
    blk_a = to_block(seeds, nbrs, counts, ws1)
    blk_b = sample_block_fused(dip, dix, ws2, seeds, fanout=10, seed=42)

    assert blk_a.num_dst_nodes == blk_b.num_dst_nodes
    assert blk_a.num_src_nodes == blk_b.num_src_nodes
    assert torch.equal(blk_a.csc_indptr, blk_b.csc_indptr)
    # same parent node per edge position (local ids may differ in order)
    assert torch.equal(
        blk_a.srcdata_nids[blk_a.csc_indices],
        blk_b.srcdata_nids[blk_b.csc_indices],
    )
    # same set of source nodes; seeds first
    assert torch.equal(blk_b.srcdata_nids[:1500], seeds)
    assert torch.equal(
        blk_a.srcdata_nids.sort().values, blk_b.srcdata_nids.sort().values
    )
    # workspace reset
    assert int((ws2.table != -1).sum()) == 0


@pytest.mark.parametrize("dtype", [torch.bfloat16, torch.float16])
def test_spmm_low_precision(dev, big_graph, dtype):
    """bf16/fp16 SpMM (fp32 accumulation) tracks the fp32 reference."""
    g = big_graph
    indptr, indices, _ = g.csc()
    x = torch.randn(g.num_nodes, 64)
    ref = _spmm_ref(indptr, indices, x, None, True)
    out = spmm_raw(
        indptr.to(dev), indices.to(dev), x.to(dev).to(dtype), None, True
    )
    assert out.dtype == dtype
    assert torch.allclose(out.float().cpu(), ref, atol=0.15, rtol=0.05)


@pytest.mark.parametrize("weighted", [False, True])
def test_spmm_backward_scatter_matches_cpu(dev, big_graph, weighted):
    """GPU backward (scatter-atomic) == CPU backward (CSR transpose)."""
    from dgl_operator_amd.ops import gspmm

    g = big_graph
    F = 32
    x_cpu = torch.randn(g.num_nodes, F, requires_grad=True)
    w_cpu = torch.rand(g.num_edges, requires_grad=True) if weighted else None
    if weighted:
        out_cpu = gspmm(g, "u_mul_e", "mean", x_cpu, w_cpu)
    else:
        out_cpu = gspmm(g, "copy_u", "mean", x_cpu)
    gout = torch.randn_like(out_cpu)
    out_cpu.backward(gout)

    gg = g.to(dev)
    x_gpu = x_cpu.detach().to(dev).requires_grad_(True)
    w_gpu = (w_cpu.detach().to(dev).requires_grad_(True) if weighted else None)
    if weighted:
        out_gpu = gspmm(gg, "u_mul_e", "mean", x_gpu, w_gpu)
    else:
        out_gpu = gspmm(gg, "copy_u", "mean", x_gpu)
    out_gpu.backward(gout.to(dev))
    assert torch.allclose(out_gpu.cpu(), out_cpu, atol=1e-4, rtol=1e-4)
    assert torch.allclose(x_gpu.grad.cpu(), x_cpu.grad, atol=1e-3, rtol=1e-3)
    if weighted:
        assert torch.allclose(w_gpu.grad.cpu(), w_cpu.grad, atol=1e-3,
                              rtol=1e-3)


def test_compact_ids_heavy_duplicates(dev):
    """Atomic-claim stress: every neighbor is one of 3 hot ids."""
    from dgl_operator_amd.ops.sampling import CompactionWorkspace

    ws = CompactionWorkspace(1000, dev)
    seeds = torch.tensor([7, 8], device=dev)
    nbrs = torch.tensor([5, 6, 9], device=dev).repeat(100_000)
    srcdata, local = ws.relabel(seeds, nbrs)
    assert srcdata.numel() == 5  # 2 seeds + 3 unique neighbors
    assert torch.equal(srcdata[:2].cpu(), torch.tensor([7, 8]))
    assert torch.equal(srcdata[local], nbrs)
    assert int((ws.table != -1).sum()) == 0


def test_gather_rows_kernel(dev):
    ext = backend.load_extension(required=True)
    feat = torch.randn(1000, 100, device=dev)
    gids = torch.randint(50, 1050, (3000,), device=dev)
    # offset form
    out = ext.gather_rows(feat, gids, None, 50)
    assert torch.equal(out, feat[gids - 50])
    # map form
    mp_full = torch.full((2000,), -1, dtype=torch.int64, device=dev)
    mp_full[:1050] = torch.randint(0, 1000, (1050,), device=dev)
    out2 = ext.gather_rows(feat, gids.clamp(max=1049), mp_full, 0)
    assert torch.equal(out2, feat[mp_full[gids.clamp(max=1049)]])
    # odd width (VEC=1 path) + bf16 (VEC=8)
    f2 = torch.randn(100, 7, device=dev)
    ids = torch.arange(100, device=dev)
    assert torch.equal(ext.gather_rows(f2, ids, None, 0), f2)
    f3 = torch.randn(100, 64, device=dev).bfloat16()
    assert torch.equal(ext.gather_rows(f3, ids, None, 0), f3)


def test_gather_mm_mfma(dev):
    """fp32 MFMA gather-GEMM vs CPU reference. ASYMMETRIC weight (guide G9:
    symmetric operands miss transposed fragment layouts)."""
    ext = backend.load_extension(required=True)
    torch.manual_seed(0)
    for M, K, N in [(1000, 100, 16), (64, 4, 16), (37, 7, 5), (130, 128, 16),
                    (5000, 100, 16)]:
        feat = torch.randn(2000, K)
        rows = torch.randint(0, 2000, (M,))
        W = torch.randn(K, N) + torch.arange(N).float() * 0.1  # asymmetric
        b = torch.randn(N)
        ref = feat[rows] @ W + b
        out = ext.gather_mm(feat.to(dev), rows.to(dev), W.to(dev), b.to(dev))
        assert torch.allclose(out.cpu(), ref, atol=1e-3, rtol=1e-3), (M, K, N)
    # no-bias form
    out = ext.gather_mm(feat.to(dev), rows.to(dev), W.to(dev), None)
    assert torch.allclose(out.cpu(), feat[rows] @ W, atol=1e-3, rtol=1e-3)


def test_sageconv_gatherview_gpu_matches_cpu(dev):
    """Fused MFMA input projection == dense CPU SAGEConv."""
    from dgl_operator_amd.nn import SAGEConv
    from dgl_operator_amd.ops import GatherView

    torch.manual_seed(3)
    g = rmat_graph(3000, 30_000, num_feats=100, seed=5)
    layer = SAGEConv(100, 16, bias=True)
    x = g.ndata["feat"]
    o_cpu = layer(g, x)
    layer_gpu = SAGEConv(100, 16, bias=True).to(dev)
    layer_gpu.load_state_dict(layer.state_dict())
    gg = g.to(dev)
    view = GatherView(x.to(dev), torch.arange(3000, device=dev))
    o_gpu = layer_gpu(gg, view)
    assert torch.allclose(o_gpu.cpu(), o_cpu, atol=1e-3, rtol=1e-3)
    # weight grads match
    go = torch.randn_like(o_cpu)
    o_cpu.backward(go)
    o_gpu.backward(go.to(dev))
    assert torch.allclose(layer_gpu.fc_neigh.weight.grad.cpu(),
                          layer.fc_neigh.weight.grad, atol=1e-2, rtol=1e-2)
    assert torch.allclose(layer_gpu.fc_self.weight.grad.cpu(),
                          layer.fc_self.weight.grad, atol=1e-2, rtol=1e-2)


def test_gat_score_fused_gpu(dev, big_graph):
    from dgl_operator_amd.ops import gat_score

    g = big_graph
    H = 4
    el = torch.randn(g.num_nodes, H)
    er = torch.randn(g.num_nodes, H)
    el_c = el.clone().requires_grad_(True)
    er_c = er.clone().requires_grad_(True)
    s_cpu = gat_score(g, el_c, er_c, 0.2)
    gout = torch.randn_like(s_cpu)
    s_cpu.backward(gout)

    gg = g.to(dev)
    el_g = el.to(dev).requires_grad_(True)
    er_g = er.to(dev).requires_grad_(True)
    s_gpu = gat_score(gg, el_g, er_g, 0.2)
    s_gpu.backward(gout.to(dev))
    assert torch.allclose(s_gpu.cpu(), s_cpu, atol=1e-5, rtol=1e-5)
    assert torch.allclose(el_g.grad.cpu(), el_c.grad, atol=1e-3, rtol=1e-3)
    assert torch.allclose(er_g.grad.cpu(), er_c.grad, atol=1e-3, rtol=1e-3)


def test_bf16_sddmm_and_segment(dev, big_graph):
    """Low-precision paths of the remaining aggregation kernels."""
    g = big_graph
    src, dst = g.edges()
    fu = torch.randn(g.num_nodes, 64)
    ref = (fu[src] * fu[dst]).sum(-1)
    out = sddmm_dot_raw(src.to(dev), dst.to(dev),
                        fu.to(dev).bfloat16(), fu.to(dev).bfloat16())
    assert out.dtype == torch.bfloat16
    assert torch.allclose(out.float().cpu(), ref, atol=2.0, rtol=0.05)
    offsets = torch.tensor([0, 10, 500, 1000])
    x = torch.randn(1000, 32)
    ref = _segment_ref(offsets, x, True)
    out = segment_reduce(offsets.to(dev), x.to(dev).bfloat16(), "mean")
    assert torch.allclose(out.float().cpu(), ref, atol=0.1, rtol=0.05)


def test_spmm_hub_rows(dev):
    """Hub rows (> 256 in-edges) go through the block-parallel LDS kernel;
    outputs must match the CPU reference including the short/long boundary."""
    torch.manual_seed(2)
    # star graph: node 0 has 5000 in-edges; plus a band of short rows
    hub_src = torch.randint(1, 4000, (5000,))
    hub_dst = torch.zeros(5000, dtype=torch.int64)
    band_src = torch.randint(0, 4000, (8000,))
    band_dst = torch.randint(1, 4000, (8000,))
    from dgl_operator_amd.graph import Graph

    g = Graph(torch.cat([hub_src, band_src]), torch.cat([hub_dst, band_dst]),
              4000)
    indptr, indices, _ = g.csc()
    for F in [16, 100, 1024, 7]:
        x = torch.randn(4000, F)
        for mean in (False, True):
            ref = _spmm_ref(indptr, indices, x, None, mean)
            out = spmm_raw(indptr.to(dev), indices.to(dev), x.to(dev), None,
                           mean)
            assert torch.allclose(out.cpu(), ref, atol=1e-3, rtol=1e-3), (
                F, mean)
    # weighted hub
    x = torch.randn(4000, 32)
    w = torch.rand(g.num_edges)
    ref = _spmm_ref(indptr, indices, x, w, True)
    out = spmm_raw(indptr.to(dev), indices.to(dev), x.to(dev), w.to(dev),
                   True)
    assert torch.allclose(out.cpu(), ref, atol=1e-3, rtol=1e-3)


def test_layerwise_inference_gpu(dev):
    """Single-rank GPU layer-wise inference (full-neighbor blocks incl. hub
    rows -> long-row SpMM) == CPU full-graph forward."""
    from dgl_operator_amd.distributed import DistGraph, PartitionBook
    from dgl_operator_amd.models import GraphSAGE
    from dgl_operator_amd.models.graphsage import inference_dist

    torch.manual_seed(5)
    g = rmat_graph(20_000, 400_000, num_feats=32, seed=6)  # has >256-deg hubs
    assert int(g.in_degrees().max()) > 256
    model = GraphSAGE(32, 16, 5, n_layers=2, dropout=0.0)
    book = PartitionBook([0, 20_000], device=dev)
    gg = g.to(dev)
    dg = DistGraph.from_full_graph(gg, book, 0)
    model_gpu = GraphSAGE(32, 16, 5, n_layers=2, dropout=0.0).to(dev)
    model_gpu.load_state_dict(model.state_dict())
    shard = inference_dist(model_gpu, dg, batch_size=3000)
    model.eval()
    with torch.no_grad():
        full = model(g, g.ndata["feat"])
    assert torch.allclose(shard.cpu(), full, atol=2e-3, rtol=2e-3)


def test_halo_cache_single_rank_gpu(dev):
    """Multi-GPU de-risk on one GPU: the ghost-zone halo cache build,
    halo-path fused sampling and pull/pull_view all execute on device and
    agree with the non-halo path (single rank: halo is the identity, so the
    two paths must match exactly)."""
    from dgl_operator_amd.distributed import DistGraph, PartitionBook

    g = rmat_graph(20_000, 200_000, num_feats=16, num_classes=5, seed=9)
    book = PartitionBook([0, 20_000], device=dev)
    gg = g.to(dev)
    dg_halo = DistGraph.from_full_graph(gg, book, 0)
    dg_halo.build_halo_cache(2, feat_keys=("feat", "label"))
    assert dg_halo.halo is not None and dg_halo.halo.hops == 2
    dg_plain = DistGraph.from_full_graph(gg, book, 0)

    seeds = torch.randperm(20_000, device=dev)[:500]
    inp_h, out_h, blocks_h = dg_halo.sample_blocks(seeds, [5, 10], seed=3)
    inp_p, out_p, blocks_p = dg_plain.sample_blocks(seeds, [5, 10], seed=3)
    assert torch.equal(out_h, out_p)
    # identical sampled-edge counts per dst (counts = min(deg, fanout) are
    # order-independent); parent ids are valid graph nodes
    assert torch.equal(blocks_h[-1].csc_indptr, blocks_p[-1].csc_indptr)
    for bh in blocks_h:
        parents = bh.srcdata_nids[bh.csc_indices]
        assert int(parents.min()) >= 0 and int(parents.max()) < 20_000
    # pull (halo feat_map gather kernel) == direct rows
    x = dg_halo.pull("feat", inp_h)
    assert torch.equal(x, gg.ndata["feat"][inp_h])
    y = dg_halo.pull("label", out_h)
    assert torch.equal(y, gg.ndata["label"][out_h])
    # pull_view materializes to the same matrix
    v = dg_halo.pull_view("feat", inp_h)
    xv = v.materialize() if hasattr(v, "materialize") else v
    assert torch.equal(xv, x)


def test_bench_capture_mode_gpu(dev):
    """--capture (whole-step hipGraph replay) produces a sane bench line.

    Some driver/runtime combinations refuse stream capture entirely
    (hipErrorStreamCaptureUnsupported was observed on the round-1 driver
    box); the bench must then fall back to eager WITHOUT corrupting the
    metric. So: capture-enabled runs are validated strictly, an explicit
    clean fallback line is tolerated, and the JSON must be sane either
    way."""
    import json
    import os
    import subprocess
    import sys

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "bench.py", "--capture", "--steps", "4",
         "--warmup", "2", "--nodes", "20000", "--edges", "150000"],
        capture_output=True, text=True, cwd=repo, timeout=600,
    )
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    enabled = "# capture: enabled" in r.stdout
    assert enabled or "# capture: disabled" in r.stdout, r.stdout
    d = json.loads([l for l in r.stdout.splitlines() if l.startswith("{")][-1])
    assert d["value"] > 0
    assert d["config"]["step_mode"] == (
        "hipGraph-captured" if enabled else "eager")
    # edges per step must be plausible: ~batch * (25 + 10*frontier)-ish, and
    # device-side accounting must not count the padded garbage region
    edges_per_step = d["value"] * d["ms_per_step"] / 1000.0
    assert 1000 < edges_per_step < 1000 * 36
