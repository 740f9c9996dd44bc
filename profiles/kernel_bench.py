#!/usr/bin/env python3
"""Per-kernel A/B microbenchmarks: each HIP kernel vs the equivalent eager
torch composition, on realistic bench-shaped inputs. Run on an MI355X:

    python profiles/kernel_bench.py [--iters 200]

Within-run interleaved timing (guide §5.4 rule 24): variants alternate in
one process; medians reported.
"""
import argparse
import os as _os
import statistics
import sys as _sys

_sys.path.insert(0, _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), ".."))

import torch


def timeit(fns, iters, warmup=20):
    """Interleaved A/B timing; returns {name: median_ms}."""
    names = list(fns)
    for _ in range(warmup):
        for n in names:
            fns[n]()
    torch.cuda.synchronize()
    samples = {n: [] for n in names}
    for _ in range(iters):
        for n in names:
            s = torch.cuda.Event(enable_timing=True)
            e = torch.cuda.Event(enable_timing=True)
            s.record()
            fns[n]()
            e.record()
            e.synchronize()
            samples[n].append(s.elapsed_time(e))
    return {n: statistics.median(v) for n, v in samples.items()}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iters", type=int, default=100)
    args = p.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    from dgl_operator_amd.graph import rmat_graph
    from dgl_operator_amd.ops import backend
    from dgl_operator_amd.ops.sampling import (
        CompactionWorkspace,
        sample_block_fused,
        sample_neighbors,
        to_block,
    )
    from dgl_operator_amd.ops.spmm import _edge_dst, spmm_raw

    ext = backend.load_extension(required=True)
    torch.manual_seed(0)

    # bench-shaped block: 16K dst, 160K edges, feat table 2.45M x 100
    g = rmat_graph(2_449_029, 10_000_000, seed=1, device=dev)  # structure only
    indptr, indices, _ = g.csc()
    feat = torch.randn(2_449_029, 100, device=dev)
    W = torch.randn(100, 16, device=dev)
    seeds = torch.randperm(2_449_029, device=dev)[:16_000]
    ws1 = CompactionWorkspace(2_449_029, dev)
    ws2 = CompactionWorkspace(2_449_029, dev)

    results = {}

    # 1. fused sample+compact vs unfused chain
    results["sample_block(16K seeds, fanout 10)"] = timeit({
        "fused": lambda: sample_block_fused(indptr, indices, ws1, seeds, 10,
                                            seed=3),
        "unfused": lambda: to_block(
            seeds, *sample_neighbors(indptr, indices, seeds, 10, seed=3), ws2
        ),
    }, args.iters)

    # 2. SpMM vs torch composition (index_select + index_add)
    blk = sample_block_fused(indptr, indices, ws1, seeds, 10, seed=4)
    bi, bx = blk.csc_indptr, blk.csc_indices
    x16 = torch.randn(blk.num_src_nodes, 16, device=dev)
    dst = _edge_dst(bi)

    def torch_spmm():
        msg = x16[bx]
        out = torch.zeros(blk.num_dst_nodes, 16, device=dev)
        out.index_add_(0, dst, msg)
        return out

    results["block SpMM sum (160K edges, F=16)"] = timeit({
        "hip": lambda: spmm_raw(bi, bx, x16, None, False),
        "torch": torch_spmm,
    }, args.iters)

    # 3. gather_mm vs index_select + mm
    rows = blk.srcdata_nids
    results["gather+project (135Kx100 -> 16)"] = timeit({
        "gather_mm(MFMA)": lambda: ext.gather_mm(feat, rows, W, None),
        "index_select+mm": lambda: feat[rows] @ W,
    }, args.iters)

    # 4. edge softmax vs torch reference
    scores = torch.randn(bx.numel(), 4, device=dev)

    def torch_esm():
        m = torch.full((blk.num_dst_nodes, 4), float("-inf"), device=dev)
        m = m.index_reduce_(0, dst, scores, "amax")
        ex = torch.exp(scores - m[dst])
        s = torch.zeros(blk.num_dst_nodes, 4, device=dev)
        s.index_add_(0, dst, ex)
        return ex / s.clamp_min(1e-30)[dst]

    results["edge_softmax (160K x 4 heads)"] = timeit({
        "hip": lambda: ext.edge_softmax_fwd(bi, scores),
        "torch": torch_esm,
    }, args.iters)

    # 5. fused KGE TransE neg vs broadcast
    B, C, NEG, D = 1024, 16, 256, 400
    base = torch.randn(C, B // C, D, device=dev)
    negs = torch.randn(C, NEG, D, device=dev)

    def bcast():
        d = base.unsqueeze(2) - negs.unsqueeze(1)
        return 12.0 - torch.norm(d, dim=-1)

    results["TransE neg (1024x256 negs, d400)"] = timeit({
        "fused": lambda: ext.pdist_neg_fwd(base, negs, 2, 12.0),
        "broadcast": bcast,
    }, args.iters)

    print(f"{'case':42s} " + "  ".join(f"{k}" for k in ("variants",)))
    for case, r in results.items():
        parts = "  ".join(f"{k}={v:.3f}ms" for k, v in r.items())
        ks = list(r.values())
        speedup = ks[1] / ks[0] if ks[0] > 0 else float("inf")
        print(f"{case:42s} {parts}  (x{speedup:.2f})")


if __name__ == "__main__":
    main()
