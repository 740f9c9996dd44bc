#!/usr/bin/env python3
"""LDG partitioner scale benchmark — papers100M-shape (BASELINE config #3:
ogbn-papers100M has 111,059,956 nodes / 1.615B directed edges; the edge
count here is scaled to fit the 62 GB CPU box, flagged in the output).

Measures: R-MAT generation, CSR+CSC build, ldg_partition wall time
(chunk-parallel stream + refinement passes), per-part balance, and
edge-cut vs a range partition of the same graph.

Run:  python profiles/ldg_scale_bench.py [--nodes N] [--edges E] [--parts P]
Writes a markdown summary to profiles/ldg_scale.md.
"""
import argparse
import os
import resource
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def rss_gb():
    return resource.getrusage(resource.RUSAGE_SELF).ru_maxrss / 1e6


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--nodes", type=int, default=111_059_956)
    p.add_argument("--edges", type=int, default=800_000_000)
    p.add_argument("--parts", type=int, default=8)
    p.add_argument("--out", default=os.path.join(
        os.path.dirname(os.path.abspath(__file__)), "ldg_scale.md"))
    args = p.parse_args()

    from dgl_operator_amd.graph.rmat import rmat_edges
    from dgl_operator_amd.ops import backend

    ext = backend.load_extension(required=True)
    torch.manual_seed(0)
    lines = [
        "# LDG partitioner at papers100M scale",
        "",
        f"Graph: R-MAT, {args.nodes:,} nodes (ogbn-papers100M node count), "
        f"{args.edges:,} directed edges "
        f"(papers100M has 1.615B; scaled to the 62 GB CPU box), "
        f"{args.parts} parts.",
        f"Host: {os.cpu_count()} CPU cores, torch {torch.__version__}.",
        "",
    ]

    t0 = time.time()
    src, dst = rmat_edges(args.nodes, args.edges, seed=11)
    t_gen = time.time() - t0
    E = src.numel()
    lines.append(f"- R-MAT generation: {t_gen:.0f} s ({E:,} edges after "
                 f"self-loop drop), rss {rss_gb():.1f} GB")
    print(lines[-1], flush=True)

    def build_csr(row, col):
        deg = torch.bincount(row, minlength=args.nodes)
        indptr = torch.zeros(args.nodes + 1, dtype=torch.int64)
        torch.cumsum(deg, 0, out=indptr[1:])
        order = torch.argsort(row)
        return indptr, col[order].contiguous()

    t0 = time.time()
    indptr, indices = build_csr(src, dst)
    cindptr, cindices = build_csr(dst, src)
    t_csr = time.time() - t0
    lines.append(f"- CSR+CSC build: {t_csr:.0f} s, rss {rss_gb():.1f} GB")
    print(lines[-1], flush=True)

    t0 = time.time()
    assign = ext.ldg_partition(indptr, indices, cindptr, cindices,
                               args.parts, None, True)
    t_ldg = time.time() - t0
    lines.append(f"- ldg_partition ({args.parts} parts): {t_ldg:.0f} s "
                 f"(chunk-parallel stream + refinement), rss "
                 f"{rss_gb():.1f} GB")
    print(lines[-1], flush=True)

    sizes = torch.bincount(assign, minlength=args.parts)
    cut = float((assign[src] != assign[dst]).float().mean())
    rng = torch.clamp(src.new_empty(0), 0, 0)  # placeholder no-op
    bounds = torch.arange(args.parts + 1, dtype=torch.int64) * args.nodes \
        // args.parts
    range_assign_src = torch.bucketize(src, bounds[1:-1], right=True)
    range_assign_dst = torch.bucketize(dst, bounds[1:-1], right=True)
    range_cut = float((range_assign_src != range_assign_dst).float().mean())
    imb = float(sizes.max()) / (args.nodes / args.parts)
    lines += [
        f"- part sizes: {[int(x) for x in sizes]} "
        f"(max imbalance {imb:.3f}; cap 1.05)",
        f"- edge-cut: LDG {cut:.3f} vs range {range_cut:.3f} "
        f"(scrambled R-MAT: a range partition IS a random partition here)",
        "",
        "LDG phases: growing-chunk stream (neighbor counts parallel over "
        "all cores, placement serial) + adaptive balance-safe refinement "
        "passes (up to 10, stop below 0.2% moves).",
    ]
    print("\n".join(lines[-4:]), flush=True)
    with open(args.out, "w") as f:
        f.write("\n".join(lines) + "\n")
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
