#!/usr/bin/env python3
"""Partitioner entry point (dglrun Phase 1) — generates/loads a graph and
writes the partition layout. Reference:
/root/reference/examples/GraphSAGE_dist/code/load_and_partition_graph.py
(which downloads ogbn-products; offline here, so --dataset rmat synthesizes
the same shape; --dataset file loads a saved COO .pt).
"""
from __future__ import annotations

import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), "..", ".."))



import argparse

import torch


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--graph-name", default="ogbn-products")
    p.add_argument("--num-partitions", type=int, required=True)
    p.add_argument("--output", required=True)
    p.add_argument("--dataset-url", default="",
                   help="dataset archive URL (reference manifest parity, "
                        "GraphSAGE_dist.yaml:30-31); offline environments "
                        "log it and fall back to --dataset synthesis")
    p.add_argument("--dataset", default="rmat",
                   help="rmat | path to a .pt with {src,dst,feat,label}")
    p.add_argument("--nodes", type=int, default=2_449_029)
    p.add_argument("--edges", type=int, default=61_859_140)
    p.add_argument("--feat", type=int, default=100)
    p.add_argument("--classes", type=int, default=47)
    p.add_argument("--algorithm", default="ldg", choices=["ldg", "range", "random"])
    p.add_argument("--train-fraction", type=float, default=0.1)
    # the reference partitions with balance_ntypes=train + balance_edges
    # (load_and_partition_graph.py:124-127)
    p.add_argument("--no-balance-train", dest="balance_train",
                   action="store_false")
    p.add_argument("--no-balance-edges", dest="balance_edges",
                   action="store_false")
    args = p.parse_args()

    from dgl_operator_amd.graph import Graph, partition_graph, rmat_graph

    if args.dataset == "rmat":
        g = rmat_graph(args.nodes, args.edges, num_feats=args.feat,
                       num_classes=args.classes, seed=0)
    else:
        d = torch.load(args.dataset, weights_only=True)
        g = Graph(d["src"], d["dst"])
        if "feat" in d:
            g.ndata["feat"] = d["feat"]
        if "label" in d:
            g.ndata["label"] = d["label"]
    gen = torch.Generator().manual_seed(42)
    g.ndata["train_mask"] = (
        torch.rand(g.num_nodes, generator=gen) < args.train_fraction
    )
    spec = partition_graph(g, args.graph_name, args.num_partitions, args.output,
                           algorithm=args.algorithm,
                           balance_train=args.balance_train,
                           balance_edges=args.balance_edges)
    print(f"partitioned {args.graph_name}: {spec.num_parts} parts, "
          f"boundaries {spec.boundaries}")


if __name__ == "__main__":
    main()
