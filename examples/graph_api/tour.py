#!/usr/bin/env python3
"""Graph API tour — parity with the reference's DGLGraph tutorial
(/root/reference/examples/*/code/2_dglgraph.py): constructing graphs,
structure queries, node/edge data, transforms, batching, and device moves."""
import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), "..", ".."))

import torch

from dgl_operator_amd.graph import Graph, batch_graphs, rmat_graph


def main():
    # construction from COO
    src = torch.tensor([0, 0, 1, 2, 3])
    dst = torch.tensor([1, 2, 2, 0, 2])
    g = Graph(src, dst, num_nodes=4)
    print(f"nodes={g.num_nodes} edges={g.num_edges}")

    # structure queries
    print("in_degrees :", g.in_degrees().tolist())
    print("out_degrees:", g.out_degrees().tolist())
    indptr, indices, eids = g.csc()
    print("in-neighbors of node 2:",
          indices[indptr[2]:indptr[3]].tolist())

    # node/edge data
    g.ndata["x"] = torch.randn(4, 8)
    g.edata["w"] = torch.rand(5)
    print("ndata keys:", list(g.ndata), "| edata keys:", list(g.edata))

    # transforms
    g2 = g.add_self_loops()
    print(f"with self loops: {g2.num_edges} edges")
    rg = g.reverse()
    print("reversed in_degrees:", rg.in_degrees().tolist())

    # batching (graph classification input) and its inverse
    from dgl_operator_amd.graph import batch_num_edges, to_bidirected, unbatch

    graphs = [rmat_graph(6, 12, seed=i) for i in range(3)]
    bg, sizes = batch_graphs(graphs)
    print(f"batched: {bg.num_nodes} nodes, batch_num_nodes={sizes.tolist()}, "
          f"batch_num_edges={batch_num_edges(bg, sizes).tolist()}")
    parts = unbatch(bg, sizes)
    print(f"unbatched back into {len(parts)} graphs of "
          f"{[p.num_nodes for p in parts]} nodes")

    # undirected view + isolated scratch space
    bd = to_bidirected(g)
    print(f"to_bidirected: {g.num_edges} -> {bd.num_edges} edges")
    with g.local_scope():
        g.ndata["scratch"] = torch.zeros(g.num_nodes)
    print("scratch survives local_scope:", "scratch" in g.ndata)

    # device moves
    if torch.cuda.is_available():
        gg = g.to("cuda:0")
        print("moved to", gg.device)


if __name__ == "__main__":
    main()
