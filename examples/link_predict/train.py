#!/usr/bin/env python3
"""GAT link prediction — parity with the reference tutorial
(/root/reference/examples/link_predict/code/4_link_predict.py): encoder over
the message graph, u_dot_v scores on positive vs negative edge graphs, AUC
report; the GAT encoder exercises the SDDMM + edge-softmax HIP path
(BASELINE config #5)."""
import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), "..", ".."))


import argparse

import torch
import torch.nn.functional as F


def auc_score(pos, neg):
    scores = torch.cat([pos, neg])
    labels = torch.cat([torch.ones_like(pos), torch.zeros_like(neg)])
    order = torch.argsort(scores)
    ranks = torch.empty_like(order, dtype=torch.float)
    ranks[order] = torch.arange(1, scores.numel() + 1, dtype=torch.float,
                                device=scores.device)
    n_pos, n_neg = pos.numel(), neg.numel()
    return float(
        (ranks[labels == 1].sum() - n_pos * (n_pos + 1) / 2) / (n_pos * n_neg)
    )


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--nodes", type=int, default=2708)
    p.add_argument("--edges", type=int, default=10556)
    p.add_argument("--feat", type=int, default=64)
    p.add_argument("--hidden", type=int, default=16)
    p.add_argument("--heads", type=int, default=4)
    p.add_argument("--epochs", type=int, default=60)
    args = p.parse_args()

    from dgl_operator_amd.graph import Graph, rmat_graph
    from dgl_operator_amd.models import GATLinkPredictor

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    g = rmat_graph(args.nodes, args.edges, num_feats=args.feat, seed=0,
                   device=dev)
    src, dst = g.edges()
    E = g.num_edges
    perm = torch.randperm(E, device=dev)
    n_test = E // 10
    test_e, train_e = perm[:n_test], perm[n_test:]
    msg_g = Graph(src[train_e], dst[train_e], g.num_nodes)
    gen = torch.Generator(device=dev).manual_seed(2)

    def neg_graph(n):
        return Graph(
            torch.randint(0, g.num_nodes, (n,), generator=gen, device=dev),
            torch.randint(0, g.num_nodes, (n,), generator=gen, device=dev),
            g.num_nodes,
        )

    pos_g = Graph(src[train_e], dst[train_e], g.num_nodes)
    test_pos_g = Graph(src[test_e], dst[test_e], g.num_nodes)
    model = GATLinkPredictor(args.feat, args.hidden, num_heads=args.heads).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    x = g.ndata["feat"]
    for epoch in range(args.epochs):
        neg_g = neg_graph(train_e.numel())
        pos_s, neg_s = model(msg_g, pos_g, neg_g, x)
        loss = F.binary_cross_entropy_with_logits(
            torch.cat([pos_s, neg_s]),
            torch.cat([torch.ones_like(pos_s), torch.zeros_like(neg_s)]),
        )
        opt.zero_grad(); loss.backward(); opt.step()
        if epoch % 20 == 0 or epoch == args.epochs - 1:
            with torch.no_grad():
                h = model.encode(msg_g, x)
                tp = model.pred(test_pos_g, h)
                tn = model.pred(neg_graph(n_test), h)
            print(f"epoch {epoch:03d} loss {loss:.4f} "
                  f"test AUC {auc_score(tp, tn):.3f}", flush=True)


if __name__ == "__main__":
    main()
