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
    p.add_argument("--minibatch", action="store_true",
                   help="edge-minibatch training with neighbor-sampled "
                        "blocks (ogbl-citation2-scale graphs; GAT over MFGs)")
    p.add_argument("--batch-edges", type=int, default=1024)
    p.add_argument("--fan-out", type=str, default="10,10")
    p.add_argument("--steps", type=int, default=0,
                   help="minibatch steps per epoch (0 = edges/batch)")
    args = p.parse_args()
    if args.minibatch:
        return main_minibatch(args)

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
    # message graph = graph minus the held-out test edges, exactly the
    # reference tutorial's dgl.remove_edges step (4_link_predict.py)
    msg_g = g.remove_edges(test_e)
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


def main_minibatch(args):
    """Edge-minibatch link prediction: per step sample positive edges + an
    equal number of uniform negatives, build MFG blocks around the involved
    endpoints with the HIP sampler, encode with GAT layers over the blocks,
    and score with u_dot_v on the block-local embeddings. This is the
    at-scale shape of BASELINE config #5 (ogbl-citation2-like)."""
    import time

    import torch.nn.functional as F

    from dgl_operator_amd.graph import rmat_graph
    from dgl_operator_amd.models.gat import GATLinkPredictor
    from dgl_operator_amd.nn import GATConv
    from dgl_operator_amd.ops import NeighborSampler, sddmm_dot
    from dgl_operator_amd.graph.graph import Graph

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    g = rmat_graph(args.nodes, args.edges, num_feats=args.feat, seed=0,
                   device=dev)
    src, dst = g.edges()
    indptr, indices, _ = g.csc()
    fanouts = [int(x) for x in args.fan_out.split(",")]
    sampler = NeighborSampler(indptr, indices, fanouts,
                              num_nodes=g.num_nodes)
    model = GATLinkPredictor(args.feat, args.hidden,
                             num_heads=args.heads).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=5e-3)
    gen = torch.Generator(device=dev).manual_seed(0)
    steps = args.steps or max(1, g.num_edges // args.batch_edges)
    x_all = g.ndata["feat"]
    for epoch in range(args.epochs):
        t0 = time.time()
        last = None
        for step in range(steps):
            eidx = torch.randint(0, g.num_edges, (args.batch_edges,),
                                 generator=gen, device=dev)
            pos_u, pos_v = src[eidx], dst[eidx]
            neg_u = torch.randint(0, g.num_nodes, (args.batch_edges,),
                                  generator=gen, device=dev)
            neg_v = torch.randint(0, g.num_nodes, (args.batch_edges,),
                                  generator=gen, device=dev)
            seeds = torch.unique(torch.cat([pos_u, pos_v, neg_u, neg_v]))
            inp, out_nodes, blocks = sampler.sample_blocks(seeds)
            h = x_all[inp]
            for i, layer in enumerate(model.layers):
                h = layer(blocks[i], h)
                if i != len(model.layers) - 1:
                    h = F.elu(h)
            # map global endpoint ids -> seed-local rows (seeds are sorted)
            loc = torch.searchsorted(seeds, torch.cat(
                [pos_u, pos_v, neg_u, neg_v]))
            pu, pv, nu, nv = loc.chunk(4)
            pos_s = (h[pu] * h[pv]).sum(-1)
            neg_s = (h[nu] * h[nv]).sum(-1)
            loss = F.binary_cross_entropy_with_logits(
                torch.cat([pos_s, neg_s]),
                torch.cat([torch.ones_like(pos_s),
                           torch.zeros_like(neg_s)]),
            )
            opt.zero_grad(); loss.backward(); opt.step()
            last = (pos_s, neg_s, loss)
        pos_s, neg_s, loss = last
        print(f"epoch {epoch:03d} loss {loss:.4f} "
              f"AUC {auc_score(pos_s.detach(), neg_s.detach()):.3f} "
              f"({time.time() - t0:.2f}s, {steps} steps)", flush=True)


if __name__ == "__main__":
    main()
