#!/usr/bin/env python3
"""Batched graph classification with mean_nodes readout — parity with
/root/reference/examples/graph_classification/code/5_graph_classification.py
(protein-graph-scale synthetic minibatches, GCN + mean readout)."""
import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), "..", ".."))


import argparse

import torch
import torch.nn.functional as F


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--num-graphs", type=int, default=200)
    p.add_argument("--feat", type=int, default=16)
    p.add_argument("--hidden", type=int, default=32)
    p.add_argument("--classes", type=int, default=2)
    p.add_argument("--epochs", type=int, default=20)
    p.add_argument("--batch", type=int, default=32)
    args = p.parse_args()

    from dgl_operator_amd.graph import batch_graphs, rmat_graph
    from dgl_operator_amd.models import GCN

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    graphs, labels = [], []
    for i in range(args.num_graphs):
        n = 20 + (i * 7) % 30
        gg = rmat_graph(n, n * 4, num_feats=args.feat, seed=i, device=dev)
        graphs.append(gg)
        labels.append(i % args.classes)
    labels = torch.tensor(labels, device=dev)
    model = GCN(args.feat, args.hidden, args.classes).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=0.01)
    for epoch in range(args.epochs):
        perm = torch.randperm(args.num_graphs)
        total, correct = 0, 0
        for s in range(0, args.num_graphs, args.batch):
            idx = perm[s : s + args.batch].tolist()
            bg, sizes = batch_graphs([graphs[i] for i in idx])
            bg = bg.add_self_loops()
            x = torch.cat([graphs[i].ndata["feat"] for i in idx])
            y = labels[idx]
            logits = model.forward_graph_readout(bg, x, sizes)
            loss = F.cross_entropy(logits, y)
            opt.zero_grad(); loss.backward(); opt.step()
            correct += int((logits.argmax(1) == y).sum())
            total += len(idx)
        if epoch % 5 == 0 or epoch == args.epochs - 1:
            print(f"epoch {epoch:02d} loss {loss:.4f} acc {correct/total:.3f}",
                  flush=True)


if __name__ == "__main__":
    main()
