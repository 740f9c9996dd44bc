#!/usr/bin/env python3
"""Full-graph GCN node classification — parity with the reference tutorial
(/root/reference/examples/node_classification/code/1_introduction.py):
2-layer GCN h=16 on a Cora-scale graph, 100 epochs, reports accuracy."""
import os as _os
import sys as _sys

_sys.path.insert(0, _os.path.join(_os.path.dirname(_os.path.abspath(__file__)), "..", ".."))


import argparse

import torch
import torch.nn.functional as F


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--nodes", type=int, default=2708)       # Cora shape
    p.add_argument("--edges", type=int, default=10556)
    p.add_argument("--feat", type=int, default=1433)
    p.add_argument("--classes", type=int, default=7)
    p.add_argument("--hidden", type=int, default=16)
    p.add_argument("--epochs", type=int, default=100)
    p.add_argument("--lr", type=float, default=0.01)
    args = p.parse_args()

    from dgl_operator_amd.graph import rmat_graph
    from dgl_operator_amd.models import GCN

    dev = "cuda:0" if torch.cuda.is_available() else "cpu"
    g = rmat_graph(args.nodes, args.edges, num_feats=args.feat,
                   num_classes=args.classes, seed=0, device=dev)
    g = g.add_self_loops()
    gen = torch.Generator(device=dev).manual_seed(1)
    train_mask = torch.rand(g.num_nodes, generator=gen, device=dev) < 0.6
    model = GCN(args.feat, args.hidden, args.classes).to(dev)
    opt = torch.optim.Adam(model.parameters(), lr=args.lr)
    x, y = g.ndata["feat"], g.ndata["label"]
    for epoch in range(args.epochs):
        logits = model(g, x)
        loss = F.cross_entropy(logits[train_mask], y[train_mask])
        opt.zero_grad(); loss.backward(); opt.step()
        if epoch % 20 == 0 or epoch == args.epochs - 1:
            acc = (logits.argmax(1) == y).float().mean().item()
            print(f"epoch {epoch:03d} loss {loss:.4f} acc {acc:.3f}", flush=True)


if __name__ == "__main__":
    main()
