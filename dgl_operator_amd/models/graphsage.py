"""GraphSAGE for minibatch (block) and full-graph training.

Architecture parity with the reference's DistSAGE
(/root/reference/examples/GraphSAGE_dist/code/train_dist.py:72-94):
n_layers SAGEConv('mean') with ReLU + dropout between layers.
"""
from __future__ import annotations

from typing import List, Sequence, Union

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..graph.graph import Block, Graph
from ..nn import SAGEConv


class GraphSAGE(nn.Module):
    def __init__(
        self,
        in_feats: int,
        n_hidden: int,
        n_classes: int,
        n_layers: int = 2,
        dropout: float = 0.5,
    ):
        super().__init__()
        self.layers = nn.ModuleList()
        if n_layers == 1:
            self.layers.append(SAGEConv(in_feats, n_classes))
        else:
            self.layers.append(SAGEConv(in_feats, n_hidden))
            for _ in range(n_layers - 2):
                self.layers.append(SAGEConv(n_hidden, n_hidden))
            self.layers.append(SAGEConv(n_hidden, n_classes))
        self.dropout = nn.Dropout(dropout)

    def forward(
        self, blocks: Union[Sequence[Block], Graph], x: torch.Tensor
    ) -> torch.Tensor:
        if isinstance(blocks, Graph):
            blocks = [blocks] * len(self.layers)
        h = x
        for i, (layer, blk) in enumerate(zip(self.layers, blocks)):
            h = layer(blk, h)
            if i != len(self.layers) - 1:
                h = F.relu(h)
                h = self.dropout(h)
        return h


@torch.no_grad()
def inference_dist(model: GraphSAGE, dg, batch_size: int = 1000,
                   feat_key: str = "feat") -> torch.Tensor:
    """Layer-wise full-neighbor distributed inference — parity with
    DistSAGE.inference (/root/reference/examples/GraphSAGE_dist/code/
    train_dist.py:96-144): per layer, each rank computes the layer output for
    its OWNED nodes in seed batches over full in-neighbor blocks, pulling the
    previous layer's output from peer shards (the reference's DistTensor
    writes + g.barrier()); a barrier separates layers. Returns this rank's
    shard of the final layer output."""
    model.eval()
    cur_key = feat_key
    owned = dg.owned_nodes()
    n_layers = len(model.layers)
    # every rank must issue the SAME number of pull collectives per layer:
    # align batch counts to the global max (ranks that run out issue empty
    # batches, which still participate in the exchange)
    import math

    n_batches = max(1, math.ceil(owned.numel() / batch_size))
    from ..distributed import comm as _comm

    if _comm.world()[1] > 1:
        import torch.distributed as dist

        # RCCL (backend 'nccl') rejects CPU tensors — reduce on the shard's
        # compute device (gloo accepts either)
        nb_dev = owned.device if dist.get_backend() == "nccl" else "cpu"
        nb = torch.tensor([n_batches], device=nb_dev)
        dist.all_reduce(nb, op=dist.ReduceOp.MAX)
        n_batches = int(nb[0])
    for li, layer in enumerate(model.layers):
        outs = []
        for bi in range(n_batches):
            seeds = owned[bi * batch_size : (bi + 1) * batch_size]
            blk = dg.full_neighbor_block(seeds)
            x = dg.pull(cur_key, blk.srcdata_nids)
            h = layer(blk, x)
            if li != n_layers - 1:
                h = F.relu(h)
            outs.append(h)
        next_key = f"__infer_h{li}"
        dg.ndata[next_key] = torch.cat(outs) if outs else None
        dg.barrier()
        cur_key = next_key
    out = dg.ndata[cur_key]
    # clean intermediate layer shards except the final
    for li in range(n_layers - 1):
        dg.ndata.pop(f"__infer_h{li}", None)
    return out
