"""R-MAT synthetic graph generator (Chakrabarti et al.) — torch-native.

Used by bench.py to build a graph of ogbn-products shape (2,449,029 nodes /
61,859,140 directed edges / 100-dim features / 47 classes) with random-init
weights, since this environment has no network for datasets
(reference config anchor: /root/reference/examples/v1alpha1/GraphSAGE_dist.yaml
+ examples/GraphSAGE_dist/code/train_dist.py defaults).

Generation is fully vectorized torch (runs on GPU in seconds): each of the
~log2(N) bit levels picks a quadrant i.i.d. per edge with probabilities
(a, b, c, d). Default skew a=0.57,b=0.19,c=0.19,d=0.05 (Graph500).
"""
from __future__ import annotations

import math
from typing import Tuple

import torch

from .graph import Graph

OGBN_PRODUCTS_NODES = 2_449_029
OGBN_PRODUCTS_EDGES = 61_859_140
OGBN_PRODUCTS_FEATS = 100
OGBN_PRODUCTS_CLASSES = 47


def ogbn_products_shape() -> Tuple[int, int, int, int]:
    return (
        OGBN_PRODUCTS_NODES,
        OGBN_PRODUCTS_EDGES,
        OGBN_PRODUCTS_FEATS,
        OGBN_PRODUCTS_CLASSES,
    )


def rmat_edges(
    num_nodes: int,
    num_edges: int,
    a: float = 0.57,
    b: float = 0.19,
    c: float = 0.19,
    seed: int = 0,
    device: str | torch.device = "cpu",
    chunk: int = 1 << 24,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Return (src, dst) int64 edge tensors. Node ids are scrambled so the
    power-law hubs are not clustered at id 0."""
    gen = torch.Generator(device=device)
    gen.manual_seed(seed)
    scale = max(1, math.ceil(math.log2(max(2, num_nodes))))
    srcs = []
    dsts = []
    remaining = num_edges
    while remaining > 0:
        m = min(chunk, remaining)
        src = torch.zeros(m, dtype=torch.int64, device=device)
        dst = torch.zeros(m, dtype=torch.int64, device=device)
        for _ in range(scale):
            r = torch.rand(m, generator=gen, device=device)
            # quadrant: 0 => (0,0), 1 => (0,1), 2 => (1,0), 3 => (1,1)
            src_bit = (r >= a + b).to(torch.int64)
            dst_bit = ((r >= a) & (r < a + b) | (r >= a + b + c)).to(torch.int64)
            src = (src << 1) | src_bit
            dst = (dst << 1) | dst_bit
        # scramble ids with an affine permutation mod 2^scale, then clamp into range
        n2 = 1 << scale
        mult = 0x9E3779B1 | 1
        src = ((src * mult) + 12345) & (n2 - 1)
        dst = ((dst * mult) + 12345) & (n2 - 1)
        src = src % num_nodes
        dst = dst % num_nodes
        keep = src != dst  # drop self loops
        srcs.append(src[keep])
        dsts.append(dst[keep])
        remaining -= m
    return torch.cat(srcs), torch.cat(dsts)


def rmat_graph(
    num_nodes: int,
    num_edges: int,
    num_feats: int = 0,
    num_classes: int = 0,
    seed: int = 0,
    device: str | torch.device = "cpu",
    feat_dtype: torch.dtype = torch.float32,
) -> Graph:
    src, dst = rmat_edges(num_nodes, num_edges, seed=seed, device=device)
    g = Graph(src, dst, num_nodes)
    if num_feats:
        fgen = torch.Generator(device=device)
        fgen.manual_seed(seed + 1)
        g.ndata["feat"] = torch.randn(
            num_nodes, num_feats, generator=fgen, device=device, dtype=feat_dtype
        )
    if num_classes:
        lgen = torch.Generator(device=device)
        lgen.manual_seed(seed + 2)
        g.ndata["label"] = torch.randint(
            0, num_classes, (num_nodes,), generator=lgen, device=device
        )
    return g
