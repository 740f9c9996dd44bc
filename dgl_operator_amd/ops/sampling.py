"""Neighbor sampling + block compaction — K7 of SURVEY.md §2.4.

Replaces ``dgl.distributed.sample_neighbors`` + ``dgl.to_block``
(/root/reference/examples/GraphSAGE_dist/code/train_dist.py:52-70) with an
entirely on-device path: a HIP sampling kernel (csrc/sampling.hip) draws
``fanout`` in-neighbors per seed (without replacement, DGL's default), and
compaction/relabeling is a fused atomic-claim kernel chain (sample_block:
one host sync per hop; sample_block_capture: zero, for hipGraph capture) —
no sampler worker processes (the reference needed --num-samplers CPU
processes; on MI355X the sampler is a kernel).
"""
from __future__ import annotations

from typing import Tuple

import torch

from . import backend
from ..graph.graph import Block


def sample_neighbors(
    indptr: torch.Tensor,
    indices: torch.Tensor,
    seeds: torch.Tensor,
    fanout: int,
    replace: bool = False,
    seed: int = 0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Sample up to ``fanout`` in-neighbors per seed from a CSC structure.

    Returns (neighbors, counts): ``neighbors`` is the concatenation of each
    seed's sampled neighbor ids (csc ``indices`` values); ``counts[i]`` is how
    many were drawn for seed i (== min(fanout, degree) when replace=False).
    """
    if seeds.is_cuda:
        ext = backend.ext_for(seeds)
        return ext.sample_neighbors(indptr, indices, seeds, fanout, replace, seed)
    return _sample_ref(indptr, indices, seeds, fanout, replace, seed)


def _sample_ref(indptr, indices, seeds, fanout, replace, seed):
    gen = torch.Generator().manual_seed(seed)
    deg = indptr[seeds + 1] - indptr[seeds]
    if replace:
        counts = torch.where(deg > 0, torch.full_like(deg, fanout), deg.new_zeros(()))
        pick = (
            torch.rand(seeds.numel(), fanout, generator=gen) * deg.clamp(min=1).unsqueeze(1).to(torch.float64)
        ).to(torch.int64)
        flat = (indptr[seeds].unsqueeze(1) + pick).reshape(-1)
        mask = torch.repeat_interleave(deg > 0, fanout)
        return indices[flat[mask]], counts
    counts = torch.minimum(deg, torch.full_like(deg, fanout))
    out = []
    for i in range(seeds.numel()):
        s, d = int(indptr[seeds[i]]), int(deg[i])
        if d <= fanout:
            out.append(indices[s : s + d])
        else:
            perm = torch.randperm(d, generator=gen)[:fanout]
            out.append(indices[s + perm])
    return (
        torch.cat(out) if out else indices.new_empty(0),
        counts,
    )


class CompactionWorkspace:
    """Reusable node-id -> block-local-id translation table (one per local graph)."""

    def __init__(self, num_nodes: int, device):
        self.table = torch.full((num_nodes,), -1, dtype=torch.int64, device=device)

    def relabel(
        self, seeds: torch.Tensor, neighbors: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Returns (srcdata_nids, local_neighbor_ids); seeds occupy local ids
        [0, len(seeds)) — the DGL block convention (dst nodes first)."""
        if seeds.is_cuda:
            ext = backend.ext_for(seeds)
            return ext.compact_ids(self.table, seeds, neighbors)
        n_seed = seeds.numel()
        t = self.table
        t[seeds] = torch.arange(n_seed, device=seeds.device)
        known = t[neighbors]
        new_global = neighbors[known < 0]
        uniq_new = torch.unique(new_global)
        t[uniq_new] = torch.arange(uniq_new.numel(), device=seeds.device) + n_seed
        local_nbrs = t[neighbors]
        # restore workspace
        t[seeds] = -1
        t[uniq_new] = -1
        return torch.cat([seeds, uniq_new]), local_nbrs


def sample_block_fused(
    indptr: torch.Tensor,
    indices: torch.Tensor,
    workspace: CompactionWorkspace,
    seeds: torch.Tensor,
    fanout: int,
    replace: bool = False,
    seed: int = 0,
    seed_dev: "torch.Tensor | None" = None,
    rows: "torch.Tensor | None" = None,
) -> Block:
    """GPU fast path: fused sample+compact with ONE host sync per hop
    (csrc/sampling.hip::sample_block). Semantically identical to
    sample_neighbors + to_block."""
    ext = backend.ext_for(seeds)
    padded, counts, srcdata, counter = ext.sample_block(
        indptr, indices, workspace.table, seeds, fanout, replace, seed,
        seed_dev, rows,
    )
    blk_indptr = torch.zeros(
        counts.numel() + 1, dtype=torch.int64, device=seeds.device
    )
    torch.cumsum(counts, 0, out=blk_indptr[1:])
    # one host sync fetches (n_new, total_edges) together
    tot = torch.stack([counter[0], blk_indptr[-1]]).cpu()
    n_new, E = int(tot[0]), int(tot[1])
    packed = ext.pack_padded(padded, counts, blk_indptr[:-1].contiguous(), E)
    n_seed = seeds.numel()
    return Block(
        blk_indptr,
        packed,
        num_src=n_seed + n_new,
        num_dst=n_seed,
        srcdata_nids=srcdata[: n_seed + n_new],
    )


def to_block(
    seeds: torch.Tensor,
    neighbors: torch.Tensor,
    counts: torch.Tensor,
    workspace: CompactionWorkspace,
) -> Block:
    """Build a bipartite block from per-seed sampled neighbors (parent ids)."""
    srcdata_nids, local_nbrs = workspace.relabel(seeds, neighbors)
    indptr = torch.zeros(seeds.numel() + 1, dtype=torch.int64, device=seeds.device)
    indptr[1:] = torch.cumsum(counts, 0)
    return Block(
        indptr,
        local_nbrs,
        num_src=srcdata_nids.numel(),
        num_dst=seeds.numel(),
        srcdata_nids=srcdata_nids,
    )


def sample_block_capture(
    indptr: torch.Tensor,
    indices: torch.Tensor,
    workspace: CompactionWorkspace,
    seeds: torch.Tensor,
    fanout: int,
    seed: int,
    seed_dev: torch.Tensor,
    rows: "torch.Tensor | None" = None,
) -> "tuple[Block, torch.Tensor]":
    """hipGraph-capturable block build: NO host synchronization. Shapes are
    worst-case (num_src = n + n*fanout, packed buffer n*fanout); the actual
    sizes live on device (the claim counter / indptr[-1]) and every consumer
    (SpMM, pull-gather) only reads the valid prefix, because the unclaimed
    srcdata tail is node id 0 (valid) and SpMM walks indptr. RNG comes from
    ``seed_dev`` (updated between replays) mixed with the static ``seed``.

    Returns (block, counter) — counter[0] is the device-side count of newly
    claimed source nodes (needed for the valid-edge metric)."""
    ext = backend.ext_for(seeds)
    padded, counts, srcdata, counter = ext.sample_block(
        indptr, indices, workspace.table, seeds, fanout, False, seed, seed_dev,
        rows,
    )
    n = seeds.numel()
    blk_indptr = torch.zeros(n + 1, dtype=torch.int64, device=seeds.device)
    torch.cumsum(counts, 0, out=blk_indptr[1:])
    packed = ext.pack_padded(padded, counts, blk_indptr[:-1].contiguous(),
                             n * fanout)
    blk = Block(
        blk_indptr,
        packed,
        num_src=n + n * fanout,  # worst case; valid prefix tracked on device
        num_dst=n,
        srcdata_nids=srcdata,
    )
    return blk, counter


class NeighborSampler:
    """Multi-layer neighbor sampler over a local CSC graph.

    ``fanouts`` lists per-layer fanouts input-layer-first, like the
    reference's --fan_out 10,25 (DGL MultiLayerNeighborSampler order), so
    sampling proceeds over ``reversed(fanouts)`` starting from the seeds.
    Produces blocks ordered input-layer-first, as model.forward expects.
    """

    def __init__(self, indptr, indices, fanouts, num_nodes: int, replace=False):
        self.indptr = indptr
        self.indices = indices
        self.fanouts = list(fanouts)
        self.replace = replace
        self.workspace = CompactionWorkspace(num_nodes, indptr.device)
        self._step = 0

    def sample_blocks(self, seeds: torch.Tensor):
        blocks = []
        cur = seeds
        self._step += 1
        for layer, fanout in enumerate(reversed(self.fanouts)):
            if cur.is_cuda:
                blk = sample_block_fused(
                    self.indptr, self.indices, self.workspace, cur, fanout,
                    self.replace, seed=(self._step * 1000003 + layer),
                )
            else:
                nbrs, counts = sample_neighbors(
                    self.indptr,
                    self.indices,
                    cur,
                    fanout,
                    self.replace,
                    seed=(self._step * 1000003 + layer),
                )
                blk = to_block(cur, nbrs, counts, self.workspace)
            blocks.insert(0, blk)
            cur = blk.srcdata_nids
        return cur, seeds, blocks  # (input_nodes, output_nodes, blocks)
