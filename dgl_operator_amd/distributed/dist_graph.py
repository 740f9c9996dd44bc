"""DistGraph: partitioned graph with distributed sampling + feature pulls.

MI355X-native replacement for dgl.distributed.DistGraph + its sampler RPC and
KVStore feature pulls (reference call sites:
/root/reference/examples/GraphSAGE_dist/code/train_dist.py:28-30,52-70,267-276).
Instead of socket RPC to graph-server processes, every exchange is a bulk
RCCL alltoallv over xGMI (gloo on CPU): one process per GPU owns one
partition — in-edges of its owned dst range (src ids kept GLOBAL) and the
owned rows of every node feature.

Distributed sampling of a frontier runs one alltoallv round per hop:
requests are bucketized by owner (partition book), each rank samples its own
seeds with the HIP sampler, and (neighbors, counts) return to the requesting
rank, which reassembles them in frontier order.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ..graph.graph import Block, _coo_to_compressed
from ..graph.partition import load_partition
from ..ops.sampling import (
    CompactionWorkspace,
    sample_block_fused,
    sample_neighbors,
    to_block,
)
from . import comm
from .partition_book import PartitionBook


def _cumsum0(t: torch.Tensor) -> torch.Tensor:
    z = torch.zeros(1, dtype=t.dtype, device=t.device)
    return torch.cat([z, torch.cumsum(t, 0)])


def _reorder_segments(
    payload: torch.Tensor,
    counts_sorted: torch.Tensor,
    perm: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Segments arrive concatenated in ``sorted`` order; segment s belongs at
    original position perm[s]. Returns (payload_orig, counts_orig)."""
    counts_orig = torch.empty_like(counts_sorted)
    counts_orig[perm] = counts_sorted
    soff = _cumsum0(counts_sorted)
    ooff = _cumsum0(counts_orig)
    total = int(soff[-1])
    if total == 0:
        return payload, counts_orig
    idx = torch.repeat_interleave(ooff[perm], counts_sorted) + (
        torch.arange(total, device=payload.device)
        - torch.repeat_interleave(soff[:-1], counts_sorted)
    )
    out = torch.empty_like(payload)
    out[idx] = payload
    return out, counts_orig


def _segment_sum_by_rank(values: torch.Tensor, rank_counts: torch.Tensor) -> torch.Tensor:
    """Sum ``values`` within consecutive segments sized ``rank_counts``."""
    P = rank_counts.numel()
    seg = torch.repeat_interleave(
        torch.arange(P, device=values.device), rank_counts.to(values.device)
    )
    out = torch.zeros(P, dtype=values.dtype, device=values.device)
    out.index_add_(0, seg, values)
    return out


class DistGraph:
    """One rank's shard of a partitioned graph."""

    def __init__(
        self,
        book: PartitionBook,
        rank: int,
        csc_indptr: torch.Tensor,  # [n_owned+1], rows are owned dst (local)
        csc_indices: torch.Tensor,  # GLOBAL src ids
        ndata: Dict[str, torch.Tensor],  # owned rows only
    ):
        self.book = book
        self.rank = rank
        self.lo, self.hi = book.owned_range(rank)
        self.csc_indptr = csc_indptr
        self.csc_indices = csc_indices
        self.ndata = ndata
        self.workspace = CompactionWorkspace(book.num_nodes, csc_indptr.device)

    # -- constructors ------------------------------------------------------
    @staticmethod
    def from_full_graph(g, book: PartitionBook, rank: int) -> "DistGraph":
        """Build this rank's shard from an in-memory full graph whose node ids
        are already relabeled to the book's contiguous ranges."""
        lo, hi = book.owned_range(rank)
        src, dst = g.edges()
        mask = (dst >= lo) & (dst < hi)
        psrc, pdst = src[mask], dst[mask] - lo
        indptr, indices, _ = _coo_to_compressed(pdst, psrc, hi - lo)
        ndata = {k: v[lo:hi] for k, v in g.ndata.items()}
        return DistGraph(book, rank, indptr, indices, ndata)

    @staticmethod
    def from_partition(json_path: str, part_id: int, device="cpu") -> "DistGraph":
        gpart, feats, spec = load_partition(json_path, part_id)
        book = PartitionBook(spec.boundaries)
        lo, hi = book.owned_range(part_id)
        src = gpart["src_global"].to(device)
        dst = (gpart["dst_global"] - lo).to(device)
        indptr, indices, _ = _coo_to_compressed(dst, src, hi - lo)
        ndata = {k: v.to(device) for k, v in feats.items()}
        return DistGraph(book, part_id, indptr, indices, ndata)

    # -- properties --------------------------------------------------------
    @property
    def device(self):
        return self.csc_indptr.device

    @property
    def num_owned(self) -> int:
        return self.hi - self.lo

    def owned_nodes(self) -> torch.Tensor:
        return torch.arange(self.lo, self.hi, device=self.device)

    # -- distributed sampling ---------------------------------------------
    def _sample_local(self, gids_local: torch.Tensor, fanout: int, replace, seed):
        return sample_neighbors(
            self.csc_indptr, self.csc_indices, gids_local, fanout, replace, seed
        )

    def sample_neighbors_dist(
        self, frontier: torch.Tensor, fanout: int, replace=False, seed: int = 0
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Sample in-neighbors of GLOBAL frontier ids wherever they are owned.
        Returns (neighbors_concat_global, counts) aligned with ``frontier``."""
        rank, ws = comm.world()
        if ws == 1:
            return self._sample_local(frontier - self.lo, fanout, replace, seed)
        sorted_ids, perm, send_counts = self.book.partition_by_owner(frontier)
        recv_counts = comm.exchange_counts(send_counts)
        reqs = comm.all_to_all_v(sorted_ids, send_counts.tolist(), recv_counts.tolist())
        nbrs, cnts = self._sample_local(reqs - self.lo, fanout, replace, seed)
        # return counts (aligned with reqs) and the neighbor payload
        cnts_back = comm.all_to_all_v(cnts, recv_counts.tolist(), send_counts.tolist())
        nb_send = _segment_sum_by_rank(cnts, recv_counts)
        nb_recv = comm.exchange_counts(nb_send)
        nbrs_back = comm.all_to_all_v(nbrs, nb_send.tolist(), nb_recv.tolist())
        # reassemble in frontier order: first per-rank segments are already in
        # ``sorted`` order (stable partition), then scatter to original order.
        return _reorder_segments(nbrs_back, cnts_back, perm)

    def sample_blocks(
        self, seeds: torch.Tensor, fanouts: Sequence[int], replace=False,
        seed: int = 0
    ):
        """Multi-layer distributed sampling; seeds are GLOBAL ids (typically
        owned by this rank). Returns (input_nodes, seeds, blocks)."""
        blocks: List[Block] = []
        cur = seeds
        rank, ws = comm.world()
        for layer, fanout in enumerate(reversed(list(fanouts))):
            if ws == 1 and cur.is_cuda:
                # single-rank GPU fast path: fused sample+compact, local ids
                # shifted into the global range
                blk = sample_block_fused(
                    self.csc_indptr, self.csc_indices, self.workspace,
                    cur - self.lo if self.lo else cur, fanout, replace,
                    seed=seed * 1000003 + layer,
                )
                if self.lo:
                    blk.srcdata_nids = blk.srcdata_nids + self.lo
            else:
                nbrs, counts = self.sample_neighbors_dist(
                    cur, fanout, replace, seed=seed * 1000003 + layer
                )
                blk = to_block(cur, nbrs, counts, self.workspace)
            blocks.insert(0, blk)
            cur = blk.srcdata_nids
        return cur, seeds, blocks

    # -- distributed feature pull (KVStore PULL, C3/C4) --------------------
    def pull(self, key: str, gids: torch.Tensor) -> torch.Tensor:
        """Gather feature rows for GLOBAL ids from their owners (alltoallv)."""
        rank, ws = comm.world()
        feat = self.ndata[key]
        if ws == 1:
            return feat[gids - self.lo]
        sorted_ids, perm, send_counts = self.book.partition_by_owner(gids)
        recv_counts = comm.exchange_counts(send_counts)
        reqs = comm.all_to_all_v(sorted_ids, send_counts.tolist(), recv_counts.tolist())
        rows = feat[reqs - self.lo]
        rows_back = comm.all_to_all_v(rows, recv_counts.tolist(), send_counts.tolist())
        out = torch.empty_like(rows_back)
        out[perm] = rows_back
        return out

    # -- distributed push (KVStore PUSH, C4) -------------------------------
    def push_accumulate(self, key: str, gids: torch.Tensor, rows: torch.Tensor):
        """Scatter-add rows into the owners' feature shards."""
        rank, ws = comm.world()
        feat = self.ndata[key]
        if ws == 1:
            feat.index_add_(0, gids - self.lo, rows)
            return
        sorted_ids, perm, send_counts = self.book.partition_by_owner(gids)
        recv_counts = comm.exchange_counts(send_counts)
        dest_ids = comm.all_to_all_v(sorted_ids, send_counts.tolist(), recv_counts.tolist())
        dest_rows = comm.all_to_all_v(
            rows[perm], send_counts.tolist(), recv_counts.tolist()
        )
        feat.index_add_(0, dest_ids - self.lo, dest_rows)

    def full_neighbor_block(self, seeds: torch.Tensor) -> Block:
        """Block over ALL in-edges of the (owned) seeds — the layer-wise
        full-neighbor inference structure (DistSAGE.inference parity)."""
        local = seeds - self.lo
        starts = self.csc_indptr[local]
        counts = self.csc_indptr[local + 1] - starts
        total = int(counts.sum())
        if total:
            off = _cumsum0(counts)
            pos = torch.repeat_interleave(starts, counts) + (
                torch.arange(total, device=seeds.device)
                - torch.repeat_interleave(off[:-1], counts)
            )
            nbrs = self.csc_indices[pos]
        else:
            nbrs = self.csc_indices.new_empty(0)
        return to_block(seeds, nbrs, counts, self.workspace)

    def barrier(self):
        comm.barrier()
