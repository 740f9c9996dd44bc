"""DistTensor + DistNodeDataLoader — the remaining dgl.distributed API
surface a migrating user reaches for.

Reference usage being mirrored:
  * dgl.distributed.DistTensor — created for layer-wise inference outputs
    and written by global id (/root/reference/examples/GraphSAGE_dist/code/
    train_dist.py:112-117,140)
  * dgl.distributed.DistDataLoader + NeighborSampler — the minibatch
    loop's (input_nodes, seeds, blocks) source (train_dist.py:52-70,215)

Both ride the same partition-book alltoallv plumbing as the rest of the
distributed plane (one RCCL/gloo process group).

Collective contract: with world_size > 1 every __getitem__/__setitem__ is
an alltoallv — ALL RANKS must call them the same number of times (the
reference has the same property through its kvstore barriers).
"""
from __future__ import annotations

from typing import Optional, Sequence, Tuple

import torch

from . import comm
from .partition_book import PartitionBook


def node_split(nodes: torch.Tensor, partition_book: PartitionBook,
               rank: Optional[int] = None) -> torch.Tensor:
    """Module-level dgl.distributed.node_split spelling (the reference
    calls it with a GLOBAL boolean mask + the partition book,
    train_dist.py:274-276): this rank's owned global ids where the mask
    is set."""
    if rank is None:
        rank, _ = comm.world()
    lo, hi = partition_book.owned_range(rank)
    local = nodes[lo:hi].bool()
    return torch.arange(lo, hi, device=nodes.device)[local]


class DistTensor:
    """Row-sharded distributed tensor keyed by GLOBAL row id.

    Each rank owns the contiguous row range of ``book``; reads/writes of
    arbitrary global ids route to the owners (alltoallv), exactly like
    ``DistGraph.pull`` / ``push_accumulate`` route feature rows.
    """

    def __init__(
        self,
        shape: Sequence[int],
        dtype: torch.dtype = torch.float32,
        name: str = "",
        book: Optional[PartitionBook] = None,
        rank: Optional[int] = None,
        device="cpu",
    ):
        rk, ws = comm.world()
        self.rank = rk if rank is None else rank
        if book is None:
            n = int(shape[0])
            bounds = [n * p // ws for p in range(ws + 1)]
            book = PartitionBook(bounds, device=device)
        self.book = book
        self.name = name
        self.lo, self.hi = book.owned_range(self.rank)
        self.shape = tuple(shape)
        assert self.shape[0] == book.num_nodes, (
            f"shape[0]={shape[0]} must equal the book's row count "
            f"{book.num_nodes}")
        self.local = torch.zeros((self.hi - self.lo,) + self.shape[1:],
                                 dtype=dtype, device=device)

    @property
    def dtype(self):
        return self.local.dtype

    @property
    def device(self):
        return self.local.device

    def __len__(self):
        return self.shape[0]

    # -- read (KVStore PULL) ----------------------------------------------
    def __getitem__(self, gids: torch.Tensor) -> torch.Tensor:
        rank, ws = comm.world()
        if ws == 1:
            return self.local[gids - self.lo]
        sorted_ids, perm, send_counts = self.book.partition_by_owner(gids)
        recv_counts = comm.exchange_counts(send_counts)
        reqs = comm.all_to_all_v(sorted_ids, send_counts.tolist(),
                                 recv_counts.tolist())
        rows = self.local[reqs - self.lo]
        rows_back = comm.all_to_all_v(rows, recv_counts.tolist(),
                                      send_counts.tolist())
        out = torch.empty_like(rows_back)
        out[perm] = rows_back
        return out

    # -- write (KVStore PUSH, overwrite semantics like DistTensor) ---------
    def __setitem__(self, gids: torch.Tensor, rows: torch.Tensor):
        rank, ws = comm.world()
        if ws == 1:
            self.local[gids - self.lo] = rows
            return
        sorted_ids, perm, send_counts = self.book.partition_by_owner(gids)
        recv_counts = comm.exchange_counts(send_counts)
        dest_ids = comm.all_to_all_v(sorted_ids, send_counts.tolist(),
                                     recv_counts.tolist())
        dest_rows = comm.all_to_all_v(rows[perm].contiguous(),
                                      send_counts.tolist(),
                                      recv_counts.tolist())
        self.local[dest_ids - self.lo] = dest_rows

    def index_add_(self, gids: torch.Tensor, rows: torch.Tensor):
        """Accumulating write (push_accumulate semantics)."""
        rank, ws = comm.world()
        if ws == 1:
            self.local.index_add_(0, gids - self.lo, rows)
            return self
        sorted_ids, perm, send_counts = self.book.partition_by_owner(gids)
        recv_counts = comm.exchange_counts(send_counts)
        dest_ids = comm.all_to_all_v(sorted_ids, send_counts.tolist(),
                                     recv_counts.tolist())
        dest_rows = comm.all_to_all_v(rows[perm].contiguous(),
                                      send_counts.tolist(),
                                      recv_counts.tolist())
        self.local.index_add_(0, dest_ids - self.lo, dest_rows)
        return self


class DistNodeDataLoader:
    """Minibatch (input_nodes, seeds, blocks) loader over a DistGraph —
    the DistDataLoader + NeighborSampler pair of the reference's training
    loop, with the multi-rank step-count alignment built in: every rank
    runs the SAME number of batches per epoch (all-reduced MIN, like
    train_dist.py's steps_per_epoch) so sampling collectives and gradient
    all-reduces cannot desynchronize.

    Seeds are drawn as contiguous slices of a fresh per-epoch permutation
    (DataLoader shuffle semantics, no per-step unique/sync).
    """

    def __init__(
        self,
        dg,
        nids: torch.Tensor,
        fanouts: Sequence[int],
        batch_size: int,
        shuffle: bool = True,
        seed: int = 0,
        epoch: int = 0,
    ):
        self.dg = dg
        self.nids = nids
        self.fanouts = list(fanouts)
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.seed = seed
        self.epoch = epoch
        rank, ws = comm.world()
        local_steps = max(1, nids.numel() // batch_size)
        if ws > 1:
            import torch.distributed as dist

            dev = nids.device if dist.get_backend() == "nccl" else "cpu"
            t = torch.tensor([local_steps], device=dev)
            dist.all_reduce(t, op=dist.ReduceOp.MIN)
            local_steps = max(1, int(t[0]))
        self.steps_per_epoch = local_steps

    def __len__(self):
        return self.steps_per_epoch

    def __iter__(self):
        n = self.nids.numel()
        if self.shuffle:
            gen = torch.Generator(device=self.nids.device)
            rank, _ = comm.world()
            gen.manual_seed(self.seed * 1_000_003 + self.epoch * 1009 + rank)
            order = torch.randperm(n, generator=gen, device=self.nids.device)
        else:
            order = torch.arange(n, device=self.nids.device)
        for step in range(self.steps_per_epoch):
            sel = order[step * self.batch_size : (step + 1) * self.batch_size]
            seeds = self.nids[sel]
            yield self.dg.sample_blocks(
                seeds, self.fanouts,
                seed=self.seed + self.epoch * 100_000 + step + 1)
        self.epoch += 1
