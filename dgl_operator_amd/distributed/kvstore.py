"""Sharded embedding KVStore with fused row-sparse Adagrad — the MI355X
replacement for the reference's parameter-server KVStore
(/root/reference/examples/DGL-KE/hotfix/dis_kvstore.py KVServer/KVClient and
kvserver.py's Adagrad push handler).

Design: no server processes, no sockets, no shared-memory handshakes (C6 of
SURVEY.md §2.5 disappears by construction) — each rank IS the server for its
contiguous shard of rows. PULL/PUSH are alltoallv exchanges over RCCL/xGMI;
the optimizer update runs as a HIP kernel on the owning GPU, which keeps the
embedding table resident in HBM (288 GB/GPU => Freebase-scale tables fit
sharded 8-way).
"""
from __future__ import annotations

import torch

from ..ops.adagrad import sparse_adagrad_update
from . import comm
from .partition_book import PartitionBook


class ShardedEmbedding:
    """Row-sharded embedding table with pull / push-grad(Adagrad) ops."""

    def __init__(
        self,
        num_rows: int,
        dim: int,
        num_parts: int,
        rank: int,
        device="cpu",
        init_range: float = 1.0,
        seed: int = 0,
        boundaries=None,
    ):
        # boundaries: explicit (possibly uneven) range bounds, e.g. the
        # degree-balanced entity partition written by tools/kg_partition.py;
        # default = even split
        if boundaries is not None:
            bounds = [int(b) for b in boundaries]
            assert len(bounds) == num_parts + 1 and bounds[-1] == num_rows, (
                f"bad boundaries {bounds} for {num_rows} rows / "
                f"{num_parts} parts")
        else:
            bounds = [num_rows * p // num_parts for p in range(num_parts + 1)]
        self.book = PartitionBook(bounds, device=device)
        self.rank = rank
        self.lo, self.hi = self.book.owned_range(rank)
        self.dim = dim
        gen = torch.Generator(device=device)
        gen.manual_seed(seed)  # same seed on every rank; slice determinism via
        # generating the full table is wasteful — instead generate only the
        # owned rows with a rank-independent per-row hash-free approach:
        # uniform init is i.i.d., so a per-rank seed is statistically fine.
        gen.manual_seed(seed * 1000003 + rank)
        self.local = (
            torch.rand(self.hi - self.lo, dim, generator=gen, device=device)
            * 2.0 - 1.0
        ) * init_range
        self.state = torch.zeros(self.hi - self.lo, device=device)

    @property
    def device(self):
        return self.local.device

    # -- PULL (dis_kvstore.py:818-902 equivalent) --------------------------
    def pull(self, gids: torch.Tensor) -> torch.Tensor:
        rank, ws = comm.world()
        if ws == 1:
            if gids.is_cuda and self.local.is_floating_point():
                from ..ops import backend

                ext = backend.ext_for(gids)
                return ext.gather_rows(self.local, gids, None, self.lo)
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

    # -- PUSH with Adagrad at the owner (kvserver.py:41-51 equivalent) ------
    def push_grad(self, gids: torch.Tensor, grads: torch.Tensor, lr: float,
                  eps: float = 1e-10):
        rank, ws = comm.world()
        if ws == 1:
            sparse_adagrad_update(self.local, self.state, gids - self.lo,
                                  grads, lr, eps)
            return
        sorted_ids, perm, send_counts = self.book.partition_by_owner(gids)
        recv_counts = comm.exchange_counts(send_counts)
        dest_ids = comm.all_to_all_v(sorted_ids, send_counts.tolist(),
                                     recv_counts.tolist())
        dest_grads = comm.all_to_all_v(grads[perm].contiguous(),
                                       send_counts.tolist(),
                                       recv_counts.tolist())
        sparse_adagrad_update(self.local, self.state, dest_ids - self.lo,
                              dest_grads, lr, eps)

    def barrier(self):
        comm.barrier()

    # -- checkpoint layout (dglkerun --save_path equivalent) ---------------
    def save_shard(self, path: str):
        torch.save(
            {
                "lo": self.lo,
                "hi": self.hi,
                "emb": self.local.cpu(),
                "state": self.state.cpu(),
            },
            path,
        )

    def load_shard(self, path: str):
        d = torch.load(path, weights_only=True)
        assert d["lo"] == self.lo and d["hi"] == self.hi
        self.local.copy_(d["emb"].to(self.device))
        self.state.copy_(d["state"].to(self.device))
