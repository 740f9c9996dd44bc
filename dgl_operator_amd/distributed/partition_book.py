"""Partition book: global-node-id -> (owner rank, local row).

After partition_graph's relabeling every part owns one contiguous global id
range, so the book is a boundary array and ownership lookup is a bucketize —
the MI355X-native replacement for DGL's GraphPartitionBook / the kvstore's
partition logic (/root/reference/examples/DGL-KE/hotfix/dis_kvstore.py
machine_id partitioning).
"""
from __future__ import annotations

from typing import Sequence, Tuple

import torch


class PartitionBook:
    def __init__(self, boundaries: Sequence[int], device=None):
        self.boundaries = torch.as_tensor(
            list(boundaries), dtype=torch.int64, device=device
        )
        assert self.boundaries[0] == 0

    @property
    def num_parts(self) -> int:
        return self.boundaries.numel() - 1

    @property
    def num_nodes(self) -> int:
        return int(self.boundaries[-1])

    def to(self, device) -> "PartitionBook":
        return PartitionBook(self.boundaries.to(device).tolist(), device=device)

    def owner(self, gids: torch.Tensor) -> torch.Tensor:
        b = self.boundaries.to(gids.device)
        return torch.bucketize(gids, b[1:-1], right=True)

    def to_local(self, gids: torch.Tensor, part: int) -> torch.Tensor:
        return gids - int(self.boundaries[part])

    def owned_range(self, part: int) -> Tuple[int, int]:
        return int(self.boundaries[part]), int(self.boundaries[part + 1])

    def partition_by_owner(
        self, gids: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """Sort ids by owning part. Returns (sorted_ids, perm, counts_per_part):
        ``sorted_ids = gids[perm]``; ``counts_per_part`` has num_parts entries."""
        owner = self.owner(gids)
        perm = torch.argsort(owner, stable=True)
        counts = torch.bincount(owner, minlength=self.num_parts)
        return gids[perm], perm, counts
