"""Process-group setup + alltoallv exchange primitives.

MI355X-native replacement for the reference's three-layer comm stack
(gloo DDP + DGL socket RPC + kvstore push/pull): one
torch.distributed process group — RCCL over xGMI on GPU boxes ("nccl"
backend IS RCCL on ROCm), gloo for CPU tests — carries dense-grad
all-reduce, feature/embedding pulls (alltoallv) and barriers.
"""
from __future__ import annotations

import os
from typing import Optional, Sequence, Tuple

import torch
import torch.distributed as dist


def init_from_env(backend: Optional[str] = None) -> Tuple[int, int]:
    """Initialize the default process group from torchrun env vars.
    Returns (rank, world_size); (0, 1) without distributed launch."""
    if "RANK" not in os.environ or int(os.environ.get("WORLD_SIZE", "1")) <= 1:
        return 0, 1
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    return dist.get_rank(), dist.get_world_size()


def initialize(ip_config: Optional[str] = None, num_servers: int = 0,
               num_workers: int = 0, backend: Optional[str] = None
               ) -> Tuple[int, int]:
    """dgl.distributed.initialize spelling (reference train_dist.py:267).
    There are no graph-server or sampler processes to start here — the
    equivalent setup is joining the RCCL/gloo process group; the ip_config
    and process-count arguments are accepted and unused (torchrun's env
    carries the rendezvous)."""
    return init_from_env(backend=backend)


def world() -> Tuple[int, int]:
    if dist.is_available() and dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def barrier():
    if dist.is_initialized():
        dist.barrier()


def all_reduce_max_(t: torch.Tensor):
    if dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return t


def all_reduce_sum_(t: torch.Tensor):
    if dist.is_initialized():
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def exchange_counts(send_counts: torch.Tensor) -> torch.Tensor:
    """all_to_all of per-rank item counts. send_counts: [P] int64 on CPU or GPU."""
    if not dist.is_initialized():
        return send_counts.clone()
    recv = torch.empty_like(send_counts)
    dist.all_to_all_single(recv, send_counts.contiguous())
    return recv


def all_to_all_v(
    data: torch.Tensor,
    send_counts: Sequence[int],
    recv_counts: Sequence[int],
) -> torch.Tensor:
    """Variable-size all_to_all along dim 0. ``data`` is the concatenation of
    per-destination-rank segments sized ``send_counts``."""
    if not dist.is_initialized():
        return data
    out_shape = (int(sum(recv_counts)),) + tuple(data.shape[1:])
    out = torch.empty(out_shape, dtype=data.dtype, device=data.device)
    dist.all_to_all_single(
        out, data.contiguous(), list(recv_counts), list(send_counts)
    )
    return out
