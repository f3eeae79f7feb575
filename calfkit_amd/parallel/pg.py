"""Process-group bootstrap and TP collectives.

One process per GPU; backend "nccl" (RCCL over xGMI) on GPU boxes, "gloo" in
CPU test lanes.  All BASELINE configs are single-node, so there is no
cross-node backend.

xGMI note: each MI355X has 7 p2p links × ~153 GB/s.  A decode
step's TP all-reduce moves batch×hidden×2 B (a few hundred KB) — pure
latency regime, so the engine issues it unbucketed and RCCL's one-shot
algorithm applies; prefill all-reduces move S×hidden×2 B and ride the ring.
"""

from __future__ import annotations

import datetime
import os
from dataclasses import dataclass

import torch
import torch.distributed as dist

__all__ = ["DistInfo", "init_distributed", "get_dist", "is_initialized",
           "all_reduce_sum", "all_to_all_rows", "barrier", "shard_dim"]


@dataclass(frozen=True)
class DistInfo:
    rank: int
    world_size: int
    local_rank: int
    backend: str
    device: torch.device

    @property
    def is_main(self) -> bool:
        return self.rank == 0


_DIST: DistInfo | None = None


def init_distributed(backend: str | None = None,
                     device_override: str | None = None) -> DistInfo:
    """Initialize from torchrun env (RANK/WORLD_SIZE/...); single-process
    fallback when absent."""
    global _DIST
    if _DIST is not None:
        return _DIST
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_cuda = torch.cuda.is_available()
    if backend is None:
        backend = "nccl" if use_cuda else "gloo"
    if device_override is not None:
        device = torch.device(device_override)
    elif use_cuda:
        device = torch.device("cuda", local_rank % max(1, torch.cuda.device_count()))
        torch.cuda.set_device(device)
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29510")
        dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=600))
    _DIST = DistInfo(rank=rank, world_size=world, local_rank=local_rank,
                     backend=backend, device=device)
    return _DIST


def get_dist() -> DistInfo:
    return _DIST or init_distributed()


def is_initialized() -> bool:
    return _DIST is not None


def all_reduce_sum(t: torch.Tensor) -> torch.Tensor:
    """In-place sum all-reduce across the TP group (no-op at world 1)."""
    d = get_dist()
    if d.world_size > 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def all_to_all_rows(t: torch.Tensor, send_splits: list[int],
                    recv_splits: list[int] | None = None,
                    ) -> tuple[torch.Tensor, list[int]]:
    """Variable-row all-to-all: rank r receives t's send_splits[r] rows from
    every peer, concatenated in rank order.

    The expert-parallel token exchange (MoE dispatch/combine).  When
    recv_splits is unknown it is exchanged first with a counts all-to-all —
    a host sync, which is why EP decode runs eager (the
    fixed-capacity capturable variant is the planned next step).  On GPU
    this rides RCCL's point-to-point xGMI sends; gloo serves the CPU lane.
    """
    d = get_dist()
    if d.world_size == 1:
        return t, [t.shape[0]]
    if recv_splits is None:
        counts = torch.tensor(send_splits, dtype=torch.int64, device=t.device)
        rcounts = torch.empty_like(counts)
        dist.all_to_all_single(rcounts, counts)
        recv_splits = [int(c) for c in rcounts]
    out = torch.empty(sum(recv_splits), *t.shape[1:], dtype=t.dtype,
                      device=t.device)
    dist.all_to_all_single(out, t.contiguous(), recv_splits, send_splits)
    return out, recv_splits


def barrier() -> None:
    d = get_dist()
    if d.world_size > 1:
        if d.backend == "nccl":
            dist.barrier(device_ids=[d.device.index])
        else:
            dist.barrier()


def shard_dim(total: int, world: int, rank: int) -> tuple[int, int]:
    """(start, size) of this rank's equal shard; total must divide evenly."""
    if total % world != 0:
        raise ValueError(f"cannot shard {total} over {world} ranks evenly")
    size = total // world
    return rank * size, size
