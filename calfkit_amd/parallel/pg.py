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
           "all_reduce_sum", "all_to_all_rows", "barrier", "shard_dim",
           "init_model_parallel", "tp_group", "ep_group", "tp_rank",
           "ep_rank", "broadcast_tensor"]


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


# ---------------------------------------------------------- TP x EP groups
#
# Composition (VERDICT r01 #2 "compose EP x TP"): world = tp * ep; rank r
# has tp_rank = r % tp (fast axis: TP peers are xGMI neighbours) and
# ep_rank = r // tp.  Attention/dense collectives run in the TP group,
# MoE token exchanges in the EP group.  With a single axis (tp == world or
# ep == world) no subgroups are created and the default WORLD group serves.

_TP_GROUP = None
_EP_GROUP = None
_TP = 1
_EP = 1


def init_model_parallel(tp: int = 1, ep: int = 1) -> None:
    global _TP_GROUP, _EP_GROUP, _TP, _EP
    d = get_dist()
    if tp * ep != d.world_size:
        raise ValueError(f"tp({tp}) x ep({ep}) != world {d.world_size}")
    _TP, _EP = tp, ep
    _TP_GROUP = _EP_GROUP = None
    if tp > 1 and ep > 1:
        # every rank must create every group, in the same order
        for e in range(ep):
            g = dist.new_group(list(range(e * tp, (e + 1) * tp)))
            if e == d.rank // tp:
                _TP_GROUP = g
        for t in range(tp):
            g = dist.new_group(list(range(t, d.world_size, tp)))
            if t == d.rank % tp:
                _EP_GROUP = g


def tp_group():
    return _TP_GROUP


def ep_group():
    return _EP_GROUP


def tp_rank() -> int:
    return get_dist().rank % _TP if _TP > 1 else 0


def ep_rank() -> int:
    return get_dist().rank // _TP if _EP > 1 else 0


def _group_size(group) -> int:
    return dist.get_world_size(group=group) if group is not None \
        else get_dist().world_size


def broadcast_tensor(t: torch.Tensor, src: int = 0) -> torch.Tensor:
    """World broadcast of a tensor (the lockstep op channel)."""
    if get_dist().world_size > 1:
        dist.broadcast(t, src=src)
    return t


def all_reduce_sum(t: torch.Tensor, group=None) -> torch.Tensor:
    """In-place sum all-reduce across the TP group (no-op at group size 1)."""
    d = get_dist()
    if d.world_size == 1:
        return t
    g = group if group is not None else _TP_GROUP
    if g is not None or _EP == 1:
        dist.all_reduce(t, op=dist.ReduceOp.SUM, group=g)
    return t


def all_to_all_rows(t: torch.Tensor, send_splits: list[int],
                    recv_splits: list[int] | None = None, group=None,
                    ) -> tuple[torch.Tensor, list[int]]:
    """Variable-row all-to-all over the EP group: rank r receives
    t's send_splits[r] rows from every peer, concatenated in rank order.

    The expert-parallel token exchange (MoE dispatch/combine).  When
    recv_splits is unknown it is exchanged first with a counts all-to-all —
    a host sync, which is why EP decode runs eager (capacity
    padding was analyzed and rejected, STATUS r01).  On GPU this rides
    RCCL's point-to-point xGMI sends; gloo serves the CPU lane.
    """
    d = get_dist()
    g = group if group is not None else _EP_GROUP
    if d.world_size == 1 or (g is None and _TP > 1 and _EP == 1):
        return t, [t.shape[0]]
    if recv_splits is None:
        counts = torch.tensor(send_splits, dtype=torch.int64, device=t.device)
        rcounts = torch.empty_like(counts)
        dist.all_to_all_single(rcounts, counts, group=g)
        recv_splits = [int(c) for c in rcounts]
    out = torch.empty(sum(recv_splits), *t.shape[1:], dtype=t.dtype,
                      device=t.device)
    dist.all_to_all_single(out, t.contiguous(), recv_splits, send_splits,
                           group=g)
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
