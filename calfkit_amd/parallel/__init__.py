"""Distributed/parallel plumbing: one process per GPU, RCCL over xGMI.

``torch.distributed`` with backend "nccl" IS RCCL on ROCm; CPU test lanes use
"gloo".  TP collectives here are shaped for the MI355X node topology: 7
point-to-point xGMI links × ~153 GB/s per GPU — decode-step
activations are latency-bound (small), prefill activations bandwidth-bound,
so the engine keeps decode all-reduces unbucketed on the compute stream and
prefill all-reduces chunked.
"""

from .pg import (
    DistInfo,
    all_reduce_sum,
    all_to_all_rows,
    barrier,
    broadcast_tensor,
    ep_group,
    ep_rank,
    get_dist,
    init_distributed,
    init_model_parallel,
    is_initialized,
    shard_dim,
    tp_group,
    tp_rank,
)

__all__ = [
    "DistInfo",
    "broadcast_tensor",
    "ep_group",
    "ep_rank",
    "init_model_parallel",
    "tp_group",
    "tp_rank",
    "all_reduce_sum",
    "all_to_all_rows",
    "barrier",
    "get_dist",
    "init_distributed",
    "is_initialized",
    "shard_dim",
]
