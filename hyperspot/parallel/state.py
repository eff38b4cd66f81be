"""Tensor/expert-parallel process-group state over RCCL (xGMI).

MI355X nodes are 8 GPUs with 7 point-to-point xGMI links per GPU
(fully-connected, ~153 GB/s per link).  We run one process per GPU with
``torch.distributed`` — backend "nccl" IS RCCL on ROCm — and carve TP / EP
subgroups out of the world.  On CPU (tests) the same code runs over gloo.

The reference has no data-plane collectives at all; the
gRPC control plane it does have is mirrored by the C++ host plane (host/),
not here.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_GROUP: Optional[dist.ProcessGroup] = None
_EP_GROUP: Optional[dist.ProcessGroup] = None
_TP_RANKS: list = []
_EP_RANKS: list = []


def initialize_model_parallel(tp_size: int = 1, ep_size: int = 1,
                              backend: Optional[str] = None) -> None:
    """Initialise torch.distributed (if needed) and build TP/EP subgroups.

    World layout: ranks [i*tp .. (i+1)*tp) form TP group i; EP groups are
    built the same way over ep_size.  Single-process (world=1, tp=1) needs
    no init at all.
    """
    global _TP_GROUP, _EP_GROUP, _TP_RANKS, _EP_RANKS
    if tp_size == 1 and ep_size == 1 and not dist.is_initialized() \
            and "RANK" not in os.environ:
        return
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend)
    world = dist.get_world_size()
    rank = dist.get_rank()
    assert world % tp_size == 0, (world, tp_size)
    for start in range(0, world, tp_size):
        ranks = list(range(start, start + tp_size))
        g = dist.new_group(ranks) if tp_size < world else dist.group.WORLD
        if rank in ranks:
            _TP_GROUP, _TP_RANKS = g, ranks
    assert world % ep_size == 0, (world, ep_size)
    for start in range(0, world, ep_size):
        ranks = list(range(start, start + ep_size))
        g = dist.new_group(ranks) if ep_size < world else dist.group.WORLD
        if rank in ranks:
            _EP_GROUP, _EP_RANKS = g, ranks


def destroy_model_parallel() -> None:
    global _TP_GROUP, _EP_GROUP, _TP_RANKS, _EP_RANKS
    _TP_GROUP = _EP_GROUP = None
    _TP_RANKS = _EP_RANKS = []


def get_tp_group():
    return _TP_GROUP


def get_tp_size() -> int:
    return len(_TP_RANKS) if _TP_RANKS else 1


def get_tp_rank() -> int:
    if not _TP_RANKS:
        return 0
    return _TP_RANKS.index(dist.get_rank())


def get_ep_group():
    return _EP_GROUP


def get_ep_size() -> int:
    return len(_EP_RANKS) if _EP_RANKS else 1


def get_ep_rank() -> int:
    if not _EP_RANKS:
        return 0
    return _EP_RANKS.index(dist.get_rank())


def tensor_model_parallel_all_reduce(x: torch.Tensor) -> torch.Tensor:
    """All-reduce across the TP group (no-op at TP=1).

    xGMI is point-to-point (7 links/GPU): RCCL picks ring for large prefill
    tensors and one-shot for the small decode-step tensors; we keep decode
    all-reduces inside the hipGraph capture so launch cost amortises.
    """
    if get_tp_size() == 1:
        return x
    dist.all_reduce(x, group=_TP_GROUP)
    return x
