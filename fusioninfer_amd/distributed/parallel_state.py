"""Process-group state: one process per GPU, torch.distributed over RCCL.

MI355X-first: backend "nccl" IS RCCL on ROCm; xGMI gives
7 p2p links per GPU, so TP all-reduce bucket/algorithm choices live in
comm.py, not here. Multi-node bootstrap consumes the controller's
LWS_LEADER_ADDRESS / LWS_WORKER_INDEX contract (reference pkg/workload/
lws.go:36-38) via torchrun-style env rendezvous instead of Ray.
"""

from __future__ import annotations

import os
from typing import Optional

import torch
import torch.distributed as dist

_TP_GROUP: Optional[dist.ProcessGroup] = None
_TP_RANK = 0
_TP_WORLD = 1
_INITIALIZED = False


def init_distributed(
    tensor_parallel_size: int = 1,
    backend: str = "nccl",
    timeout_s: int = 600,
) -> None:
    """Initialize the global process group and the TP subgroup.

    Reads RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT from env (torchrun), or
    LWS_LEADER_ADDRESS when launched by the control plane's LWS wrapper.
    """
    global _TP_GROUP, _TP_RANK, _TP_WORLD, _INITIALIZED
    if _INITIALIZED:
        return
    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size > 1 and not dist.is_initialized():
        if "MASTER_ADDR" not in os.environ and "LWS_LEADER_ADDRESS" in os.environ:
            os.environ["MASTER_ADDR"] = os.environ["LWS_LEADER_ADDRESS"]
            os.environ.setdefault("MASTER_PORT", "29500")
        import datetime

        dist.init_process_group(
            backend=backend, timeout=datetime.timedelta(seconds=timeout_s)
        )
    rank = dist.get_rank() if dist.is_initialized() else 0
    world_size = dist.get_world_size() if dist.is_initialized() else 1

    tp = tensor_parallel_size
    assert world_size % tp == 0, (world_size, tp)
    if dist.is_initialized() and tp > 1:
        for start in range(0, world_size, tp):
            ranks = list(range(start, start + tp))
            grp = dist.new_group(ranks)
            if rank in ranks:
                _TP_GROUP = grp
    _TP_RANK = rank % tp
    _TP_WORLD = tp
    _INITIALIZED = True


def ensure_single_process() -> None:
    """Initialize TP=1 state without torch.distributed (tests, 1-GPU)."""
    global _INITIALIZED
    if not _INITIALIZED:
        init_distributed(1)


def tp_rank() -> int:
    return _TP_RANK


def tp_world_size() -> int:
    return _TP_WORLD


def tp_group():
    return _TP_GROUP


def destroy() -> None:
    global _TP_GROUP, _TP_RANK, _TP_WORLD, _INITIALIZED
    if dist.is_initialized():
        dist.destroy_process_group()
    _TP_GROUP, _TP_RANK, _TP_WORLD, _INITIALIZED = None, 0, 1, False


def tp_broadcast_object(obj=None):
    """Broadcast a picklable object from the TP group's first rank."""
    if _TP_WORLD == 1:
        return obj
    lst = [obj]
    src = (dist.get_rank() // _TP_WORLD) * _TP_WORLD
    dist.broadcast_object_list(lst, src=src, group=_TP_GROUP)
    return lst[0]


def tp_all_reduce_min_int(value: int) -> int:
    if _TP_WORLD == 1 or not dist.is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.int64)
    dist.all_reduce(t, op=dist.ReduceOp.MIN, group=_TP_GROUP)
    return int(t.item())


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    if _TP_WORLD == 1:
        return t
    if t.device.type == "cpu" and t.dtype == torch.bfloat16:
        # gloo (CPU test path) lacks bf16 reduction; RCCL path is native bf16
        f = t.float()
        dist.all_reduce(f, group=_TP_GROUP)
        t.copy_(f.to(t.dtype))
        return t
    dist.all_reduce(t, group=_TP_GROUP)
    return t
