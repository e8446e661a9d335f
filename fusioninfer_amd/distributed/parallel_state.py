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
_PP_RANK = 0
_PP_WORLD = 1
_RANK = 0
_ENGINE_BASE = 0  # first global rank of this engine's tp*pp sub-world
_EXEC_GROUP = None  # the engine's whole pipeline (PP>1 sub-world group)
_INITIALIZED = False


def init_distributed(
    tensor_parallel_size: int = 1,
    backend: str = "nccl",
    timeout_s: int = 600,
    pipeline_parallel_size: int = 1,
) -> None:
    """Initialize the global process group and the TP subgroup.

    Reads RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT from env (torchrun), or
    LWS_LEADER_ADDRESS when launched by the control plane's LWS wrapper.

    TP x PP composition: world = pp * tp, stage = rank // tp (TP groups
    are stage-contiguous), and pipeline p2p moves column-wise — stage s
    tp-rank r talks to stage s+1 tp-rank r. PP-only (tp=1) reduces to
    rank == stage.
    """
    global _TP_GROUP, _TP_RANK, _TP_WORLD, _PP_RANK, _PP_WORLD, _RANK, \
        _ENGINE_BASE, _INITIALIZED
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
    _RANK = rank
    pp = pipeline_parallel_size
    if pp > 1:
        # world may hold SEVERAL tp*pp engines (PD: prefiller + decoder
        # halves); stage indices are sub-world-local
        assert world_size % (tp * pp) == 0, (world_size, tp, pp)
        _ENGINE_BASE = rank - rank % (tp * pp)
        _PP_RANK = (rank % (tp * pp)) // tp
        _PP_WORLD = pp
        # execution group = this engine's whole pipeline: broadcasts must
        # NOT use the default group when the world holds several engines
        # (PD composes prefiller + decoder sub-worlds)
        global _EXEC_GROUP
        if dist.is_initialized():
            sub = tp * pp
            for start in range(0, world_size, sub):
                ranks = list(range(start, start + sub))
                grp = dist.new_group(ranks)
                if rank in ranks:
                    _EXEC_GROUP = grp
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
    global _TP_GROUP, _TP_RANK, _TP_WORLD, _PP_RANK, _PP_WORLD, _RANK, \
        _ENGINE_BASE, _INITIALIZED
    if dist.is_initialized():
        dist.destroy_process_group()
    _TP_GROUP, _TP_RANK, _TP_WORLD, _INITIALIZED = None, 0, 1, False
    global _EXEC_GROUP
    _PP_RANK, _PP_WORLD, _RANK, _ENGINE_BASE = 0, 1, 0, 0
    _EXEC_GROUP = None


# --------------------------------------------------------------- pipeline
def pp_rank() -> int:
    return _PP_RANK


def pp_world_size() -> int:
    return _PP_WORLD


def pp_is_first() -> bool:
    return _PP_RANK == 0


def pp_is_last() -> bool:
    return _PP_RANK == _PP_WORLD - 1


def _gloo_safe_send(t: torch.Tensor, dst: int) -> None:
    if t.device.type == "cpu" and t.dtype == torch.bfloat16:
        dist.send(t.contiguous().view(torch.int16), dst)
    else:
        dist.send(t.contiguous(), dst)


def _gloo_safe_recv(t: torch.Tensor, src: int) -> None:
    if t.device.type == "cpu" and t.dtype == torch.bfloat16:
        buf = torch.empty(t.shape, dtype=torch.int16)
        dist.recv(buf, src)
        t.copy_(buf.view(torch.bfloat16))
    else:
        dist.recv(t, src)


def _stage_peer(stage: int) -> int:
    """Global rank of `stage`'s member in THIS rank's TP column
    (within this engine's sub-world — PD composes two engines)."""
    return _ENGINE_BASE + stage * _TP_WORLD + _TP_RANK


def pp_send_next(t: torch.Tensor) -> None:
    """Ship activations to the next pipeline stage (xGMI p2p under RCCL);
    column-wise under TP x PP (same tp-rank in the next stage)."""
    _gloo_safe_send(t, _RANK + _TP_WORLD)


def pp_recv_prev(shape, dtype, device) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    _gloo_safe_recv(t, _RANK - _TP_WORLD)
    return t


def pp_send_to(t: torch.Tensor, stage: int) -> None:
    _gloo_safe_send(t, _stage_peer(stage))


def pp_recv_from(shape, dtype, device, stage: int) -> torch.Tensor:
    t = torch.empty(shape, dtype=dtype, device=device)
    _gloo_safe_recv(t, _stage_peer(stage))
    return t


# ------------------------------------------- async p2p (microbatch overlap)
class _PendingSend:
    """isend handle; keeps the staging buffer alive until wait()."""

    def __init__(self, work, buf):
        self._work = work
        self._buf = buf

    def wait(self):
        self._work.wait()


class _PendingRecv:
    """irecv handle; wait() returns the received tensor."""

    def __init__(self, work, buf, bf16_view: bool):
        self._work = work
        self._buf = buf
        self._bf16 = bf16_view

    def wait(self) -> torch.Tensor:
        self._work.wait()
        return self._buf.view(torch.bfloat16) if self._bf16 else self._buf


def _gloo_safe_isend(t: torch.Tensor, dst: int) -> _PendingSend:
    if t.device.type == "cpu" and t.dtype == torch.bfloat16:
        buf = t.contiguous().view(torch.int16)
    else:
        buf = t.contiguous()
    return _PendingSend(dist.isend(buf, dst), buf)


def pp_isend_next(t: torch.Tensor) -> _PendingSend:
    return _gloo_safe_isend(t, _RANK + _TP_WORLD)


def pp_isend_to(t: torch.Tensor, stage: int) -> _PendingSend:
    return _gloo_safe_isend(t, _stage_peer(stage))


def pp_irecv_from(shape, dtype, device, stage: int) -> _PendingRecv:
    """Post a receive NOW (pre-posting on the driver is what lets pipeline
    microbatches drain without a send/recv deadlock) and collect later."""
    bf16_cpu = dtype == torch.bfloat16 and torch.device(device).type == "cpu"
    buf = torch.empty(shape, dtype=torch.int16 if bf16_cpu else dtype,
                      device=device)
    return _PendingRecv(dist.irecv(buf, _stage_peer(stage)), buf, bf16_cpu)


def tp_broadcast_object(obj=None):
    """Broadcast a picklable object from the execution group's first rank
    (the TP group, or the whole pipeline when PP > 1)."""
    if _PP_WORLD > 1:
        lst = [obj]
        dist.broadcast_object_list(lst, src=_ENGINE_BASE, group=_EXEC_GROUP)
        return lst[0]
    if _TP_WORLD == 1:
        return obj
    lst = [obj]
    src = (dist.get_rank() // _TP_WORLD) * _TP_WORLD
    dist.broadcast_object_list(lst, src=src, group=_TP_GROUP)
    return lst[0]


def tp_all_reduce_min_int(value: int) -> int:
    """Min over the execution group (TP ranks, or all pipeline stages)."""
    if not dist.is_initialized():
        return value
    if _PP_WORLD > 1:
        t = torch.tensor([value], dtype=torch.int64)
        dist.all_reduce(t, op=dist.ReduceOp.MIN, group=_EXEC_GROUP)
        return int(t.item())
    if _TP_WORLD == 1:
        return value
    t = torch.tensor([value], dtype=torch.int64)
    dist.all_reduce(t, op=dist.ReduceOp.MIN, group=_TP_GROUP)
    return int(t.item())


# One-shot threshold: xGMI is nearly fully-connected p2p (7 links/GPU),
# so for small latency-bound decode tensors an all-gather + local sum
# (1 p2p step) beats RCCL's ring all-reduce (2*(N-1) per-link steps).
# Large prefill tensors keep the bandwidth-optimal ring. 256 KiB ~= a
# decode batch of 32 rows x 4096 hidden in bf16.
_ONESHOT_MAX_BYTES = 256 * 1024


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    if _TP_WORLD == 1:
        return t
    if t.device.type == "cpu" and t.dtype == torch.bfloat16:
        # gloo (CPU test path) lacks bf16 reduction; RCCL path is native bf16
        f = t.float()
        dist.all_reduce(f, group=_TP_GROUP)
        t.copy_(f.to(t.dtype))
        return t
    if t.numel() * t.element_size() <= _ONESHOT_MAX_BYTES:
        gathered = torch.empty(
            _TP_WORLD * t.numel(), dtype=t.dtype, device=t.device
        )
        dist.all_gather_into_tensor(gathered, t.contiguous().view(-1),
                                    group=_TP_GROUP)
        torch.sum(gathered.view((_TP_WORLD,) + tuple(t.shape)), dim=0, out=t)
        return t
    dist.all_reduce(t, group=_TP_GROUP)
    return t
