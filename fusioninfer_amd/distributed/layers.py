"""Tensor-parallel linear layers (Megatron-style sharding).

Column-parallel shards the output dim (no comm on forward); row-parallel
shards the input dim and all-reduces the partial outputs over RCCL/xGMI.
GEMMs go through torch.nn.functional.linear (hipBLASLt) — plain library
GEMMs; the fused hot ops are hand-written HIP (fusioninfer_amd/ops).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from fusioninfer_amd.distributed import parallel_state as ps


def _init_weight(out_features: int, in_features: int, dtype, std: float = 0.02):
    w = torch.empty(out_features, in_features, dtype=dtype)
    with torch.no_grad():
        w.normal_(0.0, std)
    return nn.Parameter(w, requires_grad=False)


class ColumnParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the output dim."""

    def __init__(self, in_features: int, out_features: int, dtype=torch.bfloat16):
        super().__init__()
        tp = ps.tp_world_size()
        assert out_features % tp == 0, (out_features, tp)
        self.out_per_rank = out_features // tp
        self.weight = _init_weight(self.out_per_rank, in_features, dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)


class RowParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the input dim; all-reduce on forward."""

    def __init__(self, in_features: int, out_features: int, dtype=torch.bfloat16):
        super().__init__()
        tp = ps.tp_world_size()
        assert in_features % tp == 0, (in_features, tp)
        self.in_per_rank = in_features // tp
        self.weight = _init_weight(out_features, self.in_per_rank, dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = F.linear(x, self.weight)
        return ps.tp_all_reduce(out)


class ReplicatedLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int, dtype=torch.bfloat16):
        super().__init__()
        self.weight = _init_weight(out_features, in_features, dtype)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)
