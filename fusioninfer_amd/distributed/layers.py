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


def _full_weight(out_features: int, in_features: int, dtype, std: float = 0.02):
    """Draw the FULL logical matrix from the (seeded) global RNG, so every
    TP rank sees the same logical weights and shards are true slices —
    TP=N is numerically the same model as TP=1."""
    w = torch.empty(out_features, in_features, dtype=dtype)
    with torch.no_grad():
        w.normal_(0.0, std)
    return w


class ColumnParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the output dim."""

    def __init__(self, in_features: int, out_features: int, dtype=torch.bfloat16):
        super().__init__()
        tp = ps.tp_world_size()
        rank = ps.tp_rank()
        assert out_features % tp == 0, (out_features, tp)
        self.out_per_rank = out_features // tp
        full = _full_weight(out_features, in_features, dtype)
        shard = full[rank * self.out_per_rank : (rank + 1) * self.out_per_rank]
        self.weight = nn.Parameter(shard.contiguous(), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)


class MergedColumnParallelLinear(nn.Module):
    """Several column-parallel projections fused into one GEMM (qkv_proj,
    gate_up_proj). Each segment is sharded INDEPENDENTLY by rank so the
    per-rank output layout is [seg0_shard | seg1_shard | ...] — slicing a
    merged matrix contiguously would give rank 0 all of segment 0."""

    def __init__(self, in_features: int, out_sizes, dtype=torch.bfloat16,
                 bias: bool = False):
        super().__init__()
        tp = ps.tp_world_size()
        rank = ps.tp_rank()
        shards = []
        self.out_per_rank_sizes = []
        for out in out_sizes:
            assert out % tp == 0, (out, tp)
            per = out // tp
            full = _full_weight(out, in_features, dtype)
            shards.append(full[rank * per : (rank + 1) * per])
            self.out_per_rank_sizes.append(per)
        self.weight = nn.Parameter(
            torch.cat(shards, dim=0).contiguous(), requires_grad=False
        )
        # per-segment sharded bias (Qwen2.5 qkv); zero-init like HF does
        # for missing bias, overwritten by the checkpoint loader
        self.bias = (
            nn.Parameter(
                torch.zeros(sum(self.out_per_rank_sizes), dtype=dtype),
                requires_grad=False,
            )
            if bias
            else None
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight, self.bias)


class RowParallelLinear(nn.Module):
    """Y = X W^T with W sharded along the input dim; all-reduce on forward."""

    def __init__(self, in_features: int, out_features: int, dtype=torch.bfloat16):
        super().__init__()
        tp = ps.tp_world_size()
        rank = ps.tp_rank()
        assert in_features % tp == 0, (in_features, tp)
        self.in_per_rank = in_features // tp
        full = _full_weight(out_features, in_features, dtype)
        shard = full[:, rank * self.in_per_rank : (rank + 1) * self.in_per_rank]
        self.weight = nn.Parameter(shard.contiguous(), requires_grad=False)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        out = F.linear(x, self.weight)
        return ps.tp_all_reduce(out)

    def forward_with_lora(self, x, lora, layer_idx, target):
        """LoRA delta added BEFORE the TP all-reduce (A is input-sharded,
        so per-rank deltas are partial sums)."""
        out = F.linear(x, self.weight)
        if lora is not None:
            lora.apply(layer_idx, target, x, out)
        return ps.tp_all_reduce(out)


class ReplicatedLinear(nn.Module):
    def __init__(self, in_features: int, out_features: int, dtype=torch.bfloat16):
        super().__init__()
        self.weight = nn.Parameter(
            _full_weight(out_features, in_features, dtype), requires_grad=False
        )

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        return F.linear(x, self.weight)
