"""Multi-adapter LoRA serving.

Capability parity with the reference's `--enable-lora --max-loras
--max-cpu-loras` engine flags and lora-affinity routing (SURVEY.md §2.3;
reference docs/.../runtime-profile.md:370-377, pkg/router/strategy.go:100-113).

Adapters target the four projection GEMMs (qkv, o, gate_up, down). Batched
application groups the step's tokens by adapter and adds
x[rows] @ A^T @ B^T (two skinny GEMMs per adapter per target) to the base
projection output — cheap for the usual 0-2 live adapters per batch.
TP sharding follows the base weights: B is sliced on column-parallel
outputs, A on row-parallel inputs.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from fusioninfer_amd.distributed import parallel_state as ps

TARGETS = ("qkv", "o", "gate_up", "down")


class LoRAAdapter:
    def __init__(self, name: str, rank: int, alpha: float, model_cfg,
                 dtype=torch.bfloat16, device="cpu", seed: Optional[int] = None):
        self.name = name
        self.rank = rank
        self.scaling = alpha / rank
        tp = ps.tp_world_size()
        tpr = ps.tp_rank()
        H = model_cfg.hidden_size
        qkv_out = (model_cfg.num_heads + 2 * model_cfg.num_kv_heads) * model_cfg.head_dim
        inter = model_cfg.intermediate_size
        gen = torch.Generator(device="cpu")
        if seed is not None:
            gen.manual_seed(seed)
        # weights[layer][target] = (A [r, in_shard], B [out_shard, r])
        self.weights: List[Dict[str, Tuple[torch.Tensor, torch.Tensor]]] = []
        for _ in range(model_cfg.num_layers):
            layer = {}
            for tgt, inf, outf, col in (
                ("qkv", H, qkv_out, True),
                ("o", model_cfg.num_heads * model_cfg.head_dim, H, False),
                ("gate_up", H, 2 * inter, True),
                ("down", inter, H, False),
            ):
                A = torch.randn(rank, inf, generator=gen, dtype=torch.float32) * 0.02
                # seeded B std 0.1: synthetic adapters must move logits
                # enough to flip greedy argmax on random-init base weights
                # (0.02 was within sampler tie-break noise on some seeds)
                B = torch.zeros(outf, rank) if seed is None else (
                    torch.randn(outf, rank, generator=gen, dtype=torch.float32) * 0.1
                )
                if col:  # column-parallel: shard B rows (per-segment is an
                    # approximation: qkv/gate_up segments are contiguous per
                    # rank in the base layout, so shard each segment)
                    B = self._shard_col_segments(B, tgt, model_cfg, tp, tpr)
                else:   # row-parallel: shard A input dim
                    per = inf // tp
                    A = A[:, tpr * per : (tpr + 1) * per]
                layer[tgt] = (
                    A.to(dtype=dtype, device=device),
                    B.to(dtype=dtype, device=device),
                )
            self.weights.append(layer)

    @staticmethod
    def _shard_col_segments(B, tgt, cfg, tp, tpr):
        if tp == 1:
            return B
        if tgt == "qkv":
            sizes = [
                cfg.num_heads * cfg.head_dim,
                cfg.num_kv_heads * cfg.head_dim,
                cfg.num_kv_heads * cfg.head_dim,
            ]
        else:  # gate_up
            sizes = [cfg.intermediate_size, cfg.intermediate_size]
        parts = []
        off = 0
        for sz in sizes:
            per = sz // tp
            parts.append(B[off + tpr * per : off + (tpr + 1) * per])
            off += sz
        return torch.cat(parts, dim=0)

    def to(self, device):
        for layer in self.weights:
            for tgt in list(layer):
                A, B = layer[tgt]
                layer[tgt] = (A.to(device), B.to(device))
        return self


class LoRABatch:
    """Per-step grouping: [(adapter, row_index_tensor)]."""

    def __init__(self, groups: List[Tuple[LoRAAdapter, torch.Tensor]]):
        self.groups = groups

    def apply(self, layer_idx: int, target: str, x: torch.Tensor,
              out: torch.Tensor) -> None:
        """out[rows] += scaling * x[rows] @ A^T @ B^T (in place)."""
        for adapter, rows in self.groups:
            A, B = adapter.weights[layer_idx][target]
            xr = x[rows]
            delta = (xr @ A.T) @ B.T
            out[rows] += adapter.scaling * delta


class LoRARegistry:
    def __init__(self, max_loras: int = 8):
        self.max_loras = max_loras
        self._adapters: Dict[str, LoRAAdapter] = {}

    def add(self, adapter: LoRAAdapter) -> None:
        if len(self._adapters) >= self.max_loras and adapter.name not in self._adapters:
            raise RuntimeError(f"max_loras={self.max_loras} reached")
        self._adapters[adapter.name] = adapter

    def get(self, name: str) -> LoRAAdapter:
        return self._adapters[name]

    def maybe_get(self, name: Optional[str]) -> Optional[LoRAAdapter]:
        return self._adapters.get(name) if name else None

    def names(self) -> List[str]:
        return sorted(self._adapters)
