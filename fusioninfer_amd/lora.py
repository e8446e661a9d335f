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

    @classmethod
    def from_safetensors(cls, name: str, path: str, model_cfg,
                         dtype=torch.bfloat16, device="cpu"):
        """Load a PEFT adapter directory (adapter_model.safetensors +
        adapter_config.json). PEFT keeps one (A, B) pair per projection;
        the merged targets (qkv, gate_up) are reconstructed as a
        block-diagonal adapter of rank = sum of the member ranks:
        A = [A_q; A_k; A_v], B with each B_proj in its segment's rows and
        its own rank columns — mathematically identical to applying the
        projections separately. Per-projection alpha/r scaling is folded
        into B, so mixed configs compose; TP sharding then follows the
        base-weight layout (B by column segments, A on row-parallel
        inputs)."""
        import json as _json
        import os

        from safetensors.torch import safe_open

        cfg_path = os.path.join(path, "adapter_config.json")
        alpha_over_r = 1.0
        if os.path.isfile(cfg_path):
            with open(cfg_path) as f:
                pc = _json.load(f)
            if pc.get("r"):
                alpha_over_r = float(
                    pc.get("lora_alpha", pc["r"])
                ) / float(pc["r"])
        st_path = os.path.join(path, "adapter_model.safetensors")
        tensors: Dict[str, torch.Tensor] = {}
        with safe_open(st_path, framework="pt") as sf:
            for k in sf.keys():
                tensors[k] = sf.get_tensor(k)

        def pair(li: int, proj: str):
            for prefix in (
                f"base_model.model.model.layers.{li}.{proj}",
                f"model.layers.{li}.{proj}",
            ):
                a = tensors.get(prefix + ".lora_A.weight")
                b = tensors.get(prefix + ".lora_B.weight")
                if a is not None and b is not None:
                    return a.float(), b.float() * alpha_over_r
            return None

        tp = ps.tp_world_size()
        tpr = ps.tp_rank()
        H = model_cfg.hidden_size
        inter = model_cfg.intermediate_size
        self = cls.__new__(cls)
        self.name = name
        self.scaling = 1.0  # folded into B above
        self.rank = 0
        self.weights = []
        merged = {
            "qkv": (["self_attn.q_proj", "self_attn.k_proj",
                     "self_attn.v_proj"],
                    [model_cfg.num_heads * model_cfg.head_dim,
                     model_cfg.num_kv_heads * model_cfg.head_dim,
                     model_cfg.num_kv_heads * model_cfg.head_dim], H),
            "gate_up": (["mlp.gate_proj", "mlp.up_proj"],
                        [inter, inter], H),
            "o": (["self_attn.o_proj"],
                  [H], model_cfg.num_heads * model_cfg.head_dim),
            "down": (["mlp.down_proj"], [H], inter),
        }
        for li in range(model_cfg.num_layers):
            layer = {}
            for tgt, (projs, seg_sizes, inf) in merged.items():
                pairs = [pair(li, p) for p in projs]
                if not any(p is not None for p in pairs):
                    # adapter does not target this module: rank-1 zeros
                    layer[tgt] = (
                        torch.zeros(1, inf // (tp if tgt in ("o", "down")
                                               else 1), dtype=dtype,
                                    device=device),
                        torch.zeros(
                            sum(s // (tp if tgt in ("qkv", "gate_up")
                                      else 1) for s in seg_sizes), 1,
                            dtype=dtype, device=device),
                    )
                    continue
                ranks = [0 if p is None else p[0].shape[0] for p in pairs]
                R = sum(r for r in ranks) or 1
                A = torch.zeros(R, inf)
                out_total = sum(seg_sizes)
                B = torch.zeros(out_total, R)
                r_off = 0
                seg_off = 0
                for p, r, sz in zip(pairs, ranks, seg_sizes):
                    if p is not None:
                        a, b = p
                        A[r_off: r_off + r] = a
                        B[seg_off: seg_off + sz, r_off: r_off + r] = b
                        r_off += r
                    seg_off += sz
                self.rank = max(self.rank, R)
                if tgt in ("qkv", "gate_up"):
                    B = LoRAAdapter._shard_col_segments(
                        B, tgt, model_cfg, tp, tpr
                    )
                else:
                    per = inf // tp
                    A = A[:, tpr * per: (tpr + 1) * per]
                layer[tgt] = (A.to(dtype=dtype, device=device),
                              B.to(dtype=dtype, device=device))
            self.weights.append(layer)
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
    """Adapter store with vLLM's --max-loras / --max-cpu-loras tiering:
    at most `max_loras` adapters stay device-resident; the LRU ones park
    in host memory (up to `max_cpu_loras` total) and page back on use."""

    def __init__(self, max_loras: int = 8, max_cpu_loras: int = 16,
                 device: str = "cpu"):
        self.max_loras = max_loras
        self.max_cpu_loras = max(max_cpu_loras, max_loras)
        self.device = device
        self._adapters: Dict[str, LoRAAdapter] = {}
        self._resident: Dict[str, int] = {}  # name -> last-use tick
        self._tick = 0

    def _evict_lru_if_needed(self) -> None:
        while len(self._resident) > self.max_loras:
            lru = min(self._resident, key=self._resident.get)
            del self._resident[lru]
            if self.device != "cpu":
                self._adapters[lru].to("cpu")

    def add(self, adapter: LoRAAdapter) -> None:
        if (len(self._adapters) >= self.max_cpu_loras
                and adapter.name not in self._adapters):
            raise RuntimeError(f"max_cpu_loras={self.max_cpu_loras} reached")
        self._adapters[adapter.name] = adapter
        self._tick += 1
        self._resident[adapter.name] = self._tick
        self._evict_lru_if_needed()

    def get(self, name: str) -> LoRAAdapter:
        adapter = self._adapters[name]
        self._tick += 1
        if name not in self._resident:
            # page back in from host memory
            if self.device != "cpu":
                adapter.to(self.device)
            self._resident[name] = self._tick
            self._evict_lru_if_needed()
        else:
            self._resident[name] = self._tick
        return adapter

    def maybe_get(self, name: Optional[str]) -> Optional[LoRAAdapter]:
        return self.get(name) if name and name in self._adapters else None

    def names(self) -> List[str]:
        return sorted(self._adapters)

    def num_resident(self) -> int:
        return len(self._resident)

    def remove(self, name: str) -> bool:
        self._resident.pop(name, None)
        return self._adapters.pop(name, None) is not None
