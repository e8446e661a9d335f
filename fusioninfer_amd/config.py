"""Engine configuration.

The flag surface mirrors what the reference's controller passes to its
delegated engine via container args (reference
docs/.../core-design.md:88-209): --model, --tensor-parallel-size,
--kv-transfer-config {kv_connector, kv_role}, --max-model-len, ports.
"""

from __future__ import annotations

import dataclasses
from typing import Optional


@dataclasses.dataclass
class ModelConfig:
    """Architecture hyperparameters (decoder-only, Llama/Qwen family)."""

    name: str = "Qwen3-8B"
    hidden_size: int = 4096
    num_layers: int = 36
    num_heads: int = 32
    num_kv_heads: int = 8
    head_dim: int = 128
    intermediate_size: int = 12288
    vocab_size: int = 151936
    rope_theta: float = 1_000_000.0
    rms_norm_eps: float = 1e-6
    max_position_embeddings: int = 40960
    qk_norm: bool = True              # Qwen3 per-head q/k RMSNorm
    attention_bias: bool = False      # Qwen2.5-style qkv bias
    # HF-style rope_scaling dict (Llama-3.1 "llama3" remap, "linear")
    rope_scaling: Optional[dict] = None
    tie_word_embeddings: bool = False
    dtype: str = "bfloat16"
    # optional HF-layout safetensors checkpoint dir; None = random init
    model_path: Optional[str] = None
    # None = bf16; "fp8" = OCP e4m3 weights + dynamic per-token activations
    quantization: Optional[str] = None
    # MoE (Qwen3-MoE family): num_experts 0 = dense MLP
    num_experts: int = 0
    num_experts_per_tok: int = 8
    moe_intermediate_size: int = 0
    norm_topk_prob: bool = True

    @property
    def is_moe(self) -> bool:
        return self.num_experts > 0

    @property
    def q_size(self) -> int:
        return self.num_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.num_kv_heads * self.head_dim


@dataclasses.dataclass
class CacheConfig:
    block_size: int = 16
    num_gpu_blocks: Optional[int] = None   # None = derive from gpu_memory_utilization
    gpu_memory_utilization: float = 0.85
    enable_prefix_caching: bool = True
    # "auto" = bf16; "fp8" = OCP e4m3 at scale 1.0 (half the KV bytes ->
    # ~2x decode-attention bandwidth and 2x cache capacity; mirrors the
    # reference engines' --kv-cache-dtype fp8 surface)
    kv_cache_dtype: str = "auto"
    # CPU swap space for preemption (vLLM --swap-space surface): preempted
    # sequences park their KV in host memory and resume without recompute.
    # 0 = preemption-by-recompute only.
    swap_space_gb: float = 0.0


@dataclasses.dataclass
class SchedulerConfig:
    max_num_seqs: int = 256                # max sequences resident per step
    max_num_batched_tokens: int = 8192     # per-step token budget
    max_model_len: int = 8192
    # admission hysteresis: hold NEW prompts until this many prompt tokens
    # are waiting (or nothing is decoding), so most steps stay pure-decode
    # and take the hipGraph path; 0 = admit eagerly every step
    prefill_admission_tokens: int = 8192
    # aging escape for the hysteresis: never hold a prompt longer than
    # this (ms). Bounds the TTFT cost of batching. A/B on MI355X
    # (headline config): 50ms -> 42.8 req/s (TTFT 129ms), 150ms ->
    # 45.3 (+5.8%, TTFT 139ms), 300ms -> 44.9 (TTFT 146ms) — bigger
    # admission batches keep more steps on the pure-decode hipGraph
    # path and run the prefill GEMMs at larger M.
    # Fixed 150 ms default — measured best across Qwen3-8B bf16/fp8 and
    # 30B MoE (+5.8% goodput over 50 ms at +10 ms p50 TTFT). None =
    # ADAPTIVE: ~4 engine steps (EMA step time) clamped to [50, 250] ms;
    # measured -8% goodput on 30B MoE (window pinned at the 250 ms
    # clamp), so adaptive is opt-in.
    prefill_admission_ms: Optional[float] = 150.0
    # "fcfs" (default) or "priority" (vLLM --scheduling-policy): priority
    # orders admission by (priority, arrival) and preempts the
    # lowest-priority running sequence first (lower value = higher prio)
    policy: str = "fcfs"


@dataclasses.dataclass
class ParallelConfig:
    tensor_parallel_size: int = 1
    pipeline_parallel_size: int = 1
    data_parallel_size: int = 1
    rank: int = 0
    world_size: int = 1
    distributed_backend: str = "nccl"      # "nccl" is RCCL on ROCm; tests use "gloo"


@dataclasses.dataclass
class KVTransferConfig:
    """PD-disaggregation connector config.

    Mirrors the reference's --kv-transfer-config
    '{"kv_connector":"PyNcclConnector","kv_role":"kv_producer"|"kv_consumer"}'
    (reference docs/.../core-design.md:88-111). Our connector moves packed
    KV blocks with RCCL send/recv over xGMI.
    """

    kv_connector: Optional[str] = None     # e.g. "RcclConnector"
    kv_role: Optional[str] = None          # "kv_producer" | "kv_consumer"
    kv_rank: int = 0
    kv_world_size: int = 2


@dataclasses.dataclass
class EngineConfig:
    model: ModelConfig = dataclasses.field(default_factory=ModelConfig)
    cache: CacheConfig = dataclasses.field(default_factory=CacheConfig)
    scheduler: SchedulerConfig = dataclasses.field(default_factory=SchedulerConfig)
    parallel: ParallelConfig = dataclasses.field(default_factory=ParallelConfig)
    kv_transfer: KVTransferConfig = dataclasses.field(default_factory=KVTransferConfig)
    # vLLM --speculative-config parity: None = off; set to a
    # SpeculativeConfig (engine/spec_decode.py) for ngram prompt-lookup
    # or draft_model speculation
    speculative: Optional[object] = None
    # LoRA capacity tiering (vLLM --max-loras / --max-cpu-loras):
    # device-resident cap and total (host-parked) cap
    max_loras: int = 8
    max_cpu_loras: int = 16
    seed: int = 0
    enforce_eager: bool = False            # True disables hipGraph decode capture
