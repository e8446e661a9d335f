"""Named model architectures.

The reference exercises Qwen/Qwen3-8B (README.md:136-148) and multi-node
Llama-70B-class TP=8 (docs/.../user-guide/deployment.md:47-83); BASELINE.json
names both. Weights are random-initialized (no network in this environment).
"""

from __future__ import annotations

from fusioninfer_amd.config import ModelConfig

_REGISTRY = {}


def register(name: str, **kwargs) -> None:
    _REGISTRY[name.lower()] = ModelConfig(name=name, **kwargs)


register(
    "Qwen3-8B",
    hidden_size=4096,
    num_layers=36,
    num_heads=32,
    num_kv_heads=8,
    head_dim=128,
    intermediate_size=12288,
    vocab_size=151936,
    rope_theta=1_000_000.0,
    rms_norm_eps=1e-6,
    max_position_embeddings=40960,
    qk_norm=True,
)

register(
    "Qwen3-0.6B",
    hidden_size=1024,
    num_layers=28,
    num_heads=16,
    num_kv_heads=8,
    head_dim=128,
    intermediate_size=3072,
    vocab_size=151936,
    rope_theta=1_000_000.0,
    rms_norm_eps=1e-6,
    max_position_embeddings=40960,
    qk_norm=True,
)

register(
    "Qwen3-30B-A3B",
    hidden_size=2048,
    num_layers=48,
    num_heads=32,
    num_kv_heads=4,
    head_dim=128,
    intermediate_size=6144,          # dense fallback size (unused for MoE)
    vocab_size=151936,
    rope_theta=1_000_000.0,
    rms_norm_eps=1e-6,
    max_position_embeddings=40960,
    qk_norm=True,
    num_experts=128,
    num_experts_per_tok=8,
    moe_intermediate_size=768,
)

register(
    "Llama-3-8B",
    hidden_size=4096,
    num_layers=32,
    num_heads=32,
    num_kv_heads=8,
    head_dim=128,
    intermediate_size=14336,
    vocab_size=128256,
    rope_theta=500_000.0,
    rms_norm_eps=1e-5,
    max_position_embeddings=8192,
    qk_norm=False,
)

register(
    "Llama-3-70B",
    hidden_size=8192,
    num_layers=80,
    num_heads=64,
    num_kv_heads=8,
    head_dim=128,
    intermediate_size=28672,
    vocab_size=128256,
    rope_theta=500_000.0,
    rms_norm_eps=1e-5,
    max_position_embeddings=8192,
    qk_norm=False,
)

_LLAMA31_SCALING = {
    "rope_type": "llama3",
    "factor": 8.0,
    "low_freq_factor": 1.0,
    "high_freq_factor": 4.0,
    "original_max_position_embeddings": 8192,
}

register(
    "Llama-3.1-8B",
    hidden_size=4096,
    num_layers=32,
    num_heads=32,
    num_kv_heads=8,
    head_dim=128,
    intermediate_size=14336,
    vocab_size=128256,
    rope_theta=500_000.0,
    rms_norm_eps=1e-5,
    max_position_embeddings=131072,
    qk_norm=False,
    rope_scaling=_LLAMA31_SCALING,
)

register(
    "Llama-3.1-70B",
    hidden_size=8192,
    num_layers=80,
    num_heads=64,
    num_kv_heads=8,
    head_dim=128,
    intermediate_size=28672,
    vocab_size=128256,
    rope_theta=500_000.0,
    rms_norm_eps=1e-5,
    max_position_embeddings=131072,
    qk_norm=False,
    rope_scaling=_LLAMA31_SCALING,
)

register(
    "Qwen2.5-7B",
    hidden_size=3584,
    num_layers=28,
    num_heads=28,
    num_kv_heads=4,
    head_dim=128,
    intermediate_size=18944,
    vocab_size=152064,
    rope_theta=1_000_000.0,
    rms_norm_eps=1e-6,
    max_position_embeddings=32768,
    qk_norm=False,
    attention_bias=True,
)

register(
    "Qwen2.5-32B",
    hidden_size=5120,
    num_layers=64,
    num_heads=40,
    num_kv_heads=8,
    head_dim=128,
    intermediate_size=27648,
    vocab_size=152064,
    rope_theta=1_000_000.0,
    rms_norm_eps=1e-6,
    max_position_embeddings=32768,
    qk_norm=False,
    attention_bias=True,
)

# Tiny debug model: same code paths, CPU-testable sizes.
register(
    "tiny-qwen3",
    hidden_size=256,
    num_layers=2,
    num_heads=4,
    num_kv_heads=2,
    head_dim=64,
    intermediate_size=512,
    vocab_size=1024,
    rope_theta=10_000.0,
    rms_norm_eps=1e-6,
    max_position_embeddings=4096,
    qk_norm=True,
)

register(
    "tiny-qwen3-moe",
    hidden_size=256,
    num_layers=2,
    num_heads=4,
    num_kv_heads=2,
    head_dim=64,
    intermediate_size=512,
    vocab_size=1024,
    rope_theta=10_000.0,
    rms_norm_eps=1e-6,
    max_position_embeddings=4096,
    qk_norm=True,
    num_experts=8,
    num_experts_per_tok=2,
    moe_intermediate_size=128,
)


def get_model_config(name: str) -> ModelConfig:
    key = name.lower()
    if key not in _REGISTRY and "/" in key:
        # accept HF-style ids ("Qwen/Qwen3-8B" -> "Qwen3-8B")
        key = key.rsplit("/", 1)[1]
    if key not in _REGISTRY:
        raise KeyError(f"unknown model {name!r}; known: {sorted(_REGISTRY)}")
    import dataclasses

    return dataclasses.replace(_REGISTRY[key])


def list_models():
    return sorted(_REGISTRY)
