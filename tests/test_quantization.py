"""FP8 quantization tests (CPU dequant path; GPU scaled_mm path marked)."""

import pytest
import torch

from fusioninfer_amd.quantization import (
    fp8_linear,
    quantize_activation_fp8,
    quantize_weight_fp8,
)


def test_weight_quant_roundtrip_error():
    torch.manual_seed(0)
    w = torch.randn(64, 128, dtype=torch.bfloat16)
    w8, s = quantize_weight_fp8(w)
    wd = w8.float() * s.unsqueeze(1)
    rel = (wd - w.float()).abs().mean() / w.float().abs().mean()
    assert rel < 0.05


def test_fp8_linear_cpu_close_to_bf16():
    torch.manual_seed(1)
    x = torch.randn(16, 128, dtype=torch.bfloat16)
    w = torch.randn(64, 128, dtype=torch.bfloat16) * 0.05
    w8, s = quantize_weight_fp8(w)
    y8 = fp8_linear(x, w8, s).float()
    y = (x.float() @ w.float().t())
    rel = (y8 - y).norm() / y.norm()
    assert rel < 0.05


def test_engine_fp8_generates():
    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config

    torch.manual_seed(0)
    mc = get_model_config("tiny-qwen3")
    mc.quantization = "fp8"
    cfg = EngineConfig(
        model=mc,
        cache=CacheConfig(num_gpu_blocks=64),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    out = eng.generate([[3, 1, 4] * 8], SamplingParams(max_tokens=4))[0]
    assert len(out.output_token_ids) == 4


def test_fused_fp8_epilogue_references_match_eager_quant():
    """The fused CPU references (rms_norm_fp8 etc.) must agree with
    quantize_activation_fp8 applied to the bf16 op output."""
    import fusioninfer_amd.ops.reference as ref

    torch.manual_seed(3)
    x = torch.randn(8, 256, dtype=torch.bfloat16)
    w = torch.rand(256, dtype=torch.bfloat16) + 0.5
    out_bf16 = ref.rms_norm(x, w, 1e-6)
    f8, sc = ref.rms_norm_fp8(x, w, 1e-6)
    e8, es = quantize_activation_fp8(out_bf16)
    assert torch.equal(f8.float(), e8.float())
    assert torch.allclose(sc, es)

    res = torch.randn(8, 256, dtype=torch.bfloat16)
    out2, new_res = ref.fused_add_rms_norm(x, res.clone(), w, 1e-6)
    f8, sc, r2 = ref.fused_add_rms_norm_fp8(x, res.clone(), w, 1e-6)
    assert torch.equal(r2, new_res)
    e8, _ = quantize_activation_fp8(out2)
    assert torch.equal(f8.float(), e8.float())

    gu = torch.randn(8, 512, dtype=torch.bfloat16)
    f8, sc = ref.silu_and_mul_fp8(gu)
    e8, es = quantize_activation_fp8(ref.silu_and_mul(gu))
    assert torch.equal(f8.float(), e8.float())


def test_fp8_linear_accepts_prequantized_tuple():
    torch.manual_seed(4)
    x = torch.randn(16, 128, dtype=torch.bfloat16)
    w = torch.randn(64, 128, dtype=torch.bfloat16) * 0.05
    w8, s = quantize_weight_fp8(w)
    x8, xs = quantize_activation_fp8(x)
    y_tuple = fp8_linear((x8, xs), w8, s).float()
    y = x.float() @ w.float().t()
    rel = (y_tuple - y).norm() / y.norm()
    assert rel < 0.05


def test_engine_fp8_kv_cache_generates_close_to_bf16():
    """CPU reference path: fp8 KV cache engine output stays close to the
    bf16-KV engine with identical weights (exact token match is not
    guaranteed under e4m3 KV rounding, but greedy tokens on a short run
    should rarely diverge — assert the run completes and lengths match)."""
    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config

    def run(kv_dtype):
        torch.manual_seed(5)
        cfg = EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=64, kv_cache_dtype=kv_dtype),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
            ),
        )
        eng = LLMEngine(cfg, device="cpu")
        assert eng.runner.kv_caches[0][0].dtype == (
            torch.float8_e4m3fn if kv_dtype == "fp8" else torch.bfloat16
        )
        return eng.generate(
            [[2, 4, 6] * 8], SamplingParams(max_tokens=6, temperature=0.0,
                                            ignore_eos=True)
        )[0].output_token_ids

    toks_bf16 = run("auto")
    toks_fp8 = run("fp8")
    assert len(toks_fp8) == 6
    # same first token at minimum (single-step divergence would mean a bug,
    # not rounding: the first decode sees a cache written from identical
    # prefill activations)
    assert toks_fp8[0] == toks_bf16[0]


def test_lora_rejected_in_fp8_mode():
    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.models.registry import get_model_config

    mc = get_model_config("tiny-qwen3")
    mc.quantization = "fp8"
    cfg = EngineConfig(
        model=mc,
        cache=CacheConfig(num_gpu_blocks=32),
        scheduler=SchedulerConfig(
            max_num_seqs=2, max_num_batched_tokens=128, max_model_len=64
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    with pytest.raises(AssertionError):
        eng.add_lora("a", rank=4)


@pytest.mark.gpu
def test_fp8_scaled_mm_gpu():
    torch.manual_seed(2)
    x = torch.randn(32, 256, dtype=torch.bfloat16, device="cuda")
    w = torch.randn(128, 256, dtype=torch.bfloat16, device="cuda") * 0.05
    w8, s = quantize_weight_fp8(w)
    y8 = fp8_linear(x, w8, s).float()
    y = (x.float() @ w.float().t())
    rel = (y8 - y).norm() / y.norm()
    assert rel.item() < 0.05, rel.item()


def test_fp8_moe_engine_close_to_bf16():
    """fp8 MoE (per-expert _scaled_mm path; CPU reference dequant here):
    same seeded experts as bf16, outputs close, engine generates."""
    import torch

    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config

    def build(quant):
        torch.manual_seed(3)
        mc = get_model_config("tiny-qwen3-moe")
        mc.quantization = quant
        cfg = EngineConfig(
            model=mc,
            cache=CacheConfig(num_gpu_blocks=64),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=512, max_model_len=128
            ),
            seed=3,
        )
        return LLMEngine(cfg, device="cpu")

    prompt = [3, 1, 4, 1, 5, 9, 2, 6] * 3
    bf16 = build(None)
    ref = bf16.generate([prompt], SamplingParams(max_tokens=6))[0]
    fp8 = build("fp8")
    out = fp8.generate([prompt], SamplingParams(max_tokens=6))[0]
    assert len(out.output_token_ids) == 6
    # quantization perturbs logits; with a tiny random model greedy picks
    # can drift, so compare the MoE layer output directly instead
    layer_b = bf16.runner.model.layers[0].mlp
    layer_f = fp8.runner.model.layers[0].mlp
    x = torch.randn(5, bf16.cfg.model.hidden_size).to(torch.bfloat16)
    y_b = layer_b(x).float()
    from fusioninfer_amd.quantization import quantize_activation_fp8

    y_f = layer_f(quantize_activation_fp8(x)).float()
    rel = (y_b - y_f).norm() / y_b.norm().clamp(min=1e-6)
    assert rel < 0.15, f"fp8 MoE relative error {rel:.3f}"
