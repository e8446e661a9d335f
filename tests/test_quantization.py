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
