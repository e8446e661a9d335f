"""Checkpoint loading tests: HF-layout safetensors roundtrip."""

import torch

from fusioninfer_amd.distributed import parallel_state as ps
from fusioninfer_amd.models.model import CausalLM
from fusioninfer_amd.models.registry import get_model_config
from fusioninfer_amd.models.weight_loader import (
    export_hf_state_dict,
    load_hf_state_dict,
    load_safetensors_dir,
)


def _model(seed):
    ps.ensure_single_process()
    torch.manual_seed(seed)
    return CausalLM(get_model_config("tiny-qwen3")).eval()


def _forward_logits(model, tokens):
    from fusioninfer_amd.engine.metadata import AttnMetadata

    T = len(tokens)
    nblk = (T + 15) // 16
    kv = [
        (torch.zeros(nblk + 1, 2, 16, 64, dtype=torch.bfloat16),
         torch.zeros(nblk + 1, 2, 16, 64, dtype=torch.bfloat16))
        for _ in range(model.cfg.num_layers)
    ]
    meta = AttnMetadata(
        num_prefill_tokens=T,
        num_decode_tokens=0,
        positions=torch.arange(T, dtype=torch.int32),
        slot_mapping=torch.arange(T, dtype=torch.int32),
        cu_seqlens=torch.tensor([0, T], dtype=torch.int32),
        prefill_block_tables=torch.arange(nblk, dtype=torch.int32).unsqueeze(0),
        prefill_seq_lens_k=torch.tensor([T], dtype=torch.int32),
    )
    with torch.no_grad():
        hidden = model(torch.tensor(tokens), meta, kv)
        return model.compute_logits(hidden[-1:])


def test_hf_state_dict_roundtrip():
    src = _model(1)
    dst = _model(2)
    tokens = [3, 1, 4, 1, 5] * 6
    before = _forward_logits(dst, tokens)
    sd = export_hf_state_dict(src)
    assert "model.layers.0.self_attn.q_norm.weight" in sd  # qwen3 qk-norm
    n = load_hf_state_dict(dst, sd.items())
    assert n == len(sd)
    after = _forward_logits(dst, tokens)
    expected = _forward_logits(src, tokens)
    assert not torch.allclose(before.float(), expected.float())
    torch.testing.assert_close(after.float(), expected.float())


def test_safetensors_dir_loading(tmp_path):
    from safetensors.torch import save_file

    src = _model(3)
    sd = export_hf_state_dict(src)
    # split across two files like real HF checkpoints
    keys = sorted(sd)
    half = len(keys) // 2
    save_file({k: sd[k] for k in keys[:half]}, str(tmp_path / "model-1.safetensors"))
    save_file({k: sd[k] for k in keys[half:]}, str(tmp_path / "model-2.safetensors"))
    dst = _model(4)
    n = load_safetensors_dir(dst, str(tmp_path))
    assert n == len(sd)
    tokens = [9, 8, 7] * 5
    torch.testing.assert_close(
        _forward_logits(dst, tokens).float(),
        _forward_logits(src, tokens).float(),
    )


def test_fp8_kv_scale_keys_load():
    """Calibrated fp8-KV dequant scales land on the attention module
    (vLLM checkpoint key spellings, incl. the legacy combined kv_scale)."""
    m = _model(5)
    n = load_hf_state_dict(m, [
        ("model.layers.0.self_attn.attn.k_scale", torch.tensor(2.5)),
        ("model.layers.0.self_attn.attn.v_scale", torch.tensor([1.75])),
        ("model.layers.1.self_attn.kv_scale", torch.tensor(3.0)),
    ])
    assert n == 3
    assert m.layers[0].self_attn.k_scale == 2.5
    assert m.layers[0].self_attn.v_scale == 1.75
    assert m.layers[1].self_attn.k_scale == 3.0
    assert m.layers[1].self_attn.v_scale == 3.0
    # untouched layers keep the 1.0 default
    if len(m.layers) > 2:
        assert m.layers[2].self_attn.k_scale == 1.0


def test_moe_expert_weights_roundtrip():
    """HF qwen3_moe expert names load into the stacked expert tensors and
    change the model's output."""
    import torch

    from fusioninfer_amd.distributed import parallel_state as ps
    from fusioninfer_amd.models.model import CausalLM
    from fusioninfer_amd.models.registry import get_model_config
    from fusioninfer_amd.models.weight_loader import load_hf_state_dict

    ps.ensure_single_process()
    torch.manual_seed(0)
    cfg = get_model_config("tiny-qwen3-moe")
    model = CausalLM(cfg)
    mlp = model.layers[0].mlp
    E, I, H = cfg.num_experts, cfg.moe_intermediate_size, cfg.hidden_size
    tensors = []
    torch.manual_seed(99)
    want_gu = torch.randn(E, 2 * I, H) * 0.03
    want_down = torch.randn(E, H, I) * 0.03
    want_router = torch.randn(E, H) * 0.03
    tensors.append(("model.layers.0.mlp.gate.weight", want_router))
    for e in range(E):
        tensors.append(
            (f"model.layers.0.mlp.experts.{e}.gate_proj.weight", want_gu[e, :I])
        )
        tensors.append(
            (f"model.layers.0.mlp.experts.{e}.up_proj.weight", want_gu[e, I:])
        )
        tensors.append(
            (f"model.layers.0.mlp.experts.{e}.down_proj.weight", want_down[e])
        )
    n = load_hf_state_dict(model, tensors)
    assert n == 1 + 3 * E
    assert torch.allclose(
        mlp.gate_up_t.float(), want_gu.transpose(1, 2).bfloat16().float()
    )
    assert torch.allclose(
        mlp.down_t.float(), want_down.transpose(1, 2).bfloat16().float()
    )
    assert torch.allclose(
        mlp.router_weight.float(), want_router.bfloat16().float()
    )


def test_fp8_native_checkpoint_matches_quantize_on_load(tmp_path):
    """An fp8-native checkpoint (e4m3 weights + weight_scale, the HF fp8
    layout) must produce the SAME engine weights as loading the bf16
    checkpoint with --quantization fp8 (both use per-output-channel
    symmetric quantization), and fp8 mode must auto-enable."""
    from safetensors.torch import save_file

    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.weight_loader import checkpoint_is_fp8
    from fusioninfer_amd.quantization import quantize_weight_fp8

    ps.ensure_single_process()
    src = _model(5)
    sd = export_hf_state_dict(src)

    bf16_dir = tmp_path / "bf16"
    bf16_dir.mkdir()
    save_file({k: v.contiguous() for k, v in sd.items()},
              str(bf16_dir / "model.safetensors"))

    fp8_dir = tmp_path / "fp8"
    fp8_dir.mkdir()
    fp8_sd = {}
    for k, v in sd.items():
        if k.endswith("proj.weight"):
            w8, s = quantize_weight_fp8(v)
            fp8_sd[k] = w8.contiguous()
            fp8_sd[k + "_scale"] = s.contiguous()
        else:
            fp8_sd[k] = v.contiguous()
    save_file(fp8_sd, str(fp8_dir / "model.safetensors"))

    assert checkpoint_is_fp8(str(fp8_dir))
    assert not checkpoint_is_fp8(str(bf16_dir))

    def engine(path, quant):
        torch.manual_seed(9)
        mc = get_model_config("tiny-qwen3")
        mc.model_path = path
        mc.quantization = quant
        cfg = EngineConfig(
            model=mc,
            cache=CacheConfig(num_gpu_blocks=64),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=512, max_model_len=128
            ),
            seed=9,
        )
        return LLMEngine(cfg, device="cpu")

    eng_a = engine(str(bf16_dir), "fp8")          # quantize-on-load
    eng_b = engine(str(fp8_dir), None)            # fp8-native, auto-detect
    assert eng_b.cfg.model.quantization == "fp8"

    la = eng_a.runner.model.layers[0]
    lb = eng_b.runner.model.layers[0]
    assert torch.equal(
        la.self_attn.qkv_proj.weight.data.view(torch.uint8),
        lb.self_attn.qkv_proj.weight.data.view(torch.uint8),
    )
    torch.testing.assert_close(
        la.self_attn.qkv_proj.weight_scale, lb.self_attn.qkv_proj.weight_scale
    )
    torch.testing.assert_close(
        la.mlp.down_proj.weight_scale, lb.mlp.down_proj.weight_scale
    )

    prompt = [3, 1, 4, 1, 5, 9] * 4
    out_a = eng_a.generate([prompt], SamplingParams(max_tokens=5))[0]
    out_b = eng_b.generate([prompt], SamplingParams(max_tokens=5))[0]
    assert out_a.output_token_ids == out_b.output_token_ids


def test_qwen25_attention_bias_loading():
    """Qwen2.5-family qkv bias: zero-init bias is a no-op; loaded bias
    changes the logits and q/k/v biases fuse + shard like their weights."""
    ps.ensure_single_process()
    torch.manual_seed(4)
    mc = get_model_config("tiny-qwen3")
    mc.attention_bias = True
    mc.qk_norm = False
    model = CausalLM(mc).eval()
    tokens = [3, 1, 4, 1, 5] * 4
    before = _forward_logits(model, tokens)

    q_size = mc.num_heads * mc.head_dim
    kv = mc.num_kv_heads * mc.head_dim
    sd = {}
    for li in range(mc.num_layers):
        p = f"model.layers.{li}.self_attn."
        sd[p + "q_proj.bias"] = torch.randn(q_size, dtype=torch.bfloat16) * 0.1
        sd[p + "k_proj.bias"] = torch.randn(kv, dtype=torch.bfloat16) * 0.1
        sd[p + "v_proj.bias"] = torch.randn(kv, dtype=torch.bfloat16) * 0.1
    n = load_hf_state_dict(model, sd.items())
    assert n == len(sd)
    got = model.layers[0].self_attn.qkv_proj.bias.data
    expect = torch.cat([
        sd["model.layers.0.self_attn.q_proj.bias"],
        sd["model.layers.0.self_attn.k_proj.bias"],
        sd["model.layers.0.self_attn.v_proj.bias"],
    ])
    torch.testing.assert_close(got.float(), expect.float())
    after = _forward_logits(model, tokens)
    assert not torch.allclose(before.float(), after.float())

    # real Qwen2.5 registry entries exist and build
    assert get_model_config("Qwen2.5-7B").attention_bias
    assert get_model_config("Qwen2.5-32B").attention_bias
