"""LoRA serving tests (CPU, tiny model)."""

import torch

from fusioninfer_amd.engine.sequence import SamplingParams
from tests.test_engine import make_engine

PROMPT = [5, 3, 8, 1] * 10


def test_zero_b_adapter_is_identity():
    """B initialized to zero => adapter output == base model output."""
    torch.manual_seed(0)
    eng = make_engine()
    eng.add_lora("zero", rank=8, seed=None)  # B = 0
    base = eng.generate([PROMPT], SamplingParams(max_tokens=5))[0]
    tuned = eng.generate(
        [PROMPT], SamplingParams(max_tokens=5)
    )  # warm path sanity
    req = eng.add_request(PROMPT, SamplingParams(max_tokens=5), lora_name="zero")
    outs = {}
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                outs[o.request_id] = o
    assert outs[req].output_token_ids == base.output_token_ids


def test_nonzero_adapter_changes_output_only_for_its_requests():
    torch.manual_seed(0)
    eng = make_engine()
    eng.add_lora("tuned", rank=8, seed=42)  # random B: real deltas
    base = eng.generate([PROMPT], SamplingParams(max_tokens=6))[0]
    # batch a base request and an adapter request TOGETHER
    r_base = eng.add_request(PROMPT, SamplingParams(max_tokens=6))
    r_lora = eng.add_request(PROMPT, SamplingParams(max_tokens=6), lora_name="tuned")
    outs = {}
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                outs[o.request_id] = o
    assert outs[r_base].output_token_ids == base.output_token_ids
    assert outs[r_lora].output_token_ids != base.output_token_ids


def test_active_loras_listed_for_affinity_scorer():
    eng = make_engine()
    eng.add_lora("sql", rank=4)
    eng.add_lora("chat", rank=4)
    assert eng.active_loras() == ["chat", "sql"]


def _write_peft_adapter(d, model_cfg, r=4, alpha=8.0, layers=None,
                        projs=("self_attn.q_proj", "self_attn.v_proj",
                               "mlp.down_proj")):
    """Synthetic PEFT adapter dir (adapter_model.safetensors +
    adapter_config.json) in the HF naming layout."""
    import json

    from safetensors.torch import save_file

    torch.manual_seed(17)
    H = model_cfg.hidden_size
    inter = model_cfg.intermediate_size
    dims = {
        "self_attn.q_proj": (model_cfg.num_heads * model_cfg.head_dim, H),
        "self_attn.k_proj": (model_cfg.num_kv_heads * model_cfg.head_dim, H),
        "self_attn.v_proj": (model_cfg.num_kv_heads * model_cfg.head_dim, H),
        "self_attn.o_proj": (H, model_cfg.num_heads * model_cfg.head_dim),
        "mlp.gate_proj": (inter, H),
        "mlp.up_proj": (inter, H),
        "mlp.down_proj": (H, inter),
    }
    sd = {}
    for li in layers if layers is not None else range(model_cfg.num_layers):
        for p in projs:
            out, inn = dims[p]
            pre = f"base_model.model.model.layers.{li}.{p}"
            sd[pre + ".lora_A.weight"] = torch.randn(r, inn) * 0.05
            sd[pre + ".lora_B.weight"] = torch.randn(out, r) * 0.05
    d.mkdir(parents=True, exist_ok=True)
    save_file({k: v.contiguous() for k, v in sd.items()},
              str(d / "adapter_model.safetensors"))
    (d / "adapter_config.json").write_text(json.dumps({
        "r": r, "lora_alpha": alpha,
        "target_modules": sorted({p.split(".")[-1] for p in projs}),
    }))
    return sd


def test_peft_adapter_from_safetensors(tmp_path):
    """PEFT-format adapter loading: block-diagonal reconstruction of the
    merged qkv target matches applying q/v separately, and the engine
    output changes vs base (then reverts on unload)."""
    from fusioninfer_amd.lora import LoRAAdapter

    torch.manual_seed(0)
    eng = make_engine()
    mc = eng.cfg.model
    sd = _write_peft_adapter(tmp_path / "ad", mc)
    adapter = LoRAAdapter.from_safetensors("tuned", str(tmp_path / "ad"), mc)
    scale = 8.0 / 4.0

    # qkv block structure: rank rows [0:4] = q, [4:8] = v; manual check
    A, B = adapter.weights[0]["qkv"]
    x = torch.randn(3, mc.hidden_size).to(torch.bfloat16)
    got = (x @ A.T.to(x.dtype)) @ B.T.to(x.dtype)
    aq = sd["base_model.model.model.layers.0.self_attn.q_proj.lora_A.weight"]
    bq = sd["base_model.model.model.layers.0.self_attn.q_proj.lora_B.weight"]
    av = sd["base_model.model.model.layers.0.self_attn.v_proj.lora_A.weight"]
    bv = sd["base_model.model.model.layers.0.self_attn.v_proj.lora_B.weight"]
    q_size = mc.num_heads * mc.head_dim
    kv = mc.num_kv_heads * mc.head_dim
    want_q = scale * (x.float() @ aq.T @ bq.T)
    want_v = scale * (x.float() @ av.T @ bv.T)
    torch.testing.assert_close(got[:, :q_size].float(), want_q,
                               rtol=0.1, atol=0.05)
    torch.testing.assert_close(got[:, q_size + kv:].float(), want_v,
                               rtol=0.1, atol=0.05)
    # k segment untargeted -> zero contribution
    assert got[:, q_size: q_size + kv].abs().max() < 1e-3

    # engine: load -> output differs; unload -> back to base
    prompt = [3, 1, 4, 1, 5] * 5
    base = eng.generate([prompt], SamplingParams(max_tokens=5))[0]
    eng.add_lora_from_path("tuned", str(tmp_path / "ad"))
    assert "tuned" in eng.active_loras()
    rid = eng.add_request(prompt, SamplingParams(max_tokens=5),
                          lora_name="tuned")
    tuned = None
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished and o.request_id == rid:
                tuned = o
    assert tuned.output_token_ids != base.output_token_ids
    assert eng.remove_lora("tuned")
    again = eng.generate([prompt], SamplingParams(max_tokens=5))[0]
    assert again.output_token_ids == base.output_token_ids


def test_registry_lru_tiering():
    """--max-loras/--max-cpu-loras semantics: at most max_loras resident,
    LRU pages out, paging back in on use; max_cpu_loras hard-caps."""
    from fusioninfer_amd.lora import LoRAAdapter, LoRARegistry
    from fusioninfer_amd.models.registry import get_model_config

    mc = get_model_config("tiny-qwen3")
    reg = LoRARegistry(max_loras=2, max_cpu_loras=3, device="cpu")
    for n in ("a", "b", "c"):
        reg.add(LoRAAdapter(n, 2, 4.0, mc, seed=1))
    assert reg.num_resident() == 2          # 'a' was evicted (LRU)
    assert set(reg.names()) == {"a", "b", "c"}
    reg.get("a")                            # pages 'a' back in
    assert reg.num_resident() == 2
    import pytest as _p
    with _p.raises(RuntimeError):
        reg.add(LoRAAdapter("d", 2, 4.0, mc, seed=1))
    assert reg.remove("c")
    reg.add(LoRAAdapter("d", 2, 4.0, mc, seed=1))
    assert set(reg.names()) == {"a", "b", "d"}
