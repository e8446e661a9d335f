"""Checkpoint loading: HF-layout safetensors -> the FusionInfer-AMD model.

The reference treats model weight materialization as design-doc-only
(node caches / warm-up jobs); the engine itself must still
be able to load real weights. This maps HuggingFace parameter names
(Qwen3 / Llama layout) onto the fused, TP-sharded modules:

  q_proj+k_proj+v_proj -> qkv_proj     (per-segment rank shard)
  gate_proj+up_proj    -> gate_up_proj (per-segment rank shard)
  o_proj / down_proj   -> row-parallel shard (input-dim slice)
"""

from __future__ import annotations

import os
from typing import Dict, Iterable, Tuple

import torch

from fusioninfer_amd.distributed import parallel_state as ps


def _shard_rows(w: torch.Tensor, tp: int, rank: int) -> torch.Tensor:
    per = w.shape[0] // tp
    return w[rank * per : (rank + 1) * per]


def _shard_cols(w: torch.Tensor, tp: int, rank: int) -> torch.Tensor:
    per = w.shape[1] // tp
    return w[:, rank * per : (rank + 1) * per]


def _row_scale(s: torch.Tensor, rows: int) -> torch.Tensor:
    """Normalize an fp8 weight_scale to per-output-row [rows] fp32:
    checkpoints carry either a per-tensor scalar or per-channel rows."""
    s = s.reshape(-1).float()
    if s.numel() == 1:
        return s.expand(rows).clone()
    assert s.numel() == rows, (s.numel(), rows)
    return s


def checkpoint_is_fp8(path: str) -> bool:
    """True when the safetensors checkpoint stores projection weights in
    OCP e4m3 (HF fp8 checkpoints ship `weight` f8 + `weight_scale`)."""
    from safetensors import safe_open

    files = sorted(
        f for f in os.listdir(path) if f.endswith(".safetensors")
    )
    if not files:
        return False
    with safe_open(os.path.join(path, files[0]), framework="pt") as sf:
        for name in sf.keys():
            if name.endswith("proj.weight"):
                return "F8" in str(sf.get_slice(name).get_dtype()).upper()
    return False


def load_hf_state_dict(
    model, tensors: Iterable[Tuple[str, torch.Tensor]]
) -> int:
    """Load HF-named tensors into a CausalLM. Returns #tensors consumed."""
    cfg = model.cfg
    tp = ps.tp_world_size()
    rank = ps.tp_rank()
    q_size = cfg.num_heads * cfg.head_dim
    kv_size = cfg.num_kv_heads * cfg.head_dim

    # staging for fused weights: (layer, which) -> tensor
    pending: Dict[Tuple[int, str], torch.Tensor] = {}
    moe_pending: Dict[Tuple[int, int, str], torch.Tensor] = {}
    loaded = 0

    def put(param: torch.nn.Parameter, value: torch.Tensor):
        assert param.data.shape == value.shape, (param.data.shape, value.shape)
        param.data.copy_(value.to(param.dtype))

    def _is_fp8(w: torch.Tensor) -> bool:
        return w.dtype == torch.float8_e4m3fn

    def try_fuse_qkv(li: int):
        keys = [(li, "q"), (li, "k"), (li, "v")]
        if not all(k in pending for k in keys):
            return
        fp8 = _is_fp8(pending[keys[0]])
        skeys = [(li, "q_scale"), (li, "k_scale"), (li, "v_scale")]
        if fp8 and not all(k in pending for k in skeys):
            return  # wait for the weight_scale tensors
        qw = pending.pop(keys[0])
        kw = pending.pop(keys[1])
        vw = pending.pop(keys[2])
        attn = local(li).self_attn
        w = torch.cat([
            _shard_rows(qw, tp, rank),
            _shard_rows(kw, tp, rank),
            _shard_rows(vw, tp, rank),
        ], dim=0)
        if fp8:
            sc = torch.cat([
                _shard_rows(_row_scale(pending.pop(skeys[0]), qw.shape[0]),
                            tp, rank),
                _shard_rows(_row_scale(pending.pop(skeys[1]), kw.shape[0]),
                            tp, rank),
                _shard_rows(_row_scale(pending.pop(skeys[2]), vw.shape[0]),
                            tp, rank),
            ])
            attn.qkv_proj.weight_scale.copy_(sc)
        put(attn.qkv_proj.weight, w)

    def try_fuse_qkv_bias(li: int):
        keys = [(li, "q_bias"), (li, "k_bias"), (li, "v_bias")]
        if not all(k in pending for k in keys):
            return
        attn = local(li).self_attn
        b = torch.cat([
            _shard_rows(pending.pop(k), tp, rank) for k in keys
        ])
        assert attn.qkv_proj.bias is not None, (
            "checkpoint has qkv bias but the model config lacks "
            "attention_bias=True"
        )
        put(attn.qkv_proj.bias, b)

    def try_fuse_gate_up(li: int):
        keys = [(li, "gate"), (li, "up")]
        if not all(k in pending for k in keys):
            return
        fp8 = _is_fp8(pending[keys[0]])
        skeys = [(li, "gate_scale"), (li, "up_scale")]
        if fp8 and not all(k in pending for k in skeys):
            return
        gw = pending.pop(keys[0])
        uw = pending.pop(keys[1])
        mlp = local(li).mlp
        w = torch.cat([
            _shard_rows(gw, tp, rank), _shard_rows(uw, tp, rank)
        ], dim=0)
        if fp8:
            sc = torch.cat([
                _shard_rows(_row_scale(pending.pop(skeys[0]), gw.shape[0]),
                            tp, rank),
                _shard_rows(_row_scale(pending.pop(skeys[1]), uw.shape[0]),
                            tp, rank),
            ])
            mlp.gate_up_proj.weight_scale.copy_(sc)
        put(mlp.gate_up_proj.weight, w)

    def local(li: int):
        # PP: global layer index -> this stage's slice
        return model.layers[li - model.layer_start]

    def owns(li: int) -> bool:
        return model.layer_start <= li < model.layer_end

    def try_fuse_expert(li: int, e: int):
        keys = [(li, e, "gate"), (li, e, "up")]
        if all(k in moe_pending for k in keys):
            mlp = local(li).mlp
            le = e - mlp.e_start
            gw = moe_pending.pop(keys[0])
            uw = moe_pending.pop(keys[1])
            # experts are stored pre-transposed ([H, 2I]) — see MoEMLP
            mlp.gate_up_t.data[le].copy_(
                torch.cat([gw, uw], dim=0).T.to(mlp.gate_up_t.dtype)
            )

    def try_fuse_expert_fp8(li: int, e: int, le: int, mlp):
        """fp8-mode expert slots: fp8-native checkpoint tensors load
        directly (weight + weight_scale); bf16 checkpoint experts are
        quantized on the way in."""
        from fusioninfer_amd.quantization import quantize_weight_fp8

        dk = (li, e, "down.weight")
        if dk in moe_pending:
            w = moe_pending[dk]
            if w.dtype == torch.float8_e4m3fn:
                sk = (li, e, "down.weight_scale")
                if sk in moe_pending:
                    moe_pending.pop(dk)
                    s = _row_scale(moe_pending.pop(sk), w.shape[0])
                    mlp.down_fp8.data[le].copy_(w)
                    mlp.down_scale[le].copy_(s)
            else:
                moe_pending.pop(dk)
                w8, s = quantize_weight_fp8(w)
                mlp.down_fp8.data[le].copy_(w8)
                mlp.down_scale[le].copy_(s)
        gk, uk = (li, e, "gate.weight"), (li, e, "up.weight")
        if gk in moe_pending and uk in moe_pending:
            if moe_pending[gk].dtype == torch.float8_e4m3fn:
                gsk = (li, e, "gate.weight_scale")
                usk = (li, e, "up.weight_scale")
                if gsk in moe_pending and usk in moe_pending:
                    g = moe_pending.pop(gk)
                    u = moe_pending.pop(uk)
                    gs = _row_scale(moe_pending.pop(gsk), g.shape[0])
                    us = _row_scale(moe_pending.pop(usk), u.shape[0])
                    mlp.gate_up_fp8.data[le].copy_(torch.cat([g, u], dim=0))
                    mlp.gate_up_scale[le].copy_(torch.cat([gs, us]))
            else:
                g = moe_pending.pop(gk)
                u = moe_pending.pop(uk)
                w8, s = quantize_weight_fp8(torch.cat([g, u], dim=0))
                mlp.gate_up_fp8.data[le].copy_(w8)
                mlp.gate_up_scale[le].copy_(s)

    for name, w in tensors:
        loaded += 1
        if name == "model.embed_tokens.weight":
            if model.embed_tokens is not None:
                put(model.embed_tokens, w)
        elif name == "model.norm.weight":
            if model.final_norm_weight is not None:
                put(model.final_norm_weight, w)
        elif name == "lm_head.weight":
            if model.lm_head is not None:
                put(model.lm_head.weight, w)
        elif name.startswith("model.layers."):
            parts = name.split(".")
            li = int(parts[2])
            if not owns(li):
                continue
            rest = ".".join(parts[3:])
            layer = local(li)
            if rest == "self_attn.q_proj.weight":
                pending[(li, "q")] = w
                try_fuse_qkv(li)
            elif rest == "self_attn.k_proj.weight":
                pending[(li, "k")] = w
                try_fuse_qkv(li)
            elif rest == "self_attn.v_proj.weight":
                pending[(li, "v")] = w
                try_fuse_qkv(li)
            elif rest in ("self_attn.q_proj.bias",
                          "self_attn.k_proj.bias",
                          "self_attn.v_proj.bias"):
                pending[(li, rest.split(".")[1][0] + "_bias")] = w
                try_fuse_qkv_bias(li)
            elif rest in ("self_attn.q_proj.weight_scale",
                          "self_attn.k_proj.weight_scale",
                          "self_attn.v_proj.weight_scale"):
                pending[(li, rest.split(".")[1][0] + "_scale")] = w
                try_fuse_qkv(li)
            elif rest == "self_attn.o_proj.weight":
                put(layer.self_attn.o_proj.weight, _shard_cols(w, tp, rank))
            elif rest == "self_attn.o_proj.weight_scale":
                o = layer.self_attn.o_proj
                o.weight_scale.copy_(_row_scale(w, o.weight.shape[0]))
            elif rest in ("self_attn.attn.k_scale", "self_attn.k_scale"):
                # calibrated fp8-KV dequant scale (vLLM checkpoint keys)
                layer.self_attn.k_scale = float(w.reshape(-1)[0])
            elif rest in ("self_attn.attn.v_scale", "self_attn.v_scale"):
                layer.self_attn.v_scale = float(w.reshape(-1)[0])
            elif rest in ("self_attn.attn.kv_scale", "self_attn.kv_scale"):
                # legacy combined K/V scale
                layer.self_attn.k_scale = float(w.reshape(-1)[0])
                layer.self_attn.v_scale = float(w.reshape(-1)[0])
            elif rest == "self_attn.q_norm.weight":
                put(layer.self_attn.q_norm_weight, w)
            elif rest == "self_attn.k_norm.weight":
                put(layer.self_attn.k_norm_weight, w)
            elif rest == "mlp.gate_proj.weight":
                pending[(li, "gate")] = w
                try_fuse_gate_up(li)
            elif rest == "mlp.up_proj.weight":
                pending[(li, "up")] = w
                try_fuse_gate_up(li)
            elif rest == "mlp.gate_proj.weight_scale":
                pending[(li, "gate_scale")] = w
                try_fuse_gate_up(li)
            elif rest == "mlp.up_proj.weight_scale":
                pending[(li, "up_scale")] = w
                try_fuse_gate_up(li)
            elif rest == "mlp.down_proj.weight":
                put(layer.mlp.down_proj.weight, _shard_cols(w, tp, rank))
            elif rest == "mlp.down_proj.weight_scale":
                d = layer.mlp.down_proj
                d.weight_scale.copy_(_row_scale(w, d.weight.shape[0]))
            elif rest == "mlp.gate.weight":  # MoE router (qwen3_moe)
                put(layer.mlp.router_weight, w)
            elif parts[3] == "mlp" and parts[4] == "experts":
                e = int(parts[5])
                mlp = layer.mlp
                if not (mlp.e_start <= e < mlp.e_end):
                    continue  # EP: expert lives on another rank
                if hasattr(mlp, "ensure_unpacked"):
                    mlp.ensure_unpacked()  # no-op unless released
                which = parts[6].replace("_proj", "")  # gate/up/down
                kind = parts[7] if len(parts) > 7 else "weight"
                le = e - mlp.e_start
                if mlp.fp8:
                    moe_pending[(li, e, which + "." + kind)] = w
                    try_fuse_expert_fp8(li, e, le, mlp)
                elif which == "down":
                    mlp.down_t.data[le].copy_(w.T.to(mlp.down_t.dtype))
                else:
                    moe_pending[(li, e, which)] = w
                    try_fuse_expert(li, e)
            elif rest == "input_layernorm.weight":
                put(layer.input_norm_weight, w)
            elif rest == "post_attention_layernorm.weight":
                put(layer.post_norm_weight, w)
            else:
                loaded -= 1  # unknown (bias-less models shouldn't hit this)
        else:
            loaded -= 1
    assert not pending, f"unfused partial weights remain: {list(pending)}"
    assert not moe_pending, (
        f"unfused expert weights remain: {list(moe_pending)[:4]}"
    )
    # .data mutations don't bump tensor._version — explicitly invalidate
    # the MFMA-packed expert-weight cache of every MoE layer
    for layer in model.layers:
        if hasattr(layer.mlp, "invalidate_packed"):
            layer.mlp.invalidate_packed()
    return loaded


def load_safetensors_dir(model, path: str) -> int:
    """Load all *.safetensors under `path` (HF checkpoint directory)."""
    from safetensors.torch import safe_open

    files = sorted(
        os.path.join(path, f)
        for f in os.listdir(path)
        if f.endswith(".safetensors")
    )
    assert files, f"no safetensors files under {path}"

    def tensor_iter():
        for f in files:
            with safe_open(f, framework="pt") as sf:
                for name in sf.keys():
                    yield name, sf.get_tensor(name)

    return load_hf_state_dict(model, tensor_iter())


def export_hf_state_dict(model) -> Dict[str, torch.Tensor]:
    """Inverse mapping (TP=1 only) — used by tests and for producing
    checkpoints from random-init models."""
    assert ps.tp_world_size() == 1
    cfg = model.cfg
    q_size = cfg.num_heads * cfg.head_dim
    kv_size = cfg.num_kv_heads * cfg.head_dim
    out: Dict[str, torch.Tensor] = {
        "model.embed_tokens.weight": model.embed_tokens.data.clone(),
        "model.norm.weight": model.final_norm_weight.data.clone(),
    }
    if model.lm_head is not None:
        out["lm_head.weight"] = model.lm_head.weight.data.clone()
    for li, layer in enumerate(model.layers):
        p = f"model.layers.{li}."
        qkv = layer.self_attn.qkv_proj.weight.data
        out[p + "self_attn.q_proj.weight"] = qkv[:q_size].clone()
        out[p + "self_attn.k_proj.weight"] = qkv[q_size : q_size + kv_size].clone()
        out[p + "self_attn.v_proj.weight"] = qkv[q_size + kv_size :].clone()
        out[p + "self_attn.o_proj.weight"] = layer.self_attn.o_proj.weight.data.clone()
        if layer.self_attn.q_norm_weight is not None:
            out[p + "self_attn.q_norm.weight"] = layer.self_attn.q_norm_weight.data.clone()
            out[p + "self_attn.k_norm.weight"] = layer.self_attn.k_norm_weight.data.clone()
        gu = layer.mlp.gate_up_proj.weight.data
        inter = gu.shape[0] // 2
        out[p + "mlp.gate_proj.weight"] = gu[:inter].clone()
        out[p + "mlp.up_proj.weight"] = gu[inter:].clone()
        out[p + "mlp.down_proj.weight"] = layer.mlp.down_proj.weight.data.clone()
        out[p + "input_layernorm.weight"] = layer.input_norm_weight.data.clone()
        out[p + "post_attention_layernorm.weight"] = layer.post_norm_weight.data.clone()
    return out
