"""Pure-PyTorch fp32 reference implementations of every HIP op.

These are (a) the numerics oracle the GPU kernels are tested against and
(b) the CPU execution path for tests in GPU-less environments. They are
intentionally simple and allocate freely — never used on the GPU hot path
(ops/__init__.py raises if the HIP extension is missing on a GPU box).
"""

from __future__ import annotations

import torch


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    """RMSNorm over the last dim. Computes in fp32, returns x.dtype."""
    xf = x.float()
    var = xf.pow(2).mean(dim=-1, keepdim=True)
    out = xf * torch.rsqrt(var + eps) * weight.float()
    return out.to(x.dtype)


def fused_add_rms_norm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
):
    """residual = residual + x; out = rms_norm(residual). Returns (out, residual)."""
    new_residual = (residual.float() + x.float()).to(x.dtype)
    return rms_norm(new_residual, weight, eps), new_residual


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    """x: [..., 2*I] (gate | up) -> silu(gate) * up, in fp32."""
    gate, up = x.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(gate) * up).to(x.dtype)


def quant_fp8_rows(x: torch.Tensor):
    """Per-row dynamic OCP-e4m3 quantization: (x_fp8 [T, C], scale [T] fp32)."""
    xf = x.float()
    amax = xf.abs().amax(dim=-1, keepdim=True).clamp(min=1e-8)
    scale = amax / 448.0
    x_fp8 = (xf / scale).clamp(-448.0, 448.0).to(torch.float8_e4m3fn)
    return x_fp8, scale.squeeze(-1)


def rms_norm_fp8(x: torch.Tensor, weight: torch.Tensor, eps: float):
    return quant_fp8_rows(rms_norm(x, weight, eps))


def fused_add_rms_norm_fp8(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
):
    out, new_residual = fused_add_rms_norm(x, residual, weight, eps)
    x_fp8, scale = quant_fp8_rows(out)
    return x_fp8, scale, new_residual


def silu_and_mul_fp8(x: torch.Tensor):
    return quant_fp8_rows(silu_and_mul(x))


def _llama3_scale_inv_freq(inv_freq: torch.Tensor, sc: dict) -> torch.Tensor:
    """Llama-3.1 rope scaling (HF rope_scaling rope_type="llama3"):
    low-frequency bands are divided by `factor`, high-frequency bands
    kept, with a smooth ramp between (per-wavelength interpolation)."""
    import math

    factor = float(sc.get("factor", 8.0))
    low = float(sc.get("low_freq_factor", 1.0))
    high = float(sc.get("high_freq_factor", 4.0))
    orig = float(sc.get("original_max_position_embeddings", 8192))
    wavelen = 2 * math.pi / inv_freq
    low_wl = orig / low
    high_wl = orig / high
    scaled = torch.where(wavelen > low_wl, inv_freq / factor, inv_freq)
    smooth = (orig / wavelen - low) / (high - low)
    mid = (1 - smooth) * inv_freq / factor + smooth * inv_freq
    is_mid = (wavelen <= low_wl) & (wavelen >= high_wl)
    return torch.where(is_mid, mid, scaled)


def compute_cos_sin_cache(
    head_dim: int, max_positions: int, theta: float, device=None,
    rope_scaling=None,
) -> torch.Tensor:
    """[max_positions, head_dim] fp32; first half cos, second half sin (NeoX).
    rope_scaling: None, or an HF-style dict ({"rope_type": "llama3", ...}
    — the Llama-3.1 long-context frequency remap)."""
    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32, device=device) / head_dim)
    )
    if rope_scaling:
        kind = rope_scaling.get("rope_type") or rope_scaling.get("type")
        if kind == "llama3":
            inv_freq = _llama3_scale_inv_freq(inv_freq, rope_scaling)
        elif kind == "linear":
            inv_freq = inv_freq / float(rope_scaling.get("factor", 1.0))
        else:
            raise ValueError(f"unsupported rope_scaling {kind!r}")
    t = torch.arange(max_positions, dtype=torch.float32, device=device)
    freqs = torch.outer(t, inv_freq)  # [P, D/2]
    return torch.cat([freqs.cos(), freqs.sin()], dim=-1)


def _apply_rope_neox(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor) -> torch.Tensor:
    """x: [T, H, D]; cos/sin: [T, D/2]. Rotate-half (NeoX) style, fp32."""
    d2 = x.shape[-1] // 2
    x1, x2 = x[..., :d2], x[..., d2:]
    cos = cos.unsqueeze(1)
    sin = sin.unsqueeze(1)
    return torch.cat([x1 * cos - x2 * sin, x2 * cos + x1 * sin], dim=-1)


def rope_qk_norm(
    q: torch.Tensor,
    k: torch.Tensor,
    positions: torch.Tensor,
    cos_sin: torch.Tensor,
    q_weight=None,
    k_weight=None,
    eps: float = 1e-6,
):
    """Optional per-head RMSNorm (Qwen3 qk-norm) then NeoX RoPE.

    q: [T, Hq, D], k: [T, Hk, D], positions: [T] int. Returns new (q, k).
    """
    dtype = q.dtype
    qf, kf = q.float(), k.float()
    if q_weight is not None:
        qf = rms_norm(qf, q_weight, eps)
    if k_weight is not None:
        kf = rms_norm(kf, k_weight, eps)
    d2 = q.shape[-1] // 2
    cs = cos_sin[positions]  # [T, D]
    cos, sin = cs[:, :d2], cs[:, d2:]
    qf = _apply_rope_neox(qf.float(), cos, sin)
    kf = _apply_rope_neox(kf.float(), cos, sin)
    return qf.to(dtype), kf.to(dtype)


def reshape_and_cache(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
    k_inv_scale: float = 1.0,
    v_inv_scale: float = 1.0,
):
    """k/v: [T, Hk, D]; caches: [num_blocks, Hk, block_size, D]; slots: [T].
    fp8 caches quantize at the inverse per-layer static scale (matching
    the HIP kernel); bf16 caches store unscaled."""
    block_size = k_cache.shape[2]
    blk = torch.div(slot_mapping, block_size, rounding_mode="floor")
    off = slot_mapping % block_size
    if k_cache.dtype == torch.float8_e4m3fn:
        # clamp to +-448: v_cvt_pk_fp8_f32 SATURATES out-of-range values
        # where torch's .to(e4m3) would produce NaN
        k_cache[blk, :, off] = (k.float() * k_inv_scale).clamp(
            -448.0, 448.0).to(k_cache.dtype)
        v_cache[blk, :, off] = (v.float() * v_inv_scale).clamp(
            -448.0, 448.0).to(v_cache.dtype)
    else:
        k_cache[blk, :, off] = k.to(k_cache.dtype)
        v_cache[blk, :, off] = v.to(v_cache.dtype)


def paged_attention_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
) -> torch.Tensor:
    """One-token attention per sequence against the paged cache.

    q: [S, Hq, D]; caches: [B, Hk, bs, D]; block_tables: [S, max_blocks];
    seq_lens: [S] (context length INCLUDING the current token). fp32 math.
    """
    num_seqs, num_heads, head_dim = q.shape
    num_kv_heads = k_cache.shape[1]
    bs = k_cache.shape[2]
    group = num_heads // num_kv_heads
    out = torch.empty_like(q, dtype=torch.float32)
    for s in range(num_seqs):
        ctx = int(seq_lens[s])
        nblk = (ctx + bs - 1) // bs
        blocks = block_tables[s, :nblk].long()
        # [Hk, nblk*bs, D]
        keys = k_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)
        vals = v_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)
        keys = keys[:, :ctx].float()
        vals = vals[:, :ctx].float()
        qh = q[s].float()  # [Hq, D]
        for h in range(num_heads):
            kv_h = h // group
            scores = keys[kv_h] @ qh[h] * scale  # [ctx]
            p = torch.softmax(scores, dim=-1)
            out[s, h] = p @ vals[kv_h]
    return out.to(q.dtype)


def prefill_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    scale: float,
    causal: bool = True,
) -> torch.Tensor:
    """Varlen causal attention, dense (non-paged) K/V.

    q: [T, Hq, D]; k/v: [T, Hk, D]; cu_seqlens: [num_seqs+1] int32.
    """
    T, num_heads, head_dim = q.shape
    num_kv_heads = k.shape[1]
    group = num_heads // num_kv_heads
    out = torch.empty_like(q, dtype=torch.float32)
    for i in range(cu_seqlens.numel() - 1):
        s, e = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
        L = e - s
        qf = q[s:e].float()  # [L, Hq, D]
        kf = k[s:e].float()
        vf = v[s:e].float()
        for h in range(num_heads):
            kv_h = h // group
            scores = qf[:, h] @ kf[:, kv_h].T * scale  # [L, L]
            if causal:
                mask = torch.triu(
                    torch.ones(L, L, dtype=torch.bool, device=q.device), diagonal=1
                )
                scores = scores.masked_fill(mask, float("-inf"))
            p = torch.softmax(scores, dim=-1)
            out[s:e, h] = p @ vf[:, kv_h]
    return out.to(q.dtype)


def prefill_attention_paged(
    q: torch.Tensor,            # [Tnew, Hq, D] (new tokens only)
    k_cache: torch.Tensor,      # [B, Hk, bs, D]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [S, max_blocks]
    cu_seqlens_q: torch.Tensor,  # [S+1] over NEW tokens
    seq_lens_k: torch.Tensor,    # [S] total context length
    scale: float,
) -> torch.Tensor:
    """Context attention: new tokens attend over the paged cache (which
    already contains their own K/V plus any cached prefix)."""
    Tn, num_heads, head_dim = q.shape
    num_kv_heads = k_cache.shape[1]
    bs = k_cache.shape[2]
    group = num_heads // num_kv_heads
    out = torch.empty_like(q, dtype=torch.float32)
    for s in range(cu_seqlens_q.numel() - 1):
        qs, qe = int(cu_seqlens_q[s]), int(cu_seqlens_q[s + 1])
        q_len = qe - qs
        k_len = int(seq_lens_k[s])
        ctx_start = k_len - q_len
        nblk = (k_len + bs - 1) // bs
        blocks = block_tables[s, :nblk].long()
        keys = k_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)
        vals = v_cache[blocks].permute(1, 0, 2, 3).reshape(num_kv_heads, -1, head_dim)
        keys = keys[:, :k_len].float()
        vals = vals[:, :k_len].float()
        qf = q[qs:qe].float()  # [q_len, Hq, D]
        # causal mask with context offset
        kv_pos = torch.arange(k_len, device=q.device)
        q_pos = ctx_start + torch.arange(q_len, device=q.device)
        mask = kv_pos.unsqueeze(0) > q_pos.unsqueeze(1)  # [q_len, k_len]
        for h in range(num_heads):
            kv_h = h // group
            scores = qf[:, h] @ keys[kv_h].T * scale
            scores = scores.masked_fill(mask, float("-inf"))
            p = torch.softmax(scores, dim=-1)
            out[qs:qe, h] = p @ vals[kv_h]
    return out.to(q.dtype)


def gather_kv_blocks(
    k_cache: torch.Tensor, v_cache: torch.Tensor, block_ids: torch.Tensor
) -> torch.Tensor:
    """Pack the given cache blocks into one contiguous staging buffer.

    Returns [2, nblocks, Hk, bs, D] (K plane then V plane) — a single large
    contiguous tensor so the PD handoff is one RCCL send per batch of blocks.
    """
    idx = block_ids.long()
    return torch.stack([k_cache[idx], v_cache[idx]], dim=0).contiguous()


def scatter_kv_blocks(
    staging: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_ids: torch.Tensor,
):
    idx = block_ids.long()
    k_cache[idx] = staging[0].to(k_cache.dtype)
    v_cache[idx] = staging[1].to(v_cache.dtype)


# ---------------------------------------------------------------- MoE

def unpack_moe_weights(b_packed: torch.Tensor) -> torch.Tensor:
    """Inverse of ops.pack_moe_weights: [E, K/32, N/16, 64, 8] -> [E, K, N]."""
    E, K32, N16, _, _ = b_packed.shape
    return (
        b_packed.view(E, K32, N16, 4, 16, 8)
        .permute(0, 1, 3, 5, 2, 4)
        .reshape(E, K32 * 32, N16 * 16)
        .contiguous()
    )


def moe_gemm(out, a, b_packed, sorted_ids, expert_ids, n_valid,
             block_m: int, gate_up: bool):
    """fp32 reference of the grouped GEMM (CPU; same semantics as the HIP
    kernel incl. padding rows computed from the dummy token row)."""
    w = unpack_moe_weights(b_packed).float()
    n_tiles = int(n_valid.item())
    N = out.shape[1]
    for mt in range(n_tiles):
        e = int(expert_ids[mt])
        r0 = mt * block_m
        rows = slice(r0, r0 + block_m)
        if gate_up:
            src = a[sorted_ids[rows].long()].float()
        else:
            src = a[rows].float()
        acc = src @ w[e]
        if gate_up:
            g, u = acc[:, :N], acc[:, N:]
            acc = torch.nn.functional.silu(g) * u
        out[rows] = acc.to(out.dtype)
    return out


def moe_combine(out, y, pos, w):
    T, H = out.shape
    topk = pos.numel() // T
    acc = torch.zeros(T, H, dtype=torch.float32)
    yv = y.float()
    for t in range(T):
        for k in range(topk):
            p = int(pos[t * topk + k])
            if p < 0:
                continue
            acc[t] += float(w[t * topk + k]) * yv[p]
    out.copy_(acc.to(out.dtype))
    return out


def moe_gemm_fp8(out, a, a_scales, b_packed, b_scales, sorted_ids,
                 expert_ids, n_valid, block_m: int, gate_up: bool):
    """fp32 reference of the grouped fp8 GEMM (dequant then matmul)."""
    E = b_packed.shape[0]
    w = unpack_moe_weights(b_packed.view(torch.int8)).view(
        torch.float8_e4m3fn).float()          # [E, K, NB]
    NB = w.shape[2]
    ws = b_scales.view(E, NB).float()
    n_tiles = int(n_valid.item())
    N = out.shape[1]
    af = a.float() * a_scales.unsqueeze(1).float()
    for mt in range(n_tiles):
        e = int(expert_ids[mt])
        r0 = mt * block_m
        rows = slice(r0, r0 + block_m)
        if gate_up:
            src = af[sorted_ids[rows].long()]
        else:
            src = af[rows]
        acc = src @ (w[e] * ws[e].unsqueeze(0))
        if gate_up:
            g, u = acc[:, :N], acc[:, N:]
            acc = torch.nn.functional.silu(g) * u
        out[rows] = acc.to(out.dtype)
    return out
