"""Op dispatch: CDNA4 HIP kernels on GPU, PyTorch fp32 reference on CPU.

Policy (required by the build contract): on a GPU box the HIP extension is
the ONLY execution path — if ``_C.so`` is missing or fails to import, any
GPU-tensor call raises immediately instead of silently falling back to
eager PyTorch. CPU tensors (CI containers without a GPU) use the reference
implementations from ``fusioninfer_amd.ops.reference``.
"""

from __future__ import annotations

import math
from typing import Optional

import torch

from fusioninfer_amd.ops import reference as ref

_C = None
_C_IMPORT_ERROR: Optional[BaseException] = None
try:
    import importlib

    _C = importlib.import_module("fusioninfer_amd.ops._C")
except Exception as e:  # pragma: no cover - exercised only when build broken
    _C_IMPORT_ERROR = e


def has_native() -> bool:
    return _C is not None


def _require_native():
    if _C is None:
        raise RuntimeError(
            "fusioninfer_amd HIP extension (_C.so) is not available but a GPU "
            "tensor was passed — refusing to fall back to eager PyTorch on "
            "the GPU path. Build it with `python -m fusioninfer_amd.ops.build`."
        ) from _C_IMPORT_ERROR


# ---------------------------------------------------------------- norm ops

def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    if x.is_cuda:
        _require_native()
        out = torch.empty_like(x)
        _C.rms_norm(out, x, weight, eps)
        return out
    return ref.rms_norm(x, weight, eps)


def fused_add_rms_norm(x, residual, weight, eps: float):
    """In-place on GPU: residual += x; x = rmsnorm(residual). Returns (x, residual)."""
    if x.is_cuda:
        _require_native()
        _C.fused_add_rms_norm(x, residual, weight, eps)
        return x, residual
    return ref.fused_add_rms_norm(x, residual, weight, eps)


def silu_and_mul(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        _require_native()
        out = torch.empty(
            (*x.shape[:-1], x.shape[-1] // 2), dtype=x.dtype, device=x.device
        )
        _C.silu_and_mul(out, x)
        return out
    return ref.silu_and_mul(x)


# ------------------------------------------------------- fp8 fused epilogues
#
# The fp8 serving mode quantizes activations per-token for torch._scaled_mm;
# done eagerly that costs more than the fp8 GEMM saves, so quantization is
# fused into the producing kernel (norm / SwiGLU) or done in one pass
# (quant_fp8_rows for the attention output).

def rms_norm_fp8(x: torch.Tensor, weight: torch.Tensor, eps: float):
    """Returns (x_fp8 [T, H] e4m3, scales [T] fp32)."""
    if x.is_cuda:
        _require_native()
        out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
        scales = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _C.rms_norm_fp8(out, scales, x, weight, eps)
        return out, scales
    return ref.rms_norm_fp8(x, weight, eps)


def fused_add_rms_norm_fp8(x, residual, weight, eps: float):
    """residual += x in place; returns (x_fp8, scales, residual)."""
    if x.is_cuda:
        _require_native()
        out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
        scales = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _C.fused_add_rms_norm_fp8(out, scales, x, residual, weight, eps)
        return out, scales, residual
    return ref.fused_add_rms_norm_fp8(x, residual, weight, eps)


def silu_and_mul_fp8(x: torch.Tensor):
    """x: [T, 2*I] -> (act_fp8 [T, I] e4m3, scales [T] fp32)."""
    if x.is_cuda:
        _require_native()
        inter = x.shape[-1] // 2
        out = torch.empty(
            (*x.shape[:-1], inter), dtype=torch.float8_e4m3fn, device=x.device
        )
        scales = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _C.silu_and_mul_fp8(out, scales, x)
        return out, scales
    return ref.silu_and_mul_fp8(x)


def quant_fp8_rows(x: torch.Tensor):
    """Per-row dynamic fp8 quant: (x_fp8, scales [T] fp32)."""
    if x.is_cuda:
        _require_native()
        out = torch.empty(x.shape, dtype=torch.float8_e4m3fn, device=x.device)
        scales = torch.empty(x.shape[0], dtype=torch.float32, device=x.device)
        _C.quant_fp8_rows(out, scales, x)
        return out, scales
    return ref.quant_fp8_rows(x)


# ---------------------------------------------------------------- rope

def rope_qk_norm_(
    q: torch.Tensor,          # [T, Hq*D] (row-strided slice OK)
    k: torch.Tensor,          # [T, Hk*D]
    positions: torch.Tensor,  # [T] int32
    cos_sin: torch.Tensor,    # [P, D] fp32
    num_q_heads: int,
    num_kv_heads: int,
    head_dim: int,
    q_weight: Optional[torch.Tensor] = None,
    k_weight: Optional[torch.Tensor] = None,
    eps: float = 1e-6,
):
    """Fused optional per-head RMSNorm + NeoX RoPE, in-place. Returns (q, k)."""
    if q.is_cuda:
        _require_native()
        _C.rope_qk_norm(
            q, k, q_weight, k_weight, cos_sin, positions,
            num_q_heads, num_kv_heads, head_dim, eps,
        )
        return q, k
    qv = q.view(-1, num_q_heads, head_dim)
    kv = k.view(-1, num_kv_heads, head_dim)
    qo, ko = ref.rope_qk_norm(qv, kv, positions.long(), cos_sin, q_weight, k_weight, eps)
    q.copy_(qo.reshape(q.shape))
    k.copy_(ko.reshape(k.shape))
    return q, k


# ---------------------------------------------------------------- kv cache

def reshape_and_cache(k, v, k_cache, v_cache, slot_mapping,
                      k_inv_scale: float = 1.0, v_inv_scale: float = 1.0):
    """Scatter K/V rows into the paged cache. For fp8 caches the rows are
    multiplied by the INVERSE per-layer static scales before e4m3
    conversion (vLLM k_scale/v_scale); bf16 caches ignore the scales
    (the read side compensates only on the fp8 path)."""
    if k.is_cuda:
        _require_native()
        _C.reshape_and_cache(k, v, k_cache, v_cache, slot_mapping,
                             k_inv_scale, v_inv_scale)
        return
    H, D = k_cache.shape[1], k_cache.shape[3]
    ref.reshape_and_cache(
        k.view(-1, H, D), v.view(-1, H, D), k_cache, v_cache,
        slot_mapping.long(), k_inv_scale, v_inv_scale,
    )


def gather_kv_blocks(k_cache, v_cache, block_ids) -> torch.Tensor:
    if k_cache.is_cuda:
        _require_native()
        n = int(block_ids.numel())
        staging = torch.empty(
            (2, n, *k_cache.shape[1:]), dtype=k_cache.dtype, device=k_cache.device
        )
        _C.gather_kv_blocks(staging, k_cache, v_cache, block_ids)
        return staging
    return ref.gather_kv_blocks(k_cache, v_cache, block_ids.long())


def scatter_kv_blocks(staging, k_cache, v_cache, block_ids):
    if k_cache.is_cuda:
        _require_native()
        _C.scatter_kv_blocks(staging, k_cache, v_cache, block_ids)
        return
    ref.scatter_kv_blocks(staging, k_cache, v_cache, block_ids.long())


# ---------------------------------------------------------------- attention

DECODE_PARTITION_TOKENS = 512  # flash-decoding partition size (kernel contract)


def paged_attention_decode(
    q: torch.Tensor,            # [S, Hq, D]
    k_cache: torch.Tensor,      # [B, Hk, bs, D]
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [S, max_blocks] int32
    seq_lens: torch.Tensor,      # [S] int32
    scale: Optional[float] = None,
) -> torch.Tensor:
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        _require_native()
        S, Hq, D = q.shape
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        # partitions from the block-table width: static shape under
        # hipGraph capture (seq_lens vary between replays, width doesn't)
        max_tokens = int(block_tables.shape[1]) * int(k_cache.shape[2])
        num_parts = max(
            (max_tokens + DECODE_PARTITION_TOKENS - 1) // DECODE_PARTITION_TOKENS,
            1,
        )
        ml_ws = acc_ws = None
        if num_parts > 1:
            ml_ws = torch.empty(
                (S, Hq, num_parts, 2), dtype=torch.float32, device=q.device
            )
            acc_ws = torch.empty(
                (S, Hq, num_parts, D), dtype=torch.float32, device=q.device
            )
        _C.paged_attention_decode(
            out, q, k_cache, v_cache, block_tables, seq_lens,
            ml_ws, acc_ws, num_parts, scale,
        )
        return out
    return ref.paged_attention_decode(q, k_cache, v_cache, block_tables, seq_lens, scale)


# q rows per prefill workgroup (NW waves x 32) — kernel contract; the
# FI_PF_NW env selects the 4-wave variant on both sides (see
# prefill_attention.hip launcher)
import os as _os

PREFILL_TILE_ROWS = (4 if _os.environ.get("FI_PF_NW", "8").startswith("4")
                     else 8) * 32


def prefill_tile_rows(seq_lens) -> int:
    """Per-batch workgroup width: 128-row (4-wave) workgroups once any
    sequence is >=4096 new tokens — the narrower shape halves the
    causal-raggedness idle and measured +10% at 1x8192 (it loses ~5% at
    many short sequences, so the default stays 256)."""
    if seq_lens and max(seq_lens) >= 4096:
        return 128
    return PREFILL_TILE_ROWS


def build_prefill_tiles(seq_lens, device=None, rows=None):
    """Host-side tile table for the prefill kernel: one workgroup per
    `rows` q-rows of each sequence (rows must be 128 or 256 and MATCH
    the tile_rows passed to the kernel).

    seq_lens: list[int]. Returns (tile_seq, tile_row0) int32 tensors.
    """
    rows = rows or PREFILL_TILE_ROWS
    tile_seq, tile_row0 = [], []
    for s, L in enumerate(seq_lens):
        for r0 in range(0, L, rows):
            tile_seq.append(s)
            tile_row0.append(r0)
    return (
        torch.tensor(tile_seq, dtype=torch.int32, device=device),
        torch.tensor(tile_row0, dtype=torch.int32, device=device),
    )


def prefill_attention(
    q: torch.Tensor,           # [T, Hq, D]
    k: torch.Tensor,           # [T, Hk, D]
    v: torch.Tensor,           # [T, Hk, D]
    cu_seqlens: torch.Tensor,  # [S+1] int32
    scale: Optional[float] = None,
    tile_seq: Optional[torch.Tensor] = None,
    tile_row0: Optional[torch.Tensor] = None,
    tile_rows: Optional[int] = None,
) -> torch.Tensor:
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        _require_native()
        if tile_seq is None:
            lens = (cu_seqlens[1:] - cu_seqlens[:-1]).tolist()
            tile_rows = tile_rows or prefill_tile_rows(lens)
            tile_seq, tile_row0 = build_prefill_tiles(
                lens, device=q.device, rows=tile_rows
            )
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        _C.prefill_attention(out, q, k, v, tile_seq, tile_row0, cu_seqlens,
                             scale, tile_rows or PREFILL_TILE_ROWS)
        return out
    return ref.prefill_attention(q, k, v, cu_seqlens, scale, causal=True)


def prefill_attention_paged(
    q: torch.Tensor,             # [Tnew, Hq, D] new tokens only
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_tables: torch.Tensor,  # [S, max_blocks] int32
    cu_seqlens_q: torch.Tensor,  # [S+1] int32 over new tokens
    seq_lens_k: torch.Tensor,    # [S] int32 total context length
    scale: Optional[float] = None,
    tile_seq: Optional[torch.Tensor] = None,
    tile_row0: Optional[torch.Tensor] = None,
    tile_rows: Optional[int] = None,
) -> torch.Tensor:
    """Context attention: new tokens attend over the paged cache (their own
    K/V must already be written via reshape_and_cache)."""
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    if q.is_cuda:
        _require_native()
        if tile_seq is None:
            lens = (cu_seqlens_q[1:] - cu_seqlens_q[:-1]).tolist()
            tile_rows = tile_rows or prefill_tile_rows(lens)
            tile_seq, tile_row0 = build_prefill_tiles(
                lens, device=q.device, rows=tile_rows
            )
        out = torch.empty(q.shape, dtype=q.dtype, device=q.device)
        _C.prefill_attention_paged(
            out, q, k_cache, v_cache, tile_seq, tile_row0, cu_seqlens_q,
            block_tables, seq_lens_k, scale,
            tile_rows or PREFILL_TILE_ROWS,
        )
        return out
    return ref.prefill_attention_paged(
        q, k_cache, v_cache, block_tables, cu_seqlens_q, seq_lens_k, scale
    )


# ---------------------------------------------------------------- MoE

def pack_moe_weights(w: torch.Tensor) -> torch.Tensor:
    """Pack per-expert weights [E, K, N] into MFMA B-fragment order
    [E, K/32, N/16, 64, 8] for the grouped GEMM: lane l of a wave reads
    its 8 contiguous bf16 B elements (B[k=(l>>4)*8+e][n=l&15] within a
    32x16 k-by-n subtile) with one 16 B load — one fully-coalesced 1 KiB
    transaction per wave per K-step."""
    E, K, N = w.shape
    assert K % 32 == 0 and N % 16 == 0, (K, N)
    return (
        w.view(E, K // 32, 4, 8, N // 16, 16)
        .permute(0, 1, 4, 2, 5, 3)
        .reshape(E, K // 32, N // 16, 64, 8)
        .contiguous()
    )


def moe_gemm(out, a, b_packed, sorted_ids, expert_ids, n_valid,
             block_m: int, gate_up: bool):
    """Grouped GEMM over block-aligned expert segments.

    out [PM, N] bf16; a = x [T, K] (gate_up, rows gathered via sorted_ids)
    or act [PM, K]; b_packed from pack_moe_weights ([E, K/32, NB/16, 64, 8]
    with NB = 2N when gate_up). n_valid: device int32 scalar tensor with
    the real m-tile count (static grid, dynamic work — hipGraph-safe).
    gate_up=True fuses the SwiGLU epilogue: out = silu(gate) * up."""
    if out.is_cuda:
        _require_native()
        _C.moe_gemm(out, a, b_packed, sorted_ids, expert_ids, n_valid,
                    block_m, gate_up)
        return out
    return ref.moe_gemm(out, a, b_packed, sorted_ids, expert_ids, n_valid,
                        block_m, gate_up)


def moe_gemm_fp8(out, a, a_scales, b_packed, b_scales, sorted_ids,
                 expert_ids, n_valid, block_m: int, gate_up: bool):
    """Grouped fp8 (e4m3) GEMM with dequant epilogue: out[bf16] =
    (A_fp8 @ B_fp8) * a_scale[row] * b_scale[expert, col]; gate_up fuses
    SwiGLU. Same block-aligned layout as moe_gemm; b_packed from
    pack_moe_weights over the [E, K, N] e4m3 weights."""
    if out.is_cuda:
        _require_native()
        _C.moe_gemm_fp8(out, a, a_scales, b_packed, b_scales, sorted_ids,
                        expert_ids, n_valid, block_m, gate_up)
        return out
    return ref.moe_gemm_fp8(out, a, a_scales, b_packed, b_scales,
                            sorted_ids, expert_ids, n_valid, block_m,
                            gate_up)


def moe_align(topi: torch.Tensor, e_start: int, e_end: int, block_m: int):
    """Single-kernel device-side block alignment of expert assignments
    (GPU only; MoEMLP._moe_align's torch composition is the CPU
    reference and the semantic oracle). Within-expert slot order is
    arrival order rather than stable order — downstream results are
    bit-identical either way (each row's K-loop and the pos-gathered
    combine don't depend on the slot). Returns
    (sorted_ids [PM], expert_ids [PM/bm], n_valid [1], pos [T*k], PM)."""
    _require_native()
    T, k = topi.shape
    n = T * k
    E_local = e_end - e_start
    PM = ((n + block_m - 1) // block_m + E_local) * block_m
    dev = topi.device
    sorted_ids = torch.empty(PM, dtype=torch.int32, device=dev)
    expert_ids = torch.empty(PM // block_m, dtype=torch.int32, device=dev)
    n_valid = torch.empty(1, dtype=torch.int32, device=dev)
    pos = torch.empty(n, dtype=torch.int32, device=dev)
    _C.moe_align(topi.reshape(-1).int().contiguous(), sorted_ids,
                 expert_ids, n_valid, pos, k, e_start, e_end, block_m)
    return sorted_ids, expert_ids, n_valid, pos, PM


def moe_router_topk(logits: torch.Tensor, k: int, renorm: bool):
    """Fused router tail (GPU): softmax + top-k + optional top-k
    renormalization in one wave-per-token kernel. logits [T, E] fp32;
    returns (topv [T,k] fp32 weights, topi [T,k] int32 expert ids,
    descending; ties pick the smaller index)."""
    _require_native()
    T, E = logits.shape
    topv = torch.empty(T, k, dtype=torch.float32, device=logits.device)
    topi = torch.empty(T, k, dtype=torch.int32, device=logits.device)
    _C.moe_router_topk(logits.contiguous(), topv, topi, k, renorm)
    return topv, topi


def moe_combine(out, y, pos, w):
    """out[t] = sum_k w[t,k] * y[pos[t,k]] (pos < 0 skipped). Deterministic
    (no atomics) so token-exact tests stay reproducible."""
    if out.is_cuda:
        _require_native()
        _C.moe_combine(out, y, pos, w)
        return out
    return ref.moe_combine(out, y, pos, w)


compute_cos_sin_cache = ref.compute_cos_sin_cache
