"""FP8 (OCP e4m3) quantized linear path.

gfx950 runs OCP fp8 MFMA at ~2x the bf16 rate and halves weight-read
bytes — decode-time GEMMs are weight-bandwidth-bound, so fp8 weights pay
there directly. Enabled per-engine with `--quantization fp8`: projection
weights are stored fp8 with per-output-channel scales; activations are
dynamically quantized per-token; the GEMM is hipBLASLt fp8 via
torch._scaled_mm with bf16 output. The flagship benchmark stays bf16
(BASELINE dtype contract); fp8 is an opt-in serving mode.

Note: gfx950 fp8 is OCP e4m3fn — NOT the MI300X fnuz variant
(guide cdna_hip_programming.md §4).
"""

from __future__ import annotations

import torch
import torch.nn as nn

from fusioninfer_amd.distributed import parallel_state as ps

FP8_DTYPE = torch.float8_e4m3fn
FP8_MAX = 448.0  # e4m3fn max normal


def quantize_weight_fp8(w: torch.Tensor):
    """Per-output-channel symmetric quantization: returns (w_fp8 [O, I],
    scale [O] fp32) with w ~= w_fp8 * scale[:, None]."""
    absmax = w.abs().amax(dim=1, keepdim=True).float().clamp(min=1e-8)
    scale = absmax / FP8_MAX
    w_fp8 = (w.float() / scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    return w_fp8, scale.squeeze(1)


def quantize_activation_fp8(x: torch.Tensor):
    """Per-token dynamic quantization: (x_fp8 [T, I], scale [T] fp32)."""
    absmax = x.abs().amax(dim=1, keepdim=True).float().clamp(min=1e-8)
    scale = absmax / FP8_MAX
    x_fp8 = (x.float() / scale).clamp(-FP8_MAX, FP8_MAX).to(FP8_DTYPE)
    return x_fp8, scale.squeeze(1)


def fp8_linear(x, w_fp8: torch.Tensor, w_scale: torch.Tensor):
    """y[T, O] = x @ W^T with fp8 operands, bf16 output.

    `x` is either a bf16 tensor (quantized here, one fused HIP kernel on
    GPU) or an (x_fp8, x_scale) tuple already produced by a fused epilogue
    (rms_norm_fp8 / silu_and_mul_fp8 / quant_fp8_rows)."""
    if isinstance(x, tuple):
        x_fp8, x_scale = x
    elif x.is_cuda:
        from fusioninfer_amd import ops

        x_fp8, x_scale = ops.quant_fp8_rows(x)
    else:
        x_fp8, x_scale = None, None  # CPU: keep the bf16 tensor
    if x_fp8 is not None and x_fp8.is_cuda:
        return torch._scaled_mm(
            x_fp8,
            w_fp8.t(),  # [I, O], column-major view of the row-major weight
            scale_a=x_scale.unsqueeze(1),
            scale_b=w_scale.unsqueeze(0),
            out_dtype=torch.bfloat16,
        )
    # CPU reference path (tests): dequantize and matmul in fp32
    if x_fp8 is not None:  # tuple input on CPU
        x = x_fp8.float() * x_scale.unsqueeze(1).float()
    w = w_fp8.float() * w_scale.unsqueeze(1).float()
    return (x.float() @ w.t()).to(torch.bfloat16)


def convert_linear_to_fp8(module: nn.Module) -> int:
    """Convert every projection Linear in a CausalLM to fp8 storage.
    Returns the number of converted layers. lm_head/embedding stay bf16
    (vocab-scale accuracy)."""
    from fusioninfer_amd.distributed.layers import (
        MergedColumnParallelLinear,
        RowParallelLinear,
    )

    converted = 0
    for m in module.modules():
        if isinstance(m, (MergedColumnParallelLinear, RowParallelLinear)):
            w_fp8, scale = quantize_weight_fp8(m.weight.data)
            m.weight = nn.Parameter(w_fp8, requires_grad=False)
            m.register_buffer("weight_scale", scale, persistent=False)
            m.forward = _make_fp8_forward(m)
            converted += 1
    return converted


def _make_fp8_forward(m):
    is_row_parallel = hasattr(m, "in_per_rank")
    bias = getattr(m, "bias", None)

    def fwd(x):
        out = fp8_linear(x, m.weight, m.weight_scale)
        if bias is not None:
            out = out + bias
        if is_row_parallel:
            out = ps.tp_all_reduce(out)
        return out

    return fwd
