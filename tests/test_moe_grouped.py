"""MoE grouped-GEMM path: device-side block alignment, fragment packing,
and the grouped forward vs the capacity-padded torch path (CPU; the HIP
kernels themselves are covered in tests/test_ops_gpu.py).

Capability parity: expert serving the reference invokes via vLLM images
(SURVEY.md §2.3); the alignment mirrors vLLM's moe_align_block_size but
runs entirely on-device with static shapes (hipGraph-capturable).
"""

import types

import pytest
import torch

import fusioninfer_amd.ops as ops
from fusioninfer_amd.models.model import MoEMLP
from fusioninfer_amd.models.registry import get_model_config
from fusioninfer_amd.ops import reference as ref


def test_pack_unpack_roundtrip():
    w = torch.randn(3, 64, 48, dtype=torch.bfloat16)
    packed = ops.pack_moe_weights(w)
    assert packed.shape == (3, 2, 3, 64, 8)
    assert torch.equal(ref.unpack_moe_weights(packed), w)


def _align(e_start, e_end, topi, block_m):
    fake = types.SimpleNamespace(e_start=e_start, e_end=e_end)
    return MoEMLP._moe_align(fake, topi, block_m)


def test_moe_align_block_structure():
    torch.manual_seed(7)
    T, k, E = 13, 2, 4
    topi = torch.randint(0, E, (T, k))
    block_m = 16
    sorted_ids, expert_ids, n_valid, pos, PM = _align(0, E, topi, block_m)
    assert PM % block_m == 0 and sorted_ids.shape[0] == PM
    n_tiles = int(n_valid.item())
    flat = topi.reshape(-1)
    counts = torch.bincount(flat, minlength=E)
    assert n_tiles == sum(-(-int(c) // block_m) for c in counts if c > 0)
    # every assignment lands in a slot of its own expert's tile range,
    # with the tile's expert_id matching
    for t in range(T):
        for j in range(k):
            p = int(pos[t * k + j])
            assert p >= 0
            assert int(sorted_ids[p]) == t
            assert int(expert_ids[p // block_m]) == int(topi[t, j])
    # slots are unique over real assignments
    ps = pos.tolist()
    assert len(set(ps)) == len(ps)


def test_moe_align_non_local_experts():
    """EP: assignments to experts outside [e_start, e_end) get pos=-1 and
    produce no tiles."""
    topi = torch.tensor([[0, 5], [5, 6], [2, 3]])
    sorted_ids, expert_ids, n_valid, pos, PM = _align(2, 6, topi, 16)
    # local experts are 2..5 -> local ids 0..3
    assert int(n_valid.item()) == 3  # experts 5 (x2), 2, 3 -> 3 nonempty
    want_local = [False, True, True, False, True, True]
    assert [(int(p) >= 0) for p in pos] == want_local


def test_grouped_forward_matches_bmm_path():
    """The grouped path (reference semantics on CPU: same align + pack +
    combine as the HIP kernels) matches the capacity-padded bmm path."""
    torch.manual_seed(0)
    cfg = get_model_config("tiny-qwen3-moe")
    mlp = MoEMLP(cfg, layer_idx=0)
    for T in (1, 5, 37, 190):
        x = torch.randn(T, cfg.hidden_size, dtype=torch.bfloat16)
        want = mlp.forward(x)
        got = mlp._forward_grouped(x)
        torch.testing.assert_close(
            got.float(), want.float(), atol=3e-2, rtol=3e-2
        )


def test_grouped_forward_block_m_128():
    """Enough tokens to trigger the prefill (block_m=128) variant."""
    torch.manual_seed(1)
    cfg = get_model_config("tiny-qwen3-moe")
    mlp = MoEMLP(cfg, layer_idx=1)
    T = 64 * cfg.num_experts // cfg.num_experts_per_tok  # 256
    x = torch.randn(T, cfg.hidden_size, dtype=torch.bfloat16)
    assert T * mlp.top_k >= 64 * mlp.num_experts  # picks block_m=128
    want = mlp.forward(x)
    got = mlp._forward_grouped(x)
    torch.testing.assert_close(got.float(), want.float(), atol=3e-2, rtol=3e-2)


def test_packed_cache_invalidates_on_weight_mutation():
    cfg = get_model_config("tiny-qwen3-moe")
    mlp = MoEMLP(cfg, layer_idx=0)
    p1, _ = mlp._packed_weights()
    p1_again, _ = mlp._packed_weights()
    assert p1 is p1_again  # cached
    # .data mutations are invisible to tensor._version — the weight
    # loader must invalidate explicitly (weight_loader.py does)
    mlp.gate_up_t.data[0].add_(1.0)
    mlp.invalidate_packed()
    p2, _ = mlp._packed_weights()
    assert p2 is not p1
    assert torch.equal(p2, ops.pack_moe_weights(mlp.gate_up_t.data))
    # tracked in-place mutation invalidates via _version
    with torch.no_grad():
        mlp.gate_up_t[0].mul_(2.0)
    p3, _ = mlp._packed_weights()
    assert p3 is not p2


def test_release_unpacked_cpu_noop():
    """CPU never releases (the unpacked form IS the CPU compute path)."""
    cfg = get_model_config("tiny-qwen3-moe")
    mlp = MoEMLP(cfg, layer_idx=0)
    n = mlp.gate_up_t.numel()
    mlp.release_unpacked()
    assert mlp.gate_up_t.numel() == n
    x = torch.randn(4, cfg.hidden_size, dtype=torch.bfloat16)
    mlp.forward(x)


def test_ensure_unpacked_roundtrip():
    """ensure_unpacked re-materializes EXACT contents from the packed
    cache (release itself is CUDA-gated; emulate the released state)."""
    cfg = get_model_config("tiny-qwen3-moe")
    mlp = MoEMLP(cfg, layer_idx=2)
    gu0 = mlp.gate_up_t.data.clone()
    dn0 = mlp.down_t.data.clone()
    mlp._packed_weights()
    mlp.gate_up_t.data = gu0.new_empty((0,) + tuple(gu0.shape[1:]))
    mlp.down_t.data = dn0.new_empty((0,) + tuple(dn0.shape[1:]))
    # mutating released weights without re-materializing is a bug
    with pytest.raises(RuntimeError, match="released"):
        mlp.invalidate_packed()
    mlp.ensure_unpacked()
    assert torch.equal(mlp.gate_up_t.data, gu0)
    assert torch.equal(mlp.down_t.data, dn0)
    mlp.ensure_unpacked()  # idempotent
    assert torch.equal(mlp.gate_up_t.data, gu0)


def test_ensure_unpacked_roundtrip_fp8():
    """fp8 variant: the packed form stores transposed int8-viewed bytes —
    the round-trip must restore the [E, O, I] e4m3 layout bit-exactly."""
    cfg = get_model_config("tiny-qwen3-moe")
    cfg.quantization = "fp8"
    try:
        mlp = MoEMLP(cfg, layer_idx=0)
        gu0 = mlp.gate_up_fp8.data.clone()
        dn0 = mlp.down_fp8.data.clone()
        mlp._packed_weights_fp8()
        mlp.gate_up_fp8.data = gu0.new_empty((0,) + tuple(gu0.shape[1:]))
        mlp.down_fp8.data = dn0.new_empty((0,) + tuple(dn0.shape[1:]))
        mlp.ensure_unpacked()
        assert torch.equal(mlp.gate_up_fp8.data.view(torch.int8),
                           gu0.view(torch.int8))
        assert torch.equal(mlp.down_fp8.data.view(torch.int8),
                           dn0.view(torch.int8))
    finally:
        cfg.quantization = None


@pytest.mark.gpu
def test_release_unpacked_gpu_forward_parity():
    """On GPU, releasing the unpacked copy must not change the grouped
    forward (it reads only the packed cache + shapes)."""
    torch.manual_seed(2)
    cfg = get_model_config("tiny-qwen3-moe")
    with torch.device("cuda"):
        mlp = MoEMLP(cfg, layer_idx=0)
    x = torch.randn(17, cfg.hidden_size, dtype=torch.bfloat16,
                    device="cuda")
    want = mlp.forward(x)
    gu0 = mlp.gate_up_t.data.clone()
    mlp.release_unpacked()
    assert mlp.gate_up_t.numel() == 0
    torch.testing.assert_close(mlp.forward(x), want)
    mlp.ensure_unpacked()
    assert torch.equal(mlp.gate_up_t.data, gu0)
    torch.testing.assert_close(mlp.forward(x), want)


def test_fp8_grouped_matches_per_expert_loop():
    """fp8 grouped path (reference semantics on CPU) vs the per-expert
    dequant loop, same quantized weights."""
    torch.manual_seed(4)
    cfg = get_model_config("tiny-qwen3-moe")
    cfg.quantization = "fp8"
    try:
        mlp = MoEMLP(cfg, layer_idx=0)
        for T in (3, 40):
            x = torch.randn(T, cfg.hidden_size, dtype=torch.bfloat16) * 0.3
            from fusioninfer_amd.ops import quant_fp8_rows

            x8, xs = quant_fp8_rows(x)
            want = mlp._forward_fp8((x8, xs)).float()
            got = mlp._forward_fp8_grouped(x8, xs).float()
            rel = (got - want).norm() / want.norm().clamp_min(1e-6)
            assert rel.item() < 0.08, (T, rel.item())
    finally:
        cfg.quantization = None
