"""GPU numerics tests: every HIP kernel vs the PyTorch fp32 reference.

Mirrors the reference's test strategy tier 1 (table-driven unit tests,
SURVEY.md §4) applied to the data plane: each op is compared against
ops/reference.py on random inputs.
"""

import math

import pytest
import torch

import fusioninfer_amd.ops as ops
from fusioninfer_amd.ops import reference as ref

pytestmark = pytest.mark.gpu

DEV = "cuda"


def setup_module():
    assert ops.has_native(), "HIP extension must be present on a GPU box"
    torch.manual_seed(0)


def assert_close_bf16(actual, expected_f32, atol=2e-2, rtol=2e-2):
    torch.testing.assert_close(
        actual.float(), expected_f32.float(), atol=atol, rtol=rtol
    )


@pytest.mark.parametrize("tokens,hidden", [(1, 1024), (17, 4096), (256, 4096), (33, 8192)])
def test_rms_norm(tokens, hidden):
    x = torch.randn(tokens, hidden, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(hidden, dtype=torch.bfloat16, device=DEV)
    out = ops.rms_norm(x, w, 1e-6)
    expected = ref.rms_norm(x.cpu(), w.cpu(), 1e-6)
    assert_close_bf16(out.cpu(), expected)


@pytest.mark.parametrize("tokens,hidden", [(16, 4096), (130, 4096)])
def test_fused_add_rms_norm(tokens, hidden):
    x = torch.randn(tokens, hidden, dtype=torch.bfloat16, device=DEV)
    res = torch.randn(tokens, hidden, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(hidden, dtype=torch.bfloat16, device=DEV)
    x_ref, res_ref = ref.fused_add_rms_norm(x.cpu(), res.cpu(), w.cpu(), 1e-6)
    out, new_res = ops.fused_add_rms_norm(x, res, w, 1e-6)
    assert_close_bf16(new_res.cpu(), res_ref)
    assert_close_bf16(out.cpu(), x_ref)


@pytest.mark.parametrize("tokens,inter", [(7, 12288), (64, 512)])
def test_silu_and_mul(tokens, inter):
    x = torch.randn(tokens, 2 * inter, dtype=torch.bfloat16, device=DEV)
    out = ops.silu_and_mul(x)
    expected = ref.silu_and_mul(x.cpu())
    assert_close_bf16(out.cpu(), expected)


def _assert_fp8_close(got_f8, got_sc, exp_f8, exp_sc, exact=None):
    """fp8 kernel vs CPU reference. The kernel quantizes from unrounded
    f32 registers while the reference quantizes the bf16-rounded op
    output, so scales agree only to bf16 precision and elements may land
    one e4m3 quantum apart — compare dequantized values against the
    reference DEQUANTIZED output within fp8 quantization error (e4m3: 3
    mantissa bits, max rel step 2^-4; two independent quantizations of
    near-identical data differ by <~1 quantum RMS)."""
    torch.testing.assert_close(got_sc.cpu(), exp_sc, atol=1e-5, rtol=2e-2)
    got = got_f8.float().cpu() * got_sc.cpu().unsqueeze(-1)
    exp = exp_f8.float() * exp_sc.unsqueeze(-1)
    rel = (got - exp).norm() / exp.norm().clamp(min=1e-6)
    assert rel.item() < 0.05, rel.item()
    if exact is not None:  # anchor to the unquantized fp32 op output
        rel = (got - exact.float()).norm() / exact.float().norm().clamp(min=1e-6)
        assert rel.item() < 0.06, rel.item()


@pytest.mark.parametrize("tokens,hidden", [(1, 1024), (17, 4096), (64, 8192)])
def test_rms_norm_fp8(tokens, hidden):
    x = torch.randn(tokens, hidden, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(hidden, dtype=torch.bfloat16, device=DEV)
    f8, sc = ops.rms_norm_fp8(x, w, 1e-6)
    ef8, esc = ref.rms_norm_fp8(x.cpu(), w.cpu(), 1e-6)
    _assert_fp8_close(f8, sc, ef8, esc, exact=ref.rms_norm(x.cpu(), w.cpu(), 1e-6))


@pytest.mark.parametrize("tokens,hidden", [(16, 4096), (130, 4096)])
def test_fused_add_rms_norm_fp8(tokens, hidden):
    x = torch.randn(tokens, hidden, dtype=torch.bfloat16, device=DEV)
    res = torch.randn(tokens, hidden, dtype=torch.bfloat16, device=DEV)
    w = torch.randn(hidden, dtype=torch.bfloat16, device=DEV)
    exact, _ = ref.fused_add_rms_norm(x.cpu(), res.cpu(), w.cpu(), 1e-6)
    ef8, esc, eres = ref.fused_add_rms_norm_fp8(x.cpu(), res.cpu(), w.cpu(), 1e-6)
    f8, sc, new_res = ops.fused_add_rms_norm_fp8(x, res, w, 1e-6)
    assert_close_bf16(new_res.cpu(), eres)
    _assert_fp8_close(f8, sc, ef8, esc, exact=exact)


@pytest.mark.parametrize("tokens,inter", [(7, 12288), (64, 512)])
def test_silu_and_mul_fp8(tokens, inter):
    x = torch.randn(tokens, 2 * inter, dtype=torch.bfloat16, device=DEV)
    f8, sc = ops.silu_and_mul_fp8(x)
    ef8, esc = ref.silu_and_mul_fp8(x.cpu())
    _assert_fp8_close(f8, sc, ef8, esc, exact=ref.silu_and_mul(x.cpu()))


@pytest.mark.parametrize("tokens,cols", [(1, 4096), (33, 2048), (256, 4096)])
def test_quant_fp8_rows(tokens, cols):
    x = torch.randn(tokens, cols, dtype=torch.bfloat16, device=DEV)
    f8, sc = ops.quant_fp8_rows(x)
    ef8, esc = ref.quant_fp8_rows(x.cpu())
    _assert_fp8_close(f8, sc, ef8, esc, exact=x.cpu())


@pytest.mark.parametrize("qk_norm", [True, False])
@pytest.mark.parametrize("head_dim", [64, 128])
def test_rope_qk_norm(qk_norm, head_dim):
    T, Hq, Hk = 33, 8, 2
    torch.manual_seed(1)
    # row-strided slices of a fused qkv buffer, like the real model
    qkv = torch.randn(T, (Hq + 2 * Hk) * head_dim, dtype=torch.bfloat16, device=DEV)
    q = qkv[:, : Hq * head_dim]
    k = qkv[:, Hq * head_dim : (Hq + Hk) * head_dim]
    positions = torch.randint(0, 500, (T,), dtype=torch.int32, device=DEV)
    cos_sin = ops.compute_cos_sin_cache(head_dim, 512, 10000.0).to(DEV)
    qw = torch.randn(head_dim, dtype=torch.bfloat16, device=DEV) if qk_norm else None
    kw = torch.randn(head_dim, dtype=torch.bfloat16, device=DEV) if qk_norm else None

    q_ref, k_ref = ref.rope_qk_norm(
        q.cpu().view(T, Hq, head_dim),
        k.cpu().view(T, Hk, head_dim),
        positions.cpu().long(),
        cos_sin.cpu(),
        qw.cpu() if qw is not None else None,
        kw.cpu() if kw is not None else None,
        1e-6,
    )
    ops.rope_qk_norm_(q, k, positions, cos_sin, Hq, Hk, head_dim, qw, kw, 1e-6)
    assert_close_bf16(q.cpu().view(T, Hq, head_dim), q_ref)
    assert_close_bf16(k.cpu().view(T, Hk, head_dim), k_ref)


def test_reshape_and_cache():
    T, Hk, D, bs, nblocks = 37, 4, 128, 16, 12
    k = torch.randn(T, Hk * D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, Hk * D, dtype=torch.bfloat16, device=DEV)
    k_cache = torch.zeros(nblocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV)
    v_cache = torch.zeros_like(k_cache)
    slots = torch.randperm(nblocks * bs, device=DEV)[:T].to(torch.int32)
    k_ref = k_cache.cpu().clone()
    v_ref = v_cache.cpu().clone()
    ref.reshape_and_cache(
        k.cpu().view(T, Hk, D), v.cpu().view(T, Hk, D), k_ref, v_ref, slots.cpu().long()
    )
    ops.reshape_and_cache(k, v, k_cache, v_cache, slots)
    torch.testing.assert_close(k_cache.cpu(), k_ref)
    torch.testing.assert_close(v_cache.cpu(), v_ref)


def test_kv_block_gather_scatter_roundtrip():
    Hk, D, bs, nblocks = 2, 128, 16, 20
    k_cache = torch.randn(nblocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV)
    v_cache = torch.randn_like(k_cache)
    ids = torch.tensor([3, 7, 1, 19], dtype=torch.int32, device=DEV)
    staging = ops.gather_kv_blocks(k_cache, v_cache, ids)
    expected = ref.gather_kv_blocks(k_cache.cpu(), v_cache.cpu(), ids.cpu().long())
    torch.testing.assert_close(staging.cpu(), expected)

    # scatter into a fresh cache at different ids
    k2 = torch.zeros_like(k_cache)
    v2 = torch.zeros_like(v_cache)
    ids2 = torch.tensor([0, 2, 4, 6], dtype=torch.int32, device=DEV)
    ops.scatter_kv_blocks(staging, k2, v2, ids2)
    torch.testing.assert_close(k2[ids2.long()].cpu(), k_cache[ids.long()].cpu())
    torch.testing.assert_close(v2[ids2.long()].cpu(), v_cache[ids.long()].cpu())


@pytest.mark.parametrize("group", [1, 4, 8])
@pytest.mark.parametrize(
    "seq_lens", [[1], [16], [1, 5, 16, 17, 255, 1023]]
)
def test_paged_attention_decode(group, seq_lens):
    torch.manual_seed(2)
    Hk, D, bs = 2, 128, 16
    Hq = Hk * group
    S = len(seq_lens)
    max_blocks = (max(seq_lens) + bs - 1) // bs
    total_blocks = sum((L + bs - 1) // bs for L in seq_lens) + 2
    q = torch.randn(S, Hq, D, dtype=torch.bfloat16, device=DEV)
    k_cache = torch.randn(total_blocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV)
    v_cache = torch.randn_like(k_cache)
    # assign blocks sequentially
    bt = torch.zeros(S, max_blocks, dtype=torch.int32, device=DEV)
    nxt = 1
    for s, L in enumerate(seq_lens):
        n = (L + bs - 1) // bs
        bt[s, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    lens = torch.tensor(seq_lens, dtype=torch.int32, device=DEV)
    scale = 1.0 / math.sqrt(D)
    out = ops.paged_attention_decode(q, k_cache, v_cache, bt, lens, scale)
    expected = ref.paged_attention_decode(
        q.cpu(), k_cache.cpu(), v_cache.cpu(), bt.cpu(), lens.cpu(), scale
    )
    assert_close_bf16(out.cpu(), expected, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("head_dim", [64, 128])
@pytest.mark.parametrize(
    "seq_lens,hq,hk",
    [([64], 4, 4), ([128], 8, 2), ([1, 33, 64, 100, 257], 8, 2)],
)
def test_prefill_attention(head_dim, seq_lens, hq, hk):
    torch.manual_seed(3)
    T = sum(seq_lens)
    q = torch.randn(T, hq, head_dim, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(T, hk, head_dim, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, hk, head_dim, dtype=torch.bfloat16, device=DEV)
    cu = torch.tensor(
        [0] + list(torch.tensor(seq_lens).cumsum(0)), dtype=torch.int32, device=DEV
    )
    scale = 1.0 / math.sqrt(head_dim)
    out = ops.prefill_attention(q, k, v, cu, scale)
    expected = ref.prefill_attention(q.cpu(), k.cpu(), v.cpu(), cu.cpu(), scale)
    assert_close_bf16(out.cpu(), expected, atol=3e-2, rtol=3e-2)


def test_prefill_attention_spiked_scores():
    """Force large per-tile max jumps so online-softmax rescale paths fire
    (guide §5.4 rule 26: bounded random data never exercises them)."""
    torch.manual_seed(4)
    L, hq, hk, D = 256, 2, 2, 128
    q = torch.randn(L, hq, D, dtype=torch.bfloat16, device=DEV)
    k = torch.randn(L, hk, D, dtype=torch.bfloat16, device=DEV) * 0.1
    # spike some K rows late in the sequence: max jumps at a late tile
    k[200] = q[250, 0].sign() * 3.0
    k[77] = q[100, 1].sign() * 2.0
    v = torch.randn(L, hk, D, dtype=torch.bfloat16, device=DEV)
    cu = torch.tensor([0, L], dtype=torch.int32, device=DEV)
    out = ops.prefill_attention(q, k, v, cu)
    expected = ref.prefill_attention(q.cpu(), k.cpu(), v.cpu(), cu.cpu(), 1.0 / math.sqrt(D))
    assert_close_bf16(out.cpu(), expected, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize(
    "ctx_lens,new_lens",
    [([48], [48]), ([128], [32]), ([64, 100, 300], [64, 36, 44])],
)
def test_prefill_attention_paged(ctx_lens, new_lens):
    """Context attention over the paged cache vs the fp32 reference."""
    torch.manual_seed(5)
    Hq, Hk, D, bs = 8, 2, 128, 16
    S = len(ctx_lens)
    Tn = sum(new_lens)
    max_blocks = max((c + bs - 1) // bs for c in ctx_lens)
    total_blocks = sum((c + bs - 1) // bs for c in ctx_lens) + 1
    q = torch.randn(Tn, Hq, D, dtype=torch.bfloat16, device=DEV)
    k_cache = torch.randn(total_blocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV)
    v_cache = torch.randn_like(k_cache)
    bt = torch.zeros(S, max_blocks, dtype=torch.int32, device=DEV)
    nxt = 1
    for s, c in enumerate(ctx_lens):
        n = (c + bs - 1) // bs
        bt[s, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    cu_q = torch.tensor(
        [0] + list(torch.tensor(new_lens).cumsum(0)), dtype=torch.int32, device=DEV
    )
    lens_k = torch.tensor(ctx_lens, dtype=torch.int32, device=DEV)
    scale = 1.0 / math.sqrt(D)
    out = ops.prefill_attention_paged(q, k_cache, v_cache, bt, cu_q, lens_k, scale)
    expected = ref.prefill_attention_paged(
        q.cpu(), k_cache.cpu(), v_cache.cpu(), bt.cpu(), cu_q.cpu(),
        lens_k.cpu(), scale,
    )
    assert_close_bf16(out.cpu(), expected, atol=3e-2, rtol=3e-2)


# ------------------------------------------------------------ fp8 KV cache

def _fp8_cache(t):
    return t.to(torch.float8_e4m3fn)


def test_reshape_and_cache_fp8():
    T, Hk, D, bs, nblocks = 37, 4, 128, 16, 12
    k = torch.randn(T, Hk * D, dtype=torch.bfloat16, device=DEV)
    v = torch.randn(T, Hk * D, dtype=torch.bfloat16, device=DEV)
    k_cache = torch.zeros(nblocks, Hk, bs, D, dtype=torch.float8_e4m3fn,
                          device=DEV)
    v_cache = torch.zeros_like(k_cache)
    slots = torch.randperm(nblocks * bs, device=DEV)[:T].to(torch.int32)
    k_ref = torch.zeros(nblocks, Hk, bs, D, dtype=torch.float8_e4m3fn)
    v_ref = torch.zeros_like(k_ref)
    ref.reshape_and_cache(
        k.cpu().view(T, Hk, D), v.cpu().view(T, Hk, D), k_ref, v_ref,
        slots.cpu().long()
    )
    ops.reshape_and_cache(k, v, k_cache, v_cache, slots)
    # hardware cvt vs torch cast can differ 1 ulp on RNE ties -> compare
    # dequantized with one-quantum tolerance
    torch.testing.assert_close(
        k_cache.float().cpu(), k_ref.float(), atol=0.07, rtol=0.07
    )
    torch.testing.assert_close(
        v_cache.float().cpu(), v_ref.float(), atol=0.07, rtol=0.07
    )


def test_reshape_and_cache_fp8_inv_scale():
    """Per-layer static KV scales: the write multiplies by the inverse
    scale before e4m3 conversion; out-of-range inputs saturate (matching
    the reference's clamp) instead of NaN-ing."""
    T, Hk, D, bs, nblocks = 9, 2, 64, 16, 4
    k = torch.randn(T, Hk * D, dtype=torch.bfloat16, device=DEV) * 3.0
    v = torch.randn(T, Hk * D, dtype=torch.bfloat16, device=DEV) * 600.0
    k_cache = torch.zeros(nblocks, Hk, bs, D, dtype=torch.float8_e4m3fn,
                          device=DEV)
    v_cache = torch.zeros_like(k_cache)
    slots = torch.arange(T, device=DEV, dtype=torch.int32)
    k_ref = torch.zeros(nblocks, Hk, bs, D, dtype=torch.float8_e4m3fn)
    v_ref = torch.zeros_like(k_ref)
    ref.reshape_and_cache(
        k.cpu().view(T, Hk, D), v.cpu().view(T, Hk, D), k_ref, v_ref,
        slots.cpu().long(), 1.0 / 3.0, 1.0 / 4.0
    )
    ops.reshape_and_cache(k, v, k_cache, v_cache, slots, 1.0 / 3.0, 1.0 / 4.0)
    assert torch.isfinite(v_cache.float()).all()
    torch.testing.assert_close(
        k_cache.float().cpu(), k_ref.float(), atol=0.07, rtol=0.07
    )
    torch.testing.assert_close(
        v_cache.float().cpu(), v_ref.float(), atol=12.0, rtol=0.07
    )


@pytest.mark.parametrize("group", [1, 4, 8])
@pytest.mark.parametrize("seq_lens", [[1], [1, 5, 16, 17, 255, 1023]])
def test_paged_attention_decode_fp8kv(group, seq_lens):
    """Decode over an fp8 (e4m3, scale-1) cache: the kernel and the fp32
    reference read the SAME quantized cache, so tolerances stay tight."""
    torch.manual_seed(6)
    Hk, D, bs = 2, 128, 16
    Hq = Hk * group
    S = len(seq_lens)
    max_blocks = (max(seq_lens) + bs - 1) // bs
    total_blocks = sum((L + bs - 1) // bs for L in seq_lens) + 2
    q = torch.randn(S, Hq, D, dtype=torch.bfloat16, device=DEV)
    k_cache = _fp8_cache(
        torch.randn(total_blocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV))
    v_cache = _fp8_cache(
        torch.randn(total_blocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV))
    bt = torch.zeros(S, max_blocks, dtype=torch.int32, device=DEV)
    nxt = 1
    for s, L in enumerate(seq_lens):
        n = (L + bs - 1) // bs
        bt[s, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    lens = torch.tensor(seq_lens, dtype=torch.int32, device=DEV)
    scale = 1.0 / math.sqrt(D)
    out = ops.paged_attention_decode(q, k_cache, v_cache, bt, lens, scale)
    expected = ref.paged_attention_decode(
        q.cpu(), k_cache.cpu(), v_cache.cpu(), bt.cpu(), lens.cpu(), scale
    )
    # G=8 dispatches the fp8 MFA variant, which quantizes Q to e4m3
    # per head (the reference keeps fp32 Q) — one extra quantization
    # step of score error
    tol = 6e-2 if group == 8 else 3e-2
    assert_close_bf16(out.cpu(), expected, atol=tol, rtol=tol)


@pytest.mark.parametrize("head_dim", [64, 128])
@pytest.mark.parametrize(
    "ctx_lens,new_lens",
    [([48], [48]), ([64, 100, 300], [64, 36, 44])],
)
def test_prefill_attention_paged_fp8kv(head_dim, ctx_lens, new_lens):
    torch.manual_seed(7)
    Hq, Hk, bs, D = 8, 2, 16, head_dim
    S = len(ctx_lens)
    Tn = sum(new_lens)
    max_blocks = max((c + bs - 1) // bs for c in ctx_lens)
    total_blocks = sum((c + bs - 1) // bs for c in ctx_lens) + 1
    q = torch.randn(Tn, Hq, D, dtype=torch.bfloat16, device=DEV)
    k_cache = _fp8_cache(
        torch.randn(total_blocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV))
    v_cache = _fp8_cache(
        torch.randn(total_blocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV))
    bt = torch.zeros(S, max_blocks, dtype=torch.int32, device=DEV)
    nxt = 1
    for s, c in enumerate(ctx_lens):
        n = (c + bs - 1) // bs
        bt[s, :n] = torch.arange(nxt, nxt + n, dtype=torch.int32)
        nxt += n
    cu_q = torch.tensor(
        [0] + list(torch.tensor(new_lens).cumsum(0)), dtype=torch.int32,
        device=DEV
    )
    lens_k = torch.tensor(ctx_lens, dtype=torch.int32, device=DEV)
    scale = 1.0 / math.sqrt(D)
    out = ops.prefill_attention_paged(q, k_cache, v_cache, bt, cu_q, lens_k,
                                      scale)
    expected = ref.prefill_attention_paged(
        q.cpu(), k_cache.cpu(), v_cache.cpu(), bt.cpu(), cu_q.cpu(),
        lens_k.cpu(), scale
    )
    assert_close_bf16(out.cpu(), expected, atol=3e-2, rtol=3e-2)


def test_kv_block_copy_fp8_roundtrip():
    Hk, D, bs, nblocks = 2, 128, 16, 20
    k_cache = _fp8_cache(
        torch.randn(nblocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV))
    v_cache = _fp8_cache(
        torch.randn(nblocks, Hk, bs, D, dtype=torch.bfloat16, device=DEV))
    ids = torch.tensor([3, 7, 1, 19], dtype=torch.int32, device=DEV)
    staging = ops.gather_kv_blocks(k_cache, v_cache, ids)
    assert staging.dtype == torch.float8_e4m3fn
    k2 = torch.zeros_like(k_cache)
    v2 = torch.zeros_like(v_cache)
    ids2 = torch.tensor([0, 2, 4, 6], dtype=torch.int32, device=DEV)
    ops.scatter_kv_blocks(staging, k2, v2, ids2)
    torch.testing.assert_close(
        k2[ids2.long()].float().cpu(), k_cache[ids.long()].float().cpu())
    torch.testing.assert_close(
        v2[ids2.long()].float().cpu(), v_cache[ids.long()].float().cpu())


# ---------------------------------------------------------------- MoE


@pytest.mark.parametrize("block_m", [16, 128])
@pytest.mark.parametrize("gate_up", [True, False])
def test_moe_gemm_vs_reference(block_m, gate_up):
    """Grouped MFMA GEMM vs the CPU fp32 reference (same align inputs)."""
    torch.manual_seed(block_m + int(gate_up))
    E, K, N = 5, 256, 128
    T = 300
    w = torch.randn(E, K, (2 * N) if gate_up else N,
                    dtype=torch.bfloat16, device=DEV) * 0.05
    b_packed = ops.pack_moe_weights(w)
    # synthetic block-aligned assignment: uneven expert loads
    counts = [0, 7, block_m, 2 * block_m + 3, 1]
    tiles = [-(-c // block_m) for c in counts if c > 0]
    n_tiles = sum(tiles)
    PM = (n_tiles + 2) * block_m  # slack tiles past n_valid stay untouched
    sorted_ids = torch.zeros(PM, dtype=torch.int32, device=DEV)
    expert_ids = torch.zeros(PM // block_m, dtype=torch.int32, device=DEV)
    p = 0
    t = 0
    for e, c in enumerate(counts):
        if c == 0:
            continue
        nt = -(-c // block_m)
        for i in range(c):
            sorted_ids[p + i] = t % T
            t += 3
        expert_ids[p // block_m : p // block_m + nt] = e
        p += nt * block_m
    n_valid = torch.tensor([n_tiles], dtype=torch.int32, device=DEV)
    a = torch.randn(max(T, PM), K, dtype=torch.bfloat16, device=DEV)
    out = torch.full((PM, N), float("nan"), dtype=torch.bfloat16, device=DEV)
    ops.moe_gemm(out, a, b_packed, sorted_ids, expert_ids, n_valid,
                 block_m, gate_up)
    want = torch.full((PM, N), float("nan"), dtype=torch.bfloat16)
    from fusioninfer_amd.ops import reference as _r

    _r.moe_gemm(want, a.cpu(), b_packed.cpu(), sorted_ids.cpu(),
                expert_ids.cpu(), n_valid.cpu(), block_m, gate_up)
    rows = n_tiles * block_m
    torch.testing.assert_close(
        out[:rows].float().cpu(), want[:rows].float(), atol=3e-2, rtol=3e-2
    )
    # tiles past n_valid untouched on both
    assert out[rows:].isnan().all()


def test_moe_combine_vs_reference():
    torch.manual_seed(3)
    T, topk, H, PM = 23, 4, 256, 128
    y = torch.randn(PM, H, dtype=torch.bfloat16, device=DEV)
    pos = torch.randint(-1, PM, (T * topk,), dtype=torch.int32, device=DEV)
    w = torch.rand(T * topk, dtype=torch.float32, device=DEV)
    out = torch.empty(T, H, dtype=torch.bfloat16, device=DEV)
    ops.moe_combine(out, y, pos, w)
    want = torch.empty(T, H, dtype=torch.bfloat16)
    from fusioninfer_amd.ops import reference as _r

    _r.moe_combine(want, y.cpu(), pos.cpu(), w.cpu())
    torch.testing.assert_close(out.float().cpu(), want.float(),
                               atol=2e-2, rtol=2e-2)


def test_moe_grouped_forward_matches_manual():
    """Full MoEMLP grouped path on GPU vs a per-token manual reference."""
    import torch.nn.functional as F

    from fusioninfer_amd.models.model import MoEMLP
    from fusioninfer_amd.models.registry import get_model_config

    torch.manual_seed(11)
    cfg = get_model_config("tiny-qwen3-moe")
    with torch.device(DEV):
        mlp = MoEMLP(cfg, layer_idx=0)
    for T in (1, 9, 300):
        x = torch.randn(T, cfg.hidden_size, dtype=torch.bfloat16,
                        device=DEV) * 0.5
        got = mlp(x).float()
        logits = x.float() @ mlp.router_weight.float().T
        topv, topi = torch.softmax(logits, -1).topk(mlp.top_k, -1)
        topv = topv / topv.sum(-1, keepdim=True)
        exp = torch.zeros_like(got)
        for t in range(T):
            for k in range(mlp.top_k):
                e = int(topi[t, k])
                gu = x[t].float() @ mlp.gate_up_t[e].float()
                g, u = gu.chunk(2)
                y = (F.silu(g) * u) @ mlp.down_t[e].float()
                exp[t] += float(topv[t, k]) * y
        rel = (got - exp).norm() / exp.norm()
        assert rel.item() < 0.05, (T, rel.item())


def test_moe_gemm_fp8_vs_reference():
    """Grouped fp8 GEMM (dequant epilogue) vs the CPU fp32 reference."""
    torch.manual_seed(9)
    E, K, N, T = 4, 128, 128, 200
    for gate_up in (True, False):
        w = (torch.randn(E, K, (2 * N) if gate_up else N) * 0.1).to(
            torch.float8_e4m3fn).cuda()
        ws = torch.rand(E, (2 * N) if gate_up else N,
                        dtype=torch.float32, device=DEV) * 0.2 + 0.01
        b_packed = ops.pack_moe_weights(w.view(torch.int8)).view(
            torch.float8_e4m3fn)
        block_m = 16
        counts = [5, 0, 2 * block_m, 9]
        tiles = [-(-c // block_m) for c in counts if c > 0]
        n_tiles = sum(tiles)
        PM = (n_tiles + 1) * block_m
        sorted_ids = torch.zeros(PM, dtype=torch.int32, device=DEV)
        expert_ids = torch.zeros(PM // block_m, dtype=torch.int32, device=DEV)
        p = t = 0
        for e, c in enumerate(counts):
            if c == 0:
                continue
            nt = -(-c // block_m)
            for i in range(c):
                sorted_ids[p + i] = t % T
                t += 7
            expert_ids[p // block_m: p // block_m + nt] = e
            p += nt * block_m
        n_valid = torch.tensor([n_tiles], dtype=torch.int32, device=DEV)
        a = (torch.randn(max(T, PM), K) * 0.3).to(torch.float8_e4m3fn).cuda()
        a_s = torch.rand(max(T, PM), dtype=torch.float32, device=DEV) + 0.1
        out = torch.zeros(PM, N, dtype=torch.bfloat16, device=DEV)
        ops.moe_gemm_fp8(out, a, a_s, b_packed, ws, sorted_ids, expert_ids,
                         n_valid, block_m, gate_up)
        want = torch.zeros(PM, N, dtype=torch.bfloat16)
        from fusioninfer_amd.ops import reference as _r

        _r.moe_gemm_fp8(want, a.cpu(), a_s.cpu(), b_packed.cpu(), ws.cpu(),
                        sorted_ids.cpu(), expert_ids.cpu(), n_valid.cpu(),
                        block_m, gate_up)
        rows = n_tiles * block_m
        torch.testing.assert_close(out[:rows].float().cpu(),
                                   want[:rows].float(),
                                   atol=5e-2, rtol=5e-2)


def test_moe_align_kernel_matches_python():
    """The fused alignment kernel vs the torch composition (the CPU/
    semantic oracle, which also runs on GPU tensors): tile counts and
    expert_ids must be EQUAL; slot assignments may permute within an
    expert's segment (arrival vs stable order) but must satisfy the
    same invariants."""
    import types

    from fusioninfer_amd.models.model import MoEMLP

    torch.manual_seed(5)
    cases = [
        (13, 2, 8, 0, 8, 16),     # small, all local
        (300, 8, 16, 4, 12, 16),  # EP slice: half the experts non-local
        (256, 8, 8, 0, 8, 128),   # prefill block_m
        (1, 8, 128, 0, 128, 16),  # decode single token, many experts
    ]
    for T, k, E, e0, e1, bm in cases:
        topi = torch.randint(0, E, (T, k), device=DEV)
        fake = types.SimpleNamespace(e_start=e0, e_end=e1)
        p_sid, p_eid, p_nv, p_pos, p_PM = MoEMLP._moe_align(fake, topi, bm)
        sid, eid, nv, pos, PM = ops.moe_align(topi, e0, e1, bm)
        assert PM == p_PM
        assert int(nv.item()) == int(p_nv.item())
        assert torch.equal(eid, p_eid)
        flat = topi.reshape(-1).cpu()
        pos_c, sid_c = pos.cpu(), sid.cpu()
        local = [(int(e) >= e0) and (int(e) < e1) for e in flat]
        seen = set()
        for i, e in enumerate(flat.tolist()):
            p = int(pos_c[i])
            if not local[i]:
                assert p == -1
                continue
            assert 0 <= p < PM and p not in seen
            seen.add(p)
            assert int(sid_c[p]) == i // k          # row gather correct
            assert int(eid[p // bm]) == e - e0      # right expert tile
        # padding rows are zeroed (the GEMM's dummy-row contract)
        pad = [i for i in range(PM) if i not in seen]
        assert all(int(sid_c[i]) == 0 for i in pad)


def test_moe_router_topk_matches_torch():
    """Fused router tail vs the torch composition: same experts, same
    weights, same (descending) order. Random fp32 logits are distinct
    almost surely, so exact order comparison is stable."""
    torch.manual_seed(9)
    for T, E, k, renorm in [(7, 8, 2, True), (256, 128, 8, True),
                            (33, 64, 4, False), (1, 128, 8, True)]:
        logits = torch.randn(T, E, device=DEV, dtype=torch.float32)
        tv, ti = ops.moe_router_topk(logits, k, renorm)
        probs = torch.softmax(logits, -1)
        etv, eti = probs.topk(k, -1)
        if renorm:
            etv = etv / etv.sum(-1, keepdim=True)
        assert torch.equal(ti.long(), eti), (T, E, k)
        torch.testing.assert_close(tv, etv, atol=2e-6, rtol=2e-6)
