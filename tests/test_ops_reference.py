"""CPU sanity tests for the reference ops (the oracle the HIP kernels
are checked against) — validated against independent torch formulations."""

import math

import torch

from fusioninfer_amd.ops import reference as ref


def test_rms_norm_matches_formula():
    torch.manual_seed(0)
    x = torch.randn(5, 64)
    w = torch.randn(64)
    out = ref.rms_norm(x, w, 1e-6)
    expected = x / (x.pow(2).mean(-1, keepdim=True) + 1e-6).sqrt() * w
    torch.testing.assert_close(out, expected, atol=1e-5, rtol=1e-5)


def test_silu_and_mul():
    x = torch.randn(3, 32)
    out = ref.silu_and_mul(x)
    g, u = x.chunk(2, -1)
    torch.testing.assert_close(out, torch.nn.functional.silu(g) * u)


def test_prefill_attention_vs_sdpa():
    torch.manual_seed(1)
    L, H, D = 37, 4, 32
    q = torch.randn(L, H, D)
    k = torch.randn(L, H, D)
    v = torch.randn(L, H, D)
    cu = torch.tensor([0, L], dtype=torch.int32)
    out = ref.prefill_attention(q, k, v, cu, 1.0 / math.sqrt(D))
    expected = torch.nn.functional.scaled_dot_product_attention(
        q.transpose(0, 1), k.transpose(0, 1), v.transpose(0, 1), is_causal=True
    ).transpose(0, 1)
    torch.testing.assert_close(out, expected, atol=1e-4, rtol=1e-4)


def test_decode_consistent_with_prefill_last_row():
    """Decoding the last token against cached K/V must equal the last row of
    full prefill attention."""
    torch.manual_seed(2)
    L, Hq, Hk, D, bs = 33, 4, 2, 32, 16
    q = torch.randn(L, Hq, D)
    k = torch.randn(L, Hk, D)
    v = torch.randn(L, Hk, D)
    cu = torch.tensor([0, L], dtype=torch.int32)
    full = ref.prefill_attention(q, k, v, cu, 1.0 / math.sqrt(D))

    nblk = (L + bs - 1) // bs
    k_cache = torch.zeros(nblk + 1, Hk, bs, D)
    v_cache = torch.zeros_like(k_cache)
    slots = torch.arange(L)
    ref.reshape_and_cache(k, v, k_cache, v_cache, slots)
    bt = torch.arange(nblk, dtype=torch.int32).unsqueeze(0)
    lens = torch.tensor([L], dtype=torch.int32)
    dec = ref.paged_attention_decode(
        q[-1:], k_cache, v_cache, bt, lens, 1.0 / math.sqrt(D)
    )
    torch.testing.assert_close(dec[0], full[-1], atol=1e-4, rtol=1e-4)


def test_fp8_kv_static_scale_recovers_large_values():
    """Per-layer static k/v scales: writes quantize at the inverse scale;
    the read side folds k_scale into the softmax scalar and v_scale into
    a post-multiply. With |V| well beyond e4m3's 448 max, scale-1.0
    clips badly while a calibrated scale recovers the fp32 answer."""
    torch.manual_seed(3)
    L, Hq, Hk, D, bs = 24, 2, 1, 32, 16
    scale = 1.0 / math.sqrt(D)
    q = torch.randn(L, Hq, D)
    k = torch.randn(L, Hk, D) * 3.0
    v = torch.randn(L, Hk, D) * 600.0   # saturates e4m3 at scale 1.0
    cu = torch.tensor([0, L], dtype=torch.int32)
    want = ref.prefill_attention(q, k, v, cu, scale)[-1]

    nblk = (L + bs - 1) // bs
    bt = torch.arange(nblk, dtype=torch.int32).unsqueeze(0)
    lens = torch.tensor([L], dtype=torch.int32)
    slots = torch.arange(L)

    def run(k_scale, v_scale):
        k_cache = torch.zeros(nblk, Hk, bs, D, dtype=torch.float8_e4m3fn)
        v_cache = torch.zeros_like(k_cache)
        ref.reshape_and_cache(k, v, k_cache, v_cache, slots,
                              1.0 / k_scale, 1.0 / v_scale)
        out = ref.paged_attention_decode(
            q[-1:], k_cache, v_cache, bt, lens, scale * k_scale
        )[0]
        return out * v_scale

    err_unscaled = (run(1.0, 1.0) - want).norm() / want.norm()
    err_scaled = (run(3.0, 4.0) - want).norm() / want.norm()
    assert err_unscaled > 0.2          # saturation destroys the answer
    # e4m3 carries ~6% per-element quantization noise (3 mantissa bits);
    # the calibrated scale must recover to that noise floor
    assert err_scaled < 0.12, float(err_scaled)
    assert err_scaled < err_unscaled / 3


def test_kv_pack_unpack_roundtrip():
    k_cache = torch.randn(8, 2, 16, 32)
    v_cache = torch.randn(8, 2, 16, 32)
    ids = torch.tensor([5, 0, 3])
    staging = ref.gather_kv_blocks(k_cache, v_cache, ids)
    k2, v2 = torch.zeros_like(k_cache), torch.zeros_like(v_cache)
    ref.scatter_kv_blocks(staging, k2, v2, ids)
    torch.testing.assert_close(k2[ids], k_cache[ids])
    torch.testing.assert_close(v2[ids], v_cache[ids])


def test_llama31_rope_scaling_matches_transformers():
    """Our llama3 rope remap equals transformers' reference math."""
    import torch

    from fusioninfer_amd.ops.reference import _llama3_scale_inv_freq

    sc = {"rope_type": "llama3", "factor": 8.0, "low_freq_factor": 1.0,
          "high_freq_factor": 4.0,
          "original_max_position_embeddings": 8192}
    head_dim, theta = 128, 500000.0
    inv = 1.0 / (theta ** (torch.arange(0, head_dim, 2,
                                        dtype=torch.float32) / head_dim))
    ours = _llama3_scale_inv_freq(inv, sc)

    try:
        from transformers import LlamaConfig
        from transformers.modeling_rope_utils import ROPE_INIT_FUNCTIONS

        hf_cfg = LlamaConfig(
            rope_theta=theta, head_dim=head_dim, hidden_size=head_dim * 32,
            num_attention_heads=32, rope_scaling=dict(sc),
            max_position_embeddings=131072,
        )
        theirs, att = ROPE_INIT_FUNCTIONS["llama3"](hf_cfg, device="cpu")
        torch.testing.assert_close(ours, theirs.float(), rtol=1e-5,
                                   atol=1e-7)
        assert att == 1.0
    except (ImportError, KeyError, AttributeError, TypeError):
        # transformers helper interface moved: structural checks below
        pass
    # structural invariants hold either way
    assert torch.allclose(ours[:4], inv[:4])          # high-freq kept
    assert torch.allclose(ours[-4:], inv[-4:] / 8.0)  # low-freq / factor
