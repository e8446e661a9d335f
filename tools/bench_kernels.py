"""Microbenchmarks for the FusionInfer-AMD HIP kernels (run on a GPU box).

Usage: python tools/bench_kernels.py [prefill|decode|elementwise|all]
Prints per-shape timing + achieved TFLOP/s (attention) or GB/s (memory ops).
"""

import math
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

import fusioninfer_amd.ops as ops


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_prefill():
    print("== prefill attention (varlen causal, GQA 32/8, D=128) ==")
    for nseq, L in [(1, 1024), (8, 1024), (4, 2048), (1, 8192), (16, 512)]:
        Hq, Hk, D = 32, 8, 128
        T = nseq * L
        q = torch.randn(T, Hq, D, dtype=torch.bfloat16, device="cuda")
        k = torch.randn(T, Hk, D, dtype=torch.bfloat16, device="cuda")
        v = torch.randn(T, Hk, D, dtype=torch.bfloat16, device="cuda")
        cu = torch.arange(0, nseq + 1, dtype=torch.int32, device="cuda") * L
        lens = [L] * nseq
        ts, tr = ops.build_prefill_tiles(lens, device="cuda")
        scale = 1.0 / math.sqrt(D)
        t = timeit(lambda: ops.prefill_attention(q, k, v, cu, scale, ts, tr))
        # causal flops: 2 matmuls * L*(L+1)/2 * D * 2 per head
        flops = nseq * Hq * 2 * (L * (L + 1) / 2) * D * 2
        print(f"  {nseq}x{L}: {t*1e6:8.1f} us  {flops/t/1e12:7.1f} TF/s")


def bench_decode(group=4):
    Hq, Hk = 32, 32 // group
    print(f"== paged decode attention (GQA 32/{Hk}, G={group}, D=128, bs=16) ==")
    for batch, ctx in [(1, 1024), (8, 1024), (64, 1024), (256, 1024),
                       (64, 4096), (256, 2048), (8, 8192)]:
        D, bs = 128, 16
        nblk = (ctx + bs - 1) // bs
        total_blocks = batch * nblk + 1
        q = torch.randn(batch, Hq, D, dtype=torch.bfloat16, device="cuda")
        k_cache = torch.randn(total_blocks, Hk, bs, D, dtype=torch.bfloat16,
                              device="cuda")
        v_cache = torch.randn_like(k_cache)
        bt = torch.arange(1, total_blocks, dtype=torch.int32, device="cuda")
        bt = bt.view(batch, nblk)
        lens = torch.full((batch,), ctx, dtype=torch.int32, device="cuda")
        t = timeit(lambda: ops.paged_attention_decode(q, k_cache, v_cache, bt, lens))
        gb = batch * ctx * Hk * D * 2 * 2 / 1e9  # K+V bytes actually needed
        k8 = k_cache.to(torch.float8_e4m3fn)
        v8 = v_cache.to(torch.float8_e4m3fn)
        t8 = timeit(lambda: ops.paged_attention_decode(q, k8, v8, bt, lens))
        print(f"  b{batch} ctx{ctx}: {t*1e6:8.1f} us  {gb/t:7.0f} GB/s (KV) | "
              f"fp8kv {t8*1e6:8.1f} us  {gb/2/t8:7.0f} GB/s")


def bench_elementwise():
    print("== rmsnorm / silu_mul / rope (bf16) ==")
    for T in [64, 256, 2048, 8192]:
        H = 4096
        x = torch.randn(T, H, dtype=torch.bfloat16, device="cuda")
        r = torch.randn_like(x)
        w = torch.randn(H, dtype=torch.bfloat16, device="cuda")
        t = timeit(lambda: ops.fused_add_rms_norm(x, r, w, 1e-6))
        gb = T * H * 2 * 4 / 1e9  # r x read, write x2
        print(f"  fused_add_rms_norm T={T}: {t*1e6:7.1f} us {gb/t:6.0f} GB/s")
        y = torch.randn(T, 24576, dtype=torch.bfloat16, device="cuda")
        t = timeit(lambda: ops.silu_and_mul(y))
        gb = T * 24576 * 2 * 1.5 / 1e9
        print(f"  silu_and_mul T={T}:       {t*1e6:7.1f} us {gb/t:6.0f} GB/s")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    torch.manual_seed(0)
    if which in ("prefill", "all"):
        bench_prefill()
    if which in ("decode", "all"):
        bench_decode(group=4)
        bench_decode(group=8)
    if which in ("elementwise", "all"):
        bench_elementwise()
