"""Decoder-only transformer (Llama/Qwen3 family) on the FusionInfer-AMD ops.

Every hot op is a hand-written CDNA4 HIP kernel (fusioninfer_amd/ops);
GEMMs are hipBLASLt via F.linear. TP sharding is Megatron-style: heads
sharded for attention, intermediate sharded for the MLP, one RCCL
all-reduce after o_proj and one after down_proj per layer.

Capability parity: the model execution the reference delegates to
vLLM images (`vllm serve --model Qwen/Qwen3-8B`, reference README.md:136-148).
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

import fusioninfer_amd.ops as ops
from fusioninfer_amd.config import ModelConfig
from fusioninfer_amd.distributed import parallel_state as ps
from fusioninfer_amd.distributed.layers import (
    MergedColumnParallelLinear,
    ReplicatedLinear,
    RowParallelLinear,
)
from fusioninfer_amd.engine.metadata import AttnMetadata


class Attention(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        self.layer_idx = layer_idx
        tp = ps.tp_world_size()
        assert cfg.num_heads % tp == 0 and cfg.num_kv_heads % tp == 0, (
            "head counts must divide TP size"
        )
        self.num_heads = cfg.num_heads // tp
        self.num_kv_heads = cfg.num_kv_heads // tp
        self.head_dim = cfg.head_dim
        self.q_size = self.num_heads * self.head_dim
        self.kv_size = self.num_kv_heads * self.head_dim
        self.scale = 1.0 / math.sqrt(self.head_dim)
        # fp8-KV per-layer static dequant scales (vLLM k_scale/v_scale;
        # set by the weight loader from calibrated checkpoints, 1.0
        # default). Plain python floats: K folds into the softmax scalar
        # and V into a post-multiply, so no device reads — graph-safe.
        self.k_scale = 1.0
        self.v_scale = 1.0
        self.eps = cfg.rms_norm_eps
        hidden = cfg.hidden_size
        self.qkv_proj = MergedColumnParallelLinear(
            hidden,
            [
                cfg.num_heads * cfg.head_dim,
                cfg.num_kv_heads * cfg.head_dim,
                cfg.num_kv_heads * cfg.head_dim,
            ],
            bias=cfg.attention_bias,   # Qwen2.5 family
        )
        self.o_proj = RowParallelLinear(cfg.num_heads * cfg.head_dim, hidden)
        if cfg.qk_norm:
            self.q_norm_weight = nn.Parameter(
                torch.ones(cfg.head_dim, dtype=torch.bfloat16), requires_grad=False
            )
            self.k_norm_weight = nn.Parameter(
                torch.ones(cfg.head_dim, dtype=torch.bfloat16), requires_grad=False
            )
        else:
            self.q_norm_weight = None
            self.k_norm_weight = None

    def forward(
        self,
        x,  # [T, H] bf16, or (x_fp8, scales) from a fused fp8 epilogue
        meta: AttnMetadata,
        kv_cache,  # (k_cache, v_cache) [B, Hk, bs, D]
        cos_sin: torch.Tensor,
    ) -> torch.Tensor:
        qkv = self.qkv_proj(x)
        if meta.lora is not None:
            meta.lora.apply(self.layer_idx, "qkv", x, qkv)
        q = qkv[:, : self.q_size]
        k = qkv[:, self.q_size : self.q_size + self.kv_size]
        v = qkv[:, self.q_size + self.kv_size :]
        ops.rope_qk_norm_(
            q, k, meta.positions, cos_sin,
            self.num_heads, self.num_kv_heads, self.head_dim,
            self.q_norm_weight, self.k_norm_weight, self.eps,
        )
        k_cache, v_cache = kv_cache
        fp8_kv = k_cache.dtype == torch.float8_e4m3fn
        ks = self.k_scale if fp8_kv else 1.0
        vs = self.v_scale if fp8_kv else 1.0
        ops.reshape_and_cache(k, v, k_cache, v_cache, meta.slot_mapping,
                              1.0 / ks, 1.0 / vs)
        # K's dequant scale folds EXACTLY into the softmax scalar:
        # scale*(Q . ks*K8) == (scale*ks)*(Q . K8); V's is a linear
        # post-multiply on the attention output (applied below).
        scale = self.scale * ks

        T = qkv.shape[0]
        np_, nd = meta.num_prefill_tokens, meta.num_decode_tokens
        out = torch.empty(T, self.q_size, dtype=qkv.dtype, device=qkv.device)
        if np_ > 0:
            # context attention over the paged cache (the new tokens' K/V
            # were just written above); cached prefixes are never recomputed
            o = ops.prefill_attention_paged(
                q[:np_].view(np_, self.num_heads, self.head_dim),
                k_cache,
                v_cache,
                meta.prefill_block_tables,
                meta.cu_seqlens,
                meta.prefill_seq_lens_k,
                scale,
                tile_seq=meta.tile_seq,
                tile_row0=meta.tile_row0,
                tile_rows=meta.tile_rows or None,
            )
            out[:np_] = o.view(np_, self.q_size)
        if nd > 0:
            o = ops.paged_attention_decode(
                q[np_:].view(nd, self.num_heads, self.head_dim),
                k_cache,
                v_cache,
                meta.block_tables,
                meta.seq_lens,
                scale,
            )
            out[np_:] = o.view(nd, self.q_size)
        if vs != 1.0:
            out *= vs
        if meta.lora is not None:
            return self.o_proj.forward_with_lora(out, meta.lora,
                                                 self.layer_idx, "o")
        return self.o_proj(out)


class MLP(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        self.layer_idx = layer_idx
        self.fp8 = cfg.quantization == "fp8"
        self.gate_up_proj = MergedColumnParallelLinear(
            cfg.hidden_size, [cfg.intermediate_size, cfg.intermediate_size]
        )
        self.down_proj = RowParallelLinear(cfg.intermediate_size, cfg.hidden_size)
        self.inter_per_rank = cfg.intermediate_size // ps.tp_world_size()

    def forward(self, x, lora=None) -> torch.Tensor:
        gu = self.gate_up_proj(x)
        if self.fp8:
            # fused SwiGLU + fp8 row quant straight into the fp8 down_proj
            return self.down_proj(ops.silu_and_mul_fp8(gu))
        if lora is not None:
            lora.apply(self.layer_idx, "gate_up", x, gu)
        act = ops.silu_and_mul(gu)
        if lora is not None:
            return self.down_proj.forward_with_lora(
                act, lora, self.layer_idx, "down"
            )
        return self.down_proj(act)


class MoEMLP(nn.Module):
    """Qwen3-MoE sparse MLP: softmax router over E experts, top-k with
    normalized weights, SwiGLU experts (HF qwen3_moe semantics:
    norm_topk_prob renormalizes the selected probabilities).

    Expert parallelism over the TP group: experts are partitioned across
    ranks (each rank holds E/tp complete experts); tokens routed to a
    remote expert contribute nothing locally and the per-layer all-reduce
    — the same one the dense row-parallel convention already requires —
    sums expert outputs across ranks (allreduce-combine EP; all-to-all
    dispatch is the multi-node variant). Expert weights
    are drawn from per-(layer, expert) seeds so the same engine seed
    yields identical experts under any EP layout."""

    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        self.layer_idx = layer_idx
        tp = ps.tp_world_size()
        E = cfg.num_experts
        assert E % tp == 0, "num_experts must divide TP size (EP sharding)"
        self.num_experts = E
        self.top_k = cfg.num_experts_per_tok
        self.norm_topk = cfg.norm_topk_prob
        self.e_start = ps.tp_rank() * (E // tp)
        self.e_end = self.e_start + E // tp
        self.fp8 = cfg.quantization == "fp8"
        H, inter = cfg.hidden_size, cfg.moe_intermediate_size
        base = torch.initial_seed() % (2**62)
        dev = torch.empty(0).device  # honor the ambient torch.device context
        gen = torch.Generator(device="cpu")
        gen.manual_seed(base)
        # experts drawn on the CPU from per-(layer, expert) seeds so every
        # rank generates identical experts regardless of the EP layout
        self.router_weight = nn.Parameter(
            (torch.randn(E, H, generator=gen, device="cpu") * 0.02)
            .to(dtype=torch.bfloat16, device=dev),
            requires_grad=False,
        )
        # per-expert seeded draws; on GPU use the device RNG (Philox is
        # seed-deterministic and rank-independent, so EP layouts still
        # agree) — CPU-generating 128 experts x 48 layers took minutes
        if dev.type == "cuda":
            egen = torch.Generator(device=dev)
            edev = dev
        else:
            egen = gen
            edev = "cpu"
        if self.fp8:
            # fp8 expert storage ([O, I] layout + per-expert per-output-
            # channel scales): the fp8 path runs per-expert _scaled_mm
            # (no bmm), halving expert-weight bytes. Same seeded draws as
            # the bf16 path, so fp8-vs-bf16 numerics are comparable.
            from fusioninfer_amd.quantization import quantize_weight_fp8

            gu8, gus, dn8, dns = [], [], [], []
            for e in range(self.e_start, self.e_end):
                egen.manual_seed(base + 7919 * (e + 1))
                g_w = torch.randn(2 * inter, H, generator=egen,
                                  device=edev) * 0.02
                d_w = torch.randn(H, inter, generator=egen,
                                  device=edev) * 0.02
                w8, s = quantize_weight_fp8(g_w)
                gu8.append(w8)
                gus.append(s)
                w8, s = quantize_weight_fp8(d_w)
                dn8.append(w8)
                dns.append(s)
            self.gate_up_fp8 = nn.Parameter(
                torch.stack(gu8).to(dev), requires_grad=False
            )  # [E_local, 2*I, H] e4m3
            self.down_fp8 = nn.Parameter(
                torch.stack(dn8).to(dev), requires_grad=False
            )  # [E_local, H, I] e4m3
            self.register_buffer(
                "gate_up_scale", torch.stack(gus).to(dev), persistent=False
            )
            self.register_buffer(
                "down_scale", torch.stack(dns).to(dev), persistent=False
            )
            if dev.type == "cuda":
                self._packed_weights_fp8()  # before KV-cache sizing
            return
        gup, down = [], []
        for e in range(self.e_start, self.e_end):
            egen.manual_seed(base + 7919 * (e + 1))
            gup.append(
                torch.randn(2 * inter, H, generator=egen, device=edev) * 0.02
            )
            down.append(
                torch.randn(H, inter, generator=egen, device=edev) * 0.02
            )
        # stored PRE-TRANSPOSED ([in, out] per expert) so both bmm B
        # operands are contiguous: hipBLASLt/rocBLAS pick a custom
        # Stream-K kernel for some transposed-view strided-batch shapes
        # (e.g. [8,890,768]x[8,768,2048]^T) that memory-faults on gfx950
        # — reproduced standalone, ROCm 7.2
        self.gate_up_t = nn.Parameter(
            torch.stack([w.T.contiguous() for w in gup]).to(
                dtype=torch.bfloat16, device=dev),
            requires_grad=False,
        )  # [E_local, H, 2*I]
        self.down_t = nn.Parameter(
            torch.stack([w.T.contiguous() for w in down]).to(
                dtype=torch.bfloat16, device=dev),
            requires_grad=False,
        )  # [E_local, I, H]
        # pack for the grouped GEMM NOW, before the engine sizes the KV
        # cache from free HBM — lazy packing after cache allocation OOMs
        # (packed copies must be part of the model's memory footprint)
        if dev.type == "cuda":
            self._packed_weights()

    def _forward_fp8(self, x) -> torch.Tensor:
        """fp8 expert compute: per-expert torch._scaled_mm (hipBLASLt fp8)
        with the fused silu_and_mul_fp8 epilogue between the two GEMMs.
        Input is the (fp8, per-row scale) tuple the fused norm emits.
        Launch-bound at high expert counts — the round-2 GPU path is the
        grouped fp8 MFMA GEMM (_forward_fp8_grouped); this per-expert
        loop remains as the CPU reference."""
        x8, xs = x
        if x8.is_cuda:
            return self._forward_fp8_grouped(x8, xs)
        T, H = x8.shape
        xd = x8.float() * xs.unsqueeze(1).float()
        logits = xd @ self.router_weight.float().T  # [T, E]
        probs = torch.softmax(logits, dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        if self.norm_topk:
            topv = topv / topv.sum(dim=-1, keepdim=True)
        flat_e = topi.reshape(-1)
        flat_w = topv.reshape(-1)
        out = torch.zeros(T, H, dtype=torch.float32, device=x8.device)
        cuda = x8.is_cuda
        for el in range(self.e_end - self.e_start):
            sel = (flat_e == self.e_start + el).nonzero(as_tuple=True)[0]
            if sel.numel() == 0:
                continue
            rows = sel // self.top_k
            if cuda:
                gu = torch._scaled_mm(
                    x8[rows],
                    self.gate_up_fp8[el].t(),
                    scale_a=xs[rows].unsqueeze(1),
                    scale_b=self.gate_up_scale[el].unsqueeze(0),
                    out_dtype=torch.bfloat16,
                )
                a8, a_s = ops.silu_and_mul_fp8(gu)
                y = torch._scaled_mm(
                    a8,
                    self.down_fp8[el].t(),
                    scale_a=a_s.unsqueeze(1),
                    scale_b=self.down_scale[el].unsqueeze(0),
                    out_dtype=torch.bfloat16,
                )
            else:  # CPU reference: dequantize + fp32 matmul
                wg = self.gate_up_fp8[el].float() \
                    * self.gate_up_scale[el].unsqueeze(1).float()
                gu = (xd[rows] @ wg.t()).to(torch.bfloat16)
                a8, a_s = ops.silu_and_mul_fp8(gu)
                wd = self.down_fp8[el].float() \
                    * self.down_scale[el].unsqueeze(1).float()
                y = (a8.float() * a_s.unsqueeze(1).float()) @ wd.t()
            out.index_add_(0, rows, y.float() * flat_w[sel].unsqueeze(1))
        return ps.tp_all_reduce(out.to(torch.bfloat16))

    def invalidate_packed(self):
        """Drop the packed-weight cache (the weight loader calls this:
        mutations through Parameter.data don't bump tensor._version).
        Frees the stale packed copy immediately so the repack never
        doubles up, then repacks eagerly on GPU (keeps the footprint
        visible before any KV-cache sizing)."""
        stored = self.gate_up_fp8 if self.fp8 else self.gate_up_t
        if stored.numel() == 0 and self.e_end > self.e_start:
            raise RuntimeError(
                "unpacked MoE expert weights were released; call "
                "ensure_unpacked() before mutating them"
            )
        self._pack_version = None
        self._pack_cache = None
        if self.fp8:
            if self.gate_up_fp8.is_cuda:
                self._packed_weights_fp8()
        elif self.gate_up_t.is_cuda:
            self._packed_weights()

    def release_unpacked(self):
        """Free the unpacked expert weights: GPU serving reads ONLY the
        MFMA-packed layout, so keeping both resident doubles expert HBM
        (round-1 debt). Zero-expert stand-ins preserve shape[1:]/dtype/
        device (forward paths read .shape[2] only); ensure_unpacked()
        re-materializes exact contents from the packed cache if the
        loader needs to write again. The runner calls this after weight
        load and BEFORE KV-cache sizing, so the freed bytes become KV
        blocks. CPU keeps the unpacked form (it IS the compute path)."""
        if self.e_end == self.e_start:
            return
        if self.fp8:
            if not self.gate_up_fp8.is_cuda or self.gate_up_fp8.numel() == 0:
                return
            self._packed_weights_fp8()
            self.gate_up_fp8.data = self.gate_up_fp8.data.new_empty(
                (0,) + tuple(self.gate_up_fp8.shape[1:]))
            self.down_fp8.data = self.down_fp8.data.new_empty(
                (0,) + tuple(self.down_fp8.shape[1:]))
            self._pack_version = (self.gate_up_fp8._version,
                                  self.down_fp8._version)
        else:
            if not self.gate_up_t.is_cuda or self.gate_up_t.numel() == 0:
                return
            self._packed_weights()
            self.gate_up_t.data = self.gate_up_t.data.new_empty(
                (0,) + tuple(self.gate_up_t.shape[1:]))
            self.down_t.data = self.down_t.data.new_empty(
                (0,) + tuple(self.down_t.shape[1:]))
            self._pack_version = (self.gate_up_t._version,
                                  self.down_t._version)

    def ensure_unpacked(self):
        """Inverse of release_unpacked: re-materialize the unpacked
        params EXACTLY from the packed cache (unpack_moe_weights is pure
        view/permute tensor ops, so it runs on-device). No-op when the
        unpacked form is already resident."""
        if self.e_end == self.e_start:
            return
        from fusioninfer_amd.ops.reference import unpack_moe_weights

        if self.fp8:
            if self.gate_up_fp8.numel():
                return
            gu_p, _, dn_p, _ = self._pack_cache
            self.gate_up_fp8.data = (
                unpack_moe_weights(gu_p.view(torch.int8))
                .view(torch.float8_e4m3fn).transpose(1, 2).contiguous()
            )
            self.down_fp8.data = (
                unpack_moe_weights(dn_p.view(torch.int8))
                .view(torch.float8_e4m3fn).transpose(1, 2).contiguous()
            )
            self._pack_version = (self.gate_up_fp8._version,
                                  self.down_fp8._version)
        else:
            if self.gate_up_t.numel():
                return
            gu_p, dn_p = self._pack_cache
            self.gate_up_t.data = unpack_moe_weights(gu_p)
            self.down_t.data = unpack_moe_weights(dn_p)
            self._pack_version = (self.gate_up_t._version,
                                  self.down_t._version)

    def _packed_weights_fp8(self):
        """MFMA-packed fp8 expert weights ([E, K, N] order: the stored
        [O, I] weights transpose to K-major) + their per-out-channel
        scales, cached like the bf16 form."""
        v = (self.gate_up_fp8._version, self.down_fp8._version)
        if getattr(self, "_pack_version", None) != v or \
                getattr(self, "_pack_cache", None) is None:
            gu = self.gate_up_fp8.data.transpose(1, 2).contiguous()  # [E,H,2I]
            dn = self.down_fp8.data.transpose(1, 2).contiguous()     # [E,I,H]
            self._pack_cache = (
                ops.pack_moe_weights(gu.view(torch.int8))
                .view(torch.float8_e4m3fn),
                self.gate_up_scale.float().contiguous(),
                ops.pack_moe_weights(dn.view(torch.int8))
                .view(torch.float8_e4m3fn),
                self.down_scale.float().contiguous(),
            )
            self._pack_version = v
        return self._pack_cache

    def _forward_fp8_grouped(self, x8, xs) -> torch.Tensor:
        """GPU fp8 path: grouped e4m3 MFMA GEMMs with dequant epilogues —
        half the expert-weight bytes of bf16 (decode is weight-BW-bound)
        and no per-expert host loops, so fp8 MoE decode captures too."""
        T, H = x8.shape
        xd = x8.float() * xs.unsqueeze(1).float()
        logits = xd @ self.router_weight.float().T
        topv, topi = self._router_topk(logits)
        E_local = self.e_end - self.e_start
        block_m = 128 if T * self.top_k >= 64 * E_local else 16
        sorted_ids, expert_ids, n_valid, pos, PM = self._moe_align(
            topi, block_m
        )
        gu_p, gu_s, dn_p, dn_s = self._packed_weights_fp8()
        inter = self.down_fp8.shape[2]
        act = torch.empty(PM, inter, dtype=torch.bfloat16, device=x8.device)
        ops.moe_gemm_fp8(act, x8, xs.float(), gu_p, gu_s, sorted_ids,
                         expert_ids, n_valid, block_m, gate_up=True)
        a8, a_s = ops.quant_fp8_rows(act)
        y = torch.empty(PM, H, dtype=torch.bfloat16, device=x8.device)
        ops.moe_gemm_fp8(y, a8, a_s, dn_p, dn_s, sorted_ids, expert_ids,
                         n_valid, block_m, gate_up=False)
        out = torch.empty(T, H, dtype=torch.bfloat16, device=x8.device)
        ops.moe_combine(out, y, pos, topv.reshape(-1).float().contiguous())
        return ps.tp_all_reduce(out)

    def _packed_weights(self):
        """MFMA-fragment-packed expert weights for the grouped GEMM,
        cached; invalidated explicitly by the weight loader and on any
        tracked in-place mutation (tensor._version). After weight load
        the runner drops the unpacked copy (release_unpacked), leaving
        only this layout resident on GPU."""
        v = (self.gate_up_t._version, self.down_t._version)
        if getattr(self, "_pack_version", None) != v:
            self._pack_cache = (
                ops.pack_moe_weights(self.gate_up_t.data),
                ops.pack_moe_weights(self.down_t.data),
            )
            self._pack_version = v
        return self._pack_cache

    def _router_topk(self, logits: torch.Tensor):
        """Router tail: softmax + top-k + optional renormalize. GPU uses
        the fused wave-per-token kernel; the torch composition is the
        CPU path and the semantic reference."""
        if logits.is_cuda:
            return ops.moe_router_topk(logits, self.top_k, self.norm_topk)
        probs = torch.softmax(logits, dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        if self.norm_topk:
            topv = topv / topv.sum(dim=-1, keepdim=True)
        return topv, topi

    def _moe_align(self, topi: torch.Tensor, block_m: int):
        """Device-side block alignment (no host sync, static shapes —
        hipGraph-capturable): sort token assignments by local expert, pad
        each expert's segment to a multiple of block_m, and emit
        - sorted_ids [PM]: token row per padded slot (padding -> row 0)
        - expert_ids [PM/block_m]: local expert per m-tile
        - n_valid [1] int32: real tile count (device scalar)
        - pos [T*topk] int32: padded slot per assignment (-1 = non-local)
        """
        if topi.is_cuda:
            # one fused kernel instead of this ~12-op torch composition
            # (same contract; within-expert slot order is arrival order,
            # which is output-invariant — see ops.moe_align)
            return ops.moe_align(topi, self.e_start, self.e_end, block_m)
        T, k = topi.shape
        dev = topi.device
        E_local = self.e_end - self.e_start
        flat = topi.reshape(-1)
        local = (flat >= self.e_start) & (flat < self.e_end)
        key = torch.where(local, flat - self.e_start,
                          torch.full_like(flat, E_local))
        order = torch.argsort(key, stable=True)
        # scatter_add, not bincount: bincount is hipGraph-capture-unsafe
        # (hipErrorStreamCaptureUnsupported on ROCm 7.2)
        counts = torch.zeros(E_local + 1, dtype=torch.long,
                             device=dev).scatter_add_(
            0, key, torch.ones_like(key)
        )[:E_local]
        tiles = (counts + block_m - 1) // block_m
        pad_starts = (tiles.cumsum(0) - tiles) * block_m
        seg_starts = torch.cat(
            [counts.cumsum(0) - counts, counts.sum().unsqueeze(0)]
        )
        key_sorted = key[order]
        rank = torch.arange(T * k, device=dev) - seg_starts[key_sorted]
        PM = ((T * k + block_m - 1) // block_m + E_local) * block_m
        pad_pos = torch.where(
            key_sorted < E_local,
            pad_starts[key_sorted.clamp(max=E_local - 1)] + rank,
            torch.full_like(rank, PM),
        )
        sorted_ids = torch.zeros(PM + 1, dtype=torch.int32, device=dev)
        sorted_ids[pad_pos] = (order // k).int()
        pos = torch.empty(T * k, dtype=torch.int32, device=dev)
        pos[order] = torch.where(
            key_sorted < E_local, pad_pos, torch.full_like(pad_pos, -1)
        ).int()
        tile_cum = tiles.cumsum(0)
        expert_ids = torch.searchsorted(
            tile_cum, torch.arange(PM // block_m, device=dev), right=True
        ).clamp_(max=max(E_local - 1, 0)).int()
        n_valid = tile_cum[-1:].int()
        return sorted_ids[:PM].contiguous(), expert_ids, n_valid, pos, PM

    def _forward_grouped(self, x: torch.Tensor) -> torch.Tensor:
        """GPU path: hand-written grouped MFMA GEMM over block-aligned
        expert segments with fused SwiGLU, then a deterministic weighted
        combine. No host syncs — MoE decode is hipGraph-capturable."""
        T, H = x.shape
        E_local = self.e_end - self.e_start
        logits = x.float() @ self.router_weight.float().T
        topv, topi = self._router_topk(logits)
        # block_m: 16 in the few-rows-per-expert (decode) regime, 128 when
        # segments are long enough for the 4x in-register B-reuse variant
        block_m = 128 if T * self.top_k >= 64 * E_local else 16
        sorted_ids, expert_ids, n_valid, pos, PM = self._moe_align(
            topi, block_m
        )
        gu_packed, dn_packed = self._packed_weights()
        inter = self.gate_up_t.shape[2] // 2
        act = torch.empty(PM, inter, dtype=x.dtype, device=x.device)
        ops.moe_gemm(act, x, gu_packed, sorted_ids, expert_ids, n_valid,
                     block_m, gate_up=True)
        y = torch.empty(PM, H, dtype=x.dtype, device=x.device)
        ops.moe_gemm(y, act, dn_packed, sorted_ids, expert_ids, n_valid,
                     block_m, gate_up=False)
        out = torch.empty(T, H, dtype=x.dtype, device=x.device)
        ops.moe_combine(out, y, pos, topv.reshape(-1).float().contiguous())
        return ps.tp_all_reduce(out)

    def forward(self, x: torch.Tensor, lora=None) -> torch.Tensor:
        if self.fp8:
            return self._forward_fp8(x)
        if x.is_cuda:
            return self._forward_grouped(x)
        T, H = x.shape
        E_local = self.e_end - self.e_start
        logits = (x.float() @ self.router_weight.float().T)  # [T, E]
        probs = torch.softmax(logits, dim=-1)
        topv, topi = probs.topk(self.top_k, dim=-1)
        if self.norm_topk:
            topv = topv / topv.sum(dim=-1, keepdim=True)
        # capacity-padded batched expert compute: token-slots sorted by
        # expert, scattered into [E_local, cap, H], TWO bmm launches per
        # layer regardless of E (a per-expert GEMM loop is launch-bound
        # at 128 experts; fused grouped MFMA GEMM is the round-2 step up)
        flat_e = topi.reshape(-1)
        order = torch.argsort(flat_e, stable=True)
        counts = torch.bincount(flat_e, minlength=self.num_experts)
        sorted_e = flat_e[order]
        tok_of = order // self.top_k
        w_of = topv.reshape(-1)[order]
        starts = counts.cumsum(0) - counts
        pos_in_e = (
            torch.arange(sorted_e.numel(), device=x.device) - starts[sorted_e]
        )
        local = (sorted_e >= self.e_start) & (sorted_e < self.e_end)
        cap = int(counts[self.e_start : self.e_end].max())
        out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
        if cap > 0:
            dest = ((sorted_e - self.e_start) * cap + pos_in_e)[local]
            rows = tok_of[local]
            xpad = torch.zeros(E_local * cap, H, dtype=x.dtype, device=x.device)
            xpad[dest] = x[rows]
            gu = torch.bmm(xpad.view(E_local, cap, H), self.gate_up_t)
            act = ops.silu_and_mul(gu.view(E_local * cap, -1))
            y = torch.bmm(
                act.view(E_local, cap, -1), self.down_t
            ).view(E_local * cap, H)
            out.index_add_(0, rows, y[dest].float() * w_of[local, None])
        out = out.to(x.dtype)
        return ps.tp_all_reduce(out)


class DecoderLayer(nn.Module):
    def __init__(self, cfg: ModelConfig, layer_idx: int = 0):
        super().__init__()
        self.input_norm_weight = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=torch.bfloat16), requires_grad=False
        )
        self.post_norm_weight = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=torch.bfloat16), requires_grad=False
        )
        self.self_attn = Attention(cfg, layer_idx)
        self.mlp = (MoEMLP(cfg, layer_idx) if cfg.is_moe
                    else MLP(cfg, layer_idx))
        self.eps = cfg.rms_norm_eps
        self.fp8 = cfg.quantization == "fp8"

    def forward(self, hidden, residual, meta, kv_cache, cos_sin):
        if self.fp8:
            # fused epilogues: the norms emit (fp8, per-row scale) directly
            # into the fp8 GEMMs — the bf16 normalized row never hits HBM
            if residual is None:
                residual = hidden.clone()
                x = ops.rms_norm_fp8(hidden, self.input_norm_weight, self.eps)
            else:
                f8, sc, residual = ops.fused_add_rms_norm_fp8(
                    hidden, residual, self.input_norm_weight, self.eps
                )
                x = (f8, sc)
            hidden = self.self_attn(x, meta, kv_cache, cos_sin)
            f8, sc, residual = ops.fused_add_rms_norm_fp8(
                hidden, residual, self.post_norm_weight, self.eps
            )
            hidden = self.mlp((f8, sc))
            return hidden, residual
        if residual is None:
            residual = hidden.clone()
            hidden = ops.rms_norm(hidden, self.input_norm_weight, self.eps)
        else:
            hidden, residual = ops.fused_add_rms_norm(
                hidden, residual, self.input_norm_weight, self.eps
            )
        hidden = self.self_attn(hidden, meta, kv_cache, cos_sin)
        hidden, residual = ops.fused_add_rms_norm(
            hidden, residual, self.post_norm_weight, self.eps
        )
        hidden = self.mlp(hidden, lora=meta.lora)
        return hidden, residual


class CausalLM(nn.Module):
    """The full model: embedding -> N decoder layers -> final norm -> lm_head.

    Pipeline parallelism: stage r holds a contiguous slice
    of the layers; the first stage owns the embedding, the last the final
    norm + lm_head. Random init draws every module from a per-module seed
    derived from the caller's seed, so the SAME seed produces the SAME
    weights regardless of the PP/TP layout (stages skip modules without
    desyncing the stream; TP ranks still slice identical full matrices)."""

    def __init__(self, cfg: ModelConfig):
        super().__init__()
        self.cfg = cfg
        pp = ps.pp_world_size()
        ppr = ps.pp_rank()
        self.pp_first = ppr == 0
        self.pp_last = ppr == pp - 1
        self.layer_start = cfg.num_layers * ppr // pp
        self.layer_end = cfg.num_layers * (ppr + 1) // pp

        base_seed = torch.initial_seed() % (2**62)

        def seeded(i):
            torch.manual_seed(base_seed + 1000003 * (i + 1))

        if self.pp_first or (self.pp_last and cfg.tie_word_embeddings):
            seeded(0)
            self.embed_tokens = nn.Parameter(
                torch.empty(cfg.vocab_size, cfg.hidden_size,
                            dtype=torch.bfloat16).normal_(0.0, 0.02),
                requires_grad=False,
            )
        else:
            self.embed_tokens = None
        layers = []
        for i in range(self.layer_start, self.layer_end):
            seeded(1 + i)
            layers.append(DecoderLayer(cfg, i))
        self.layers = nn.ModuleList(layers)
        if self.pp_last:
            self.final_norm_weight = nn.Parameter(
                torch.ones(cfg.hidden_size, dtype=torch.bfloat16),
                requires_grad=False,
            )
            if cfg.tie_word_embeddings:
                self.lm_head = None
            else:
                seeded(1 + cfg.num_layers)
                self.lm_head = ReplicatedLinear(cfg.hidden_size, cfg.vocab_size)
        else:
            self.final_norm_weight = None
            self.lm_head = None
        torch.manual_seed(base_seed + 777)  # same post-init state on all ranks
        cos_sin = ops.compute_cos_sin_cache(
            cfg.head_dim, cfg.max_position_embeddings, cfg.rope_theta,
            rope_scaling=cfg.rope_scaling,
        )
        self.register_buffer("cos_sin", cos_sin, persistent=False)

    @property
    def num_local_layers(self) -> int:
        return self.layer_end - self.layer_start

    def forward(
        self,
        input_ids,                   # [T] int (first stage; None otherwise)
        meta: AttnMetadata,
        kv_caches,                   # list of (k_cache, v_cache) per LOCAL layer
        hidden: torch.Tensor = None,     # PP stages > 0: received activations
        residual: torch.Tensor = None,
    ):
        if self.pp_first:
            hidden = self.embed_tokens[input_ids]
            residual = None
        for layer, kv in zip(self.layers, kv_caches):
            hidden, residual = layer(hidden, residual, meta, kv, self.cos_sin)
        if not self.pp_last:
            return hidden, residual
        hidden, _ = ops.fused_add_rms_norm(
            hidden, residual, self.final_norm_weight, self.cfg.rms_norm_eps
        )
        return hidden

    def compute_logits(self, hidden: torch.Tensor) -> torch.Tensor:
        """hidden: [S, H] (already gathered to the sampled rows)."""
        if self.lm_head is None:
            return hidden @ self.embed_tokens.T
        return self.lm_head(hidden)
