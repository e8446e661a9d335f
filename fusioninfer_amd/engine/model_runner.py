"""Model runner: batch preparation + forward execution + hipGraph decode.

Pure-decode steps have static shapes (fixed-width block tables), so they
are captured once per batch-size bucket into hipGraphs
(torch.cuda.CUDAGraph == hipGraph on ROCm) and replayed — removing launch
overhead from the latency-critical decode loop (guide: "capture
launch-bound inner loops in hipGraphs"). Prefill and mixed steps run
eager; batches with live LoRA adapters also run eager. Payloads are plain
host lists so the TP driver can broadcast them to worker ranks.

Capability parity: the execution layer of the continuous-batching
engine the reference delegates to its vLLM containers (SURVEY.md §2.3
"Continuous-batching scheduler + paged KV cache").
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from fusioninfer_amd.config import EngineConfig
from fusioninfer_amd.engine.block_manager import BlockManager
from fusioninfer_amd.engine.metadata import AttnMetadata
from fusioninfer_amd.engine.sequence import Sequence
from fusioninfer_amd.models.model import CausalLM
import fusioninfer_amd.ops as ops_mod

_DECODE_BUCKETS = [1, 2, 4, 8, 16, 24, 32, 48, 64, 96, 128, 160, 192, 224, 256, 320, 384, 448, 512]


class ModelRunner:
    def __init__(self, cfg: EngineConfig, device: str):
        self.cfg = cfg
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        mc = cfg.model
        torch.manual_seed(cfg.seed)
        fp8_ckpt = False
        if mc.model_path:
            from fusioninfer_amd.models.weight_loader import checkpoint_is_fp8

            fp8_ckpt = checkpoint_is_fp8(mc.model_path)
            if fp8_ckpt and mc.quantization is None:
                # fp8-native checkpoint implies fp8 serving (vLLM
                # auto-detects this from quantization_config too)
                mc.quantization = "fp8"
        with torch.device(self.device):
            self.model = CausalLM(mc).eval()
        if mc.quantization == "fp8" and fp8_ckpt:
            # fp8-native checkpoint: convert modules FIRST (fp8 storage +
            # scale buffers exist), then load e4m3 weights + weight_scale
            from fusioninfer_amd.models.weight_loader import load_safetensors_dir
            from fusioninfer_amd.quantization import convert_linear_to_fp8

            assert convert_linear_to_fp8(self.model) > 0
            load_safetensors_dir(self.model, mc.model_path)
        else:
            if mc.model_path:
                from fusioninfer_amd.models.weight_loader import (
                    load_safetensors_dir,
                )

                load_safetensors_dir(self.model, mc.model_path)
            if mc.quantization == "fp8":
                # bf16 weights (checkpoint or random init) quantized here;
                # MoE experts are born fp8 inside MoEMLP
                from fusioninfer_amd.quantization import convert_linear_to_fp8

                n = convert_linear_to_fp8(self.model)
                assert n > 0
        if mc.quantization not in (None, "fp8"):
            raise ValueError(f"unknown quantization {mc.quantization!r}")
        if self.is_cuda and os.environ.get(
                "FI_MOE_KEEP_UNPACKED", "0") != "1":
            # weights are final: keep only the MFMA-packed expert layout
            # resident — the freed HBM is picked up by the KV-cache
            # sizing (profile_num_blocks reads mem_get_info after this)
            for m in self.model.modules():
                if hasattr(m, "release_unpacked"):
                    m.release_unpacked()
            torch.cuda.empty_cache()
        self.block_size = cfg.cache.block_size
        self.max_blocks_per_seq = (
            cfg.scheduler.max_model_len + self.block_size - 1
        ) // self.block_size
        self.kv_caches: List[Tuple[torch.Tensor, torch.Tensor]] = []
        self.num_gpu_blocks = 0
        self._graphs: Dict[int, Tuple[object, torch.Tensor]] = {}
        self._graph_pool = None
        self._static: Dict[str, torch.Tensor] = {}
        self.lora_registry = None  # set by the engine when LoRA is enabled

    # ------------------------------------------------------------ kv cache
    def kv_block_bytes(self) -> int:
        mc = self.cfg.model
        import fusioninfer_amd.distributed.parallel_state as ps

        kv_heads = mc.num_kv_heads // ps.tp_world_size()
        esize = 1 if self.cfg.cache.kv_cache_dtype == "fp8" else 2
        layers = self.model.num_local_layers  # PP stages hold a layer slice
        return layers * 2 * kv_heads * self.block_size * mc.head_dim * esize

    def profile_num_blocks(self) -> int:
        import fusioninfer_amd.distributed.parallel_state as ps

        if self.cfg.cache.num_gpu_blocks is not None:
            return ps.tp_all_reduce_min_int(self.cfg.cache.num_gpu_blocks)
        if not self.is_cuda:  # CPU debug serving: small fixed pool
            return 2048
        free_b, total_b = torch.cuda.mem_get_info(self.device)
        usable = int(
            total_b * self.cfg.cache.gpu_memory_utilization
            - (total_b - free_b)
        )
        # headroom for activations / graphs
        usable -= 4 << 30
        n = max(usable // self.kv_block_bytes(), 16)
        # TP ranks must agree on the cache size
        return ps.tp_all_reduce_min_int(int(n))

    def allocate_kv_caches(self) -> None:
        import fusioninfer_amd.distributed.parallel_state as ps

        mc = self.cfg.model
        self.num_gpu_blocks = self.profile_num_blocks()
        kv_heads = mc.num_kv_heads // ps.tp_world_size()
        shape = (self.num_gpu_blocks, kv_heads, self.block_size, mc.head_dim)
        kv_dtype = (
            torch.float8_e4m3fn
            if self.cfg.cache.kv_cache_dtype == "fp8"
            else torch.bfloat16
        )
        self.kv_caches = [
            (
                torch.zeros(shape, dtype=kv_dtype, device=self.device),
                torch.zeros(shape, dtype=kv_dtype, device=self.device),
            )
            for _ in range(self.model.num_local_layers)
        ]

    # ------------------------------------------------------------ prefill
    def build_prefill_payload(
        self,
        seqs: List[Sequence],
        bm: BlockManager,
        chunks: Optional[List[int]] = None,
    ):
        """Host-side batch over each sequence's scheduled prompt chunk
        [num_computed_tokens, +chunk) — prefix-cache hits and previously
        processed chunks are skipped; attention runs over the paged cache
        (context attention). The payload is plain lists so the TP driver
        can broadcast it to worker ranks. `sample` marks sequences whose
        chunk completes the prompt (only those produce a token)."""
        if chunks is None:
            chunks = [
                s.num_prompt_tokens - s.num_computed_tokens for s in seqs
            ]
        input_ids: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        cu = [0]
        new_lens = []
        total_lens = []
        sample = []
        bs = bm.block_size
        for seq, chunk in zip(seqs, chunks):
            toks = seq.all_token_ids
            C = seq.num_computed_tokens or seq.num_cached_tokens
            end = C + chunk
            input_ids.extend(toks[C:end])
            positions.extend(range(C, end))
            # vectorized slot mapping (a per-token Python loop costs
            # milliseconds at 8k-token admission steps)
            pos = np.arange(C, end)
            blocks = np.asarray(seq.block_ids, dtype=np.int64)
            slots.extend((blocks[pos // bs] * bs + pos % bs).tolist())
            cu.append(cu[-1] + chunk)
            new_lens.append(chunk)
            total_lens.append(end)
            sample.append(end >= seq.num_prompt_tokens)
        max_blocks = max(len(s.block_ids) for s in seqs)
        bt = [
            s.block_ids + [0] * (max_blocks - len(s.block_ids)) for s in seqs
        ]
        return {
            "kind": "prefill",
            "ids": input_ids,
            "positions": positions,
            "slots": slots,
            "cu": cu,
            "new_lens": new_lens,
            "total_lens": total_lens,
            "bt": bt,
            "sample": sample,
            "lora_names": [s.lora_name for s in seqs],
        }

    def _lora_batch(self, prefill_payload, decode_payload, np_=0):
        """Group this step's token rows by adapter -> LoRABatch or None."""
        if self.lora_registry is None:
            return None
        groups: Dict[str, List[int]] = {}
        if prefill_payload is not None:
            cu = prefill_payload["cu"]
            for i, name in enumerate(prefill_payload.get("lora_names") or []):
                if name:
                    groups.setdefault(name, []).extend(
                        range(cu[i], cu[i + 1])
                    )
        if decode_payload is not None:
            for i, name in enumerate(decode_payload.get("lora_names") or []):
                if name:
                    groups.setdefault(name, []).append(np_ + i)
        if not groups:
            return None
        from fusioninfer_amd.lora import LoRABatch

        return LoRABatch(
            [
                (
                    self.lora_registry.get(name),
                    torch.tensor(rows, dtype=torch.long, device=self.device),
                )
                for name, rows in groups.items()
            ]
        )

    def run_prefill(self, payload) -> torch.Tensor:
        import fusioninfer_amd.distributed.parallel_state as ps

        if ps.pp_world_size() > 1:
            subs = self._split_prefill_payload(payload)
            if len(subs) > 1:
                return self._run_prefill_pp_pipelined(subs)
        ids, meta, logits_idx = self._prefill_inputs(payload)
        with torch.no_grad():
            return self._forward_and_logits(ids, meta, logits_idx)

    def _prefill_inputs(self, payload):
        dev = self.device
        cu = payload["cu"]
        tile_rows = ops_mod.prefill_tile_rows(payload["new_lens"])
        tile_seq, tile_row0 = ops_mod.build_prefill_tiles(
            payload["new_lens"], device=dev, rows=tile_rows
        )
        meta = AttnMetadata(
            num_prefill_tokens=cu[-1],
            num_decode_tokens=0,
            positions=torch.tensor(
                payload["positions"], dtype=torch.int32, device=dev
            ),
            slot_mapping=torch.tensor(
                payload["slots"], dtype=torch.int32, device=dev
            ),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            tile_seq=tile_seq,
            tile_rows=tile_rows,
            tile_row0=tile_row0,
            prefill_block_tables=torch.tensor(
                payload["bt"], dtype=torch.int32, device=dev
            ),
            prefill_seq_lens_k=torch.tensor(
                payload["total_lens"], dtype=torch.int32, device=dev
            ),
        )
        meta.lora = self._lora_batch(payload, None)
        ids = torch.tensor(payload["ids"], dtype=torch.long, device=dev)
        if payload.get("logits_rows") is not None:
            # spec-decode verification asks for explicit rows (every draft
            # position, not just chunk ends)
            logits_idx = torch.tensor(
                payload["logits_rows"], dtype=torch.long, device=dev
            )
        else:
            sample = payload.get("sample") or [True] * (len(cu) - 1)
            logits_idx = torch.tensor(
                [c - 1 for c, smp in zip(cu[1:], sample) if smp],
                dtype=torch.long,
                device=dev,
            )
        return ids, meta, logits_idx

    # ------------------------------------------- PP microbatch pipelining
    def _split_prefill_payload(self, payload, min_mb_tokens: int = 256):
        """Split a prefill payload into per-stage microbatches (contiguous
        sequence ranges, balanced by token count). Returns [payload] when
        splitting is not worth it (small batches pay comm overhead without
        filling the pipeline)."""
        import fusioninfer_amd.distributed.parallel_state as ps

        cu = payload["cu"]
        n_seqs = len(cu) - 1
        M = min(ps.pp_world_size(), n_seqs, cu[-1] // min_mb_tokens)
        if M < 2:
            return [payload]
        target = cu[-1] / M
        bounds = [0]
        for i in range(1, M):
            want = i * target
            j = bounds[-1] + 1
            while j < n_seqs - (M - 1 - i) and cu[j + 1] <= want:
                j += 1
            bounds.append(j)
        bounds.append(n_seqs)
        rows = payload.get("logits_rows")
        subs = []
        for a, b in zip(bounds, bounds[1:]):
            lo, hi = cu[a], cu[b]
            sp = {
                "kind": "prefill",
                "ids": payload["ids"][lo:hi],
                "positions": payload["positions"][lo:hi],
                "slots": payload["slots"][lo:hi],
                "cu": [c - lo for c in cu[a: b + 1]],
                "new_lens": payload["new_lens"][a:b],
                "total_lens": payload["total_lens"][a:b],
                "bt": payload["bt"][a:b],
                "sample": payload["sample"][a:b],
                "lora_names": (payload.get("lora_names") or [None] * n_seqs)[a:b],
            }
            if rows is not None:
                sp["logits_rows"] = [r - lo for r in rows if lo <= r < hi]
            subs.append(sp)
        return subs

    def _run_prefill_pp_pipelined(self, subs) -> torch.Tensor:
        """Microbatch-overlapped pipeline prefill: every stage loops over
        the microbatches in order, so stage s computes microbatch i while
        stage s+1 computes i-1 (activation p2p over xGMI under RCCL; on
        gloo this is a pure correctness path). The driver pre-posts every
        logits irecv BEFORE its compute loop — sends stay isend — so the
        pipeline drains without rendezvous deadlock."""
        import fusioninfer_amd.distributed.parallel_state as ps

        V = self.cfg.model.vocab_size
        H = self.cfg.model.hidden_size
        dt = torch.bfloat16
        dev = self.device
        last = ps.pp_world_size() - 1
        inputs = [self._prefill_inputs(sp) for sp in subs]
        sends = []
        with torch.no_grad():
            if ps.pp_rank() == 0:
                pending = [
                    ps.pp_irecv_from((int(li.numel()), V), dt, dev, last)
                    for _, _, li in inputs
                    if li.numel() > 0
                ]
                for ids, meta, _ in inputs:
                    hidden, residual = self.model(ids, meta, self.kv_caches)
                    sends.append(ps.pp_isend_next(hidden))
                    sends.append(ps.pp_isend_next(residual))
                for s in sends:
                    s.wait()
                outs = [p.wait() for p in pending]
                if not outs:
                    return torch.empty((0, V), dtype=dt, device=dev)
                return torch.cat(outs, dim=0)
            for ids, meta, logits_idx in inputs:
                T = int(meta.num_prefill_tokens)
                hidden = ps.pp_recv_prev((T, H), dt, dev)
                residual = ps.pp_recv_prev((T, H), dt, dev)
                out = self.model(None, meta, self.kv_caches,
                                 hidden=hidden, residual=residual)
                if ps.pp_is_last():
                    if logits_idx.numel() > 0:
                        logits = self.model.compute_logits(out[logits_idx])
                        sends.append(ps.pp_isend_to(logits, 0))
                else:
                    h2, r2 = out
                    sends.append(ps.pp_isend_next(h2))
                    sends.append(ps.pp_isend_next(r2))
            for s in sends:
                s.wait()
        return torch.empty((0, V), dtype=dt, device=dev)

    def _forward_and_logits(self, ids, meta, logits_idx):
        """Run the model (all pipeline stages) and return logits on the
        driver. PP: activations flow stage-to-stage with p2p send/recv
        (xGMI under RCCL); the last stage computes logits over the sampled
        rows and ships them back to rank 0. Middle stages return an empty
        placeholder."""
        import fusioninfer_amd.distributed.parallel_state as ps

        V = self.cfg.model.vocab_size
        if ps.pp_world_size() == 1:
            hidden = self.model(ids, meta, self.kv_caches)
            if logits_idx.numel() == 0:
                return hidden.new_empty((0, V))
            return self.model.compute_logits(hidden[logits_idx])
        T = ids.shape[0]
        H = self.cfg.model.hidden_size
        dt = torch.bfloat16
        final = None
        if ps.pp_is_first():
            hidden, residual = self.model(ids, meta, self.kv_caches)
            ps.pp_send_next(hidden)
            ps.pp_send_next(residual)
        else:
            hidden = ps.pp_recv_prev((T, H), dt, self.device)
            residual = ps.pp_recv_prev((T, H), dt, self.device)
            out = self.model(None, meta, self.kv_caches,
                             hidden=hidden, residual=residual)
            if ps.pp_is_last():
                final = out
            else:
                h2, r2 = out
                ps.pp_send_next(h2)
                ps.pp_send_next(r2)
        last = ps.pp_world_size() - 1
        S = int(logits_idx.numel())
        if ps.pp_rank() == last:
            if S == 0:
                return torch.empty((0, V), dtype=dt, device=self.device)
            logits = self.model.compute_logits(final[logits_idx])
            ps.pp_send_to(logits, 0)
            return logits
        if ps.pp_rank() == 0:
            if S == 0:
                return torch.empty((0, V), dtype=dt, device=self.device)
            return ps.pp_recv_from((S, V), dt, self.device, last)
        return torch.empty((0, V), dtype=dt, device=self.device)

    def execute_prefill(self, seqs: List[Sequence], bm: BlockManager):
        return self.run_prefill(self.build_prefill_payload(seqs, bm))

    def execute_prefill_hidden(self, seqs: List[Sequence], bm: BlockManager,
                               pooling: str = "last") -> torch.Tensor:
        """Embeddings path (vLLM embed-task analog): one full prefill
        forward, return the POOLED final hidden state per sequence
        (pre-lm_head, [len(seqs), H]). pooling: "last" (decoder-LM
        default) or "mean". PP unsupported (hidden lives on the last
        stage only)."""
        import fusioninfer_amd.distributed.parallel_state as ps

        assert ps.pp_world_size() == 1, "embeddings: PP unsupported"
        payload = self.build_prefill_payload(seqs, bm)
        ids, meta, _ = self._prefill_inputs(payload)
        with torch.no_grad():
            hidden = self.model(ids, meta, self.kv_caches)
        cu = payload["cu"]
        outs = []
        for j in range(len(seqs)):
            h = hidden[cu[j]:cu[j + 1]]
            outs.append(h.float().mean(0) if pooling == "mean"
                        else h[-1].float())
        return torch.stack(outs)

    # ------------------------------------------------------------- mixed
    def build_batch_payload(
        self,
        prefill_seqs: List[Sequence],
        chunks: List[int],
        decode_seqs: List[Sequence],
        bm: BlockManager,
    ):
        """One payload for a mixed step: [prefill chunk tokens | decode
        tokens]. Either segment may be empty."""
        payload = (
            self.build_prefill_payload(prefill_seqs, bm, chunks)
            if prefill_seqs
            else {"kind": "prefill", "ids": [], "positions": [], "slots": [],
                  "cu": [0], "new_lens": [], "total_lens": [], "bt": [],
                  "sample": []}
        )
        payload["kind"] = "mixed"
        d = (
            self.build_decode_payload(decode_seqs, bm)
            if decode_seqs
            else {"ids": [], "positions": [], "slots": [], "lens": [], "bt": []}
        )
        payload["decode"] = d
        return payload

    def build_spec_payload(
        self,
        prefill_seqs: List[Sequence],
        chunks: List[int],
        decode_seqs: List[Sequence],
        drafts: List[List[int]],
        bm: BlockManager,
    ):
        """One prefill-kind payload for a speculative step: real prefill
        chunks first, then each decode sequence as a (1 + k_i)-token chunk
        [last_sampled_token, draft_0 .. draft_{k_i-1}] verified through the
        paged context-attention path. `logits_rows` lists the absolute
        token rows the forward must score: the completing prefill chunks'
        last rows, then EVERY row of each decode chunk (row j predicts the
        token after input j — the acceptance test needs all of them).
        Layout contract with LLMEngine._finish_spec_step: logits come back
        as [prefill-sampled rows | seq0's 1+k_0 rows | seq1's ... ]."""
        input_ids: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        cu = [0]
        new_lens: List[int] = []
        total_lens: List[int] = []
        sample: List[bool] = []
        logits_rows: List[int] = []
        bs = bm.block_size
        all_seqs = list(prefill_seqs) + list(decode_seqs)

        def _extend(seq, toks, C):
            chunk = len(toks)
            input_ids.extend(toks)
            positions.extend(range(C, C + chunk))
            pos = np.arange(C, C + chunk)
            blocks = np.asarray(seq.block_ids, dtype=np.int64)
            slots.extend((blocks[pos // bs] * bs + pos % bs).tolist())
            cu.append(cu[-1] + chunk)
            new_lens.append(chunk)
            total_lens.append(C + chunk)

        for seq, chunk in zip(prefill_seqs, chunks):
            C = seq.num_computed_tokens or seq.num_cached_tokens
            _extend(seq, seq.all_token_ids[C: C + chunk], C)
            done = C + chunk >= seq.num_prompt_tokens
            sample.append(done)
            if done:
                logits_rows.append(cu[-1] - 1)
        for seq, draft in zip(decode_seqs, drafts):
            row0 = cu[-1]
            _extend(seq, [seq.all_token_ids[-1]] + list(draft),
                    seq.num_tokens - 1)
            sample.append(True)
            logits_rows.extend(range(row0, cu[-1]))
        max_blocks = max(len(s.block_ids) for s in all_seqs)
        bt = [
            s.block_ids + [0] * (max_blocks - len(s.block_ids))
            for s in all_seqs
        ]
        return {
            "kind": "prefill",
            "ids": input_ids,
            "positions": positions,
            "slots": slots,
            "cu": cu,
            "new_lens": new_lens,
            "total_lens": total_lens,
            "bt": bt,
            "sample": sample,
            "logits_rows": logits_rows,
            "lora_names": [s.lora_name for s in all_seqs],
        }

    def run_batch(self, payload) -> torch.Tensor:
        """Mixed forward. Pure-decode payloads take the hipGraph path."""
        d = payload["decode"]
        if not payload["ids"]:
            return self.run_decode(d)
        if not d["ids"]:
            return self.run_prefill(payload)
        dev = self.device
        cu = payload["cu"]
        np_ = cu[-1]
        nd = len(d["ids"])
        tile_rows = ops_mod.prefill_tile_rows(payload["new_lens"])
        tile_seq, tile_row0 = ops_mod.build_prefill_tiles(
            payload["new_lens"], device=dev, rows=tile_rows
        )
        max_blocks = max(len(b) for b in d["bt"])
        dbt_np = np.zeros((nd, max_blocks), dtype=np.int32)
        for i, ids in enumerate(d["bt"]):
            dbt_np[i, : len(ids)] = ids
        dbt = torch.from_numpy(dbt_np)
        meta = AttnMetadata(
            num_prefill_tokens=np_,
            num_decode_tokens=nd,
            positions=torch.tensor(
                payload["positions"] + d["positions"],
                dtype=torch.int32, device=dev,
            ),
            slot_mapping=torch.tensor(
                payload["slots"] + d["slots"], dtype=torch.int32, device=dev
            ),
            cu_seqlens=torch.tensor(cu, dtype=torch.int32, device=dev),
            tile_seq=tile_seq,
            tile_rows=tile_rows,
            tile_row0=tile_row0,
            prefill_block_tables=torch.tensor(
                payload["bt"], dtype=torch.int32, device=dev
            ),
            prefill_seq_lens_k=torch.tensor(
                payload["total_lens"], dtype=torch.int32, device=dev
            ),
            block_tables=dbt.to(dev),
            seq_lens=torch.tensor(d["lens"], dtype=torch.int32, device=dev),
        )
        meta.lora = self._lora_batch(payload, d, np_=np_)
        ids = torch.tensor(
            payload["ids"] + d["ids"], dtype=torch.long, device=dev
        )
        if payload.get("logits_rows") is not None:
            # prompt_logprobs asks for explicit rows (prompt positions
            # plus the standard sampling rows); the engine re-maps them
            logits_idx = torch.tensor(
                payload["logits_rows"], dtype=torch.long, device=dev
            )
        else:
            # logits rows: completing prefill chunks first, then every
            # decode row
            pf_idx = [
                c - 1 for c, smp in zip(cu[1:], payload["sample"]) if smp
            ]
            logits_idx = torch.tensor(
                pf_idx + list(range(np_, np_ + nd)), dtype=torch.long,
                device=dev,
            )
        with torch.no_grad():
            return self._forward_and_logits(ids, meta, logits_idx)

    # ------------------------------------------------------------ decode
    def _alloc_static(self, max_bs: int) -> None:
        dev = self.device
        self._static = {
            "ids": torch.zeros(max_bs, dtype=torch.long, device=dev),
            "positions": torch.zeros(max_bs, dtype=torch.int32, device=dev),
            "slots": torch.full((max_bs,), -1, dtype=torch.int32, device=dev),
            "block_tables": torch.zeros(
                (max_bs, self.max_blocks_per_seq), dtype=torch.int32, device=dev
            ),
            "seq_lens": torch.ones(max_bs, dtype=torch.int32, device=dev),
        }

    def _decode_forward(self, bs: int, lora=None) -> torch.Tensor:
        s = self._static
        meta = AttnMetadata(
            num_prefill_tokens=0,
            num_decode_tokens=bs,
            lora=lora,
            positions=s["positions"][:bs],
            slot_mapping=s["slots"][:bs],
            block_tables=s["block_tables"][:bs],
            seq_lens=s["seq_lens"][:bs],
        )
        return self._forward_and_logits(
            s["ids"][:bs], meta,
            torch.arange(bs, dtype=torch.long, device=self.device),
        )

    def capture_decode_graphs(self) -> None:
        import fusioninfer_amd.distributed.parallel_state as ps

        if not self.is_cuda or self.cfg.enforce_eager:
            return
        if ps.pp_world_size() > 1:
            return  # p2p sends inside capture: PP decode runs eager
        # MoE decode captures in both dtypes: the grouped-GEMM paths
        # (bf16 and fp8) are static-shaped — device-side block alignment,
        # static grid + device tile count, no host syncs.
        max_bs = min(self.cfg.scheduler.max_num_seqs, _DECODE_BUCKETS[-1])
        buckets = [b for b in _DECODE_BUCKETS if b <= max_bs]
        if not self._static:
            self._alloc_static(max(buckets))
        torch.cuda.synchronize()
        with torch.no_grad():
            for bs in reversed(buckets):  # largest first shares the pool best
                self._decode_forward(bs)  # warm-up (workspaces, autotuners)
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                if self._graph_pool is None:
                    with torch.cuda.graph(g):
                        out = self._decode_forward(bs)
                    self._graph_pool = g.pool()
                else:
                    with torch.cuda.graph(g, pool=self._graph_pool):
                        out = self._decode_forward(bs)
                self._graphs[bs] = (g, out)
        torch.cuda.synchronize()

    def build_decode_payload(self, seqs: List[Sequence], bm: BlockManager):
        return {
            "kind": "decode",
            "ids": [s.all_token_ids[-1] for s in seqs],
            "positions": [s.num_tokens - 1 for s in seqs],
            "slots": [bm.slot_for(s, s.num_tokens - 1) for s in seqs],
            "lens": [s.num_tokens for s in seqs],
            "bt": [list(s.block_ids) for s in seqs],
            "lora_names": [s.lora_name for s in seqs],
        }

    def _fill_decode_inputs(self, payload, bs: int) -> None:
        n = len(payload["lens"])
        st = self._static
        if payload["ids"] is not None:
            st["ids"][:n].copy_(
                torch.tensor(payload["ids"], dtype=torch.long),
                non_blocking=True,
            )
        st["positions"][:n].copy_(
            torch.tensor(payload["positions"], dtype=torch.int32),
            non_blocking=True,
        )
        st["slots"][:n].copy_(
            torch.tensor(payload["slots"], dtype=torch.int32),
            non_blocking=True,
        )
        st["seq_lens"][:n].copy_(
            torch.tensor(payload["lens"], dtype=torch.int32), non_blocking=True
        )
        # numpy row fill: a per-row torch.tensor loop cost 2-4 ms/step at
        # batch 256 (measured with FI_STEP_TIMING)
        bt_np = np.zeros((n, self.max_blocks_per_seq), dtype=np.int32)
        for i, ids in enumerate(payload["bt"]):
            bt_np[i, : len(ids)] = ids
        st["block_tables"][:n].copy_(
            torch.from_numpy(bt_np), non_blocking=True
        )
        if bs > n:  # padding rows: decode block 0, pos 0, len 1, slot -1
            st["ids"][n:bs].zero_()
            st["positions"][n:bs].zero_()
            st["slots"][n:bs].fill_(-1)
            st["seq_lens"][n:bs].fill_(1)
            st["block_tables"][n:bs].zero_()

    def run_decode_device_ids(self, payload, ids_dev: torch.Tensor):
        """Graph-replay decode with ON-DEVICE input ids: the pipelined
        engine feeds the previous step's sampled ids straight into the
        static graph buffer (device-to-device copy, no host sync).
        payload["ids"] is None; the batch size comes from "lens"."""
        n = len(payload["lens"])
        if not self._static:
            self._alloc_static(max(self.cfg.scheduler.max_num_seqs, n))
        st = self._static
        st["ids"][:n].copy_(ids_dev[:n])
        bucket = next((b for b in _DECODE_BUCKETS if b >= n), None)
        use_graph = (
            bucket is not None and bucket in self._graphs and self.is_cuda
        )
        bs = bucket if use_graph else n
        self._fill_decode_inputs(payload, bs)
        if use_graph:
            g, out = self._graphs[bucket]
            g.replay()
            return out[:n]
        with torch.no_grad():
            return self._decode_forward(n)[:n]

    def run_decode(self, payload) -> torch.Tensor:
        n = len(payload["ids"])
        if not self._static:
            self._alloc_static(max(self.cfg.scheduler.max_num_seqs, n))
        lora = self._lora_batch(None, payload, np_=0)
        bucket = next((b for b in _DECODE_BUCKETS if b >= n), None)
        use_graph = (
            bucket is not None and bucket in self._graphs and self.is_cuda
            and lora is None  # LoRA steps run eager
        )
        bs = bucket if use_graph else n
        self._fill_decode_inputs(payload, bs)
        if use_graph:
            g, out = self._graphs[bucket]
            g.replay()
            return out[:n]
        with torch.no_grad():
            return self._decode_forward(n, lora=lora)[:n]

    def execute_decode(self, seqs: List[Sequence], bm: BlockManager):
        return self.run_decode(self.build_decode_payload(seqs, bm))
