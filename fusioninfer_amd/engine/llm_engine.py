"""LLMEngine: the per-GPU serving engine (continuous batching over paged KV).

One engine instance per GPU process (TP ranks share one logical engine:
rank 0 drives scheduling; all ranks execute the same batch — v1 runs
TP via symmetric SPMD stepping, see distributed/parallel_state.py).
"""

from __future__ import annotations

import itertools
import os
import time
from typing import Dict, List, Optional

import torch

from fusioninfer_amd.config import EngineConfig
from fusioninfer_amd.engine.block_manager import BlockManager
from fusioninfer_amd.engine.model_runner import ModelRunner
from fusioninfer_amd.engine.sampler import Sampler
from fusioninfer_amd.engine.scheduler import Scheduler
from fusioninfer_amd.engine.sequence import SamplingParams, Sequence, SeqStatus


class InsufficientBlocksError(RuntimeError):
    """PD consumer has no room for an imported KV batch right now —
    backpressure signal (NOT a crash): the recv loop waits for decode
    completions to free blocks, and past its wait budget the transfer is
    drained and the request rejected (HTTP 429)."""


class RequestOutput:
    def __init__(self, seq: Sequence, new_token_ids=None):
        self.request_id = seq.seq_id
        self.prompt_token_ids = seq.prompt_token_ids
        self.output_token_ids = list(seq.output_token_ids)
        # tokens produced THIS step: one for regular decode, up to k+1 for
        # an accepted speculative draft (streaming pushes each)
        self.new_token_ids = list(new_token_ids) if new_token_ids else (
            self.output_token_ids[-1:] if self.output_token_ids else []
        )
        self.finished = seq.is_finished()
        self.finish_reason = (
            (seq.finish_reason or "stop") if self.finished else None
        )
        self.logprobs = list(seq.logprobs)
        self.prompt_logprobs = list(seq.prompt_logprobs)
        self.ttft = seq.ttft
        self.arrival_time = seq.arrival_time
        self.finish_time = seq.finish_time


class LLMEngine:
    def __init__(self, cfg: EngineConfig, device: Optional[str] = None):
        from fusioninfer_amd.distributed import parallel_state as ps

        if (cfg.parallel.tensor_parallel_size > 1
                or cfg.parallel.pipeline_parallel_size > 1):
            ps.init_distributed(
                cfg.parallel.tensor_parallel_size,
                backend=cfg.parallel.distributed_backend,
                pipeline_parallel_size=cfg.parallel.pipeline_parallel_size,
            )
        else:
            ps.ensure_single_process()
        self._ps = ps
        self.is_driver = ps.tp_rank() == 0 and ps.pp_rank() == 0
        self.cfg = cfg
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.runner = ModelRunner(cfg, device)
        self.runner.allocate_kv_caches()
        self.block_manager = BlockManager(
            self.runner.num_gpu_blocks,
            cfg.cache.block_size,
            enable_prefix_caching=cfg.cache.enable_prefix_caching,
        )
        self.scheduler = Scheduler(cfg.scheduler, self.block_manager)
        self._swapped: Dict[str, "torch.Tensor"] = {}
        self._swap_bytes = 0
        self.num_swap_outs = 0
        self.num_preemptions = 0
        if (cfg.cache.swap_space_gb > 0
                and cfg.parallel.tensor_parallel_size == 1
                and cfg.parallel.pipeline_parallel_size == 1):
            self.scheduler.swap_out_fn = self._swap_out
        self.sampler = Sampler(cfg.seed, device)
        from fusioninfer_amd.engine.spec_decode import build_proposer

        if (cfg.speculative is not None
                and getattr(cfg.speculative, "method", "ngram") != "ngram"):
            assert (cfg.parallel.tensor_parallel_size
                    * cfg.parallel.pipeline_parallel_size) == 1, (
                "draft-model speculation runs single-process; use the "
                "ngram method under TP/PP"
            )
        self.proposer = build_proposer(cfg.speculative, cfg, device)
        self.num_spec_draft_tokens = 0
        self.num_spec_accepted_tokens = 0
        self.runner.capture_decode_graphs()
        self._req_counter = itertools.count()
        self.seqs: Dict[str, Sequence] = {}
        self._held: Dict[str, Sequence] = {}
        # PD: the KV connector (set by the server/pd wiring on every TP
        # rank; worker ranks use it for pd_send / pd_recv commands)
        self.kv_connector = None
        self._import_holder = None
        self._timing = os.environ.get("FI_STEP_TIMING") == "1"
        self._tacc = {}
        # pipelined decode: the last launched-but-unprocessed pure-decode
        # step (seqs, on-device sampled ids, pinned host copy, event).
        # Host emission of step N overlaps the GPU running step N+1; the
        # next step's input ids are fed device-to-device, so the chain
        # never waits on a host sync to launch. FI_ASYNC_DECODE=0 opts
        # out (every step then drains synchronously, round-1 behavior).
        self._pending = None
        self._async_decode = os.environ.get("FI_ASYNC_DECODE", "1") == "1"
        self._drained_backlog: List[RequestOutput] = []
        self.num_async_steps = 0  # pipelined continuations taken
        # running stats for metrics / EPP scorers
        self.num_finished = 0
        self.num_generated_tokens = 0
        self.num_prefilled_tokens = 0
        self.ttft_sum = 0.0
        self.e2e_latency_sum = 0.0
        # Prometheus-style histogram counts (vLLM bucket boundaries)
        self.ttft_buckets = [0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0,
                             2.5, 5.0, 10.0]
        self.e2e_buckets = [0.3, 0.5, 0.8, 1.0, 1.5, 2.0, 2.5, 5.0,
                            10.0, 15.0, 30.0, 60.0]
        self.ttft_hist = [0] * (len(self.ttft_buckets) + 1)
        self.e2e_hist = [0] * (len(self.e2e_buckets) + 1)
        self.tpot_buckets = [0.01, 0.025, 0.05, 0.075, 0.1, 0.15, 0.2,
                             0.3, 0.4, 0.5, 0.75, 1.0]
        self.tpot_hist = [0] * (len(self.tpot_buckets) + 1)
        self.step_time_sum = 0.0
        self.num_steps = 0

    # ------------------------------------------------------------ requests
    def add_lora(self, name: str, rank: int = 16, alpha: float = 32.0,
                 seed: Optional[int] = None):
        """Register a LoRA adapter (random-init when seed given, zero-B
        otherwise — zero-B adapters are numerically the base model).

        TP > 1: the driver broadcasts the registration over the payload
        channel so every worker holds its shard BEFORE any batch
        references the adapter by name (adapters are generated from the
        seed on the CPU identically on all ranks, then sharded by
        tp_rank — same scheme as the base weights)."""
        # fp8 mode feeds (fp8, scale) tuples through the layers; LoRA's
        # bf16 shrink/expand GEMMs would need the bf16 activations
        assert self.cfg.model.quantization != "fp8", (
            "LoRA is not supported with --quantization fp8"
        )
        if (self.cfg.parallel.tensor_parallel_size
                * self.cfg.parallel.pipeline_parallel_size) > 1:
            assert self.is_driver
            self._ps.tp_broadcast_object(
                {"kind": "add_lora", "name": name, "rank": rank,
                 "alpha": alpha, "seed": seed}
            )
        return self._register_lora(name, rank, alpha, seed)

    def _lora_registry(self):
        from fusioninfer_amd.lora import LoRARegistry

        if self.runner.lora_registry is None:
            self.runner.lora_registry = LoRARegistry(
                max_loras=self.cfg.max_loras,
                max_cpu_loras=self.cfg.max_cpu_loras,
                device=self.device,
            )
        return self.runner.lora_registry

    def _register_lora(self, name, rank, alpha, seed):
        """Per-rank adapter construction (driver and TP workers)."""
        from fusioninfer_amd.lora import LoRAAdapter

        self._lora_registry()
        adapter = LoRAAdapter(
            name, rank, alpha, self.cfg.model, device=self.device, seed=seed
        )
        self.runner.lora_registry.add(adapter)
        return adapter

    def add_lora_from_path(self, name: str, path: str):
        """Load a PEFT adapter directory (vLLM /v1/load_lora_adapter).
        TP > 1: the registration broadcasts; all ranks read the shared
        path and keep their own shard."""
        from fusioninfer_amd.lora import LoRAAdapter

        if (self.cfg.parallel.tensor_parallel_size
                * self.cfg.parallel.pipeline_parallel_size) > 1:
            assert self.is_driver
            self._ps.tp_broadcast_object(
                {"kind": "add_lora_path", "name": name, "path": path}
            )
        self._lora_registry()
        adapter = LoRAAdapter.from_safetensors(
            name, path, self.cfg.model, device=self.device
        )
        self.runner.lora_registry.add(adapter)
        return adapter

    def remove_lora(self, name: str) -> bool:
        reg = self.runner.lora_registry
        return bool(reg and reg.remove(name))

    def active_loras(self) -> List[str]:
        reg = self.runner.lora_registry
        return reg.names() if reg else []

    def add_request(
        self,
        prompt_token_ids: List[int],
        sampling: Optional[SamplingParams] = None,
        request_id: Optional[str] = None,
        lora_name: Optional[str] = None,
        priority: int = 0,
    ) -> str:
        if request_id is None:
            request_id = f"req-{next(self._req_counter)}"
        seq = Sequence(request_id, prompt_token_ids, sampling or SamplingParams())
        seq.lora_name = lora_name
        seq.priority = priority
        if seq.num_prompt_tokens > self.cfg.scheduler.max_model_len:
            raise ValueError(
                f"prompt length {seq.num_prompt_tokens} exceeds "
                f"max_model_len {self.cfg.scheduler.max_model_len}"
            )
        self.seqs[request_id] = seq
        self.scheduler.add(seq)
        return request_id

    def has_unfinished(self) -> bool:
        return self.scheduler.has_work()

    def embed(self, prompts: List[List[int]], pooling: str = "last"):
        """Embeddings (vLLM embed task / OpenAI /v1/embeddings): one
        prefill forward per prompt, pooled final hidden state
        (pre-lm_head), L2-normalized. Temporary blocks come from the
        engine's pool and are freed immediately. Driver-only single-rank
        path (TP/PP embedding is not wired through the worker loop)."""
        tp = (self.cfg.parallel.tensor_parallel_size
              * self.cfg.parallel.pipeline_parallel_size)
        if tp != 1:
            raise ValueError("embeddings unsupported with TP/PP")
        outs = []
        for p in prompts:
            seq = Sequence(f"embed-{next(self._req_counter)}", p,
                           SamplingParams())
            self.block_manager.allocate(seq)
            # force full recompute: a prefix-cache hit would shrink the
            # payload to the uncached suffix (empty for a full hit) and
            # the pooled hidden must cover the whole prompt
            seq.num_cached_tokens = 0
            try:
                h = self.runner.execute_prefill_hidden(
                    [seq], self.block_manager, pooling=pooling
                )[0]
            finally:
                self.block_manager.free(seq)
            v = h / h.norm().clamp_min(1e-12)
            outs.append(v.cpu().tolist())
        return outs

    def abort_request(self, request_id: str) -> bool:
        """Cancel a request wherever it is (waiting / running); frees its
        cache blocks. Returns True if it was found."""
        from fusioninfer_amd.engine.sequence import SeqStatus

        if self._pending is not None:
            # flush the pipelined in-flight step: removing a seq from
            # running mid-chain would desync the device-id row mapping.
            # The drained outputs belong to OTHER requests — surface
            # them through the next step() call
            self._drained_backlog.extend(self._drain_pending())

        seq = self.seqs.pop(request_id, None)
        if seq is None:
            return False
        if self.proposer is not None:
            self.proposer.release(seq)
        staging = self._swapped.pop(request_id, None)
        if staging is not None:
            self._swap_bytes -= staging.numel() * staging.element_size()
        if seq in self.scheduler.waiting:
            self.scheduler.waiting.remove(seq)
        if seq in self.scheduler.running:
            self.scheduler.running.remove(seq)
            self.block_manager.free(seq)
        elif seq.block_ids:
            self.block_manager.free(seq)
        seq.status = SeqStatus.FINISHED
        return True

    # ----------------------------------------------------- swap preemption
    def _swap_out(self, seq: Sequence) -> bool:
        """Park a preempted sequence's KV blocks in host memory (the
        vLLM --swap-space behavior). Called by the scheduler BEFORE the
        victim's blocks are freed. Returns False when the swap budget is
        exhausted (scheduler falls back to recompute)."""
        from fusioninfer_amd.distributed.kv_transfer import pack_kv_blocks

        cap = int(self.cfg.cache.swap_space_gb * (1 << 30))
        ids = torch.tensor(seq.block_ids, dtype=torch.int32,
                           device=self.runner.device)
        staging = pack_kv_blocks(self.runner.kv_caches, ids).to(
            "cpu", non_blocking=False
        )
        nbytes = staging.numel() * staging.element_size()
        if self._swap_bytes + nbytes > cap:
            return False
        self._swapped[seq.seq_id] = staging
        self._swap_bytes += nbytes
        self.num_swap_outs += 1
        return True

    def _swap_in(self, seq: Sequence) -> None:
        from fusioninfer_amd.distributed.kv_transfer import unpack_kv_blocks

        staging = self._swapped.pop(seq.seq_id)
        self._swap_bytes -= staging.numel() * staging.element_size()
        ids = torch.tensor(seq.block_ids, dtype=torch.int32,
                           device=self.runner.device)
        unpack_kv_blocks(staging.to(self.runner.device), self.runner.kv_caches,
                         ids)

    # ------------------------------------------------------- PD interfaces
    def prefill_export(self, prompt_token_ids: List[int]):
        """PD producer: prefill one request, sample its first token, and
        return (first_token, block_ids) with the blocks HELD for KV export.
        Call release_held() after the connector has shipped the blocks."""
        from fusioninfer_amd.engine.sequence import SamplingParams

        req_id = self.add_request(
            prompt_token_ids, SamplingParams(max_tokens=1, temperature=0.0)
        )
        seq = self.seqs[req_id]
        seq.hold_blocks = True
        while req_id in self.seqs:
            self.step()
        assert len(seq.output_token_ids) == 1
        self._held[req_id] = seq
        return req_id, seq.output_token_ids[0], seq.block_ids

    def release_held(self, req_id: str) -> None:
        seq = self._held.pop(req_id)
        self.block_manager.free(seq)

    def add_export_request(self, prompt_token_ids: List[int],
                           priority: int = 0) -> str:
        """PD producer, ASYNC form: admit a prefill-only request whose
        blocks are held for KV export. It batches with other prefills in
        the normal serving loop; when its single token arrives, ship the
        KV via pd_send_held() / release with release_held()."""
        from fusioninfer_amd.engine.sequence import SamplingParams

        req_id = self.add_request(
            prompt_token_ids,
            SamplingParams(max_tokens=1, temperature=0.0),
            priority=priority,
        )
        seq = self.seqs[req_id]
        seq.hold_blocks = True
        self._held[req_id] = seq
        return req_id

    def pd_send_held(self, connector, req_id: str, prompt_len: int,
                     first_token: int, tag: int) -> None:
        """Ship a held export's KV (this rank's shard) and, under TP,
        direct worker ranks to ship theirs (same block ids — block
        layout is replicated across TP ranks)."""
        seq = self._held[req_id]
        if self.is_driver and self.exec_world_size() > 1:
            self._ps.tp_broadcast_object({
                "kind": "pd_send",
                "block_ids": list(seq.block_ids),
                "prompt_len": prompt_len,
                "first_token": first_token,
                "tag": tag,
            })
        connector.send_kv(self.runner.kv_caches, seq.block_ids, prompt_len,
                          first_token, tag)

    def pd_recv_broadcast(self, block_ids: List[int]) -> None:
        """PD consumer driver under TP: direct worker ranks to post their
        shard recvs into the same block ids."""
        if self.is_driver and self.exec_world_size() > 1:
            self._ps.tp_broadcast_object({
                "kind": "pd_recv", "block_ids": list(block_ids),
            })

    def exec_world_size(self) -> int:
        return (self.cfg.parallel.tensor_parallel_size
                * self.cfg.parallel.pipeline_parallel_size)

    def allocate_import_blocks(self, num_blocks: int) -> List[int]:
        """PD consumer: reserve blocks the connector will scatter KV into.
        Raises InsufficientBlocksError (backpressure) when full."""
        from fusioninfer_amd.engine.sequence import SamplingParams, Sequence

        if self.block_manager.num_free() < num_blocks:
            raise InsufficientBlocksError(
                f"need {num_blocks} KV blocks, "
                f"{self.block_manager.num_free()} free"
            )
        holder = Sequence("_import", [0] * (num_blocks * self.cfg.cache.block_size),
                          SamplingParams())
        self.block_manager.allocate(holder)
        self._import_holder = holder
        return holder.block_ids

    def take_import_holder(self):
        """PD consumer: detach the block-holder sequence created by the last
        allocate_import_blocks() call so admission can happen later (the
        HTTP decode request that claims this KV may not have arrived yet)."""
        holder = self._import_holder
        self._import_holder = None
        return holder

    def add_imported_request(
        self,
        prompt_len: int,
        first_token: int,
        sampling: Optional["SamplingParams"] = None,
        request_id: Optional[str] = None,
        holder=None,
    ) -> str:
        """PD consumer: register a request whose prompt KV was imported into
        the blocks reserved by allocate_import_blocks(). The sequence joins
        the RUNNING set directly (its prefill happened on the prefiller)."""
        from fusioninfer_amd.engine.sequence import SamplingParams, Sequence, SeqStatus

        if holder is None:
            holder = self.take_import_holder()
        if request_id is None:
            request_id = f"req-{next(self._req_counter)}"
        seq = Sequence(request_id, [0] * prompt_len, sampling or SamplingParams())
        seq.block_ids = holder.block_ids
        seq.append_token(first_token)
        seq.status = SeqStatus.RUNNING
        seq.num_computed_tokens = prompt_len  # KV imported, nothing to prefill
        self.seqs[request_id] = seq
        self.scheduler.running.append(seq)
        return request_id

    # ------------------------------------------------------------ metrics
    def gpu_cache_usage(self) -> float:
        total = self.block_manager.num_blocks
        return 1.0 - self.block_manager.num_free() / max(total, 1)

    def num_waiting(self) -> int:
        return self.scheduler.num_waiting

    def num_running(self) -> int:
        return self.scheduler.num_running

    # ------------------------------------------------------------ stepping
    def step(self) -> List[RequestOutput]:
        """One engine iteration (TP driver / single rank). Returns outputs
        for sequences that produced a token this step (finished flagged).

        TP > 1: rank 0 schedules and broadcasts the batch payload; worker
        ranks (in worker_loop) execute the same forward so the per-layer
        RCCL all-reduces line up."""
        assert self.is_driver, "only TP rank 0 steps; others run worker_loop"
        pre: List[RequestOutput] = self._drained_backlog
        if pre:
            self._drained_backlog = []
        if self._pending is not None:
            outs = self._try_continue_async()
            if outs is not None:
                return pre + outs
            pre = pre + self._drain_pending()
        step_t0 = time.monotonic()
        tp = (self.cfg.parallel.tensor_parallel_size
              * self.cfg.parallel.pipeline_parallel_size)
        batch = self.scheduler.schedule()
        t_sched = time.monotonic()
        for seq in batch.swap_in:
            self._swap_in(seq)
        if batch.is_empty:
            return pre
        self.num_preemptions += len(batch.preempted)
        want_plp = any(
            s.sampling.prompt_logprobs is not None for s in batch.prefill_seqs
        )
        spec_drafts = None
        if self.proposer is not None and batch.decode_seqs and not want_plp:
            cap = self.proposer.cfg.disable_by_batch_size
            if not cap or len(batch.decode_seqs) <= cap:
                spec_drafts = self._propose_drafts(batch.decode_seqs)
        chunk_starts = [
            s.num_computed_tokens or s.num_cached_tokens
            for s in batch.prefill_seqs
        ]
        if spec_drafts is not None:
            payload = self.runner.build_spec_payload(
                batch.prefill_seqs, batch.prefill_chunks, batch.decode_seqs,
                spec_drafts, self.block_manager,
            )
        else:
            payload = self.runner.build_batch_payload(
                batch.prefill_seqs, batch.prefill_chunks, batch.decode_seqs,
                self.block_manager,
            )
            if want_plp:
                rows = set()
                cu = payload["cu"]
                for j, s in enumerate(batch.prefill_seqs):
                    if s.sampling.prompt_logprobs is not None:
                        rows.update(range(cu[j], cu[j + 1]))
                    if payload["sample"][j]:
                        rows.add(cu[j + 1] - 1)
                nd = len(payload["decode"]["ids"])
                rows.update(range(cu[-1], cu[-1] + nd))
                payload["logits_rows"] = sorted(rows)
        t_payload = time.monotonic()
        if tp > 1:
            self._ps.tp_broadcast_object(payload)
        logits = (
            self.runner.run_prefill(payload)
            if spec_drafts is not None
            else self.runner.run_batch(payload)
        )
        if self._timing:
            import torch as _t

            if logits.is_cuda:
                _t.cuda.synchronize()
            t_fwd = time.monotonic()
            kind = "P" if batch.prefill_seqs else "D"
            acc = self._tacc.setdefault(kind, [0.0, 0.0, 0.0, 0.0, 0])
            acc[0] += t_sched - step_t0
            acc[1] += t_payload - t_sched
            acc[2] += t_fwd - t_payload
            acc[4] += 1
            self._t_fwd_mark = t_fwd
        for seq, chunk in zip(batch.prefill_seqs, batch.prefill_chunks):
            seq.num_computed_tokens = (
                seq.num_computed_tokens or seq.num_cached_tokens
            ) + chunk
            self.num_prefilled_tokens += chunk
        if (
            self._async_decode
            and self.proposer is None  # spec engines keep the sync path
            and spec_drafts is None
            and not batch.prefill_seqs
            and batch.decode_seqs
            and tp == 1
            and payload.get("logits_rows") is None
            and all(self._seq_async_ok(s) for s in batch.decode_seqs)
        ):
            # enter the pipelined-decode chain: sample WITHOUT a host
            # sync and defer emission to the next step() call (which
            # launches step N+1 first, then processes these tokens
            # while the GPU runs)
            self._stash_pending(batch.decode_seqs, logits)
            elapsed_step = time.monotonic() - step_t0
            self.step_time_sum += elapsed_step
            self.scheduler._step_ema_s = (
                0.9 * self.scheduler._step_ema_s + 0.1 * elapsed_step
            )
            self.num_steps += 1
            return pre
        if spec_drafts is not None:
            outputs = self._finish_spec_step(batch, payload, logits,
                                             spec_drafts)
        else:
            # logits rows: completing prefill chunks first, then decode rows
            sample_seqs = [
                s for s, smp in zip(batch.prefill_seqs, payload["sample"])
                if smp
            ] + batch.decode_seqs
            logits_f = logits.float()
            if payload.get("logits_rows") is not None:
                logits_f = self._consume_prompt_logprobs(
                    batch, payload, logits_f, chunk_starts
                )
            if not sample_seqs:
                return pre
            outputs = self._sample_and_emit(sample_seqs, logits_f)
        if self._timing and hasattr(self, "_t_fwd_mark"):
            kind = "P" if batch.prefill_seqs else "D"
            self._tacc[kind][3] += time.monotonic() - self._t_fwd_mark
            n = self._tacc[kind][4]
            if n % 50 == 0:
                a = self._tacc[kind]
                print(
                    f"[timing {kind}] n={n} sched={a[0]/n*1e3:.2f}ms "
                    f"payload={a[1]/n*1e3:.2f}ms fwd={a[2]/n*1e3:.2f}ms "
                    f"sample+book={a[3]/n*1e3:.2f}ms", flush=True,
                )
        elapsed_step = time.monotonic() - step_t0
        self.step_time_sum += elapsed_step
        # feed the scheduler's adaptive admission window (EMA)
        self.scheduler._step_ema_s = (
            0.9 * self.scheduler._step_ema_s + 0.1 * elapsed_step
        )
        self.num_steps += 1
        return pre + outputs

    # ------------------------------------------------- sampling / emission
    @staticmethod
    def _observe(hist, buckets, value) -> None:
        for i, b in enumerate(buckets):
            if value <= b:
                hist[i] += 1
                return
        hist[-1] += 1

    def _finish_seq(self, seq: Sequence) -> None:
        if self.proposer is not None:
            self.proposer.release(seq)
        seq.finish_time = time.monotonic()
        self.scheduler.finish(seq)
        self.num_finished += 1
        if seq.ttft is not None:
            self.ttft_sum += seq.ttft
            self._observe(self.ttft_hist, self.ttft_buckets, seq.ttft)
        e2e = seq.finish_time - seq.arrival_time
        self.e2e_latency_sum += e2e
        self._observe(self.e2e_hist, self.e2e_buckets, e2e)
        del self.seqs[seq.seq_id]

    def _emit_tokens(self, seq: Sequence, toks) -> RequestOutput:
        """Append this step's tokens (one for regular decode, up to k+1
        for an accepted speculative draft) with per-token stop checks."""
        guided = seq.sampling.guided
        new = []
        for tok in toks:
            now = time.monotonic()
            if seq.last_token_time is not None:
                # inter-token latency (vLLM time_per_output_token)
                self._observe(self.tpot_hist, self.tpot_buckets,
                              now - seq.last_token_time)
            seq.last_token_time = now
            seq.append_token(int(tok))
            new.append(int(tok))
            self.num_generated_tokens += 1
            if guided is not None:
                try:
                    guided.advance_token(int(tok))
                except ValueError:
                    # grammar blow-up: finish THIS request gracefully
                    seq.finish_reason = seq.finish_reason or "stop"
                    self._finish_seq(seq)
                    break
            if seq.check_stop():
                self._finish_seq(seq)
                break
            if guided is not None and guided.is_terminal():
                seq.finish_reason = seq.finish_reason or "stop"
                self._finish_seq(seq)
                break
        return RequestOutput(seq, new_token_ids=new)

    # ------------------------------------------------- pipelined decode
    @staticmethod
    def _seq_async_ok(s: Sequence) -> bool:
        """Pipeline-eligible: nothing about next-step sampling may depend
        on THIS step's token value on the host (guided masks, penalties
        over emitted tokens) and nothing extra is read back (logprobs)."""
        sp = s.sampling
        return (
            sp.guided is None
            and not sp.logit_bias
            and sp.logprobs is None
            and sp.prompt_logprobs is None
            and sp.repetition_penalty == 1.0
            and sp.presence_penalty == 0.0
            and sp.frequency_penalty == 0.0
            and s.lora_name is None
            # seeded draws key on len(output) — one short pre-drain
            and (sp.temperature == 0 or sp.seed is None)
        )

    def _stash_pending(self, seqs, logits) -> None:
        """Sample on-device and record the in-flight step: the sampled
        ids tensor feeds the next launch device-to-device; the pinned
        host copy (awaited via the event at drain time) feeds emission."""
        logits_f = logits.float()
        sampled = self.sampler.sample(logits_f, seqs)
        if sampled.is_cuda:
            host = torch.empty(
                sampled.shape, dtype=sampled.dtype, pin_memory=True
            )
            host.copy_(sampled, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()
        else:
            host, ev = sampled, None
        self._pending = {
            "seqs": list(seqs), "sampled": sampled, "host": host, "event": ev,
        }

    def _drain_pending(self) -> List[RequestOutput]:
        """Emit the in-flight step's tokens (stop checks, block frees,
        stats). Safe against seqs that finished since the stash: their
        late token is discarded."""
        p = self._pending
        if p is None:
            return []
        self._pending = None
        if p["event"] is not None:
            p["event"].synchronize()
        toks = p["host"].tolist()
        return [
            self._emit_tokens(s, [t])
            for s, t in zip(p["seqs"], toks)
            if not s.is_finished()
        ]

    def _try_continue_async(self):
        """Continue the pipelined pure-decode chain: launch step N+1
        from step N's ON-DEVICE sampled ids, then overlap N's host-side
        emission with the GPU. Returns N's outputs, or None when the
        chain must break to the sync path (admission due, composition
        changed, block pressure, or every seq deterministically
        finishes). Deterministic finishers (in-flight token completes
        max_tokens) are dropped from the continued batch up front; data-
        dependent stops (eos/stop tokens) cost one wasted decode row and
        break the chain at the next composition check."""
        p = self._pending
        sch = self.scheduler
        if not self._async_decode or not sch.decode_only_next():
            return None
        if len(sch.running) != len(p["seqs"]) or any(
            a is not b for a, b in zip(sch.running, p["seqs"])
        ):
            return None
        bm = self.block_manager
        max_len = self.cfg.scheduler.max_model_len
        keep: List[Sequence] = []
        keep_rows: List[int] = []
        for i, s in enumerate(p["seqs"]):
            out_after = s.num_tokens + 1 - s.num_prompt_tokens
            if (
                out_after >= s.sampling.max_tokens
                or s.num_tokens + 1 >= max_len
                or not self._seq_async_ok(s)
            ):
                continue
            keep.append(s)
            keep_rows.append(i)
        if not keep:
            return None
        need = sum(bm.extra_blocks_for(s, s.num_tokens) for s in keep)
        if bm.num_free() < need:
            return None  # sync path can preempt
        step_t0 = time.monotonic()
        for s in keep:
            bm.append_slots_upto(s, s.num_tokens)
        payload = {
            "kind": "decode",
            "ids": None,  # fed on-device below
            "positions": [s.num_tokens for s in keep],
            "slots": [bm.slot_for(s, s.num_tokens) for s in keep],
            "lens": [s.num_tokens + 1 for s in keep],
            "bt": [list(s.block_ids) for s in keep],
            "lora_names": [None] * len(keep),
        }
        sampled = p["sampled"]
        if len(keep) != len(p["seqs"]):
            idx = torch.tensor(
                keep_rows, dtype=torch.long, device=sampled.device
            )
            sampled = sampled.index_select(0, idx)
        logits = self.runner.run_decode_device_ids(payload, sampled)
        # N+1 is on the GPU; process N on the host in the shadow
        outputs = self._drain_pending()
        self._stash_pending(keep, logits)
        self.num_async_steps += 1
        elapsed_step = time.monotonic() - step_t0
        self.step_time_sum += elapsed_step
        self.scheduler._step_ema_s = (
            0.9 * self.scheduler._step_ema_s + 0.1 * elapsed_step
        )
        self.num_steps += 1
        return outputs

    def _sample_and_emit(self, sample_seqs, logits_f) -> List[RequestOutput]:
        """Regular sampling path: logits_f row i belongs to sample_seqs[i].
        Guided sequences get their grammar mask applied first; a sequence
        whose grammar admits no token is finished without emitting."""
        outputs: List[RequestOutput] = []
        keep = []
        for i, s in enumerate(sample_seqs):
            if s.sampling.logit_bias:
                V = logits_f.shape[1]
                # drop out-of-vocab ids: a bad request must not raise a
                # device-side index fault in the engine loop
                ids = [t for t in s.sampling.logit_bias if 0 <= t < V]
                if ids:
                    logits_f[i, ids] += torch.tensor(
                        [s.sampling.logit_bias[t] for t in ids],
                        dtype=logits_f.dtype, device=logits_f.device,
                    )
            g = s.sampling.guided
            if g is not None:
                try:
                    allowed = g.allowed_mask(logits_f.device)
                except ValueError:
                    # grammar blow-up (state-explosion cap, expansion
                    # bound): fail THIS request, not the engine loop
                    allowed = None
                if allowed is None:
                    s.finish_reason = s.finish_reason or "stop"
                    self._finish_seq(s)
                    outputs.append(RequestOutput(s, new_token_ids=[]))
                    continue
                logits_f[i].masked_fill_(~allowed, float("-inf"))
            keep.append(i)
        if not keep:
            return outputs
        if len(keep) < len(sample_seqs):
            sample_seqs = [sample_seqs[i] for i in keep]
            logits_f = logits_f[keep]
        next_tokens = self.sampler.sample(logits_f, sample_seqs).tolist()
        # optional logprobs for requests that asked
        lp_rows = [
            i for i, s in enumerate(sample_seqs)
            if s.sampling.logprobs is not None
        ]
        if lp_rows:
            lp = torch.log_softmax(logits_f[lp_rows], dim=-1)
            for j, i in enumerate(lp_rows):
                s = sample_seqs[i]
                k = max(int(s.sampling.logprobs), 0)
                entry_top = {}
                if k > 0:
                    vals, idx = lp[j].topk(k)
                    entry_top = {
                        int(t): float(v) for t, v in zip(idx, vals)
                    }
                s.logprobs.append(
                    (float(lp[j, next_tokens[i]]), entry_top)
                )
        outputs.extend(
            self._emit_tokens(seq, [tok])
            for seq, tok in zip(sample_seqs, next_tokens)
        )
        return outputs

    def _consume_prompt_logprobs(self, batch, payload, logits_f,
                                 chunk_starts):
        """prompt_logprobs (vLLM parity): the payload carried explicit
        logits_rows covering every prompt position of requesting
        sequences; fill seq.prompt_logprobs (index 0 and prefix-cached
        positions are None) and return the standard sampling-row subset
        in the engine's expected order. log_softmax runs in row slices
        so a full 8k-token admission never materializes a [T, V] fp32."""
        rows = payload["logits_rows"]
        pos_of = {r: i for i, r in enumerate(rows)}
        cu = payload["cu"]
        for j, seq in enumerate(batch.prefill_seqs):
            k = seq.sampling.prompt_logprobs
            if k is None:
                continue
            C = chunk_starts[j]
            n = batch.prefill_chunks[j]
            if not seq.prompt_logprobs:
                # position 0 has no logprob; prefix-cached positions are
                # skipped (their logits were never computed)
                seq.prompt_logprobs.extend(
                    [None] * min(C + 1, seq.num_prompt_tokens)
                )
            toks = seq.all_token_ids
            idx = [pos_of[r] for r in range(cu[j], cu[j + 1])]
            for lo in range(0, n, 256):
                hi = min(lo + 256, n)
                sub = torch.log_softmax(logits_f[idx[lo:hi]], dim=-1)
                for i in range(lo, hi):
                    pos_next = C + i + 1
                    if pos_next >= seq.num_prompt_tokens:
                        break
                    row = sub[i - lo]
                    entry_top = {}
                    if k > 0:
                        vals, tix = row.topk(k)
                        entry_top = {
                            int(t): float(v) for t, v in zip(tix, vals)
                        }
                    seq.prompt_logprobs.append(
                        (float(row[toks[pos_next]]), entry_top)
                    )
        pf_rows = [
            cu[j + 1] - 1 for j, smp in enumerate(payload["sample"]) if smp
        ]
        nd = len(payload["decode"]["ids"])
        sel = [pos_of[r] for r in pf_rows] + [
            pos_of[cu[-1] + i] for i in range(nd)
        ]
        return logits_f[sel]

    # ------------------------------------------------ speculative decoding
    def _propose_drafts(self, decode_seqs):
        """Drafts per decode sequence (ngram lookup or the draft model);
        a draft is dropped when the block pool cannot cover its tail.
        None = nothing to speculate (step falls back to the fast
        pure-decode / hipGraph path)."""
        bm = self.block_manager
        drafts = [list(d) for d in self.proposer.propose_all(decode_seqs)]
        any_d = False
        for s, d in zip(decode_seqs, drafts):
            if d:
                last_pos = s.num_tokens - 1 + len(d)
                if bm.extra_blocks_for(s, last_pos) > bm.num_free():
                    d.clear()
                else:
                    bm.append_slots_upto(s, last_pos)
            any_d = any_d or bool(d)
        return drafts if any_d else None

    def _finish_spec_step(self, batch, payload, logits, drafts):
        """Greedy draft acceptance. Logits row layout (see
        build_spec_payload): [completing-prefill rows | per decode seq,
        1 + k_i rows]. Draft-less decode seqs go through the regular
        sampler; drafted seqs accept the longest matching prefix plus the
        corrected token — token-exact with non-speculative greedy."""
        logits_f = logits.float()
        n_prefill = len(batch.prefill_seqs)
        pf_flags = payload["sample"][:n_prefill]
        samp_seqs = [s for s, smp in zip(batch.prefill_seqs, pf_flags) if smp]
        samp_rows = list(range(len(samp_seqs)))
        spec = []  # (seq, draft, first logits row of its span)
        r = len(samp_seqs)
        for s, d in zip(batch.decode_seqs, drafts):
            if d:
                spec.append((s, d, r))
            else:
                samp_rows.append(r)
                samp_seqs.append(s)
            r += 1 + len(d)
        outputs: List[RequestOutput] = []
        if samp_seqs:
            outputs.extend(
                self._sample_and_emit(samp_seqs, logits_f[samp_rows])
            )
        if spec:
            greedy = logits_f.argmax(dim=-1).tolist()
            for s, d, r0 in spec:
                k = len(d)
                g = greedy[r0: r0 + k + 1]
                m = 0
                while m < k and d[m] == g[m]:
                    m += 1
                self.num_spec_draft_tokens += k
                self.num_spec_accepted_tokens += m
                outputs.append(self._emit_tokens(s, d[:m] + [g[m]]))
                if not s.is_finished():
                    # draft-model proposers roll their KV back to the
                    # accepted prefix (no-op for ngram)
                    self.proposer.commit(s, m)
        return outputs

    # --------------------------------------------------------- TP workers
    def worker_loop(self) -> None:
        """TP rank > 0: execute broadcast batches until the driver stops."""
        assert not self.is_driver
        while True:
            payload = self._ps.tp_broadcast_object(None)
            if payload is None or payload.get("kind") == "stop":
                return
            if payload["kind"] == "add_lora":
                self._register_lora(payload["name"], payload["rank"],
                                    payload["alpha"], payload["seed"])
            elif payload["kind"] == "add_lora_path":
                from fusioninfer_amd.lora import LoRAAdapter

                self._lora_registry().add(LoRAAdapter.from_safetensors(
                    payload["name"], payload["path"], self.cfg.model,
                    device=self.device,
                ))
            elif payload["kind"] == "pd_send":
                # ship this rank's KV shard for a PD export (same block
                # ids on every TP rank)
                self.kv_connector.send_kv(
                    self.runner.kv_caches, payload["block_ids"],
                    payload["prompt_len"], payload["first_token"],
                    payload["tag"],
                )
            elif payload["kind"] == "pd_recv":
                self.kv_connector.recv_kv(
                    self.runner.kv_caches,
                    lambda n: payload["block_ids"],
                )
            elif payload["kind"] == "mixed":
                self.runner.run_batch(payload)
            elif payload["kind"] == "prefill":
                self.runner.run_prefill(payload)
            else:
                self.runner.run_decode(payload)

    def stop_workers(self) -> None:
        exec_world = (self.cfg.parallel.tensor_parallel_size
                      * self.cfg.parallel.pipeline_parallel_size)
        if self.is_driver and exec_world > 1:
            self._ps.tp_broadcast_object({"kind": "stop"})

    # ------------------------------------------------------------ sync API
    def generate(
        self,
        prompts: List[List[int]],
        sampling: Optional[SamplingParams] = None,
    ) -> List[RequestOutput]:
        ids = [self.add_request(p, sampling) for p in prompts]
        done: Dict[str, RequestOutput] = {}
        while self.has_unfinished() and len(done) < len(ids):
            for out in self.step():
                if out.finished:
                    done[out.request_id] = out
        return [done[i] for i in ids]
