"""Continuous-batching scheduler (token-level, unified mixed batches).

Each step schedules ONE mixed batch: a decode token for every
fully-prefilled running sequence PLUS prompt chunks (chunked prefill)
under a shared max_num_batched_tokens budget — decode-ready sequences
never stall behind prefills. New-prompt admission uses hysteresis
(prefill_admission_tokens): prompts are held until enough tokens queue
up, so steady-state steps stay pure-decode and take the hipGraph path.
Preemption-by-recompute frees the newest sequence when the cache runs
out of blocks mid-decode.

Capability parity: the "continuous-batching scheduler + paged KV cache"
the reference delegates to vLLM and whose kv-util / queue-depth metrics
its EPP scorers consume (reference pkg/router/
strategy.go:70-98).
"""

from __future__ import annotations

import dataclasses
import time
from collections import deque
from typing import Deque, List

from fusioninfer_amd.config import SchedulerConfig
from fusioninfer_amd.engine.block_manager import BlockManager
from fusioninfer_amd.engine.sequence import Sequence, SeqStatus


@dataclasses.dataclass
class ScheduledBatch:
    prefill_seqs: List[Sequence]
    prefill_chunks: List[int]          # tokens scheduled per prefill seq
    decode_seqs: List[Sequence]
    preempted: List[Sequence]
    # swapped-out sequences re-admitted this step: the engine restores
    # their KV from CPU swap into the freshly allocated blocks BEFORE the
    # forward runs; they rejoin decode/prefill next step
    swap_in: List[Sequence] = dataclasses.field(default_factory=list)

    @property
    def is_empty(self) -> bool:
        return not self.prefill_seqs and not self.decode_seqs

    @property
    def is_prefill(self) -> bool:
        return bool(self.prefill_seqs)


class Scheduler:
    #: adaptive admission window = this many steps of EMA step time
    ADMISSION_STEPS = 4.0

    def __init__(self, cfg: SchedulerConfig, block_manager: BlockManager):
        self.cfg = cfg
        self.bm = block_manager
        self._step_ema_s = 0.05  # updated by the engine after each step
        self.waiting: Deque[Sequence] = deque()
        self.running: List[Sequence] = []
        # engine-provided: callable(seq) -> bool, True = KV parked in CPU
        # swap (preemption keeps computed state); False/None = recompute
        self.swap_out_fn = None

    # ----------------------------------------------------------- queue ops
    def add(self, seq: Sequence) -> None:
        self.waiting.append(seq)

    @property
    def num_waiting(self) -> int:
        return len(self.waiting)

    @property
    def num_running(self) -> int:
        return len(self.running)

    def has_work(self) -> bool:
        return bool(self.waiting or self.running)

    # ----------------------------------------------------------- schedule
    def decode_only_next(self) -> bool:
        """Read-only preview: would schedule() produce a full-composition
        pure-decode batch (no admissions, swap-ins, prefills, or
        preemptions by composition)? The engine's pipelined decode path
        uses this to continue a graph-replay chain WITHOUT invoking
        schedule()'s side effects (slot appends). Mirrors the admission
        hysteresis below exactly; conservative on swapped heads."""
        if not self.running:
            return False
        for s in self.running:
            if (s.status != SeqStatus.RUNNING
                    or s.num_computed_tokens < s.num_prompt_tokens):
                return False
        if len(self.running) >= self.cfg.max_num_seqs:
            # full: neither the swap-in loop nor the admission loop can
            # run (both gate on len(running) < max_num_seqs)
            return True
        if not self.waiting:
            return True
        if self.waiting[0].status == SeqStatus.SWAPPED:
            return False
        budget = self.cfg.max_num_batched_tokens
        threshold = min(self.cfg.prefill_admission_tokens, budget)
        if sum(s.num_prompt_tokens for s in self.waiting) >= threshold:
            return False
        if self.cfg.prefill_admission_ms is not None:
            window_s = self.cfg.prefill_admission_ms / 1000.0
        else:
            window_s = min(max(
                self.ADMISSION_STEPS * self._step_ema_s, 0.05), 0.25)
        return time.monotonic() - self.waiting[0].arrival_time <= window_s

    def schedule(self) -> ScheduledBatch:
        """One MIXED batch per step (vLLM-v1-style unified scheduling):
        decode tokens for every fully-prefilled running sequence PLUS
        prompt chunks (continuing chunked prefills, then new admissions)
        up to the shared max_num_batched_tokens budget. Decode-ready
        sequences never stall behind prefills."""
        preempted: List[Sequence] = []
        budget = self.cfg.max_num_batched_tokens
        if self.cfg.policy == "priority" and len(self.waiting) > 1:
            self.waiting = deque(sorted(
                self.waiting, key=lambda s: (s.priority, s.arrival_time)
            ))

        # decode: every fully-prefilled running sequence, one token each
        decode: List[Sequence] = []
        for seq in list(self.running):
            if seq.status != SeqStatus.RUNNING:
                continue  # preempted earlier in this same pass
            if seq.num_computed_tokens < seq.num_prompt_tokens:
                continue  # still prefilling
            if not self.bm.can_append_slot(seq):
                victim = self._preempt_newest()
                preempted.append(victim)
                if victim is seq:
                    continue
            self.bm.append_slot(seq)
            decode.append(seq)
            budget -= 1

        # prefill chunks: continue partially-prefilled running seqs first
        prefill: List[Sequence] = []
        chunks: List[int] = []
        for seq in self.running:
            if seq.status != SeqStatus.RUNNING:
                continue
            remaining = seq.num_prompt_tokens - seq.num_computed_tokens
            if remaining <= 0 or budget <= 0:
                continue
            chunk = min(remaining, budget)
            prefill.append(seq)
            chunks.append(chunk)
            budget -= chunk
        # resume swapped-out sequences first: they are at the queue head,
        # carry computed KV in host swap, and need neither token budget nor
        # the admission hysteresis — just free blocks (+1 headroom so the
        # next decode append does not instantly re-preempt them)
        swap_in: List[Sequence] = []
        while (
            self.waiting
            and self.waiting[0].status == SeqStatus.SWAPPED
            and self.waiting[0] not in preempted
            and len(self.running) < self.cfg.max_num_seqs
        ):
            seq = self.waiting[0]
            if self.bm.num_free() < seq.swap_num_blocks + 1:
                break
            self.waiting.popleft()
            self.bm.allocate_raw(seq, seq.swap_num_blocks)
            seq.status = SeqStatus.RUNNING
            swap_in.append(seq)
            self.running.append(seq)

        # then admit waiting prompts — with hysteresis: when decodes are
        # running, hold admissions until enough prompt tokens queue up so
        # most steps stay pure-decode (hipGraph path)
        waiting_tokens = sum(s.num_prompt_tokens for s in self.waiting)
        threshold = min(self.cfg.prefill_admission_tokens, budget)
        if not decode or prefill:
            # idle decode path, or this step is mixed anyway: admit freely
            admit = True
        else:
            admit = waiting_tokens >= threshold
            # aging escape: at low load the batching threshold may never
            # be reached — never hold a prompt longer than the admission
            # window (explicit prefill_admission_ms, or adaptive:
            # ADMISSION_STEPS x the EMA step time, clamped [50, 250] ms)
            if not admit and self.waiting:
                if self.cfg.prefill_admission_ms is not None:
                    window_s = self.cfg.prefill_admission_ms / 1000.0
                else:
                    window_s = min(max(
                        self.ADMISSION_STEPS * self._step_ema_s, 0.05), 0.25)
                admit = (
                    time.monotonic() - self.waiting[0].arrival_time
                    > window_s
                )
        while (
            admit
            and budget > 0
            and self.waiting
            and len(self.running) < self.cfg.max_num_seqs
        ):
            seq = self.waiting[0]
            if seq.status == SeqStatus.SWAPPED:
                # the head must resume through the swap-in path above (it
                # lacked blocks this step); admitting it as a fresh prompt
                # would discard its computed state and leak its swap staging
                break
            if not self.bm.can_allocate(seq.num_prompt_tokens):
                break
            self.waiting.popleft()
            self.bm.allocate(seq)
            seq.status = SeqStatus.RUNNING
            seq.num_computed_tokens = seq.num_cached_tokens
            chunk = min(seq.num_prompt_tokens - seq.num_computed_tokens, budget)
            prefill.append(seq)
            chunks.append(chunk)
            budget -= chunk
            self.running.append(seq)
        return ScheduledBatch(prefill, chunks, decode, preempted, swap_in)

    def _preempt_newest(self) -> Sequence:
        if self.cfg.policy == "priority":
            # lowest-priority (then newest) sequence yields first
            victim = max(self.running,
                         key=lambda s: (s.priority, s.arrival_time))
            self.running.remove(victim)
        else:
            victim = self.running.pop()  # newest
        if self.swap_out_fn is not None and self.swap_out_fn(victim):
            # KV parked in CPU swap: keep all computed state; resume
            # restores the blocks instead of re-prefilling
            victim.swap_num_blocks = len(victim.block_ids)
            self.bm.free(victim)
            victim.status = SeqStatus.SWAPPED
            self.waiting.appendleft(victim)
            return victim
        self.bm.free(victim)
        victim.status = SeqStatus.PREEMPTED
        # recompute: generated tokens become part of the prompt
        victim.prompt_token_ids.extend(victim.output_token_ids)
        victim.output_token_ids = []
        victim.num_computed_tokens = 0
        victim.num_cached_tokens = 0
        self.waiting.appendleft(victim)
        return victim

    # ----------------------------------------------------------- finish
    def finish(self, seq: Sequence) -> None:
        seq.status = SeqStatus.FINISHED
        if not seq.hold_blocks:
            self.bm.free(seq)
        self.running.remove(seq)
