"""Speculative decoding: n-gram (prompt-lookup) drafting + verification.

Capability parity with the reference engines' `--speculative-config
{"method": "ngram", ...}` surface (vLLM prompt-lookup decoding): each
decode step, a draft of up to `num_speculative_tokens` tokens is proposed
by matching the sequence's trailing n-gram against its earlier context
(prompt + generated). The engine verifies the draft in ONE forward by
expressing the sequence as a (1 + k)-token chunk through the paged
context-attention prefill path — the same kernel chunked prefill uses —
so the feature needs no extra model, no extra KV cache, and works
unchanged under TP and PP (the verify payload is a plain "prefill"
payload the driver already broadcasts).

Acceptance is greedy (drafting is only enabled for temperature==0
requests without penalties/logprobs/guided grammars): draft token j is
accepted while it equals the argmax of the verification logits at row
j-1; the first mismatch row contributes the corrected token, so every
spec step emits between 1 and k+1 tokens and the output is TOKEN-EXACT
with non-speculative greedy decoding.

KV bookkeeping: verification writes draft KV into the sequence's own
slots (slot = block[pos//bs]*bs + pos%bs is deterministic per position),
so rejected positions are simply re-written when the sequence reaches
them again — no rollback pass is needed. Blocks are pre-extended for the
draft tail and trimmed when the pool is tight.
"""

from __future__ import annotations

import dataclasses
from typing import List, Optional

import numpy as np

from fusioninfer_amd.engine.sequence import Sequence


@dataclasses.dataclass
class SpeculativeConfig:
    """vLLM --speculative-config parity (methods "ngram" and
    "draft_model")."""

    method: str = "ngram"
    num_speculative_tokens: int = 4
    prompt_lookup_max: int = 4   # longest trailing n-gram to match
    prompt_lookup_min: int = 2   # shortest n-gram worth trusting
    # draft_model method: registry name of the small proposer model
    # (e.g. "Qwen3-0.6B" drafting for "Qwen3-8B"; vocab must match)
    model: Optional[str] = None
    # draft KV pool: cache blocks for the draft model (None = target/4)
    draft_gpu_blocks: Optional[int] = None
    # vLLM speculative_disable_by_batch_size: above this many concurrent
    # decode sequences, skip drafting — a spec step runs the WHOLE decode
    # batch through the eager multi-token verify path (no hipGraph, no
    # decode kernel), which loses at high batch (measured: 21.4 vs 44
    # req/s at concurrency 256 with accept_rate 0.22). Speculation is a
    # LATENCY feature for small batches. 0 disables the guard.
    disable_by_batch_size: int = 32


def seq_is_draftable(seq: Sequence) -> bool:
    """Drafting needs greedy argmax acceptance: temperature 0, no
    penalties (they perturb the argmax), no per-token logprobs (rows for
    rejected drafts would be meaningless), no grammar mask."""
    sp = seq.sampling
    return (
        sp.temperature == 0.0
        and sp.repetition_penalty == 1.0
        and sp.presence_penalty == 0.0
        and sp.frequency_penalty == 0.0
        and sp.logprobs is None
        and getattr(sp, "guided", None) is None
        and not getattr(sp, "logit_bias", None)
    )


def propose_ngram(
    token_ids: List[int],
    max_n: int,
    min_n: int,
    k: int,
) -> List[int]:
    """Prompt-lookup proposal: find the most recent earlier occurrence of
    the trailing n-gram (longest n first) and return the k tokens that
    followed it. Empty when nothing matches."""
    L = len(token_ids)
    if L < min_n + 1 or k <= 0:
        return []
    arr = np.asarray(token_ids, dtype=np.int64)
    for n in range(min(max_n, L - 1), min_n - 1, -1):
        pattern = arr[L - n:]
        windows = np.lib.stride_tricks.sliding_window_view(arr[: L - 1], n)
        hits = np.nonzero((windows == pattern).all(axis=1))[0]
        if hits.size == 0:
            continue
        start = int(hits[-1]) + n          # token right after the match
        draft = arr[start: start + k]
        if draft.size == 0:
            continue
        return draft.tolist()
    return []


def _seq_budget(seq: Sequence, k: int) -> int:
    # never draft past max_tokens: the tail would be dead work
    return min(k, seq.sampling.max_tokens - len(seq.output_token_ids) - 1)


class NgramProposer:
    def __init__(self, cfg: SpeculativeConfig):
        self.cfg = cfg

    def propose(self, seq: Sequence) -> List[int]:
        if not seq_is_draftable(seq):
            return []
        k = _seq_budget(seq, self.cfg.num_speculative_tokens)
        if k <= 0:
            return []
        return propose_ngram(
            seq.all_token_ids,
            self.cfg.prompt_lookup_max,
            self.cfg.prompt_lookup_min,
            k,
        )

    def propose_all(self, seqs) -> List[List[int]]:
        return [self.propose(s) for s in seqs]

    # draft-model lifecycle hooks are no-ops for ngram
    def commit(self, seq: Sequence, accepted: int) -> None:
        pass

    def release(self, seq: Sequence) -> None:
        pass


class DraftModelProposer:
    """Classic draft-model speculation: a small model of the same
    tokenizer family proposes k greedy tokens per step. The draft keeps
    its OWN paged KV (separate ModelRunner + BlockManager) and syncs to
    the target lazily: every proposal round first consumes the gap
    between what the draft has seen and the target's current stream —
    so target prefill, preemption-recompute and non-drafted steps all
    reduce to "catch up on the token stream" with no extra plumbing.

    After acceptance the engine calls commit(seq, m): the draft's KV is
    valid for the stream prefix it consumed, which matches the target
    for L + min(m, k-1) tokens (the rejected proposal's KV slots are
    position-deterministic and get overwritten on the next catch-up).

    v1 scope: single-process engines (TP/PP run the ngram proposer)."""

    def __init__(self, cfg: SpeculativeConfig, target_cfg, device):
        from fusioninfer_amd.config import (
            CacheConfig,
            EngineConfig,
            SchedulerConfig,
        )
        from fusioninfer_amd.engine.block_manager import BlockManager
        from fusioninfer_amd.engine.model_runner import ModelRunner
        from fusioninfer_amd.models.registry import get_model_config

        self.cfg = cfg
        mc = get_model_config(cfg.model)
        assert mc.vocab_size == target_cfg.model.vocab_size, (
            "draft model must share the target's vocabulary"
        )
        blocks = cfg.draft_gpu_blocks
        if blocks is None:
            blocks = max((target_cfg.cache.num_gpu_blocks or 2048) // 4, 64)
        dcfg = EngineConfig(
            model=mc,
            cache=CacheConfig(
                block_size=target_cfg.cache.block_size,
                num_gpu_blocks=blocks,
                enable_prefix_caching=False,
            ),
            scheduler=SchedulerConfig(
                max_num_seqs=target_cfg.scheduler.max_num_seqs,
                max_num_batched_tokens=target_cfg.scheduler.max_num_batched_tokens,
                max_model_len=target_cfg.scheduler.max_model_len,
            ),
            # same seed as the target: random-init dev/test engines then
            # get draft==target weights for a same-arch draft (always
            # accepts), and real checkpoints ignore init anyway
            seed=target_cfg.seed,
            enforce_eager=True,  # draft decode batches are tiny; eager
        )
        self.runner = ModelRunner(dcfg, device)
        self.runner.allocate_kv_caches()
        self.bm = BlockManager(self.runner.num_gpu_blocks,
                               dcfg.cache.block_size)
        # seq_id -> [block_ids, consumed_token_count]
        self._state = {}

    # ------------------------------------------------------------ helpers
    def _ensure_blocks(self, st, upto_position: int) -> bool:
        bs = self.bm.block_size
        need = upto_position // bs + 1
        while len(st[0]) < need:
            if self.bm.num_free() == 0:
                return False
            blk = self.bm._pop_free_block()
            self.bm.ref_count[blk] = 1
            st[0].append(blk)
        return True

    def _slots(self, st, lo: int, hi: int) -> List[int]:
        bs = self.bm.block_size
        blocks = st[0]
        return [blocks[p // bs] * bs + p % bs for p in range(lo, hi)]

    # ------------------------------------------------------------- public
    def propose_all(self, seqs) -> List[List[int]]:
        k = self.cfg.num_speculative_tokens
        live = []   # (idx, seq, st, budget)
        drafts: List[List[int]] = [[] for _ in seqs]
        for i, s in enumerate(seqs):
            if not seq_is_draftable(s):
                continue
            budget = _seq_budget(s, k)
            if budget <= 0:
                continue
            st = self._state.setdefault(s.seq_id, [[], 0])
            L = s.num_tokens
            if st[1] >= L:  # stale KV past the stream (shouldn't happen)
                st[1] = max(L - 1, 0)
            if not self._ensure_blocks(st, L - 1 + budget):
                continue
            live.append((i, s, st, budget))
        if not live:
            return drafts

        dev = self.runner.device
        # round 0: catch-up chunk [consumed, L) per seq -> first proposal
        ids: List[int] = []
        positions: List[int] = []
        slots: List[int] = []
        cu = [0]
        new_lens, total_lens, bt = [], [], []
        for _, s, st, _b in live:
            toks = s.all_token_ids
            C, L = st[1], s.num_tokens
            ids.extend(toks[C:L])
            positions.extend(range(C, L))
            slots.extend(self._slots(st, C, L))
            cu.append(cu[-1] + (L - C))
            new_lens.append(L - C)
            total_lens.append(L)
            bt.append(st[0])
        max_blocks = max(len(b) for b in bt)
        payload = {
            "kind": "prefill",
            "ids": ids,
            "positions": positions,
            "slots": slots,
            "cu": cu,
            "new_lens": new_lens,
            "total_lens": total_lens,
            "bt": [b + [0] * (max_blocks - len(b)) for b in bt],
            "sample": [True] * len(live),
            "lora_names": [None] * len(live),
        }
        logits = self.runner.run_prefill(payload)
        cur = logits.float().argmax(dim=-1).tolist()
        for j, (i, _s, _st, _b) in enumerate(live):
            drafts[i].append(int(cur[j]))

        # rounds 1..k-1: single-token draft decodes for seqs with budget
        for r in range(1, max(b for _, _, _, b in live)):
            active = [j for j, (_, _, _, b) in enumerate(live) if b > r]
            if not active:
                break
            d = {
                "kind": "decode",
                "ids": [drafts[live[j][0]][-1] for j in active],
                "positions": [live[j][1].num_tokens - 1 + r for j in active],
                "slots": [
                    self._slots(live[j][2],
                                live[j][1].num_tokens - 1 + r,
                                live[j][1].num_tokens + r)[0]
                    for j in active
                ],
                "lens": [live[j][1].num_tokens + r for j in active],
                "bt": [list(live[j][2][0]) for j in active],
                "lora_names": [None] * len(active),
            }
            logits = self.runner.run_decode(d)
            nxt = logits.float().argmax(dim=-1).tolist()
            for jj, j in enumerate(active):
                drafts[live[j][0]].append(int(nxt[jj]))

        # the draft has consumed [stream..L-1] + its own proposals[:-1]
        for j, (i, s, st, b) in enumerate(live):
            st[1] = s.num_tokens + len(drafts[i]) - 1
        return drafts

    def commit(self, seq: Sequence, accepted: int) -> None:
        """Engine reports m accepted draft tokens: the draft's consumed
        stream matches the target for L_prev + min(m, k_prop-1) tokens
        (proposal m onward diverged)."""
        st = self._state.get(seq.seq_id)
        if st is None:
            return
        # seq.num_tokens already includes the emitted tokens; the draft
        # consumed up to st[1]; roll back to the longest matching prefix
        st[1] = min(st[1], seq.num_tokens - 1)

    def release(self, seq: Sequence) -> None:
        st = self._state.pop(seq.seq_id, None)
        if st is None:
            return
        for blk in st[0]:
            self.bm.ref_count.pop(blk, None)
            self.bm.free_blocks.append(blk)


def build_proposer(cfg: Optional[SpeculativeConfig], target_cfg=None,
                   device=None):
    if cfg is None:
        return None
    if cfg.method == "ngram":
        return NgramProposer(cfg)
    if cfg.method in ("draft_model", "draft"):
        assert cfg.model, "draft_model method needs a draft model name"
        return DraftModelProposer(cfg, target_cfg, device)
    raise ValueError(f"unknown speculative method {cfg.method!r}")
