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
    """vLLM --speculative-config parity (method "ngram")."""

    method: str = "ngram"
    num_speculative_tokens: int = 4
    prompt_lookup_max: int = 4   # longest trailing n-gram to match
    prompt_lookup_min: int = 2   # shortest n-gram worth trusting
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


class NgramProposer:
    def __init__(self, cfg: SpeculativeConfig):
        assert cfg.method == "ngram", f"unknown speculative method {cfg.method!r}"
        self.cfg = cfg

    def propose(self, seq: Sequence) -> List[int]:
        if not seq_is_draftable(seq):
            return []
        k = self.cfg.num_speculative_tokens
        # never draft past max_tokens: the tail would be dead work
        k = min(k, seq.sampling.max_tokens - len(seq.output_token_ids) - 1)
        if k <= 0:
            return []
        return propose_ngram(
            seq.all_token_ids,
            self.cfg.prompt_lookup_max,
            self.cfg.prompt_lookup_min,
            k,
        )


def build_proposer(cfg: Optional[SpeculativeConfig]):
    if cfg is None:
        return None
    return NgramProposer(cfg)
