"""Request/sequence state for the continuous-batching engine."""

from __future__ import annotations

import enum
import time
from typing import List, Optional


class SeqStatus(enum.Enum):
    WAITING = "waiting"
    RUNNING = "running"
    PREEMPTED = "preempted"
    SWAPPED = "swapped"      # KV saved to CPU swap space; resume skips recompute
    FINISHED = "finished"


class SamplingParams:
    def __init__(
        self,
        max_tokens: int = 128,
        min_tokens: int = 0,
        temperature: float = 0.0,
        top_p: float = 1.0,
        top_k: int = 0,
        min_p: float = 0.0,
        repetition_penalty: float = 1.0,
        presence_penalty: float = 0.0,
        frequency_penalty: float = 0.0,
        seed: Optional[int] = None,
        logprobs: Optional[int] = None,
        prompt_logprobs: Optional[int] = None,
        ignore_eos: bool = True,
        stop_token_ids: Optional[List[int]] = None,
        guided=None,
        logit_bias=None,
    ):
        self.max_tokens = max_tokens
        self.min_tokens = min_tokens  # suppress stop tokens until reached
        self.temperature = temperature
        self.top_p = top_p
        self.top_k = top_k
        self.min_p = min_p  # drop tokens with p < min_p * max(p)
        self.repetition_penalty = repetition_penalty
        self.presence_penalty = presence_penalty
        self.frequency_penalty = frequency_penalty
        # normalize into torch.Generator.manual_seed's accepted range —
        # a huge JSON seed must not raise inside the engine loop
        self.seed = None if seed is None else int(seed) % (2**63)
        self.logprobs = logprobs
        self.prompt_logprobs = prompt_logprobs
        self.ignore_eos = ignore_eos
        self.stop_token_ids = stop_token_ids or []
        # guided decoding: a guided.GuidedDecoder enforcing a grammar via
        # logits masks (None = unconstrained)
        self.guided = guided
        # OpenAI logit_bias: {token_id: additive bias in [-100, 100]}
        self.logit_bias = {int(k): float(v)
                           for k, v in (logit_bias or {}).items()}


class Sequence:
    """One request = one sequence (no beam/parallel sampling in v1)."""

    def __init__(self, seq_id: str, prompt_token_ids: List[int],
                 sampling: SamplingParams):
        self.seq_id = seq_id
        self.prompt_token_ids = list(prompt_token_ids)
        self.output_token_ids: List[int] = []
        self.sampling = sampling
        self.status = SeqStatus.WAITING
        self.block_ids: List[int] = []
        self.arrival_time = time.monotonic()
        self.first_token_time: Optional[float] = None
        self.last_token_time: Optional[float] = None
        self.finish_time: Optional[float] = None
        # number of prompt tokens whose KV was satisfied by prefix cache
        self.num_cached_tokens = 0
        # prompt tokens processed so far (chunked prefill); set to
        # num_cached_tokens at admission, advances per prefill chunk
        self.num_computed_tokens = 0
        # PD producer: keep cache blocks alive after finish for KV export
        self.hold_blocks = False
        # OpenAI finish_reason: "length" (max_tokens) or "stop" (eos /
        # stop token / grammar terminal); None while running
        self.finish_reason = None
        self.swap_num_blocks = 0
        # LoRA adapter name (None = base model)
        self.lora_name = None
        # scheduling priority (lower = more urgent; "priority" policy)
        self.priority = 0
        # per-output-token logprob entries when sampling.logprobs is set:
        # [(logprob_of_sampled, {token_id: logprob, ...top-k}), ...]
        self.logprobs = []
        # per-PROMPT-position entries when sampling.prompt_logprobs is
        # set: index 0 and prefix-cached positions are None
        self.prompt_logprobs = []

    @property
    def num_prompt_tokens(self) -> int:
        return len(self.prompt_token_ids)

    @property
    def num_tokens(self) -> int:
        return len(self.prompt_token_ids) + len(self.output_token_ids)

    @property
    def all_token_ids(self) -> List[int]:
        return self.prompt_token_ids + self.output_token_ids

    def append_token(self, token_id: int) -> None:
        if self.first_token_time is None:
            self.first_token_time = time.monotonic()
        self.output_token_ids.append(token_id)

    def is_finished(self) -> bool:
        return self.status == SeqStatus.FINISHED

    def check_stop(self) -> bool:
        s = self.sampling
        if (
            not s.ignore_eos
            and self.output_token_ids
            and len(self.output_token_ids) >= s.min_tokens
            and self.output_token_ids[-1] in s.stop_token_ids
        ):
            self.finish_reason = "stop"
            return True
        if len(self.output_token_ids) >= s.max_tokens:
            self.finish_reason = self.finish_reason or "length"
            return True
        return False

    @property
    def ttft(self) -> Optional[float]:
        if self.first_token_time is None:
            return None
        return self.first_token_time - self.arrival_time
