"""Speculative decoding (ngram prompt-lookup): proposer unit tests and
token-exactness of the engine's verify/accept path vs plain greedy decode.

Parity target: the reference engines' --speculative-config
{"method": "ngram", "num_speculative_tokens": k, "prompt_lookup_max": n}.
"""

import torch

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams
from fusioninfer_amd.engine.spec_decode import SpeculativeConfig, propose_ngram
from fusioninfer_amd.models.registry import get_model_config


def make_engine(speculative=None, seed=0, **sched):
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=1024,
            max_model_len=256, **sched,
        ),
        speculative=speculative,
        seed=seed,
    )
    return LLMEngine(cfg, device="cpu")


# ------------------------------------------------------------- proposer
def test_propose_ngram_basic():
    # trailing [1, 2] matched earlier; the 3 tokens after it follow
    toks = [1, 2, 7, 8, 9, 5, 1, 2]
    assert propose_ngram(toks, max_n=3, min_n=2, k=3) == [7, 8, 9]


def test_propose_ngram_prefers_longest_and_most_recent():
    # trailing [4, 1, 2] occurs earlier twice; the MOST RECENT match wins
    toks = [4, 1, 2, 6, 0, 4, 1, 2, 9, 9, 4, 1, 2]
    assert propose_ngram(toks, max_n=3, min_n=2, k=2) == [9, 9]


def test_propose_ngram_no_match():
    assert propose_ngram([1, 2, 3, 4, 5, 6], max_n=3, min_n=2, k=4) == []
    assert propose_ngram([1, 2], max_n=3, min_n=2, k=4) == []


def test_propose_ngram_truncated_tail():
    # match is near the end: draft is whatever tokens exist after it
    toks = [5, 6, 7, 5, 6]
    assert propose_ngram(toks, max_n=2, min_n=2, k=4) == [7, 5, 6]


# ------------------------------------------------------- engine exactness
def _repetitive_prompts():
    # repetition makes the ngram proposer fire; random-init weights make
    # the model's continuations arbitrary but deterministic
    return [
        [3, 1, 4, 1, 5, 9] * 6,
        [2, 7, 2, 7, 2, 7, 2, 7] * 4,
        [11, 12, 13] * 10,
    ]


def test_spec_matches_plain_greedy():
    torch.manual_seed(0)
    base = make_engine()
    expected = [
        o.output_token_ids
        for o in base.generate(_repetitive_prompts(),
                               SamplingParams(max_tokens=24))
    ]
    spec = make_engine(SpeculativeConfig(num_speculative_tokens=4))
    outs = spec.generate(_repetitive_prompts(), SamplingParams(max_tokens=24))
    for o, exp in zip(outs, expected):
        assert o.output_token_ids == exp
    # the speculative path must actually have run (not silently fallen
    # back to plain decode)
    assert spec.num_spec_draft_tokens > 0
    assert spec.num_spec_accepted_tokens >= 0


def test_spec_respects_max_tokens_and_stop():
    torch.manual_seed(0)
    spec = make_engine(SpeculativeConfig(num_speculative_tokens=4))
    outs = spec.generate(_repetitive_prompts()[:1],
                         SamplingParams(max_tokens=7))
    assert len(outs[0].output_token_ids) == 7

    # stop token mid-draft: find what plain greedy emits, then stop on its
    # 3rd token — spec must cut at exactly the same place
    base = make_engine()
    full = base.generate(_repetitive_prompts()[:1],
                         SamplingParams(max_tokens=10))[0].output_token_ids
    stop_tok = full[2]
    want = full[: full.index(stop_tok) + 1]
    spec2 = make_engine(SpeculativeConfig(num_speculative_tokens=4))
    outs2 = spec2.generate(
        _repetitive_prompts()[:1],
        SamplingParams(max_tokens=10, ignore_eos=False,
                       stop_token_ids=[stop_tok]),
    )
    assert outs2[0].output_token_ids == want


def test_spec_mixed_with_sampled_and_logprobs_requests():
    """Non-greedy / logprobs sequences ride the same step draft-less and
    keep their exact non-spec behavior."""
    torch.manual_seed(0)
    base = make_engine(seed=7)
    prompts = _repetitive_prompts()
    sp_greedy = SamplingParams(max_tokens=16)
    sp_seeded = SamplingParams(max_tokens=16, temperature=0.8, seed=123)
    sp_logprob = SamplingParams(max_tokens=16, logprobs=2)
    ids = [
        base.add_request(prompts[0], sp_greedy),
        base.add_request(prompts[1], sp_seeded),
        base.add_request(prompts[2], sp_logprob),
    ]
    done = {}
    while base.has_unfinished():
        for o in base.step():
            if o.finished:
                done[o.request_id] = o
    exp = [done[i] for i in ids]

    spec = make_engine(SpeculativeConfig(num_speculative_tokens=4), seed=7)
    sp_seeded2 = SamplingParams(max_tokens=16, temperature=0.8, seed=123)
    sp_logprob2 = SamplingParams(max_tokens=16, logprobs=2)
    ids2 = [
        spec.add_request(prompts[0], SamplingParams(max_tokens=16)),
        spec.add_request(prompts[1], sp_seeded2),
        spec.add_request(prompts[2], sp_logprob2),
    ]
    done2 = {}
    while spec.has_unfinished():
        for o in spec.step():
            if o.finished:
                done2[o.request_id] = o
    got = [done2[i] for i in ids2]
    # greedy request: token-exact
    assert got[0].output_token_ids == exp[0].output_token_ids
    # per-request-seeded request: seeded draws depend only on
    # (seed, position), so they are reproducible across engines
    assert got[1].output_token_ids == exp[1].output_token_ids
    # logprobs request: entries present, one per emitted token
    assert len(got[2].logprobs) == len(got[2].output_token_ids)
    assert spec.num_spec_draft_tokens > 0


def test_spec_new_token_ids_stream_contract():
    """RequestOutput.new_token_ids concatenated over steps == the full
    output (what serving streams to SSE clients)."""
    torch.manual_seed(0)
    spec = make_engine(SpeculativeConfig(num_speculative_tokens=4))
    rid = spec.add_request(_repetitive_prompts()[0],
                           SamplingParams(max_tokens=20))
    streamed = []
    final = None
    while spec.has_unfinished():
        for o in spec.step():
            streamed.extend(o.new_token_ids)
            if o.finished:
                final = o
    assert final is not None
    assert streamed == final.output_token_ids


def test_spec_under_tiny_block_pool():
    """Draft tails that don't fit the block pool are dropped, not fatal."""
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=12),
        scheduler=SchedulerConfig(max_num_seqs=4,
                                  max_num_batched_tokens=1024,
                                  max_model_len=128),
        speculative=SpeculativeConfig(num_speculative_tokens=4),
    )
    eng = LLMEngine(cfg, device="cpu")
    base = make_engine()
    prompts = [[5, 6, 7] * 11, [9, 8] * 16]
    expected = [
        o.output_token_ids
        for o in base.generate(prompts, SamplingParams(max_tokens=8))
    ]
    outs = eng.generate(prompts, SamplingParams(max_tokens=8))
    for o, exp in zip(outs, expected):
        assert o.output_token_ids == exp


def test_disable_by_batch_size():
    """vLLM speculative_disable_by_batch_size parity: above the cap the
    step takes the plain decode path (no drafts)."""
    torch.manual_seed(0)
    spec = make_engine(
        SpeculativeConfig(num_speculative_tokens=4, disable_by_batch_size=1)
    )
    outs = spec.generate(_repetitive_prompts()[:2],
                         SamplingParams(max_tokens=12))
    assert all(len(o.output_token_ids) == 12 for o in outs)
    assert spec.num_spec_draft_tokens == 0  # 2 seqs > cap of 1

    base = make_engine()
    ref = [o.output_token_ids
           for o in base.generate(_repetitive_prompts()[:2],
                                  SamplingParams(max_tokens=12))]
    assert [o.output_token_ids for o in outs] == ref


# --------------------------------------------------- draft-model method
def make_draft_engine(draft_model="tiny-qwen3", k=4, blocks=None, seed=0):
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=1024, max_model_len=256
        ),
        speculative=SpeculativeConfig(
            method="draft_model", model=draft_model,
            num_speculative_tokens=k, draft_gpu_blocks=blocks,
        ),
        seed=seed,
    )
    return LLMEngine(cfg, device="cpu")


def test_draft_model_same_arch_always_accepts():
    """Draft == target (same registry config + seed => identical random
    weights): every draft token matches the target argmax, so each spec
    step emits k+1 tokens and acceptance is 100%."""
    torch.manual_seed(0)
    base = make_engine()
    prompts = [[3, 9, 27, 4] * 4, [8, 8, 1] * 5]
    expected = [
        o.output_token_ids
        for o in base.generate(prompts, SamplingParams(max_tokens=21))
    ]
    eng = make_draft_engine()
    outs = eng.generate(prompts, SamplingParams(max_tokens=21))
    for o, exp in zip(outs, expected):
        assert o.output_token_ids == exp
    assert eng.num_spec_draft_tokens > 0
    assert eng.num_spec_accepted_tokens == eng.num_spec_draft_tokens
    # draft KV state is released when sequences finish
    assert eng.proposer._state == {}


def test_draft_model_exact_with_mixed_requests():
    """Sampled/logprobs sequences ride draft-less; greedy stays exact."""
    torch.manual_seed(0)
    base = make_engine(seed=5)
    prompts = [[3, 9, 27, 4] * 4, [1, 2, 3] * 6]
    a = base.add_request(prompts[0], SamplingParams(max_tokens=12))
    b = base.add_request(prompts[1],
                         SamplingParams(max_tokens=12, temperature=0.7,
                                        seed=77))
    done = {}
    while base.has_unfinished():
        for o in base.step():
            if o.finished:
                done[o.request_id] = o
    eng = make_draft_engine(seed=5)
    a2 = eng.add_request(prompts[0], SamplingParams(max_tokens=12))
    b2 = eng.add_request(prompts[1],
                         SamplingParams(max_tokens=12, temperature=0.7,
                                        seed=77))
    done2 = {}
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                done2[o.request_id] = o
    assert done2[a2].output_token_ids == done[a].output_token_ids
    assert done2[b2].output_token_ids == done[b].output_token_ids


def test_draft_model_tiny_pool_degrades_gracefully():
    """Draft block pool exhaustion drops drafts, never breaks output."""
    torch.manual_seed(0)
    base = make_engine()
    prompts = [[5, 6, 7] * 10, [9, 8] * 12]
    expected = [
        o.output_token_ids
        for o in base.generate(prompts, SamplingParams(max_tokens=10))
    ]
    eng = make_draft_engine(blocks=2)  # room for ~1 sequence's draft KV
    outs = eng.generate(prompts, SamplingParams(max_tokens=10))
    for o, exp in zip(outs, expected):
        assert o.output_token_ids == exp


def test_draft_model_abort_releases_state():
    torch.manual_seed(0)
    eng = make_draft_engine()
    rid = eng.add_request([3, 9, 27, 4] * 4, SamplingParams(max_tokens=50))
    for _ in range(4):
        eng.step()
    assert rid in eng.seqs
    assert eng.abort_request(rid)
    assert eng.proposer._state == {}
