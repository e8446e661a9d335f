"""Randomized workload fuzzing: block-accounting and scheduler invariants
hold under arbitrary add/step/abort interleavings (CPU, tiny model)."""

import random

import torch

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams
from fusioninfer_amd.models.registry import get_model_config


def check_block_invariants(eng: LLMEngine):
    bm = eng.block_manager
    # every live sequence's blocks have refcount >= 1
    live_blocks = {}
    for seq in list(eng.scheduler.running) + list(eng.scheduler.waiting):
        for b in seq.block_ids:
            live_blocks[b] = live_blocks.get(b, 0) + 1
    for b, n in live_blocks.items():
        assert bm.ref_count.get(b, 0) >= 1, (b, n)
        assert b not in bm.free_blocks
    # free + referenced partitions the pool (cached_free are ref-0 blocks)
    accounted = (
        len(bm.free_blocks) + len(bm.cached_free) + len(bm.ref_count)
    )
    assert accounted == bm.num_blocks, (
        len(bm.free_blocks), len(bm.cached_free), len(bm.ref_count),
        bm.num_blocks,
    )
    # no block simultaneously free and referenced
    assert not (set(bm.free_blocks) & set(bm.ref_count))
    assert not (set(bm.cached_free) & set(bm.ref_count))


def test_fuzz_workload_invariants():
    torch.manual_seed(0)
    rng = random.Random(1234)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=48, enable_prefix_caching=True),
        scheduler=SchedulerConfig(
            max_num_seqs=6, max_num_batched_tokens=128, max_model_len=192
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    shared = [rng.randrange(1000) for _ in range(48)]
    pending = []
    finished = 0
    for it in range(300):
        op = rng.random()
        if op < 0.30 and len(eng.seqs) < 12:
            # mix of shared-prefix and fresh prompts, varied lengths
            if rng.random() < 0.5:
                prompt = shared[: rng.randrange(16, 48)] + [
                    rng.randrange(1000) for _ in range(rng.randrange(1, 30))
                ]
            else:
                prompt = [rng.randrange(1000) for _ in range(rng.randrange(1, 120))]
            rid = eng.add_request(
                prompt, SamplingParams(max_tokens=rng.randrange(1, 12))
            )
            pending.append(rid)
        elif op < 0.38 and pending:
            rid = pending.pop(rng.randrange(len(pending)))
            eng.abort_request(rid)
        else:
            for out in eng.step():
                if out.finished:
                    finished += 1
                    if out.request_id in pending:
                        pending.remove(out.request_id)
                    assert len(out.output_token_ids) <= 12
        check_block_invariants(eng)
    # drain
    guard = 0
    while eng.has_unfinished() and guard < 2000:
        for out in eng.step():
            if out.finished:
                finished += 1
        guard += 1
        check_block_invariants(eng)
    assert not eng.has_unfinished()
    assert finished > 20
    # all blocks returned (cached_free blocks count as free)
    assert eng.block_manager.num_free() == eng.block_manager.num_blocks


def test_fuzz_with_spec_guided_priority():
    """The same invariants hold with the late-round features live:
    ngram speculation (draft block tails), guided grammars, priority
    scheduling, logit_bias, min_tokens — plus draft-model proposer state
    is empty once everything drains."""
    from fusioninfer_amd.engine.spec_decode import SpeculativeConfig
    from fusioninfer_amd.guided import Vocabulary, build_guided

    torch.manual_seed(0)
    rng = random.Random(99)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=48, enable_prefix_caching=True),
        scheduler=SchedulerConfig(
            max_num_seqs=6, max_num_batched_tokens=128, max_model_len=192,
            policy="priority",
        ),
        speculative=SpeculativeConfig(method="draft_model",
                                      model="tiny-qwen3",
                                      num_speculative_tokens=3,
                                      draft_gpu_blocks=24),
    )
    eng = LLMEngine(cfg, device="cpu")
    vocab = Vocabulary(
        cfg.model.vocab_size,
        lambda t: bytes([max(t - 3, 0) & 0xFF]).decode("utf-8", "replace"),
    )
    pending = []
    finished = 0
    for it in range(250):
        op = rng.random()
        if op < 0.30 and len(eng.seqs) < 10:
            kind = rng.random()
            sp_kw = {"max_tokens": rng.randrange(1, 12)}
            if kind < 0.2:
                sp_kw["guided"] = build_guided(
                    "choice", ["yes", "no"], vocab
                )
            elif kind < 0.4:
                sp_kw["temperature"] = 0.8
                sp_kw["seed"] = rng.randrange(1000)
            elif kind < 0.5:
                sp_kw["logit_bias"] = {str(rng.randrange(500)): 5.0}
            elif kind < 0.6:
                sp_kw["min_tokens"] = 2
                sp_kw["ignore_eos"] = False
                sp_kw["stop_token_ids"] = [rng.randrange(500)]
            # repetitive prompts so the ngram/draft proposers fire
            chunk = [rng.randrange(500) for _ in range(8)]
            reps = rng.randrange(2, 12)
            rid = eng.add_request(chunk * reps, SamplingParams(**sp_kw),
                                  priority=rng.randrange(3))
            pending.append(rid)
        elif op < 0.38 and pending:
            eng.abort_request(pending.pop(rng.randrange(len(pending))))
        else:
            for out in eng.step():
                if out.finished:
                    finished += 1
                    if out.request_id in pending:
                        pending.remove(out.request_id)
        check_block_invariants(eng)
    guard = 0
    while eng.has_unfinished() and guard < 2000:
        for out in eng.step():
            if out.finished:
                finished += 1
        guard += 1
        check_block_invariants(eng)
    assert not eng.has_unfinished()
    assert finished > 15
    assert eng.block_manager.num_free() == eng.block_manager.num_blocks
    # draft proposer released every sequence's KV state and blocks
    assert eng.proposer._state == {}
    assert len(eng.proposer.bm.free_blocks) == eng.proposer.bm.num_blocks
