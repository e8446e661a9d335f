"""Engine correctness tests (CPU, tiny model, reference ops).

The key invariant: greedy generation through the paged-KV decode path must
match an oracle that re-runs full prefill attention over (prompt + generated)
at every step — i.e. the cache path computes the same attention.
"""

import torch

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.engine.block_manager import BlockManager
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams, Sequence
from fusioninfer_amd.models.registry import get_model_config


def make_engine(num_blocks=256, max_seqs=8, max_len=256, seed=0):
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=num_blocks),
        scheduler=SchedulerConfig(
            max_num_seqs=max_seqs, max_num_batched_tokens=1024, max_model_len=max_len
        ),
        seed=seed,
    )
    return LLMEngine(cfg, device="cpu")


def oracle_greedy(engine, prompt, n_tokens):
    """Re-runs the model as pure prefill over the growing sequence."""
    runner = engine.runner
    bm = BlockManager(runner.num_gpu_blocks, engine.cfg.cache.block_size)
    toks = list(prompt)
    out = []
    for _ in range(n_tokens):
        seq = Sequence("oracle", toks, SamplingParams())
        bm.allocate(seq)
        logits = runner.execute_prefill([seq], bm)
        tok = int(logits.float().argmax(dim=-1)[0])
        out.append(tok)
        toks.append(tok)
        bm.free(seq)
    return out


def test_decode_matches_prefill_oracle():
    torch.manual_seed(0)
    eng = make_engine()
    prompts = [[3, 1, 4, 1, 5, 9, 2, 6] * 3, [2, 7, 1, 8] * 9, [1] * 17]
    expected = [oracle_greedy(eng, p, 6) for p in prompts]
    outs = eng.generate(prompts, SamplingParams(max_tokens=6))
    for o, exp in zip(outs, expected):
        assert o.output_token_ids == exp


def test_preemption_recompute_still_correct():
    torch.manual_seed(0)
    # tiny cache: 12 blocks of 16 = 192 token slots; prompts force eviction
    eng = make_engine(num_blocks=12, max_seqs=4, max_len=128)
    prompts = [[5, 6, 7] * 11, [9, 8] * 16, [4] * 40]
    expected = [oracle_greedy(eng, p, 8) for p in prompts]
    outs = eng.generate(prompts, SamplingParams(max_tokens=8))
    for o, exp in zip(outs, expected):
        assert o.output_token_ids == exp
    assert eng.num_finished == 3


def test_continuous_batching_join_midstream():
    torch.manual_seed(0)
    eng = make_engine()
    a = eng.add_request([1, 2, 3] * 5, SamplingParams(max_tokens=10))
    # a few steps before the second request arrives
    for _ in range(4):
        eng.step()
    b = eng.add_request([7, 7, 1] * 4, SamplingParams(max_tokens=5))
    while eng.has_unfinished():
        eng.step()
    assert eng.num_finished == 2


def test_block_manager_prefix_caching_reuse():
    bm = BlockManager(16, 16, enable_prefix_caching=True)
    s1 = Sequence("a", list(range(40)), SamplingParams())
    bm.allocate(s1)
    assert len(s1.block_ids) == 3
    s1_blocks = list(s1.block_ids)
    bm.free(s1)
    # identical prompt: the two full blocks should be reused
    s2 = Sequence("b", list(range(40)), SamplingParams())
    bm.allocate(s2)
    assert s2.num_cached_tokens == 32
    assert s2.block_ids[:2] == s1_blocks[:2]
    # different prompt: no reuse
    s3 = Sequence("c", [99] * 40, SamplingParams())
    bm.allocate(s3)
    assert s3.num_cached_tokens == 0


def test_block_manager_accounting():
    bm = BlockManager(8, 16)
    s = Sequence("a", list(range(33)), SamplingParams())
    assert bm.can_allocate(33)
    bm.allocate(s)
    assert len(s.block_ids) == 3
    assert bm.num_free() == 5
    # appending within the last block needs no new block
    s.output_token_ids = [1]  # num_tokens=34 -> position 33 in block 2
    bm.append_slot(s)
    assert len(s.block_ids) == 3
    # position 48 (num_tokens=49) crosses into a 4th block
    s.output_token_ids = list(range(16))
    bm.append_slot(s)
    assert len(s.block_ids) == 4
    bm.free(s)
    assert bm.num_free() == 8


def test_prefix_caching_end_to_end():
    """Second identical-prompt request reuses cached blocks and produces
    identical greedy tokens (context-attention path)."""
    torch.manual_seed(0)
    eng = make_engine()
    assert eng.block_manager.enable_prefix_caching
    prompt = [7, 3, 9, 1] * 12  # 48 tokens = 3 full blocks
    out1 = eng.generate([prompt], SamplingParams(max_tokens=5))[0]
    out2 = eng.generate([prompt], SamplingParams(max_tokens=5))[0]
    assert out1.output_token_ids == out2.output_token_ids
    # the second request must have hit the prefix cache (2 full blocks;
    # the 3rd is capped so >=1 token is recomputed)
    # (seq objects are gone; verify via the hash table state)
    assert len(eng.block_manager.hash_to_block) >= 2


def test_prefix_caching_shared_prefix_divergent_tail():
    torch.manual_seed(0)
    eng = make_engine()
    base = [5, 1, 2, 6] * 10          # 40 tokens, 2 full blocks cacheable
    p1 = base + [11, 12, 13]
    p2 = base + [21, 22, 23, 24]
    r1 = eng.generate([p1], SamplingParams(max_tokens=4))[0]
    # second shares the 2-block prefix but diverges after
    r2 = eng.generate([p2], SamplingParams(max_tokens=4))[0]
    # oracle without caching
    eng2 = make_engine()
    eng2.block_manager.enable_prefix_caching = False
    e1 = eng2.generate([p1], SamplingParams(max_tokens=4))[0]
    e2 = eng2.generate([p2], SamplingParams(max_tokens=4))[0]
    assert r1.output_token_ids == e1.output_token_ids
    assert r2.output_token_ids == e2.output_token_ids


def test_chunked_prefill_matches_unchunked():
    """A prompt longer than max_num_batched_tokens is prefilled in chunks
    (context attention over the cache) and yields the same greedy tokens."""
    torch.manual_seed(0)
    # budget 48 tokens per step, prompt 100 -> 3 chunks
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=48, max_model_len=256
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    prompt = [3, 7, 2, 9, 4] * 20  # 100 tokens
    expected = oracle_greedy(eng, prompt, 5)
    out = eng.generate([prompt], SamplingParams(max_tokens=5))[0]
    assert out.output_token_ids == expected


def test_chunked_prefill_interleaves_decode():
    """While one prompt is mid-chunk, already-running sequences keep
    decoding between chunks (prefill-priority per step, chunk-bounded)."""
    torch.manual_seed(0)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=32, max_model_len=256
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    a = eng.add_request([1, 2] * 8, SamplingParams(max_tokens=20))
    for _ in range(3):
        eng.step()
    b = eng.add_request([4, 5, 6] * 30, SamplingParams(max_tokens=3))  # 90 toks
    while eng.has_unfinished():
        eng.step()
    assert eng.num_finished == 2


def test_abort_request():
    torch.manual_seed(0)
    eng = make_engine()
    a = eng.add_request([1, 2, 3] * 10, SamplingParams(max_tokens=50))
    b = eng.add_request([4, 5] * 10, SamplingParams(max_tokens=5))
    for _ in range(3):
        eng.step()
    free_before = eng.block_manager.num_free()
    assert eng.abort_request(a)
    assert eng.block_manager.num_free() > free_before
    assert not eng.abort_request(a)  # already gone
    while eng.has_unfinished():
        eng.step()
    assert eng.num_finished == 1  # only b completed


def test_logprobs():
    torch.manual_seed(0)
    eng = make_engine()
    rid = eng.add_request(
        [3, 1, 4] * 8, SamplingParams(max_tokens=4, logprobs=3)
    )
    outs = {}
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                outs[o.request_id] = o
    out = outs[rid]
    assert len(out.logprobs) == 4
    for sampled_lp, top in out.logprobs:
        assert sampled_lp <= 0.0
        assert len(top) == 3
        # greedy sampling: the sampled token's logprob equals the max
        assert abs(max(top.values()) - sampled_lp) < 1e-5


def test_admission_aging_escape_under_low_load(monkeypatch):
    """A waiting prompt must not starve behind long-running decodes when
    the batching threshold is never reached (low-load latency bug)."""
    import time as _time

    torch.manual_seed(0)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=4096, max_model_len=512,
        ),
    )
    cfg.scheduler.prefill_admission_tokens = 4096
    eng = LLMEngine(cfg, device="cpu")
    # one long decode in flight
    eng.add_request([1, 2, 3] * 5, SamplingParams(max_tokens=400))
    for _ in range(3):
        eng.step()
    # a small prompt arrives: threshold (4096) will never accumulate
    b = eng.add_request([9, 9] * 8, SamplingParams(max_tokens=2))
    arrival = eng.seqs[b].arrival_time
    # age it past the 50 ms escape
    eng.seqs[b].arrival_time = arrival - 1.0
    eng.step()
    assert eng.seqs[b].num_computed_tokens > 0  # admitted, not starved


def test_swap_preemption_is_transparent():
    """With CPU swap space, a preempted sequence parks its KV and resumes
    with the EXACT same cache state — greedy outputs must be identical to
    a run with a cache big enough to never preempt (recompute-preemption
    cannot promise this: re-prefilling decode tokens changes op order and
    can flip argmax). At least one swap must happen under the tiny cache."""
    import torch

    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config

    def run(swap_gb, num_blocks):
        torch.manual_seed(9)
        cfg = EngineConfig(
            model=get_model_config("tiny-qwen3"),
            # 12 blocks of 16 = 192 token slots: three 40-token prompts
            # generating 40 tokens each cannot all stay resident
            cache=CacheConfig(num_gpu_blocks=num_blocks,
                              enable_prefix_caching=False,
                              swap_space_gb=swap_gb),
            scheduler=SchedulerConfig(
                max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
            ),
        )
        eng = LLMEngine(cfg, device="cpu")
        prompts = [
            [(11 * i + j) % 500 for j in range(40)] for i in range(3)
        ]
        outs = eng.generate(
            prompts,
            SamplingParams(max_tokens=40, temperature=0.0, ignore_eos=True),
        )
        assert eng.block_manager.num_free() == eng.block_manager.num_blocks
        return [o.output_token_ids for o in outs], eng.num_swap_outs

    toks_unconstrained, s0 = run(0.0, 64)
    assert s0 == 0
    toks_swap, swaps = run(1.0, 12)
    assert swaps > 0, "tiny cache never triggered a swap"
    assert toks_swap == toks_unconstrained


def test_moe_engine_generates_and_routes():
    """Qwen3-MoE family: tiny MoE model generates; the router actually
    spreads tokens over multiple experts; MoE layer matches a hand
    reference on one forward."""
    import torch
    import torch.nn.functional as F

    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config

    torch.manual_seed(4)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3-moe"),
        cache=CacheConfig(num_gpu_blocks=64),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    out = eng.generate(
        [[5, 3, 1] * 10], SamplingParams(max_tokens=5, temperature=0.0)
    )[0]
    assert len(out.output_token_ids) == 5

    # MoE layer oracle: dense per-expert compute, weighted-sum combine
    moe = eng.runner.model.layers[0].mlp
    x = torch.randn(7, cfg.model.hidden_size, dtype=torch.bfloat16)
    got = moe(x).float()
    logits = x.float() @ moe.router_weight.float().T
    probs = torch.softmax(logits, -1)
    topv, topi = probs.topk(moe.top_k, -1)
    topv = topv / topv.sum(-1, keepdim=True)
    exp = torch.zeros_like(x, dtype=torch.float32)
    for t in range(x.shape[0]):
        for k in range(moe.top_k):
            e = int(topi[t, k])
            gu = x[t].float() @ moe.gate_up_t[e].float()
            g, u = gu.chunk(2)
            y = (F.silu(g) * u) @ moe.down_t[e].float()
            exp[t] += float(topv[t, k]) * y
    # the layer computes in bf16 GEMMs; compare loosely
    rel = (got - exp).norm() / exp.norm()
    assert rel.item() < 0.05, rel.item()
    # routing uses more than one expert across tokens
    assert len(set(topi.flatten().tolist())) > 1


def test_swap_budget_exhaustion_falls_back_to_recompute():
    """A swap budget too small for one sequence's KV forces the recompute
    path; generation still completes and frees all blocks."""
    import torch

    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config

    torch.manual_seed(9)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=12, enable_prefix_caching=False,
                          swap_space_gb=1e-9),  # ~1 byte: never fits
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    prompts = [[(11 * i + j) % 500 for j in range(40)] for i in range(3)]
    outs = eng.generate(
        prompts, SamplingParams(max_tokens=40, temperature=0.0,
                                ignore_eos=True)
    )
    assert all(len(o.output_token_ids) == 40 for o in outs)
    assert eng.num_swap_outs == 0
    assert eng.block_manager.num_free() == eng.block_manager.num_blocks


def test_abort_swapped_request_frees_swap_space():
    import torch

    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.engine.llm_engine import LLMEngine
    from fusioninfer_amd.engine.sequence import SamplingParams
    from fusioninfer_amd.models.registry import get_model_config

    torch.manual_seed(9)
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=12, enable_prefix_caching=False,
                          swap_space_gb=1.0),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=256, max_model_len=128
        ),
    )
    eng = LLMEngine(cfg, device="cpu")
    ids = [
        eng.add_request([(7 * i + j) % 500 for j in range(40)],
                        SamplingParams(max_tokens=60, temperature=0.0,
                                       ignore_eos=True))
        for i in range(3)
    ]
    # run until something swaps, then abort the swapped request
    for _ in range(200):
        eng.step()
        if eng._swapped:
            break
    assert eng._swapped, "no swap happened"
    victim = next(iter(eng._swapped))
    assert eng.abort_request(victim)
    assert victim not in eng._swapped
    while eng.has_unfinished():
        eng.step()
    assert eng._swap_bytes == 0
    assert eng.block_manager.num_free() == eng.block_manager.num_blocks


def test_min_tokens_suppresses_stop():
    """vLLM min_tokens parity: stop tokens ignored until min reached."""
    from fusioninfer_amd.engine.sequence import SamplingParams, Sequence

    sp = SamplingParams(max_tokens=8, min_tokens=3, ignore_eos=False,
                        stop_token_ids=[7])
    seq = Sequence("s", [1, 2], sp)
    seq.append_token(7)
    assert not seq.check_stop()      # 1 < min_tokens
    seq.append_token(5)
    assert not seq.check_stop()
    seq.append_token(7)
    assert seq.check_stop()          # 3 >= min_tokens and stop token


def test_prompt_logprobs_match_oracle():
    """prompt_logprobs (vLLM parity): per-position logprob of each prompt
    token under the model, computed during (chunked) prefill; index 0 is
    None. Oracle: one full prefill forward scoring every position."""
    import torch.nn.functional as F

    from fusioninfer_amd.engine.block_manager import BlockManager
    from fusioninfer_amd.engine.sequence import SamplingParams, Sequence

    torch.manual_seed(0)
    eng = make_engine()
    prompt = [3, 1, 4, 1, 5, 9, 2, 6] * 5  # 40 tokens

    # oracle logits over the whole prompt
    runner = eng.runner
    bm = BlockManager(runner.num_gpu_blocks, eng.cfg.cache.block_size)
    oseq = Sequence("o", prompt, SamplingParams())
    bm.allocate(oseq)
    payload = runner.build_prefill_payload([oseq], bm)
    payload["logits_rows"] = list(range(len(prompt)))
    logits = runner.run_prefill(payload).float()
    lp = torch.log_softmax(logits, dim=-1)
    expect = [None] + [
        float(lp[i - 1, prompt[i]]) for i in range(1, len(prompt))
    ]
    bm.free(oseq)

    rid = eng.add_request(
        prompt, SamplingParams(max_tokens=2, prompt_logprobs=2)
    )
    out = None
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                out = o
    got = out.prompt_logprobs
    assert len(got) == len(prompt)
    assert got[0] is None
    for e, g in zip(expect[1:], got[1:]):
        assert abs(e - g[0]) < 1e-3
        assert len(g[1]) == 2  # top-2 entries


def test_prompt_logprobs_chunked_prefill():
    """Chunked prefill produces the same prompt logprobs as one chunk."""
    torch.manual_seed(0)
    big = make_engine()
    prompt = [7, 2, 9] * 20  # 60 tokens
    r1 = big.add_request(prompt, SamplingParams(max_tokens=1,
                                                prompt_logprobs=0))
    o1 = None
    while big.has_unfinished():
        for o in big.step():
            o1 = o
    # tiny token budget forces multi-chunk prefill
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=8, max_num_batched_tokens=16, max_model_len=256
        ),
        seed=0,
    )
    torch.manual_seed(0)
    small = LLMEngine(cfg, device="cpu")
    r2 = small.add_request(prompt, SamplingParams(max_tokens=1,
                                                  prompt_logprobs=0))
    o2 = None
    while small.has_unfinished():
        for o in small.step():
            o2 = o
    assert len(o1.prompt_logprobs) == len(o2.prompt_logprobs) == len(prompt)
    for a, b in zip(o1.prompt_logprobs[1:], o2.prompt_logprobs[1:]):
        assert abs(a[0] - b[0]) < 1e-3


def test_priority_scheduling_admission_and_preemption():
    """--scheduling-policy priority: lower value admits first and the
    lowest-priority running sequence is preempted first."""
    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=256),
        scheduler=SchedulerConfig(
            max_num_seqs=1, max_num_batched_tokens=1024, max_model_len=256,
            policy="priority",
        ),
    )
    torch.manual_seed(0)
    eng = LLMEngine(cfg, device="cpu")
    # low-priority arrives FIRST, high-priority second; with one slot the
    # high-priority one must run first
    lo = eng.add_request([5, 6, 7] * 4, SamplingParams(max_tokens=3),
                         priority=10)
    hi = eng.add_request([9, 8, 7] * 4, SamplingParams(max_tokens=3),
                         priority=0)
    order = []
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                order.append(o.request_id)
    assert order == [hi, lo]

    # preemption picks the lowest-priority victim (tiny pool forces it)
    cfg2 = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=8),
        scheduler=SchedulerConfig(
            max_num_seqs=4, max_num_batched_tokens=1024, max_model_len=128,
            policy="priority",
        ),
    )
    torch.manual_seed(0)
    eng2 = LLMEngine(cfg2, device="cpu")
    a = eng2.add_request([4] * 40, SamplingParams(max_tokens=30), priority=0)
    b = eng2.add_request([5] * 40, SamplingParams(max_tokens=30), priority=9)
    done = []
    steps = 0
    while eng2.has_unfinished() and steps < 500:
        steps += 1
        for o in eng2.step():
            if o.finished:
                done.append(o.request_id)
    assert set(done) == {a, b}
    # the low-priority request was the preemption victim
    assert eng2.num_preemptions > 0
    assert done[0] == a


# ---------------------------------------------------- pipelined decode
def test_async_decode_matches_sync():
    """The pipelined decode chain (deferred emission, device-fed ids)
    must produce byte-identical greedy streams to the sync path, across
    ragged max_tokens (deterministic finishers shrink the chain)."""
    torch.manual_seed(0)
    prompts = [[3, 1, 4, 1, 5] * 4, [2, 7] * 9, [11, 12, 13] * 7]
    lens = [6, 17, 11]
    sync = make_engine()
    sync._async_decode = False
    a = [
        sync.generate([p], SamplingParams(max_tokens=n))[0].output_token_ids
        for p, n in zip(prompts, lens)
    ]
    # batched, async (default): all three in flight together
    eng = make_engine()
    assert eng._async_decode
    ids = [
        eng.add_request(p, SamplingParams(max_tokens=n))
        for p, n in zip(prompts, lens)
    ]
    done = {}
    while eng.has_unfinished():
        for out in eng.step():
            if out.finished:
                done[out.request_id] = out
    assert eng.num_async_steps > 0  # the pipeline actually engaged
    for rid, exp in zip(ids, a):
        assert done[rid].output_token_ids == exp
        assert done[rid].finish_reason == "length"


def test_async_decode_abort_mid_chain():
    """Aborting while a step is in flight drains the pipeline; other
    requests' tokens survive via the backlog and finish correctly."""
    torch.manual_seed(0)
    eng = make_engine()
    keep_exp = None
    sync = make_engine()
    sync._async_decode = False
    keep_exp = sync.generate(
        [[5, 6, 7] * 5], SamplingParams(max_tokens=12)
    )[0].output_token_ids
    r1 = eng.add_request([5, 6, 7] * 5, SamplingParams(max_tokens=12))
    r2 = eng.add_request([9, 9, 2] * 5, SamplingParams(max_tokens=40))
    for _ in range(6):
        eng.step()
    assert eng._pending is not None
    assert eng.abort_request(r2)
    assert eng._pending is None  # chain flushed
    done = {}
    while eng.has_unfinished():
        for out in eng.step():
            if out.finished:
                done[out.request_id] = out
    assert done[r1].output_token_ids == keep_exp


def test_async_decode_mixed_admission_breaks_chain():
    """A prompt arriving mid-chain breaks the pipeline at the admission
    window and still prefills + finishes correctly."""
    torch.manual_seed(0)
    eng = make_engine()
    r1 = eng.add_request([1, 2, 3] * 6, SamplingParams(max_tokens=30))
    for _ in range(5):
        eng.step()
    assert eng._pending is not None
    r2 = eng.add_request([4, 5] * 8, SamplingParams(max_tokens=5))
    done = {}
    steps = 0
    while eng.has_unfinished() and steps < 200:
        steps += 1
        for out in eng.step():
            if out.finished:
                done[out.request_id] = out
    assert set(done) == {r1, r2}
    assert len(done[r2].output_token_ids) == 5
    assert len(done[r1].output_token_ids) == 30


def test_async_decode_under_block_pressure():
    """Block exhaustion mid-chain: the pipeline refuses continuation
    (capacity check), drains, and the sync path preempts/recomputes —
    all requests still complete with the right lengths."""
    torch.manual_seed(0)
    # tiny pool: 3 seqs x (prompt 32 + gen 40) over 16-token blocks
    # cannot all stay resident
    eng = make_engine(num_blocks=10, max_seqs=4, max_len=128)
    assert eng._async_decode
    ids = [eng.add_request([i + 1, i + 2, i + 3] * 6,
                           SamplingParams(max_tokens=40))
           for i in range(3)]
    done = {}
    steps = 0
    while eng.has_unfinished() and steps < 500:
        steps += 1
        for out in eng.step():
            if out.finished:
                done[out.request_id] = out
    assert set(done) == set(ids)
    assert all(len(done[r].output_token_ids) == 40 for r in ids)
    assert eng.num_preemptions > 0  # the pressure actually materialized


def test_decode_only_next_is_read_only():
    """The pipeline's scheduler preview must not mutate state (no slot
    appends, no admissions) — schedule() afterwards sees a clean view."""
    torch.manual_seed(0)
    eng = make_engine()
    eng.add_request([1, 2, 3] * 6, SamplingParams(max_tokens=8))
    eng.step()  # prefill
    free0 = eng.block_manager.num_free()
    blocks0 = {
        s.seq_id: list(s.block_ids) for s in eng.scheduler.running
    }
    for _ in range(3):
        eng.scheduler.decode_only_next()
    assert eng.block_manager.num_free() == free0
    assert {
        s.seq_id: list(s.block_ids) for s in eng.scheduler.running
    } == blocks0


def test_engine_fault_marks_unhealthy():
    """An exception inside engine.step flips the serving loop unhealthy
    and fails outstanding streams (complements the hang watchdog)."""
    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
    from fusioninfer_amd.server.serving import ServingEngine

    cfg = EngineConfig(
        model=get_model_config("tiny-qwen3"),
        cache=CacheConfig(num_gpu_blocks=64),
        scheduler=SchedulerConfig(
            max_num_seqs=2, max_num_batched_tokens=256, max_model_len=64
        ),
    )
    s = ServingEngine(cfg, device="cpu")
    try:
        def boom():
            raise RuntimeError("injected fault")

        s.engine.step = boom
        _, q = s.submit([5, 6, 7], SamplingParams(max_tokens=4))
        tok, finished = q.get(timeout=10)
        assert finished and tok is None  # stream failed, not hung
        import time as _t

        for _ in range(100):
            if not s.healthy:
                break
            _t.sleep(0.02)
        assert not s.healthy
        assert "injected fault" in s.last_error
    finally:
        s.shutdown()


def test_async_decode_env_kill_switch(monkeypatch):
    monkeypatch.setenv("FI_ASYNC_DECODE", "0")
    eng = make_engine()
    assert not eng._async_decode
    out = eng.generate([[1, 2, 3] * 5], SamplingParams(max_tokens=5))[0]
    assert len(out.output_token_ids) == 5
    assert eng.num_async_steps == 0


def test_async_decode_with_prefix_caching():
    """Pipelined decode + prefix caching: shared-prefix requests reuse
    cached blocks while the chain runs; streams match the sync engine."""
    from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig

    def build(async_on):
        cfg = EngineConfig(
            model=get_model_config("tiny-qwen3"),
            cache=CacheConfig(num_gpu_blocks=256,
                              enable_prefix_caching=True),
            scheduler=SchedulerConfig(
                max_num_seqs=8, max_num_batched_tokens=1024,
                max_model_len=256,
            ),
        )
        torch.manual_seed(0)
        e = LLMEngine(cfg, device="cpu")
        e._async_decode = async_on
        return e

    shared = [7, 3, 9] * 12
    prompts = [shared + [i] for i in range(4)]
    exp = [build(False).generate([p], SamplingParams(max_tokens=9))[0]
           .output_token_ids for p in prompts]
    eng = build(True)
    ids = [eng.add_request(p, SamplingParams(max_tokens=9))
           for p in prompts]
    done = {}
    while eng.has_unfinished():
        for out in eng.step():
            if out.finished:
                done[out.request_id] = out
    assert eng.num_async_steps > 0
    assert eng.block_manager.cache_hit_tokens > 0  # prefixes reused
    for rid, e in zip(ids, exp):
        assert done[rid].output_token_ids == e
