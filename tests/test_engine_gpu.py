"""GPU engine tests: decode path (incl. hipGraph replay) vs prefill oracle,
on the real HIP kernels with a small Qwen3-architecture model."""

import pytest
import torch

from fusioninfer_amd.config import CacheConfig, EngineConfig, SchedulerConfig
from fusioninfer_amd.engine.block_manager import BlockManager
from fusioninfer_amd.engine.llm_engine import LLMEngine
from fusioninfer_amd.engine.sequence import SamplingParams, Sequence
from fusioninfer_amd.models.registry import get_model_config

pytestmark = pytest.mark.gpu


def make_engine(enforce_eager, num_blocks=512, max_seqs=16):
    mc = get_model_config("Qwen3-0.6B")
    mc.num_layers = 4
    cfg = EngineConfig(
        model=mc,
        cache=CacheConfig(num_gpu_blocks=num_blocks),
        scheduler=SchedulerConfig(
            max_num_seqs=max_seqs, max_num_batched_tokens=2048, max_model_len=512
        ),
        seed=7,
        enforce_eager=enforce_eager,
    )
    return LLMEngine(cfg, device="cuda:0")


def oracle_greedy(engine, prompt, n_tokens):
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


@pytest.mark.parametrize("enforce_eager", [True, False])
def test_gpu_decode_matches_prefill_oracle(enforce_eager):
    torch.manual_seed(0)
    eng = make_engine(enforce_eager)
    prompts = [
        list(range(10, 150)),
        [3, 1, 4, 1, 5] * 13,
        [2] * 31,
    ]
    expected = [oracle_greedy(eng, p, 6) for p in prompts]
    outs = eng.generate(prompts, SamplingParams(max_tokens=6))
    mismatches = 0
    for o, exp in zip(outs, expected):
        # bf16 decode vs bf16 prefill can tie-break argmax differently on the
        # first token in rare cases; require exact match of the sequence
        if o.output_token_ids != exp:
            mismatches += 1
    assert mismatches == 0, (outs[0].output_token_ids, expected)


def test_gpu_graph_replay_consistent_with_eager():
    torch.manual_seed(0)
    prompts = [list(range(50, 120)), [11, 13, 17] * 20]
    eager = make_engine(True).generate(prompts, SamplingParams(max_tokens=8))
    graphed = make_engine(False).generate(prompts, SamplingParams(max_tokens=8))
    for a, b in zip(eager, graphed):
        assert a.output_token_ids == b.output_token_ids
