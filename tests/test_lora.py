"""LoRA serving tests (CPU, tiny model)."""

import torch

from fusioninfer_amd.engine.sequence import SamplingParams
from tests.test_engine import make_engine

PROMPT = [5, 3, 8, 1] * 10


def test_zero_b_adapter_is_identity():
    """B initialized to zero => adapter output == base model output."""
    torch.manual_seed(0)
    eng = make_engine()
    eng.add_lora("zero", rank=8, seed=None)  # B = 0
    base = eng.generate([PROMPT], SamplingParams(max_tokens=5))[0]
    tuned = eng.generate(
        [PROMPT], SamplingParams(max_tokens=5)
    )  # warm path sanity
    req = eng.add_request(PROMPT, SamplingParams(max_tokens=5), lora_name="zero")
    outs = {}
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                outs[o.request_id] = o
    assert outs[req].output_token_ids == base.output_token_ids


def test_nonzero_adapter_changes_output_only_for_its_requests():
    torch.manual_seed(0)
    eng = make_engine()
    eng.add_lora("tuned", rank=8, seed=42)  # random B: real deltas
    base = eng.generate([PROMPT], SamplingParams(max_tokens=6))[0]
    # batch a base request and an adapter request TOGETHER
    r_base = eng.add_request(PROMPT, SamplingParams(max_tokens=6))
    r_lora = eng.add_request(PROMPT, SamplingParams(max_tokens=6), lora_name="tuned")
    outs = {}
    while eng.has_unfinished():
        for o in eng.step():
            if o.finished:
                outs[o.request_id] = o
    assert outs[r_base].output_token_ids == base.output_token_ids
    assert outs[r_lora].output_token_ids != base.output_token_ids


def test_active_loras_listed_for_affinity_scorer():
    eng = make_engine()
    eng.add_lora("sql", rank=4)
    eng.add_lora("chat", rank=4)
    assert eng.active_loras() == ["chat", "sql"]
