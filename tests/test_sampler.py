"""Sampler tests: penalties, top-k/top-p, per-request seeds."""

import torch

from fusioninfer_amd.engine.sampler import Sampler
from fusioninfer_amd.engine.sequence import SamplingParams, Sequence


def _seq(**kw):
    s = Sequence("s", [1, 2, 3], SamplingParams(**kw))
    return s


def test_greedy():
    logits = torch.tensor([[0.1, 2.0, 0.3], [5.0, 0.0, 0.0]])
    out = Sampler().sample(logits, [_seq(), _seq()])
    assert out.tolist() == [1, 0]


def test_repetition_penalty_discourages_repeats():
    s = _seq(repetition_penalty=10.0)
    s.output_token_ids = [1]
    # token 1 slightly preferred, but heavily penalized (in prompt+output)
    logits = torch.tensor([[0.9, 1.0, 0.1, 0.0]])
    out = Sampler().sample(logits, [s])
    assert out.item() != 1


def test_frequency_penalty():
    s = _seq(frequency_penalty=0.6)
    s.output_token_ids = [0, 0, 0]
    logits = torch.tensor([[1.0, 0.1, 0.0, 0.0]])
    out = Sampler().sample(logits, [s])
    assert out.item() != 0  # 3 repeats x 0.6 pushes token 0 below token 1


def test_top_k_limits_support():
    torch.manual_seed(0)
    s = _seq(temperature=1.0, top_k=2, seed=7)
    logits = torch.tensor([[10.0, 9.0, -1.0, -2.0]])
    for trial in range(20):
        out = Sampler().sample(logits.clone(), [s])
        assert out.item() in (0, 1)
        s.output_token_ids.append(0)  # shift the per-step seed


def test_per_request_seed_reproducible():
    logits = torch.randn(1, 100)
    a = Sampler().sample(logits.clone(), [_seq(temperature=1.0, seed=123)])
    b = Sampler().sample(logits.clone(), [_seq(temperature=1.0, seed=123)])
    c = Sampler().sample(logits.clone(), [_seq(temperature=1.0, seed=999)])
    assert a.item() == b.item()
    # different seed USUALLY differs; retry logic avoided: just check type
    assert isinstance(c.item(), int)


def test_mixed_greedy_and_sampled_batch():
    torch.manual_seed(0)
    logits = torch.randn(3, 50)
    seqs = [_seq(), _seq(temperature=1.0, seed=5), _seq(temperature=0.7)]
    out = Sampler().sample(logits.clone(), seqs)
    assert out[0].item() == int(logits[0].argmax())


def test_min_p_filters_tail():
    """min_p drops tokens below min_p * max-prob (vLLM surface)."""
    import torch

    from fusioninfer_amd.engine.sampler import Sampler
    from fusioninfer_amd.engine.sequence import SamplingParams, Sequence

    torch.manual_seed(0)
    logits = torch.tensor([[5.0, 4.9, 0.0, -2.0, -50.0]])
    sampler = Sampler()
    picks = set()
    for i in range(50):  # fresh seed per draw (seeded draws are keyed on
        s = Sequence("a", [1, 2],  # seed + output length)
                     SamplingParams(temperature=1.0, min_p=0.5, seed=i))
        picks.add(int(sampler.sample(logits, [s])[0]))
    # only tokens 0 and 1 survive (p1/p0 ~ 0.90; others < 0.5*max)
    assert picks <= {0, 1} and picks
    # min_p=0 keeps the tail reachable at high temperature
    picks2 = set()
    for i in range(200):
        s2 = Sequence("b", [1, 2],
                      SamplingParams(temperature=10.0, min_p=0.0, seed=i))
        picks2.add(int(sampler.sample(logits, [s2])[0]))
    assert len(picks2) >= 3
