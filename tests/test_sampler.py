"""Sampling parity with the reference's generation defaults
(/root/reference/bee2bee/hf.py:94-103: temperature 0.7, top_p 0.95,
repetition_penalty 1.15) and the batched repetition-penalty path."""
import torch

from bee2bee_amd.engine.sampler import (
    SamplingParams,
    apply_repetition_penalty,
    sample,
)


def test_from_request_reference_defaults():
    sp = SamplingParams.from_request(None)
    assert sp.temperature == 0.7
    assert sp.top_p == 0.95
    assert sp.repetition_penalty == 1.15
    assert not sp.greedy


def test_from_request_greedy_keeps_penalty():
    sp = SamplingParams.from_request(0.0)
    assert sp.greedy
    assert sp.repetition_penalty == 1.15  # do_sample=False + penalty, like HF


def test_from_request_explicit_knobs():
    sp = SamplingParams.from_request(1.2, top_p=0.5, top_k=7,
                                     repetition_penalty=1.0)
    assert (sp.temperature, sp.top_p, sp.top_k, sp.repetition_penalty) == (
        1.2, 0.5, 7, 1.0)


def _loop_reference(logits, prev_ids, penalty):
    out = logits.clone()
    for b in range(logits.shape[0]):
        ids = prev_ids[b]
        ids = ids[ids >= 0]
        if ids.numel() == 0:
            continue
        vals = out[b, ids]
        out[b, ids] = torch.where(vals > 0, vals / penalty, vals * penalty)
    return out


def test_repetition_penalty_matches_loop_reference():
    g = torch.Generator().manual_seed(3)
    B, V, L = 9, 128, 17
    logits = torch.randn(B, V, generator=g)
    prev = torch.randint(0, V, (B, L), generator=g)
    prev[torch.rand(B, L, generator=g) < 0.3] = -1  # pad
    prev[2] = -1  # a row with no history
    got = apply_repetition_penalty(logits.clone(), prev, 1.15)
    want = _loop_reference(logits, prev, 1.15)
    assert torch.allclose(got, want)


def test_repetition_penalty_pad_does_not_touch_token_zero():
    # token 0 emitted by row 0 only; row 1 is all pad — its token-0 logit
    # must be untouched (the pad index is routed to a sacrificial column)
    logits = torch.ones(2, 4)
    prev = torch.tensor([[0, -1], [-1, -1]])
    out = apply_repetition_penalty(logits.clone(), prev, 2.0)
    assert out[0, 0] == 0.5
    assert out[1, 0] == 1.0


def test_repetition_penalty_noop_fastpaths():
    logits = torch.randn(2, 8)
    assert apply_repetition_penalty(logits, torch.empty(2, 0, dtype=torch.int64), 1.15) is logits
    assert apply_repetition_penalty(logits, torch.zeros(2, 3, dtype=torch.int64), 1.0) is logits


def test_penalized_greedy_avoids_repeats():
    # a logit landscape where unpenalized greedy loops on token 5
    logits = torch.zeros(1, 10)
    logits[0, 5] = 2.0
    logits[0, 3] = 1.9
    prev = torch.tensor([[5]])
    out = apply_repetition_penalty(logits.clone(), prev, 1.15)
    assert int(torch.argmax(out, -1)) == 3


def test_sample_greedy_ignores_generator():
    logits = torch.randn(4, 32)
    sp = SamplingParams(greedy=True)
    assert torch.equal(sample(logits, sp), logits.argmax(-1))


def test_from_request_hostile_values_sanitized():
    """NaN/inf/negative knobs (json.loads accepts NaN/Infinity) fall back
    to defaults instead of poisoning softmax/multinomial downstream."""
    import math

    from bee2bee_amd.engine.sampler import SamplingParams

    p = SamplingParams.from_request(float("nan"), float("inf"),
                                    -3, float("-inf"))
    assert math.isfinite(p.temperature) and p.temperature > 0
    assert 0.0 < p.top_p <= 1.0 or p.top_p == 0.95
    assert p.top_k == 0
    assert p.repetition_penalty == 1.15
    p2 = SamplingParams.from_request(float("inf"))
    assert math.isfinite(p2.temperature)
