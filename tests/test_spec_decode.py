"""Speculative decoding (prompt-lookup proposer + exact greedy verification):
the accepted tokens must equal plain greedy decode exactly, in fewer forward
steps. Invariance is exact on CPU (fp32 reference ops)."""
import pytest

from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
from bee2bee_amd.engine.sampler import SamplingParams


def _drain(req):
    while True:
        x = req.out_queue.get(timeout=60)
        if not isinstance(x, int):
            break


def _run(spec_decode, prompt, n, seed=7, **kw):
    eng = InferenceEngine(
        "tiny", device="cpu", max_batch=4, max_seq_len=256, seed=seed,
        spec_decode=spec_decode, **kw
    )
    try:
        r = GenerationRequest(
            prompt_ids=list(prompt), max_new_tokens=n,
            sampling=SamplingParams(greedy=True),
        )
        eng.submit(r)
        _drain(r)
        assert r.error is None, r.error
        return r.output_ids, dict(eng.spec_stats), eng
    finally:
        eng.shutdown()


def test_spec_decode_output_invariant_and_faster():
    prompt = [7, 99, 23] * 6
    base, _, _ = _run(False, prompt, 40)
    spec, stats, _ = _run(True, prompt, 40)
    assert spec == base  # exact greedy invariance
    assert len(spec) == 40
    # random tiny models cycle under greedy decode, so the n-gram proposer
    # must accept on this repetitive trajectory and cut the step count
    assert stats["accepted"] > 0
    assert stats["steps"] < 40


def test_spec_decode_batch_and_sampled_mix():
    """Greedy + sampled requests coexist: greedy rows speculate, sampled
    rows take the one-token path, outputs stay request-isolated."""
    eng = InferenceEngine("tiny", device="cpu", max_batch=4, max_seq_len=256,
                          seed=7, spec_decode=True)
    ref = InferenceEngine("tiny", device="cpu", max_batch=4, max_seq_len=256,
                          seed=7)
    try:
        p1, p2 = [7, 99, 23] * 5, [4, 4, 8]
        r_ref = ref.submit(GenerationRequest(
            prompt_ids=list(p1), max_new_tokens=12,
            sampling=SamplingParams(greedy=True)))
        _drain(r_ref)
        g = eng.submit(GenerationRequest(
            prompt_ids=list(p1), max_new_tokens=12,
            sampling=SamplingParams(greedy=True)))
        s = eng.submit(GenerationRequest(
            prompt_ids=list(p2), max_new_tokens=12,
            sampling=SamplingParams(greedy=False, temperature=0.9)))
        _drain(g)
        _drain(s)
        assert g.output_ids == r_ref.output_ids
        assert len(s.output_ids) == 12
    finally:
        eng.shutdown()
        ref.shutdown()


def test_spec_decode_stop_token_mid_acceptance():
    """A stop token inside an accepted run must terminate the request at
    exactly the same index as plain decode."""
    prompt = [7, 99, 23] * 6
    base, _, _ = _run(False, prompt, 40)
    stop = base[10]
    def run_stop(spec):
        eng = InferenceEngine("tiny", device="cpu", max_batch=4,
                              max_seq_len=256, seed=7, spec_decode=spec)
        try:
            r = GenerationRequest(
                prompt_ids=list(prompt), max_new_tokens=40,
                sampling=SamplingParams(greedy=True), stop_token_ids=(stop,),
            )
            eng.submit(r)
            _drain(r)
            return r.output_ids
        finally:
            eng.shutdown()
    assert run_stop(True) == run_stop(False)


def test_spec_decode_max_new_tokens_respected():
    prompt = [7, 99, 23] * 6
    out, _, _ = _run(True, prompt, 7)
    assert len(out) == 7


def test_spec_decode_delegates_without_proposals():
    """A prompt with no self-similarity must delegate to the plain decode
    path (spec mode costs ~nothing when there is nothing to speculate)."""
    prompt = [3, 17, 91, 204, 55]  # no repeated bigram
    base, _, _ = _run(False, prompt, 3)
    spec, stats, _ = _run(True, prompt, 3)
    assert spec == base
    assert stats["delegated"] >= 1
