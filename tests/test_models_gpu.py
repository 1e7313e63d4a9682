"""Model-family smoke on MI355X: every BASELINE-named family builds with
random weights and decodes greedily on the HIP path."""
import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires MI355X", allow_module_level=True)

from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
from bee2bee_amd.engine.sampler import SamplingParams


def _generate(model, n=4, **kw):
    eng = InferenceEngine(model, device="cuda:0", max_batch=2,
                          max_seq_len=256, seed=3, **kw)
    try:
        req = GenerationRequest(prompt_ids=[1, 2, 3, 4], max_new_tokens=n,
                                sampling=SamplingParams(greedy=True))
        eng.submit(req)
        while True:
            item = req.out_queue.get(timeout=600)
            if not isinstance(item, int):
                break
        assert req.error is None, req.error
        assert len(req.output_ids) == n
        return req.output_ids
    finally:
        eng.shutdown()


def test_zephyr_7b_decodes():
    toks = _generate("zephyr-7b")
    assert all(0 <= t < 32000 for t in toks)


@pytest.mark.timeout(900)
def test_mixtral_8x7b_decodes():
    """46.7B-param MoE (93 GB bf16) on one MI355X — 288 GB HBM holds it."""
    toks = _generate("mixtral-8x7b")
    assert all(0 <= t < 32000 for t in toks)
    # MoE path must be deterministic under greedy
    toks2 = _generate("mixtral-8x7b")
    assert toks == toks2


@pytest.mark.timeout(300)
def test_distilgpt2_decodes_on_gpu():
    """The real GPT-2 architecture (layernorm/gelu/learned-pos kernels) on
    the HIP path, deterministic greedy."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("distilgpt2", device="cuda:0", max_batch=4,
                          max_seq_len=256, seed=3)
    try:
        assert eng.spec.arch == "gpt2"
        outs = []
        for _ in range(2):
            req = GenerationRequest(prompt_ids=[50, 51, 52, 99],
                                    max_new_tokens=8,
                                    sampling=SamplingParams(greedy=True))
            eng.submit(req)
            while True:
                item = req.out_queue.get(timeout=120)
                if not isinstance(item, int):
                    break
            assert req.error is None, req.error
            outs.append(req.output_ids)
        assert outs[0] == outs[1] and len(outs[0]) == 8
    finally:
        eng.shutdown()
