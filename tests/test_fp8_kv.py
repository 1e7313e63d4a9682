"""fp8 (OCP e4m3) KV cache — opt-in. CPU: reference dequant path; numerics
stay close to the bf16-KV engine (fp8 has ~6% relative quantization error,
so compare logits loosely and require the pipeline to run end to end)."""
import pytest
import torch

from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
from bee2bee_amd.engine.sampler import SamplingParams
from bee2bee_amd.ops import reference as R


def test_fp8_quant_roundtrip_reference():
    x = torch.randn(8, 4, 16)
    cache = torch.zeros(2, 4, 32, 16, dtype=torch.uint8)
    q = R._kv_quant_like(x, cache)
    back = q.view(torch.float8_e4m3fn).float()
    err = (back - x.float()).abs() / x.float().abs().clamp_min(1e-3)
    assert float(err.median()) < 0.07  # e4m3: 3 mantissa bits


def test_fp8_kv_attn_decode_close():
    torch.manual_seed(4)
    B, nkv, G, hd, bs, L = 2, 2, 2, 16, 32, 60
    W = (L + bs - 1) // bs
    nb = B * W + 1
    bt = (torch.arange(1, B * W + 1).reshape(B, W)).to(torch.int32)
    kc8 = torch.zeros(nb, nkv, bs, hd, dtype=torch.uint8)
    vc8 = torch.zeros(nb, nkv, bs, hd, dtype=torch.uint8)
    kc = torch.zeros(nb, nkv, bs, hd)
    vc = torch.zeros(nb, nkv, bs, hd)
    k = torch.randn(B * L, nkv, hd) * 0.5
    v = torch.randn(B * L, nkv, hd) * 0.5
    slots = torch.cat([
        torch.tensor(
            [int(bt[b, p // bs]) * bs + p % bs for p in range(L)],
            dtype=torch.int32,
        )
        for b in range(B)
    ])
    R.kv_cache_store(k, v, kc, vc, slots)
    R.kv_cache_store(k, v, kc8, vc8, slots)
    q = torch.randn(B, nkv * G, hd)
    lens = torch.tensor([L, L - 7], dtype=torch.int32)
    out16 = R.attn_decode(q, kc, vc, bt, lens, hd**-0.5)
    out8 = R.attn_decode(q, kc8, vc8, bt, lens, hd**-0.5)
    assert torch.isfinite(out8).all()
    # attention outputs are convex combinations of V rows: fp8 noise stays
    # bounded
    assert float((out16 - out8).abs().max()) < 0.15


def test_fp8_kv_engine_runs_cpu():
    eng8 = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=64,
                           seed=7, kv_dtype="fp8")
    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=64,
                          seed=7)
    try:
        def gen(e):
            r = GenerationRequest(prompt_ids=[5, 6, 7, 8], max_new_tokens=8,
                                  sampling=SamplingParams(greedy=True))
            e.submit(r)
            while True:
                x = r.out_queue.get(timeout=60)
                if not isinstance(x, int):
                    break
            assert r.error is None, r.error
            return r.output_ids
        o8 = gen(eng8)
        o16 = gen(eng)
        assert len(o8) == 8
        # trajectories may diverge after quantization noise; require the
        # first token (pure prefill logits, fresh bf16 K/V) to agree
        assert o8[0] == o16[0]
    finally:
        eng8.shutdown()
        eng.shutdown()


def test_fp8_rejects_spec_decode():
    with pytest.raises(ValueError):
        InferenceEngine("tiny", device="cpu", kv_dtype="fp8", spec_decode=True)


def test_fp8_kv_chunked_prefill_cpu():
    """fp8 KV + chunked prefill together (the serving combination): the
    engine must complete through the paged-history reference path with a
    quantized cache."""
    prompt = [(i * 7 + 3) % 500 for i in range(50)]
    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=128,
                          seed=7, kv_dtype="fp8", max_prefill_tokens=16)
    try:
        r = GenerationRequest(prompt_ids=list(prompt), max_new_tokens=6,
                              sampling=SamplingParams(greedy=True))
        eng.submit(r)
        while True:
            x = r.out_queue.get(timeout=60)
            if not isinstance(x, int):
                break
        assert r.error is None, r.error
        assert len(r.output_ids) == 6
    finally:
        eng.shutdown()
