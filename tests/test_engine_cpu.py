"""Engine correctness on CPU (fp32 reference ops): incremental decode must
equal full-context recompute; continuous batching must be request-isolated."""
import queue

import pytest
import torch

from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
from bee2bee_amd.engine.sampler import SamplingParams
from bee2bee_amd.models.spec import PRESETS


@pytest.fixture(scope="module")
def engine():
    eng = InferenceEngine("tiny", device="cpu", max_batch=4, max_seq_len=128, seed=7)
    yield eng
    eng.shutdown()


def _greedy_req(prompt_ids, n):
    return GenerationRequest(
        prompt_ids=prompt_ids,
        max_new_tokens=n,
        sampling=SamplingParams(greedy=True),
    )


def test_greedy_deterministic(engine):
    r1 = engine.submit(_greedy_req([5, 6, 7, 8], 8))
    _drain(r1)
    r2 = engine.submit(_greedy_req([5, 6, 7, 8], 8))
    _drain(r2)
    assert r1.output_ids == r2.output_ids
    assert len(r1.output_ids) == 8


def _drain(req):
    items = []
    while True:
        x = req.out_queue.get(timeout=60)
        if not isinstance(x, int):
            break
        items.append(x)
    return items


def test_incremental_equals_full_context(engine):
    """Tokens decoded one-by-one through the paged KV cache must equal
    greedy decoding recomputed from scratch each step (the KV-cache
    correctness invariant)."""
    prompt = [9, 10, 11, 12, 13]
    req = engine.submit(_greedy_req(list(prompt), 6))
    _drain(req)
    got = req.output_ids

    # recompute: full forward over growing context, argmax each step
    from bee2bee_amd.engine.kv import PagedKV
    from bee2bee_amd.engine.runner import Runner

    spec = engine.spec
    ids = list(prompt)
    expect = []
    for _ in range(6):
        kv = PagedKV(spec, torch.device("cpu"), torch.float32, n_blocks=16)
        runner = Runner(spec, engine.weights, kv, torch.device("cpu"), torch.float32)
        kv.new_seq(0)
        kv.extend_seq(0, len(ids))
        T = len(ids)
        slots = torch.tensor(kv.slot_mapping(0, range(T)), dtype=torch.int32)
        pos = torch.arange(T, dtype=torch.int32)
        cu = torch.tensor([0, T], dtype=torch.int32)
        hidden = runner.forward_prefill(
            torch.tensor(ids, dtype=torch.int64), pos, slots, cu, T
        )
        logits = runner.lm_head(hidden[-1:])
        nxt = int(torch.argmax(logits[0]))
        expect.append(nxt)
        ids.append(nxt)
    assert got == expect


def test_continuous_batching_isolation(engine):
    """Several concurrent requests must produce the same outputs as solo."""
    solo = []
    for p in ([3, 4, 5], [20, 21], [30, 31, 32, 33]):
        r = engine.submit(_greedy_req(list(p), 5))
        _drain(r)
        solo.append(r.output_ids)
    reqs = [
        engine.submit(_greedy_req(list(p), 5))
        for p in ([3, 4, 5], [20, 21], [30, 31, 32, 33])
    ]
    for r in reqs:
        _drain(r)
    assert [r.output_ids for r in reqs] == solo


def test_stop_token(engine):
    r1 = engine.submit(_greedy_req([5, 6, 7, 8], 8))
    _drain(r1)
    stop = r1.output_ids[2]
    req = GenerationRequest(
        prompt_ids=[5, 6, 7, 8],
        max_new_tokens=8,
        sampling=SamplingParams(greedy=True),
        stop_token_ids=(stop,),
    )
    engine.submit(req)
    _drain(req)
    assert req.output_ids[-1] == stop
    # generation halts at the FIRST occurrence of the stop token
    assert len(req.output_ids) == r1.output_ids.index(stop) + 1


def test_generate_text_roundtrip(engine):
    res = engine.generate_text("hello mesh", max_new_tokens=4, temperature=0.0)
    assert isinstance(res["text"], str)
    assert res["tokens"] == 4
    assert res["latency_ms"] >= 0


def test_streaming_callback(engine):
    deltas = []
    res = engine.generate_text(
        "stream me", max_new_tokens=4, temperature=0.0, on_text=deltas.append
    )
    assert "".join(deltas) == res["text"]


def test_moe_engine_runs():
    eng = InferenceEngine("tiny-moe", device="cpu", max_batch=2, max_seq_len=64, seed=1)
    try:
        req = eng.submit(_greedy_req([1, 2, 3], 4))
        _drain(req)
        assert len(req.output_ids) == 4
        # determinism
        req2 = eng.submit(_greedy_req([1, 2, 3], 4))
        _drain(req2)
        assert req.output_ids == req2.output_ids
    finally:
        eng.shutdown()


def test_kv_blocks_recycled(engine):
    free0 = engine.kv.free_blocks
    r = engine.submit(_greedy_req([1] * 40, 4))
    _drain(r)
    import time

    for _ in range(100):
        if engine.kv.free_blocks == free0:
            break
        time.sleep(0.02)
    assert engine.kv.free_blocks == free0


def test_engine_serves_hf_checkpoint(tmp_path):
    """export-model style checkpoint -> engine(model_path) must reproduce
    the random-init engine exactly (the HF-checkpoint serving path)."""
    import torch

    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(23)
    save_hf(w, str(tmp_path))

    eng_ckpt = InferenceEngine("tiny", device="cpu", model_path=str(tmp_path),
                               max_batch=2, max_seq_len=64)
    eng_rand = InferenceEngine("tiny", device="cpu", max_batch=2,
                               max_seq_len=64, seed=23)
    try:
        r1 = eng_ckpt.submit(_greedy_req([4, 5, 6], 5))
        _drain(r1)
        r2 = eng_rand.submit(_greedy_req([4, 5, 6], 5))
        _drain(r2)
        assert r1.output_ids == r2.output_ids
    finally:
        eng_ckpt.shutdown()
        eng_rand.shutdown()


def test_chunked_prefill_matches_whole():
    """A prompt longer than the prefill budget must prefill in chunks
    against the paged history and produce exactly the whole-prefill output
    (runner chunked path -> ops.attn_prefill_paged)."""
    prompt = [(i * 7 + 3) % 500 for i in range(50)]
    eng_small = InferenceEngine("tiny", device="cpu", max_batch=4,
                                max_seq_len=128, seed=7, max_prefill_tokens=16)
    try:
        r1 = eng_small.submit(_greedy_req(list(prompt), 6))
        _drain(r1)
        r2 = engine_whole = InferenceEngine(
            "tiny", device="cpu", max_batch=4, max_seq_len=128, seed=7
        )
        try:
            rw = engine_whole.submit(_greedy_req(list(prompt), 6))
            _drain(rw)
            assert r1.output_ids == rw.output_ids
            assert len(r1.output_ids) == 6
        finally:
            engine_whole.shutdown()
    finally:
        eng_small.shutdown()


def test_chunked_prefill_interleaves_with_decode():
    """While a long prompt prefills chunk-by-chunk, already-active requests
    keep decoding and all outputs stay solo-equal."""
    long_prompt = [(i * 11 + 1) % 500 for i in range(60)]
    short = [9, 8, 7]
    ref = InferenceEngine("tiny", device="cpu", max_batch=4,
                          max_seq_len=128, seed=7)
    try:
        solo_long = ref.submit(_greedy_req(list(long_prompt), 5))
        _drain(solo_long)
        solo_short = ref.submit(_greedy_req(list(short), 8))
        _drain(solo_short)
    finally:
        ref.shutdown()

    eng = InferenceEngine("tiny", device="cpu", max_batch=4,
                          max_seq_len=128, seed=7, max_prefill_tokens=8)
    try:
        rs = eng.submit(_greedy_req(list(short), 8))
        rl = eng.submit(_greedy_req(list(long_prompt), 5))
        _drain(rs)
        _drain(rl)
        assert rs.output_ids == solo_short.output_ids
        assert rl.output_ids == solo_long.output_ids
    finally:
        eng.shutdown()


def test_active_set_never_exceeds_max_batch():
    """Chunked prefills joining the active set must not push it past
    max_batch (regression: _admit did not count _prefilling, so the
    decode batch overflowed the graph bucket at high concurrency)."""
    eng = InferenceEngine("tiny", device="cpu", max_batch=3, max_seq_len=128,
                          seed=7, max_prefill_tokens=8)
    try:
        reqs = [
            eng.submit(_greedy_req([(i * 13 + j) % 500 for j in range(30)], 6))
            for i in range(8)
        ]
        peak = 0
        import time as _t

        for _ in range(400):
            peak = max(peak, len(eng._active))
            assert len(eng._active) <= 3, f"active set overflowed: {len(eng._active)}"
            if all(r.done_ts is not None for r in reqs):
                break
            _t.sleep(0.01)
        for r in reqs:
            _drain(r)
            assert len(r.output_ids) == 6
    finally:
        eng.shutdown()


def test_request_stage_timing(engine):
    res = engine.generate_text("trace me", max_new_tokens=4, temperature=0.0)
    t = res["timing"]
    assert set(t) == {"queue_ms", "prefill_ms", "decode_ms"}
    assert all(v >= 0 for v in t.values())
    assert t["queue_ms"] + t["prefill_ms"] + t["decode_ms"] <= res["latency_ms"] + 50


def test_bench_prefill_fills_every_prompt():
    """bench_setup must put EVERY prompt token in the cache even when the
    batch exceeds one prefill budget (regression: chunked prefill left all
    but the first max_prefill_tokens prompts empty)."""
    eng = InferenceEngine("tiny", device="cpu", max_batch=8, max_seq_len=64,
                          seed=7, max_prefill_tokens=32)
    try:
        eng.bench_setup(8, 16, steps_budget=4)
        for a in eng._bench_acts:
            assert a.prefilled == 16
            assert len(a.req.output_ids) == 1  # first sampled token emitted
            assert eng.kv.seq_len(a.seq_id) >= 16
        assert eng._prefilling == []
    finally:
        eng.shutdown()


def test_repetition_penalty_seen_mask_equals_history_rebuild():
    """The pooled seen-mask penalty (incremental) must emit exactly the
    tokens of the host-rebuilt-history fallback path."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    def run(exhaust_pool):
        eng = InferenceEngine("tiny", device="cpu", max_batch=2,
                              max_seq_len=96, seed=5)
        try:
            if exhaust_pool:
                # force the fallback branch by leaving no free slots
                eng._pen_assign  # noqa: B018 — ensure attr exists
                eng._pen_pool = None
                eng._pen_free = []
                eng._pen_assign = lambda a: None
            req = GenerationRequest(
                prompt_ids=[7, 8, 9, 7, 8], max_new_tokens=12,
                sampling=SamplingParams(greedy=True, repetition_penalty=1.3),
            )
            eng.submit(req)
            while True:
                item = req.out_queue.get(timeout=60)
                if not isinstance(item, int):
                    break
            assert req.error is None, req.error
            return req.output_ids
        finally:
            eng.shutdown()

    assert run(False) == run(True)


def test_engine_stats_surface():
    """stats() exposes the observability keys the API home endpoint and
    the serve bench rely on (incl. the r2 busy-time counters)."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=96,
                          seed=1)
    try:
        req = GenerationRequest(prompt_ids=[5, 6, 7], max_new_tokens=4,
                                sampling=SamplingParams(greedy=True))
        eng.submit(req)
        while True:
            item = req.out_queue.get(timeout=60)
            if not isinstance(item, int):
                break
        s = eng.stats()
        for key in ("model", "device", "kv_free_blocks", "kv_total_blocks",
                    "tokens_total", "tokens_per_sec_10s", "decode_graphs",
                    "engine_busy_s", "engine_steps", "engine_ms_per_step"):
            assert key in s, key
        assert s["tokens_total"] >= 4
        assert s["engine_steps"] >= 1
        assert s["engine_busy_s"] > 0
    finally:
        eng.shutdown()
