"""Engine serving robustness: KV exhaustion queueing, oversized prompts,
cancellation, malformed mesh frames."""
import json
import queue
import time

import pytest

from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
from bee2bee_amd.engine.sampler import SamplingParams


@pytest.fixture(scope="module")
def small_engine():
    # tiny KV pool: 2 seqs of 64 max
    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=64, seed=5)
    yield eng
    eng.shutdown()


def _req(n=4, prompt=(1, 2, 3), max_new=None):
    return GenerationRequest(
        prompt_ids=list(prompt), max_new_tokens=max_new or n,
        sampling=SamplingParams(greedy=True),
    )


def _drain(r, timeout=120):
    while True:
        item = r.out_queue.get(timeout=timeout)
        if not isinstance(item, int):
            break


def test_burst_beyond_batch_queues(small_engine):
    """6 requests into a max_batch=2 engine: all must complete."""
    reqs = [small_engine.submit(_req(4, (i, i + 1))) for i in range(6)]
    for r in reqs:
        _drain(r)
        assert r.error is None and len(r.output_ids) == 4


def test_oversized_prompt_truncated(small_engine):
    r = small_engine.submit(_req(4, tuple(range(1, 200))))
    _drain(r)
    assert r.error is None
    assert len(r.output_ids) == 4


def test_cancellation_stops_generation(small_engine):
    r = _req(4, (5, 6), max_new=10_000)
    small_engine.submit(r)
    # wait for the first token, then cancel
    tok = r.out_queue.get(timeout=60)
    assert isinstance(tok, int)
    r.cancelled = True
    t0 = time.time()
    _drain(r, timeout=60)
    assert time.time() - t0 < 30
    assert len(r.output_ids) < 10_000


def test_empty_prompt_gets_bos(small_engine):
    r = small_engine.submit(_req(3, ()))
    _drain(r)
    assert r.error is None and len(r.output_ids) == 3


def test_malformed_wire_frames_do_not_crash_node():
    import asyncio

    import aiohttp

    from bee2bee_amd.mesh.node import MeshNode

    async def run():
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        async with aiohttp.ClientSession() as session:
            async with session.ws_connect(f"ws://127.0.0.1:{node.port}") as ws:
                await ws.send_str("this is not json")
                await ws.send_str(json.dumps({"no": "type"}))
                await ws.send_str(json.dumps({"type": "gen_request"}))  # no rid
                await ws.send_str(json.dumps({"type": "pong"}))  # no ts
                await ws.send_str(json.dumps({"type": "hello"}))  # no peer_id
                await ws.send_str(json.dumps(
                    {"type": "piece_request", "hash": None, "index": "x"}
                ))
                await asyncio.sleep(0.3)
        # node still serves new connections afterwards
        async with aiohttp.ClientSession() as session:
            async with session.ws_connect(f"ws://127.0.0.1:{node.port}") as ws:
                await ws.send_str(json.dumps({
                    "type": "hello", "peer_id": "p2", "addr": "",
                }))
                msg = await asyncio.wait_for(ws.receive(), timeout=5)
                assert json.loads(msg.data).get("type") == "hello"
        await node.stop()

    asyncio.run(run())


def test_kv_pool_exhaustion_backpressure():
    """More concurrent demand than KV blocks: admission must defer (push
    back to pending) rather than crash, and every request must finish."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    # pool sized for ~2 sequences; submit 6 long ones
    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=64,
                          seed=3, kv_margin_blocks=0)
    try:
        reqs = [
            eng.submit(GenerationRequest(
                prompt_ids=[(i * 7 + j) % 500 for j in range(40)],
                max_new_tokens=10,
                sampling=SamplingParams(greedy=True),
            ))
            for i in range(6)
        ]
        for r in reqs:
            while True:
                x = r.out_queue.get(timeout=120)
                if not isinstance(x, int):
                    break
            assert r.error is None, r.error
            assert len(r.output_ids) == 10
    finally:
        eng.shutdown()


def test_concurrent_submit_thread_safety():
    """submit() from many threads concurrently: the queue-based admission
    must keep every request isolated and complete."""
    import threading

    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=4, max_seq_len=64,
                          seed=3)
    reqs = []
    lock = threading.Lock()

    def submit_some(base):
        for i in range(3):
            r = GenerationRequest(
                prompt_ids=[base + i, base, 7], max_new_tokens=4,
                sampling=SamplingParams(greedy=True),
            )
            eng.submit(r)
            with lock:
                reqs.append(r)

    try:
        threads = [threading.Thread(target=submit_some, args=(b,))
                   for b in (10, 50, 90, 130)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        assert len(reqs) == 12
        for r in reqs:
            while True:
                x = r.out_queue.get(timeout=120)
                if not isinstance(x, int):
                    break
            assert r.error is None
            assert len(r.output_ids) == 4
    finally:
        eng.shutdown()


def test_kv_freed_for_requests_finishing_at_prefill():
    """max_new_tokens=1 requests finish at prefill completion and must
    release their KV blocks (regression: they skipped retirement and the
    pool drained)."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=96,
                          seed=2)
    try:
        base = eng.kv.free_blocks
        for round_ in range(6):  # 6 rounds x 2 blocks would exhaust a
            req = GenerationRequest(  # leaky pool long before this
                prompt_ids=list(range(4, 40)), max_new_tokens=1,
                sampling=SamplingParams(greedy=True))
            eng.submit(req)
            while True:
                item = req.out_queue.get(timeout=60)
                if not isinstance(item, int):
                    break
            assert req.error is None, req.error
            assert len(req.output_ids) == 1
        for _ in range(100):
            if eng.kv.free_blocks == base:
                break
            import time as _t
            _t.sleep(0.02)
        assert eng.kv.free_blocks == base, (eng.kv.free_blocks, base)
    finally:
        eng.shutdown()


def test_cancel_during_chunked_prefill():
    """Cancelling a request mid-chunked-prefill must free its KV and not
    disturb other requests."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=480,
                          seed=4, max_prefill_tokens=32)
    try:
        base = eng.kv.free_blocks
        big = GenerationRequest(prompt_ids=list(range(4, 404)),
                                max_new_tokens=8,
                                sampling=SamplingParams(greedy=True))
        small = GenerationRequest(prompt_ids=[7, 8, 9],
                                  max_new_tokens=4,
                                  sampling=SamplingParams(greedy=True))
        eng.submit(big)
        eng.submit(small)
        big.cancelled = True  # cancel while the 400-token prompt chunks in
        while True:
            item = small.out_queue.get(timeout=60)
            if not isinstance(item, int):
                break
        assert small.error is None and len(small.output_ids) == 4
        # big either errored/cancelled or finished; wait for it to settle
        import time as _t
        for _ in range(200):
            if big.done_ts is not None or big.error is not None:
                break
            _t.sleep(0.02)
        for _ in range(200):
            if eng.kv.free_blocks == base:
                break
            _t.sleep(0.02)
        assert eng.kv.free_blocks == base, (eng.kv.free_blocks, base)
    finally:
        eng.shutdown()


def test_randomized_workload_soak():
    """Property test: a burst of randomized requests (lengths, sampling
    params, penalties, cancels, stop tokens) completes with no errors and
    the KV pool returns exactly to baseline."""
    import random

    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    rng = random.Random(42)
    eng = InferenceEngine("tiny", device="cpu", max_batch=4, max_seq_len=256,
                          seed=7, max_prefill_tokens=64)
    try:
        base = eng.kv.free_blocks
        reqs = []
        for i in range(24):
            n = rng.randrange(1, 180)
            sp = rng.choice([
                SamplingParams(greedy=True),
                SamplingParams(greedy=True, repetition_penalty=1.15),
                SamplingParams(temperature=0.8, top_p=0.9, top_k=20),
                SamplingParams(temperature=1.2, repetition_penalty=1.3),
            ])
            req = GenerationRequest(
                prompt_ids=[rng.randrange(4, 500) for _ in range(n)],
                max_new_tokens=rng.randrange(1, 12),
                sampling=sp,
                stop_token_ids=(rng.randrange(4, 500),) if rng.random() < 0.3
                else (),
            )
            eng.submit(req)
            if rng.random() < 0.2:
                req.cancelled = True
            reqs.append(req)
        for req in reqs:
            while True:
                item = req.out_queue.get(timeout=120)
                if not isinstance(item, int):
                    break
            assert req.error is None, req.error
            assert len(req.output_ids) <= req.max_new_tokens
        import time as _t
        for _ in range(300):
            if eng.kv.free_blocks == base and not eng._active \
                    and not eng._prefilling:
                break
            _t.sleep(0.02)
        assert eng.kv.free_blocks == base, (eng.kv.free_blocks, base)
    finally:
        eng.shutdown()


def test_context_boundary_exact_fit():
    """prompt + max_new == max_seq_len must work (off-by-one guard)."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=64,
                          seed=2)
    try:
        prompt = list(range(5, 5 + 48))
        req = GenerationRequest(prompt_ids=prompt, max_new_tokens=16,
                                sampling=SamplingParams(greedy=True))
        eng.submit(req)
        while True:
            item = req.out_queue.get(timeout=60)
            if not isinstance(item, int):
                break
        assert req.error is None, req.error
        assert len(req.output_ids) == 16
    finally:
        eng.shutdown()


def test_admission_is_fifo_when_saturated():
    """With max_batch=1, queued requests complete in submission order."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=1, max_seq_len=64,
                          seed=4)
    try:
        reqs = [GenerationRequest(prompt_ids=[7, 8, 9 + i],
                                  max_new_tokens=4,
                                  sampling=SamplingParams(greedy=True))
                for i in range(4)]
        for r in reqs:
            eng.submit(r)
        for r in reqs:
            while True:
                item = r.out_queue.get(timeout=60)
                if not isinstance(item, int):
                    break
            assert r.error is None
        done_order = sorted(range(4), key=lambda i: reqs[i].done_ts)
        assert done_order == [0, 1, 2, 3], done_order
    finally:
        eng.shutdown()


def test_oversized_prompt_rejected_typed():
    """A prompt that cannot fit even alone must error, not hang."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=32,
                          seed=6)
    try:
        req = GenerationRequest(prompt_ids=list(range(5, 5 + 64)),
                                max_new_tokens=8,
                                sampling=SamplingParams(greedy=True))
        eng.submit(req)
        deadline = 60
        import time as _t

        t0 = _t.time()
        while _t.time() - t0 < deadline:
            item = req.out_queue.get(timeout=deadline)
            if not isinstance(item, int):
                break
        assert req.error is not None or len(req.output_ids) <= 32
    finally:
        eng.shutdown()


def test_prefill_completion_does_not_clobber_decode_inputs():
    """Deterministic regression for the _last_sampled clobber: request B
    finishes AT prefill completion (max_new=1) between two decode steps of
    request A (membership unchanged) — A's next decode input must be A's
    own last token, not B's. Found by the scheduling-invariance property."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    def run(mpt):
        eng = InferenceEngine("tiny", device="cpu", max_batch=4,
                              max_seq_len=128, seed=5,
                              max_prefill_tokens=mpt)
        try:
            reqs = [GenerationRequest(prompt_ids=[6], max_new_tokens=3,
                                      sampling=SamplingParams(greedy=True)),
                    GenerationRequest(prompt_ids=[5] * 16, max_new_tokens=1,
                                      sampling=SamplingParams(greedy=True))]
            for r in reqs:
                eng.submit(r)
            outs = []
            for r in reqs:
                while True:
                    item = r.out_queue.get(timeout=60)
                    if not isinstance(item, int):
                        break
                assert r.error is None, r.error
                outs.append(list(r.output_ids))
            return outs
        finally:
            eng.shutdown()

    # the tiny chunk budget forces B's final chunk + completion between A's
    # decode steps; outputs must equal the unconstrained schedule
    assert run(16) == run(4096)


def test_kv_pressure_queues_and_completes_all():
    """More concurrent requests than the KV pool can hold at once: the
    engine must admit in waves (block reservation) and complete every
    request correctly — never deadlock, never error."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    # max_batch 8 but a pool that only fits ~3 active sequences
    eng = InferenceEngine("tiny", device="cpu", max_batch=8, max_seq_len=256,
                          seed=8, kv_margin_blocks=0)
    try:
        total = eng.kv.n_blocks
        reqs = [GenerationRequest(prompt_ids=[5 + i] * 100,
                                  max_new_tokens=6,
                                  sampling=SamplingParams(greedy=True))
                for i in range(10)]
        for r in reqs:
            eng.submit(r)
        for r in reqs:
            while True:
                item = r.out_queue.get(timeout=120)
                if not isinstance(item, int):
                    break
            assert r.error is None, r.error
            assert len(r.output_ids) == 6
        # pool fully restored
        import time as _t

        for _ in range(100):
            if eng.kv.free_blocks == total - 1:  # -1: scratch seq
                break
            _t.sleep(0.05)
        assert eng.kv.free_blocks >= total - 1
    finally:
        eng.shutdown()
