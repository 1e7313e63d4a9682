"""NativeEngineService on CPU with the tiny model — the full service
contract (execute / execute_stream JSON-lines) over the real engine."""
import json

import pytest

from bee2bee_amd.services.base import ServiceError
from bee2bee_amd.services.native import NativeEngineService


@pytest.fixture(scope="module")
def svc():
    s = NativeEngineService("tiny", price_per_token=0.002, device="cpu", max_batch=2,
                            max_seq_len=128)
    s.load_sync()
    yield s
    s.engine.shutdown()


def test_metadata(svc):
    meta = svc.get_metadata()
    assert meta["models"] == ["tiny"]
    assert meta["backend"] == "bee2bee-amd-native"
    assert svc.name == "hf"  # wire-compatible service name


def test_execute(svc):
    res = svc.execute({"prompt": "hello", "max_new_tokens": 4, "temperature": 0.0})
    assert isinstance(res["text"], str)
    assert res["tokens"] == 4
    assert res["cost"] == pytest.approx(0.002 * 4)
    assert res["latency_ms"] >= 0


def test_execute_missing_prompt(svc):
    with pytest.raises(ServiceError):
        svc.execute({"max_new_tokens": 4})


def test_execute_stream_jsonlines(svc):
    lines = list(
        svc.execute_stream({"prompt": "abc", "max_new_tokens": 4, "temperature": 0.0})
    )
    parsed = [json.loads(l) for l in lines]
    assert parsed[-1] == {"done": True}
    streamed = "".join(p.get("text", "") for p in parsed[:-1])
    buffered = svc.execute({"prompt": "abc", "max_new_tokens": 4, "temperature": 0.0})
    assert streamed == buffered["text"]


def test_execute_stream_async(svc):
    import asyncio

    async def run():
        lines = []
        async for line in svc.execute_stream_async(
            {"prompt": "abc", "max_new_tokens": 6, "temperature": 0.0}
        ):
            lines.append(line)
        return lines

    lines = asyncio.run(run())
    parsed = [json.loads(l) for l in lines]
    assert parsed[-1] == {"done": True}
    streamed = "".join(p.get("text", "") for p in parsed[:-1])
    buffered = svc.execute({"prompt": "abc", "max_new_tokens": 6, "temperature": 0.0})
    assert streamed == buffered["text"]


def test_stop_strings(svc):
    full = svc.execute({"prompt": "xy", "max_new_tokens": 8, "temperature": 0.0})
    if len(full["text"]) >= 3:
        stopw = full["text"][1:3]
        res = svc.execute(
            {"prompt": "xy", "max_new_tokens": 8, "temperature": 0.0,
             "stop": [stopw]}
        )
        assert stopw not in res["text"]


def test_generate_text_cancel_event():
    """cancel stops generation at the next emitted token — a dead client
    must not keep burning decode steps."""
    import threading

    from bee2bee_amd.engine.engine import InferenceEngine

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=480,
                          seed=3)
    try:
        cancel = threading.Event()
        res = eng.generate_text("a b c", max_new_tokens=400, temperature=0.0,
                                on_text=lambda _t: cancel.set(),
                                cancel=cancel)
        assert 0 < res["tokens"] < 400, res["tokens"]
    finally:
        eng.shutdown()


def test_execute_stream_close_cancels_engine(svc):
    """Closing the JSON-lines generator mid-stream (consumer disconnected)
    cancels the underlying engine request."""
    import time

    before = svc.engine.stats()["tokens_total"]
    gen = svc.execute_stream({"prompt": "x y", "max_new_tokens": 100,
                              "temperature": 0.0})
    first = next(gen)
    assert json.loads(first).get("text") is not None
    gen.close()  # consumer gone

    # the engine request winds down quickly instead of running to 100
    for _ in range(100):
        stats = svc.engine.stats()
        if stats["active_requests"] == 0 and stats["queued_requests"] == 0:
            break
        time.sleep(0.05)
    assert stats["active_requests"] == 0
    assert stats["tokens_total"] - before < 100


def test_admission_control_sheds_load(svc, monkeypatch):
    """BEE2BEE_MAX_QUEUE=0 makes every request shed with the typed
    server_overloaded error (admission control for production serving)."""
    monkeypatch.setenv("BEE2BEE_MAX_QUEUE", "0")
    with pytest.raises(ServiceError, match="server_overloaded"):
        svc.execute({"prompt": "x", "max_new_tokens": 2})
    monkeypatch.delenv("BEE2BEE_MAX_QUEUE")
    assert svc.execute({"prompt": "x", "max_new_tokens": 2})["tokens"] == 2
