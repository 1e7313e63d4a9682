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
