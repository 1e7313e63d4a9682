"""OpenAI-compatible gateway surface (/v1/*) + Prometheus /metrics, and the
sampling-knob passthrough on the P2P request path."""
import json

import pytest
from fastapi.testclient import TestClient

KEY = {"X-API-KEY": "secret-key"}


@pytest.fixture()
def client(monkeypatch):
    monkeypatch.setenv("BEE2BEE_API_KEY", "secret-key")
    monkeypatch.setenv("BEE2BEE_DISABLE_NAT", "1")
    monkeypatch.setenv("BEE2BEE_PORT", "0")
    monkeypatch.setenv("BEE2BEE_HOST", "127.0.0.1")
    monkeypatch.delenv("BEE2BEE_BOOTSTRAP", raising=False)
    from bee2bee_amd.gateway import api as gateway_api

    gateway_api.node = None
    with TestClient(gateway_api.app) as c:
        yield c
    gateway_api.node = None


def _attach_echo(model="oai-model"):
    from bee2bee_amd.gateway import api as gateway_api
    from tests.test_mesh import EchoService

    svc = EchoService(model=model)
    gateway_api.node.local_services[svc.name] = svc
    return svc


def test_models_requires_key_and_lists_local(client):
    assert client.get("/v1/models").status_code == 401
    _attach_echo()
    r = client.get("/v1/models", headers=KEY)
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "list"
    assert any(m["id"] == "oai-model" for m in body["data"])
    assert body["data"][0]["owned_by"] == "bee2bee-amd"


def test_completions_buffered(client):
    _attach_echo()
    r = client.post("/v1/completions", headers=KEY,
                    json={"model": "oai-model", "prompt": "hello world",
                          "max_tokens": 16})
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "text_completion"
    assert body["id"].startswith("cmpl-")
    assert body["choices"][0]["text"] == "echo:hello world"
    assert body["choices"][0]["finish_reason"] == "stop"
    usage = body["usage"]
    assert usage["total_tokens"] == (usage["prompt_tokens"]
                                     + usage["completion_tokens"])


def test_completions_stop_sequence(client):
    _attach_echo()
    r = client.post("/v1/completions", headers=KEY,
                    json={"model": "oai-model", "prompt": "abc XYZ def",
                          "stop": ["XYZ"]})
    assert r.json()["choices"][0]["text"] == "echo:abc "


def test_completions_stream_sse(client):
    _attach_echo()
    with client.stream("POST", "/v1/completions", headers=KEY,
                       json={"model": "oai-model", "prompt": "a b",
                             "stream": True}) as r:
        assert r.status_code == 200
        assert r.headers["content-type"].startswith("text/event-stream")
        lines = [l for l in r.iter_lines() if l]
    assert lines[-1] == "data: [DONE]"
    text = ""
    for line in lines[:-1]:
        ev = json.loads(line[len("data: "):])
        assert ev["object"] == "text_completion"
        text += ev["choices"][0]["text"]
    # EchoService streams word-split with trailing spaces
    assert text.replace(" ", "") == "echo:ab"


def test_chat_completions_transcript_and_shape(client):
    _attach_echo()
    r = client.post("/v1/chat/completions", headers=KEY,
                    json={"model": "oai-model",
                          "messages": [
                              {"role": "system", "content": "be brief"},
                              {"role": "user", "content": "hi"},
                          ]})
    assert r.status_code == 200
    body = r.json()
    assert body["object"] == "chat.completion"
    assert body["id"].startswith("chatcmpl-")
    msg = body["choices"][0]["message"]
    assert msg["role"] == "assistant"
    # the reference web-chat transcript flattening reached the service
    assert "system: be brief" in msg["content"]
    assert "user: hi" in msg["content"]
    assert msg["content"].rstrip().endswith("assistant:")


def test_chat_completions_stream_role_in_first_chunk(client):
    _attach_echo()
    with client.stream("POST", "/v1/chat/completions", headers=KEY,
                       json={"model": "oai-model", "stream": True,
                             "messages": [{"role": "user",
                                           "content": "x"}]}) as r:
        lines = [l for l in r.iter_lines() if l]
    events = [json.loads(l[len("data: "):]) for l in lines[:-1]]
    assert events, "no stream chunks"
    assert events[0]["choices"][0]["delta"].get("role") == "assistant"
    assert all(e["object"] == "chat.completion.chunk" for e in events)
    assert lines[-1] == "data: [DONE]"


def test_rejections(client):
    _attach_echo()
    r = client.post("/v1/completions", headers=KEY,
                    json={"prompt": "x", "n": 2})
    assert r.status_code == 400
    r = client.post("/v1/chat/completions", headers=KEY,
                    json={"messages": []})
    assert r.status_code == 400


def test_metrics_exposition(client):
    _attach_echo()
    client.post("/v1/completions", headers=KEY,
                json={"model": "oai-model", "prompt": "x"})
    r = client.get("/metrics")
    assert r.status_code == 200
    text = r.text
    assert "bee2bee_mesh_peers" in text
    assert "bee2bee_uptime_seconds" in text
    assert 'bee2bee_http_requests_total{' in text
    assert 'path="/v1/completions"' in text


def test_gen_request_sampling_knobs_roundtrip():
    """Knobs placed by gen_request are read back by request_params — the
    wire contract the engine's reference-default sampling relies on."""
    from bee2bee_amd.mesh import wire

    frame = wire.gen_request(
        "r1", "p", "m", max_new_tokens=8, temperature=0.5,
        sampling={"top_p": 0.9, "top_k": 40, "repetition_penalty": 1.1})
    params = wire.request_params(frame)
    assert params["top_p"] == 0.9
    assert params["top_k"] == 40
    assert params["repetition_penalty"] == 1.1
    # absent knobs stay absent (engine applies reference defaults)
    frame2 = wire.gen_request("r2", "p", "m", sampling={"top_p": None})
    params2 = wire.request_params(frame2)
    assert "top_p" not in params2 and "repetition_penalty" not in params2


def test_chat_forwards_sampling_to_service(client):
    """/chat hands the sampling knobs to the executing service's params."""
    from bee2bee_amd.gateway import api as gateway_api
    from tests.test_mesh import EchoService

    class Capture(EchoService):
        def execute(self, params):
            self.last_params = params
            return super().execute(params)

    svc = Capture(model="cap-model")
    gateway_api.node.local_services[svc.name] = svc

    r = client.post("/chat", headers=KEY,
                    json={"prompt": "hi", "model": "cap-model",
                          "temperature": 0.3, "top_p": 0.8,
                          "repetition_penalty": 1.2})
    assert r.json()["status"] == "ok"
    assert svc.last_params["top_p"] == 0.8
    assert svc.last_params["repetition_penalty"] == 1.2


def test_openai_completions_p2p_fallback(client):
    """/v1/completions on a node WITHOUT a local service for the model:
    the request routes over the mesh to a provider peer and comes back in
    OpenAI shape."""
    import asyncio
    import threading
    import time

    from bee2bee_amd.mesh.node import MeshNode
    from tests.test_mesh import EchoService

    # provider node on its own loop/thread (the gateway node lives on the
    # TestClient app loop; mesh wiring goes through the HTTP /connect)
    loop = asyncio.new_event_loop()
    t = threading.Thread(target=loop.run_forever, daemon=True)
    t.start()

    def on_loop(coro, timeout=15):
        return asyncio.run_coroutine_threadsafe(coro, loop).result(timeout)

    provider = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
    on_loop(provider.start())
    on_loop(provider.add_service(EchoService(model="remote-model")))
    try:
        r = client.get("/connect", headers=KEY,
                       params={"addr": provider.addr})
        assert r.json().get("status") == "connected", r.json()
        for _ in range(100):  # service_announce propagation
            rows = client.get("/providers", headers=KEY).json()
            if any("remote-model" in p["models"] for p in rows):
                break
            time.sleep(0.05)
        else:
            raise AssertionError(f"provider never advertised: {rows}")

        r = client.post("/v1/completions", headers=KEY,
                        json={"model": "remote-model", "prompt": "mesh hop"})
        assert r.status_code == 200, r.text
        assert r.json()["choices"][0]["text"] == "echo:mesh hop"
    finally:
        on_loop(provider.stop())
        loop.call_soon_threadsafe(loop.stop)
        t.join(timeout=5)


def test_metrics_engine_gauges_with_native_service(client):
    """With a real engine attached, /metrics exports the engine gauges
    under the exact stats() keys (guards against key drift)."""
    from bee2bee_amd.gateway import api as gateway_api
    from bee2bee_amd.services.native import NativeEngineService

    svc = NativeEngineService("tiny", device="cpu", max_batch=2,
                              max_seq_len=128)
    svc.load_sync()
    try:
        gateway_api.node.local_services["hf"] = svc
        r = client.post("/chat", headers=KEY,
                        json={"prompt": "hi", "model": "tiny",
                              "max_new_tokens": 4, "temperature": 0.0})
        assert r.json()["status"] == "ok"
        text = client.get("/metrics").text
        for g in ("bee2bee_engine_tokens_total",
                  "bee2bee_engine_kv_free_blocks",
                  "bee2bee_engine_engine_ms_per_step"):
            assert g in text, g
        # generated tokens actually counted
        line = next(l for l in text.splitlines()
                    if l.startswith("bee2bee_engine_tokens_total "))
        assert float(line.split()[-1]) >= 4.0
    finally:
        svc.engine.shutdown()


def test_openai_overloaded_returns_429(client, monkeypatch):
    from bee2bee_amd.gateway import api as gateway_api
    from bee2bee_amd.services.native import NativeEngineService

    svc = NativeEngineService("tiny", device="cpu", max_batch=2,
                              max_seq_len=128)
    svc.load_sync()
    try:
        gateway_api.node.local_services["hf"] = svc
        monkeypatch.setenv("BEE2BEE_MAX_QUEUE", "0")
        r = client.post("/v1/completions", headers=KEY,
                        json={"model": "tiny", "prompt": "x"})
        assert r.status_code == 429
        assert "overloaded" in r.json()["detail"]
    finally:
        svc.engine.shutdown()


def test_openai_concurrent_requests_soak(client):
    """20 concurrent /v1 requests (mixed chat/completions, some streamed)
    against a live tiny engine: all succeed, outputs non-empty, engine
    drains back to idle."""
    import concurrent.futures as cf
    import time

    from bee2bee_amd.gateway import api as gateway_api
    from bee2bee_amd.services.native import NativeEngineService

    svc = NativeEngineService("tiny", device="cpu", max_batch=8,
                              max_seq_len=128)
    svc.load_sync()
    try:
        gateway_api.node.local_services["hf"] = svc

        def one(i):
            if i % 3 == 0:
                with client.stream(
                    "POST", "/v1/completions", headers=KEY,
                    json={"model": "tiny", "prompt": f"p{i}",
                          "max_tokens": 6, "stream": True},
                ) as r:
                    assert r.status_code == 200
                    lines = [l for l in r.iter_lines() if l]
                return lines[-1] == "data: [DONE]" and len(lines) > 1
            if i % 3 == 1:
                r = client.post("/v1/chat/completions", headers=KEY,
                                json={"model": "tiny", "max_tokens": 6,
                                      "messages": [{"role": "user",
                                                    "content": f"m{i}"}]})
                return (r.status_code == 200 and
                        r.json()["choices"][0]["message"]["content"] != "")
            r = client.post("/v1/completions", headers=KEY,
                            json={"model": "tiny", "prompt": f"q{i}",
                                  "max_tokens": 6})
            return (r.status_code == 200 and
                    r.json()["usage"]["completion_tokens"] > 0)

        with cf.ThreadPoolExecutor(max_workers=8) as pool:
            results = list(pool.map(one, range(20)))
        assert all(results), results

        for _ in range(100):
            st = svc.engine.stats()
            if st["active_requests"] == 0 and st["queued_requests"] == 0:
                break
            time.sleep(0.05)
        assert st["active_requests"] == 0
    finally:
        svc.engine.shutdown()


def test_completions_batched_prompts(client):
    """OpenAI batch shape: prompt as a LIST yields one indexed choice per
    prompt (concurrent through the engine's continuous batching)."""
    _attach_echo()
    r = client.post("/v1/completions", headers=KEY,
                    json={"model": "oai-model",
                          "prompt": ["alpha", "beta", "gamma"]})
    assert r.status_code == 200
    body = r.json()
    assert [c["index"] for c in body["choices"]] == [0, 1, 2]
    assert [c["text"] for c in body["choices"]] == [
        "echo:alpha", "echo:beta", "echo:gamma"]
    assert body["usage"]["completion_tokens"] > 0

    # streaming rejects multi-prompt; non-string entries rejected
    r = client.post("/v1/completions", headers=KEY,
                    json={"prompt": ["a", "b"], "stream": True})
    assert r.status_code == 400
    r = client.post("/v1/completions", headers=KEY, json={"prompt": [1, 2]})
    assert r.status_code == 400
    r = client.post("/v1/completions", headers=KEY,
                    json={"prompt": "x", "max_tokens": 0})
    assert r.status_code == 400


def test_metrics_spec_decode_gauges(client):
    from bee2bee_amd.gateway import api as gateway_api
    from bee2bee_amd.services.native import NativeEngineService

    import os

    os.environ["BEE2BEE_SPEC_DECODE"] = "1"
    try:
        svc = NativeEngineService("tiny", device="cpu", max_batch=2,
                                  max_seq_len=128)
        svc.load_sync()
    finally:
        del os.environ["BEE2BEE_SPEC_DECODE"]
    try:
        gateway_api.node.local_services["hf"] = svc
        client.post("/chat", headers=KEY,
                    json={"prompt": "a b a b a b", "model": "tiny",
                          "max_new_tokens": 6, "temperature": 0.0,
                          "repetition_penalty": 1.0})
        text = client.get("/metrics").text
        assert "bee2bee_engine_spec_steps" in text
        assert "bee2bee_engine_spec_accepted" in text
    finally:
        svc.engine.shutdown()


def test_open_mode_without_api_key(monkeypatch):
    """With no BEE2BEE_API_KEY configured the /v1 surface is open (dev
    mode), matching the reference's optional-auth behavior."""
    monkeypatch.delenv("BEE2BEE_API_KEY", raising=False)
    monkeypatch.setenv("BEE2BEE_DISABLE_NAT", "1")
    monkeypatch.setenv("BEE2BEE_PORT", "0")
    monkeypatch.setenv("BEE2BEE_HOST", "127.0.0.1")
    monkeypatch.delenv("BEE2BEE_BOOTSTRAP", raising=False)
    from bee2bee_amd.gateway import api as gateway_api

    gateway_api.node = None
    with TestClient(gateway_api.app) as c:
        from tests.test_mesh import EchoService

        svc = EchoService(model="open-model")
        gateway_api.node.local_services[svc.name] = svc
        assert c.get("/v1/models").status_code == 200  # no key needed
        r = c.post("/v1/completions", json={"model": "open-model",
                                            "prompt": "open"})
        assert r.status_code == 200
        assert r.json()["choices"][0]["text"] == "echo:open"
    gateway_api.node = None


def test_metrics_keys_match_engine_stats_contract():
    """Pin the coupling: every engine gauge key in gateway/metrics.py must
    exist in InferenceEngine.stats() output (key drift silently zeroes the
    dashboards — it happened once)."""
    from bee2bee_amd.engine.engine import InferenceEngine
    from bee2bee_amd.gateway.metrics import _ENGINE_STATS

    eng = InferenceEngine("tiny", device="cpu", max_batch=1, max_seq_len=64,
                          seed=1)
    try:
        stats = eng.stats()
        for key, _doc in _ENGINE_STATS:
            assert key in stats, key
            assert isinstance(stats[key], (int, float)), key
    finally:
        eng.shutdown()
