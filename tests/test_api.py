"""FastAPI gateway tests with a pre-seeded node (reference test shape:
tests/test_api.py there; here the node is seeded to avoid NAT probes)."""
import asyncio

import pytest
from fastapi.testclient import TestClient


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


def test_home_open(client):
    r = client.get("/")
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "ok"
    assert body["peer_id"].startswith("peer-")
    assert body["metrics"]["uptime"] >= 0  # Q4 fix: real uptime


def test_peers_requires_key(client):
    assert client.get("/peers").status_code == 401
    assert client.get("/peers", headers={"X-API-KEY": "wrong"}).status_code == 401
    r = client.get("/peers", headers={"X-API-KEY": "secret-key"})
    assert r.status_code == 200
    assert r.json() == []


def test_providers_shape(client):
    r = client.get("/providers", headers={"X-API-KEY": "secret-key"})
    assert r.status_code == 200
    assert isinstance(r.json(), list)


def test_chat_with_local_service(client):
    from tests.test_mesh import EchoService
    from bee2bee_amd.gateway import api as gateway_api

    svc = EchoService(model="chat-model")
    gateway_api.node.local_services[svc.name] = svc

    r = client.post(
        "/chat",
        headers={"X-API-KEY": "secret-key"},
        json={"prompt": "hi there", "model": "chat-model"},
    )
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "ok"
    assert body["text"] == "echo:hi there"
    assert body["metadata"]["service"] == "hf"


def test_chat_fuzzy_model_match(client):
    from tests.test_mesh import EchoService
    from bee2bee_amd.gateway import api as gateway_api

    svc = EchoService(model="gemma4:31b-cloud")
    gateway_api.node.local_services[svc.name] = svc
    r = client.post(
        "/chat",
        headers={"X-API-KEY": "secret-key"},
        json={"prompt": "x", "model": "gemma4"},
    )
    assert r.json()["status"] == "ok"


def test_generate_alias(client):
    from tests.test_mesh import EchoService
    from bee2bee_amd.gateway import api as gateway_api

    svc = EchoService()
    gateway_api.node.local_services[svc.name] = svc
    r = client.post(
        "/generate",
        headers={"X-API-KEY": "secret-key"},
        json={"prompt": "ab"},
    )
    assert r.json()["status"] == "ok"


def test_chat_streaming(client):
    from tests.test_mesh import EchoService
    from bee2bee_amd.gateway import api as gateway_api

    svc = EchoService()
    gateway_api.node.local_services[svc.name] = svc
    with client.stream(
        "POST",
        "/chat",
        headers={"X-API-KEY": "secret-key"},
        json={"prompt": "one two", "stream": True},
    ) as r:
        assert r.status_code == 200
        lines = [l for l in r.iter_lines() if l]
    import json

    parsed = [json.loads(l) for l in lines]
    assert parsed[-1] == {"done": True}
    text = "".join(p.get("text", "") for p in parsed)
    assert text.strip() == "echo:one two"


def test_chat_no_service_error(client):
    r = client.post(
        "/chat", headers={"X-API-KEY": "secret-key"}, json={"prompt": "x"}
    )
    body = r.json()
    assert body["status"] == "error"


def test_chat_p2p_fallback_to_remote_provider(client):
    """No local service for the model -> the gateway resolves a remote mesh
    provider and relays the request."""
    import asyncio

    from tests.test_mesh import EchoService, _wait_for
    from bee2bee_amd.gateway import api as gateway_api
    from bee2bee_amd.mesh.node import MeshNode

    node = gateway_api.node
    loop = node._tasks[0].get_loop()  # the node's running loop (uvicorn's)

    async def setup():
        provider = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await provider.start()
        await provider.add_service(EchoService(model="remote-model"))
        await node.connect_bootstrap(provider.addr)
        await _wait_for(lambda: provider.peer_id in node.providers)
        return provider

    fut = asyncio.run_coroutine_threadsafe(setup(), loop)
    provider = fut.result(timeout=30)
    try:
        r = client.post(
            "/chat",
            headers={"X-API-KEY": "secret-key"},
            json={"prompt": "over the mesh", "model": "remote-model"},
        )
        body = r.json()
        assert body["status"] == "ok", body
        assert body["text"] == "echo:over the mesh"
        assert body["metadata"]["engine"] == "bee2bee-amd-p2p"
    finally:
        asyncio.run_coroutine_threadsafe(provider.stop(), loop).result(timeout=15)


def test_home_exposes_engine_stats(client):
    """With a native service attached, '/' exposes live engine stats
    (real numbers — the reference fabricated throughput)."""
    import asyncio

    from bee2bee_amd.gateway import api as gateway_api
    from bee2bee_amd.services.native import NativeEngineService

    node = gateway_api.node
    svc = NativeEngineService("tiny", max_batch=2, max_seq_len=64,
                              device="cpu")
    svc.load_sync()
    try:
        loop = node._tasks[0].get_loop()
        fut = asyncio.run_coroutine_threadsafe(node.add_service(svc), loop)
        fut.result(timeout=10)
        r = client.get("/")
        eng = r.json().get("engine")
        assert eng is not None
        assert eng["model"] == "tiny"
        assert eng["active_requests"] >= 0
        assert eng["kv_free_blocks"] <= eng["kv_total_blocks"]
        assert "tokens_per_sec_10s" in eng
    finally:
        if svc.engine is not None:
            svc.engine.shutdown()


def test_connect_endpoint_joins_peer(client):
    """GET /connect dials a live peer and it appears in /peers (reference
    api.py:170 semantics)."""
    import asyncio
    import time as _t

    from bee2bee_amd.gateway import api as gateway_api
    from bee2bee_amd.mesh.node import MeshNode

    node = gateway_api.node
    loop = node._tasks[0].get_loop()

    async def mk_peer():
        p = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await p.start()
        return p

    peer = asyncio.run_coroutine_threadsafe(mk_peer(), loop).result(timeout=15)
    try:
        r = client.get(
            f"/connect?addr={peer.addr}", headers={"X-API-KEY": "secret-key"}
        )
        assert r.status_code == 200, r.text
        deadline = _t.time() + 5
        found = False
        while _t.time() < deadline:
            peers = client.get(
                "/peers", headers={"X-API-KEY": "secret-key"}
            ).json()
            if any(p["peer_id"] == peer.peer_id for p in peers):
                found = True
                break
            _t.sleep(0.05)
        assert found, peers
    finally:
        asyncio.run_coroutine_threadsafe(peer.stop(), loop).result(timeout=15)


def test_generate_forwards_sampling_knobs(client):
    """HTTP sampling knobs reach the service params (reference generation
    surface: temperature/top_p/repetition_penalty)."""
    from bee2bee_amd.gateway import api as gateway_api

    seen = {}

    class Probe:
        name = "hf"
        price_per_token = 0.0

        def get_metadata(self):
            return {"models": ["probe-model"], "price_per_token": 0.0}

        def execute(self, params):
            seen.update(params)
            return {"text": "ok", "tokens": 1, "latency_ms": 1,
                    "price_per_token": 0.0, "cost": 0.0}

    gateway_api.node.local_services["hf"] = Probe()
    r = client.post(
        "/generate",
        headers={"X-API-KEY": "secret-key"},
        json={"prompt": "x", "model": "probe-model", "temperature": 0.9,
              "top_p": 0.5, "top_k": 11, "repetition_penalty": 1.3},
    )
    assert r.status_code == 200 and r.json()["status"] == "ok"
    assert seen["temperature"] == 0.9
    assert seen["top_p"] == 0.5
    assert seen["top_k"] == 11
    assert seen["repetition_penalty"] == 1.3


def test_connect_failure_is_typed(client):
    r = client.get("/connect", headers={"X-API-KEY": "secret-key"},
                   params={"addr": "ws://127.0.0.1:9"})  # nothing listens
    body = r.json()
    assert body["status"] == "error"
    assert "message" in body


def test_chat_unknown_model_p2p_error_shape(client):
    """No local service and no providers: the P2P fallback returns the
    typed error envelope, never a 500."""
    r = client.post("/chat", headers={"X-API-KEY": "secret-key"},
                    json={"prompt": "x", "model": "ghost-model"})
    assert r.status_code == 200
    body = r.json()
    assert body["status"] == "error"
    assert "no_local_service" in body["message"] or \
        "no_node_available" in body["message"]
