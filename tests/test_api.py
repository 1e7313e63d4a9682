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
