"""L6 web layer e2e: a browser-shaped client registers a node through the
gateway, streams a chat, and the token tally lands in the (mocked)
directory database — the reference flow of app/api/index.js:16-98 +
bridge.js:259-349 + SUPABASE_SCHEMA.sql `messages`."""
import asyncio
import json
import math

import httpx
import pytest
from aiohttp import web as aioweb

from bee2bee_amd.mesh.links import generate_join_link
from bee2bee_amd.mesh.node import MeshNode
from bee2bee_amd.web.bridge import MeshBridge, _http_addr, _ws_addr
from bee2bee_amd.web.gateway import create_app
from bee2bee_amd.web.store import GLOBAL_METRICS_NODE, WebStore

from tests.test_mesh import EchoService, _wait_for


class MockDirectory:
    """In-memory stand-in for the Supabase REST surface the web layer
    touches: messages insert, active_nodes upsert/list, system_stats."""

    def __init__(self):
        self.messages = []
        self.nodes = {}
        self.runner = None
        self.url = None

    async def start(self):
        app = aioweb.Application()
        app.router.add_post("/rest/v1/messages", self._post_messages)
        app.router.add_post("/rest/v1/active_nodes", self._post_nodes)
        app.router.add_get("/rest/v1/active_nodes", self._get_nodes)
        app.router.add_get("/rest/v1/system_stats", self._get_stats)
        self.runner = aioweb.AppRunner(app)
        await self.runner.setup()
        site = aioweb.TCPSite(self.runner, "127.0.0.1", 0)
        await site.start()
        port = site._server.sockets[0].getsockname()[1]  # noqa: SLF001
        self.url = f"http://127.0.0.1:{port}"

    async def stop(self):
        if self.runner:
            await self.runner.cleanup()

    async def _post_messages(self, req):
        self.messages.append(await req.json())
        return aioweb.json_response({}, status=201)

    async def _post_nodes(self, req):
        row = await req.json()
        self.nodes[row.get("peer_id") or row.get("addr")] = row
        return aioweb.json_response({}, status=201)

    async def _get_nodes(self, req):
        return aioweb.json_response(list(self.nodes.values()))

    async def _get_stats(self, req):
        return aioweb.json_response([{
            "total_tokens": sum(m.get("tokens", 0) for m in self.messages),
            "total_chats": sum(1 for m in self.messages
                               if m.get("role") == "user"),
            "total_users": 0,
        }])


def test_addr_conversions():
    assert _ws_addr("1.2.3.4:4001") == "ws://1.2.3.4:4001"
    assert _ws_addr("http://h:1") == "ws://h:1"
    assert _ws_addr("https://h:1") == "wss://h:1"
    assert _ws_addr("ws://h:1") == "ws://h:1"
    assert _http_addr("ws://h:1") == "http://h:1"
    assert _http_addr("h:1") == "http://h:1"
    assert _http_addr("https://h:1") == "https://h:1"


@pytest.mark.timeout(180)
def test_web_gateway_e2e():
    async def run():
        directory = MockDirectory()
        await directory.start()
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        node.local_services["hf"] = EchoService(model="echo-model")
        store = WebStore(base_url=directory.url, key="test-key")
        app = create_app(store=store)
        try:
            async with app.router.lifespan_context(app):
                transport = httpx.ASGITransport(app=app)
                async with httpx.AsyncClient(
                        transport=transport,
                        base_url="http://gw") as client:
                    # ---- register via join link (browser one-click flow)
                    link = generate_join_link(
                        "main", "echo-model", "h" * 8, [node.addr])
                    r = await client.post("/api/p2p/register",
                                          json={"link": link})
                    assert r.status_code == 200, r.text
                    body = r.json()
                    assert body["success"] is True
                    assert body["connected"] is True
                    assert body["mode"] == "fusion-serverless"
                    assert body["node"] == node.addr
                    # registration pushed the node into the directory
                    assert any(row["addr"] == node.addr
                               for row in directory.nodes.values())

                    # missing link -> 400 (reference index.js:18)
                    r = await client.post("/api/p2p/register", json={})
                    assert r.status_code == 400

                    # ---- streamed chat through the WS tunnel
                    r = await client.post(
                        "/api/p2p/generate",
                        json={"prompt": "hello world",
                              "model": "echo-model"})
                    assert r.status_code == 200
                    text = r.text
                    assert "echo:hello" in text and "world" in text

                    # missing prompt -> 400 (index.js:41)
                    r = await client.post("/api/p2p/generate", json={})
                    assert r.status_code == 400

                    # ---- token tally persisted (index.js:66-86)
                    tallies = [m for m in directory.messages
                               if m["content"] == "[Metric Log]"]
                    assert len(tallies) == 1
                    tally = tallies[0]
                    assert tally["node_id"] == GLOBAL_METRICS_NODE
                    assert tally["role"] == "assistant"
                    streamed = text.lstrip(" ")
                    assert tally["tokens"] == math.ceil(len(streamed) / 4)

                    # ---- status reflects the live mesh
                    r = await client.get("/api/p2p/status")
                    s = r.json()
                    assert s["status"] == "active"
                    assert s["connected"] is True
                    assert s["activeNode"] == node.addr
                    assert any(peers for peers in s["mesh"].values())

                    # dynamic discovery action (index.js:153-161)
                    r = await client.post(
                        "/api/p2p/status",
                        json={"action": "discover_peer",
                              "peer": {"addr": node.addr}})
                    assert r.json() == {"status": "discovery_initiated"}

                    # ---- global metrics aggregate the tally
                    r = await client.get("/api/p2p/global_metrics")
                    m = r.json()
                    assert m["tokens"] == tally["tokens"]

                    r = await client.post("/api/p2p/global_metrics",
                                          json={"tokens": 7})
                    assert r.json() == {"success": True}
                    r = await client.get("/api/p2p/global_metrics")
                    assert r.json()["tokens"] == tally["tokens"] + 7

                    r = await client.post("/api/p2p/global_metrics",
                                          json={"tokens": 0})
                    assert r.json() == {"success": False}

                    # ---- dashboard serves a browser page
                    r = await client.get("/")
                    assert r.status_code == 200
                    assert "bee2bee" in r.text
        finally:
            await node.stop()
            await directory.stop()

    asyncio.run(run())


@pytest.mark.timeout(180)
def test_bridge_direct_http_fallback_to_tunnel():
    """targetNode whose HTTP API is unreachable: the bridge must fall back
    to the WS tunnel (reference bridge.js:271-309)."""
    async def run():
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        node.local_services["hf"] = EchoService(model="echo-model")
        store = WebStore(base_url=None, key=None)  # offline mode
        bridge = MeshBridge(seeds=[], store=store, auto_reconnect=False)
        await bridge.start()
        try:
            chunks = []
            # target's HTTP side is the WS port: /generate 404s, so the
            # direct path fails and the tunnel is used
            result = await bridge.request(
                {"prompt": "ping", "model": "echo-model"},
                on_chunk=chunks.append,
                target_node=node.addr,
            )
            assert "echo:ping" in result["text"]
            assert "".join(chunks) == result["text"]
        finally:
            await bridge.stop()
            await node.stop()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_bridge_registry_driven_discovery():
    """With no seeds and no join link, the bridge finds the node through
    the directory (bridge.js syncGlobalMesh -> connect)."""
    async def run():
        directory = MockDirectory()
        await directory.start()
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        node.local_services["hf"] = EchoService(model="echo-model")
        directory.nodes["seed"] = {"peer_id": node.peer_id,
                                   "addr": node.addr,
                                   "models": ["echo-model"],
                                   "region": "Test"}
        store = WebStore(base_url=directory.url, key="k")
        bridge = MeshBridge(seeds=[], store=store, auto_reconnect=False)
        await bridge.start()
        try:
            ok = await bridge.connect()
            assert ok and bridge.connected
            meta = bridge.peer_meta.get(node.addr)
            assert meta and meta["models"] == ["echo-model"]
            stats = bridge.get_stats()
            assert stats["activeNode"] == node.addr
        finally:
            await bridge.stop()
            await node.stop()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_bridge_pool_rotation_drops_dead_seed():
    """Dead seeds are pruned and the next candidate is used
    (bridge.js:83-92)."""
    async def run():
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        store = WebStore(base_url=None, key=None)
        dead = "ws://127.0.0.1:9"  # discard port: connection refused
        bridge = MeshBridge(seeds=[dead, node.addr], store=store,
                            auto_reconnect=False)
        await bridge.start()
        try:
            ok = await bridge.connect()
            assert ok
            assert dead not in bridge.pool
            assert bridge.get_stats()["activeNode"] == node.addr
        finally:
            await bridge.stop()
            await node.stop()

    asyncio.run(run())


@pytest.mark.timeout(120)
def test_bridge_reconnects_after_node_restart():
    """The bridge must survive its active node dying and reattach when a
    node at the same address returns (reference bridge.js:217-222)."""
    async def run():
        from bee2bee_amd.web import bridge as bridge_mod

        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        port = node.port
        node.local_services["hf"] = EchoService(model="echo-model")
        store = WebStore(base_url=None, key=None)
        bridge = MeshBridge(seeds=[f"ws://127.0.0.1:{port}"], store=store,
                            auto_reconnect=True)
        # fast reconnect for the test
        orig_delay = bridge_mod.RECONNECT_DELAY_S
        bridge_mod.RECONNECT_DELAY_S = 0.2
        await bridge.start()
        try:
            assert await bridge.connect()
            r1 = await bridge.request({"prompt": "a", "model": "echo-model"})
            assert "echo:a" in r1["text"]
            await node.stop()
            await asyncio.sleep(0.5)
            assert not bridge.connected
            # node comes back on the SAME port
            node2 = MeshNode(host="127.0.0.1", port=port, enable_nat=False)
            await node2.start()
            node2.local_services["hf"] = EchoService(model="echo-model")
            for _ in range(100):
                if bridge.connected:
                    break
                await asyncio.sleep(0.1)
            # even if the auto-reconnect timer hasn't fired, a request
            # must re-dial and succeed
            r2 = await bridge.request({"prompt": "b", "model": "echo-model"})
            assert "echo:b" in r2["text"]
            await node2.stop()
        finally:
            bridge_mod.RECONNECT_DELAY_S = orig_delay
            await bridge.stop()

    asyncio.run(run())


@pytest.mark.timeout(180)
def test_bridge_direct_http_success_path():
    """targetNode with a LIVE FastAPI /generate: the bridge must stream via
    direct HTTP and never open a tunnel (reference bridge.js:271-289)."""
    import threading

    import uvicorn

    async def run():
        from bee2bee_amd.gateway import api as gateway_api

        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        node.local_services["hf"] = EchoService(model="echo-model")
        gateway_api.node = node
        config = uvicorn.Config(gateway_api.app, host="127.0.0.1",
                                port=18461, log_level="error")
        server = uvicorn.Server(config)
        task = asyncio.ensure_future(server.serve())
        for _ in range(100):
            if server.started:
                break
            await asyncio.sleep(0.05)

        store = WebStore(base_url=None, key=None)
        bridge = MeshBridge(seeds=[], store=store, auto_reconnect=False)
        await bridge.start()
        try:
            chunks = []
            result = await bridge.request(
                {"prompt": "direct hit", "model": "echo-model"},
                on_chunk=chunks.append,
                target_node="127.0.0.1:18461",
            )
            assert "echo:direct" in result["text"]
            assert result["metadata"]["transport"] == "direct-http"
            assert not bridge.connected  # no tunnel was needed
            assert "".join(chunks) == result["text"]
        finally:
            await bridge.stop()
            server.should_exit = True
            await asyncio.sleep(0.2)
            task.cancel()
            await node.stop()

    asyncio.run(run())


def test_web_store_offline_guards():
    """Offline store (no creds): every call degrades to no-op/zeros."""
    async def run():
        store = WebStore(base_url=None, key=None)
        assert not store.enabled
        assert await store.insert_message("n", 5) is False
        assert await store.system_stats() == {"users": 0, "chats": 0,
                                              "tokens": 0}
        assert await store.active_nodes() == []
        assert await store.upsert_node({"peer_id": "x"}) is False
        # zero tokens never writes even when enabled-looking
        store2 = WebStore(base_url="http://127.0.0.1:9", key="k")
        assert await store2.insert_message("n", 0) is False

    asyncio.run(run())


@pytest.mark.timeout(180)
def test_gateway_generate_with_target_node_direct_http():
    """Full L6 path with targetNode: gateway -> bridge -> DIRECT HTTP to
    the node's FastAPI /generate (no WS tunnel), tally persisted under the
    target's id (reference index.js:80: node_id = targetNode)."""
    import threading

    import uvicorn

    async def run():
        from bee2bee_amd.gateway import api as gateway_api

        directory = MockDirectory()
        await directory.start()
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        node.local_services["hf"] = EchoService(model="echo-model")
        gateway_api.node = node
        config = uvicorn.Config(gateway_api.app, host="127.0.0.1",
                                port=18462, log_level="error")
        server = uvicorn.Server(config)
        task = asyncio.ensure_future(server.serve())
        for _ in range(100):
            if server.started:
                break
            await asyncio.sleep(0.05)

        store = WebStore(base_url=directory.url, key="k")
        app = create_app(store=store)
        try:
            async with app.router.lifespan_context(app):
                transport = httpx.ASGITransport(app=app)
                async with httpx.AsyncClient(transport=transport,
                                             base_url="http://gw") as client:
                    r = await client.post(
                        "/api/p2p/generate",
                        json={"task": {"prompt": "hit the target",
                                       "model": "echo-model",
                                       "targetNode": "127.0.0.1:18462"}})
                    assert r.status_code == 200
                    assert "echo:hit" in r.text
            tallies = [m for m in directory.messages
                       if m["content"] == "[Metric Log]"]
            assert len(tallies) == 1
            assert tallies[0]["node_id"] == "127.0.0.1:18462"
        finally:
            server.should_exit = True
            await asyncio.sleep(0.2)
            task.cancel()
            await node.stop()
            await directory.stop()

    asyncio.run(run())


def test_bridge_link_loss_fails_inflight_tunnel_fast():
    """The mesh link dropping mid-tunnel-request resolves the pending
    future immediately (ConnectionError, or the partial stream if chunks
    already arrived) instead of waiting out the 90 s request timeout."""
    class SlowStream(EchoService):
        def execute_stream(self, params):
            import time as _t

            _t.sleep(3.0)  # in the provider's executor thread
            yield from super().execute_stream(params)

    async def run():
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        node.local_services["hf"] = SlowStream(model="echo-model")
        store = WebStore(base_url=None, key=None)
        bridge = MeshBridge(seeds=[node.addr], store=store,
                            auto_reconnect=False)
        await bridge.start()
        try:
            assert await bridge.connect()
            t0 = asyncio.get_event_loop().time()
            task = asyncio.create_task(bridge.request(
                {"prompt": "x", "model": "echo-model"}))
            await _wait_for(lambda: bridge._pending)  # request in flight
            # abrupt link loss: close the transport under the WS (no
            # graceful close handshake — the peer is still mid-request)
            bridge._ws._writer.transport.close()
            with pytest.raises(ConnectionError, match="mesh link lost"):
                await asyncio.wait_for(task, timeout=6.0)
            assert asyncio.get_event_loop().time() - t0 < 6.0
            assert not bridge._pending
        finally:
            await bridge.stop()
            await node.stop()

    asyncio.run(run())


def test_bridge_marks_stale_peers():
    """Directory rows not refreshed within STALE_AFTER_S surface as
    status=stale in stats and the regional mesh (the reference shows every
    directory row as active forever)."""
    from bee2bee_amd.web.store import WebStore

    bridge = MeshBridge(seeds=[], store=WebStore(base_url=None, key=None),
                        auto_reconnect=False)
    bridge._merge_meta("ws://1.2.3.4:1", {"region": "EU"})
    bridge._merge_meta("ws://5.6.7.8:2", {"region": "EU"})
    bridge.peer_meta["ws://1.2.3.4:1"]["last_seen"] -= 10_000  # long ago

    stats = bridge.get_stats()
    by_addr = {p["addr"]: p for p in stats["peers"]}
    assert by_addr["ws://1.2.3.4:1"]["status"] == "stale"
    assert by_addr["ws://5.6.7.8:2"]["status"] == "active"
    assert stats["poolSize"] == 1 and stats["totalPeers"] == 2

    mesh = bridge.get_regional_mesh()
    statuses = {e["addr"]: e["status"] for e in mesh["EU"]}
    assert statuses == {"ws://1.2.3.4:1": "stale", "ws://5.6.7.8:2": "active"}
