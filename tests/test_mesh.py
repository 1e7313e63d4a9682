"""Loopback mesh tests: real WS connections on 127.0.0.1 (the multi-node test
story the reference lacked — SURVEY.md §4)."""
import asyncio
import json
from typing import Any, Dict, Iterator

import pytest

from bee2bee_amd.mesh.node import MeshNode
from bee2bee_amd.services.base import BaseService


class EchoService(BaseService):
    """Deterministic test backend speaking the service contract."""

    def __init__(self, name="hf", model="echo-model", price=0.001, latency_s=0.0):
        super().__init__(name)
        self.model = model
        self.price = price
        self.latency_s = latency_s
        self.calls = 0

    def get_metadata(self) -> Dict[str, Any]:
        return {"models": [self.model], "price_per_token": self.price}

    def execute(self, params: Dict[str, Any]) -> Dict[str, Any]:
        import time

        self.calls += 1
        if self.latency_s:
            time.sleep(self.latency_s)
        text = f"echo:{params['prompt']}"
        return {
            "text": text,
            "tokens": len(text.split()),
            "latency_ms": 1,
            "price_per_token": self.price,
            "cost": 0.0,
        }

    def execute_stream(self, params: Dict[str, Any]) -> Iterator[str]:
        for word in f"echo:{params['prompt']}".split():
            yield json.dumps({"text": word + " "}) + "\n"
        yield json.dumps({"done": True}) + "\n"


async def _start_node(**kw) -> MeshNode:
    node = MeshNode(host="127.0.0.1", port=0, enable_nat=False, **kw)
    await node.start()
    return node


async def _wait_for(cond, timeout=5.0):
    deadline = asyncio.get_event_loop().time() + timeout
    while not cond():
        if asyncio.get_event_loop().time() > deadline:
            raise TimeoutError("condition not reached")
        await asyncio.sleep(0.02)


def test_two_node_handshake_and_providers():
    async def run():
        a = await _start_node()
        b = await _start_node()
        await b.add_service(EchoService())
        await a.connect_bootstrap(b.addr)
        # hello exchange registers real peer ids and services on both sides
        await _wait_for(lambda: b.peer_id in a.peers and a.peer_id in b.peers)
        await _wait_for(lambda: b.peer_id in a.providers)
        provs = a.list_providers()
        assert len(provs) == 1
        assert provs[0]["models"] == ["echo-model"]
        assert provs[0]["peer_id"] == b.peer_id
        await a.stop()
        await b.stop()

    asyncio.run(run())


def test_request_generation_buffered():
    async def run():
        a = await _start_node()
        b = await _start_node()
        await b.add_service(EchoService())
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.providers)
        res = await a.request_generation(
            b.peer_id, "hello world", 16, "echo-model", timeout=10
        )
        assert res["text"] == "echo:hello world"
        assert "tokens" in res and "latency_ms" in res
        await a.stop()
        await b.stop()

    asyncio.run(run())


def test_request_generation_streaming_chunks():
    async def run():
        a = await _start_node()
        b = await _start_node()
        await b.add_service(EchoService())
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.providers)
        chunks = []
        res = await a.request_generation(
            b.peer_id,
            "one two three",
            16,
            "echo-model",
            stream=True,
            on_chunk=chunks.append,
            timeout=10,
        )
        assert "".join(chunks).strip() == "echo:one two three"
        assert res.get("text") == ""  # chunks carried the payload
        await a.stop()
        await b.stop()

    asyncio.run(run())


def test_self_request_short_circuit():
    async def run():
        a = await _start_node()
        await a.add_service(EchoService())
        res = await a.request_generation(a.peer_id, "self", 8, "echo-model")
        assert res["text"] == "echo:self"
        await a.stop()

    asyncio.run(run())


def test_relay_one_hop():
    """C asks B (no local service); B relays to A (provider)."""

    async def run():
        a = await _start_node()  # provider
        b = await _start_node()  # relay
        c = await _start_node()  # requester
        await a.add_service(EchoService())
        await b.connect_bootstrap(a.addr)
        await _wait_for(lambda: a.peer_id in b.providers)
        await c.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in c.peers)
        # strip c's direct knowledge of a so the request must relay via b
        res = await c.request_generation(b.peer_id, "via relay", 8, "echo-model", timeout=10)
        assert res["text"] == "echo:via relay"
        await c.stop()
        await b.stop()
        await a.stop()

    asyncio.run(run())


def test_no_provider_error():
    async def run():
        a = await _start_node()
        b = await _start_node()
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.peers)
        with pytest.raises(RuntimeError, match="consensus_deadlock"):
            await a.request_generation(b.peer_id, "x", 8, "ghost-model", timeout=10)
        await a.stop()
        await b.stop()

    asyncio.run(run())


def test_pick_provider_price_then_latency():
    async def run():
        a = await _start_node()
        cheap = await _start_node()
        pricey = await _start_node()
        await cheap.add_service(EchoService(model="m", price=0.001))
        await pricey.add_service(EchoService(model="m", price=0.9))
        await a.connect_bootstrap(cheap.addr)
        await a.connect_bootstrap(pricey.addr)
        await _wait_for(
            lambda: cheap.peer_id in a.providers and pricey.peer_id in a.providers
        )
        pid, meta = a.pick_provider("m")
        assert pid == cheap.peer_id
        assert meta["_svc_name"] == "hf"
        await a.stop()
        await cheap.stop()
        await pricey.stop()

    asyncio.run(run())


def test_gossip_peer_list():
    """A connects to B; B knows C; A should auto-connect to C via peer_list."""

    async def run():
        b = await _start_node()
        c = await _start_node()
        await b.connect_bootstrap(c.addr)
        await _wait_for(lambda: c.peer_id in b.peers)
        a = await _start_node()
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: c.peer_id in a.peers, timeout=8)
        await a.stop()
        await b.stop()
        await c.stop()

    asyncio.run(run())


def test_piece_transfer():
    async def run():
        from bee2bee_amd.mesh.pieces import piece_hashes, split_pieces

        a = await _start_node()
        b = await _start_node()
        data = b"weights-shard-" * 1000
        pieces = split_pieces(data, 4096)
        b.share_pieces("h123", pieces)
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.peers)
        got = [
            await a.request_piece(b.peer_id, "h123", i) for i in range(len(pieces))
        ]
        assert b"".join(got) == data
        with pytest.raises(RuntimeError, match="piece_not_found"):
            await a.request_piece(b.peer_id, "nope", 0)
        await a.stop()
        await b.stop()

    asyncio.run(run())


def test_disconnect_reaps_provider():
    async def run():
        a = await _start_node()
        b = await _start_node()
        await b.add_service(EchoService())
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.providers)
        await b.stop()
        await _wait_for(lambda: b.peer_id not in a.peers, timeout=8)
        assert b.peer_id not in a.providers
        await a.stop()

    asyncio.run(run())


def test_example_mesh_client():
    """The standalone example client (examples/mesh_client.py) speaks the
    wire protocol against a live node."""
    import sys
    sys.path.insert(0, ".")
    from examples.mesh_client import MeshClient

    async def run():
        node = await _start_node()
        await node.add_service(EchoService())
        client = MeshClient()
        await client.connect(f"ws://127.0.0.1:{node.port}")
        assert "hf" in client.providers
        res = await client.generate("ping", model="echo-model", max_new_tokens=8)
        assert res["text"] == "echo:ping"
        chunks = []
        await client.generate("a b c", model="echo-model", stream=True,
                              on_chunk=chunks.append)
        assert "".join(chunks).strip() == "echo:a b c"
        await client.close()
        await node.stop()

    asyncio.run(run())


def test_bootstrap_reconnect():
    """If the bootstrap peer restarts, the node re-dials it from the
    monitoring loop (elastic recovery)."""

    async def run():
        b = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await b.start()
        fixed_port = b.port
        a = await _start_node()
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.peers)
        await b.stop()
        await _wait_for(lambda: b.peer_id not in a.peers, timeout=8)
        # restart the bootstrap on the same port
        b2 = MeshNode(host="127.0.0.1", port=fixed_port, enable_nat=False)
        await b2.start()
        await a._reconnect_bootstraps()
        await _wait_for(lambda: b2.peer_id in a.peers, timeout=8)
        await a.stop()
        await b2.stop()

    asyncio.run(run())


def test_reference_quirk_messages_handled():
    """A peer sending reference-shaped frames (task_id ids, max_tokens,
    no-services hello) gets a correct response."""

    async def run():
        import aiohttp

        node = await _start_node()
        await node.add_service(EchoService())
        async with aiohttp.ClientSession() as session:
            async with session.ws_connect(f"ws://127.0.0.1:{node.port}") as ws:
                await ws.send_str(json.dumps(
                    {"type": "hello", "peer_id": "ref-peer", "addr": ""}
                ))
                # drain hello/peer_list/ping
                await ws.send_str(json.dumps({
                    "type": "gen_request", "task_id": "t-1", "svc": "hf",
                    "prompt": "ref style", "max_tokens": 16,
                }))
                deadline = asyncio.get_event_loop().time() + 10
                got = None
                while asyncio.get_event_loop().time() < deadline:
                    msg = await asyncio.wait_for(ws.receive(), timeout=10)
                    data = json.loads(msg.data)
                    if data.get("type") == "ping":
                        await ws.send_str(json.dumps(
                            {"type": "pong", "ts": data.get("ts")}
                        ))
                    if data.get("type") == "gen_success" and data.get("rid") == "t-1":
                        got = data
                        break
                assert got is not None, "no gen_success for task_id request"
                assert got["text"] == "echo:ref style"
        await node.stop()

    asyncio.run(run())


def test_mesh_dht_rendezvous():
    """Two loopback-connected nodes rendezvous into a (virtual) RCCL group
    through the mesh-replicated DHT: both must converge on the same rank
    order and master endpoint without kademlia."""
    import asyncio

    from bee2bee_amd.mesh.node import MeshNode
    from bee2bee_amd.parallel.rendezvous import join_group_mesh

    async def run():
        a = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        b = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await a.start()
        await b.start()
        try:
            await b.connect_bootstrap(a.addr)
            await asyncio.sleep(0.3)  # hello handshake settles
            ra, rb = await asyncio.gather(
                join_group_mesh(a, "pp-test", a.peer_id, "127.0.0.1", 29900,
                                world_size=2, timeout_s=10),
                join_group_mesh(b, "pp-test", b.peer_id, "127.0.0.1", 29901,
                                world_size=2, timeout_s=10),
            )
            (rank_a, master_a), (rank_b, master_b) = ra, rb
            assert {rank_a, rank_b} == {0, 1}
            assert master_a == master_b  # both agree on the TCP store
        finally:
            await a.stop()
            await b.stop()

    asyncio.run(run())


def test_dht_set_merge_on_receive():
    """Concurrent rendezvous announces from two peers must MERGE dict
    records, not clobber (the one-hop replication handler)."""
    import asyncio

    from bee2bee_amd.mesh.node import MeshNode

    async def run():
        a = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        b = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await a.start()
        await b.start()
        try:
            await b.connect_bootstrap(a.addr)
            await asyncio.sleep(0.3)
            await a.dht_set("rccl:g", {"peer-a": {"host": "1.1.1.1"}})
            await b.dht_set("rccl:g", {"peer-b": {"host": "2.2.2.2"}})
            for _ in range(100):
                ra = await a.dht.get("rccl:g") or {}
                rb = await b.dht.get("rccl:g") or {}
                if set(ra) == {"peer-a", "peer-b"} == set(rb):
                    break
                await asyncio.sleep(0.02)
            assert set(ra) == {"peer-a", "peer-b"}, ra
            assert set(rb) == {"peer-a", "peer-b"}, rb
        finally:
            await a.stop()
            await b.stop()

    asyncio.run(run())


def test_outbound_only_provider_tunnel():
    """Tunnel-hosting parity (reference notebooks served Colab nodes via
    ngrok/Bore tunnels): a provider with NO inbound reachability dials OUT
    to a public relay and serves through that one established connection —
    requester -> relay -> (inbound ws) -> provider. This is the native
    equivalent of the reference's tunnel capability: the outbound WS is
    the tunnel."""

    async def run():
        relay = await _start_node()          # the only publicly dialable node
        # behind NAT: announces a black-hole TEST-NET address, so gossiped
        # dial attempts at it can never succeed — only its outbound link
        # into the relay can carry traffic
        provider = await _start_node(announce_host="203.0.113.7",
                                     announce_port=9)
        requester = await _start_node()
        await provider.add_service(EchoService())
        # provider initiates the only connection it has (outbound)
        await provider.connect_bootstrap(relay.addr)
        await _wait_for(lambda: provider.peer_id in relay.providers)
        await requester.connect_bootstrap(relay.addr)
        await _wait_for(lambda: relay.peer_id in requester.peers)
        res = await requester.request_generation(
            relay.peer_id, "through the tunnel", 8, "echo-model", timeout=10
        )
        assert res["text"] == "echo:through the tunnel"
        # the relay reached the provider over the provider-initiated ws —
        # no peer holds a live connection AT the provider's (unreachable)
        # announce addr except via that inbound tunnel
        assert provider.addr.startswith("ws://203.0.113.7")
        await requester.stop()
        await provider.stop()
        await relay.stop()

    asyncio.run(run())


def test_node_survives_adversarial_frames():
    """Malformed/hostile frames from a peer must never kill the node:
    non-JSON, wrong-typed fields, unknown types, oversized rids, dht_set
    garbage — the node stays responsive afterwards."""

    async def run():
        import aiohttp

        node = await _start_node()
        await node.add_service(EchoService())
        session = aiohttp.ClientSession()
        try:
            ws = await session.ws_connect(node.addr)
            evil = [
                "this is not json",
                "[]",
                '"just a string"',
                json.dumps({"no_type": 1}),
                json.dumps({"type": 42}),
                json.dumps({"type": "unknown_fancy_type", "x": [1, 2]}),
                json.dumps({"type": "ping", "ts": "not-a-number"}),
                json.dumps({"type": "pong", "ts": {"nested": True}}),
                json.dumps({"type": "gen_result", "rid": "R" * 100000}),
                json.dumps({"type": "gen_chunk"}),  # no rid at all
                json.dumps({"type": "service_announce", "service": None,
                            "meta": "nope"}),
                json.dumps({"type": "peer_list", "peers": [123, None, {}]}),
                json.dumps({"type": "dht_set", "key": None, "value": object
                            .__class__.__name__}),
                json.dumps({"type": "piece_request", "hash": True,
                            "index": "NaN"}),
                json.dumps({"type": "hello", "peer_id": {"k": 1},
                            "services": 7}),
            ]
            for frame in evil:
                await ws.send_str(frame)
            await asyncio.sleep(0.3)
            # the node is still alive and serves a real request
            await ws.close()
            client = await session.ws_connect(node.addr)
            rid = "probe-1"
            await client.send_str(json.dumps({
                "type": "gen_request", "rid": rid, "svc": "hf",
                "prompt": "still alive", "max_new_tokens": 4,
            }))
            text = None
            for _ in range(50):
                msg = await asyncio.wait_for(client.receive(), timeout=5)
                if msg.type != 1:  # TEXT
                    continue
                d = json.loads(msg.data)
                if d.get("type") in ("gen_success", "gen_result") and \
                        d.get("rid") == rid and d.get("text"):
                    text = d["text"]
                    break
            assert text == "echo:still alive", text
            await client.close()
        finally:
            await session.close()
            await node.stop()

    asyncio.run(run())


def test_mesh_churn_requests_keep_flowing():
    """Peer churn: providers joining and leaving mid-traffic. Requests to
    live providers keep succeeding; requests to dead ones fail with typed
    errors, never hang."""

    async def run():
        hub = await _start_node()
        requester = await _start_node()
        await requester.connect_bootstrap(hub.addr)
        await _wait_for(lambda: hub.peer_id in requester.peers)

        for round_ in range(3):
            prov = await _start_node()
            await prov.add_service(EchoService(model=f"churn-{round_}"))
            await prov.connect_bootstrap(hub.addr)
            def has_model(name):
                return any(
                    name in m.get("models", [])
                    for svcs in hub.providers.values()
                    for m in svcs.values() if isinstance(m, dict)
                )

            await _wait_for(lambda: has_model(f"churn-{round_}"))
            res = await requester.request_generation(
                hub.peer_id, f"ping {round_}", 8, f"churn-{round_}",
                timeout=10)
            assert res["text"] == f"echo:ping {round_}"
            await prov.stop()  # churn out
            await _wait_for(lambda: not has_model(f"churn-{round_}"),
                            timeout=10)
            # now the model is gone: typed failure, not a hang
            try:
                await requester.request_generation(
                    hub.peer_id, "x", 4, f"churn-{round_}", timeout=5)
                raise AssertionError("expected a typed failure")
            except AssertionError:
                raise
            except Exception as e:
                assert "consensus_deadlock" in str(e) or "no_local" in str(e) \
                    or "timed_out" in str(e), e
        await requester.stop()
        await hub.stop()

    asyncio.run(run())


def test_provider_death_fails_inflight_request_fast():
    """A provider dying MID-REQUEST rejects the requester's future
    immediately with the typed provider_not_connected error — the reference
    (and round-1) behavior was a silent 300 s timeout wait."""

    async def run():
        a = await _start_node()
        b = await _start_node()
        # slow service: the request is guaranteed in-flight when b dies
        await b.add_service(EchoService(latency_s=8.0))
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.providers)

        t0 = asyncio.get_event_loop().time()
        task = asyncio.create_task(
            a.request_generation(b.peer_id, "x", 8, "echo-model", timeout=60)
        )
        await asyncio.sleep(0.3)  # let the gen_request land on b
        await b.stop()
        with pytest.raises(RuntimeError, match="provider_not_connected"):
            await task
        assert asyncio.get_event_loop().time() - t0 < 5.0, "waited out timeout"
        assert not a._pending and not a._pending_ws  # no leaked entries
        await a.stop()

    asyncio.run(run())


def test_concurrent_requests_on_one_link_interleave():
    """Two simultaneous requests over the SAME WS connection execute in
    parallel on the provider (the reader loop is not blocked by a
    long-running gen_request), and pings keep flowing mid-generation."""

    async def run():
        a = await _start_node()
        b = await _start_node()
        await b.add_service(EchoService(latency_s=1.0))
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: b.peer_id in a.providers)

        t0 = asyncio.get_event_loop().time()
        r1, r2 = await asyncio.gather(
            a.request_generation(b.peer_id, "one", 8, "echo-model", timeout=15),
            a.request_generation(b.peer_id, "two", 8, "echo-model", timeout=15),
        )
        elapsed = asyncio.get_event_loop().time() - t0
        assert r1["text"] == "echo:one" and r2["text"] == "echo:two"
        assert elapsed < 1.9, f"requests serialized: {elapsed:.2f}s"
        await a.stop()
        await b.stop()

    asyncio.run(run())


def test_relay_failover_to_next_provider():
    """Relay failover: requester asks R (no local service); R knows two
    providers of the model — the cheaper one is DEAD. The relay must fail
    over to the live provider instead of erroring (the reference fails the
    request on the first dead provider)."""

    async def run():
        relay = await _start_node()
        dead = await _start_node()
        live = await _start_node()
        requester = await _start_node()

        # dead is CHEAPER so the ranking tries it first
        await dead.add_service(EchoService(model="fo-model", price=0.0001))
        await live.add_service(EchoService(model="fo-model", price=0.01))
        await relay.connect_bootstrap(dead.addr)
        await relay.connect_bootstrap(live.addr)
        await _wait_for(lambda: len(relay.providers) == 2)
        await requester.connect_bootstrap(relay.addr)
        await _wait_for(lambda: relay.peer_id in requester.peers)

        dead_pid = dead.peer_id
        await dead.stop()
        await _wait_for(lambda: dead_pid not in relay.peers)
        # provider table may lag the peer table; the relay's attempt to the
        # dead pid raises provider_not_connected and fails over regardless

        res = await requester.request_generation(
            relay.peer_id, "ping", 8, "fo-model", timeout=15)
        assert res["text"] == "echo:ping"
        for n in (requester, relay, live):
            await n.stop()

    asyncio.run(run())


def test_relay_streams_chunks_through():
    """A relayed streaming request forwards gen_chunk frames end-to-end:
    requester -> relay -> provider, chunks come back with the original rid."""

    async def run():
        relay = await _start_node()
        provider = await _start_node()
        requester = await _start_node()
        await provider.add_service(EchoService(model="st-model"))
        await relay.connect_bootstrap(provider.addr)
        await _wait_for(lambda: provider.peer_id in relay.providers)
        await requester.connect_bootstrap(relay.addr)
        await _wait_for(lambda: relay.peer_id in requester.peers)

        chunks = []
        res = await requester.request_generation(
            relay.peer_id, "a b c", 8, "st-model", stream=True,
            on_chunk=chunks.append, timeout=15)
        assert "".join(chunks).strip() == "echo:a b c"
        assert isinstance(res, dict)
        for n in (requester, relay, provider):
            await n.stop()

    asyncio.run(run())


def test_node_survives_seeded_frame_fuzz():
    """500 pseudo-random frames (seeded — deterministic CI): every known
    frame type with randomized/garbage field values, interleaved with raw
    junk. The node must stay alive and serve a real request afterwards."""
    import random

    rng = random.Random(0xB2B)
    TYPES = ["hello", "peer_list", "ping", "pong", "service_announce",
             "gen_request", "gen_chunk", "gen_success", "gen_error",
             "gen_result", "piece_request", "piece_data", "dht_set",
             "mystery"]

    def rand_value(depth=0):
        kinds = ["int", "float", "str", "bool", "none", "list", "dict"]
        k = rng.choice(kinds if depth < 2 else kinds[:5])
        if k == "int":
            return rng.randint(-2**40, 2**40)
        if k == "float":
            return rng.choice([0.0, -1.5, 1e308, float(rng.randint(0, 99))])
        if k == "str":
            return "".join(rng.choice("ab:/{}[]\"'\\é☃ ")
                           for _ in range(rng.randint(0, 24)))
        if k == "bool":
            return rng.choice([True, False])
        if k == "none":
            return None
        if k == "list":
            return [rand_value(depth + 1) for _ in range(rng.randint(0, 4))]
        return {str(rng.randint(0, 9)): rand_value(depth + 1)
                for _ in range(rng.randint(0, 4))}

    def rand_frame():
        if rng.random() < 0.1:
            return "".join(rng.choice("{}[]\",:x") for _ in range(20))
        frame = {"type": rng.choice(TYPES)}
        for key in rng.sample(["rid", "task_id", "peer_id", "addr", "svc",
                               "model", "prompt", "max_new_tokens",
                               "temperature", "stream", "peers", "service",
                               "meta", "ts", "metrics", "hash", "index",
                               "data", "key", "value", "text", "error"],
                              rng.randint(0, 8)):
            frame[key] = rand_value()
        return json.dumps(frame)

    async def run():
        import aiohttp

        node = await _start_node()
        await node.add_service(EchoService())
        session = aiohttp.ClientSession()
        try:
            ws = await session.ws_connect(node.addr)
            for _ in range(500):
                await ws.send_str(rand_frame())
            await asyncio.sleep(0.5)
            await ws.close()

            probe = await session.ws_connect(node.addr)
            await probe.send_str(json.dumps({
                "type": "gen_request", "rid": "fuzz-probe", "svc": "hf",
                "prompt": "alive", "max_new_tokens": 4,
            }))
            text = None
            for _ in range(50):
                msg = await asyncio.wait_for(probe.receive(), timeout=10)
                if msg.type != aiohttp.WSMsgType.TEXT:
                    break
                data = json.loads(msg.data)
                if data.get("type") in ("gen_success", "gen_result"):
                    text = data.get("text")
                    break
            assert text == "echo:alive"
            await probe.close()
        finally:
            await session.close()
            await node.stop()

    asyncio.run(run())


def test_relay_hop_limit_prevents_loops():
    """A gen_request that already relayed once (hops=1) is never relayed
    again — nodes without a local service answer consensus_deadlock instead
    of bouncing the request around the mesh forever."""

    async def run():
        import aiohttp

        a = await _start_node()
        b = await _start_node()
        # b advertises a provider (a itself) for the model but has no local
        # service: a hops=1 frame must NOT be relayed back out
        await a.connect_bootstrap(b.addr)
        await _wait_for(lambda: a.peer_id in b.peers)
        b.providers[a.peer_id] = {"hf": {"models": ["loop-model"],
                                         "price_per_token": 0.001}}

        session = aiohttp.ClientSession()
        try:
            ws = await session.ws_connect(b.addr)
            await ws.send_str(json.dumps({
                "type": "gen_request", "rid": "loop-1", "svc": "hf",
                "model": "loop-model", "prompt": "x", "max_new_tokens": 4,
                "hops": 1,
            }))
            data = None
            for _ in range(50):
                msg = await asyncio.wait_for(ws.receive(), timeout=10)
                if msg.type != aiohttp.WSMsgType.TEXT:
                    break
                data = json.loads(msg.data)
                if data.get("type") in ("gen_result", "gen_error"):
                    break
            assert data is not None and "error" in data
            assert "no_node_available" in data["error"]
            await ws.close()
        finally:
            await session.close()
            await a.stop()
            await b.stop()

    asyncio.run(run())


def test_malformed_gen_request_gets_typed_error():
    """Non-numeric sampling knobs answer bad_request immediately instead of
    silently timing out the requester."""

    async def run():
        import aiohttp

        node = await _start_node()
        await node.add_service(EchoService())
        session = aiohttp.ClientSession()
        try:
            ws = await session.ws_connect(node.addr)
            await ws.send_str(json.dumps({
                "type": "gen_request", "rid": "bad-1", "svc": "hf",
                "prompt": "x", "temperature": "hot",
            }))
            data = None
            for _ in range(20):
                msg = await asyncio.wait_for(ws.receive(), timeout=5)
                if msg.type != aiohttp.WSMsgType.TEXT:
                    break
                data = json.loads(msg.data)
                if data.get("rid") == "bad-1":
                    break
            assert data and "bad_request" in (data.get("error") or "")
            await ws.close()
        finally:
            await session.close()
            await node.stop()

    asyncio.run(run())


def test_hello_rename_reaps_old_provider_rows():
    """A peer re-helloing with a NEW peer_id on the same connection must not
    leave provider rows under the old id (they would route requests at a
    dead identity)."""

    async def run():
        import aiohttp

        node = await _start_node()
        session = aiohttp.ClientSession()
        try:
            ws = await session.ws_connect(node.addr)
            for pid in ("id-one", "id-two"):
                await ws.send_str(json.dumps({
                    "type": "hello", "peer_id": pid, "addr": "",
                    "region": "T", "metrics": {},
                    "services": {"hf": {"models": ["ren-model"],
                                        "price_per_token": 0.001}},
                }))
                await asyncio.sleep(0.2)
            assert "id-one" not in node.peers
            assert "id-one" not in node.providers
            assert "id-two" in node.peers
            assert "ren-model" in node.providers["id-two"]["hf"]["models"]
            await ws.close()
        finally:
            await session.close()
            await node.stop()

    asyncio.run(run())
