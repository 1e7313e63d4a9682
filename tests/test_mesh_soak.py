"""Randomized (seeded) mesh soak: churn + relays + piece fetches + renames
interleaved. The properties under test are liveness ones — no hang, no
unhandled exception, typed errors only, and the surviving mesh still
serves — not specific outputs (operation interleavings are timing
dependent by nature; the SEED pins the operation schedule itself).
"""
import asyncio
import json
import random

import pytest

from bee2bee_amd.mesh.node import MeshNode
from tests.test_mesh import EchoService, _start_node, _wait_for


@pytest.mark.timeout(180)
def test_mesh_soak_mixed_operations():
    rng = random.Random(0x50AC)

    async def run():
        provider = await _start_node()
        await provider.add_service(EchoService(model="soak-model"))
        relay = await _start_node()
        await relay.connect_bootstrap(provider.addr)
        await _wait_for(lambda: provider.peer_id in relay.providers)

        churners = []

        async def op_request_direct():
            res = await relay.request_generation(
                provider.peer_id, "direct", 8, "soak-model", timeout=20)
            assert res["text"] == "echo:direct"

        async def op_request_relayed():
            node = await _start_node()
            churners.append(node)
            await node.connect_bootstrap(relay.addr)
            await _wait_for(lambda: relay.peer_id in node.peers)
            res = await node.request_generation(
                relay.peer_id, "hop", 8, "soak-model", timeout=20)
            assert res["text"] == "echo:hop"

        async def op_join_leave():
            node = await _start_node()
            await node.connect_bootstrap(provider.addr)
            await _wait_for(lambda: provider.peer_id in node.peers)
            await node.stop()

        async def op_dead_model():
            try:
                await relay.request_generation(
                    relay.peer_id, "x", 4, "missing-model", timeout=10)
            except RuntimeError as e:
                assert any(t in str(e) for t in
                           ("no_node_available", "no_local_service",
                            "provider_not_connected", "relay_link_failure",
                            "request_timed_out")), e
            else:
                raise AssertionError("dead model request succeeded")

        async def op_stream():
            chunks = []
            await relay.request_generation(
                provider.peer_id, "a b c d", 8, "soak-model",
                stream=True, on_chunk=chunks.append, timeout=20)
            assert "".join(chunks).strip() == "echo:a b c d"

        ops = [op_request_direct, op_request_relayed, op_join_leave,
               op_dead_model, op_stream]
        # 24 operations, up to 3 concurrent, schedule pinned by the seed
        schedule = [rng.choice(ops) for _ in range(24)]
        i = 0
        while i < len(schedule):
            batch = schedule[i:i + 3]
            i += 3
            results = await asyncio.gather(*(op() for op in batch),
                                           return_exceptions=True)
            for op, r in zip(batch, results):
                assert not isinstance(r, BaseException), (op.__name__, r)

        # the core pair still healthy after the soak
        res = await relay.request_generation(
            provider.peer_id, "final", 8, "soak-model", timeout=20)
        assert res["text"] == "echo:final"

        for node in churners:
            await node.stop()
        await relay.stop()
        await provider.stop()

    asyncio.run(run())
