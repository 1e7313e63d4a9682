"""End-to-end mesh + native engine on MI355X: two WS peers on one GPU, a
generation relayed over the wire protocol into the HIP engine."""
import asyncio
import json

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("requires MI355X", allow_module_level=True)


@pytest.mark.timeout(600)
def test_mesh_two_peers_native_engine():
    from bee2bee_amd.mesh.node import MeshNode
    from bee2bee_amd.services.native import NativeEngineService

    async def run():
        provider = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        client = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await provider.start()
        await client.start()
        svc = NativeEngineService("llama3.2-1b", device="cuda:0", max_batch=4,
                                  max_seq_len=256)
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(None, svc.load_sync)
        await provider.add_service(svc)
        await client.connect_bootstrap(provider.addr)
        for _ in range(200):
            if provider.peer_id in client.providers:
                break
            await asyncio.sleep(0.05)
        res = await client.request_generation(
            provider.peer_id, "hello gpu mesh", 8, "llama3.2-1b", timeout=120
        )
        assert res.get("tokens") == 8
        assert isinstance(res.get("text"), str)
        # streaming over the wire
        chunks = []
        await client.request_generation(
            provider.peer_id, "stream", 8, "llama3.2-1b", stream=True,
            on_chunk=chunks.append, timeout=120,
        )
        assert chunks, "no gen_chunk frames arrived"
        svc.engine.shutdown()
        await client.stop()
        await provider.stop()

    asyncio.run(run())
