"""Programmatic mesh client: connect to a node, list providers, request a
generation (buffered and streaming) over the P2P wire protocol.

Parity with the reference's examples/p2p_request_demo.py client wrapper.

Usage: python examples/mesh_client.py ws://127.0.0.1:4001 [model-name]
"""
import asyncio
import json
import sys

import aiohttp

sys.path.insert(0, ".")

from bee2bee_amd.mesh import wire
from bee2bee_amd.utils import new_id


class MeshClient:
    """Minimal standalone peer speaking the wire protocol (no server side)."""

    def __init__(self) -> None:
        self.peer_id = new_id("client")
        self.providers = {}
        self._pending = {}
        self._chunks = {}
        self._ws = None

    async def connect(self, addr: str) -> None:
        self._session = aiohttp.ClientSession()
        self._ws = await self._session.ws_connect(addr, max_msg_size=wire.MAX_FRAME)
        await self._send(wire.hello(self.peer_id, "", "client", {}, {}))
        self._reader = asyncio.create_task(self._read())
        await asyncio.sleep(0.5)  # let hello/providers arrive

    async def close(self) -> None:
        self._reader.cancel()
        await self._ws.close()
        await self._session.close()

    async def _send(self, obj) -> None:
        await self._ws.send_str(json.dumps(obj))

    async def _read(self) -> None:
        async for msg in self._ws:
            if msg.type != aiohttp.WSMsgType.TEXT:
                break
            data = json.loads(msg.data)
            t = data.get("type")
            if t == wire.HELLO:
                self.providers = data.get("services", {})
            elif t in wire.TERMINAL_TYPES:
                fut = self._pending.pop(data.get("rid"), None)
                if fut and not fut.done():
                    if "error" in data:
                        fut.set_exception(RuntimeError(data["error"]))
                    else:
                        fut.set_result(data)
            elif t == wire.GEN_CHUNK:
                cb = self._chunks.get(data.get("rid"))
                if cb:
                    cb(data.get("text", ""))
            elif t == wire.PING:
                await self._send(wire.pong(data.get("ts")))

    async def generate(self, prompt: str, model=None, max_new_tokens=32,
                       stream=False, on_chunk=None):
        rid = new_id("req")
        fut = asyncio.get_running_loop().create_future()
        self._pending[rid] = fut
        if on_chunk:
            self._chunks[rid] = on_chunk
        await self._send(
            wire.gen_request(rid, prompt, model, max_new_tokens=max_new_tokens,
                             stream=stream)
        )
        try:
            return await asyncio.wait_for(fut, timeout=300)
        finally:
            self._chunks.pop(rid, None)


async def main() -> None:
    addr = sys.argv[1] if len(sys.argv) > 1 else "ws://127.0.0.1:4001"
    model = sys.argv[2] if len(sys.argv) > 2 else None
    client = MeshClient()
    await client.connect(addr)
    print("connected; provider services:", list(client.providers))

    print("\n-- buffered request --")
    res = await client.generate("hello mesh", model=model, max_new_tokens=16)
    print("text:", res.get("text", "")[:200])
    print("tokens:", res.get("tokens"), "latency_ms:", res.get("latency_ms"))

    print("\n-- streaming request --")
    await client.generate(
        "stream this", model=model, max_new_tokens=16, stream=True,
        on_chunk=lambda t: print(t, end="", flush=True),
    )
    print("\ndone")
    await client.close()


if __name__ == "__main__":
    asyncio.run(main())
