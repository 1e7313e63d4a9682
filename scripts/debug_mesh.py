"""Manual mesh debugging harness (reference scripts/test_connection.py +
debug_p2p_request.py rolled into one): dial a node, complete the hello
handshake, print its services/metrics, optionally send a generation
request and stream the reply.

Usage:
  python scripts/debug_mesh.py ws://host:port
  python scripts/debug_mesh.py ws://host:port --model llama3-8b \
      --prompt "hello" --max-new 16
"""
import argparse
import asyncio
import json
import sys

sys.path.insert(0, ".")

import aiohttp

from bee2bee_amd.mesh import wire
from bee2bee_amd.utils import new_id


async def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("addr", help="node WS address (ws://host:port)")
    ap.add_argument("--model", default=None)
    ap.add_argument("--prompt", default=None)
    ap.add_argument("--max-new", type=int, default=16)
    ap.add_argument("--timeout", type=float, default=30.0)
    args = ap.parse_args()

    peer_id = new_id("debug")
    async with aiohttp.ClientSession() as session:
        try:
            ws = await asyncio.wait_for(
                session.ws_connect(args.addr, max_msg_size=wire.MAX_FRAME),
                timeout=args.timeout)
        except Exception as e:  # noqa: BLE001
            print(f"DIAL FAILED: {e}")
            return 1
        print(f"connected to {args.addr}")
        await ws.send_str(json.dumps(
            wire.hello(peer_id, "", "debug", {}, {})))

        rid = None
        if args.prompt:
            rid = new_id("req")
            frame = wire.gen_request(
                rid, args.prompt, args.model,
                max_new_tokens=args.max_new, stream=True)
            await ws.send_str(json.dumps(frame))
            print(f"sent gen_request rid={rid}")

        deadline = asyncio.get_event_loop().time() + args.timeout
        hello_seen = False
        while asyncio.get_event_loop().time() < deadline:
            try:
                msg = await asyncio.wait_for(ws.receive(), timeout=5)
            except asyncio.TimeoutError:
                if hello_seen and rid is None:
                    break
                continue
            if msg.type != aiohttp.WSMsgType.TEXT:
                print(f"connection closed ({msg.type})")
                break
            data = json.loads(msg.data)
            t = data.get("type")
            if t == wire.HELLO:
                hello_seen = True
                print(f"HELLO from {data.get('peer_id')} "
                      f"region={data.get('region')}")
                for name, meta in (data.get("services") or {}).items():
                    print(f"  service {name}: models={meta.get('models')} "
                          f"price={meta.get('price_per_token')}")
                if rid is None:
                    # nothing else expected beyond gossip; linger briefly
                    deadline = min(deadline,
                                   asyncio.get_event_loop().time() + 2)
            elif t == wire.PING:
                await ws.send_str(json.dumps(wire.pong(data.get("ts"))))
            elif t == wire.GEN_CHUNK and data.get("rid") == rid:
                print(f"  chunk: {data.get('text')!r}")
            elif t in wire.TERMINAL_TYPES and data.get("rid") == rid:
                if data.get("error"):
                    print(f"ERROR: {data['error']}")
                    await ws.close()
                    return 1
                print(f"RESULT: {data.get('text')!r} "
                      f"tokens={data.get('tokens')}")
                await ws.close()
                return 0
            elif t == wire.PEER_LIST:
                print(f"peer_list: {data.get('peers')}")
        await ws.close()
        return 0 if hello_seen else 1


if __name__ == "__main__":
    sys.exit(asyncio.run(main()))
