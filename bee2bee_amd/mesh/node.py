"""MeshNode — the asyncio WebSocket peer-mesh runtime.

Behavior parity with the reference's P2PNode (bee2bee/p2p_runtime.py:33-840):
WS server + outbound peer connections, hello/peer_list/ping/pong gossip,
provider table with price+latency selection, gen_request local-first with
one-hop swarm relay, health monitoring, Supabase registry sync, and the
`run_mesh_node` orchestrator (reference run_p2p_node :843-954).

Architectural differences (deliberate, MI355X-first):
  * Transport is aiohttp (server and client WS in one lib) instead of the
    `websockets` package; frames stay JSON text, max 32 MiB, so the wire is
    compatible with reference peers and the JS web bridge.
  * Q1 fix — a pending request resolves on gen_result OR gen_success OR
    gen_error (reference only handled gen_result, p2p_runtime.py:660; Python
    peer↔peer buffered requests would otherwise hit the 300 s timeout).
  * Streaming consumption — `request_generation(..., on_chunk=...)` consumes
    gen_chunk frames (the reference had no Python-side consumer; only the JS
    bridge did, app/api/bridge.js:181-187).
  * Blocking service work (model execution) runs in a thread executor; the
    event loop (pings, health checks) never blocks on compute. The GPU
    engine itself runs in its own threads/streams — see engine/engine.py.
  * Data plane: activations between GPU peers never travel over WS; peers
    that co-schedule a pipeline/expert group rendezvous into RCCL
    (parallel/rendezvous.py). The WS mesh stays control-plane only.
"""
from __future__ import annotations

import asyncio
import json
import logging
import time
from typing import Any, Callable, Dict, List, Optional, Tuple

import aiohttp
from aiohttp import web

from ..utils import get_lan_ip, get_system_metrics, new_id
from . import wire
from .links import generate_join_link, parse_join_link
from .registry import RegistryClient

logger = logging.getLogger("bee2bee_amd.mesh")


class Peer:
    """A live WS connection to one peer (either direction)."""

    __slots__ = ("pid", "ws", "addr", "last_pong_ms", "metrics", "health_status", "last_audit")

    def __init__(self, pid: str, ws: Any, addr: Optional[str]) -> None:
        self.pid = pid
        self.ws = ws  # aiohttp WebSocketResponse or ClientWebSocketResponse
        self.addr = addr
        self.last_pong_ms: float = 0.0
        self.metrics: Optional[Dict[str, Any]] = None
        self.health_status: str = "unknown"
        self.last_audit: int = 0

    def view(self) -> Dict[str, Any]:
        return {
            "addr": self.addr,
            "last_pong_ms": self.last_pong_ms,
            "metrics": self.metrics,
            "health_status": self.health_status,
            "last_audit": self.last_audit,
        }


class MeshNode:
    def __init__(
        self,
        host: str = "0.0.0.0",
        port: int = 4001,
        announce_host: Optional[str] = None,
        announce_port: Optional[int] = None,
        entrypoint_url: Optional[str] = None,
        region: str = "Auto",
        enable_nat: bool = True,
    ) -> None:
        self.host = host
        self.port = port
        self.announce_host = announce_host
        self.announce_port = announce_port
        self.peer_id = new_id("peer")
        self.region = region
        self.registry = RegistryClient(entrypoint_url=entrypoint_url)
        self.enable_nat = enable_nat

        self.addr: str = ""
        self.public_host: Optional[str] = None
        self.external_port: Optional[int] = None
        self.api_port: Optional[int] = None
        self.api_host: Optional[str] = None
        self.start_time: float = time.time()

        # state
        self.peers: Dict[str, Peer] = {}
        # mesh-replicated KV (rendezvous records etc.): local DHT store whose
        # writes broadcast one hop over the control plane (wire.DHT_SET)
        from .dht import DHTNode

        self.dht = DHTNode()
        self.local_services: Dict[str, Any] = {}  # name -> BaseService
        self.providers: Dict[str, Dict[str, Any]] = {}  # pid -> {svc: meta, _latency}
        self.pieces: Dict[str, Dict[str, Any]] = {}

        self._lock = asyncio.Lock()
        self._bootstrap_addrs: List[str] = []  # reconnect targets
        self._pending: Dict[str, asyncio.Future] = {}
        self._pending_ws: Dict[str, Any] = {}  # rid -> ws it was sent over
        self._req_tasks: set = set()  # in-flight gen/piece request handlers
        # short-lived tasks (per-connection readers, gossip dials) live in a
        # self-pruning set; _tasks keeps only the persistent loops — a
        # long-running node under churn must not accumulate task objects
        self._chunk_cbs: Dict[str, Callable[[str], None]] = {}
        self._running = False
        self._monitor_active = False
        self._runner: Optional[web.AppRunner] = None
        self._site: Optional[web.TCPSite] = None
        self._session: Optional[aiohttp.ClientSession] = None
        self._tasks: List[asyncio.Task] = []

    # ------------------------------------------------------------------ life

    async def start(self) -> None:
        logger.info("starting mesh node on %s:%s", self.host, self.port)
        app = web.Application()
        app.router.add_get("/", self._ws_endpoint)
        app.router.add_get("/ws", self._ws_endpoint)
        self._runner = web.AppRunner(app)
        await self._runner.setup()
        # short shutdown grace: don't hang stop() on live WS handlers
        self._site = web.TCPSite(
            self._runner, self.host, self.port, shutdown_timeout=2.0
        )
        await self._site.start()
        self._running = True

        # resolve OS-assigned port
        server = self._site._server  # noqa: SLF001 — aiohttp exposes no accessor
        if self.port == 0 and server is not None and server.sockets:
            self.port = server.sockets[0].getsockname()[1]

        self._session = aiohttp.ClientSession()
        await self.dht.start()

        # announce address resolution: explicit > NAT/STUN discovery > LAN IP
        display_host = self.announce_host
        if not display_host:
            display_host = get_lan_ip() if self.host == "0.0.0.0" else self.host
            if self.enable_nat and self.host == "0.0.0.0":
                display_host = await self._try_nat(display_host)
        display_port = self.announce_port or self.external_port or self.port
        self.addr = f"ws://{display_host}:{display_port}"
        self.public_host = display_host

        self._monitor_active = True
        self._tasks.append(asyncio.create_task(self._monitor_loop(wire.PING_INTERVAL)))
        logger.info("mesh node %s listening at %s", self.peer_id, self.addr)

    async def _try_nat(self, fallback_host: str) -> str:
        """UPnP → NAT-PMP → PCP → STUN discovery chain (mesh/nat.py)."""
        try:
            from .nat import auto_port_forward

            res = await auto_port_forward(self.port, "TCP")
            if res and res.success and res.external_ip:
                if res.external_port and res.external_port != self.port:
                    self.external_port = res.external_port
                return res.external_ip
        except Exception as e:  # NAT failure is never fatal
            logger.debug("NAT traversal failed: %s", e)
        try:
            from .stun import try_stun

            stun_res = await try_stun()
            if stun_res:
                ip, port = stun_res
                if port and port != self.port:
                    self.external_port = port
                return ip
        except Exception as e:
            logger.debug("STUN failed: %s", e)
        return fallback_host

    async def stop(self) -> None:
        self._running = False
        self._monitor_active = False
        # fail in-flight requests immediately instead of letting callers
        # wait out the 300 s request timeout
        for fut in list(self._pending.values()):
            if not fut.done():
                fut.set_exception(RuntimeError("node stopped"))
        self._pending.clear()
        self._pending_ws.clear()
        for t in list(self._req_tasks):
            t.cancel()
        for t in self._tasks:
            t.cancel()
        async with self._lock:
            for peer in list(self.peers.values()):
                try:
                    await peer.ws.close()
                except Exception:
                    pass
            self.peers.clear()
        if self._session:
            await self._session.close()
        if self._runner:
            await self._runner.cleanup()
        logger.info("mesh node stopped")

    # ------------------------------------------------------------- transport

    async def _ws_endpoint(self, request: web.Request) -> web.WebSocketResponse:
        ws = web.WebSocketResponse(max_msg_size=wire.MAX_FRAME)
        await ws.prepare(request)
        logger.info("inbound connection from %s", request.remote)
        await self._reader(ws)
        return ws

    async def connect_bootstrap(self, link_or_addr: str) -> None:
        if any(link_or_addr.startswith(p + "://") for p in ("p2pnet", "coithub", "coithub.org")):
            addrs = parse_join_link(link_or_addr).get("bootstrap", [])
        else:
            addrs = [link_or_addr]
        for addr in addrs:
            if addr not in self._bootstrap_addrs:
                self._bootstrap_addrs.append(addr)
        for addr in addrs:
            try:
                await self._connect_peer(addr)
                return
            except Exception as e:
                logger.warning("bootstrap connect failed %s: %s", addr, e)
        logger.error("all bootstrap connections failed (will keep retrying)")

    async def _connect_peer(self, addr: str) -> None:
        if addr == self.addr:
            return
        assert self._session is not None, "node not started"
        try:
            ws = await asyncio.wait_for(
                self._session.ws_connect(
                    addr, max_msg_size=wire.MAX_FRAME, heartbeat=None
                ),
                timeout=10.0,
            )
        except Exception as e:
            # wss→ws fallback for local/dev SSL mismatch (reference :354-361)
            if addr.startswith("wss://"):
                ws = await asyncio.wait_for(
                    self._session.ws_connect(
                        addr.replace("wss://", "ws://"),
                        max_msg_size=wire.MAX_FRAME,
                    ),
                    timeout=10.0,
                )
            else:
                raise IOError(f"could not connect to {addr}: {e}") from e

        pid = new_id("peer")  # temporary until the hello handshake
        async with self._lock:
            self.peers[pid] = Peer(pid, ws, addr)
        await self._send(ws, self._make_hello())
        # prune finished reader tasks so long-lived nodes with many
        # reconnects don't accumulate task objects forever
        self._tasks = [t for t in self._tasks if not t.done()]
        self._track(asyncio.create_task(self._reader(ws)))

    async def _reader(self, ws: Any) -> None:
        try:
            async for msg in ws:
                if msg.type == aiohttp.WSMsgType.TEXT:
                    try:
                        data = json.loads(msg.data)
                    except json.JSONDecodeError:
                        continue
                    try:
                        await self._on_message(ws, data)
                    except Exception:
                        logger.exception("error handling %s", data.get("type"))
                elif msg.type in (aiohttp.WSMsgType.ERROR, aiohttp.WSMsgType.CLOSE):
                    break
        except Exception as e:
            logger.debug("reader ended: %s", e)
        finally:
            await self._on_disconnect(ws)

    async def _on_disconnect(self, ws: Any) -> None:
        async with self._lock:
            for pid, peer in list(self.peers.items()):
                if peer.ws is ws:
                    self.peers.pop(pid, None)
                    self.providers.pop(pid, None)
                    logger.info("peer disconnected: %s", pid)
                    break
        # fail-fast: in-flight requests routed over this link error NOW with
        # a typed message instead of waiting out the 300 s request timeout
        for rid, routed_ws in list(self._pending_ws.items()):
            if routed_ws is ws:
                self._pending_ws.pop(rid, None)
                fut = self._pending.pop(rid, None)
                self._chunk_cbs.pop(rid, None)
                if fut is not None and not fut.done():
                    fut.set_exception(RuntimeError(wire.ERR_NOT_CONNECTED))

    async def _send(self, ws: Any, obj: Dict[str, Any]) -> None:
        try:
            await ws.send_str(json.dumps(obj))
        except Exception as e:
            logger.warning("send failed: %s", e)

    async def _broadcast(self, obj: Dict[str, Any]) -> None:
        async with self._lock:
            targets = [p.ws for p in self.peers.values()]
        if targets:
            await asyncio.gather(
                *(self._send(ws, obj) for ws in targets), return_exceptions=True
            )

    # -------------------------------------------------------------- protocol

    def _make_hello(self) -> Dict[str, Any]:
        services_meta = {
            name: svc.get_metadata() for name, svc in self.local_services.items()
        }
        api_host = self.public_host or self.announce_host or self.host
        return wire.hello(
            peer_id=self.peer_id,
            addr=self.addr,
            region=self.region,
            metrics=get_system_metrics(),
            services=services_meta,
            api_port=self.api_port or 8000,
            api_host=api_host,
            public_ip=self.public_host,
        )

    async def _on_message(self, ws: Any, data: Dict[str, Any]) -> None:
        mtype = data.get("type")
        handler = {
            wire.HELLO: self._handle_hello,
            wire.PEER_LIST: self._handle_peer_list,
            wire.PING: self._handle_ping,
            wire.PONG: self._handle_pong,
            wire.SERVICE_ANNOUNCE: self._handle_service_announce,
            wire.GEN_REQUEST: self._handle_gen_request,
            wire.GEN_RESULT: self._handle_terminal,
            wire.GEN_SUCCESS: self._handle_terminal,  # Q1 fix
            wire.GEN_ERROR: self._handle_terminal,  # Q1 fix
            wire.GEN_CHUNK: self._handle_gen_chunk,
            wire.PIECE_REQUEST: self._handle_piece_request,
            wire.PIECE_DATA: self._handle_piece_data,
            wire.DHT_SET: self._handle_dht_set,
        }.get(mtype)
        if handler is None:
            logger.warning("unknown message type: %s", mtype)
            return
        if mtype in (wire.GEN_REQUEST, wire.PIECE_REQUEST):
            # long-running handlers must not block this connection's read
            # loop: a provider mid-generation still answers pings and serves
            # concurrent requests on the same link (chunks interleave by rid)
            self._track(asyncio.create_task(self._run_request_handler(
                handler, ws, data)))
            return
        await handler(ws, data)

    def _track(self, task: "asyncio.Task") -> "asyncio.Task":
        self._req_tasks.add(task)
        task.add_done_callback(self._req_tasks.discard)
        return task

    async def _run_request_handler(self, handler, ws: Any,
                                   data: Dict[str, Any]) -> None:
        try:
            await handler(ws, data)
        except Exception:
            logger.exception("error handling %s", data.get("type"))

    async def _handle_hello(self, ws: Any, data: Dict[str, Any]) -> None:
        pid = data.get("peer_id")
        addr = data.get("addr")
        if not pid:
            return
        first_contact = False
        async with self._lock:
            old_pid = next(
                (p for p, peer in self.peers.items() if peer.ws is ws), None
            )
            if old_pid and old_pid != pid:
                self.peers.pop(old_pid)
                # the provider rows travel with the connection's identity:
                # leaving them under the old pid would route requests at a
                # peer id that no longer answers
                self.providers.pop(old_pid, None)
            if pid not in self.peers:
                first_contact = True
            existing = self.peers.get(pid)
            peer = Peer(pid, ws, addr)
            if existing is not None:
                peer.metrics = existing.metrics
            self.peers[pid] = peer
            svcs = data.get("services") or {}
            if svcs:
                self.providers[pid] = dict(svcs)
        if first_contact:
            await self._send(ws, self._make_hello())
        peer_addrs = [p.addr for p in self.peers.values() if p.addr]
        await self._send(ws, wire.peer_list(peer_addrs))
        await self._send(ws, wire.ping())

    async def _handle_peer_list(self, ws: Any, data: Dict[str, Any]) -> None:
        for addr in data.get("peers", []):
            if addr == self.addr:
                continue
            if not any(p.addr == addr for p in self.peers.values()):
                self._track(asyncio.create_task(self._safe_connect(addr)))

    async def _safe_connect(self, addr: str) -> None:
        try:
            await self._connect_peer(addr)
        except Exception as e:
            logger.debug("gossip connect to %s failed: %s", addr, e)

    async def _handle_ping(self, ws: Any, data: Dict[str, Any]) -> None:
        metrics = data.get("metrics")
        if metrics:
            async with self._lock:
                for peer in self.peers.values():
                    if peer.ws is ws:
                        peer.metrics = metrics
                        break
        await self._send(ws, wire.pong(data.get("ts")))

    async def _handle_pong(self, ws: Any, data: Dict[str, Any]) -> None:
        ts = data.get("ts") or time.time()
        rtt = (time.time() - float(ts)) * 1000.0
        async with self._lock:
            for pid, peer in self.peers.items():
                if peer.ws is ws:
                    peer.last_pong_ms = rtt
                    if pid in self.providers:
                        self.providers[pid]["_latency"] = rtt
                    break

    async def _handle_service_announce(self, ws: Any, data: Dict[str, Any]) -> None:
        svc = data.get("service")
        meta = data.get("meta", {})
        async with self._lock:
            for pid, peer in self.peers.items():
                if peer.ws is ws:
                    self.providers.setdefault(pid, {})[svc] = meta
                    logger.info("registered service %s from %s", svc, pid)
                    break

    # ------------------------------------------------------------ generation

    async def _handle_gen_request(self, ws: Any, data: Dict[str, Any]) -> None:
        rid = wire.request_id(data)
        svc_name = data.get("svc", "hf")
        model_name = data.get("model")
        try:
            params = wire.request_params(data)
        except (TypeError, ValueError, OverflowError) as e:
            # malformed knobs (non-numeric temperature, ...): answer a typed
            # error NOW — silence would cost the requester the full timeout
            await self._send(ws, wire.gen_error(rid, f"bad_request: {e}"))
            await self._send(ws, wire.gen_result_error(rid, f"bad_request: {e}"))
            return
        loop = asyncio.get_running_loop()

        # 1. local execution primary
        svc = self.local_services.get(svc_name)
        if svc is None and model_name:
            for name, inst in self.local_services.items():
                if model_name in inst.get_metadata().get("models", []):
                    svc, svc_name = inst, name
                    break

        if svc is not None:
            try:
                if data.get("stream") and hasattr(svc, "execute_stream_async"):
                    async for chunk_raw in svc.execute_stream_async(params):
                        try:
                            text = json.loads(chunk_raw).get("text", "")
                        except Exception:
                            text = ""
                        if text:
                            await self._send(ws, wire.gen_chunk(rid, text))
                    closing = {"text": "", "backend": "bee2bee-amd"}
                    await self._send(ws, wire.gen_success(rid, closing))
                    await self._send(ws, wire.gen_result(rid, closing))  # Q1
                elif data.get("stream"):
                    # pump the (sync, blocking) stream generator from a
                    # thread; a requester that disconnects mid-stream closes
                    # the generator, which cancels the engine request —
                    # dead clients must not keep burning decode steps
                    queue: asyncio.Queue = asyncio.Queue()
                    gone = False

                    def _pump() -> None:
                        gen = svc.execute_stream(params)
                        try:
                            for chunk_raw in gen:
                                if gone:
                                    break
                                loop.call_soon_threadsafe(
                                    queue.put_nowait, chunk_raw)
                        finally:
                            gen.close()
                            loop.call_soon_threadsafe(queue.put_nowait, None)

                    pump_fut = loop.run_in_executor(None, _pump)
                    while True:
                        chunk_raw = await queue.get()
                        if chunk_raw is None:
                            break
                        if getattr(ws, "closed", False):
                            gone = True
                            continue  # drain until the pump notices
                        try:
                            text = json.loads(chunk_raw).get("text", "")
                        except Exception:
                            text = ""
                        if text:
                            await self._send(ws, wire.gen_chunk(rid, text))
                    await pump_fut
                    closing = {"text": "", "backend": "bee2bee-amd"}
                    await self._send(ws, wire.gen_success(rid, closing))
                    await self._send(ws, wire.gen_result(rid, closing))  # Q1
                else:
                    result = await loop.run_in_executor(None, svc.execute, params)
                    await self._send(ws, wire.gen_success(rid, result))
                    await self._send(ws, wire.gen_result(rid, result))  # Q1
            except Exception as e:
                logger.exception("local execution failed for %s", rid)
                await self._send(ws, wire.gen_error(rid, f"local_error: {e}"))
                await self._send(ws, wire.gen_result_error(rid, f"local_error: {e}"))
            return

        # 2. one-hop swarm relay: failover down the (price, latency) ranking,
        # never relaying back to the requester; sampling knobs and stream
        # chunks pass through (the reference relays prompt/max_tokens only
        # and has no failover — one dead provider fails the request)
        try:
            hops = int(data.get("hops") or 0)
        except (TypeError, ValueError):
            hops = 0
        if model_name and hops < 1:
            requester_pid = next(
                (p for p, peer in self.peers.items() if peer.ws is ws), None)
            last_err: Optional[Exception] = None
            want_stream = bool(data.get("stream"))

            def _forward_chunk(text: str) -> None:
                self._track(asyncio.create_task(
                    self._send(ws, wire.gen_chunk(rid, text))))

            for pid, _meta in self.pick_providers(
                model_name, limit=3, exclude=(requester_pid,)
            ):
                try:
                    result = await self.request_generation(
                        provider_id=pid,
                        prompt=params["prompt"],
                        max_new_tokens=params["max_new_tokens"],
                        model_name=model_name,
                        temperature=params.get("temperature", 0.7),
                        sampling={k: params.get(k) for k in
                                  ("top_p", "top_k", "repetition_penalty")},
                        stream=want_stream,
                        on_chunk=_forward_chunk if want_stream else None,
                        hops=hops + 1,
                    )
                    await self._send(ws, wire.gen_result(rid, result))
                    return
                except Exception as e:  # noqa: BLE001 — try the next provider
                    logger.info("relay to %s failed (%s); failing over", pid, e)
                    last_err = e
            if last_err is not None:
                await self._send(
                    ws, wire.gen_result_error(rid, f"{wire.ERR_RELAY}: {last_err}")
                )
                return

        await self._send(ws, wire.gen_result_error(rid, wire.ERR_NO_NODE))

    async def _handle_terminal(self, ws: Any, data: Dict[str, Any]) -> None:
        """gen_result / gen_success / gen_error all settle a pending rid."""
        rid = data.get("rid")
        fut = self._pending.pop(rid, None) if rid else None
        self._pending_ws.pop(rid, None)
        self._chunk_cbs.pop(rid, None)
        if fut is None or fut.done():
            return
        if data.get("type") == wire.GEN_ERROR or "error" in data:
            fut.set_exception(RuntimeError(data.get("error", "unknown_error")))
        else:
            # strip the wire envelope so relays re-wrapping this result don't
            # have their own type/rid clobbered by **result expansion (the
            # reference has exactly that bug on its relay path)
            fut.set_result(
                {k: v for k, v in data.items() if k not in ("type", "rid")}
            )

    async def _handle_gen_chunk(self, ws: Any, data: Dict[str, Any]) -> None:
        rid = data.get("rid")
        cb = self._chunk_cbs.get(rid)
        if cb is not None:
            try:
                cb(data.get("text", ""))
            except Exception:
                logger.exception("chunk callback failed")

    async def _handle_dht_set(self, ws: Any, data: Dict[str, Any]) -> None:
        key = data.get("key")
        if not isinstance(key, str):
            return
        value = data.get("value")
        # dict values merge (rendezvous maps peer-id -> record: two peers
        # announcing concurrently must not clobber each other's entries)
        if isinstance(value, dict):
            cur = await self.dht.get(key)
            if isinstance(cur, dict):
                value = {**cur, **value}
        await self.dht.set(key, value)

    async def dht_set(self, key: str, value: Any) -> None:
        """Store locally AND replicate one hop to every connected peer —
        enough for same-mesh RCCL rendezvous (parallel/rendezvous.py) without
        the optional kademlia dependency. With kademlia present the DHT
        itself replicates and the broadcast is a harmless no-op overlay."""
        await self.dht.set(key, value)
        async with self._lock:
            targets = [p.ws for p in self.peers.values()]
        for ws in targets:
            await self._send(ws, {"type": wire.DHT_SET, "key": key,
                                  "value": value})

    async def _handle_piece_request(self, ws: Any, data: Dict[str, Any]) -> None:
        """Serve a locally-held content piece (reference left this a stub,
        p2p_runtime.py:675-678)."""
        content_hash = data.get("hash")
        index = int(data.get("index", 0))
        info = self.pieces.get(content_hash)
        if not info:
            await self._send(
                ws,
                {"type": wire.PIECE_DATA, "hash": content_hash, "index": index,
                 "error": "piece_not_found"},
            )
            return
        import base64

        if "path" in info:  # disk-backed share_file registration
            if index >= info["n"]:
                await self._send(
                    ws,
                    {"type": wire.PIECE_DATA, "hash": content_hash,
                     "index": index, "error": "index_out_of_range"},
                )
                return

            def _read_slice() -> bytes:
                with open(info["path"], "rb") as f:
                    f.seek(index * info["piece_size"])
                    return f.read(info["piece_size"])

            try:
                blob = await asyncio.get_running_loop().run_in_executor(
                    None, _read_slice)
            except OSError as e:
                await self._send(
                    ws,
                    {"type": wire.PIECE_DATA, "hash": content_hash,
                     "index": index, "error": f"read_failed: {e}"},
                )
                return
        else:
            pieces = info.get("pieces", [])
            if index >= len(pieces):
                await self._send(
                    ws,
                    {"type": wire.PIECE_DATA, "hash": content_hash,
                     "index": index, "error": "index_out_of_range"},
                )
                return
            blob = pieces[index]
        await self._send(
            ws,
            {
                "type": wire.PIECE_DATA,
                "hash": content_hash,
                "index": index,
                "data": base64.b64encode(blob).decode(),
            },
        )

    async def _handle_piece_data(self, ws: Any, data: Dict[str, Any]) -> None:
        rid = f"piece:{data.get('hash')}:{data.get('index')}"
        fut = self._pending.pop(rid, None)
        self._pending_ws.pop(rid, None)
        if fut is None or fut.done():
            return
        if data.get("error"):
            fut.set_exception(RuntimeError(data["error"]))
        else:
            import base64

            fut.set_result(base64.b64decode(data.get("data", "")))

    async def request_piece(self, peer_id: str, content_hash: str, index: int) -> bytes:
        peer = self.peers.get(peer_id)
        if peer is None:
            raise RuntimeError(wire.ERR_NOT_CONNECTED)
        rid = f"piece:{content_hash}:{index}"
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._pending[rid] = fut
        self._pending_ws[rid] = peer.ws
        await self._send(
            peer.ws,
            {"type": wire.PIECE_REQUEST, "hash": content_hash, "index": index},
        )
        try:
            return await asyncio.wait_for(fut, timeout=60.0)
        except asyncio.TimeoutError:
            self._pending.pop(rid, None)
            self._pending_ws.pop(rid, None)
            raise RuntimeError(wire.ERR_TIMEOUT) from None

    def share_pieces(self, content_hash: str, pieces: List[bytes]) -> None:
        """Register in-memory pieces for serving to the mesh."""
        self.pieces[content_hash] = {"pieces": pieces}

    def share_file(self, content_hash: str, path: str, piece_size: int,
                   n_pieces: int) -> None:
        """Register a DISK-BACKED file for piece serving: the handler reads
        the requested slice on demand (executor), so seeding a multi-GB
        checkpoint holds no payload bytes in host RAM."""
        self.pieces[content_hash] = {
            "path": path, "piece_size": int(piece_size), "n": int(n_pieces)
        }

    # -------------------------------------------------------------- services

    async def add_service(self, service: Any) -> None:
        self.local_services[service.name] = service
        await self._broadcast(wire.service_announce(service.name, service.get_metadata()))
        logger.info("added service: %s", service.name)

    # --------------------------------------------------------------- routing

    def list_providers(self) -> List[Dict[str, Any]]:
        out = []
        for pid, svcs in self.providers.items():
            all_models: List[str] = []
            min_price = float("inf")
            found = False
            tag = None
            for svc_name, meta in svcs.items():
                if svc_name.startswith("_") or not isinstance(meta, dict):
                    continue
                if "models" in meta:
                    found = True
                    all_models.extend(meta.get("models", []))
                    price = meta.get("price_per_token", 0.0)
                    min_price = min(min_price, price)
                    if "tag" in meta and tag is None:
                        tag = meta["tag"]
            if found:
                peer = self.peers.get(pid)
                out.append(
                    {
                        "peer_id": pid,
                        "addr": peer.addr if peer else None,
                        "latency_ms": svcs.get("_latency"),
                        "models": sorted(set(all_models)),
                        "price_per_token": 0.0 if min_price == float("inf") else min_price,
                        "tag": tag,
                    }
                )
        return out

    def pick_providers(
        self,
        model_name: str,
        limit: int = 1,
        exclude: Tuple[Optional[str], ...] = (),
    ) -> List[Tuple[str, Dict[str, Any]]]:
        """Up to `limit` providers advertising the model, cheapest-then-
        fastest (reference sort key :745) — the ranking the relay's
        failover walks."""
        candidates = []
        for pid, svcs in self.providers.items():
            if pid in exclude:
                continue
            for svc_name, meta in svcs.items():
                if svc_name.startswith("_") or not isinstance(meta, dict):
                    continue
                if model_name in meta.get("models", []):
                    candidates.append(
                        (
                            meta.get("price_per_token", 0.0),
                            svcs.get("_latency", 99999.0),
                            pid,
                            svc_name,
                        )
                    )
                    break
        candidates.sort(key=lambda c: (c[0], c[1]))
        out = []
        for _price, _lat, pid, svc_name in candidates[:limit]:
            meta = dict(self.providers[pid][svc_name])
            meta["_svc_name"] = svc_name
            out.append((pid, meta))
        return out

    def pick_provider(self, model_name: str) -> Optional[Tuple[str, Dict[str, Any]]]:
        """Cheapest-then-fastest provider advertising the model."""
        picked = self.pick_providers(model_name, limit=1)
        return picked[0] if picked else None

    async def request_generation(
        self,
        provider_id: str,
        prompt: str,
        max_new_tokens: int = 32,
        model_name: Optional[str] = None,
        temperature: float = 0.7,
        stream: bool = False,
        on_chunk: Optional[Callable[[str], None]] = None,
        timeout: float = wire.REQUEST_TIMEOUT,
        sampling: Optional[Dict[str, Any]] = None,
        hops: int = 0,
    ) -> Dict[str, Any]:
        # self-request short-circuits to the local service
        if provider_id in (self.peer_id, "local"):
            svc = None
            for _name, inst in self.local_services.items():
                if not model_name or model_name in inst.get_metadata().get("models", []):
                    svc = inst
                    break
            if svc is None and self.local_services:
                svc = next(iter(self.local_services.values()))
            if svc is not None:
                params = {
                    "prompt": prompt,
                    "max_new_tokens": max_new_tokens,
                    "temperature": temperature,
                }
                for key in ("top_p", "top_k", "repetition_penalty"):
                    if sampling and sampling.get(key) is not None:
                        params[key] = sampling[key]
                loop = asyncio.get_running_loop()
                return await loop.run_in_executor(None, svc.execute, params)
            raise RuntimeError(wire.ERR_NO_LOCAL_SERVICE)

        peer = self.peers.get(provider_id)
        if peer is None:
            raise RuntimeError(wire.ERR_NOT_CONNECTED)

        rid = new_id("req")
        fut: asyncio.Future = asyncio.get_running_loop().create_future()
        self._pending[rid] = fut
        self._pending_ws[rid] = peer.ws
        if on_chunk is not None:
            self._chunk_cbs[rid] = on_chunk

        # resolve the remote service name from the provider table
        target_svc = "hf"
        svcs = self.providers.get(provider_id, {})
        if model_name:
            for name, meta in svcs.items():
                if not name.startswith("_") and isinstance(meta, dict) and model_name in meta.get("models", []):
                    target_svc = name
                    break
        if target_svc == "hf" and svcs:
            for name in svcs:
                if not name.startswith("_"):
                    target_svc = name
                    break

        await self._send(
            peer.ws,
            wire.gen_request(
                rid=rid,
                prompt=prompt,
                model=model_name,
                svc=target_svc,
                max_new_tokens=max_new_tokens,
                temperature=temperature,
                stream=stream,
                sampling=sampling,
                hops=hops,
            ),
        )
        try:
            return await asyncio.wait_for(fut, timeout=timeout)
        except asyncio.TimeoutError:
            self._pending.pop(rid, None)
            self._pending_ws.pop(rid, None)
            self._chunk_cbs.pop(rid, None)
            raise RuntimeError(wire.ERR_TIMEOUT) from None

    # ------------------------------------------------------------ monitoring

    async def enable_monitoring(self, interval_seconds: float = 30.0) -> None:
        if self._monitor_active:
            return
        self._monitor_active = True
        self._tasks.append(asyncio.create_task(self._monitor_loop(interval_seconds)))

    async def _monitor_loop(self, interval: float) -> None:
        while self._monitor_active and self._running:
            try:
                await self._run_health_checks()
                await self._reconnect_bootstraps()
                if self.registry.enabled:
                    await self.sync_with_registry()
            except Exception:
                logger.exception("monitoring error")
            await asyncio.sleep(interval)

    async def _reconnect_bootstraps(self) -> None:
        """Elastic recovery: re-dial bootstrap peers whose connection
        dropped (the reference's JS bridge reconnects after 5 s,
        app/api/bridge.js:83-95; its Python node never did)."""
        for addr in list(self._bootstrap_addrs):
            alive = any(
                p.addr == addr and not p.ws.closed for p in self.peers.values()
            )
            if not alive:
                try:
                    await self._connect_peer(addr)
                    logger.info("reconnected bootstrap %s", addr)
                except Exception as e:
                    logger.debug("bootstrap %s still down: %s", addr, e)

    async def _run_health_checks(self) -> None:
        from ..utils import now_ms

        timestamp = now_ms()
        local_metrics = get_system_metrics()
        for pid, peer in list(self.peers.items()):
            if peer.ws.closed:
                peer.health_status = "unreachable"
                if pid in self.providers:
                    self.providers[pid]["_health"] = "degraded"
                continue
            try:
                await self._send(peer.ws, wire.ping(local_metrics))
                peer.last_audit = timestamp
                peer.health_status = "online"
                if pid in self.providers:
                    self.providers[pid]["_health"] = "good"
            except Exception:
                peer.health_status = "unreachable"
                if pid in self.providers:
                    self.providers[pid]["_health"] = "degraded"

    async def sync_with_registry(self) -> None:
        if not self.addr or not self.registry.enabled:
            return
        metrics = get_system_metrics()
        metrics["api_port"] = self.api_port or 8000
        metrics["backend"] = "bee2bee-amd"
        models: List[str] = []
        for svc in self.local_services.values():
            meta = svc.get_metadata()
            models.extend(meta.get("models", []))
            if "model" in meta:
                models.append(meta["model"])
        await self.registry.sync_node(
            peer_id=self.peer_id,
            address=self.addr,
            models=sorted(set(models)),
            tag="bee2bee-amd",
            region=self.region,
            metrics=metrics,
        )


# ---------------------------------------------------------------------------


async def run_mesh_node(
    host: Optional[str] = None,
    port: Optional[int] = None,
    bootstrap_link: Optional[str] = None,
    model_name: Optional[str] = None,
    price_per_token: float = 0.0,
    announce_host: Optional[str] = None,
    backend: str = "native",
    api_port: Optional[int] = None,
    entrypoint_url: Optional[str] = None,
    region: str = "Auto",
    model_path: Optional[str] = None,
    device: Optional[str] = None,
    enable_nat: bool = True,
    ready_event: Optional[asyncio.Event] = None,
) -> None:
    """Start one peer node: WS mesh + optional FastAPI gateway + one service.

    Backends: "native" (the MI355X HIP engine — replaces the reference's
    transformers path), "ollama", "hf_remote". Reference orchestration:
    bee2bee/p2p_runtime.py run_p2p_node :843-954.
    """
    node = MeshNode(
        host=host or "0.0.0.0",
        port=port or 0,
        announce_host=announce_host,
        entrypoint_url=entrypoint_url,
        region=region,
        enable_nat=enable_nat,
    )
    node.api_port = api_port or 8000
    node.api_host = node.announce_host or node.host

    await node.start()

    api_server = None
    if api_port:
        import uvicorn

        from ..gateway import api as gateway_api

        gateway_api.node = node
        config = uvicorn.Config(
            gateway_api.app, host=node.host, port=api_port, log_level="warning"
        )
        api_server = uvicorn.Server(config)
        asyncio.get_running_loop().create_task(api_server.serve())
        logger.info("API gateway on http://%s:%s", node.host, api_port)

    if bootstrap_link:
        await node.connect_bootstrap(bootstrap_link)

    if model_name:
        svc = _build_service(backend, model_name, price_per_token, model_path, device)
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(None, svc.load_sync)
        await node.add_service(svc)

        from ..utils import sha256_hex_bytes

        join_link = generate_join_link(
            "connectit", model_name, sha256_hex_bytes(model_name.encode()), [node.addr]
        )
        logger.info("model %s (%s) serving; join link: %s", model_name, backend, join_link)
        print(f"JOIN_LINK {join_link}", flush=True)

    if node.registry.enabled:
        await node.enable_monitoring()
        await node.sync_with_registry()

    if ready_event is not None:
        ready_event.set()

    # graceful termination: SIGTERM (systemd/k8s stop) drains like Ctrl+C —
    # peers see a clean close instead of a cut socket
    stop_requested = asyncio.Event()
    try:
        import signal

        asyncio.get_running_loop().add_signal_handler(
            signal.SIGTERM, stop_requested.set)
    except (NotImplementedError, RuntimeError):
        pass  # non-unix loops / nested loops: Ctrl+C path still works

    try:
        await stop_requested.wait()
    except (asyncio.CancelledError, KeyboardInterrupt):
        pass
    finally:
        if api_server is not None:
            api_server.should_exit = True
        await node.stop()


def _build_service(
    backend: str,
    model_name: str,
    price_per_token: float,
    model_path: Optional[str],
    device: Optional[str],
) -> Any:
    if backend in ("native", "hf"):
        # "hf" maps to the native engine: this framework's replacement for the
        # reference's transformers backend. It loads HF-format checkpoints.
        from ..services.native import NativeEngineService

        return NativeEngineService(
            model_name,
            price_per_token=float(price_per_token or 0.0),
            model_path=model_path,
            device=device,
        )
    if backend == "ollama":
        from ..services.ollama import OllamaService

        return OllamaService(model_name)
    if backend == "hf_remote":
        from ..services.hf_remote import HFRemoteService

        return HFRemoteService(model_name, price_per_token=float(price_per_token or 0.005))
    raise ValueError(f"unknown backend: {backend}")
