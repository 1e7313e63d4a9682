"""Browser-facing web gateway: the reference's Express API, natively.

Endpoint parity (reference app/api/index.js):
  POST /api/p2p/register        :16-32   join-link registration
  POST /api/p2p/generate        :35-98   streamed generation proxy +
                                         `messages` token-tally persistence
  GET/POST /api/p2p/status      :100-161 mesh telemetry + dynamic discovery
  GET/POST /api/p2p/global_metrics :164-216 aggregate usage

plus GET / serving a minimal chat dashboard (stand-in for the reference's
React SPA: one-click node registration + streaming chat + live metrics).

Run: `python -m bee2bee_amd serve-web --port 8080` or embed via
`create_app(bridge)`.
"""
from __future__ import annotations

import asyncio
import json
import logging
import math
from contextlib import asynccontextmanager
from typing import Any, Dict, Optional

import httpx
from fastapi import FastAPI, Request
from fastapi.responses import HTMLResponse, JSONResponse, StreamingResponse

from .bridge import MeshBridge
from .store import GLOBAL_METRICS_NODE, NETWORK_PULSE_CONTENT, WebStore

logger = logging.getLogger("bee2bee_amd.web")

MODE = "fusion-serverless"  # reported mode string (reference parity)
TARGET_PROBE_TIMEOUT_S = 2.0


def create_app(bridge: Optional[MeshBridge] = None,
               store: Optional[WebStore] = None) -> FastAPI:
    owned = bridge is None
    the_store = store if store is not None else WebStore()

    @asynccontextmanager
    async def lifespan(app: FastAPI):
        nonlocal bridge
        if bridge is None:
            bridge = MeshBridge(store=the_store)
            await bridge.start()
        app.state.bridge = bridge
        app.state.store = the_store
        yield
        if owned and bridge is not None:
            await bridge.stop()

    app = FastAPI(title="bee2bee-amd web gateway", lifespan=lifespan)

    # ------------------------------------------------------------- register

    @app.post("/api/p2p/register")
    async def register(req: Request):
        body = await _json_body(req)
        link = body.get("link")
        if not link:
            return JSONResponse({"error": "Missing join link"}, status_code=400)
        try:
            result = await app.state.bridge.register_join_link(link)
        except Exception as e:  # noqa: BLE001
            return JSONResponse({"error": str(e)}, status_code=500)
        stats = app.state.bridge.get_stats()
        return {**result, "connected": stats["connected"],
                "activeNode": stats["activeNode"], "mode": MODE}

    # ------------------------------------------------------------- generate

    @app.post("/api/p2p/generate")
    async def generate(req: Request):
        body = await _json_body(req)
        task = body.get("task") or {}
        prompt = task.get("prompt") or body.get("prompt")
        model = task.get("model") or body.get("model") or "default"
        target = task.get("targetNode") or body.get("targetNode")
        if not prompt:
            return JSONResponse({"error": "Prompt is required"},
                                status_code=400)
        payload = {
            "prompt": prompt,
            "model": model,
            "max_tokens": body.get("max_tokens"),
            "temperature": body.get("temperature"),
        }
        bridge: MeshBridge = app.state.bridge
        store: WebStore = app.state.store
        queue: asyncio.Queue = asyncio.Queue()
        DONE = object()

        async def run_request() -> None:
            emitted = 0

            def on_chunk(text: str) -> None:
                nonlocal emitted
                emitted += len(text)
                queue.put_nowait(text)

            try:
                result = await bridge.request(payload, on_chunk, target)
                text = result.get("text", "")
                # buffered transports emit nothing chunk-wise: flush the
                # final text so the client always sees content
                if not emitted and text:
                    queue.put_nowait(text)
                full_len = max(emitted, len(text))
                tokens_est = math.ceil(full_len / 4)
                if tokens_est > 0:
                    await store.insert_message(
                        node_id=target or GLOBAL_METRICS_NODE,
                        tokens=tokens_est,
                        metadata={"model": model},
                    )
            except Exception as e:  # noqa: BLE001
                queue.put_nowait(e)
            finally:
                queue.put_nowait(DONE)

        runner = asyncio.ensure_future(run_request())

        async def stream():
            # leading flush mirrors the reference's `res.write(' ')` so
            # proxies start forwarding immediately
            yield " "
            try:
                while True:
                    item = await queue.get()
                    if item is DONE:
                        break
                    if isinstance(item, Exception):
                        yield f"\n\n[Error]: {item}"
                        break
                    yield item
            finally:
                runner.cancel()

        return StreamingResponse(stream(), media_type="text/event-stream",
                                 headers={"Cache-Control": "no-cache"})

    # --------------------------------------------------------------- status

    async def _status(target: Optional[str]) -> Dict[str, Any]:
        bridge: MeshBridge = app.state.bridge
        target_status = None
        if target:
            probe = target if target.startswith("http") else f"http://{target}"
            try:
                async with httpx.AsyncClient(
                        timeout=TARGET_PROBE_TIMEOUT_S) as client:
                    resp = await client.get(f"{probe.rstrip('/')}/")
                if resp.status_code == 200:
                    target_status = resp.json()
            except Exception as e:  # noqa: BLE001
                logger.info("target %s unreachable: %s", target, e)
        await bridge.sync_global_mesh()
        stats = bridge.get_stats()
        mesh = bridge.get_regional_mesh()
        if target_status and target:
            region = target_status.get("region") or "Local-Probe"
            rows = mesh.setdefault(region, [])
            known = any(r.get("addr") == target
                        or r.get("peer_id") == target_status.get("peer_id")
                        for r in rows)
            if not known:
                rows.append({**target_status, "addr": target,
                             "status": "active", "tag": "direct-ingress"})
        active = stats["connected"] or stats["poolSize"] > 0 or bool(target_status)
        return {**stats, "mesh": mesh, "mode": MODE,
                "status": "active" if active else "idle"}

    @app.get("/api/p2p/status")
    async def status_get(target: Optional[str] = None):
        return await _status(target)

    @app.post("/api/p2p/status")
    async def status_post(req: Request):
        body = await _json_body(req)
        peer = body.get("peer") or {}
        if body.get("action") == "discover_peer" and peer.get("addr"):
            await app.state.bridge.connect_to_peer(peer["addr"])
            return {"status": "discovery_initiated"}
        return await _status(body.get("target"))

    # ------------------------------------------------------- global metrics

    @app.get("/api/p2p/global_metrics")
    async def global_metrics_get():
        return await app.state.store.system_stats()

    @app.post("/api/p2p/global_metrics")
    async def global_metrics_post(req: Request):
        body = await _json_body(req)
        tokens = int(body.get("tokens") or 0)
        if tokens <= 0:
            return {"success": False}
        ok = await app.state.store.insert_message(
            node_id=GLOBAL_METRICS_NODE, tokens=tokens,
            content=NETWORK_PULSE_CONTENT)
        return {"success": bool(ok)}

    # ------------------------------------------------------------ dashboard

    @app.get("/", response_class=HTMLResponse)
    async def dashboard():
        return DASHBOARD_HTML

    return app


async def _json_body(req: Request) -> Dict[str, Any]:
    try:
        body = await req.json()
        return body if isinstance(body, dict) else {}
    except Exception:  # noqa: BLE001
        return {}


# Minimal single-file dashboard: node registration, mesh status, streaming
# chat and the global token tally — the reference SPA's core flows without
# a JS build step.
DASHBOARD_HTML = """<!doctype html>
<html><head><meta charset="utf-8"><title>bee2bee-amd mesh</title>
<style>
 body{font-family:system-ui,sans-serif;margin:0;background:#0d1117;color:#e6edf3}
 main{max-width:860px;margin:0 auto;padding:24px}
 h1{font-size:20px} section{background:#161b22;border:1px solid #30363d;
 border-radius:8px;padding:16px;margin:16px 0}
 input,button{font:inherit;padding:8px;border-radius:6px;border:1px solid #30363d;
 background:#0d1117;color:inherit} input{width:60%}
 button{cursor:pointer;background:#238636;border:none;color:#fff}
 #log{white-space:pre-wrap;background:#0d1117;border:1px solid #30363d;
 border-radius:6px;padding:12px;min-height:120px;margin-top:8px}
 .stat{display:inline-block;margin-right:24px;color:#7d8590}
 .stat b{color:#e6edf3}
</style></head><body><main>
<h1>bee2bee-amd &mdash; decentralized inference mesh</h1>
<section><h3>Network</h3><div id="stats">loading&hellip;</div>
<table id="mesh" style="width:100%;margin-top:12px;border-collapse:collapse">
</table></section>
<section><h3>Register a node</h3>
<input id="link" placeholder="coithub.org://join?..."/>
<button onclick="registerNode()">Register</button>
<div id="regout" class="stat"></div></section>
<section><h3>Chat</h3>
<input id="prompt" placeholder="Ask the mesh&hellip;"/>
<button onclick="send()">Send</button>
<div id="log"></div></section>
<script>
async function refresh(){
 try{
  const s=await (await fetch('api/p2p/status')).json();
  const m=await (await fetch('api/p2p/global_metrics')).json();
  document.getElementById('stats').innerHTML=
   `<span class=stat>status <b>${s.status}</b></span>`+
   `<span class=stat>active node <b>${s.activeNode||'-'}</b></span>`+
   `<span class=stat>peers <b>${s.poolSize}</b></span>`+
   `<span class=stat>total tokens <b>${m.tokens||0}</b></span>`+
   `<span class=stat>chats <b>${m.chats||0}</b></span>`;
  const rows=[['region','node','models','latency','status']];
  for(const [region,peers] of Object.entries(s.mesh||{}))
   for(const p of peers)
    rows.push([region,p.peer_id||p.addr||'-',
     (p.models||[]).join(', ')||'-',
     p.latency!=null?p.latency+' ms':'-',p.status||'-']);
  document.getElementById('mesh').innerHTML=rows.map((r,i)=>
   `<tr style="border-bottom:1px solid #30363d">`+r.map(c=>
    i?`<td style="padding:4px 8px">${c}</td>`
     :`<th style="text-align:left;padding:4px 8px;color:#7d8590">${c}</th>`)
   .join('')+`</tr>`).join('');
 }catch(e){}
}
async function registerNode(){
 const link=document.getElementById('link').value.trim();
 if(!link)return;
 const r=await fetch('api/p2p/register',{method:'POST',
  headers:{'Content-Type':'application/json'},body:JSON.stringify({link})});
 const j=await r.json();
 document.getElementById('regout').textContent=
  j.success?`connected: ${j.node}`:`failed: ${j.error}`;
 refresh();
}
async function send(){
 const p=document.getElementById('prompt').value.trim();
 if(!p)return;
 const log=document.getElementById('log');
 log.textContent+='\\n> '+p+'\\n';
 const r=await fetch('api/p2p/generate',{method:'POST',
  headers:{'Content-Type':'application/json'},body:JSON.stringify({prompt:p})});
 const reader=r.body.getReader();const dec=new TextDecoder();
 for(;;){const {done,value}=await reader.read();if(done)break;
  log.textContent+=dec.decode(value);log.scrollTop=log.scrollHeight;}
 log.textContent+='\\n';refresh();
}
refresh();setInterval(refresh,10000);
</script></main></body></html>
"""
