"""FastAPI gateway: node status, peers, providers, connect, chat/generate.

Endpoint parity with reference bee2bee/api.py (GET /, /peers, /providers,
/connect; POST /chat and /generate with optional streaming; X-API-KEY auth
:19-32; lifespan boots a node when not pre-seeded :35-96). Fixes reference
quirk Q4: uptime is real (node.start_time is set by MeshNode.__init__).
"""
from __future__ import annotations

import asyncio
import os
import time
from typing import List, Optional

from fastapi import Depends, FastAPI, HTTPException, status
from fastapi.middleware.cors import CORSMiddleware
from fastapi.responses import StreamingResponse
from fastapi.security import APIKeyHeader
from pydantic import BaseModel

from ..mesh.node import MeshNode

node: Optional[MeshNode] = None

API_KEY_NAME = "X-API-KEY"
api_key_header = APIKeyHeader(name=API_KEY_NAME, auto_error=False)


async def get_api_key(header_key: Optional[str] = Depends(api_key_header)):
    config_key = os.getenv("BEE2BEE_API_KEY")
    if not config_key:
        return None  # open API when no key is configured (dev mode)
    if header_key == config_key:
        return header_key
    raise HTTPException(
        status_code=status.HTTP_401_UNAUTHORIZED,
        detail="Invalid or missing API Key",
    )


from contextlib import asynccontextmanager


@asynccontextmanager
async def lifespan(app: FastAPI):
    global node
    if node is None:
        port = int(os.getenv("BEE2BEE_PORT", "4001"))
        host = os.getenv("BEE2BEE_HOST", "0.0.0.0")
        announce_host = os.getenv("BEE2BEE_ANNOUNCE_HOST")
        announce_port_str = os.getenv("BEE2BEE_ANNOUNCE_PORT")
        announce_port = int(announce_port_str) if announce_port_str else None
        node = MeshNode(
            host=host,
            port=port,
            announce_host=announce_host,
            announce_port=announce_port,
            enable_nat=os.getenv("BEE2BEE_DISABLE_NAT") is None,
        )
        await node.start()
    bootstrap = os.getenv("BEE2BEE_BOOTSTRAP")
    if bootstrap:
        await node.connect_bootstrap(bootstrap)
    await node.enable_monitoring(interval_seconds=15)
    yield
    if node is not None:
        await node.stop()


app = FastAPI(title="Bee2Bee-AMD Node API", lifespan=lifespan)

app.add_middleware(
    CORSMiddleware,
    allow_origins=os.getenv("CORS_ORIGINS", "*").split(","),
    allow_credentials=True,
    allow_methods=["*"],
    allow_headers=["*"],
)

# OpenAI-compatible surface (/v1/models, /v1/completions,
# /v1/chat/completions) behind the same optional API key
from . import metrics as _metrics  # noqa: E402
from .openai_compat import router as _openai_router  # noqa: E402

app.include_router(_openai_router, dependencies=[Depends(get_api_key)])


@app.middleware("http")
async def _count_requests(request, call_next):
    response = await call_next(request)
    if request.url.path != "/metrics":
        _metrics.HTTP_REQUESTS.labels(
            path=request.url.path, method=request.method,
            status=str(response.status_code)).inc()
    return response


@app.get("/metrics")
def metrics_endpoint():
    from fastapi import Response

    _metrics.refresh(node)
    return Response(_metrics.render(), media_type=_metrics.CONTENT_TYPE)


class PeerInfo(BaseModel):
    peer_id: str
    addr: str
    latency_ms: Optional[float]


class ProviderInfo(BaseModel):
    peer_id: str
    addr: Optional[str]
    latency_ms: Optional[float]
    models: List[str]
    price_per_token: Optional[float]
    tag: Optional[str] = None


class ChatRequest(BaseModel):
    provider_id: Optional[str] = "local"
    prompt: str
    model: Optional[str] = None
    max_new_tokens: Optional[int] = None
    temperature: Optional[float] = 0.7
    # None -> the engine applies the reference generation defaults
    # (top_p 0.95, repetition_penalty 1.15 — bee2bee/hf.py:94-103)
    top_p: Optional[float] = None
    top_k: Optional[int] = None
    repetition_penalty: Optional[float] = None
    stream: Optional[bool] = False


@app.get("/")
def home():
    if node is None:
        return {"status": "starting", "node_id": "not_started"}
    services_meta = {}
    all_models: List[str] = []
    for name, svc in node.local_services.items():
        meta = svc.get_metadata()
        services_meta[name] = meta
        all_models.extend(meta.get("models", []))
    return {
        "status": "ok",
        "node_id": node.peer_id,
        "peer_id": node.peer_id,
        "region": node.region or "Global",
        "models": sorted(set(all_models)),
        "services": services_meta,
        "metrics": {
            "uptime": int(time.time() - node.start_time),
            "pool_size": len(node.peers),
            "status": "active",
        },
        # live engine state (queue depth, KV pool, tok/s) — real numbers,
        # unlike the reference's simulated throughput (bee2bee/utils.py:129)
        "engine": next(
            (
                svc.engine.stats()
                for svc in node.local_services.values()
                if getattr(svc, "engine", None) is not None
                and hasattr(svc.engine, "stats")
            ),
            None,
        ),
    }


@app.get("/peers", dependencies=[Depends(get_api_key)])
def get_peers():
    if node is None:
        return []
    return [
        {
            "peer_id": pid,
            "addr": peer.addr or "",
            "latency_ms": peer.last_pong_ms,
            "health_status": peer.health_status,
            "last_audit": peer.last_audit,
            "metrics": peer.metrics,
        }
        for pid, peer in node.peers.items()
    ]


@app.get("/providers", response_model=List[ProviderInfo], dependencies=[Depends(get_api_key)])
def list_providers():
    if node is None:
        return []
    return node.list_providers()


@app.get("/connect", dependencies=[Depends(get_api_key)])
async def connect_peer(addr: str):
    if node is None:
        return {"error": "Node not running"}
    try:
        if any(addr.startswith(p) for p in ("p2pnet", "coithub")):
            await node.connect_bootstrap(addr)
        else:
            await node._connect_peer(addr)  # noqa: SLF001 - parity with reference
        return {"status": "connected", "addr": addr}
    except Exception as e:
        return {"status": "error", "message": str(e)}


def _model_matches(requested: Optional[str], models: List[str]) -> bool:
    if not requested:
        return True
    for m in models:
        if requested == m or requested in m or m in requested:
            return True
    return False


@app.post("/chat", dependencies=[Depends(get_api_key)])
@app.post("/generate", dependencies=[Depends(get_api_key)])
async def chat(req: ChatRequest):
    if node is None:
        return {"error": "Node not running"}
    loop = asyncio.get_running_loop()
    try:
        # local-first with fuzzy model matching (reference api.py:203-216)
        for svc_name, svc in node.local_services.items():
            meta = svc.get_metadata()
            if not _model_matches(req.model, meta.get("models", [])):
                continue
            params = {
                "prompt": req.prompt,
                "max_new_tokens": req.max_new_tokens or 2048,
                "temperature": req.temperature or 0.7,
                "top_p": req.top_p,
                "top_k": req.top_k,
                "repetition_penalty": req.repetition_penalty,
            }
            if req.stream:
                if hasattr(svc, "execute_stream_async"):
                    return StreamingResponse(
                        svc.execute_stream_async(params), media_type="text/plain"
                    )
                # pump the sync generator in a worker thread
                queue: asyncio.Queue = asyncio.Queue()

                def _pump() -> None:
                    try:
                        for chunk in svc.execute_stream(params):
                            loop.call_soon_threadsafe(queue.put_nowait, chunk)
                    finally:
                        loop.call_soon_threadsafe(queue.put_nowait, None)

                loop.run_in_executor(None, _pump)

                async def agen():
                    while True:
                        item = await queue.get()
                        if item is None:
                            break
                        yield item

                return StreamingResponse(agen(), media_type="text/plain")

            result = await loop.run_in_executor(None, svc.execute, params)
            return {
                "status": "ok",
                "text": result.get("text", ""),
                "rid": f"local-{int(time.time() * 1000)}",
                "metadata": {
                    "engine": "bee2bee-amd-local",
                    "node": node.addr,
                    "service": svc_name,
                    "latency_ms": result.get("latency_ms"),
                    "tokens": result.get("tokens"),
                    "timing": result.get("timing"),
                },
            }

        # P2P fallback
        pid = req.provider_id
        if not pid or pid == "local":
            if req.model:
                picked = node.pick_provider(req.model)
                pid = picked[0] if picked else node.peer_id
            else:
                pid = node.peer_id
        res = await node.request_generation(
            pid, req.prompt, req.max_new_tokens or 2048, req.model,
            temperature=req.temperature or 0.7,
            sampling={"top_p": req.top_p, "top_k": req.top_k,
                      "repetition_penalty": req.repetition_penalty},
        )
        return {
            "status": "ok",
            "text": res.get("text", ""),
            "rid": res.get("rid"),
            "metadata": {
                "engine": "bee2bee-amd-p2p",
                "node": node.addr,
                "latency_ms": res.get("latency_ms"),
            },
        }
    except Exception as e:
        return {"status": "error", "message": str(e)}
