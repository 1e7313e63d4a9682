"""OpenAI-compatible API on the node gateway: /v1/models, /v1/completions,
/v1/chat/completions (buffered + SSE streaming).

Beyond reference parity (the reference exposes only its own /chat and
/generate shapes): existing OpenAI-client tooling points at a mesh node
unchanged. Routing reuses the gateway's local-first-then-P2P semantics
(api.py /chat); chat messages are flattened to the `user:`/`assistant:`
transcript the reference's web chat sends as its prompt
(app/src/App.jsx:993-997, parsed by bee2bee/hf.py:56-81).
"""
from __future__ import annotations

import asyncio
import json
import time
from typing import Any, AsyncIterator, Dict, List, Optional

from fastapi import APIRouter, HTTPException
from fastapi.responses import StreamingResponse
from pydantic import BaseModel

from ..services.base import ServiceError
from ..utils import new_id

router = APIRouter()


def _http_error(e: Exception) -> HTTPException:
    msg = str(e)
    code = 429 if "overloaded" in msg else 502
    return HTTPException(status_code=code, detail=msg)


class CompletionRequest(BaseModel):
    model: Optional[str] = None
    # str, or a LIST of prompts (OpenAI batch shape — one choice per prompt,
    # run concurrently through the engine's continuous batching)
    prompt: Any = ""
    max_tokens: Optional[int] = 256
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    stream: Optional[bool] = False
    n: Optional[int] = 1
    stop: Optional[Any] = None  # str | List[str]


class ChatMessage(BaseModel):
    role: str
    content: str


class ChatCompletionRequest(BaseModel):
    model: Optional[str] = None
    messages: List[ChatMessage]
    max_tokens: Optional[int] = 256
    temperature: Optional[float] = None
    top_p: Optional[float] = None
    stream: Optional[bool] = False
    n: Optional[int] = 1
    stop: Optional[Any] = None


def _gateway():
    """The live gateway module (node + auth are its globals)."""
    from . import api as gateway_api

    if gateway_api.node is None:
        raise HTTPException(status_code=503, detail="node not running")
    return gateway_api


def _stop_list(stop: Any) -> List[str]:
    if stop is None:
        return []
    return [stop] if isinstance(stop, str) else [s for s in stop if s]


def _truncate_at_stop(text: str, stops: List[str]) -> str:
    """Cut at the EARLIEST occurrence of any stop (OpenAI semantics; also
    what engine.generate_text and StopStringFilter compute — sequential
    per-stop cuts would be order-dependent when stops overlap)."""
    cut = min((text.find(s) for s in stops if s and text.find(s) >= 0),
              default=-1)
    return text[:cut] if cut >= 0 else text


def _est_tokens(text: str) -> int:
    # the directory tally estimate (web/gateway.py): ceil(len/4)
    return max(0, -(-len(text) // 4))


def _pick_service(node, model: Optional[str]):
    """Local-first with the gateway's fuzzy matching; None -> P2P."""
    from .api import _model_matches

    for _name, svc in node.local_services.items():
        if _model_matches(model, svc.get_metadata().get("models", [])):
            return svc
    return None


async def _run_buffered(node, model: Optional[str], prompt: str,
                        req) -> Dict[str, Any]:
    params = {
        "prompt": prompt,
        "max_new_tokens": req.max_tokens or 256,
        "temperature": 0.7 if req.temperature is None else req.temperature,
    }
    if req.top_p is not None:
        params["top_p"] = req.top_p
    svc = _pick_service(node, model)
    loop = asyncio.get_running_loop()
    if svc is not None:
        return await loop.run_in_executor(None, svc.execute, params)
    picked = node.pick_provider(model) if model else None
    pid = picked[0] if picked else node.peer_id
    return await node.request_generation(
        pid, prompt, params["max_new_tokens"], model,
        temperature=params["temperature"],
        sampling={"top_p": req.top_p},
    )


async def _sse(events: AsyncIterator[Dict[str, Any]]) -> AsyncIterator[str]:
    async for ev in events:
        yield f"data: {json.dumps(ev)}\n\n"
    yield "data: [DONE]\n\n"


async def _stream_deltas(node, model, prompt, req) -> AsyncIterator[str]:
    """Text deltas from the local service stream (or one buffered burst
    when only the P2P path exists)."""
    svc = _pick_service(node, model)
    stops = _stop_list(req.stop)
    if svc is not None:
        params = {
            "prompt": prompt,
            "max_new_tokens": req.max_tokens or 256,
            "temperature": 0.7 if req.temperature is None else req.temperature,
        }
        if req.top_p is not None:
            params["top_p"] = req.top_p
        loop = asyncio.get_running_loop()
        queue: asyncio.Queue = asyncio.Queue()

        def _pump() -> None:
            try:
                for line in svc.execute_stream(params):
                    loop.call_soon_threadsafe(queue.put_nowait, line)
            finally:
                loop.call_soon_threadsafe(queue.put_nowait, None)

        loop.run_in_executor(None, _pump)
        # exact incremental truncation (straddling and overlapping stops
        # included): engine.StopStringFilter is the single implementation
        from ..engine.engine import StopStringFilter

        filt = StopStringFilter(stops)
        while True:
            line = await queue.get()
            if line is None:
                tail = filt.flush()
                if tail:
                    yield tail
                return
            try:
                obj = json.loads(line)
            except (TypeError, json.JSONDecodeError):
                continue
            out = filt.feed(obj.get("text") or "")
            if out:
                yield out
            if filt.done:
                return
    else:
        result = await _run_buffered(node, model, prompt, req)
        text = _truncate_at_stop(result.get("text", ""), stops)
        if text:
            yield text


def _reject_n(n: Optional[int]) -> None:
    if n is not None and n != 1:
        raise HTTPException(status_code=400, detail="only n=1 is supported")


def _check_max_tokens(v: Optional[int]) -> None:
    if v is not None and v < 1:
        raise HTTPException(status_code=400,
                            detail="max_tokens must be >= 1")


@router.get("/v1/models")
async def list_models():
    gw = _gateway()
    node = gw.node
    ids = set()
    for svc in node.local_services.values():
        ids.update(svc.get_metadata().get("models", []))
    for entry in node.list_providers():
        ids.update(entry.get("models", []))
    now = int(time.time())
    return {
        "object": "list",
        "data": [{"id": m, "object": "model", "created": now,
                  "owned_by": "bee2bee-amd"} for m in sorted(ids)],
    }


@router.post("/v1/completions")
async def completions(req: CompletionRequest):
    gw = _gateway()
    _reject_n(req.n)
    _check_max_tokens(req.max_tokens)
    cid = new_id("cmpl")
    created = int(time.time())
    model = req.model
    prompts = req.prompt if isinstance(req.prompt, list) else [req.prompt]
    if not all(isinstance(p, str) for p in prompts):
        raise HTTPException(status_code=400,
                            detail="prompt must be a string or list of strings")

    if req.stream:
        if len(prompts) != 1:
            raise HTTPException(
                status_code=400,
                detail="stream=true supports a single prompt")

        async def events():
            async for delta in _stream_deltas(gw.node, model, prompts[0], req):
                yield {
                    "id": cid, "object": "text_completion",
                    "created": created, "model": model or "auto",
                    "choices": [{"index": 0, "text": delta,
                                 "finish_reason": None}],
                }

        return StreamingResponse(_sse(events()),
                                 media_type="text/event-stream")

    try:
        # one choice per prompt; concurrent calls ride the engine's
        # continuous batching (they share decode steps)
        results = await asyncio.gather(
            *(_run_buffered(gw.node, model, p, req) for p in prompts))
    except ServiceError as e:
        raise _http_error(e) from e
    stops = _stop_list(req.stop)
    choices = []
    completion_total = 0
    for i, (p, result) in enumerate(zip(prompts, results)):
        text = _truncate_at_stop(result.get("text", ""), stops)
        completion_total += result.get("tokens") or _est_tokens(text)
        choices.append({"index": i, "text": text, "logprobs": None,
                        "finish_reason": "stop"})
    prompt_total = sum(_est_tokens(p) for p in prompts)
    return {
        "id": cid,
        "object": "text_completion",
        "created": created,
        "model": model or "auto",
        "choices": choices,
        "usage": {
            "prompt_tokens": prompt_total,
            "completion_tokens": completion_total,
            "total_tokens": prompt_total + completion_total,
        },
    }


def _transcript(messages: List[ChatMessage]) -> str:
    lines = []
    for m in messages:
        role = m.role if m.role in ("user", "assistant", "system") else "user"
        lines.append(f"{role}: {m.content}")
    lines.append("assistant:")
    return "\n".join(lines)


@router.post("/v1/chat/completions")
async def chat_completions(req: ChatCompletionRequest):
    gw = _gateway()
    _reject_n(req.n)
    _check_max_tokens(req.max_tokens)
    if not req.messages:
        raise HTTPException(status_code=400, detail="messages must not be empty")
    cid = new_id("chatcmpl")
    created = int(time.time())
    prompt = _transcript(req.messages)
    model = req.model

    if req.stream:
        async def events():
            first = True
            async for delta in _stream_deltas(gw.node, model, prompt, req):
                d: Dict[str, Any] = {"content": delta}
                if first:
                    d["role"] = "assistant"
                    first = False
                yield {
                    "id": cid, "object": "chat.completion.chunk",
                    "created": created, "model": model or "auto",
                    "choices": [{"index": 0, "delta": d,
                                 "finish_reason": None}],
                }

        return StreamingResponse(_sse(events()),
                                 media_type="text/event-stream")

    try:
        result = await _run_buffered(gw.node, model, prompt, req)
    except ServiceError as e:
        raise _http_error(e) from e
    text = _truncate_at_stop(result.get("text", ""), _stop_list(req.stop))
    completion_tokens = result.get("tokens") or _est_tokens(text)
    return {
        "id": cid,
        "object": "chat.completion",
        "created": created,
        "model": model or "auto",
        "choices": [{"index": 0,
                     "message": {"role": "assistant", "content": text},
                     "finish_reason": "stop"}],
        "usage": {
            "prompt_tokens": _est_tokens(prompt),
            "completion_tokens": completion_tokens,
            "total_tokens": _est_tokens(prompt) + completion_tokens,
        },
    }
