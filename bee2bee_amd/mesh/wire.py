"""The mesh wire protocol — JSON text frames over WebSockets.

This module is the de-facto spec of the reference network's protocol
(reference: bee2bee/p2p_runtime.py dispatch :456-476 and handlers :478-683;
fields documented in SURVEY.md §2.1), written down once as typed builders so
every producer/consumer in this package agrees on field names.

Wire compatibility notes (deliberate fixes of reference quirks, kept
receive-compatible):
  * Q1: the reference requester resolves pending futures only on `gen_result`
    but a provider answering a buffered request sends `gen_success`
    (p2p_runtime.py:625 vs :660). We TREAT gen_result, gen_success and
    gen_error all as terminal for a pending rid, and we SEND gen_result in
    addition to the reference-shaped gen_success so both old Python peers and
    the JS web bridge get what they listen for.
  * Max frame size stays 32 MiB (p2p_runtime.py:176).
"""
from __future__ import annotations

import time
from typing import Any, Dict, List, Optional

MAX_FRAME = 32 * 1024 * 1024  # bytes, matches the reference mesh

# message types
HELLO = "hello"
PEER_LIST = "peer_list"
PING = "ping"
PONG = "pong"
SERVICE_ANNOUNCE = "service_announce"
GEN_REQUEST = "gen_request"
GEN_CHUNK = "gen_chunk"
GEN_SUCCESS = "gen_success"
GEN_ERROR = "gen_error"
GEN_RESULT = "gen_result"
PIECE_REQUEST = "piece_request"
PIECE_DATA = "piece_data"
# extension beyond the reference wire set: one-hop DHT record replication so
# mesh-connected peers can rendezvous into RCCL groups without kademlia
# (parallel/rendezvous.py); unknown types are ignored by reference peers
DHT_SET = "dht_set"

# terminal message types for a pending request id (Q1 fix: all three)
TERMINAL_TYPES = (GEN_RESULT, GEN_SUCCESS, GEN_ERROR)

# typed error strings (reference p2p_runtime.py:658,:654,:837,:792)
ERR_NO_NODE = "consensus_deadlock: no_node_available"
ERR_RELAY = "relay_link_failure"
ERR_TIMEOUT = "request_timed_out"
ERR_NOT_CONNECTED = "provider_not_connected"
ERR_NO_LOCAL_SERVICE = "no_local_service"

# timeouts / intervals (seconds)
REQUEST_TIMEOUT = 300.0
PING_INTERVAL = 15.0


def hello(
    peer_id: str,
    addr: str,
    region: str,
    metrics: Dict[str, Any],
    services: Dict[str, Dict[str, Any]],
    api_port: Optional[int] = None,
    api_host: Optional[str] = None,
    public_ip: Optional[str] = None,
) -> Dict[str, Any]:
    return {
        "type": HELLO,
        "peer_id": peer_id,
        "addr": addr,
        "region": region,
        "metrics": metrics,
        "services": services,
        "api_port": api_port,
        "api_host": api_host,
        "public_ip": public_ip,
    }


def peer_list(addrs: List[str]) -> Dict[str, Any]:
    return {"type": PEER_LIST, "peers": addrs}


def ping(metrics: Optional[Dict[str, Any]] = None) -> Dict[str, Any]:
    msg: Dict[str, Any] = {"type": PING, "ts": time.time()}
    if metrics is not None:
        msg["metrics"] = metrics
    return msg


def pong(ts: Any) -> Dict[str, Any]:
    return {"type": PONG, "ts": ts}


def service_announce(name: str, meta: Dict[str, Any]) -> Dict[str, Any]:
    return {"type": SERVICE_ANNOUNCE, "service": name, "meta": meta}


def gen_request(
    rid: str,
    prompt: str,
    model: Optional[str],
    svc: str = "hf",
    max_new_tokens: int = 2048,
    temperature: float = 0.7,
    stream: bool = False,
    sampling: Optional[Dict[str, Any]] = None,
    hops: int = 0,
) -> Dict[str, Any]:
    frame = {
        "type": GEN_REQUEST,
        "rid": rid,
        "prompt": prompt,
        "model": model,
        "svc": svc,
        "max_new_tokens": max_new_tokens,
        # duplicate under the legacy key some reference peers read
        "max_tokens": max_new_tokens,
        "temperature": temperature,
        "stream": stream,
    }
    # optional sampling knobs ride alongside (request_params reads the same
    # keys on the receiving side; absent -> reference generation defaults)
    for key in ("top_p", "top_k", "repetition_penalty"):
        if sampling and sampling.get(key) is not None:
            frame[key] = sampling[key]
    if hops:
        # relay-loop guard: relays forward hops+1; a frame that already
        # relayed once is never relayed again (one-hop semantics, matching
        # the reference's design). Extra key is ignored by reference peers.
        frame["hops"] = int(hops)
    return frame


def gen_chunk(rid: str, text: str) -> Dict[str, Any]:
    return {"type": GEN_CHUNK, "rid": rid, "text": text}


def gen_success(rid: str, result: Dict[str, Any]) -> Dict[str, Any]:
    return {"type": GEN_SUCCESS, "rid": rid, **result}


def gen_error(rid: str, error: str) -> Dict[str, Any]:
    return {"type": GEN_ERROR, "rid": rid, "error": error}


def gen_result(rid: str, result: Dict[str, Any]) -> Dict[str, Any]:
    return {"type": GEN_RESULT, "rid": rid, **result}


def gen_result_error(rid: str, error: str) -> Dict[str, Any]:
    return {"type": GEN_RESULT, "rid": rid, "error": error}


def request_params(data: Dict[str, Any]) -> Dict[str, Any]:
    """Normalize an incoming gen_request into service-execute params.

    Accepts both `max_new_tokens` and the legacy `max_tokens` key and both
    `rid` and legacy `task_id` (reference :574-586)."""
    out = {
        "prompt": data.get("prompt", ""),
        "max_new_tokens": int(
            data.get("max_new_tokens") or data.get("max_tokens") or 2048
        ),
        "temperature": float(data.get("temperature", 0.7)),
    }
    # optional sampling knobs pass through when present (None -> the
    # engine applies the reference generation defaults)
    for key in ("top_p", "top_k", "repetition_penalty"):
        if data.get(key) is not None:
            out[key] = data[key]
    return out


def request_id(data: Dict[str, Any]) -> Optional[str]:
    return data.get("rid") or data.get("task_id")
