"""Prometheus metrics for the node gateway (`GET /metrics`).

Beyond reference parity (the reference exposes metrics only as JSON inside
its own status payloads, and fabricates throughput — bee2bee/utils.py:129):
this exports the REAL engine counters (queue depth, KV pool, busy-time,
tok/s) plus mesh state in the standard exposition format, scrape-ready for
production monitoring. Gauges are refreshed from live node/engine state at
scrape time; the request counter is bumped by the gateway middleware.
"""
from __future__ import annotations

from typing import Any, Dict, Optional

from prometheus_client import (
    CONTENT_TYPE_LATEST,
    CollectorRegistry,
    Counter,
    Gauge,
    generate_latest,
)

REGISTRY = CollectorRegistry()

HTTP_REQUESTS = Counter(
    "bee2bee_http_requests_total",
    "Gateway HTTP requests",
    ["path", "method", "status"],
    registry=REGISTRY,
)

_PEERS = Gauge("bee2bee_mesh_peers", "Connected mesh peers",
               registry=REGISTRY)
_PROVIDERS = Gauge("bee2bee_mesh_providers", "Known model providers",
                   registry=REGISTRY)
_UPTIME = Gauge("bee2bee_uptime_seconds", "Node uptime", registry=REGISTRY)
_ENGINE = {}  # stat key -> Gauge, created lazily from engine.stats() keys

# keys exactly as InferenceEngine.stats() emits them
_ENGINE_STATS = (
    ("queued_requests", "requests waiting for admission"),
    ("active_requests", "sequences in the running batch"),
    ("prefilling_requests", "requests in chunked prefill"),
    ("kv_free_blocks", "KV pool blocks free"),
    ("kv_total_blocks", "KV pool blocks total"),
    ("tokens_total", "tokens generated since start"),
    ("tokens_per_sec_10s", "generation throughput (10s window)"),
    ("engine_busy_s", "engine thread busy seconds"),
    ("engine_steps", "engine busy-step count"),
    ("engine_ms_per_step", "mean engine step latency (ms)"),
)


def _engine_gauge(key: str, doc: str) -> Gauge:
    if key not in _ENGINE:
        _ENGINE[key] = Gauge(f"bee2bee_engine_{key}", doc, registry=REGISTRY)
    return _ENGINE[key]


def _engine_stats(node) -> Optional[Dict[str, Any]]:
    for svc in node.local_services.values():
        eng = getattr(svc, "engine", None)
        if eng is not None and hasattr(eng, "stats"):
            return eng.stats()
    return None


def refresh(node) -> None:
    """Pull live node/engine state into the gauges (called per scrape)."""
    if node is None:
        return
    import time

    _PEERS.set(len(node.peers))
    _PROVIDERS.set(len(node.list_providers()))
    _UPTIME.set(time.time() - node.start_time)
    stats = _engine_stats(node)
    if stats:
        for key, doc in _ENGINE_STATS:
            val = stats.get(key)
            if isinstance(val, (int, float)):
                _engine_gauge(key, doc).set(val)
        spec = stats.get("spec_decode")
        if isinstance(spec, dict):
            for k, v in spec.items():
                if isinstance(v, (int, float)):
                    _engine_gauge(f"spec_{k}",
                                  f"speculative decoding: {k}").set(v)


def render() -> bytes:
    return generate_latest(REGISTRY)


CONTENT_TYPE = CONTENT_TYPE_LATEST
