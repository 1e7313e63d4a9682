"""L6 web layer: browser-facing gateway + mesh bridge + usage persistence.

Reimplements the reference's JS web stack natively (Python/FastAPI):
  * gateway.py — /api/p2p/{register,generate,status,global_metrics}
    (reference: app/api/index.js:16-216) + a static chat dashboard
  * bridge.py  — registry-driven WS client with direct-HTTP-first request
    routing and seed rotation (reference: app/api/bridge.js:23-349)
  * store.py   — `messages` / `active_nodes` / `system_stats` persistence
    (reference schema: SUPABASE_SCHEMA.sql:9-38)
"""
from .bridge import MeshBridge  # noqa: F401
from .store import WebStore  # noqa: F401
