"""Global node directory client.

A node publishes itself to the public directory so browsers / bridges can
find it. Two transports exist, resolved from the environment:

  * direct Supabase REST upsert into the `active_nodes` table, or
  * a relay POST to a cluster entrypoint's `/api/nodes/register`.

The row shape, env-var names, REST paths and the
`Prefer: resolution=merge-duplicates` upsert header are pinned by directory
compatibility (reference: bee2bee/registry.py, app SUPABASE_SCHEMA.sql:70-79
— a node registered by this framework appears in the same public listing).
The implementation follows this package's typed-builder idiom (mesh/wire.py):
the row is a frozen dataclass, the transport a resolved-once value object.
"""
from __future__ import annotations

import dataclasses
import logging
import os
from datetime import datetime, timezone
from typing import List, Mapping, Optional, Sequence, Tuple

import httpx

logger = logging.getLogger("bee2bee_amd.registry")

# directory-compatibility constants (category-b pinned surface)
_ENV_SUPABASE_URL = ("VITE_SUPABASE_URL", "SUPABASE_URL")
_ENV_SUPABASE_KEY = ("VITE_SUPABASE_ANON_KEY", "SUPABASE_ANON_KEY")
_ENV_ENTRYPOINT = "BEE2BEE_ENTRYPOINT"
_SUPABASE_TABLE_PATH = "/rest/v1/active_nodes"
_ENTRYPOINT_REGISTER_PATH = "/api/nodes/register"
_SYNC_TIMEOUT_S = 5.0


def _first_env(names: Sequence[str]) -> Optional[str]:
    for n in names:
        v = os.getenv(n)
        if v:
            return v
    return None


@dataclasses.dataclass(frozen=True)
class NodeRow:
    """One `active_nodes` directory row (field names are the wire contract)."""

    peer_id: str
    addr: str
    models: Tuple[str, ...]
    latency_ms: float = 0.0
    region: str = "Auto"
    tag: str = "global"
    metrics: Optional[Mapping] = None

    def payload(self) -> dict:
        return {
            "peer_id": self.peer_id,
            "addr": self.addr,
            "models": list(self.models),
            "latency_ms": self.latency_ms,
            "region": self.region,
            "tag": self.tag,
            "metrics": dict(self.metrics) if self.metrics is not None else None,
            "last_seen": datetime.now(timezone.utc).isoformat(),
        }


@dataclasses.dataclass(frozen=True)
class _Transport:
    """Where rows go and with which headers (resolved once at startup)."""

    url: str
    headers: Mapping[str, str]
    kind: str  # "supabase" | "entrypoint"


def resolve_transport(entrypoint_url: Optional[str] = None) -> Optional[_Transport]:
    """Pick the directory transport from explicit arg / environment.

    Supabase credentials win; an entrypoint relay is the fallback; neither
    means the node runs private/offline (returns None)."""
    base = _first_env(_ENV_SUPABASE_URL)
    key = _first_env(_ENV_SUPABASE_KEY)
    if base and key:
        return _Transport(
            url=base.rstrip("/") + _SUPABASE_TABLE_PATH,
            headers={
                "apikey": key,
                "Authorization": f"Bearer {key}",
                "Content-Type": "application/json",
                "Prefer": "resolution=merge-duplicates",
            },
            kind="supabase",
        )
    entry = entrypoint_url or os.getenv(_ENV_ENTRYPOINT)
    if entry:
        return _Transport(
            url=entry.rstrip("/") + _ENTRYPOINT_REGISTER_PATH,
            headers={"Content-Type": "application/json"},
            kind="entrypoint",
        )
    return None


class RegistryClient:
    """Async upsert client for the public node directory."""

    def __init__(self, entrypoint_url: Optional[str] = None) -> None:
        self._transport = resolve_transport(entrypoint_url)
        if self._transport is None:
            logger.info("no registry credentials; node runs in private/offline mode")

    @property
    def enabled(self) -> bool:
        return self._transport is not None

    @property
    def api_url(self) -> Optional[str]:
        return self._transport.url if self._transport else None

    @property
    def headers(self) -> Mapping[str, str]:
        return self._transport.headers if self._transport else {}

    async def sync_row(self, row: NodeRow) -> bool:
        """Upsert one directory row; False on any failure (never raises)."""
        t = self._transport
        if t is None:
            return False
        try:
            async with httpx.AsyncClient(timeout=_SYNC_TIMEOUT_S) as client:
                resp = await client.post(t.url, json=row.payload(), headers=t.headers)
        except Exception as e:  # noqa: BLE001 — directory sync is best-effort
            logger.error("registry connection error: %s", e)
            return False
        if resp.status_code in (200, 201):
            return True
        logger.error("registry sync failed: %s %s", resp.status_code, resp.text)
        return False

    async def sync_node(
        self,
        peer_id: str,
        address: str,
        models: List[str],
        latency: float = 0.0,
        tag: str = "global",
        region: str = "Auto",
        metrics: Optional[dict] = None,
    ) -> bool:
        """Compatibility wrapper keeping the reference call signature."""
        return await self.sync_row(
            NodeRow(
                peer_id=peer_id,
                addr=address,
                models=tuple(models),
                latency_ms=latency,
                region=region,
                tag=tag,
                metrics=metrics,
            )
        )
