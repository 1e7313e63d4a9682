"""Usage/message persistence against the public directory database.

The reference persists per-chat token estimates into a Supabase `messages`
table and aggregates them through the `system_stats` view (schema:
SUPABASE_SCHEMA.sql:9-38; writes: app/api/index.js:66-86,197-210; reads:
:164-188). This client speaks the same REST surface; the base URL/key are
injectable so tests run against a local mock.

All calls are best-effort: persistence failures never fail a chat.
"""
from __future__ import annotations

import logging
import os
from typing import Any, Dict, List, Optional

import httpx

logger = logging.getLogger("bee2bee_amd.web")

# row sentinels shared with the reference gateway
GLOBAL_METRICS_NODE = "GLOBAL_METRICS"
METRIC_LOG_CONTENT = "[Metric Log]"
NETWORK_PULSE_CONTENT = "[Network Pulse]"


def _env_first(*names: str) -> Optional[str]:
    for n in names:
        v = os.getenv(n)
        if v:
            return v
    return None


class WebStore:
    """REST client for the `messages`/`active_nodes`/`system_stats` tables."""

    def __init__(self, base_url: Optional[str] = None,
                 key: Optional[str] = None) -> None:
        self.base_url = (base_url or
                         _env_first("VITE_SUPABASE_URL", "SUPABASE_URL"))
        self.key = key or _env_first("VITE_SUPABASE_ANON_KEY",
                                     "SUPABASE_ANON_KEY")
        if self.base_url:
            self.base_url = self.base_url.rstrip("/")

    @property
    def enabled(self) -> bool:
        return bool(self.base_url and self.key)

    def _headers(self, upsert: bool = False) -> Dict[str, str]:
        h = {
            "apikey": self.key or "",
            "Authorization": f"Bearer {self.key}",
            "Content-Type": "application/json",
        }
        if upsert:
            h["Prefer"] = "resolution=merge-duplicates"
        return h

    async def insert_message(
        self,
        node_id: str,
        tokens: int,
        content: str = METRIC_LOG_CONTENT,
        role: str = "assistant",
        cost: float = 0.0,
        metadata: Optional[Dict[str, Any]] = None,
    ) -> bool:
        """One `messages` row — the per-chat token tally."""
        if not self.enabled or tokens <= 0:
            return False
        row = {
            "node_id": node_id,
            "content": content,
            "role": role,
            "tokens": int(tokens),
        }
        if cost:
            row["cost"] = cost
        if metadata:
            row["metadata"] = metadata
        return await self._post("/rest/v1/messages", row)

    async def system_stats(self) -> Dict[str, int]:
        """Aggregate usage via the system_stats view; zeros when offline."""
        empty = {"users": 0, "chats": 0, "tokens": 0}
        if not self.enabled:
            return empty
        try:
            async with httpx.AsyncClient(timeout=5.0) as client:
                resp = await client.get(
                    f"{self.base_url}/rest/v1/system_stats",
                    params={"select": "*"}, headers=self._headers())
            if resp.status_code == 200:
                rows = resp.json()
                if rows:
                    return {
                        "tokens": rows[0].get("total_tokens", 0),
                        "chats": rows[0].get("total_chats", 0),
                        "users": rows[0].get("total_users", 0),
                    }
        except Exception as e:  # noqa: BLE001
            logger.debug("system_stats fetch failed: %s", e)
        return empty

    async def active_nodes(self, limit: int = 20) -> List[Dict[str, Any]]:
        """Freshest directory rows (reference bridge.js:140)."""
        if not self.enabled:
            return []
        try:
            async with httpx.AsyncClient(timeout=5.0) as client:
                resp = await client.get(
                    f"{self.base_url}/rest/v1/active_nodes",
                    params={"select": "*", "order": "last_seen.desc",
                            "limit": str(limit)},
                    headers=self._headers())
            if resp.status_code == 200:
                return list(resp.json())
        except Exception as e:  # noqa: BLE001
            logger.debug("active_nodes fetch failed: %s", e)
        return []

    async def upsert_node(self, payload: Dict[str, Any]) -> bool:
        """Directory upsert (same row shape as mesh/registry.py NodeRow)."""
        if not self.enabled:
            return False
        return await self._post("/rest/v1/active_nodes", payload, upsert=True)

    async def _post(self, path: str, payload: Dict[str, Any],
                    upsert: bool = False) -> bool:
        try:
            async with httpx.AsyncClient(timeout=5.0) as client:
                resp = await client.post(
                    f"{self.base_url}{path}", json=payload,
                    headers=self._headers(upsert))
            if resp.status_code in (200, 201):
                return True
            logger.warning("store POST %s -> %s", path, resp.status_code)
        except Exception as e:  # noqa: BLE001
            logger.debug("store POST %s failed: %s", path, e)
        return False
