"""Global node directory client (Supabase `active_nodes` table or an
entrypoint relay).

Parity: reference bee2bee/registry.py — same env vars, REST path, payload row
shape and `Prefer: resolution=merge-duplicates` upsert header, so a node
registered by this framework appears in the same public directory.
"""
from __future__ import annotations

import logging
import os
from datetime import datetime, timezone
from typing import List, Optional

import httpx

logger = logging.getLogger("bee2bee_amd.registry")


class RegistryClient:
    def __init__(self, entrypoint_url: Optional[str] = None) -> None:
        self.supabase_url = os.getenv("VITE_SUPABASE_URL") or os.getenv("SUPABASE_URL")
        self.supabase_key = os.getenv("VITE_SUPABASE_ANON_KEY") or os.getenv(
            "SUPABASE_ANON_KEY"
        )
        self.entrypoint_url = entrypoint_url or os.getenv("BEE2BEE_ENTRYPOINT")
        self.enabled = bool(
            (self.supabase_url and self.supabase_key) or self.entrypoint_url
        )
        self.api_url: Optional[str] = None
        self.headers = {}
        if self.enabled:
            if self.supabase_url and self.supabase_key:
                self.api_url = f"{self.supabase_url.rstrip('/')}/rest/v1/active_nodes"
                self.headers = {
                    "apikey": self.supabase_key,
                    "Authorization": f"Bearer {self.supabase_key}",
                    "Content-Type": "application/json",
                    "Prefer": "resolution=merge-duplicates",
                }
            else:
                self.api_url = f"{self.entrypoint_url.rstrip('/')}/api/nodes/register"
                self.headers = {"Content-Type": "application/json"}
        else:
            logger.info("no registry credentials; node runs in private/offline mode")

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
        if not self.enabled:
            return False
        payload = {
            "peer_id": peer_id,
            "addr": address,
            "models": models,
            "latency_ms": latency,
            "region": region,
            "tag": tag,
            "metrics": metrics,
            "last_seen": datetime.now(timezone.utc).isoformat(),
        }
        try:
            async with httpx.AsyncClient() as client:
                resp = await client.post(
                    self.api_url, json=payload, headers=self.headers, timeout=5.0
                )
                if resp.status_code in (200, 201):
                    return True
                logger.error("registry sync failed: %s %s", resp.status_code, resp.text)
        except Exception as e:
            logger.error("registry connection error: %s", e)
        return False
