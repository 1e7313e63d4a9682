"""DHT key-value store for piece/provider discovery.

Parity: reference bee2bee/dht.py — a kademlia-backed store when the optional
dependency is importable, otherwise an in-memory dict with the same async
interface; piece providers are announced under `piece:<hash>` keys (:53-64).

Extension beyond the reference: `announce_rank` / `find_ranks` map mesh
peer-ids to RCCL rank/endpoint records so a set of GPU peers on one node can
rendezvous into a torch.distributed (RCCL) group — the data-plane counterpart
of the WS control plane (see parallel/rendezvous.py).
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional, Tuple


class InMemoryDHT:
    def __init__(self) -> None:
        self.store: Dict[str, Any] = {}

    async def set(self, key: str, value: Any) -> None:
        self.store[key] = value

    async def get(self, key: str) -> Any:
        return self.store.get(key)


class DHTNode:
    """Async KV node; kademlia if available, in-memory otherwise."""

    def __init__(self, host: str = "0.0.0.0", port: int = 8468) -> None:
        self.host = host
        self.port = port
        self.backend: Any = None
        self._server = None

    async def start(self, bootstrap: Optional[List[Tuple[str, int]]] = None) -> None:
        try:
            from kademlia.network import Server  # type: ignore
        except Exception:
            self.backend = InMemoryDHT()
            return
        self._server = Server()
        await self._server.listen(self.port)
        self.backend = self._server
        if bootstrap:
            try:
                await self._server.bootstrap(bootstrap)
            except Exception:
                pass

    async def stop(self) -> None:
        if self._server is not None:
            try:
                self._server.stop()
            except Exception:
                pass

    async def set(self, key: str, value: Any) -> None:
        await self.backend.set(key, value)

    async def get(self, key: str) -> Any:
        return await self.backend.get(key)


async def announce_piece(dht: DHTNode, content_hash: str, addr: str) -> None:
    key = f"piece:{content_hash}"
    cur = await dht.get(key) or []
    if addr not in cur:
        cur.append(addr)
    await dht.set(key, cur)


async def find_providers(dht: DHTNode, content_hash: str) -> List[str]:
    return await dht.get(f"piece:{content_hash}") or []


# --- RCCL rendezvous records (MI355X extension) ---------------------------


async def announce_rank(
    dht: DHTNode, group: str, peer_id: str, record: Dict[str, Any]
) -> None:
    """Publish a peer's data-plane record (host, port, gpu index, shard range)
    under `rccl:<group>` so peers can form a torch.distributed group."""
    key = f"rccl:{group}"
    cur = await dht.get(key) or {}
    cur[peer_id] = record
    await dht.set(key, cur)


async def find_ranks(dht: DHTNode, group: str) -> Dict[str, Dict[str, Any]]:
    return await dht.get(f"rccl:{group}") or {}
