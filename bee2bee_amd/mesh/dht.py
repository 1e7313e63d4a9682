"""DHT key-value store for piece/provider discovery.

Behavior parity: reference bee2bee/dht.py — a kademlia-backed store when the
optional dependency is importable, otherwise an in-memory dict with the same
async interface; piece providers live under `piece:<hash>` keys. The key
namespaces and record shapes are the compatibility surface; the structure
here is this package's own (namespace builders + a read-modify-write helper
instead of inline get/set pairs).

Extension beyond the reference: `announce_rank` / `find_ranks` map mesh
peer-ids to RCCL rank/endpoint records so a set of GPU peers on one node can
rendezvous into a torch.distributed (RCCL) group — the data-plane counterpart
of the WS control plane (see parallel/rendezvous.py).
"""
from __future__ import annotations

from typing import Any, Awaitable, Callable, Dict, List, Optional, Tuple

# key namespaces (wire contract)
PIECE_NS = "piece:"
RANK_NS = "rccl:"


def piece_key(content_hash: str) -> str:
    return PIECE_NS + content_hash


def rank_key(group: str) -> str:
    return RANK_NS + group


class InMemoryDHT:
    """Dict-backed fallback exposing the same async surface as kademlia's
    Server (the de-facto test backend, as in the reference)."""

    __slots__ = ("store",)

    def __init__(self) -> None:
        self.store: Dict[str, Any] = {}

    async def set(self, key: str, value: Any) -> None:
        self.store[key] = value

    async def get(self, key: str) -> Any:
        return self.store.get(key)


async def _open_kademlia(port: int, bootstrap: Optional[List[Tuple[str, int]]]):
    """Return a listening kademlia Server, or None when the optional
    dependency is absent (offline images, like this one)."""
    try:
        from kademlia.network import Server  # type: ignore
    except Exception:
        return None
    server = Server()
    await server.listen(port)
    if bootstrap:
        try:
            await server.bootstrap(bootstrap)
        except Exception:
            pass
    return server


class DHTNode:
    """Async KV node; kademlia if available, in-memory otherwise."""

    def __init__(self, host: str = "0.0.0.0", port: int = 8468) -> None:
        self.host = host
        self.port = port
        self.backend: Any = None
        self._server = None

    async def start(self, bootstrap: Optional[List[Tuple[str, int]]] = None) -> None:
        self._server = await _open_kademlia(self.port, bootstrap)
        self.backend = self._server if self._server is not None else InMemoryDHT()

    async def stop(self) -> None:
        server, self._server = self._server, None
        if server is not None:
            try:
                server.stop()
            except Exception:
                pass

    async def set(self, key: str, value: Any) -> None:
        await self.backend.set(key, value)

    async def get(self, key: str) -> Any:
        return await self.backend.get(key)

    async def update(self, key: str, fn: Callable[[Any], Any]) -> Any:
        """Read-modify-write: fn(current) -> new value, which is stored and
        returned. Last-writer-wins under true concurrency, like any DHT —
        callers needing merge semantics use merge-shaped fn's (see
        announce_piece) so concurrent writers converge."""
        value = fn(await self.get(key))
        await self.set(key, value)
        return value


async def announce_piece(dht: DHTNode, content_hash: str, addr: str) -> None:
    def add(current: Optional[List[str]]) -> List[str]:
        providers = list(current or [])
        if addr not in providers:
            providers.append(addr)
        return providers

    await dht.update(piece_key(content_hash), add)


async def find_providers(dht: DHTNode, content_hash: str) -> List[str]:
    return await dht.get(piece_key(content_hash)) or []


# --- RCCL rendezvous records (MI355X extension) ---------------------------


async def announce_rank(
    dht: DHTNode, group: str, peer_id: str, record: Dict[str, Any]
) -> None:
    """Publish a peer's data-plane record (host, port, gpu index, shard range)
    under `rccl:<group>` so peers can form a torch.distributed group."""

    def add(current: Optional[Dict[str, Any]]) -> Dict[str, Any]:
        records = dict(current or {})
        records[peer_id] = record
        return records

    await dht.update(rank_key(group), add)


async def find_ranks(dht: DHTNode, group: str) -> Dict[str, Dict[str, Any]]:
    return await dht.get(rank_key(group)) or {}
