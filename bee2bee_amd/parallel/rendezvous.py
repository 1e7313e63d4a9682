"""Mesh peer-id -> RCCL rank rendezvous.

Peers that agree to form a data-plane group (pipeline stages, expert
shards) publish records to the DHT under `rccl:<group>` (mesh/dht.py
announce_rank) and then initialize one torch.distributed process group.
The rank order is deterministic: sorted peer-ids; the lowest peer-id's
host:port is the TCP store endpoint. Backend is "nccl" (RCCL on ROCm) for
GPU groups and "gloo" for CPU tests — both exercised by
tests/test_parallel_cpu.py with world_size 2.
"""
from __future__ import annotations

import datetime
import logging
from typing import Any, Dict, Optional, Tuple

import torch
import torch.distributed as dist

from ..mesh.dht import DHTNode, announce_rank, find_ranks

logger = logging.getLogger("bee2bee_amd.parallel")


def rank_order(records: Dict[str, Dict[str, Any]]) -> Tuple[list, str]:
    """Deterministic rank assignment: sorted peer-ids; returns (ordered
    peer ids, master endpoint 'host:port')."""
    peers = sorted(records)
    if not peers:
        raise ValueError("no rendezvous records")
    master = records[peers[0]]
    return peers, f"{master.get('host', '127.0.0.1')}:{master.get('port', 29500)}"


async def join_group(
    dht: DHTNode,
    group: str,
    peer_id: str,
    host: str,
    port: int,
    world_size: int,
    gpu: int = 0,
    poll_s: float = 0.2,
    timeout_s: float = 120.0,
) -> Tuple[int, str]:
    """Announce this peer and wait until `world_size` peers are present.
    Returns (rank, master_endpoint). Call init_process_group after."""
    import asyncio

    await announce_rank(
        dht, group, peer_id, {"host": host, "port": port, "gpu": gpu}
    )
    deadline = asyncio.get_event_loop().time() + timeout_s
    while True:
        records = await find_ranks(dht, group)
        if len(records) >= world_size:
            peers, master = rank_order(records)
            return peers.index(peer_id), master
        if asyncio.get_event_loop().time() > deadline:
            raise TimeoutError(
                f"rendezvous {group}: {len(records)}/{world_size} peers"
            )
        await asyncio.sleep(poll_s)


async def join_group_mesh(
    node: Any,
    group: str,
    peer_id: str,
    host: str,
    port: int,
    world_size: int,
    gpu: int = 0,
    poll_s: float = 0.2,
    timeout_s: float = 120.0,
) -> Tuple[int, str]:
    """join_group over a MeshNode's replicated DHT: the announce broadcasts
    one hop over the WS control plane (wire.DHT_SET), so mesh-connected
    peers rendezvous without the optional kademlia dependency."""
    import asyncio

    key = f"rccl:{group}"
    cur = await node.dht.get(key) or {}
    cur = dict(cur)
    cur[peer_id] = {"host": host, "port": port, "gpu": gpu}
    await node.dht_set(key, cur)
    deadline = asyncio.get_event_loop().time() + timeout_s
    while True:
        records = await find_ranks(node.dht, group)
        if len(records) >= world_size:
            peers, master = rank_order(records)
            return peers.index(peer_id), master
        if asyncio.get_event_loop().time() > deadline:
            raise TimeoutError(
                f"rendezvous {group}: {len(records)}/{world_size} peers"
            )
        # re-broadcast our record: a peer that connected after our announce
        # missed the one-hop replication
        await node.dht_set(key, {peer_id: cur[peer_id]})
        await asyncio.sleep(poll_s)


def init_distributed(
    rank: int,
    world_size: int,
    master: str = "127.0.0.1:29500",
    backend: Optional[str] = None,
    device: Optional[torch.device] = None,
) -> None:
    if backend is None:
        backend = "nccl" if (device is not None and device.type == "cuda") else "gloo"
    host, port = master.rsplit(":", 1)
    dist.init_process_group(
        backend=backend,
        init_method=f"tcp://{host}:{port}",
        rank=rank,
        world_size=world_size,
        timeout=datetime.timedelta(seconds=300),
    )
    logger.info("distributed group up: rank %d/%d via %s", rank, world_size, backend)
