"""Minimal RFC 5389 STUN client: public (IP, port) discovery + NAT typing.

Capability parity with reference bee2bee/stun_client.py (binding request :37,
XOR-MAPPED-ADDRESS decode :46, parallel multi-server query :122, NAT-type
detection :138) — implemented from the RFC, asyncio datagram based.
"""
from __future__ import annotations

import asyncio
import os
import secrets
import struct
from typing import List, Optional, Tuple

MAGIC_COOKIE = 0x2112A442
BIND_REQUEST = 0x0001
BIND_RESPONSE = 0x0101
ATTR_MAPPED = 0x0001
ATTR_XOR_MAPPED = 0x0020

DEFAULT_SERVERS = [
    ("stun.l.google.com", 19302),
    ("stun1.l.google.com", 19302),
    ("stun.cloudflare.com", 3478),
]


def create_binding_request() -> Tuple[bytes, bytes]:
    txn = secrets.token_bytes(12)
    header = struct.pack("!HHI12s", BIND_REQUEST, 0, MAGIC_COOKIE, txn)
    return header, txn


def parse_binding_response(data: bytes, txn: bytes) -> Optional[Tuple[str, int]]:
    if len(data) < 20:
        return None
    msg_type, msg_len, cookie, rtxn = struct.unpack("!HHI12s", data[:20])
    if msg_type != BIND_RESPONSE or cookie != MAGIC_COOKIE or rtxn != txn:
        return None
    off = 20
    end = min(len(data), 20 + msg_len)
    while off + 4 <= end:
        atype, alen = struct.unpack("!HH", data[off : off + 4])
        aval = data[off + 4 : off + 4 + alen]
        if atype == ATTR_XOR_MAPPED and len(aval) >= 8:
            family = aval[1]
            if family == 0x01:  # IPv4
                xport = struct.unpack("!H", aval[2:4])[0] ^ (MAGIC_COOKIE >> 16)
                xip = struct.unpack("!I", aval[4:8])[0] ^ MAGIC_COOKIE
                ip = ".".join(str((xip >> s) & 0xFF) for s in (24, 16, 8, 0))
                return ip, xport
        if atype == ATTR_MAPPED and len(aval) >= 8 and aval[1] == 0x01:
            port = struct.unpack("!H", aval[2:4])[0]
            ip = ".".join(str(b) for b in aval[4:8])
            return ip, port
        off += 4 + alen + ((4 - alen % 4) % 4)  # attrs are 32-bit padded
    return None


class _StunProtocol(asyncio.DatagramProtocol):
    def __init__(self, txn: bytes, fut: asyncio.Future) -> None:
        self.txn = txn
        self.fut = fut

    def datagram_received(self, data: bytes, addr) -> None:
        res = parse_binding_response(data, self.txn)
        if res and not self.fut.done():
            self.fut.set_result(res)

    def error_received(self, exc) -> None:
        if not self.fut.done():
            self.fut.set_exception(exc)


async def stun_query(
    server: str, port: int, timeout: float = 2.0, local_port: int = 0
) -> Optional[Tuple[str, int]]:
    """One binding round-trip against one server."""
    loop = asyncio.get_running_loop()
    req, txn = create_binding_request()
    fut: asyncio.Future = loop.create_future()
    try:
        transport, _proto = await loop.create_datagram_endpoint(
            lambda: _StunProtocol(txn, fut),
            local_addr=("0.0.0.0", local_port),
            remote_addr=(server, port),
        )
    except Exception:
        return None
    try:
        transport.sendto(req)
        return await asyncio.wait_for(fut, timeout=timeout)
    except Exception:
        return None
    finally:
        transport.close()


async def get_public_info(
    servers: Optional[List[Tuple[str, int]]] = None, timeout: float = 2.0
) -> Optional[Tuple[str, int]]:
    """Query several servers in parallel, first answer wins."""
    servers = servers or DEFAULT_SERVERS
    tasks = [asyncio.create_task(stun_query(s, p, timeout)) for s, p in servers]
    try:
        for done in asyncio.as_completed(tasks, timeout=timeout + 1.0):
            try:
                res = await done
            except Exception:
                continue
            if res:
                return res
    except asyncio.TimeoutError:
        pass
    finally:
        for t in tasks:
            t.cancel()
    return None


async def detect_nat_type(
    servers: Optional[List[Tuple[str, int]]] = None, timeout: float = 2.0
) -> str:
    """Coarse NAT classification: Blocked / Cone / Symmetric.

    Same local socket against two servers: same mapped endpoint → cone NAT
    (hole punching viable); different → symmetric (relay needed)."""
    servers = (servers or DEFAULT_SERVERS)[:2]
    if len(servers) < 2:
        return "Unknown"
    port = int(os.environ.get("BEE2BEE_STUN_LOCAL_PORT", "0")) or 54320
    a = await stun_query(*servers[0], timeout=timeout, local_port=port)
    b = await stun_query(*servers[1], timeout=timeout, local_port=port)
    if a is None and b is None:
        return "Blocked"
    if a is None or b is None:
        return "Unknown"
    return "Cone" if a == b else "Symmetric"


async def try_stun() -> Optional[Tuple[str, int]]:
    """Reference-named wrapper (bee2bee/nat.py:591)."""
    return await get_public_info()
