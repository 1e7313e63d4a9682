"""Automated port forwarding: UPnP → NAT-PMP → PCP → STUN fallback chain.

Capability parity with reference bee2bee/nat.py (chain :59-64, UPnP :118,
NAT-PMP :207, hand-rolled PCP MAP :258-288, STUN fallback :322, public-IP
cache :411, legacy wrappers :584-609). Implemented from the protocols
directly (SSDP/SOAP for UPnP IGD, RFC 6886 NAT-PMP, RFC 6887 PCP) with no
external dependencies; every step is offline-tolerant and returns a typed
result instead of raising.
"""
from __future__ import annotations

import asyncio
import logging
import re
import socket
import struct
import time
from dataclasses import dataclass, field
from typing import List, Optional, Tuple

logger = logging.getLogger("bee2bee_amd.nat")


@dataclass
class PortForwardingResult:
    success: bool
    method: str = ""
    external_ip: Optional[str] = None
    external_port: Optional[int] = None
    internal_port: Optional[int] = None
    message: str = ""
    manual_instructions: List[str] = field(default_factory=list)


# ------------------------------------------------------------------- helpers

_public_ip_cache: Tuple[float, Optional[str]] = (0.0, None)
_PUBLIC_IP_TTL = 300.0  # 5-min cache, as the reference


async def get_public_ip() -> Optional[str]:
    global _public_ip_cache
    ts, ip = _public_ip_cache
    if ip and time.time() - ts < _PUBLIC_IP_TTL:
        return ip
    from ..utils import get_public_ip as _fetch

    ip = await asyncio.get_running_loop().run_in_executor(None, _fetch)
    if ip:
        _public_ip_cache = (time.time(), ip)
    return ip


def _default_gateway() -> Optional[str]:
    """Default gateway from /proc/net/route (Linux)."""
    try:
        with open("/proc/net/route") as f:
            for line in f.readlines()[1:]:
                parts = line.split()
                if len(parts) >= 3 and parts[1] == "00000000":
                    gw = int(parts[2], 16)
                    return socket.inet_ntoa(struct.pack("<I", gw))
    except Exception:
        pass
    return None


# ---------------------------------------------------------------------- UPnP

_SSDP_ADDR = ("239.255.255.250", 1900)
_SSDP_SEARCH = (
    "M-SEARCH * HTTP/1.1\r\n"
    "HOST: 239.255.255.250:1900\r\n"
    'MAN: "ssdp:discover"\r\n'
    "MX: 2\r\n"
    "ST: urn:schemas-upnp-org:device:InternetGatewayDevice:1\r\n\r\n"
)


async def _ssdp_discover(timeout: float = 2.5) -> Optional[str]:
    """Return the IGD description URL via SSDP multicast, or None."""
    loop = asyncio.get_running_loop()

    def _search() -> Optional[str]:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.settimeout(timeout)
        try:
            s.sendto(_SSDP_SEARCH.encode(), _SSDP_ADDR)
            deadline = time.time() + timeout
            while time.time() < deadline:
                try:
                    data, _ = s.recvfrom(4096)
                except socket.timeout:
                    break
                m = re.search(rb"(?im)^location:\s*(\S+)", data)
                if m:
                    return m.group(1).decode()
        except Exception:
            return None
        finally:
            s.close()
        return None

    return await loop.run_in_executor(None, _search)


async def _upnp_control_url(desc_url: str) -> Optional[Tuple[str, str]]:
    """Fetch the device description and find the WANIPConnection control URL.

    Returns (control_url, service_type)."""
    import httpx
    from urllib.parse import urljoin

    try:
        async with httpx.AsyncClient(timeout=3.0) as client:
            resp = await client.get(desc_url)
            xml = resp.text
    except Exception:
        return None
    for svc_type in (
        "urn:schemas-upnp-org:service:WANIPConnection:1",
        "urn:schemas-upnp-org:service:WANPPPConnection:1",
    ):
        idx = xml.find(svc_type)
        if idx < 0:
            continue
        tail = xml[idx:]
        m = re.search(r"<controlURL>([^<]+)</controlURL>", tail)
        if m:
            return urljoin(desc_url, m.group(1).strip()), svc_type
    return None


async def _upnp_soap(control_url: str, svc_type: str, action: str, args: str) -> Optional[str]:
    import httpx

    body = (
        '<?xml version="1.0"?>'
        '<s:Envelope xmlns:s="http://schemas.xmlsoap.org/soap/envelope/" '
        's:encodingStyle="http://schemas.xmlsoap.org/soap/encoding/">'
        f'<s:Body><u:{action} xmlns:u="{svc_type}">{args}</u:{action}></s:Body>'
        "</s:Envelope>"
    )
    headers = {
        "Content-Type": 'text/xml; charset="utf-8"',
        "SOAPAction": f'"{svc_type}#{action}"',
    }
    try:
        async with httpx.AsyncClient(timeout=4.0) as client:
            resp = await client.post(control_url, content=body, headers=headers)
            if resp.status_code == 200:
                return resp.text
    except Exception:
        pass
    return None


async def try_upnp(port: int, protocol: str = "TCP") -> Optional[PortForwardingResult]:
    desc_url = await _ssdp_discover()
    if not desc_url:
        return None
    ctrl = await _upnp_control_url(desc_url)
    if not ctrl:
        return None
    control_url, svc_type = ctrl
    from ..utils import get_lan_ip

    lan_ip = get_lan_ip()
    args = (
        "<NewRemoteHost></NewRemoteHost>"
        f"<NewExternalPort>{port}</NewExternalPort>"
        f"<NewProtocol>{protocol}</NewProtocol>"
        f"<NewInternalPort>{port}</NewInternalPort>"
        f"<NewInternalClient>{lan_ip}</NewInternalClient>"
        "<NewEnabled>1</NewEnabled>"
        "<NewPortMappingDescription>bee2bee-amd</NewPortMappingDescription>"
        "<NewLeaseDuration>0</NewLeaseDuration>"
    )
    added = await _upnp_soap(control_url, svc_type, "AddPortMapping", args)
    if added is None:
        return None
    ext = await _upnp_soap(control_url, svc_type, "GetExternalIPAddress", "")
    ext_ip = None
    if ext:
        m = re.search(r"<NewExternalIPAddress>([^<]+)</NewExternalIPAddress>", ext)
        if m:
            ext_ip = m.group(1).strip()
    return PortForwardingResult(
        success=True,
        method="UPnP",
        external_ip=ext_ip or await get_public_ip(),
        external_port=port,
        internal_port=port,
    )


async def delete_upnp_mapping(port: int, protocol: str = "TCP") -> bool:
    desc_url = await _ssdp_discover()
    if not desc_url:
        return False
    ctrl = await _upnp_control_url(desc_url)
    if not ctrl:
        return False
    control_url, svc_type = ctrl
    args = (
        "<NewRemoteHost></NewRemoteHost>"
        f"<NewExternalPort>{port}</NewExternalPort>"
        f"<NewProtocol>{protocol}</NewProtocol>"
    )
    return await _upnp_soap(control_url, svc_type, "DeletePortMapping", args) is not None


# -------------------------------------------------------------------- NATPMP


async def try_natpmp(port: int, protocol: str = "TCP", lifetime: int = 3600) -> Optional[PortForwardingResult]:
    """RFC 6886 NAT-PMP: external-address request + mapping request."""
    gw = _default_gateway()
    if not gw:
        return None
    loop = asyncio.get_running_loop()

    def _query() -> Optional[PortForwardingResult]:
        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.settimeout(1.5)
        try:
            # opcode 0: public address
            s.sendto(struct.pack("!BB", 0, 0), (gw, 5351))
            data, _ = s.recvfrom(16)
            if len(data) < 12 or data[1] != 128:
                return None
            ext_ip = socket.inet_ntoa(data[8:12])
            # opcode 1=UDP, 2=TCP mapping
            op = 2 if protocol.upper() == "TCP" else 1
            req = struct.pack("!BBHHHI", op, 0, 0, port, port, lifetime)
            s.sendto(req, (gw, 5351))
            data, _ = s.recvfrom(16)
            if len(data) < 16 or data[1] != 128 + op:
                return None
            result = struct.unpack("!H", data[2:4])[0]
            if result != 0:
                return None
            ext_port = struct.unpack("!H", data[10:12])[0]
            return PortForwardingResult(
                success=True,
                method="NAT-PMP",
                external_ip=ext_ip,
                external_port=ext_port,
                internal_port=port,
            )
        except Exception:
            return None
        finally:
            s.close()

    return await loop.run_in_executor(None, _query)


# ----------------------------------------------------------------------- PCP


async def try_pcp(port: int, protocol: str = "TCP", lifetime: int = 3600) -> Optional[PortForwardingResult]:
    """RFC 6887 PCP MAP request (the NAT-PMP successor)."""
    gw = _default_gateway()
    if not gw:
        return None
    from ..utils import get_lan_ip

    lan_ip = get_lan_ip()
    loop = asyncio.get_running_loop()

    def _query() -> Optional[PortForwardingResult]:
        import secrets

        s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
        s.settimeout(1.5)
        try:
            # PCP header: version 2, opcode MAP(1), reserved, lifetime,
            # client IP as IPv4-mapped IPv6
            client_ip6 = b"\x00" * 10 + b"\xff\xff" + socket.inet_aton(lan_ip)
            nonce = secrets.token_bytes(12)
            proto_num = 6 if protocol.upper() == "TCP" else 17
            # MAP opcode payload: nonce(12) proto(1) rsv(3) int_port(2)
            # ext_port(2) ext_ip(16)
            payload = (
                nonce
                + struct.pack("!B3xHH", proto_num, port, port)
                + b"\x00" * 16
            )
            req = struct.pack("!BBxxI", 2, 1, lifetime) + client_ip6 + payload
            s.sendto(req, (gw, 5351))
            data, _ = s.recvfrom(1100)
            if len(data) < 60 or data[0] != 2 or data[1] != 0x81:
                return None
            result = data[3]
            if result != 0:
                return None
            ext_port = struct.unpack("!H", data[42:44])[0]
            ext_ip_raw = data[44:60]
            if ext_ip_raw[:12] == b"\x00" * 10 + b"\xff\xff":
                ext_ip = socket.inet_ntoa(ext_ip_raw[12:16])
            else:
                ext_ip = socket.inet_ntop(socket.AF_INET6, ext_ip_raw)
            return PortForwardingResult(
                success=True,
                method="PCP",
                external_ip=ext_ip,
                external_port=ext_port,
                internal_port=port,
            )
        except Exception:
            return None
        finally:
            s.close()

    return await loop.run_in_executor(None, _query)


# --------------------------------------------------------------------- chain


def manual_instructions(port: int, protocol: str = "TCP") -> List[str]:
    from ..utils import get_lan_ip

    return [
        "Automatic port forwarding failed. To accept inbound peers:",
        f"1. Open your router admin page (usually http://{_default_gateway() or '192.168.1.1'}).",
        f"2. Forward external {protocol} port {port} to {get_lan_ip()}:{port}.",
        "3. Or run behind a relay/tunnel (the mesh still works outbound-only).",
    ]


class PortForwarder:
    """Fallback chain UPnP → NAT-PMP → PCP → STUN discovery."""

    async def auto_forward_port(self, port: int, protocol: str = "TCP") -> PortForwardingResult:
        for attempt in (try_upnp, try_natpmp, try_pcp):
            try:
                res = await attempt(port, protocol)
            except Exception as e:
                logger.debug("%s failed: %s", attempt.__name__, e)
                res = None
            if res and res.success:
                logger.info("%s mapped port %s -> %s:%s", res.method, port, res.external_ip, res.external_port)
                return res
        # STUN: discovery only (no mapping), still yields the public endpoint
        try:
            from .stun import try_stun

            stun_res = await try_stun()
        except Exception:
            stun_res = None
        if stun_res:
            ip, sport = stun_res
            return PortForwardingResult(
                success=True,
                method="STUN",
                external_ip=ip,
                external_port=sport or port,
                internal_port=port,
                message="discovered via STUN; inbound reachability depends on NAT type",
            )
        return PortForwardingResult(
            success=False,
            method="none",
            internal_port=port,
            manual_instructions=manual_instructions(port, protocol),
        )


async def auto_port_forward(port: int, protocol: str = "TCP") -> PortForwardingResult:
    return await PortForwarder().auto_forward_port(port, protocol)


# legacy wrapper names kept from the reference API (bee2bee/nat.py:584-609)
async def try_upnp_map(port: int, protocol: str = "TCP") -> bool:
    res = await try_upnp(port, protocol)
    return bool(res and res.success)
