"""NAT/STUN units: packet build/parse are pure functions; network probes are
offline-tolerant type-contract checks (reference test shape:
tests/test_nat_optional.py)."""
import asyncio
import struct

from bee2bee_amd.mesh import stun
from bee2bee_amd.mesh.nat import PortForwardingResult, manual_instructions


def test_binding_request_shape():
    req, txn = stun.create_binding_request()
    assert len(req) == 20 and len(txn) == 12
    mtype, mlen, cookie, rtxn = struct.unpack("!HHI12s", req)
    assert mtype == stun.BIND_REQUEST and mlen == 0
    assert cookie == stun.MAGIC_COOKIE and rtxn == txn


def test_parse_xor_mapped_response():
    _req, txn = stun.create_binding_request()
    ip = (203, 0, 113, 7)
    port = 54321
    xport = port ^ (stun.MAGIC_COOKIE >> 16)
    xip = struct.unpack("!I", bytes(ip))[0] ^ stun.MAGIC_COOKIE
    attr = struct.pack("!HHBBH I"[:10], stun.ATTR_XOR_MAPPED, 8, 0, 0x01, xport, xip)
    resp = struct.pack("!HHI12s", stun.BIND_RESPONSE, len(attr), stun.MAGIC_COOKIE, txn) + attr
    parsed = stun.parse_binding_response(resp, txn)
    assert parsed == ("203.0.113.7", port)


def test_parse_rejects_wrong_txn():
    _req, txn = stun.create_binding_request()
    resp = struct.pack("!HHI12s", stun.BIND_RESPONSE, 0, stun.MAGIC_COOKIE, b"x" * 12)
    assert stun.parse_binding_response(resp, txn) is None


def test_stun_query_offline_tolerant():
    # unroutable server: must return None, never raise
    res = asyncio.run(stun.stun_query("192.0.2.1", 3478, timeout=0.3))
    assert res is None


def test_manual_instructions():
    lines = manual_instructions(4001)
    assert any("4001" in l for l in lines)


def test_port_forwarding_result_dataclass():
    r = PortForwardingResult(success=True, method="UPnP", external_ip="1.2.3.4",
                             external_port=4001)
    assert r.success and r.manual_instructions == []


def test_auto_forward_offline_tolerant():
    from bee2bee_amd.mesh.nat import auto_port_forward

    async def run():
        return await asyncio.wait_for(auto_port_forward(4019), timeout=30)

    res = asyncio.run(run())
    assert isinstance(res, PortForwardingResult)
    assert res.success in (True, False)
