from bee2bee_amd.mesh import wire


def test_hello_shape():
    msg = wire.hello("peer-1", "ws://h:1", "EU", {"cpu": 1}, {"hf": {"models": ["m"]}})
    assert msg["type"] == "hello"
    # field names are the reference wire protocol's
    for key in ("peer_id", "addr", "region", "metrics", "services", "api_port"):
        assert key in msg


def test_gen_request_legacy_keys():
    msg = wire.gen_request("rid-1", "hi", "m", max_new_tokens=7)
    assert msg["max_new_tokens"] == 7
    assert msg["max_tokens"] == 7  # legacy duplicate for old peers


def test_request_params_normalization():
    # legacy max_tokens key
    p = wire.request_params({"prompt": "x", "max_tokens": 5})
    assert p["max_new_tokens"] == 5
    # modern key wins
    p = wire.request_params({"prompt": "x", "max_new_tokens": 9, "max_tokens": 5})
    assert p["max_new_tokens"] == 9
    # defaults
    p = wire.request_params({})
    assert p["max_new_tokens"] == 2048 and p["temperature"] == 0.7


def test_request_id_legacy():
    assert wire.request_id({"rid": "a"}) == "a"
    assert wire.request_id({"task_id": "b"}) == "b"


def test_terminal_types():
    assert set(wire.TERMINAL_TYPES) == {"gen_result", "gen_success", "gen_error"}


def test_reference_shaped_messages_parse():
    """Messages exactly as the reference emits them must normalize cleanly
    (quirk coverage: legacy task_id ids, max_tokens-only, no-services
    hello)."""
    # reference _handle_gen_request reads rid or task_id, max_tokens default
    # 2048, temperature default 0.7 (p2p_runtime.py:574-586)
    ref_req = {
        "type": "gen_request",
        "task_id": "task-42",
        "svc": "hf",
        "model": "llama3",
        "prompt": "hi",
        "max_tokens": 128,
        "temperature": 0.2,
        "stream": True,
    }
    assert wire.request_id(ref_req) == "task-42"
    p = wire.request_params(ref_req)
    assert p["max_new_tokens"] == 128 and p["temperature"] == 0.2

    # reference buffered provider answers with gen_success (Q1): it must be
    # terminal for us
    assert "gen_success" in wire.TERMINAL_TYPES

    # hello without services/metrics (early-handshake JS bridge shape)
    bare = {"type": "hello", "peer_id": "p1", "addr": "ws://x:1"}
    assert bare.get("services") is None  # nodes must tolerate absence


def test_request_params_optional_sampling_knobs():
    from bee2bee_amd.mesh import wire

    p = wire.request_params({"prompt": "x", "top_p": 0.4,
                             "repetition_penalty": 1.2})
    assert p["top_p"] == 0.4 and p["repetition_penalty"] == 1.2
    p2 = wire.request_params({"prompt": "x"})
    assert "top_p" not in p2  # absent stays absent (engine defaults apply)


def test_wire_constants_pinned_to_reference():
    """Protocol constants are the compatibility surface (reference
    p2p_runtime.py:176 frame size, :831 timeout, :658/:654/:837/:792 error
    strings) — pin them so refactors cannot drift the wire."""
    assert wire.MAX_FRAME == 32 * 1024 * 1024
    assert wire.REQUEST_TIMEOUT == 300.0
    assert wire.PING_INTERVAL == 15.0
    assert wire.ERR_NO_NODE == "consensus_deadlock: no_node_available"
    assert wire.ERR_RELAY == "relay_link_failure"
    assert wire.ERR_TIMEOUT == "request_timed_out"
    assert wire.ERR_NOT_CONNECTED == "provider_not_connected"
    assert set(wire.TERMINAL_TYPES) == {"gen_result", "gen_success",
                                        "gen_error"}
