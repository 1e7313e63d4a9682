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
