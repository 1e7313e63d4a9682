import json

from click.testing import CliRunner

from bee2bee_amd.__main__ import cli
from bee2bee_amd.utils import (
    get_lan_ip,
    get_system_metrics,
    new_id,
    save_json,
    load_json,
    sha256_hex,
)


def test_new_id_unique_and_shaped():
    ids = {new_id("peer") for _ in range(200)}
    assert len(ids) == 200
    assert all(i.startswith("peer-") and len(i) == 13 for i in ids)


def test_sha256_deterministic():
    assert sha256_hex("abc") == sha256_hex("abc")
    assert sha256_hex("abc") != sha256_hex("abd")


def test_atomic_json_roundtrip(tmp_path):
    p = tmp_path / "x.json"
    save_json(p, {"a": [1, 2]})
    assert load_json(p, None) == {"a": [1, 2]}
    assert load_json(tmp_path / "missing.json", 42) == 42


def test_lan_ip_shape():
    ip = get_lan_ip()
    assert ip.count(".") == 3


def test_metrics_keys():
    m = get_system_metrics()
    # dashboard key names preserved from the reference
    for k in ("throughput", "memory_percent", "gpu_percent", "trust_score"):
        assert k in m


def test_cli_help_lists_commands():
    res = CliRunner().invoke(cli, ["--help"])
    assert res.exit_code == 0
    for cmd in ("serve-hf", "serve-ollama", "serve-hf-remote", "register", "config"):
        assert cmd in res.output


def test_cli_config_set(tmp_path, monkeypatch):
    monkeypatch.setenv("BEE2BEE_HOME", str(tmp_path))
    res = CliRunner().invoke(cli, ["config", "bootstrap_url", "ws://1.1.1.1:9"])
    assert res.exit_code == 0
    cfg = json.loads((tmp_path / "config.json").read_text())
    assert cfg["bootstrap_url"] == "ws://1.1.1.1:9"
