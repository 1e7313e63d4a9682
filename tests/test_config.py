from bee2bee_amd.config import (
    DEFAULT_CONFIG,
    get_bootstrap_url,
    load_config,
    save_config,
    set_bootstrap_url,
)


def test_defaults(monkeypatch):
    monkeypatch.delenv("BEE2BEE_BOOTSTRAP", raising=False)
    cfg = load_config()
    assert cfg["bootstrap_url"] == DEFAULT_CONFIG["bootstrap_url"]


def test_file_persistence(monkeypatch):
    monkeypatch.delenv("BEE2BEE_BOOTSTRAP", raising=False)
    set_bootstrap_url("ws://1.2.3.4:9999")
    assert get_bootstrap_url() == "ws://1.2.3.4:9999"


def test_env_overrides_file(monkeypatch):
    set_bootstrap_url("ws://file:1")
    monkeypatch.setenv("BEE2BEE_BOOTSTRAP", "ws://env:2")
    assert get_bootstrap_url() == "ws://env:2"
    cfg = load_config()
    assert cfg["bootstrap_url"] == "ws://env:2"


def test_save_load_extra_keys():
    cfg = load_config()
    cfg["custom"] = 42
    save_config(cfg)
    assert load_config()["custom"] == 42
