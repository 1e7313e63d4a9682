"""Three-tier configuration: env > ~/.bee2bee/config.json > defaults.

Parity: reference bee2bee/config.py (defaults :11-17, env override :37-39).
Same file path and key names so a node config written by the reference keeps
working here.
"""
from __future__ import annotations

import os
from pathlib import Path
from typing import Any, Dict

from .utils import bee2bee_home

CONFIG_FILE = "config.json"

DEFAULT_CONFIG: Dict[str, Any] = {
    "bootstrap_url": "ws://127.0.0.1:4003",
    "p2p_port": 0,  # 0 = OS-assigned
    "api_port": 4002,
}

# env var -> config key (env always wins)
_ENV_KEYS = {
    "BEE2BEE_BOOTSTRAP": "bootstrap_url",
    "BEE2BEE_PORT": "p2p_port",
    "BEE2BEE_API_PORT": "api_port",
}


def get_config_path() -> Path:
    return bee2bee_home() / CONFIG_FILE


def load_config() -> Dict[str, Any]:
    import json

    path = get_config_path()
    cfg = DEFAULT_CONFIG.copy()
    if path.exists():
        try:
            cfg.update(json.loads(path.read_text(encoding="utf-8")))
        except Exception:
            pass
    for env, key in _ENV_KEYS.items():
        val = os.getenv(env)
        if val is not None:
            cfg[key] = int(val) if key.endswith("port") and val.isdigit() else val
    return cfg


def save_config(config: Dict[str, Any]) -> None:
    from .utils import save_json

    save_json(get_config_path(), config)


def get_bootstrap_url() -> str:
    env = os.getenv("BEE2BEE_BOOTSTRAP")
    if env:
        return env
    return load_config().get("bootstrap_url", DEFAULT_CONFIG["bootstrap_url"])


def set_bootstrap_url(url: str) -> None:
    cfg = load_config()
    cfg["bootstrap_url"] = url
    save_config(cfg)
