"""Cross-cutting utilities: home dir, ids, atomic JSON, host metrics.

Capability parity with reference bee2bee/utils.py (home dir :11, atomic save
:37, new_id :43, LAN IP :68, system metrics :120-135) — but metrics here are
REAL: GPU utilization comes from amd-smi/rocm-smi (not nvidia-smi), and
`throughput` is the engine's measured tokens/sec (reference simulated it as
cpu%*0.85).
"""
from __future__ import annotations

import hashlib
import json
import logging
import os
import shutil
import socket
import subprocess
import time
import uuid
from pathlib import Path
from typing import Any, Dict

logger = logging.getLogger("bee2bee_amd")


def setup_logging(level: str | None = None) -> None:
    lvl = (level or os.environ.get("LOG_LEVEL", "INFO")).upper()
    logging.basicConfig(
        level=getattr(logging, lvl, logging.INFO),
        format="%(asctime)s %(levelname)s %(name)s: %(message)s",
    )


def bee2bee_home() -> Path:
    """~/.bee2bee (BEE2BEE_HOME override) — kept identical to the reference
    so configs/join-state written by either implementation interoperate."""
    base = os.environ.get("BEE2BEE_HOME")
    p = Path(base) if base else Path.home() / ".bee2bee"
    p.mkdir(parents=True, exist_ok=True)
    return p


def load_json(path: Path, default: Any) -> Any:
    if not path.exists():
        return default
    try:
        return json.loads(path.read_text(encoding="utf-8"))
    except Exception:
        return default


def save_json(path: Path, obj: Any) -> None:
    """Atomic write (tmp + rename)."""
    tmp = path.with_suffix(path.suffix + ".tmp")
    tmp.write_text(json.dumps(obj, indent=2, ensure_ascii=False), encoding="utf-8")
    tmp.replace(path)


def new_id(prefix: str) -> str:
    """Short unique id, `<prefix>-<8 hex>` (wire-compatible shape with the
    reference's peer/request ids)."""
    return f"{prefix}-{uuid.uuid4().hex[:8]}"


def now_ms() -> int:
    return int(time.time() * 1000)


def sha256_hex(s: str) -> str:
    return hashlib.sha256(s.encode("utf-8")).hexdigest()


def sha256_hex_bytes(data: bytes) -> str:
    return hashlib.sha256(data).hexdigest()


def get_lan_ip() -> str:
    """LAN IP via the connected-UDP-socket trick (no traffic is sent)."""
    s = socket.socket(socket.AF_INET, socket.SOCK_DGRAM)
    try:
        s.connect(("10.255.255.255", 1))
        return s.getsockname()[0]
    except Exception:
        return "127.0.0.1"
    finally:
        s.close()


def get_public_ip(timeout: float = 3.0) -> str | None:
    """Public IP via ipify (None when offline)."""
    import urllib.request

    try:
        return (
            urllib.request.urlopen("https://api.ipify.org", timeout=timeout)
            .read()
            .decode("utf-8")
        )
    except Exception:
        return None


# ---------------------------------------------------------------------------
# Host / GPU metrics (real measurements; dashboard key names kept from the
# reference so Supabase rows and /peers consumers keep working).
# ---------------------------------------------------------------------------

_engine_throughput_tps: float = 0.0


def report_engine_throughput(tokens_per_sec: float) -> None:
    """Engines call this so mesh metrics carry real tokens/sec."""
    global _engine_throughput_tps
    _engine_throughput_tps = float(tokens_per_sec)


def get_gpu_usage() -> float:
    """GPU utilization percent via amd-smi / rocm-smi (0.0 when absent)."""
    for tool, args, parse in (
        ("amd-smi", ["metric", "--usage", "--csv"], "amdsmi"),
        ("rocm-smi", ["--showuse", "--csv"], "rocmsmi"),
    ):
        if not shutil.which(tool):
            continue
        try:
            out = subprocess.check_output(
                [tool, *args], stderr=subprocess.DEVNULL, timeout=5
            ).decode()
            vals = []
            for line in out.splitlines():
                for cell in line.split(","):
                    cell = cell.strip().rstrip("%")
                    try:
                        v = float(cell)
                    except ValueError:
                        continue
                    if 0.0 <= v <= 100.0:
                        vals.append(v)
                        break
            if vals:
                return max(vals)
        except Exception:
            continue
    return 0.0


def get_system_metrics() -> Dict[str, float]:
    """Real-time metrics with the dashboard's key names.

    `throughput` is the engine-measured tokens/sec (reference fabricated this
    from CPU load, bee2bee/utils.py:129); `trust_score` is a static 1.0 here
    — we do not simulate trust."""
    try:
        import psutil

        cpu = psutil.cpu_percent(interval=None)
        ram = psutil.virtual_memory().percent
    except Exception:
        cpu, ram = 0.0, 0.0
    return {
        "throughput": round(_engine_throughput_tps, 1),
        "cpu_percent": cpu,
        "memory_percent": ram,
        "gpu_percent": get_gpu_usage(),
        "trust_score": 1.0,
    }


def is_colab() -> bool:
    import sys

    return "google.colab" in sys.modules
