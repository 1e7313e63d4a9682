"""`bee2bee-amd doctor` — environment diagnosis for a serving node.

The checks mirror what actually breaks deployments, in the order they
break: interpreter/torch/ROCm versions, GPU visibility, the in-tree HIP
extension, RCCL availability, the multi-process IPC env contract, state-dir
writability and mesh-port bindability. Each check returns (name, ok,
detail); `run_doctor` renders them and exit-codes 1 on any hard failure
(GPU-dependent checks degrade to warnings on CPU-only hosts).
"""
from __future__ import annotations

import os
import shutil
import socket
import sys
from typing import List, Optional, Tuple

Check = Tuple[str, str, str]  # name, status ("ok"|"warn"|"fail"), detail


def _check_python() -> Check:
    v = sys.version_info
    ok = v >= (3, 9)
    return ("python", "ok" if ok else "fail", sys.version.split()[0])


def _check_torch() -> Check:
    try:
        import torch

        hip = getattr(torch.version, "hip", None)
        detail = f"{torch.__version__} (ROCm {hip})" if hip else torch.__version__
        return ("pytorch-rocm", "ok" if hip else "warn", detail)
    except Exception as e:  # noqa: BLE001
        return ("pytorch-rocm", "fail", str(e))


def _check_gpu() -> Check:
    try:
        import torch

        if not torch.cuda.is_available():
            return ("gpu", "warn", "no GPU visible (CPU-only host)")
        n = torch.cuda.device_count()
        name = torch.cuda.get_device_name(0)
        arch = torch.cuda.get_device_properties(0).gcnArchName
        return ("gpu", "ok", f"{n}x {name} ({arch})")
    except Exception as e:  # noqa: BLE001
        return ("gpu", "fail", str(e))


def _check_extension() -> Check:
    try:
        from . import ops

        if ops.hip_available():
            import bee2bee_amd.ops._bee2bee_hip as ext

            return ("hip-extension", "ok", ext.__file__)
        return ("hip-extension", "warn",
                "not built — run `python setup.py build_ext --inplace` "
                "(GPU nodes refuse eager fallback)")
    except Exception as e:  # noqa: BLE001
        return ("hip-extension", "fail", str(e))


def _check_rccl() -> Check:
    try:
        import torch.distributed as dist

        if dist.is_nccl_available():
            return ("rccl", "ok", "torch.distributed nccl backend (RCCL)")
        return ("rccl", "warn", "nccl backend unavailable (gloo only)")
    except Exception as e:  # noqa: BLE001
        return ("rccl", "fail", str(e))


def _check_ipc_env() -> Check:
    val = os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY")
    if val == "0":
        return ("ipc-env", "ok", "HSA_ENABLE_IPC_MODE_LEGACY=0")
    return ("ipc-env", "warn",
            f"HSA_ENABLE_IPC_MODE_LEGACY={val!r} — export 0 for "
            "multi-process GPU work (dmabuf IPC)")


def _check_rocm_tools() -> Check:
    found = [t for t in ("rocm-smi", "rocprofv3", "hipcc")
             if shutil.which(t)]
    missing = [t for t in ("rocm-smi", "rocprofv3", "hipcc")
               if t not in found]
    if not missing:
        return ("rocm-tools", "ok", ", ".join(found))
    return ("rocm-tools", "warn", f"missing: {', '.join(missing)}")


def _check_home() -> Check:
    try:
        from .utils import bee2bee_home

        home = str(bee2bee_home())
        probe = os.path.join(home, ".doctor-probe")
        with open(probe, "w") as f:
            f.write("ok")
        os.remove(probe)
        return ("state-dir", "ok", home)
    except Exception as e:  # noqa: BLE001
        return ("state-dir", "fail", str(e))


def _check_port(port: Optional[int]) -> Check:
    try:
        s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        s.bind(("0.0.0.0", port or 0))
        bound = s.getsockname()[1]
        s.close()
        return ("mesh-port", "ok", f"bindable (probe {bound})")
    except OSError as e:
        return ("mesh-port", "fail", f"port {port}: {e}")


def _check_models() -> Check:
    try:
        from .models.spec import PRESETS

        real = [n for n in PRESETS if not n.startswith(("tiny", "demo"))]
        return ("model-catalog", "ok",
                f"{len(real)} presets ({', '.join(sorted(real)[:4])}, ...)")
    except Exception as e:  # noqa: BLE001
        return ("model-catalog", "fail", str(e))


def collect_checks(port: Optional[int] = None) -> List[Check]:
    return [
        _check_python(),
        _check_torch(),
        _check_gpu(),
        _check_extension(),
        _check_rccl(),
        _check_ipc_env(),
        _check_rocm_tools(),
        _check_home(),
        _check_port(port),
        _check_models(),
    ]


def run_doctor(port: Optional[int] = None, echo=print) -> int:
    """Render all checks; exit code 1 iff any hard failure."""
    checks = collect_checks(port)
    icon = {"ok": "[ok]  ", "warn": "[warn]", "fail": "[FAIL]"}
    width = max(len(c[0]) for c in checks)
    for name, status, detail in checks:
        echo(f"{icon[status]} {name.ljust(width)}  {detail}")
    failures = [c for c in checks if c[1] == "fail"]
    warns = [c for c in checks if c[1] == "warn"]
    echo(f"{len(checks) - len(failures) - len(warns)} ok, "
         f"{len(warns)} warnings, {len(failures)} failures")
    return 1 if failures else 0
