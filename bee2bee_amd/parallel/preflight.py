"""Scale-run preflight: catch every multi-GPU failure mode that can be
triggered on ONE GPU (or none) before burning an 8-GPU run on it.

The round-1 review flagged that the `backend="nccl"` path had never
executed on hardware; first-contact RCCL failures (library load, IPC env,
store rendezvous, port clashes, device mapping) are the classic way to lose
a scale run. `python bench.py --preflight [--gpus N]` runs these checks and
prints one JSON line; every check that can run on the current box does.

Checks:
  env          HSA_ENABLE_IPC_MODE_LEGACY=0 (dmabuf IPC — RCCL/CUDA-tensor
               sharing across processes fails without it on this pool)
  master       MASTER_ADDR resolves and MASTER_PORT (or 29500) is bindable
  devices      visible GPU count vs the requested world size
  rccl_self    single-rank nccl init + all_reduce + all_gather + barrier on
               cuda:0 — executes the real RCCL library end to end
  gloo_wiring  2-process torch.distributed rank wiring over loopback
  tunableop    rank-0 tuning table present and seedable to ranks 1..7
"""
from __future__ import annotations

import json
import os
import socket
import sys
from typing import Dict, List

import torch

PASS, WARN, FAIL = "pass", "warn", "fail"


def _check_env() -> Dict:
    v = os.environ.get("HSA_ENABLE_IPC_MODE_LEGACY")
    if v == "0":
        return {"status": PASS, "detail": "HSA_ENABLE_IPC_MODE_LEGACY=0"}
    return {
        "status": WARN,
        "detail": f"HSA_ENABLE_IPC_MODE_LEGACY={v!r}; set it to 0 or RCCL "
                  "cross-process sharing fails with hipIpcGetMemHandle",
    }


def _check_master() -> Dict:
    addr = os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = int(os.environ.get("MASTER_PORT", "29500"))
    try:
        resolved = socket.gethostbyname(addr)
    except OSError as e:
        return {"status": FAIL,
                "detail": f"MASTER_ADDR {addr!r} does not resolve: {e}; "
                          "use --master-addr 127.0.0.1"}
    try:
        with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
            s.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
            s.bind((resolved, port))
    except OSError as e:
        return {"status": FAIL,
                "detail": f"cannot bind {resolved}:{port}: {e}"}
    return {"status": PASS, "detail": f"{addr} -> {resolved}:{port} bindable"}


def _check_devices(n_gpus: int) -> Dict:
    if not torch.cuda.is_available():
        return {"status": WARN, "detail": "no GPU visible (CPU box)"}
    n = torch.cuda.device_count()
    name = torch.cuda.get_device_name(0)
    if n >= n_gpus:
        return {"status": PASS, "detail": f"{n} x {name}"}
    return {"status": WARN,
            "detail": f"{n} x {name} visible, {n_gpus} requested "
                      "(fine if this is the 1-GPU preflight box)"}


def _check_rccl_self() -> Dict:
    """World-1 nccl group on cuda:0: loads RCCL, creates a communicator and
    runs real collectives — catches library/env breakage without a second
    GPU."""
    if not torch.cuda.is_available():
        return {"status": WARN, "detail": "skipped: no GPU"}
    import torch.distributed as dist

    if dist.is_initialized():
        return {"status": WARN, "detail": "skipped: group already initialized"}
    port = _free_port()
    try:
        dist.init_process_group(
            backend="nccl",
            init_method=f"tcp://127.0.0.1:{port}",
            rank=0,
            world_size=1,
        )
        torch.cuda.set_device(0)
        x = torch.ones(1 << 20, device="cuda:0")
        dist.all_reduce(x)
        out = [torch.empty_like(x)]
        dist.all_gather(out, x)
        dist.barrier()
        torch.cuda.synchronize()
        ok = bool((x == 1.0).all()) and bool((out[0] == 1.0).all())
        nccl_ver = ".".join(str(p) for p in torch.cuda.nccl.version())
        return {"status": PASS if ok else FAIL,
                "detail": f"rccl {nccl_ver}: all_reduce/all_gather/barrier ok"}
    except Exception as e:  # noqa: BLE001 — preflight reports, never raises
        return {"status": FAIL, "detail": f"RCCL init/collective failed: {e}"}
    finally:
        import torch.distributed as dist2

        if dist2.is_initialized():
            dist2.destroy_process_group()


def _gloo_worker(rank: int, world: int, port: int) -> None:
    import torch.distributed as dist

    dist.init_process_group(
        backend="gloo",
        init_method=f"tcp://127.0.0.1:{port}",
        rank=rank,
        world_size=world,
    )
    t = torch.tensor([float(rank + 1)])
    dist.all_reduce(t)
    assert t.item() == sum(range(1, world + 1))
    dist.destroy_process_group()


def _check_gloo_wiring() -> Dict:
    """2-process rank wiring over loopback — the same store rendezvous the
    nccl launch uses, minus the GPU."""
    import torch.multiprocessing as mp

    port = _free_port()
    try:
        mp.spawn(_gloo_worker, args=(2, port), nprocs=2, join=True)
        return {"status": PASS, "detail": "2-rank store rendezvous + all_reduce"}
    except Exception as e:  # noqa: BLE001
        return {"status": FAIL, "detail": f"gloo wiring failed: {e}"}


def _check_tunableop(tune_dir: str) -> Dict:
    base = os.path.join(tune_dir, "tunableop_0.csv")
    if not os.path.exists(base):
        return {"status": WARN,
                "detail": "no rank-0 TunableOp table yet (first bench run "
                          "tunes; later runs replay)"}
    missing = [d for d in range(1, 8)
               if not os.path.exists(os.path.join(tune_dir, f"tunableop_{d}.csv"))]
    if missing:
        return {"status": WARN,
                "detail": f"rank-0 table present; ranks {missing} get seeded "
                          "at bench start"}
    return {"status": PASS, "detail": "all 8 per-rank tables present"}


def _free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def run_preflight(n_gpus: int = 8, tune_dir: str = "gpurun_out",
                  skip: List[str] = ()) -> Dict:
    checks = {
        "env": _check_env,
        "master": _check_master,
        "devices": lambda: _check_devices(n_gpus),
        "rccl_self": _check_rccl_self,
        "gloo_wiring": _check_gloo_wiring,
        "tunableop": lambda: _check_tunableop(tune_dir),
    }
    results = {name: fn() for name, fn in checks.items() if name not in skip}
    ok = all(r["status"] != FAIL for r in results.values())
    return {"preflight": "ok" if ok else "failed", "n_gpus": n_gpus,
            "checks": results}


def main() -> int:
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 8
    report = run_preflight(n)
    print(json.dumps(report), flush=True)
    return 0 if report["preflight"] == "ok" else 1


if __name__ == "__main__":
    sys.exit(main())
