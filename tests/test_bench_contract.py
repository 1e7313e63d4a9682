"""Driver contract tests for bench.py: single-process JSON line shape and
the torch.distributed.run multi-process launch (gloo on CPU; the GPU box
takes the same path over RCCL)."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _last_json(out: str):
    for line in reversed(out.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out[-2000:]}")


REQUIRED = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


@pytest.mark.timeout(300)
def test_bench_single_process_contract():
    out = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--batch", "2",
         "--prompt-len", "8", "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    res = _last_json(out.stdout)
    assert REQUIRED <= set(res)
    assert res["n_gpus"] == 1 and res["steps"] == 2 and res["warmup"] == 1
    assert res["scaling"] == "weak" and res["higher_is_better"] is True
    assert res["config"]["global_batch"] == 2
    assert res["value"] > 0


@pytest.mark.timeout(300)
def test_bench_multi_process_contract():
    """The driver's N>1 launch shape: torch.distributed.run, one JSON line
    from rank 0, whole-job aggregate value, MAX-over-ranks timing."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29753", "bench.py", "--gpus", "2", "--model",
         "tiny", "--batch", "2", "--prompt-len", "8", "--steps", "2",
         "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    res = _last_json(out.stdout)
    assert res["n_gpus"] == 2
    assert res["config"]["parallelism"] == "replica-dp2"
    assert res["config"]["global_batch"] == 4  # whole-job aggregate


@pytest.mark.timeout(300)
@pytest.mark.parametrize("mode,model", [("pp", "tiny"), ("tp", "tiny"), ("cp", "tiny"), ("ep", "tiny-moe")])
def test_bench_pp_contract(mode, model):
    """scripts/bench_pp.py (the 8-GPU model-parallel bench) launches under
    torch.distributed.run and emits the JSON contract (gloo dry run)."""
    port = {"pp": "29771", "tp": "29772", "cp": "29773", "ep": "29774"}[mode]
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", port, "scripts/bench_pp.py", "--mode", mode,
         "--model", model, "--batch", "2", "--prompt-len", "8",
         "--steps", "2", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    res = _last_json(out.stdout)
    assert res["scaling"] == "strong"
    assert res["config"]["parallelism"] == f"{mode}2"
    assert res["value"] > 0


@pytest.mark.timeout(240)
def test_preflight_cpu():
    """`bench.py --preflight` must pass on a GPU-less box (rccl/device
    checks degrade to warn; master/env/gloo wiring are hard checks)."""
    import json
    import subprocess

    r = subprocess.run(
        [sys.executable, "bench.py", "--preflight", "--gpus", "8"],
        cwd=REPO, capture_output=True, text=True, timeout=220,
    )
    assert r.returncode == 0, r.stdout + r.stderr
    report = json.loads(r.stdout.strip().splitlines()[-1])
    assert report["preflight"] == "ok"
    assert report["checks"]["gloo_wiring"]["status"] == "pass"
    assert report["checks"]["master"]["status"] == "pass"


def test_preflight_unit_checks():
    """Individual preflight checks behave sanely off-GPU."""
    from bee2bee_amd.parallel import preflight as pf

    assert pf._check_env()["status"] in ("pass", "warn")
    m = pf._check_master()
    assert m["status"] == "pass", m
    d = pf._check_devices(8)
    assert d["status"] in ("pass", "warn")
    r = pf._check_rccl_self()
    assert r["status"] in ("pass", "warn")  # warn: no GPU here
    t = pf._check_tunableop(str(REPO) + "/definitely-missing-dir")
    assert t["status"] == "warn"
    report = pf.run_preflight(2, skip=["gloo_wiring"])
    assert "gloo_wiring" not in report["checks"]
    assert report["preflight"] in ("ok", "failed")


@pytest.mark.timeout(300)
@pytest.mark.parametrize("mode", ["tp", "cp", "pp", "ep"])
def test_serve_parallel_oneshot_contract(mode):
    """Distributed-serving launch shape: torch.distributed.run world 2,
    rank 0 drives one request through LockstepServer, followers serve it
    and drain cleanly."""
    port = {"tp": 29881, "cp": 29882, "pp": 29883, "ep": 29884}[mode]
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port), "scripts/serve_parallel.py",
         "--mode", mode, "--max-batch", "2", "--max-seq-len", "64",
         "--oneshot-prompt", "hello mesh"],
        cwd=REPO, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "ONESHOT_RESULT tokens=8 world=2" in out.stdout
    assert "rank 1: served 1 requests" in out.stdout


def test_gpu_test_files_are_properly_gated():
    """Meta-test: every tests/*_gpu.py file must mark itself `gpu` AND
    module-level-skip off-GPU — an unguarded file would break CPU CI, an
    unmarked one would silently run (and fail) in the CPU lane."""
    import glob

    files = sorted(glob.glob(os.path.join(REPO, "tests", "test_*_gpu.py")))
    assert files, "gpu test files disappeared?"
    for f in files:
        src = open(f).read()
        assert "pytestmark = pytest.mark.gpu" in src, f
        assert "allow_module_level=True" in src, f
        assert "torch.cuda.is_available()" in src, f


@pytest.mark.timeout(300)
def test_serve_parallel_cli_launch():
    """`python -m bee2bee_amd serve-parallel` wraps the torchrun launch of
    the packaged entry (oneshot via module path, gloo world 2)."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29885", "-m", "bee2bee_amd.parallel.serve_main",
         "--mode", "tp", "--max-batch", "2", "--max-seq-len", "64",
         "--oneshot-prompt", "cli shim"],
        cwd=REPO, capture_output=True, text=True, timeout=240,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    assert "ONESHOT_RESULT tokens=8 world=2" in out.stdout


def test_bench_defaults_match_baseline_json():
    """The driver benches the metric/config BASELINE.json names — pin the
    defaults (llama3-8b, weak-scaling replicas, bf16) so they cannot drift
    from the declared headline."""
    import json

    baseline = json.loads(open(os.path.join(REPO, "BASELINE.json")).read())
    assert "Llama-3-8B" in baseline["metric"]
    src = open(os.path.join(REPO, "bench.py")).read()
    assert '"--model", default="llama3-8b"' in src.replace("'", '"')
    assert '"scaling": "weak"' in src
    assert "llama3-8b" in src
