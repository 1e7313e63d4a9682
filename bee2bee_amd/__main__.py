"""CLI: serve-native / serve-hf / serve-ollama / serve-hf-remote / serve-web /
register / config / doctor / export-model / seed-model / bench — click group,
same command names as the reference (bee2bee/__main__.py:30-123) plus
native-engine extras (doctor, export-model, seed-model, serve-web).
"""
from __future__ import annotations

import asyncio
import os

import click

from .config import get_bootstrap_url, set_bootstrap_url
from .utils import setup_logging


@click.group()
@click.version_option(package_name=None, version=__import__(
    "bee2bee_amd").__version__, prog_name="bee2bee-amd")
def cli() -> None:
    """Bee2Bee-AMD: MI355X-native decentralized inference mesh."""
    setup_logging()


def _serve(backend: str, **kwargs) -> None:
    from .mesh.node import run_mesh_node

    bootstrap = get_bootstrap_url()
    asyncio.run(run_mesh_node(bootstrap_link=bootstrap, backend=backend, **kwargs))


@cli.command("serve-hf")
@click.option("--model", default="distilgpt2", help="Model name (preset or HF id)")
@click.option("--model-path", default=None, help="Local HF checkpoint dir (safetensors)")
@click.option("--port", default=0, type=int, help="Mesh bind port")
@click.option("--region", default="Auto", help="Region name")
@click.option("--api-port", default=8000, type=int, help="FastAPI port")
@click.option("--device", default=None, help="cuda:N / cpu (auto when omitted)")
def serve_hf(model, model_path, port, region, api_port, device):
    """Serve a model on the native MI355X engine (HF-checkpoint compatible).

    Named serve-hf for reference-CLI compatibility; the execution engine is
    the hand-written CDNA4 HIP runtime, not transformers."""
    _serve(
        "native",
        model_name=model,
        model_path=model_path,
        port=port,
        region=region,
        api_port=api_port,
        device=device,
    )


@cli.command("serve-native")
@click.option("--model", default="llama3-8b", help="Model preset or HF id")
@click.option("--model-path", default=None, help="Local HF checkpoint dir")
@click.option("--port", default=0, type=int)
@click.option("--region", default="Auto")
@click.option("--api-port", default=8000, type=int)
@click.option("--device", default=None)
@click.option("--spec-decode", is_flag=True,
              help="speculative decoding (prompt-lookup + exact greedy "
                   "verification; output-invariant)")
@click.option("--kv-fp8", is_flag=True,
              help="fp8 (OCP e4m3) KV cache: half the decode-attention "
                   "bytes at a small quantization cost")
def serve_native(model, model_path, port, region, api_port, device,
                 spec_decode, kv_fp8):
    """Serve a model on the native MI355X HIP engine (alias of serve-hf)."""
    import os

    if spec_decode:
        os.environ["BEE2BEE_SPEC_DECODE"] = "1"
    if kv_fp8:
        os.environ["BEE2BEE_KV_FP8"] = "1"
    _serve(
        "native",
        model_name=model,
        model_path=model_path,
        port=port,
        region=region,
        api_port=api_port,
        device=device,
    )


@cli.command("serve-ollama")
@click.option("--model", default="llama3", help="Ollama model name")
@click.option("--host", default="0.0.0.0", help="Bind host")
@click.option("--port", default=0, type=int, help="Bind port")
@click.option("--public-host", default=None, help="Public IP/hostname")
@click.option("--region", default="Auto")
@click.option("--api-port", default=8000, type=int)
def serve_ollama(model, host, port, public_host, region, api_port):
    """Serve a local Ollama model with P2P connectivity."""
    _serve(
        "ollama",
        model_name=model,
        host=host,
        port=port,
        announce_host=public_host,
        region=region,
        api_port=api_port,
    )


@cli.command("serve-web")
@click.option("--host", default="0.0.0.0", help="Bind host")
@click.option("--port", default=8080, type=int, help="HTTP port")
@click.option("--seeds", default=None,
              help="Comma-separated seed node WS addrs (else BEE2BEE_SEEDS)")
def serve_web(host, port, seeds):
    """Serve the browser gateway + chat dashboard (L6 web layer):
    /api/p2p/{register,generate,status,global_metrics} over a mesh bridge."""
    import uvicorn

    from .web.bridge import MeshBridge
    from .web.gateway import create_app

    seed_list = [s.strip() for s in seeds.split(",")] if seeds else None

    async def _run() -> None:
        bridge = MeshBridge(seeds=seed_list)
        await bridge.start()
        try:
            config = uvicorn.Config(create_app(bridge), host=host, port=port,
                                    log_level="info")
            await uvicorn.Server(config).serve()
        finally:
            await bridge.stop()

    asyncio.run(_run())


@cli.command("serve-hf-remote")
@click.option("--model", default="meta-llama/Llama-2-7b-hf", help="HF model name")
@click.option("--token", required=True, help="HF API token")
@click.option("--region", default="Cloud")
@click.option("--api-port", default=8000, type=int)
def serve_hf_remote(model, token, region, api_port):
    """Serve via the HF Inference API with a local FastAPI proxy."""
    os.environ["HUGGING_FACE_HUB_TOKEN"] = token
    _serve("hf_remote", model_name=model, region=region, api_port=api_port)


@cli.command("serve-parallel")
@click.option("--mode", default="tp",
              type=click.Choice(["pp", "tp", "ep", "cp"]))
@click.option("--nproc", default=8, type=int, help="ranks (one per GPU)")
@click.option("--model", default=None)
@click.option("--model-path", default=None)
@click.option("--api-port", default=8000, type=int)
@click.option("--max-batch", default=8, type=int)
@click.option("--max-seq-len", default=2048, type=int)
@click.option("--master-port", default=29500, type=int)
def serve_parallel(mode, nproc, model, model_path, api_port, max_batch,
                   max_seq_len, master_port):
    """Serve a SHARDED model (pipeline/tensor/expert/context parallel)
    behind the standard node + gateway: wraps the torch.distributed.run
    launch of bee2bee_amd.parallel.serve_main (one rank per GPU; rank 0
    owns the mesh/API, ranks 1+ follow in lockstep)."""
    import subprocess
    import sys as _sys

    cmd = [
        _sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", str(nproc), "--master-addr", "127.0.0.1",
        "--master-port", str(master_port),
        "-m", "bee2bee_amd.parallel.serve_main",
        "--mode", mode, "--api-port", str(api_port),
        "--max-batch", str(max_batch), "--max-seq-len", str(max_seq_len),
    ]
    if model:
        cmd += ["--model", model]
    if model_path:
        cmd += ["--model-path", model_path]
    raise SystemExit(subprocess.run(cmd, check=False).returncode)


@cli.command("doctor")
@click.option("--port", default=None, type=int,
              help="Also probe a specific mesh port for bindability")
def doctor_cmd(port):
    """Diagnose the node environment (torch/ROCm, GPU, HIP extension,
    RCCL, IPC env, state dir, ports) — run before first serve."""
    import sys as _sys

    from .doctor import run_doctor

    _sys.exit(run_doctor(port, echo=click.echo))


@cli.command("config")
@click.argument("key")
@click.argument("value")
def config_cmd(key, value):
    """Set a config value (e.g. `config bootstrap_url ws://host:port`)."""
    if key == "bootstrap_url":
        set_bootstrap_url(value)
    else:
        from .config import load_config, save_config

        cfg = load_config()
        cfg[key] = value
        save_config(cfg)
    click.echo(f"set {key} = {value}")


@cli.command("register")
@click.option("--node-url", default=None, help="Specific node URL to register")
@click.option("--network", default="connectit", help="Network name")
@click.option("--region", prompt="Node Region", default="US-West")
@click.option("--test/--no-test", default=True, help="Run handshake test")
def register(node_url, network, region, test):
    """Register a node in the global directory (Supabase / entrypoint)."""

    async def _reg():
        from .mesh.node import MeshNode
        from .mesh.registry import RegistryClient

        target_addr = node_url
        peer_id = f"ext-{os.urandom(4).hex()}"
        local_node = None
        if not target_addr:
            local_node = MeshNode(port=0, enable_nat=False)
            await local_node.start()
            target_addr = local_node.addr
            peer_id = local_node.peer_id
        click.echo(f"target region: {region}")
        click.echo(f"node address:  {target_addr}")
        if test and node_url:
            # real handshake: open a WS, send hello, expect a hello back
            import json

            import aiohttp

            from .mesh import wire

            ok = False
            try:
                async with aiohttp.ClientSession() as session:
                    async with session.ws_connect(
                        node_url, max_msg_size=wire.MAX_FRAME, timeout=10
                    ) as ws:
                        await ws.send_str(
                            json.dumps(
                                wire.hello(peer_id, "", region, {}, {})
                            )
                        )
                        msg = await asyncio.wait_for(ws.receive(), timeout=10)
                        data = json.loads(msg.data)
                        ok = data.get("type") == wire.HELLO
            except Exception as e:
                click.echo(f"handshake failed: {e}")
            click.echo("handshake OK" if ok else "handshake FAILED")
        reg = RegistryClient()
        if reg.enabled:
            await reg.sync_node(
                peer_id=peer_id,
                address=target_addr,
                models=["manual-entry" if node_url else "system-test"],
                tag=f"cli-{network}",
                region=region,
            )
            click.echo("node registered")
        else:
            click.echo("registry unavailable (no credentials); skipped")
        if local_node is not None:
            await local_node.stop()

    asyncio.run(_reg())


@cli.command("export-model")
@click.option("--model", required=True, help="Model preset or HF id")
@click.option("--out", required=True, help="Output checkpoint dir")
@click.option("--seed", default=0, type=int)
def export_model(model, out, seed):
    """Write an HF-format safetensors checkpoint (random-init of the named
    architecture when offline) — pairs with seed-model for mesh
    distribution."""
    import torch

    from .models.spec import resolve_spec
    from .models.weights import ModelWeights, save_hf

    spec = resolve_spec(model)
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(seed)
    save_hf(w, out)
    click.echo(f"wrote {spec.name} ({spec.n_params() / 1e9:.2f}B params) to {out}")


@cli.command("seed-model")
@click.option("--name", required=True, help="Checkpoint name to publish")
@click.option("--path", required=True, help="HF checkpoint dir to seed")
@click.option("--port", default=0, type=int)
def seed_model(name, path, port):
    """Seed a checkpoint's pieces to the mesh (torrent-style distribution)."""

    async def _run():
        from .mesh.dht import DHTNode
        from .mesh.node import MeshNode
        from .mesh.weightshare import seed_checkpoint

        dht = DHTNode()
        await dht.start()
        node = MeshNode(port=port, enable_nat=False)
        await node.start()
        manifest = await seed_checkpoint(node, dht, name, path)
        total = sum(f["bytes"] for f in manifest["files"])
        click.echo(f"seeding '{name}': {len(manifest['files'])} files, "
                   f"{total / 1e6:.1f} MB at {node.addr}")
        click.echo("Ctrl+C to stop seeding")
        while True:
            await asyncio.sleep(15)

    asyncio.run(_run())


@cli.command("bench")
@click.option("--model", default="llama3-8b")
@click.option("--batch", default=64, type=int)
@click.option("--steps", default=32, type=int)
@click.option("--warmup", default=8, type=int)
def bench_cmd(model, batch, steps, warmup):
    """Run the single-process decode benchmark (see bench.py for the full
    multi-GPU contract)."""
    import subprocess
    import sys

    subprocess.run(
        [
            sys.executable,
            os.path.join(os.path.dirname(os.path.dirname(__file__)), "bench.py"),
            "--model",
            model,
            "--batch",
            str(batch),
            "--steps",
            str(steps),
            "--warmup",
            str(warmup),
        ],
        check=False,
    )


if __name__ == "__main__":
    cli()
