"""Developer guide: a rich-console walkthrough of the mesh stack.

Parity with the reference's examples/p2p_solution_guide.py — but where that
guide mocks its discovery table rows, this one runs everything live: two
real nodes on loopback, real hello/peer_list/service_announce traffic, real
price+latency provider selection, and a real request (local-first, then a
one-hop relay through a node that does NOT have the model).

Usage: python examples/guide.py
"""
import asyncio
import sys
from typing import Any, Dict, Iterator

sys.path.insert(0, ".")

from rich.console import Console
from rich.panel import Panel
from rich.table import Table

from bee2bee_amd.mesh.links import generate_join_link
from bee2bee_amd.mesh.node import MeshNode
from bee2bee_amd.services.base import BaseService

console = Console()


class DemoService(BaseService):
    """Tiny deterministic backend so the guide runs with no GPU/model."""

    def __init__(self, model: str, price: float) -> None:
        super().__init__("hf")
        self.model = model
        self.price = price

    def get_metadata(self) -> Dict[str, Any]:
        return {"models": [self.model], "price_per_token": self.price}

    def execute(self, params: Dict[str, Any]) -> Dict[str, Any]:
        text = f"[{self.model}] answered: {params['prompt'][:48]}"
        return {"text": text, "tokens": len(text.split()),
                "latency_ms": 3, "price_per_token": self.price,
                "cost": self.price * len(text.split())}

    def execute_stream(self, params: Dict[str, Any]) -> Iterator[str]:
        import json

        for word in self.execute(params)["text"].split():
            yield json.dumps({"text": word + " "}) + "\n"
        yield json.dumps({"done": True}) + "\n"


async def wait_until(cond, timeout=10.0):
    for _ in range(int(timeout / 0.05)):
        if cond():
            return True
        await asyncio.sleep(0.05)
    return False


async def main() -> None:
    console.print(Panel.fit(
        "[bold yellow]bee2bee-amd developer guide[/bold yellow] — "
        "live mesh walkthrough", border_style="cyan"))

    console.print("\n[bold]1. Start two nodes[/bold] (loopback, port 0)")
    provider = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
    edge = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
    await provider.start()
    await edge.start()
    console.print(f"   provider [green]{provider.addr}[/green]  id={provider.peer_id}")
    console.print(f"   edge     [green]{edge.addr}[/green]  id={edge.peer_id}")

    console.print("\n[bold]2. Attach a service[/bold] (announces to peers)")
    await provider.add_service(DemoService("demo-125m", price=0.0005))

    console.print("\n[bold]3. Join via deep link[/bold] (what QR codes / "
                  "the web dashboard carry)")
    link = generate_join_link("main", "demo-125m", "0" * 8, [provider.addr])
    console.print(f"   {link[:76]}...")
    await edge.connect_bootstrap(provider.addr)
    assert await wait_until(lambda: provider.peer_id in edge.peers)
    assert await wait_until(lambda: edge.providers)

    console.print("\n[bold]4. Live provider table[/bold] (hello + "
                  "service_announce + ping RTT)")
    table = Table(show_header=True, header_style="bold magenta")
    table.add_column("peer")
    table.add_column("models", style="cyan")
    table.add_column("$/token", justify="right")
    table.add_column("status")
    for entry in edge.list_providers():
        table.add_row(entry["peer_id"][:16], ",".join(entry["models"]),
                      f"{entry['price_per_token']:.4f}",
                      "[green]" + entry.get("status", "good"))
    console.print(table)

    console.print("[bold]5. Route a request[/bold] — edge has no local "
                  "service, so pick_provider sorts by (price, latency) and "
                  "relays one hop:")
    pid, meta = edge.pick_provider("demo-125m")
    console.print(f"   selected provider: [green]{pid[:16]}[/green] "
                  f"(${meta['price_per_token']}/token)")
    result = await edge.request_generation(
        provider_id=pid, prompt="What is the future of P2P AI?",
        max_new_tokens=24, model_name="demo-125m")
    console.print(Panel(result["text"], title="gen_result",
                        border_style="green"))

    console.print("[bold]6. Torrent-style weight distribution[/bold] — seed "
                  "a checkpoint on the provider, fetch it over the wire:")
    import tempfile

    import torch

    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf
    from bee2bee_amd.mesh.weightshare import fetch_checkpoint, seed_checkpoint

    src = tempfile.mkdtemp()
    dst = tempfile.mkdtemp()
    w = ModelWeights(PRESETS["tiny"], torch.device("cpu"),
                     torch.float32).random_init(1)
    save_hf(w, src)
    manifest = await seed_checkpoint(provider, provider.dht, "demo-ckpt",
                                     src, piece_size=65536)
    await asyncio.sleep(0.3)  # one-hop DHT replication
    await fetch_checkpoint(edge, edge.dht, "demo-ckpt", dst)
    files = ", ".join(f["name"] for f in manifest["files"])
    console.print(f"   fetched + hash-verified: [green]{files}[/green]")

    console.print("[bold]7. Teardown[/bold]")
    await edge.stop()
    await provider.stop()
    console.print("[green]done[/green] — next steps: serve a real model with "
                  "`python -m bee2bee_amd serve-hf --model llama3-8b`, or the "
                  "browser dashboard with `serve-web`.")


if __name__ == "__main__":
    asyncio.run(main())
