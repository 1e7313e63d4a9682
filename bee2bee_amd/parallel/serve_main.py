"""Distributed serving entry point (launch under torch.distributed.run;
`python -m bee2bee_amd serve-parallel` wraps exactly that): a sharded model (pp/tp/ep/cp) behind
the standard mesh node + HTTP gateway.

Launch (one process per GPU, same shape as bench_pp.py):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
      --master-addr 127.0.0.1 scripts/serve_parallel.py \
      --mode tp --model llama3-70b --api-port 8000

Rank 0 runs the mesh node + FastAPI gateway and broadcasts each request to
the group (parallel/serve.py LockstepServer); ranks 1+ follow in lockstep.
Requests arrive through every surface the single-GPU node has: /chat,
/generate, /v1/*, the wire protocol, the web bridge.

On CPU (no GPU visible) this runs the same code over gloo — the CI dry-run
in tests/test_bench_contract.py uses exactly that.
"""
import argparse
import asyncio
import os
import threading
import time
from typing import Any, Dict, Iterator

import torch
import torch.distributed as dist

from ..engine.sampler import SamplingParams
from ..models.spec import resolve_spec
from ..models.tokenizer import load_tokenizer
from .serve import LockstepServer, build_engine
from ..services.base import BaseService, ServiceError


class LockstepService(BaseService):
    """BaseService adapter over a rank-0 LockstepServer (text in/out)."""

    def __init__(self, server: LockstepServer, model_name: str,
                 tokenizer, max_seq_len: int,
                 price_per_token: float = 0.0) -> None:
        super().__init__("hf")  # reference wire name: peers route unchanged
        self.server = server
        self.model_name = model_name
        self.tokenizer = tokenizer
        self.max_seq_len = max_seq_len
        self.price = price_per_token
        # torch.distributed is not thread-safe: one in-flight lockstep
        # request at a time (the gateway executes services in a thread pool)
        self._lock = threading.Lock()

    def get_metadata(self) -> Dict[str, Any]:
        return {
            "models": [self.model_name],
            "price_per_token": self.price,
            "backend": "bee2bee-amd-parallel",
            "world_size": dist.get_world_size(),
        }

    def execute(self, params: Dict[str, Any]) -> Dict[str, Any]:
        prompt = params.get("prompt")
        if not prompt:
            raise ServiceError("Missing prompt")
        max_new = max(1, min(int(params.get("max_new_tokens", 128)),
                             self.max_seq_len // 2))
        t = params.get("temperature", 0.7)
        sp = SamplingParams.from_request(
            t, params.get("top_p"), params.get("top_k"),
            params.get("repetition_penalty"))
        ids = self.tokenizer.encode(prompt)[-(self.max_seq_len - max_new - 1):]
        t0 = time.time()
        with self._lock:
            outs = self.server.generate([ids], max_new, sp)
        text = self.tokenizer.decode(outs[0])
        return {
            "text": text,
            "tokens": len(outs[0]),
            "latency_ms": int((time.time() - t0) * 1000),
            "price_per_token": self.price,
            "cost": self.price * len(outs[0]),
            "backend": "bee2bee-amd-parallel",
        }

    def execute_stream(self, params: Dict[str, Any]) -> Iterator[str]:
        import json

        res = self.execute(params)  # lockstep engines are batch-synchronous
        yield json.dumps({"text": res["text"]}) + "\n"
        yield json.dumps({"done": True}) + "\n"


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", default="tp", choices=["pp", "tp", "ep", "cp"])
    ap.add_argument("--model", default=None)
    ap.add_argument("--model-path", default=None)
    ap.add_argument("--max-batch", type=int, default=8)
    ap.add_argument("--max-seq-len", type=int, default=2048)
    ap.add_argument("--api-port", type=int, default=8000)
    ap.add_argument("--mesh-port", type=int, default=0)
    ap.add_argument("--price", type=float, default=0.0)
    ap.add_argument("--oneshot-prompt", default=None,
                    help="serve nothing: run ONE request through the group "
                         "and exit (launch-contract dry runs)")
    args = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group(backend="nccl" if on_gpu else "gloo")
    if on_gpu:
        torch.cuda.set_device(local_rank)

    model = args.model or ("tiny-moe" if args.mode == "ep" else "tiny")
    engine = build_engine(args.mode, model,
                          device=f"cuda:{local_rank}" if on_gpu else "cpu",
                          max_batch=args.max_batch,
                          max_seq_len=args.max_seq_len,
                          model_path=args.model_path)
    server = LockstepServer(engine)

    if rank != 0:
        served = server.serve_follower()
        print(f"rank {rank}: served {served} requests", flush=True)
        dist.destroy_process_group()
        return

    spec = resolve_spec(model, args.model_path)
    tok = load_tokenizer(args.model_path, spec.vocab_size,
                         spec.bos_token_id, spec.eos_token_id)
    svc = LockstepService(server, model, tok,
                          min(args.max_seq_len, spec.max_seq_len),
                          args.price)

    if args.oneshot_prompt is not None:
        res = svc.execute({"prompt": args.oneshot_prompt,
                           "max_new_tokens": 8, "temperature": 0.0})
        print(f"ONESHOT_RESULT tokens={res['tokens']} "
              f"world={dist.get_world_size()} mode={args.mode}", flush=True)
        server.shutdown()
        dist.destroy_process_group()
        return

    async def run() -> None:
        from ..mesh.node import MeshNode

        node = MeshNode(host="0.0.0.0", port=args.mesh_port,
                        enable_nat=False)
        await node.start()
        await node.add_service(svc)

        import uvicorn

        from ..gateway import api as gateway_api

        gateway_api.node = node
        config = uvicorn.Config(gateway_api.app, host="0.0.0.0",
                                port=args.api_port, log_level="warning")
        print(f"serving {model} [{args.mode}{dist.get_world_size()}] at "
              f"http://0.0.0.0:{args.api_port} and {node.addr}", flush=True)
        try:
            await uvicorn.Server(config).serve()
        finally:
            server.shutdown()
            await node.stop()

    asyncio.run(run())
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
