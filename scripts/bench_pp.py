"""Model-parallel benchmark (BASELINE config 4): Llama-3-70B sharded
across N GPUs over RCCL/xGMI.

  --mode pp  (default): layer shards, point-to-point hidden-state hops
  --mode tp:            Megatron head/intermediate shards, two all-reduces
                        per layer (lower latency per token at small batch)
  --mode cp:            context parallelism — KV pages sharded across
                        ranks, one (m,l,o) all_gather per layer; servable
                        context scales with world x 288 GB

Launch (one rank per GPU):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 scripts/bench_pp.py --steps 16 --warmup 4

Rank 0 prints one JSON line (same schema as bench.py; scaling is "strong":
the model is fixed and split across N GPUs).
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-70b")
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--steps", type=int, default=16)
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--mode", default="pp", choices=["pp", "tp", "cp", "ep"])
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    on_gpu = torch.cuda.is_available()
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group(
        backend="nccl" if on_gpu else "gloo", rank=rank, world_size=world
    )
    if on_gpu:
        torch.cuda.set_device(local_rank)

    if args.mode == "tp":
        from bee2bee_amd.parallel.tp import TPEngine as Engine
    elif args.mode == "cp":
        from bee2bee_amd.parallel.cp import CPEngine as Engine
    elif args.mode == "ep":
        from bee2bee_amd.parallel.moe_engine import MoEEngine as Engine

        if args.model == "llama3-70b":  # ep needs a MoE model
            args.model = "mixtral-8x7b" if torch.cuda.is_available() \
                else "tiny-moe"
    else:
        from bee2bee_amd.parallel.pp import PipelineEngine as Engine

    eng = Engine(
        args.model,
        device=f"cuda:{local_rank}" if on_gpu else "cpu",
        max_batch=args.batch,
        max_seq_len=args.prompt_len + args.steps + args.warmup + 8,
        seed=7,
    )
    import random

    rng = random.Random(0)
    prompts = [
        [rng.randrange(4, eng.spec.vocab_size) for _ in range(args.prompt_len)]
        for _ in range(args.batch)
    ]
    ids = eng.prefill(prompts)
    for _ in range(args.warmup):
        ids = eng.decode_step(ids)
    if on_gpu:
        torch.cuda.synchronize()
    dist.barrier()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        ids = eng.decode_step(ids)
    if on_gpu:
        torch.cuda.synchronize()
    dist.barrier()
    elapsed = time.perf_counter() - t0
    t = torch.tensor([elapsed], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    elapsed = float(t.item())

    if rank == 0:
        print(json.dumps({
            "metric": f"output tokens/sec ({eng.spec.name} bf16 greedy decode, "
                      f"{dict(tp='tensor', cp='context', pp='pipeline', ep='expert')[args.mode]}"
                      f"-parallel {args.mode}{world})",
            "value": round(args.batch * args.steps / elapsed, 1),
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 3),
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "bf16" if on_gpu else "fp32-cpu",
            "data": "synthetic",
            "config": {
                "model": eng.spec.name,
                "global_batch": args.batch,
                "seq_len": args.prompt_len,
                "parallelism": f"{args.mode}{world}",
            },
        }), flush=True)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
