"""Programmatic multi-GPU demo: the three model-parallel engines + the
serving opt-ins, launched the way a deployment would.

Run (one rank per GPU; gloo on CPU works for a dry run):

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 examples/parallel_demo.py --mode tp

Modes: tp (tensor parallel), pp (pipeline parallel), cp (context
parallel: KV pages sharded across ranks), ep (Mixtral expert
parallel). All three engines share the lockstep generate() API and are
exact-match tested against the single-process engine (tests/test_*_cpu.py).
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.distributed as dist


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--mode", default="tp", choices=["tp", "pp", "ep", "cp"])
    ap.add_argument("--model", default=None,
                    help="preset (defaults: tp/pp=tiny or llama3-70b on GPU, "
                         "ep=tiny-moe or mixtral-8x7b on GPU)")
    ap.add_argument("--max-new", type=int, default=8)
    args = ap.parse_args()

    on_gpu = torch.cuda.is_available()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    dist.init_process_group("nccl" if on_gpu else "gloo",
                            rank=rank, world_size=world)
    device = f"cuda:{local_rank}" if on_gpu else "cpu"
    if on_gpu:
        torch.cuda.set_device(local_rank)

    if args.mode == "ep":
        from bee2bee_amd.parallel.moe_engine import MoEEngine as Engine

        model = args.model or ("mixtral-8x7b" if on_gpu else "tiny-moe")
    elif args.mode == "cp":
        from bee2bee_amd.parallel.cp import CPEngine as Engine

        model = args.model or ("llama3-8b" if on_gpu else "tiny")
    elif args.mode == "pp":
        from bee2bee_amd.parallel.pp import PipelineEngine as Engine

        model = args.model or ("llama3-70b" if on_gpu else "tiny")
    else:
        from bee2bee_amd.parallel.tp import TPEngine as Engine

        model = args.model or ("llama3-70b" if on_gpu else "tiny")

    eng = Engine(model, device=device, max_batch=4, max_seq_len=256, seed=7)
    prompts = [[5, 6, 7, 8], [100, 101, 102]]
    outs = eng.generate(prompts, args.max_new)
    if rank == 0 or args.mode == "pp":  # pp emits on the last stage
        print(f"[{args.mode}{world} rank {rank}] generated:", outs)
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
