"""Skinny-M decode GEMM go/no-go: hipBLASLt (torch.mm,
TunableOp off = the engine's default dispatch) vs the hand-written
grouped-GEMM kernel driven as a single-segment GEMM (E=1), on the four
Llama-3-8B projection shapes at decode batch sizes.

The decode projections are W-streaming-bound at small M (W bytes = N*K*2
read once per step); at flagship M=1536 they are MFMA-bound. Reports
achieved W TB/s for each.

Usage (GPU box): python scripts/bench_skinny.py
"""
from __future__ import annotations

import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bee2bee_amd import ops

SHAPES = [  # (name, N, K) — llama3-8b qkv / o / gate_up / down
    ("qkv", 6144, 4096),
    ("o", 4096, 4096),
    ("gate_up", 28672, 4096),
    ("down", 4096, 14336),
]
BATCHES = [16, 64, 256, 1536]


def bench(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main() -> None:
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    g = torch.Generator(device=dev).manual_seed(3)
    rows = []
    for name, N, K in SHAPES:
        w = (torch.randn(N, K, generator=g, device=dev) * 0.05).bfloat16()
        w_e = w.unsqueeze(0).contiguous()  # [1, N, K] for grouped
        wb = N * K * 2
        for M in BATCHES:
            x = (torch.randn(M, K, generator=g, device=dev) * 0.3).bfloat16()
            offs = torch.tensor([0, M], dtype=torch.int32, device=dev)
            # correctness first
            got = ops.grouped_gemm(x, w_e, offs)
            ref = (x.float() @ w.float().T).bfloat16()
            err = (got.float() - ref.float()).abs().max().item()
            assert err < 0.25, (name, M, err)
            t_lib = bench(lambda: x @ w.t())
            t_gg = bench(lambda: ops.grouped_gemm(x, w_e, offs))
            rows.append({
                "proj": name, "M": M, "N": N, "K": K,
                "hipblaslt_us": round(t_lib * 1e6, 1),
                "grouped_us": round(t_gg * 1e6, 1),
                "hipblaslt_W_TBps": round(wb / t_lib / 1e12, 2),
                "grouped_W_TBps": round(wb / t_gg / 1e12, 2),
                "grouped_speedup": round(t_lib / t_gg, 3),
            })
            print(json.dumps(rows[-1]), flush=True)
    wins = [r for r in rows if r["grouped_speedup"] > 1.02]
    print(f"# grouped wins {len(wins)}/{len(rows)} shapes", flush=True)


if __name__ == "__main__":
    main()
