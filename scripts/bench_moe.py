"""MoE expert-MLP microbenchmark: hand-written grouped-GEMM kernel vs the
padded torch.bmm path, on the Mixtral decode shapes.

Reports effective W bandwidth (expert weights read once per projection =
the roofline for decode-sized batches) and ms per projection pair.

Usage (GPU box): python scripts/bench_moe.py [--tokens 512] [--iters 30]
"""
from __future__ import annotations

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from bee2bee_amd import ops


def bench(fn, iters, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--tokens", type=int, default=512)
    ap.add_argument("--iters", type=int, default=30)
    ap.add_argument("--experts", type=int, default=8)
    ap.add_argument("--topk", type=int, default=2)
    ap.add_argument("--hidden", type=int, default=4096)
    ap.add_argument("--inter", type=int, default=14336)
    args = ap.parse_args()
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    E, H, I, k = args.experts, args.hidden, args.inter, args.topk
    T = args.tokens
    S = T * k

    g = torch.Generator(device=dev).manual_seed(7)
    x = (torch.randn(S, H, generator=g, device=dev) * 0.3).bfloat16()
    w_gu = (torch.randn(E, 2 * I, H, generator=g, device=dev) * 0.05).bfloat16()
    w_dn = (torch.randn(E, H, I, generator=g, device=dev) * 0.05).bfloat16()
    # even routing (the expected case under load); the kernel itself is
    # tested on ragged/empty segments in tests/test_ops_gpu.py
    counts = torch.full((E,), S // E, dtype=torch.int64)
    counts[-1] += S - int(counts.sum())
    offs = torch.zeros(E + 1, dtype=torch.int32)
    offs[1:] = counts.cumsum(0).to(torch.int32)
    offs = offs.to(dev)
    act = (torch.randn(S, I, generator=g, device=dev) * 0.3).bfloat16()

    w_bytes = (w_gu.numel() + w_dn.numel()) * 2  # read-once roofline

    def grouped():
        gu = ops.grouped_gemm(x, w_gu, offs)
        y = ops.grouped_gemm(ops.swiglu(gu), w_dn, offs)
        return y

    C = int(counts.max())
    def padded_bmm():
        padded = torch.zeros(E, C, H, dtype=x.dtype, device=dev)
        # static fill stands in for the scatter (cheap next to the GEMMs)
        padded.view(E * C, H)[: S] = x
        gu = torch.bmm(padded, w_gu.transpose(1, 2))
        a = ops.swiglu(gu.reshape(E * C, 2 * I)).view(E, C, I)
        y = torch.bmm(a, w_dn.transpose(1, 2))
        return y

    # correctness spot-check vs the fp32 reference
    from bee2bee_amd.ops import reference as R
    got = ops.grouped_gemm(x, w_gu, offs)
    ref = R.grouped_gemm(x, w_gu, offs)
    err = (got.float() - ref.float()).abs().max().item()
    assert err < 0.05, f"grouped_gemm mismatch: {err}"

    t_g = bench(grouped, args.iters)
    t_b = bench(padded_bmm, args.iters)

    # single-projection timings for per-kernel attribution
    t_gu = bench(lambda: ops.grouped_gemm(x, w_gu, offs), args.iters)
    t_dn = bench(lambda: ops.grouped_gemm(act, w_dn, offs), args.iters)

    out = {
        "tokens": T, "slots": S, "experts": E,
        "grouped_ms": round(t_g * 1e3, 3),
        "bmm_ms": round(t_b * 1e3, 3),
        "grouped_gu_ms": round(t_gu * 1e3, 3),
        "grouped_dn_ms": round(t_dn * 1e3, 3),
        "grouped_W_TBps": round(w_bytes / t_g / 1e12, 2),
        "bmm_W_TBps": round(w_bytes / t_b / 1e12, 2),
        "gu_W_TBps": round(w_gu.numel() * 2 / t_gu / 1e12, 2),
        "dn_W_TBps": round(w_dn.numel() * 2 / t_dn / 1e12, 2),
        "speedup_vs_bmm": round(t_b / t_g, 3),
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
