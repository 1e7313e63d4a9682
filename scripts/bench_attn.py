"""Microbenchmark for the paged decode-attention kernel.

Usage: python scripts/bench_attn.py [--batch 64] [--len 1024] [--iters 50]
Prints achieved KV-read bandwidth (the kernel is memory-bound; the roofline
is ~6.3 TB/s achievable HBM on MI355X).
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=64)
    ap.add_argument("--len", type=int, dest="seqlen", default=1024)
    ap.add_argument("--nkv", type=int, default=8)
    ap.add_argument("--group", type=int, default=4)
    ap.add_argument("--hd", type=int, default=128)
    ap.add_argument("--bs", type=int, default=32)
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--prefill", action="store_true",
                    help="bench the MFMA prefill kernel instead")
    ap.add_argument("--fp8", action="store_true",
                    help="fp8 (e4m3) KV cache variant of the decode bench")
    args = ap.parse_args()

    from bee2bee_amd import ops

    dev = "cuda:0"
    torch.manual_seed(0)
    B, L, nkv, G, hd, bs = (
        args.batch, args.seqlen, args.nkv, args.group, args.hd, args.bs
    )
    nq = nkv * G

    if args.prefill:
        T = B * L
        q = torch.randn(T, nq, hd, device=dev).bfloat16()
        k = torch.randn(T, nkv, hd, device=dev).bfloat16()
        v = torch.randn(T, nkv, hd, device=dev).bfloat16()
        cu = torch.arange(0, T + 1, L, dtype=torch.int32, device=dev)
        scale = hd**-0.5
        for _ in range(3):
            out = ops.attn_prefill(q, k, v, cu, L, scale, True)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            out = ops.attn_prefill(q, k, v, cu, L, scale, True)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / args.iters
        # causal flops: 4 * nq * hd * L^2/2 per seq (QK^T + PV)
        flops = 4 * nq * hd * L * L / 2 * B
        print(f"prefill B{B} L{L} nq{nq} hd{hd}: {dt * 1e3:.3f} ms  "
              f"{flops / dt / 1e12:.1f} TF/s")
        return

    W = (L + bs - 1) // bs
    nb = B * W + 1
    bt = torch.arange(1, B * W + 1, dtype=torch.int32).reshape(B, W).to(dev)
    if args.fp8:
        kc = torch.randint(0, 127, (nb, nkv, bs, hd), dtype=torch.uint8,
                           device=dev)
        vc = torch.randint(0, 127, (nb, nkv, bs, hd), dtype=torch.uint8,
                           device=dev)
    else:
        kc = torch.randn(nb, nkv, bs, hd, device=dev).bfloat16()
        vc = torch.randn(nb, nkv, bs, hd, device=dev).bfloat16()
    q = torch.randn(B, nq, hd, device=dev).bfloat16()
    lens = torch.full((B,), L, dtype=torch.int32, device=dev)
    scale = hd**-0.5

    for _ in range(5):
        out = ops.attn_decode(q, kc, vc, bt, lens, scale)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        out = ops.attn_decode(q, kc, vc, bt, lens, scale)
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters
    kv_bytes = B * L * 2 * nkv * hd * (1 if args.fp8 else 2)
    print(
        f"decode{' fp8' if args.fp8 else ''} "
        f"B{B} L{L} nkv{nkv} G{G} hd{hd}: {dt * 1e6:.1f} us  "
        f"KV {kv_bytes / 1e6:.0f} MB  {kv_bytes / dt / 1e12:.2f} TB/s"
    )
    _ = out


if __name__ == "__main__" and sys.argv[1:2] not in (["probe"], ["paged"]):
    main()


def probe():
    """Bandwidth probes: python scripts/bench_attn.py probe"""
    from bee2bee_amd import ops

    hip = ops.require_hip()
    dev = "cuda:0"
    pool = torch.randn(1 << 29, device=dev).bfloat16()  # 1 GiB
    nbytes = pool.numel() * 2
    for name, mode, arg in (
        ("linear dwordx4", 0, 0),
        ("row-per-lane (serial)", 1, 0),
        ("row-per-lane (batch8)", 1, 1),
        ("row-per-instr rpw256", 2, 256),
        ("row-per-instr rpw1024", 2, 1024),
    ):
        for _ in range(2):
            hip.bw_probe(pool, mode, 128, arg)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            hip.bw_probe(pool, mode, 128, arg)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"{name:26s} {nbytes / dt / 1e12:6.2f} TB/s")


if __name__ == "__main__" and "probe" in sys.argv[1:2]:
    probe()
    sys.exit(0)


def probe_paged():
    from bee2bee_amd import ops

    hip = ops.require_hip()
    dev = "cuda:0"
    hd = 128
    pool = torch.randn(1 << 29, device=dev).bfloat16()
    nbytes = pool.numel() * 2
    for rpp in (256, 64, 32, 16):
        n_pages = pool.numel() // hd // rpp
        for name, perm in (("seq", False), ("perm", True)):
            table = (
                torch.randperm(n_pages, device=dev)
                if perm else torch.arange(n_pages, device=dev)
            ).to(torch.int32)
            for _ in range(2):
                hip.bw_probe_paged(pool, table, hd, rpp)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(10):
                hip.bw_probe_paged(pool, table, hd, rpp)
            torch.cuda.synchronize()
            dt = (time.perf_counter() - t0) / 10
            print(f"paged rows/page={rpp:4d} {name:5s}"
                  f" ({rpp * hd * 2 // 1024:3d} KB contig) "
                  f"{nbytes / dt / 1e12:6.2f} TB/s")


if __name__ == "__main__" and "paged" in sys.argv[1:2]:
    probe_paged()
    sys.exit(0)
