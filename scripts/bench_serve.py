"""End-to-end serving benchmark: p50/p95 request latency + tokens/sec
through the FULL stack (HTTP gateway -> mesh node -> service -> engine).

This measures the second half of BASELINE.json's metric ("output tokens/sec
+ p50 e2e request latency"): concurrent clients POST /generate against a
live node serving the native engine, exactly like an external user.

The server runs in its OWN process (round 2): with client and server in
one interpreter the benchmark measured its own GIL contention — separate
processes measure what an external user actually sees.

Usage:
  python scripts/bench_serve.py --model llama3-8b --clients 16 \
      --requests 64 --max-new 64 [--device cuda:0] [--prompt-len 512]
Prints one JSON line with latency percentiles and aggregate throughput.
"""
import argparse
import asyncio
import json
import multiprocessing
import os
import random
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def _server_main(opts: dict) -> None:
    """Child process: mesh node + native engine + uvicorn gateway.
    BEE2BEE_PROFILE_SERVER=1 cProfiles this process and dumps the top
    cumulative entries to gpurun_out/serve_profile.txt on SIGTERM."""
    prof = None
    if os.environ.get("BEE2BEE_PROFILE_SERVER") == "1":
        import cProfile
        import pstats
        import signal

        prof = cProfile.Profile()
        prof.enable()

        def _dump(_sig, _frm):
            prof.disable()
            os.makedirs("gpurun_out", exist_ok=True)
            with open("gpurun_out/serve_profile.txt", "w") as f:
                st = pstats.Stats(prof, stream=f)
                st.sort_stats("cumulative").print_stats(50)
                st.sort_stats("tottime").print_stats(50)
            os._exit(0)

        signal.signal(signal.SIGTERM, _dump)
    elif os.environ.get("BEE2BEE_PROFILE_ENGINE") == "1":
        import signal

        def _stop_engine(_sig, _frm):
            # shut the engine thread down so its profiler dump runs
            from bee2bee_amd.gateway import api as gateway_api

            node = gateway_api.node
            if node is not None:
                for svc in list(getattr(node, "local_services", {}).values()):
                    eng = getattr(svc, "engine", None)
                    if eng is not None:
                        eng.shutdown()
            os._exit(0)

        signal.signal(signal.SIGTERM, _stop_engine)

    async def serve() -> None:
        import uvicorn

        from bee2bee_amd.gateway import api as gateway_api
        from bee2bee_amd.mesh.node import MeshNode
        from bee2bee_amd.services.native import NativeEngineService

        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        node.api_port = opts["api_port"]
        svc = NativeEngineService(
            opts["model"], device=opts["device"],
            max_batch=opts["max_batch"], max_seq_len=2048,
        )
        loop = asyncio.get_running_loop()
        await loop.run_in_executor(None, svc.load_sync)
        await node.add_service(svc)
        gateway_api.node = node
        config = uvicorn.Config(gateway_api.app, host="127.0.0.1",
                                port=opts["api_port"], log_level="error")
        await uvicorn.Server(config).serve()

    asyncio.run(serve())


async def run() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama3-8b")
    ap.add_argument("--device", default=None)
    ap.add_argument("--clients", type=int, default=16)
    ap.add_argument("--requests", type=int, default=64)
    ap.add_argument("--max-new", type=int, default=64)
    ap.add_argument("--prompt-len", type=int, default=512, help="chars of prompt")
    ap.add_argument("--repetitive", action="store_true",
                    help="self-similar prompts (speculative-decoding-friendly "
                         "workload: code/template-like text)")
    ap.add_argument("--max-batch", type=int, default=32)
    ap.add_argument("--api-port", type=int, default=18321)
    args = ap.parse_args()

    import aiohttp

    t0 = time.time()
    proc = multiprocessing.get_context("spawn").Process(
        target=_server_main,
        args=({"model": args.model, "device": args.device,
               "max_batch": args.max_batch, "api_port": args.api_port},),
        daemon=True,
    )
    proc.start()

    base = f"http://127.0.0.1:{args.api_port}"
    device = "?"
    async with aiohttp.ClientSession() as session:
        for _ in range(600):  # engine build + server bind
            try:
                async with session.get(f"{base}/", timeout=aiohttp.ClientTimeout(total=2)) as r:
                    body = await r.json()
                    if body.get("services"):
                        device = (body.get("engine") or {}).get("device", "?")
                        break
            except Exception:
                pass
            await asyncio.sleep(0.5)
        else:
            proc.terminate()
            raise RuntimeError("server did not become ready")
    load_s = time.time() - t0

    rng = random.Random(0)
    words = ["alpha", "beta", "gamma", "delta", "mesh", "gpu", "tensor", "ring"]

    def mk_prompt() -> str:
        if args.repetitive:
            # one short phrase repeated (template/code-like self-similarity)
            phrase = " ".join(rng.choice(words) for _ in range(4))
            reps = max(1, args.prompt_len // (len(phrase) + 1))
            return " ".join([phrase] * reps)
        out = []
        while sum(len(w) + 1 for w in out) < args.prompt_len:
            out.append(rng.choice(words))
        return " ".join(out)

    lat: list = []
    ttft: list = []
    tokens_total = 0
    sem = asyncio.Semaphore(args.clients)
    url = f"{base}/generate"

    async def one_request(session) -> None:
        nonlocal tokens_total
        async with sem:
            t_start = time.perf_counter()
            first = None
            async with session.post(
                url,
                json={
                    "prompt": mk_prompt(),
                    "max_new_tokens": args.max_new,
                    "temperature": 0.0,
                    "stream": True,
                },
            ) as resp:
                assert resp.status == 200, await resp.text()
                n_tok = 0
                async for line in resp.content:
                    if not line.strip():
                        continue
                    if first is None:
                        first = time.perf_counter()
                    try:
                        d = json.loads(line)
                    except Exception:
                        continue
                    if d.get("done"):
                        break
                    if "text" in d:
                        n_tok += 1
            t_end = time.perf_counter()
            lat.append(t_end - t_start)
            if first is not None:
                ttft.append(first - t_start)
            tokens_total += args.max_new

    try:
        t_bench = time.perf_counter()
        async with aiohttp.ClientSession() as session:
            # small warmup
            await one_request(session)
            lat.clear(); ttft.clear()
            tokens_total = 0
            await asyncio.gather(
                *(one_request(session) for _ in range(args.requests)))
        wall = time.perf_counter() - t_bench
        engine_stats = {}
        try:
            async with aiohttp.ClientSession() as session:
                async with session.get(f"{base}/", timeout=aiohttp.ClientTimeout(total=3)) as r:
                    engine_stats = (await r.json()).get("engine") or {}
        except Exception:
            pass
    finally:
        proc.terminate()
        proc.join(timeout=10)

    lat.sort()
    def pct(xs, p):
        return xs[min(len(xs) - 1, int(p * len(xs)))] if xs else None

    result = {
        "metric": "e2e request latency + tokens/sec via /generate (streaming)",
        "model": args.model,
        "device": device,
        "requests": args.requests,
        "concurrency": args.clients,
        "max_new_tokens": args.max_new,
        "p50_s": round(pct(lat, 0.50), 3),
        "p95_s": round(pct(lat, 0.95), 3),
        "ttft_p50_s": round(pct(sorted(ttft), 0.50), 3) if ttft else None,
        "tokens_per_sec": round(args.requests * args.max_new / wall, 1),
        "wall_s": round(wall, 2),
        "model_load_s": round(load_s, 1),
        "data": ("repetitive" if args.repetitive else "random")
                + " synthetic prompts, random-init weights",
        "engine_busy_s": engine_stats.get("engine_busy_s"),
        "engine_steps": engine_stats.get("engine_steps"),
        "engine_ms_per_step": engine_stats.get("engine_ms_per_step"),
    }
    print(json.dumps(result), flush=True)


if __name__ == "__main__":
    asyncio.run(run())
