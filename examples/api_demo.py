"""HTTP API walkthrough: boot one native-engine node with its FastAPI
sidecar, hit every endpoint, stream a generation, then tear down.

Parity with the reference's examples/api_demo.py (subprocess server +
requests walkthrough), on the MI355X-native stack: the backing service is
the HIP engine (or its fp32 CPU reference off-GPU) serving a tiny
random-init model, so the demo runs anywhere.

Usage: python examples/api_demo.py [port]
"""
import json
import multiprocessing
import sys
import time

import requests

sys.path.insert(0, ".")

PORT = int(sys.argv[1]) if len(sys.argv) > 1 else 8077
BASE = f"http://127.0.0.1:{PORT}"


def serve() -> None:
    import asyncio

    from bee2bee_amd.mesh.node import run_mesh_node

    asyncio.run(run_mesh_node(
        host="127.0.0.1", port=0, api_port=PORT, backend="native",
        model_name="tiny", enable_nat=False,
    ))


def main() -> None:
    proc = multiprocessing.Process(target=serve, daemon=True)
    proc.start()
    try:
        for _ in range(240):  # server bind + engine build + service announce
            try:
                if requests.get(f"{BASE}/", timeout=1).json().get("services"):
                    break
            except requests.RequestException:
                pass
            time.sleep(0.5)

        print("== GET / (node status, live engine stats)")
        print(json.dumps(requests.get(f"{BASE}/").json(), indent=2)[:600])

        print("\n== GET /peers")
        print(requests.get(f"{BASE}/peers").json())

        print("\n== GET /providers")
        print(requests.get(f"{BASE}/providers").json())

        print("\n== POST /generate (buffered)")
        r = requests.post(f"{BASE}/generate", json={
            "prompt": "hello mesh", "max_new_tokens": 12, "temperature": 0.0,
        })
        print(json.dumps(r.json(), indent=2)[:400])

        print("\n== POST /generate (streaming JSON-lines)")
        with requests.post(f"{BASE}/generate", json={
            "prompt": "stream me", "max_new_tokens": 12, "stream": True,
        }, stream=True) as resp:
            for line in resp.iter_lines():
                if line:
                    print("  chunk:", line.decode()[:80])

        print("\n== GET /v1/models (OpenAI-compatible surface)")
        print(requests.get(f"{BASE}/v1/models").json())

        print("\n== POST /v1/chat/completions (SSE stream)")
        with requests.post(f"{BASE}/v1/chat/completions", json={
            "model": "tiny", "stream": True, "max_tokens": 12,
            "messages": [{"role": "user", "content": "hello"}],
        }, stream=True) as resp:
            for line in resp.iter_lines():
                if line:
                    print("  sse:", line.decode()[:80])

        print("\n== GET /metrics (Prometheus, first lines)")
        print("\n".join(requests.get(f"{BASE}/metrics").text.splitlines()[:8]))

        print("\ndemo complete")
    finally:
        proc.terminate()
        proc.join(timeout=10)


if __name__ == "__main__":
    main()
