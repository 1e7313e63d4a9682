"""Proxy service backends against a local mock Ollama server (the
reference's OllamaService surface: /api/tags probe with fuzzy tag
resolution, buffered and NDJSON-streamed /api/generate)."""
import json
import threading
from http.server import BaseHTTPRequestHandler, HTTPServer

import pytest

from bee2bee_amd.services.base import ServiceError
from bee2bee_amd.services.ollama import OllamaService


class MockOllama(BaseHTTPRequestHandler):
    def log_message(self, *a):  # quiet
        pass

    def do_GET(self):
        if self.path == "/api/tags":
            body = json.dumps({"models": [
                {"name": "gemma3:270m"}, {"name": "llama3:8b-instruct"},
            ]}).encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)
        else:
            self.send_response(404)
            self.end_headers()

    def do_POST(self):
        n = int(self.headers.get("Content-Length", 0))
        req = json.loads(self.rfile.read(n))
        if req.get("stream"):
            self.send_response(200)
            self.end_headers()
            for word in ["hello", " from", " ollama"]:
                line = json.dumps({"response": word, "done": False}) + "\n"
                self.wfile.write(line.encode())
            self.wfile.write(json.dumps(
                {"response": "", "done": True,
                 "eval_count": 3, "total_duration": 5000000}).encode() + b"\n")
        else:
            body = json.dumps({
                "response": f"echo:{req['prompt']}",
                "eval_count": 2,
                "total_duration": 7000000,
            }).encode()
            self.send_response(200)
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)


@pytest.fixture()
def mock_ollama():
    server = HTTPServer(("127.0.0.1", 0), MockOllama)
    t = threading.Thread(target=server.serve_forever, daemon=True)
    t.start()
    yield f"http://127.0.0.1:{server.server_port}"
    server.shutdown()


def test_ollama_tag_resolution_and_execute(mock_ollama):
    svc = OllamaService("gemma3", host=mock_ollama)
    svc.load_sync()
    assert svc.actual_model == "gemma3:270m"  # fuzzy substring resolution
    meta = svc.get_metadata()
    assert "gemma3:270m" in meta["models"] and meta["backend"] == "ollama"

    res = svc.execute({"prompt": "hi"})
    assert res["text"] == "echo:hi"
    assert res["tokens"] == 2
    assert res["latency_ms"] >= 0


def test_ollama_stream_json_lines(mock_ollama):
    svc = OllamaService("llama3", host=mock_ollama)
    svc.load_sync()
    assert svc.actual_model == "llama3:8b-instruct"
    chunks = [json.loads(line) for line in svc.execute_stream({"prompt": "x"})]
    text = "".join(c.get("text", "") for c in chunks)
    assert text == "hello from ollama"
    assert chunks[-1].get("done") is True


def test_ollama_unreachable_raises():
    svc = OllamaService("gemma3", host="http://127.0.0.1:9")
    with pytest.raises(ServiceError):
        svc.load_sync()
    with pytest.raises(ServiceError):
        svc.execute({"prompt": "x"})


def test_ollama_missing_prompt():
    svc = OllamaService("gemma3", host="http://127.0.0.1:9")
    with pytest.raises(ServiceError):
        svc.execute({})


class _FakeHFClient:
    def __init__(self, fail=False):
        self.fail = fail
        self.calls = []

    def text_generation(self, prompt, **kw):
        self.calls.append((prompt, kw))
        if self.fail:
            raise RuntimeError("rate limited")
        return "remote says: " + prompt


def test_hf_remote_execute_and_pricing():
    from bee2bee_amd.services.hf_remote import HFRemoteService

    svc = HFRemoteService("some/model", token="t", price_per_token=0.01)
    svc.client = _FakeHFClient()
    res = svc.execute({"prompt": "ping", "max_new_tokens": 8})
    assert res["text"] == "remote says: ping"
    assert res["cost"] == pytest.approx(res["tokens"] * 0.01)
    assert svc.client.calls[0][1]["max_new_tokens"] == 8

    chunks = [json.loads(x) for x in svc.execute_stream({"prompt": "s"})]
    assert chunks[0]["text"].startswith("remote says")
    assert chunks[-1]["done"] is True


def test_hf_remote_error_paths():
    from bee2bee_amd.services.hf_remote import HFRemoteService

    svc = HFRemoteService("m", token="t")
    with pytest.raises(ServiceError):
        svc.execute({"prompt": "x"})  # not initialized
    svc.client = _FakeHFClient(fail=True)
    with pytest.raises(ServiceError):
        svc.execute({"prompt": "x"})
    stream = list(svc.execute_stream({"prompt": "x"}))
    assert json.loads(stream[0])["status"] == "error"
