"""Ollama proxy backend (reference bee2bee/services.py:118-245): probes
/api/tags, resolves tagged model names by substring, and proxies /api/generate
in buffered and streaming form."""
from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, Iterator

import requests

from .base import BaseService, ServiceError


class OllamaService(BaseService):
    def __init__(self, model_name: str, host: str | None = None) -> None:
        super().__init__("ollama")
        self.model_name = model_name
        self.host = host or os.getenv("OLLAMA_HOST", "http://localhost:11434")
        self.price_per_token = 0.0
        self.actual_model = model_name

    def load_sync(self) -> None:
        try:
            res = requests.get(f"{self.host}/api/tags", timeout=5)
        except Exception as e:
            raise ServiceError(f"ollama connection failed: {e}") from e
        if res.status_code != 200:
            raise ServiceError(f"ollama reachable but returned {res.status_code}")
        models = [m.get("name", "") for m in res.json().get("models", [])]
        for m in models:
            if self.model_name == m or self.model_name in m or m in self.model_name:
                self.actual_model = m
                break

    def get_metadata(self) -> Dict[str, Any]:
        return {
            "models": sorted({self.model_name, self.actual_model}),
            "price_per_token": self.price_per_token,
            "backend": "ollama",
        }

    def execute(self, params: Dict[str, Any]) -> Dict[str, Any]:
        prompt = params.get("prompt")
        if not prompt:
            raise ServiceError("missing prompt")
        payload = {
            "model": self.actual_model,
            "prompt": prompt,
            "stream": False,
            "options": {
                "num_predict": int(params.get("max_new_tokens", 2048)),
                "temperature": float(params.get("temperature", 0.7)),
            },
        }
        t0 = time.time()
        try:
            res = requests.post(f"{self.host}/api/generate", json=payload, timeout=300)
        except Exception as e:
            raise ServiceError(f"ollama exec error: {e}") from e
        if res.status_code != 200:
            raise ServiceError(f"ollama error: {res.text}")
        data = res.json()
        duration_ns = data.get("total_duration", 0)
        latency_ms = duration_ns / 1e6 if duration_ns else (time.time() - t0) * 1000.0
        return {
            "text": data.get("response", ""),
            "tokens": data.get("eval_count", 0),
            "latency_ms": latency_ms,
            "price_per_token": self.price_per_token,
            "cost": 0.0,
        }

    def execute_stream(self, params: Dict[str, Any]) -> Iterator[str]:
        payload = {
            "model": self.actual_model,
            "prompt": params.get("prompt", ""),
            "stream": True,
            "options": {
                "num_predict": int(params.get("max_new_tokens", 2048)),
                "temperature": float(params.get("temperature", 0.7)),
            },
        }
        try:
            res = requests.post(
                f"{self.host}/api/generate", json=payload, stream=True, timeout=300
            )
        except Exception as e:
            yield json.dumps({"status": "error", "message": str(e)}) + "\n"
            return
        if res.status_code != 200:
            yield json.dumps({"status": "error", "message": res.text}) + "\n"
            return
        for line in res.iter_lines():
            if not line:
                continue
            try:
                data = json.loads(line.decode("utf-8"))
            except Exception:
                continue
            chunk = data.get("response", "")
            if chunk:
                yield json.dumps({"text": chunk}) + "\n"
            if data.get("done"):
                break
        yield json.dumps({"done": True}) + "\n"
