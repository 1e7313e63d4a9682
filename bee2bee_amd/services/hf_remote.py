"""HF Inference-API proxy backend (reference bee2bee/services.py:247-308)."""
from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, Iterator, Optional

from .base import BaseService, ServiceError


class HFRemoteService(BaseService):
    def __init__(
        self,
        model_name: str,
        token: Optional[str] = None,
        price_per_token: float = 0.005,
    ) -> None:
        super().__init__("hf_remote")
        self.model_name = model_name
        self.token = token or os.getenv("HUGGING_FACE_HUB_TOKEN")
        self.price_per_token = price_per_token
        self.client = None

    def load_sync(self) -> None:
        try:
            from huggingface_hub import InferenceClient
        except ImportError as e:
            raise ServiceError("huggingface_hub not installed") from e
        try:
            self.client = InferenceClient(model=self.model_name, token=self.token)
        except Exception as e:
            raise ServiceError(f"failed to init HF remote client: {e}") from e

    def get_metadata(self) -> Dict[str, Any]:
        return {
            "models": [self.model_name],
            "price_per_token": self.price_per_token,
            "tag": "remote",
            "backend": "hf_remote",
        }

    def execute(self, params: Dict[str, Any]) -> Dict[str, Any]:
        if self.client is None:
            raise ServiceError("remote client not initialized")
        prompt = params.get("prompt")
        if not prompt:
            raise ServiceError("missing prompt")
        t0 = time.time()
        try:
            response = self.client.text_generation(
                prompt,
                max_new_tokens=int(params.get("max_new_tokens", 32)),
                temperature=params.get("temperature", 0.7),
                do_sample=bool(params.get("do_sample", True)),
            )
        except Exception as e:
            raise ServiceError(f"HF remote execution error: {e}") from e
        latency_ms = int((time.time() - t0) * 1000.0)
        tokens = len(response) // 4  # API does not report counts; estimate
        return {
            "text": response,
            "tokens": tokens,
            "latency_ms": latency_ms,
            "price_per_token": self.price_per_token,
            "cost": self.price_per_token * tokens,
            "backend": "hf_remote",
        }

    def execute_stream(self, params: Dict[str, Any]) -> Iterator[str]:
        # The Inference API is buffered here; emit one chunk then done.
        try:
            result = self.execute(params)
            yield json.dumps({"text": result["text"]}) + "\n"
            yield json.dumps({"done": True}) + "\n"
        except Exception as e:
            yield json.dumps({"status": "error", "message": str(e)}) + "\n"
