from __future__ import annotations

from typing import Any, Dict, Iterator


class ServiceError(Exception):
    pass


class BaseService:
    """One inference backend. Sync methods; the mesh runs them in executor
    threads (reference contract: bee2bee/services.py:13-25).

    execute(params) -> {"text", "tokens", "latency_ms", "price_per_token",
    "cost"}; execute_stream(params) yields JSON-line strings
    '{"text": ...}\\n' ending with '{"done": true}\\n'.
    """

    def __init__(self, name: str) -> None:
        self.name = name

    def load_sync(self) -> None:  # pragma: no cover - interface
        pass

    def get_metadata(self) -> Dict[str, Any]:
        return {}

    def execute(self, params: Dict[str, Any]) -> Dict[str, Any]:
        raise NotImplementedError

    def execute_stream(self, params: Dict[str, Any]) -> Iterator[str]:
        raise NotImplementedError
