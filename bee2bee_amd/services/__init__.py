"""Pluggable inference backends behind one interface.

Parity: reference bee2bee/services.py — BaseService.execute/execute_stream/
get_metadata/load_sync (:13-25) with concrete backends. The reference's
HFService (transformers.generate) is replaced by NativeEngineService: the
hand-written CDNA4 HIP engine.
"""

from .base import BaseService, ServiceError  # noqa: F401
