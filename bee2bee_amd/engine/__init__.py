"""The MI355X inference engine: paged KV cache, layer runner, hipGraph-
captured decode, sampling, and the request-serving front end."""

from .engine import InferenceEngine, GenerationRequest  # noqa: F401
