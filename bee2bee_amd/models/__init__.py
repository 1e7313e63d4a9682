"""Model family definitions: specs, HF-format weights, tokenizers.

Covers the llama family (Llama-3 8B/70B, Llama-3.2 1B, Mistral-7B/Zephyr) and
the Mixtral 8x7B MoE family — the model set named by BASELINE.json's configs.
"""

from .spec import ModelSpec, resolve_spec, PRESETS  # noqa: F401
