"""ModelSpec: one dataclass describing a decoder-only transformer
(llama/mistral/mixtral families), plus named presets and HF config.json
parsing.

The reference delegated architecture knowledge to transformers
(bee2bee/hf.py:23-32); here the architecture is explicit because the engine
executes it layer by layer with HIP kernels.
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, Optional


@dataclass
class ModelSpec:
    name: str
    vocab_size: int
    hidden_size: int
    intermediate_size: int
    n_layers: int
    n_heads: int
    n_kv_heads: int
    head_dim: int
    rope_theta: float = 10000.0
    rms_eps: float = 1e-5
    max_seq_len: int = 8192
    tie_embeddings: bool = False
    # MoE (0 experts = dense)
    n_experts: int = 0
    top_k_experts: int = 2
    # Qwen2-style attention bias on the fused QKV projection
    qkv_bias: bool = False
    # architecture family knobs (gpt2: layernorm + learned positions +
    # gelu MLP + biases everywhere + tied embeddings)
    arch: str = "llama"          # "llama" | "gpt2" (HF weight mapping)
    norm_type: str = "rmsnorm"   # "rmsnorm" | "layernorm"
    act_type: str = "swiglu"     # "swiglu" | "gelu" (gelu: fc -> act -> proj)
    pos_type: str = "rope"       # "rope" | "learned"
    attn_out_bias: bool = False
    mlp_bias: bool = False
    # llama3-type RoPE scaling (Llama-3.1/3.2 checkpoints):
    # {"rope_type": "llama3", "factor", "low_freq_factor",
    #  "high_freq_factor", "original_max_position_embeddings"}
    rope_scaling: Optional[Dict[str, Any]] = None
    # vocab specials (byte-tokenizer defaults; overridden by a real tokenizer)
    bos_token_id: int = 1
    eos_token_id: int = 2

    @property
    def is_moe(self) -> bool:
        return self.n_experts > 0

    @property
    def q_size(self) -> int:
        return self.n_heads * self.head_dim

    @property
    def kv_size(self) -> int:
        return self.n_kv_heads * self.head_dim

    def n_params(self) -> int:
        """Parameter count (embeddings + layers + head)."""
        h, i = self.hidden_size, self.intermediate_size
        attn = h * (self.q_size + 2 * self.kv_size) + self.q_size * h
        if self.is_moe:
            mlp = self.n_experts * 3 * h * i + h * self.n_experts
        elif self.act_type == "gelu":
            mlp = 2 * h * i
        else:
            mlp = 3 * h * i
        norms = 2 * h
        per_layer = attn + mlp + norms
        emb = self.vocab_size * h
        head = 0 if self.tie_embeddings else self.vocab_size * h
        return emb + self.n_layers * per_layer + head + h

    def active_params_per_token(self) -> int:
        """Params touched per decoded token (MoE: only top-k experts)."""
        if not self.is_moe:
            return self.n_params()
        h, i = self.hidden_size, self.intermediate_size
        attn = h * (self.q_size + 2 * self.kv_size) + self.q_size * h
        mlp = self.top_k_experts * 3 * h * i + h * self.n_experts
        per_layer = attn + mlp + 2 * h
        emb = self.vocab_size * h
        head = 0 if self.tie_embeddings else self.vocab_size * h
        return emb + self.n_layers * per_layer + head + h


def _llama(name: str, **kw: Any) -> ModelSpec:
    base: Dict[str, Any] = dict(
        rope_theta=500000.0, rms_eps=1e-5, bos_token_id=128000, eos_token_id=128009
    )
    base.update(kw)
    return ModelSpec(name=name, **base)


PRESETS: Dict[str, ModelSpec] = {
    # test-size model for CPU tests and smoke runs
    "tiny": ModelSpec(
        name="tiny",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        n_layers=2,
        n_heads=4,
        n_kv_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        max_seq_len=512,
        tie_embeddings=True,
    ),
    # 3-layer variant so a 3-stage pipeline has a pure-middle rank to test
    "tiny3": ModelSpec(
        name="tiny3",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=128,
        n_layers=3,
        n_heads=4,
        n_kv_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        max_seq_len=512,
        tie_embeddings=True,
    ),
    "tiny-moe": ModelSpec(
        name="tiny-moe",
        vocab_size=512,
        hidden_size=64,
        intermediate_size=96,
        n_layers=2,
        n_heads=4,
        n_kv_heads=2,
        head_dim=16,
        rope_theta=10000.0,
        max_seq_len=512,
        tie_embeddings=True,
        n_experts=4,
        top_k_experts=2,
    ),
    # small demo model served under UNKNOWN names (fallback spec; known
    # families — incl. the real distilgpt2/gpt2 since round 2 — have their
    # own presets below)
    "demo-125m": ModelSpec(
        name="demo-125m",
        vocab_size=32000,
        hidden_size=768,
        intermediate_size=2048,
        n_layers=12,
        n_heads=12,
        n_kv_heads=4,
        head_dim=64,
        rope_theta=10000.0,
        max_seq_len=2048,
        tie_embeddings=True,
    ),
    "llama3.2-1b": _llama(
        "llama3.2-1b",
        vocab_size=128256,
        hidden_size=2048,
        intermediate_size=8192,
        n_layers=16,
        n_heads=32,
        n_kv_heads=8,
        head_dim=64,
        tie_embeddings=True,
        # NOTE: real Llama-3.2 checkpoints ship llama3-type rope_scaling;
        # it is applied automatically when loading such a checkpoint
        # (config.json parsing). The PRESET stays unscaled: synthetic
        # serving at <=8k contexts is numerically indistinguishable and
        # the hardware-validated test corpus pins these numerics.
    ),
    "llama3-8b": _llama(
        "llama3-8b",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        n_layers=32,
        n_heads=32,
        n_kv_heads=8,
        head_dim=128,
    ),
    "llama3-70b": _llama(
        "llama3-70b",
        vocab_size=128256,
        hidden_size=8192,
        intermediate_size=28672,
        n_layers=80,
        n_heads=64,
        n_kv_heads=8,
        head_dim=128,
    ),
    # Llama-3.1: same dims as Llama-3 at 8B/70B. Real 3.1 checkpoints ship
    # llama3-type rope_scaling + 128k context in config.json, which the
    # checkpoint-loading path applies (spec_from_hf_config); presets stay
    # unscaled per the llama3.2-1b note above.
    "llama3.1-8b": _llama(
        "llama3.1-8b",
        vocab_size=128256,
        hidden_size=4096,
        intermediate_size=14336,
        n_layers=32,
        n_heads=32,
        n_kv_heads=8,
        head_dim=128,
    ),
    "llama3.1-70b": _llama(
        "llama3.1-70b",
        vocab_size=128256,
        hidden_size=8192,
        intermediate_size=28672,
        n_layers=80,
        n_heads=64,
        n_kv_heads=8,
        head_dim=128,
    ),
    "llama3.2-3b": _llama(
        "llama3.2-3b",
        vocab_size=128256,
        hidden_size=3072,
        intermediate_size=8192,
        n_layers=28,
        n_heads=24,
        n_kv_heads=8,
        head_dim=128,
        tie_embeddings=True,
    ),
    # Qwen2.5-7B-Instruct class: llama-shaped + QKV bias, large vocab
    "qwen2.5-7b": ModelSpec(
        name="qwen2.5-7b",
        vocab_size=152064,
        hidden_size=3584,
        intermediate_size=18944,
        n_layers=28,
        n_heads=28,
        n_kv_heads=4,
        head_dim=128,
        rope_theta=1000000.0,
        rms_eps=1e-6,
        max_seq_len=32768,
        tie_embeddings=False,
        qkv_bias=True,
    ),
    # Qwen2.5 larger sizes (14B/32B/72B): same family, GQA groups 5/5/8 —
    # all decode-kernel-instantiated group sizes
    "qwen2.5-14b": ModelSpec(
        name="qwen2.5-14b",
        vocab_size=152064,
        hidden_size=5120,
        intermediate_size=13824,
        n_layers=48,
        n_heads=40,
        n_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        rms_eps=1e-6,
        max_seq_len=32768,
        tie_embeddings=False,
        qkv_bias=True,
    ),
    "qwen2.5-32b": ModelSpec(
        name="qwen2.5-32b",
        vocab_size=152064,
        hidden_size=5120,
        intermediate_size=27648,
        n_layers=64,
        n_heads=40,
        n_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        rms_eps=1e-6,
        max_seq_len=32768,
        tie_embeddings=False,
        qkv_bias=True,
    ),
    "qwen2.5-72b": ModelSpec(
        name="qwen2.5-72b",
        vocab_size=152064,
        hidden_size=8192,
        intermediate_size=29568,
        n_layers=80,
        n_heads=64,
        n_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        rms_eps=1e-6,
        max_seq_len=32768,
        tie_embeddings=False,
        qkv_bias=True,
    ),
    # Zephyr-7B = Mistral-7B architecture (BASELINE config 3)
    "zephyr-7b": ModelSpec(
        name="zephyr-7b",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        n_layers=32,
        n_heads=32,
        n_kv_heads=8,
        head_dim=128,
        rope_theta=10000.0,
        max_seq_len=8192,
    ),
    "mixtral-8x7b": ModelSpec(
        name="mixtral-8x7b",
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        n_layers=32,
        n_heads=32,
        n_kv_heads=8,
        head_dim=128,
        rope_theta=1000000.0,
        max_seq_len=8192,
        n_experts=8,
        top_k_experts=2,
    ),
}

def _gpt2(name: str, n_layers: int, hidden: int, n_heads: int) -> ModelSpec:
    return ModelSpec(
        name=name,
        vocab_size=50257,
        hidden_size=hidden,
        intermediate_size=4 * hidden,
        n_layers=n_layers,
        n_heads=n_heads,
        n_kv_heads=n_heads,
        head_dim=hidden // n_heads,
        rms_eps=1e-5,
        max_seq_len=1024,
        tie_embeddings=True,
        qkv_bias=True,
        arch="gpt2",
        norm_type="layernorm",
        act_type="gelu",
        pos_type="learned",
        attn_out_bias=True,
        mlp_bias=True,
        bos_token_id=50256,
        eos_token_id=50256,
    )


PRESETS["distilgpt2"] = _gpt2("distilgpt2", 6, 768, 12)
PRESETS["gpt2"] = _gpt2("gpt2", 12, 768, 12)
PRESETS["gpt2-medium"] = _gpt2("gpt2-medium", 24, 1024, 16)
PRESETS["tiny-gpt2"] = _gpt2("tiny-gpt2", 2, 64, 4)
PRESETS["tiny-gpt2"].vocab_size = 512
PRESETS["tiny-gpt2"].max_seq_len = 512

_ALIASES = {
    "distilbert/distilgpt2": "distilgpt2",
    "openai-community/gpt2": "gpt2",
    "openai-community/gpt2-medium": "gpt2-medium",
    "meta-llama/meta-llama-3-8b": "llama3-8b",
    "meta-llama/llama-3-8b": "llama3-8b",
    "llama-3-8b": "llama3-8b",
    "llama3": "llama3-8b",
    "meta-llama/meta-llama-3-70b": "llama3-70b",
    "llama-3-70b": "llama3-70b",
    "llama-3.2-1b": "llama3.2-1b",
    "meta-llama/llama-3.2-1b": "llama3.2-1b",
    "llama3.2": "llama3.2-1b",
    "llama-3.2-3b": "llama3.2-3b",
    "meta-llama/llama-3.2-3b": "llama3.2-3b",
    "meta-llama/llama-3.1-8b": "llama3.1-8b",
    "meta-llama/meta-llama-3.1-8b": "llama3.1-8b",
    "llama-3.1-8b": "llama3.1-8b",
    "llama3.1": "llama3.1-8b",
    "meta-llama/llama-3.1-70b": "llama3.1-70b",
    "meta-llama/meta-llama-3.1-70b": "llama3.1-70b",
    "llama-3.1-70b": "llama3.1-70b",
    "huggingfaceh4/zephyr-7b-beta": "zephyr-7b",
    "qwen/qwen2.5-7b-instruct": "qwen2.5-7b",
    "qwen/qwen2.5-7b": "qwen2.5-7b",
    "qwen/qwen2.5-14b": "qwen2.5-14b",
    "qwen/qwen2.5-32b": "qwen2.5-32b",
    "qwen/qwen2.5-72b": "qwen2.5-72b",
    "qwen2.5": "qwen2.5-7b",
    "zephyr": "zephyr-7b",
    "zephyr-7b-beta": "zephyr-7b",
    "mistral-7b": "zephyr-7b",
    "mistralai/mistral-7b-v0.1": "zephyr-7b",
    "mistralai/mixtral-8x7b-v0.1": "mixtral-8x7b",
    "mixtral": "mixtral-8x7b",
    "mixtral-8x7b-v0.1": "mixtral-8x7b",
}


def resolve_spec(name: str, model_path: Optional[str] = None) -> ModelSpec:
    """Resolve a model name (preset / alias / HF id / local dir) to a spec.

    A local checkpoint dir's config.json wins; otherwise presets/aliases;
    unknown names fall back to the demo spec (served with random weights —
    there is no network for checkpoint downloads in this environment)."""
    if model_path and os.path.isfile(os.path.join(model_path, "config.json")):
        return spec_from_hf_config(model_path, name)
    key = name.lower().strip()
    if key in PRESETS:
        return PRESETS[key]
    if key in _ALIASES:
        return PRESETS[_ALIASES[key]]
    # Prefix match only, and only across a variant-suffix boundary ("-instruct",
    # "_chat", ...). A bare substring fallback mapped e.g. "llama3.2-1b" to the
    # llama3-8b preset: a remainder that continues with digits is a
    # different model size, not a variant, so it must fall through to the demo
    # spec rather than silently serve the wrong architecture.
    # preset keys participate too: "llama3-8b-instruct" is a variant of
    # the llama3-8b preset even though no alias spells it out
    candidates = set(_ALIASES) | {p for p in PRESETS
                                  if not p.startswith(("tiny", "demo"))}
    for cand in sorted(candidates, key=len, reverse=True):
        if key.startswith(cand):
            rest = key[len(cand):]
            if rest and rest[0] in "-_/." and len(rest) > 1 and rest[1].isalpha():
                return PRESETS[_ALIASES.get(cand, cand)]
    spec = PRESETS["demo-125m"]
    return ModelSpec(**{**spec.__dict__, "name": name})


def spec_from_hf_config(model_path: str, name: Optional[str] = None) -> ModelSpec:
    """Parse a HuggingFace config.json (Llama/Mistral/Mixtral/GPT-2 keys)."""
    with open(os.path.join(model_path, "config.json")) as f:
        cfg = json.load(f)
    if cfg.get("model_type") == "gpt2" or "n_embd" in cfg:
        spec = _gpt2(
            name or cfg.get("_name_or_path", os.path.basename(model_path)),
            cfg.get("n_layer", 12), cfg.get("n_embd", 768),
            cfg.get("n_head", 12),
        )
        spec.vocab_size = cfg.get("vocab_size", 50257)
        spec.max_seq_len = cfg.get("n_positions", 1024)
        spec.rms_eps = cfg.get("layer_norm_epsilon", 1e-5)
        return spec
    n_heads = cfg.get("num_attention_heads", 32)
    hidden = cfg.get("hidden_size", 4096)
    return ModelSpec(
        name=name or cfg.get("_name_or_path", os.path.basename(model_path)),
        vocab_size=cfg.get("vocab_size", 32000),
        hidden_size=hidden,
        intermediate_size=cfg.get("intermediate_size", 11008),
        n_layers=cfg.get("num_hidden_layers", 32),
        n_heads=n_heads,
        n_kv_heads=cfg.get("num_key_value_heads", n_heads),
        head_dim=cfg.get("head_dim", hidden // n_heads),
        rope_theta=cfg.get("rope_theta", 10000.0),
        rms_eps=cfg.get("rms_norm_eps", 1e-5),
        max_seq_len=min(cfg.get("max_position_embeddings", 8192), 131072),
        tie_embeddings=cfg.get("tie_word_embeddings", False),
        rope_scaling=(
            cfg.get("rope_scaling")
            if (cfg.get("rope_scaling") or {}).get(
                "rope_type", (cfg.get("rope_scaling") or {}).get("type"))
            == "llama3" else None
        ),
        n_experts=cfg.get("num_local_experts", 0),
        top_k_experts=cfg.get("num_experts_per_tok", 2),
        # Qwen2 sets attention_bias (or ships q/k/v bias tensors implicitly)
        qkv_bias=bool(
            cfg.get("attention_bias", False)
            or cfg.get("model_type") == "qwen2"
        ),
        bos_token_id=cfg.get("bos_token_id", 1) or 1,
        eos_token_id=(
            cfg.get("eos_token_id")[0]
            if isinstance(cfg.get("eos_token_id"), list)
            else cfg.get("eos_token_id", 2)
        )
        or 2,
    )
