import json

from bee2bee_amd.models.spec import PRESETS, resolve_spec, spec_from_hf_config


def test_llama3_8b_param_count():
    spec = PRESETS["llama3-8b"]
    # Llama-3-8B is 8.03B parameters
    assert abs(spec.n_params() / 1e9 - 8.03) < 0.1


def test_llama3_70b_param_count():
    spec = PRESETS["llama3-70b"]
    assert abs(spec.n_params() / 1e9 - 70.6) < 1.0


def test_mixtral_param_counts():
    spec = PRESETS["mixtral-8x7b"]
    assert abs(spec.n_params() / 1e9 - 46.7) < 1.0
    # ~12.9B active per token (2 of 8 experts)
    assert abs(spec.active_params_per_token() / 1e9 - 12.9) < 0.5


def test_alias_resolution():
    assert resolve_spec("meta-llama/Meta-Llama-3-8B").name == "llama3-8b"
    assert resolve_spec("HuggingFaceH4/zephyr-7b-beta").name == "zephyr-7b"
    assert resolve_spec("mistralai/Mixtral-8x7B-v0.1").is_moe


def test_unknown_falls_back_to_demo():
    spec = resolve_spec("some-model-nobody-knows")
    assert spec.hidden_size == 768  # demo spec under the requested name
    assert spec.name == "some-model-nobody-knows"


def test_distilgpt2_is_a_real_gpt2_spec():
    # round 2: distilgpt2 (BASELINE config 1) is a REAL architecture now,
    # not the llama-shaped demo spec under a borrowed name
    spec = resolve_spec("distilgpt2")
    assert (spec.arch, spec.norm_type, spec.act_type, spec.pos_type) == (
        "gpt2", "layernorm", "gelu", "learned")
    assert spec.n_layers == 6 and spec.n_heads == 12
    assert resolve_spec("distilbert/distilgpt2").name == "distilgpt2"


def test_hf_config_parsing(tmp_path):
    cfg = {
        "vocab_size": 32000,
        "hidden_size": 4096,
        "intermediate_size": 14336,
        "num_hidden_layers": 32,
        "num_attention_heads": 32,
        "num_key_value_heads": 8,
        "rope_theta": 10000.0,
        "rms_norm_eps": 1e-5,
        "max_position_embeddings": 32768,
        "num_local_experts": 8,
        "num_experts_per_tok": 2,
    }
    (tmp_path / "config.json").write_text(json.dumps(cfg))
    spec = spec_from_hf_config(str(tmp_path))
    assert spec.n_experts == 8 and spec.head_dim == 128
    spec2 = resolve_spec("whatever", model_path=str(tmp_path))
    assert spec2.n_experts == 8


def test_qwen2_preset_and_bias_roundtrip(tmp_path):
    """Qwen2 family: QKV-bias weights roundtrip through HF save/load and
    the engine decodes with the bias applied."""
    import dataclasses

    import torch

    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams
    from bee2bee_amd.models.spec import PRESETS, resolve_spec
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    assert PRESETS["qwen2.5-7b"].qkv_bias
    assert resolve_spec("Qwen/Qwen2.5-7B-Instruct").name == "qwen2.5-7b"

    tiny_q = dataclasses.replace(PRESETS["tiny"], name="tiny-qwen", qkv_bias=True)
    w = ModelWeights(tiny_q, torch.device("cpu"), torch.float32).random_init(4)
    assert w.layers[0].wqkv_bias is not None
    save_hf(w, str(tmp_path))
    w2 = ModelWeights(tiny_q, torch.device("cpu"), torch.float32).load_hf(str(tmp_path))
    assert torch.allclose(w.layers[1].wqkv_bias, w2.layers[1].wqkv_bias)

    # bias must change the output vs the same weights without bias
    def gen(spec, seed=4):
        eng = InferenceEngine(spec, device="cpu", max_batch=2,
                              max_seq_len=64, seed=seed)
        try:
            r = GenerationRequest(prompt_ids=[4, 5, 6], max_new_tokens=5,
                                  sampling=SamplingParams(greedy=True))
            eng.submit(r)
            while True:
                x = r.out_queue.get(timeout=60)
                if not isinstance(x, int):
                    break
            assert r.error is None, r.error
            return r.output_ids
        finally:
            eng.shutdown()

    with_bias = gen(tiny_q)
    assert len(with_bias) == 5


def test_llama3_rope_scaling_matches_transformers():
    """rope_tables with the llama3 rope_type must reproduce transformers'
    ROPE_INIT_FUNCTIONS['llama3'] frequencies exactly (Llama-3.1/3.2
    checkpoint compatibility)."""
    import pytest
    import torch

    transformers = pytest.importorskip("transformers")
    from transformers.modeling_rope_utils import ROPE_INIT_FUNCTIONS

    from bee2bee_amd.ops.reference import rope_tables

    scaling = {"rope_type": "llama3", "factor": 32.0,
               "low_freq_factor": 1.0, "high_freq_factor": 4.0,
               "original_max_position_embeddings": 8192}
    cfg = transformers.LlamaConfig(
        hidden_size=2048, num_attention_heads=32, head_dim=64,
        rope_theta=500000.0, max_position_embeddings=131072,
        rope_scaling=dict(scaling))
    hf_inv, _ = ROPE_INIT_FUNCTIONS["llama3"](cfg, "cpu")
    cos, sin = rope_tables(64, 64, 500000.0, "cpu", scaling=scaling)
    ours = torch.atan2(sin[1], cos[1])  # inv_freq from position-1 angles
    assert (ours - hf_inv).abs().max().item() < 1e-6


def test_rope_scaling_parsed_from_hf_config(tmp_path):
    import json

    cfg = {"vocab_size": 1024, "hidden_size": 64, "intermediate_size": 128,
           "num_hidden_layers": 2, "num_attention_heads": 4,
           "num_key_value_heads": 2, "head_dim": 16,
           "rope_theta": 500000.0,
           "rope_scaling": {"rope_type": "llama3", "factor": 8.0,
                            "low_freq_factor": 1.0, "high_freq_factor": 4.0,
                            "original_max_position_embeddings": 8192}}
    (tmp_path / "config.json").write_text(json.dumps(cfg))
    from bee2bee_amd.models.spec import resolve_spec

    spec = resolve_spec("x", model_path=str(tmp_path))
    assert spec.rope_scaling["factor"] == 8.0
    # non-llama3 types are ignored (unsupported) rather than misapplied
    cfg["rope_scaling"] = {"rope_type": "yarn", "factor": 4.0}
    (tmp_path / "config.json").write_text(json.dumps(cfg))
    assert resolve_spec("x", model_path=str(tmp_path)).rope_scaling is None


def test_llama31_32_and_qwen_size_presets():
    """HF ids of the Llama-3.1/3.2 and larger Qwen2.5 sizes resolve to the
    real architectures (previously fell through to the demo spec), with
    parameter counts matching the published models and GQA group sizes the
    decode kernel instantiates (1..8)."""
    from bee2bee_amd.models.spec import resolve_spec

    cases = {
        "meta-llama/Llama-3.1-8B-Instruct": ("llama3.1-8b", 8.03),
        "meta-llama/Meta-Llama-3.1-70B": ("llama3.1-70b", 70.55),
        "meta-llama/Llama-3.2-3B-Instruct": ("llama3.2-3b", 3.21),
        "meta-llama/Llama-3.2-1B": ("llama3.2-1b", 1.24),
        "Qwen/Qwen2.5-14B-Instruct": ("qwen2.5-14b", 14.77),
        "Qwen/Qwen2.5-32B": ("qwen2.5-32b", 32.76),
        "Qwen/Qwen2.5-72B-Instruct": ("qwen2.5-72b", 72.71),
    }
    for hf_id, (name, billions) in cases.items():
        spec = resolve_spec(hf_id)
        assert spec.name == name, (hf_id, spec.name)
        assert abs(spec.n_params() / 1e9 - billions) < 0.05, (hf_id, spec.n_params())
        assert spec.head_dim in (64, 128)
        assert 1 <= spec.n_heads // spec.n_kv_heads <= 8
    # qwen family traits hold at every size
    for n in ("qwen2.5-14b", "qwen2.5-32b", "qwen2.5-72b"):
        s = resolve_spec(n)
        assert s.qkv_bias and not s.tie_embeddings and s.rope_theta == 1e6
