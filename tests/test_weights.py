import torch

from bee2bee_amd.models.spec import PRESETS
from bee2bee_amd.models.weights import ModelWeights, save_hf


def test_random_init_shapes():
    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init()
    assert w.embed.shape == (spec.vocab_size, spec.hidden_size)
    assert w.lm_head is w.embed  # tied
    lw = w.layers[0]
    assert lw.wqkv.shape == (spec.q_size + 2 * spec.kv_size, spec.hidden_size)
    assert lw.w_gate_up.shape == (2 * spec.intermediate_size, spec.hidden_size)


def test_hf_roundtrip(tmp_path):
    """save_hf (unfused HF names) -> load_hf (refused) must be identical —
    the checkpoint-compatibility guarantee."""
    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(seed=3)
    save_hf(w, str(tmp_path))
    assert (tmp_path / "model.safetensors").exists()
    assert (tmp_path / "config.json").exists()

    w2 = ModelWeights(spec, torch.device("cpu"), torch.float32).load_hf(str(tmp_path))
    assert torch.equal(w.embed, w2.embed)
    for a, b in zip(w.layers, w2.layers):
        assert torch.equal(a.wqkv, b.wqkv)
        assert torch.equal(a.wo, b.wo)
        assert torch.equal(a.w_gate_up, b.w_gate_up)
        assert torch.equal(a.w_down, b.w_down)


def test_moe_roundtrip(tmp_path):
    spec = PRESETS["tiny-moe"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(seed=5)
    save_hf(w, str(tmp_path))
    w2 = ModelWeights(spec, torch.device("cpu"), torch.float32).load_hf(str(tmp_path))
    for a, b in zip(w.layers, w2.layers):
        assert torch.equal(a.moe_gate, b.moe_gate)
        assert torch.equal(a.moe_w_gate_up, b.moe_w_gate_up)
        assert torch.equal(a.moe_w_down, b.moe_w_down)


def test_layer_range_partial_load():
    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(
        layer_range=(1, 2)
    )
    assert w.layers[0].wqkv is None
    assert w.layers[1].wqkv is not None
    assert w.embed is None  # not first stage
    assert w.final_norm is not None  # last stage


def test_expert_range_shard_matches_full():
    """An EP rank's expert shard must hold exactly the full init's values
    for its expert slice (per-expert seeded draws), for both random_init
    and the HF checkpoint loader."""
    import torch

    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    spec = PRESETS["tiny-moe"]
    dev = torch.device("cpu")
    full = ModelWeights(spec, dev, torch.float32).random_init(3)
    shard = ModelWeights(spec, dev, torch.float32).random_init(
        3, expert_range=(2, 4)
    )
    assert shard.expert_range == (2, 4)
    assert shard.layers[0].moe_w_gate_up.shape[0] == 2
    assert torch.equal(
        shard.layers[0].moe_w_gate_up, full.layers[0].moe_w_gate_up[2:4]
    )
    assert torch.equal(
        shard.layers[1].moe_w_down, full.layers[1].moe_w_down[2:4]
    )
    # gate replicated
    assert torch.equal(shard.layers[0].moe_gate, full.layers[0].moe_gate)


def test_expert_range_load_hf(tmp_path):
    import torch

    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    spec = PRESETS["tiny-moe"]
    dev = torch.device("cpu")
    full = ModelWeights(spec, dev, torch.float32).random_init(3)
    save_hf(full, str(tmp_path))
    shard = ModelWeights(spec, dev, torch.float32).load_hf(
        str(tmp_path), expert_range=(1, 3)
    )
    assert shard.layers[0].moe_w_gate_up.shape[0] == 2
    assert torch.allclose(
        shard.layers[0].moe_w_gate_up, full.layers[0].moe_w_gate_up[1:3]
    )
    assert torch.allclose(
        shard.layers[1].moe_w_down, full.layers[1].moe_w_down[1:3]
    )


def test_tied_head_loads_on_pp_last_stage(tmp_path):
    """PP last stages of tied-embedding checkpoints must get lm_head from
    the embedding tensor (regression: both loaders left it None)."""
    import torch

    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    for preset in ("tiny", "tiny-gpt2"):  # llama-format and gpt2-format
        spec = PRESETS[preset]
        assert spec.tie_embeddings
        w = ModelWeights(spec, torch.device("cpu"),
                         torch.float32).random_init(6)
        out = tmp_path / preset
        save_hf(w, str(out))
        last = ModelWeights(spec, torch.device("cpu"), torch.float32).load_hf(
            str(out), layer_range=(1, spec.n_layers))
        assert last.embed is None, preset
        assert last.lm_head is not None, preset
        assert torch.equal(last.lm_head, w.embed), preset


def test_save_load_bf16_exact(tmp_path):
    """bf16 checkpoints round-trip bit-exactly (the serving dtype)."""
    import torch

    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.bfloat16).random_init(13)
    save_hf(w, str(tmp_path))
    w2 = ModelWeights(spec, torch.device("cpu"), torch.bfloat16).load_hf(
        str(tmp_path))
    assert w2.embed.dtype == torch.bfloat16
    assert torch.equal(w.embed, w2.embed)
    assert torch.equal(w.layers[1].w_down, w2.layers[1].w_down)
    assert torch.equal(w.final_norm, w2.final_norm)
