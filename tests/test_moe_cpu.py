"""MoE path equivalence: dense all-experts (decode) vs per-expert gather
(prefill) must agree; engine-level MoE determinism."""
import torch

from bee2bee_amd.engine.kv import PagedKV
from bee2bee_amd.engine.runner import Runner
from bee2bee_amd.models.spec import PRESETS
from bee2bee_amd.models.weights import ModelWeights


def _runner():
    spec = PRESETS["tiny-moe"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(17)
    kv = PagedKV(spec, torch.device("cpu"), torch.float32, n_blocks=8)
    return Runner(spec, w, kv, torch.device("cpu"), torch.float32), w


def test_dense_equals_padded_equals_gather():
    runner, w = _runner()
    x = torch.randn(9, w.spec.hidden_size,
                    generator=torch.Generator().manual_seed(4))
    dense = runner._moe_mlp(w.layers[0], x)  # T=9 <= threshold -> dense
    saved_d, saved_p = Runner.MOE_DENSE_MAX_TOKENS, Runner.MOE_BMM_MAX_TOKENS
    try:
        Runner.MOE_DENSE_MAX_TOKENS = 0  # force sorted (padded-bmm on CPU)
        padded = runner._moe_mlp(w.layers[0], x)
        Runner.MOE_BMM_MAX_TOKENS = 0  # force per-expert gather path
        gather = runner._moe_mlp(w.layers[0], x)
    finally:
        Runner.MOE_DENSE_MAX_TOKENS = saved_d
        Runner.MOE_BMM_MAX_TOKENS = saved_p
    assert torch.allclose(dense, padded, atol=1e-5), (dense - padded).abs().max()
    assert torch.allclose(dense, gather, atol=1e-5), (dense - gather).abs().max()


def test_dense_path_handles_all_tokens_one_expert():
    runner, w = _runner()
    # degenerate gate: force expert 0 to dominate
    w.layers[0].moe_gate.data = torch.zeros_like(w.layers[0].moe_gate)
    w.layers[0].moe_gate.data[0] = 1.0
    x = torch.ones(5, w.spec.hidden_size)
    out = runner._moe_mlp(w.layers[0], x)
    assert torch.isfinite(out).all()
