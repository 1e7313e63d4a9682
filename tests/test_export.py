"""TorchScript/ONNX export parity (reference bee2bee/hf.py:139-158): the
exported pure-torch model must match the engine's forward, and the traced
TorchScript artifact must load and run."""
import numpy as np
import pytest
import torch

from bee2bee_amd.models.export import ExportableModel, export_torchscript
from bee2bee_amd.models.spec import PRESETS
from bee2bee_amd.models.weights import ModelWeights


def _engine_logits(spec, weights, ids):
    from bee2bee_amd.engine.kv import PagedKV
    from bee2bee_amd.engine.runner import Runner

    kv = PagedKV(spec, torch.device("cpu"), torch.float32, n_blocks=16)
    runner = Runner(spec, weights, kv, torch.device("cpu"), torch.float32)
    T = len(ids)
    kv.new_seq(0)
    kv.extend_seq(0, T)
    slots = torch.tensor(kv.slot_mapping(0, range(T)), dtype=torch.int32)
    pos = torch.arange(T, dtype=torch.int32)
    cu = torch.tensor([0, T], dtype=torch.int32)
    hidden = runner.forward_prefill(
        torch.tensor(ids, dtype=torch.int64), pos, slots, cu, T
    )
    return runner.lm_head(hidden)


def test_exportable_model_matches_engine():
    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(9)
    ids = [5, 9, 100, 42, 7]
    ref = _engine_logits(spec, w, ids)
    model = ExportableModel(spec, w).eval()
    with torch.no_grad():
        got = model(torch.tensor([ids], dtype=torch.int64))[0]
    assert torch.allclose(got, ref, atol=1e-4), (got - ref).abs().max()


def test_torchscript_roundtrip(tmp_path):
    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(9)
    path = str(tmp_path / "tiny.pt")
    export_torchscript(spec, w, path)
    loaded = torch.jit.load(path)
    ids = torch.tensor([[5, 9, 100]], dtype=torch.int64)
    with torch.no_grad():
        a = loaded(ids)
        b = ExportableModel(spec, w).eval()(ids)
    assert torch.allclose(a, b, atol=1e-5)


def test_onnx_gated():
    from bee2bee_amd.models.export import export_onnx

    spec = PRESETS["tiny"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(9)
    try:
        import onnx  # noqa: F401

        has = True
    except Exception:
        has = False
    if not has:
        with pytest.raises(RuntimeError, match="onnx_support_missing"):
            export_onnx(spec, w, "/tmp/x.onnx")


def test_export_with_qkv_bias_matches_engine():
    import dataclasses

    spec = dataclasses.replace(PRESETS["tiny"], name="tiny-qb", qkv_bias=True)
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(9)
    ids = [5, 9, 100]
    ref = _engine_logits(spec, w, ids)
    model = ExportableModel(spec, w).eval()
    with torch.no_grad():
        got = model(torch.tensor([ids], dtype=torch.int64))[0]
    assert torch.allclose(got, ref, atol=1e-4), (got - ref).abs().max()


def test_gpt2_export_matches_transformers(tmp_path):
    """TorchScript export of a GPT-2-family model: traced logits equal
    transformers' GPT2LMHeadModel on the same checkpoint."""
    import pytest
    import torch

    transformers = pytest.importorskip("transformers")
    from bee2bee_amd.models.export import export_torchscript
    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.weights import ModelWeights, save_hf

    spec = PRESETS["tiny-gpt2"]
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(5)
    path = str(tmp_path / "m.pt")
    export_torchscript(spec, w, path, example_len=6)
    traced = torch.jit.load(path)

    save_hf(w, str(tmp_path / "hf"))
    hf = transformers.GPT2LMHeadModel.from_pretrained(
        str(tmp_path / "hf"), torch_dtype=torch.float32).eval()
    ids = torch.tensor([[3, 9, 100, 7, 45, 2]])
    with torch.no_grad():
        got = traced(ids)
        want = hf(ids).logits
    assert torch.allclose(got, want, atol=2e-3), (got - want).abs().max()
