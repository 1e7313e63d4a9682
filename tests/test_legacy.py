"""Legacy coordinator surface: MLP math gradients + worker task execution
(reference bee2bee/node.py task kinds, bee2bee/model.py math)."""
import numpy as np
import pytest

from bee2bee_amd.legacy import mlp, protocol
from bee2bee_amd.legacy.worker import LegacyWorker


def test_mlp_shapes_and_json_roundtrip():
    layers = mlp.random_mlp([4, 8, 2], act="gelu", seed=1)
    x = np.random.default_rng(0).standard_normal((3, 4))
    y = x
    for l in layers:
        y = mlp.layer_forward(l, y)
    assert y.shape == (3, 2)
    restored = mlp.mlp_from_json(mlp.mlp_to_json(layers))
    y2 = x
    for l in restored:
        y2 = mlp.layer_forward(l, y2)
    assert np.allclose(y, y2)


@pytest.mark.parametrize("act", ["relu", "gelu", "none"])
def test_act_derivative_numerically(act):
    layer = mlp.Layer(
        w=np.eye(3), b=np.zeros(3), act=act
    )
    z = np.array([[-1.5, 0.3, 2.0]])
    eps = 1e-6

    def f(v):
        return mlp.layer_forward(mlp.Layer(np.eye(3), np.zeros(3), act), v)

    num = (f(z + eps) - f(z - eps)) / (2 * eps)
    ana = mlp.act_derivative(layer, z)
    assert np.allclose(num, ana, atol=1e-5)


def test_layer_backward_matches_numeric_grad():
    rng = np.random.default_rng(2)
    layer = mlp.Layer(w=rng.standard_normal((4, 3)), b=rng.standard_normal(3),
                      act="relu")
    x = rng.standard_normal((2, 4))
    g = rng.standard_normal((2, 3))
    z = x @ layer.w + layer.b
    dx, gw, gb = mlp.layer_backward(layer, x, z, g)
    eps = 1e-6
    # numeric dL/dw[0,0] with L = sum(out * g)
    wp = layer.w.copy(); wp[0, 0] += eps
    wm = layer.w.copy(); wm[0, 0] -= eps
    lp = (mlp.layer_forward(mlp.Layer(wp, layer.b, "relu"), x) * g).sum()
    lm = (mlp.layer_forward(mlp.Layer(wm, layer.b, "relu"), x) * g).sum()
    assert abs(gw[0, 0] - (lp - lm) / (2 * eps)) < 1e-4
    assert dx.shape == x.shape and gb.shape == (3,)


def test_worker_layer_tasks():
    w = LegacyWorker()
    layer = mlp.random_mlp([3, 2], seed=5)[0]
    x = [[0.5, -1.0, 2.0]]
    out = w.execute_task(
        protocol.TASK_LAYER_FORWARD, {"layer": layer.to_json(), "x": x}
    )
    expect = mlp.layer_forward(layer, np.asarray(x))
    assert np.allclose(out["y"], expect)

    # train forward caches; backward consumes the cache
    out = w.execute_task(
        protocol.TASK_LAYER_FORWARD_TRAIN,
        {"layer": layer.to_json(), "x": x, "cache_id": "c1"},
    )
    back = w.execute_task(
        protocol.TASK_LAYER_BACKWARD,
        {"cache_id": "c1", "grad": [[1.0, 1.0]]},
    )
    assert np.asarray(back["dX"]).shape == (1, 3)
    assert np.asarray(back["gW"]).shape == (3, 2)


def test_worker_hf_tasks_via_native_engine():
    w = LegacyWorker()
    w.execute_task(protocol.TASK_HF_LOAD, {"model": "tiny"})
    try:
        res = w.execute_task(
            protocol.TASK_HF_INFER,
            {"model": "tiny", "prompt": "hi", "max_new_tokens": 3,
             "temperature": 0.0},
        )
        assert res["tokens"] == 3
    finally:
        w.execute_task(protocol.TASK_HF_UNLOAD, {"model": "tiny"})


def test_worker_unknown_kind():
    w = LegacyWorker()
    with pytest.raises(ValueError):
        w.execute_task("bogus", {})


def test_hf_part_load_forward_chain():
    """Two layer-range partials chained (text -> stage0 hidden -> stage1
    hidden) must equal the full-stack forward — the reference's DistilBERT
    partial workflow (bee2bee/node.py:236-277) on our native stack."""
    import numpy as np
    import torch

    from bee2bee_amd.engine.kv import PagedKV
    from bee2bee_amd.engine.runner import Runner
    from bee2bee_amd.legacy.worker import LegacyWorker
    from bee2bee_amd.models.spec import PRESETS
    from bee2bee_amd.models.tokenizer import load_tokenizer
    from bee2bee_amd.models.weights import ModelWeights

    w = LegacyWorker()
    r0 = w.execute_task("hf_part_load", {
        "model_name": "tiny", "start": 0, "end": 1, "seed": 5,
    })
    r1 = w.execute_task("hf_part_load", {
        "model_name": "tiny", "start": 1, "end": 2, "seed": 5,
    })
    text = "partial forward"
    h0 = w.execute_task("hf_part_forward", {
        "model_id": r0["model_id"], "text": text,
    })["hidden"]
    h1 = w.execute_task("hf_part_forward", {
        "model_id": r1["model_id"], "hidden": h0,
    })["hidden"]

    spec = PRESETS["tiny"]
    dev = torch.device("cpu")
    full = ModelWeights(spec, dev, torch.float32).random_init(5)
    kv = PagedKV(spec, dev, torch.float32, n_blocks=16)
    runner = Runner(spec, full, kv, dev, torch.float32)
    tok = load_tokenizer(None, spec.vocab_size, spec.bos_token_id,
                         spec.eos_token_id)
    ids = torch.tensor(tok.encode(text), dtype=torch.int64)
    T = ids.shape[0]
    kv.new_seq(0)
    kv.extend_seq(0, T)
    slots = torch.tensor(kv.slot_mapping(0, range(T)), dtype=torch.int32)
    pos = torch.arange(T, dtype=torch.int32)
    cu = torch.tensor([0, T], dtype=torch.int32)
    ref = runner.forward_prefill(ids, pos, slots, cu, T)
    got = np.array(h1, dtype=np.float32)
    assert np.allclose(got, ref.numpy(), atol=1e-4), np.abs(got - ref.numpy()).max()


def test_onnx_tasks_gated():
    """onnx_* tasks behave like the reference: a clean error when
    onnxruntime is absent, full function when present."""
    import pytest as _pytest

    from bee2bee_amd.legacy.worker import LegacyWorker

    w = LegacyWorker()
    try:
        import onnxruntime  # noqa: F401

        has_ort = True
    except Exception:
        has_ort = False
    if not has_ort:
        with _pytest.raises(RuntimeError, match="onnx_support_missing"):
            w.execute_task("onnx_load", {"path": "/nonexistent.onnx"})
    # unload of an unknown model is a no-op either way
    assert w.execute_task("onnx_unload", {"model_id": "nope"}) == {"ok": True}
