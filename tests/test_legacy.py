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
