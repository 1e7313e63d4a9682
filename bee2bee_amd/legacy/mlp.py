"""Numpy MLP layer math (reference bee2bee/model.py:7-71): the payload
format of the coordinator-era layer tasks. Kept numerically identical so
stored layer JSON and remote layer tasks interoperate."""
from __future__ import annotations

import math
from dataclasses import dataclass, field
from typing import Any, Dict, List

import numpy as np


@dataclass
class Layer:
    w: np.ndarray  # [in, out]
    b: np.ndarray  # [out]
    act: str = "relu"  # relu | gelu | none

    def to_json(self) -> Dict[str, Any]:
        return {"w": self.w.tolist(), "b": self.b.tolist(), "act": self.act}

    @classmethod
    def from_json(cls, obj: Dict[str, Any]) -> "Layer":
        return cls(
            w=np.asarray(obj["w"], dtype=np.float64),
            b=np.asarray(obj["b"], dtype=np.float64),
            act=obj.get("act", "relu"),
        )


def relu(x: np.ndarray) -> np.ndarray:
    return np.maximum(x, 0.0)


def gelu(x: np.ndarray) -> np.ndarray:
    # tanh approximation (matches the reference's activation)
    return 0.5 * x * (1.0 + np.tanh(math.sqrt(2.0 / math.pi) * (x + 0.044715 * x**3)))


def layer_forward(layer: Layer, x: np.ndarray) -> np.ndarray:
    z = x @ layer.w + layer.b
    if layer.act == "relu":
        return relu(z)
    if layer.act == "gelu":
        return gelu(z)
    return z


def act_derivative(layer: Layer, z: np.ndarray) -> np.ndarray:
    if layer.act == "relu":
        return (z > 0).astype(z.dtype)
    if layer.act == "gelu":
        c = math.sqrt(2.0 / math.pi)
        t = np.tanh(c * (z + 0.044715 * z**3))
        return 0.5 * (1.0 + t) + 0.5 * z * (1.0 - t**2) * c * (1.0 + 3 * 0.044715 * z**2)
    return np.ones_like(z)


def layer_backward(layer: Layer, x: np.ndarray, z_or_out: np.ndarray,
                   grad_out: np.ndarray):
    """Returns (dX, gW, gb) for z = x @ w + b, out = act(z)."""
    dz = grad_out * act_derivative(layer, z_or_out)
    return dz @ layer.w.T, x.T @ dz, dz.sum(axis=0)


def random_mlp(sizes: List[int], act: str = "relu", seed: int = 0) -> List[Layer]:
    rng = np.random.default_rng(seed)
    layers = []
    for i in range(len(sizes) - 1):
        w = rng.standard_normal((sizes[i], sizes[i + 1])) * (1.0 / math.sqrt(sizes[i]))
        b = np.zeros(sizes[i + 1])
        layers.append(Layer(w=w, b=b, act=act if i < len(sizes) - 2 else "none"))
    return layers


def mlp_to_json(layers: List[Layer]) -> List[Dict[str, Any]]:
    return [l.to_json() for l in layers]


def mlp_from_json(objs: List[Dict[str, Any]]) -> List[Layer]:
    return [Layer.from_json(o) for o in objs]
