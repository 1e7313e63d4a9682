"""Coordinator-era protocol constants (reference bee2bee/protocol.py:9-53).

Envelope: {"type": <MSG_*>, ...fields}; tasks: {"type": "task", "task_id",
"kind": <TASK_*>, "payload": {...}} answered by {"type": "result",
"task_id", "payload"} or {"type": "error", "task_id", "message"}.
Tensors travel as JSON lists of floats (the reason the live runtime moved
to RCCL — parallel/pp.py — but the wire shape is preserved here).
"""
from __future__ import annotations

from typing import Any, Dict

# envelope types
MSG_REGISTER = "register"
MSG_INFO = "info"
MSG_TASK = "task"
MSG_RESULT = "result"
MSG_ERROR = "error"

# numpy MLP layer tasks
TASK_LAYER_FORWARD = "layer_forward"
TASK_LAYER_FORWARD_TRAIN = "layer_forward_train"
TASK_LAYER_BACKWARD = "layer_backward"

# HF model tasks
TASK_HF_LOAD = "hf_load"
TASK_HF_UNLOAD = "hf_unload"
TASK_HF_INFER = "hf_infer"

# ONNX model tasks
TASK_ONNX_LOAD = "onnx_load"
TASK_ONNX_UNLOAD = "onnx_unload"
TASK_ONNX_INFER = "onnx_infer"

# partial-model (layer-range) tasks — the embryonic pipeline parallelism
TASK_HF_PART_LOAD = "hf_part_load"
TASK_HF_PART_FORWARD = "hf_part_forward"


def msg(mtype: str, **fields: Any) -> Dict[str, Any]:
    out = {"type": mtype}
    out.update(fields)
    return out
