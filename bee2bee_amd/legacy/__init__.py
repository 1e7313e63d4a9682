"""Legacy coordinator-era compatibility surface.

The reference ships a vestigial coordinator-protocol worker (bee2bee/node.py,
protocol.py, model.py — SURVEY.md §1 row Lx): a WS client that registers
with a coordinator and executes layer-level numpy-MLP tasks and HF
load/infer tasks, shipping tensors as JSON float lists. The coordinator
itself no longer exists in the reference repo, but the task surface is part
of its public API; this package keeps that surface available (same message
types, task kinds and payload shapes) so anything speaking the old protocol
still works against our nodes.
"""

from .protocol import msg  # noqa: F401
from .mlp import Layer, layer_forward, act_derivative, random_mlp  # noqa: F401
