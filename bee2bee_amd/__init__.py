"""bee2bee_amd — an MI355X-native decentralized inference mesh.

A from-scratch framework with the capabilities of Bee2Bee (reference:
Chatit-cloud/BEE2BEE): a WebSocket peer mesh with DHT discovery, content
sharding, NAT traversal and a FastAPI gateway — whose per-node compute path
is a hand-written CDNA4 (gfx950) HIP inference engine (paged-KV attention,
fused RMSNorm/RoPE/SwiGLU, MFMA prefill, hipGraph-captured decode) with
RCCL-over-xGMI transport for multi-GPU parallelism.

Control plane:  bee2bee_amd.mesh   (asyncio WS mesh, wire-compatible with the
                reference protocol, see mesh/wire.py)
Gateway:        bee2bee_amd.gateway.api (FastAPI; /chat, /generate, /peers, ...)
Compute plane:  bee2bee_amd.engine + bee2bee_amd.ops (HIP/CDNA4 kernels)
Parallelism:    bee2bee_amd.parallel (RCCL pipeline/replica/expert parallel)
"""

from .version import __version__

# Reference-compatible public surface (reference: bee2bee/__init__.py:1-12).
from .mesh.node import MeshNode, run_mesh_node

# Aliases matching the reference package's exported names so downstream users
# of the reference can switch imports 1:1.
P2PNode = MeshNode
run_p2p_node = run_mesh_node

__all__ = [
    "__version__",
    "MeshNode",
    "run_mesh_node",
    "P2PNode",
    "run_p2p_node",
]
