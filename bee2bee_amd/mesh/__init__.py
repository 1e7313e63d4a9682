"""Peer mesh: wire protocol, WS node, DHT, pieces, NAT traversal, registry."""

from .node import MeshNode, run_mesh_node  # noqa: F401
from .links import generate_join_link, parse_join_link  # noqa: F401
