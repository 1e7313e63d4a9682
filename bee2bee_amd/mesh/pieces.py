"""Content-addressed sharding: split blobs into hash-verified pieces.

Parity: reference bee2bee/pieces.py (split :7, hashes :11, verify+reassemble
:15-21, `{hash}_{i:08d}.part` persistence :24-32) plus helpers from p2p.py
(chunk_bytes :43, bitfield :47). On MI355X this is also the weight-shard
distribution primitive: a safetensors shard is split into pieces, announced
to the DHT, and fetched piece-wise by joining peers (see mesh/dht.py and
parallel/planner.py for the layer-shard mapping).
"""
from __future__ import annotations

import os
from typing import List

from ..utils import sha256_hex_bytes


def split_pieces(data: bytes, piece_size: int) -> List[bytes]:
    if piece_size <= 0:
        raise ValueError("piece_size must be positive")
    return [data[i : i + piece_size] for i in range(0, len(data), piece_size)]


# chunk_bytes is the same operation under the reference's other name
chunk_bytes = split_pieces


def piece_hashes(pieces: List[bytes]) -> List[str]:
    return [sha256_hex_bytes(p) for p in pieces]


def verify_and_reassemble(pieces: List[bytes], hashes: List[str]) -> bytes:
    if len(pieces) != len(hashes):
        raise ValueError("length_mismatch")
    for i, p in enumerate(pieces):
        if sha256_hex_bytes(p) != hashes[i]:
            raise ValueError(f"hash_mismatch_at_{i}")
    return b"".join(pieces)


def bitfield_from_pieces(total_pieces: int, have_indices: List[int]) -> List[int]:
    field = [0] * total_pieces
    for i in have_indices:
        if 0 <= i < total_pieces:
            field[i] = 1
    return field


def save_pieces(folder: str, content_hash: str, pieces: List[bytes]) -> List[str]:
    os.makedirs(folder, exist_ok=True)
    paths = []
    for i, p in enumerate(pieces):
        path = os.path.join(folder, f"{content_hash}_{i:08d}.part")
        with open(path, "wb") as f:
            f.write(p)
        paths.append(path)
    return paths


def load_pieces(folder: str, content_hash: str) -> List[bytes]:
    """Load all persisted pieces for a content hash, in index order."""
    out = []
    i = 0
    while True:
        path = os.path.join(folder, f"{content_hash}_{i:08d}.part")
        if not os.path.exists(path):
            break
        with open(path, "rb") as f:
            out.append(f.read())
        i += 1
    return out
