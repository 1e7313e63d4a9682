"""Content-addressed sharding: split blobs into hash-verified pieces.

Behavior parity: reference bee2bee/pieces.py (split/hash/verify/reassemble,
`{hash}_{i:08d}.part` persistence) plus p2p.py's chunk/bitfield helpers —
the function names, part-file naming scheme and error strings are the
compatibility surface. On MI355X this is also the weight-shard distribution
primitive: a safetensors shard is split into pieces, announced to the DHT,
and fetched piece-wise by joining peers (see mesh/weightshare.py and
parallel/planner.py for the layer-shard mapping).
"""
from __future__ import annotations

import os
from typing import Iterator, List, Sequence

from ..utils import sha256_hex_bytes


def iter_pieces(data: bytes, piece_size: int) -> Iterator[bytes]:
    """Yield fixed-size pieces (last one may be short) without copying the
    whole blob twice — weight shards run to gigabytes."""
    if piece_size <= 0:
        raise ValueError("piece_size must be positive")
    view = memoryview(data)
    offset = 0
    while offset < len(view):
        yield bytes(view[offset : offset + piece_size])
        offset += piece_size


def split_pieces(data: bytes, piece_size: int) -> List[bytes]:
    return list(iter_pieces(data, piece_size))


# chunk_bytes is the same operation under the reference's other name
chunk_bytes = split_pieces


def piece_hashes(pieces: Sequence[bytes]) -> List[str]:
    return [sha256_hex_bytes(p) for p in pieces]


def verify_and_reassemble(pieces: Sequence[bytes], hashes: Sequence[str]) -> bytes:
    """Reassemble pieces after checking every hash; error strings are part
    of the wire contract ("length_mismatch", "hash_mismatch_at_<i>")."""
    if len(pieces) != len(hashes):
        raise ValueError("length_mismatch")
    bad = next(
        (i for i, (p, h) in enumerate(zip(pieces, hashes))
         if sha256_hex_bytes(p) != h),
        None,
    )
    if bad is not None:
        raise ValueError(f"hash_mismatch_at_{bad}")
    return b"".join(pieces)


def bitfield_from_pieces(total_pieces: int, have_indices: Sequence[int]) -> List[int]:
    have = {i for i in have_indices if 0 <= i < total_pieces}
    return [1 if i in have else 0 for i in range(total_pieces)]


def part_path(folder: str, content_hash: str, index: int) -> str:
    """`{hash}_{i:08d}.part` — the on-disk naming scheme shared with the
    reference so part caches interoperate."""
    return os.path.join(folder, f"{content_hash}_{index:08d}.part")


def save_pieces(folder: str, content_hash: str, pieces: Sequence[bytes]) -> List[str]:
    os.makedirs(folder, exist_ok=True)
    paths = [part_path(folder, content_hash, i) for i in range(len(pieces))]
    for path, piece in zip(paths, pieces):
        with open(path, "wb") as f:
            f.write(piece)
    return paths


def load_pieces(folder: str, content_hash: str) -> List[bytes]:
    """Load all persisted pieces for a content hash, in index order."""
    out: List[bytes] = []
    for i in range(1 << 31):
        path = part_path(folder, content_hash, i)
        if not os.path.exists(path):
            break
        with open(path, "rb") as f:
            out.append(f.read())
    return out
