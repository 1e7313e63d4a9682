"""Torrent-style model-weight distribution over the mesh.

The reference's tagline capability ("torrent-like LLM deployment",
README.md.backup:4) existed as disconnected primitives — pieces.py content
shards, dht.py piece announcements, and STUBBED piece transfer handlers
(p2p_runtime.py:675-683). This module completes the loop:

  seed_checkpoint(node, dht, dir)  -> split every file into hash-verified
                                      pieces, serve them, announce to DHT
  fetch_checkpoint(node, dht, ...) -> discover providers, pull pieces over
                                      the wire protocol, verify hashes,
                                      reassemble the checkpoint dir

A fetched checkpoint is a normal HF-format dir that NativeEngineService /
ModelWeights.load_hf consumes. Piece size defaults to 8 MiB (frames are
base64 inside the 32 MiB WS limit).
"""
from __future__ import annotations

import asyncio
import json
import os
from typing import Dict, List, Optional

from ..utils import sha256_hex_bytes
from . import dht as dht_mod
from .pieces import part_path, verify_and_reassemble

DEFAULT_PIECE = 8 * 1024 * 1024

# Hard bounds on peer-supplied manifests. A manifest arrives over the
# replicated DHT (any connected peer can write it), so every field in it is
# attacker-controlled: file names, counts and sizes must all be validated
# before they touch the filesystem.
MAX_MANIFEST_FILES = 4096
MAX_FILE_BYTES = 512 * 1024 * 1024 * 1024  # 512 GiB — a 70B bf16 shard set fits
MAX_PIECES_PER_FILE = 1 << 20


def manifest_key(name: str) -> str:
    return f"manifest:{name}"


def _safe_dest(out_dir: str, name: str) -> str:
    """Resolve a peer-supplied file name to a path strictly inside out_dir.

    Names come from an untrusted DHT manifest; anything that is not a plain
    file name (path separators, '..', absolute paths, empty) is rejected, and
    the resolved real path is checked to stay under out_dir even in the
    presence of symlinks."""
    if (
        not name
        or name != os.path.basename(name)
        or name in (".", "..")
        or "/" in name
        or "\\" in name
        or "\x00" in name
    ):
        raise ValueError(f"unsafe file name in manifest: {name!r}")
    dest = os.path.join(out_dir, name)
    root = os.path.realpath(out_dir)
    resolved = os.path.realpath(dest)
    if resolved != root and not resolved.startswith(root + os.sep):
        raise ValueError(f"manifest file escapes output dir: {name!r}")
    return dest


def _validate_manifest(manifest) -> None:
    """Structural + bounds validation of an untrusted manifest."""
    if not isinstance(manifest, dict) or not isinstance(manifest.get("files"), list):
        raise ValueError("malformed manifest")
    files = manifest["files"]
    if len(files) > MAX_MANIFEST_FILES:
        raise ValueError(f"manifest lists {len(files)} files (max {MAX_MANIFEST_FILES})")
    for entry in files:
        if not isinstance(entry, dict):
            raise ValueError("malformed manifest entry")
        hashes = entry.get("piece_hashes")
        if not isinstance(hashes, list) or len(hashes) > MAX_PIECES_PER_FILE:
            raise ValueError("manifest entry piece list missing or oversized")
        nbytes = entry.get("bytes")
        if not isinstance(nbytes, int) or nbytes < 0 or nbytes > MAX_FILE_BYTES:
            raise ValueError("manifest entry byte count out of bounds")


async def seed_checkpoint(
    node,
    dht: dht_mod.DHTNode,
    name: str,
    directory: str,
    piece_size: int = DEFAULT_PIECE,
) -> Dict:
    """Split every file of a checkpoint dir into pieces, register them with
    the node for serving, and publish a manifest + provider records."""
    # piece_data frames carry base64 (4/3 overhead) inside the 32 MiB WS
    # limit; leave headroom for the JSON envelope
    if piece_size > 20 * 1024 * 1024:
        raise ValueError(
            f"piece_size {piece_size} exceeds the wire frame budget "
            "(max 20 MiB: base64 must fit the 32 MiB WS limit)")
    import hashlib

    files = []
    for fname in sorted(os.listdir(directory)):
        path = os.path.join(directory, fname)
        if not os.path.isfile(path):
            continue
        # stream: per-piece + whole-file hashes in one pass, never holding
        # the file in memory (safetensors shards run to tens of GB)
        hashes: List[str] = []
        whole = hashlib.sha256()
        total = 0
        with open(path, "rb") as f:
            while True:
                chunk = f.read(piece_size)
                if not chunk:
                    break
                hashes.append(sha256_hex_bytes(chunk))
                whole.update(chunk)
                total += len(chunk)
        if not hashes:
            hashes = [sha256_hex_bytes(b"")]  # empty file = one empty piece
        content_hash = whole.hexdigest()
        node.share_file(content_hash, path, piece_size, len(hashes))
        await _publish(node, dht, f"piece:{content_hash}",
                       lambda cur: sorted(set((cur or []) + [node.addr])))
        files.append(
            {
                "name": fname,
                "bytes": total,
                "content_hash": content_hash,
                "piece_hashes": hashes,
                "piece_size": piece_size,
            }
        )
    manifest = {"name": name, "files": files}
    await _publish(node, dht, manifest_key(name),
                   lambda cur: json.dumps(manifest))
    return manifest


async def _publish(node, dht, key: str, update) -> None:
    """Write a DHT record; when publishing through the node's OWN store,
    use the mesh-replicated write (node.dht_set) so connected peers can
    fetch without a shared/kademlia DHT."""
    cur = await dht.get(key)
    value = update(cur)
    if getattr(node, "dht", None) is dht and hasattr(node, "dht_set"):
        await node.dht_set(key, value)
    else:
        await dht.set(key, value)


def _drop_parts(parts_dir: str, content_hash: str, n: int) -> None:
    for i in range(n):
        try:
            os.remove(part_path(parts_dir, content_hash, i))
        except OSError:
            pass


async def fetch_checkpoint(
    node,
    dht: dht_mod.DHTNode,
    name: str,
    out_dir: str,
    provider_peer_id: Optional[str] = None,
    keep_parts: bool = False,
) -> str:
    """Pull a seeded checkpoint from the mesh into out_dir (hash-verified).

    Torrent semantics:
      * pieces stripe across EVERY connected provider of a file's content
        hash, with per-piece failover to the next provider on error or a
        hash-mismatched (corrupt) piece;
      * each verified piece persists to `<out_dir>/.parts/` in the
        reference's `{hash}_{i:08d}.part` naming (bee2bee/pieces.py:24-32),
        so an interrupted fetch RESUMES — already-verified pieces are never
        re-requested — and part caches interoperate with reference peers;
      * files already complete in out_dir (content hash matches) are
        skipped entirely.
    The caller can pin one provider peer id instead (must be connected)."""
    raw = await dht.get(manifest_key(name))
    if not raw:
        raise FileNotFoundError(f"no manifest for '{name}' in DHT")
    manifest = json.loads(raw)
    _validate_manifest(manifest)
    os.makedirs(out_dir, exist_ok=True)
    parts_dir = os.path.join(out_dir, ".parts")

    for entry in manifest["files"]:
        dest = _safe_dest(out_dir, entry["name"])
        chash = entry["content_hash"]
        want_hashes = entry["piece_hashes"]
        n = len(want_hashes)

        # complete-file short-circuit (idempotent re-fetch)
        if os.path.isfile(dest):
            with open(dest, "rb") as f:
                if sha256_hex_bytes(f.read()) == chash:
                    if not keep_parts:
                        _drop_parts(parts_dir, chash, n)
                    continue

        if provider_peer_id is not None:
            pids = [provider_peer_id]
        else:
            providers = await dht_mod.find_providers(dht, chash)
            pids = [p for p, peer in node.peers.items()
                    if peer.addr in providers]
        if not pids:
            raise RuntimeError(
                f"no connected provider for {entry['name']} ({chash[:12]})"
            )

        # resume: verified pieces from the part cache
        pieces: List[Optional[bytes]] = [None] * n
        os.makedirs(parts_dir, exist_ok=True)
        for i in range(n):
            ppath = part_path(parts_dir, chash, i)
            if os.path.isfile(ppath):
                with open(ppath, "rb") as f:
                    blob = f.read()
                if sha256_hex_bytes(blob) == want_hashes[i]:
                    pieces[i] = blob

        sem = asyncio.Semaphore(min(8, 2 * len(pids)))

        async def fetch_piece(i: int) -> None:
            async with sem:
                last_err: Optional[Exception] = None
                for k in range(len(pids)):
                    pid = pids[(i + k) % len(pids)]  # stripe + failover
                    try:
                        cand = await node.request_piece(pid, chash, i)
                    except Exception as e:  # noqa: BLE001 — next provider
                        last_err = e
                        continue
                    if sha256_hex_bytes(cand) != want_hashes[i]:
                        last_err = ValueError(
                            f"corrupt piece {i} from {pid[:12]}")
                        continue
                    pieces[i] = cand
                    with open(part_path(parts_dir, chash, i), "wb") as f:
                        f.write(cand)
                    return
                raise RuntimeError(
                    f"piece {i} of {entry['name']} unavailable: {last_err}")

        missing = [i for i in range(n) if pieces[i] is None]
        # up to 8 pieces in flight across the provider set
        results = await asyncio.gather(
            *(fetch_piece(i) for i in missing), return_exceptions=True)
        for r in results:
            if isinstance(r, BaseException):
                raise r

        data = verify_and_reassemble(pieces, want_hashes)
        if sha256_hex_bytes(data) != chash:
            raise ValueError(f"content hash mismatch for {entry['name']}")
        if len(data) != entry["bytes"]:
            raise ValueError(f"size mismatch for {entry['name']}")
        with open(dest, "wb") as f:
            f.write(data)
        if not keep_parts:
            _drop_parts(parts_dir, chash, n)

    if not keep_parts and os.path.isdir(parts_dir) and not os.listdir(parts_dir):
        os.rmdir(parts_dir)
    return out_dir
