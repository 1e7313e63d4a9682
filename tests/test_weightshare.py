"""Torrent-style weight distribution e2e: seed a real checkpoint on one
peer, fetch it over the wire on another, load it, and verify identical
model outputs."""
import asyncio

import torch

from bee2bee_amd.mesh.dht import DHTNode
from bee2bee_amd.mesh.node import MeshNode
from bee2bee_amd.mesh.weightshare import fetch_checkpoint, seed_checkpoint
from bee2bee_amd.models.spec import PRESETS
from bee2bee_amd.models.weights import ModelWeights, save_hf


def test_seed_fetch_load_roundtrip(tmp_path):
    spec = PRESETS["tiny"]
    src_dir = tmp_path / "src"
    dst_dir = tmp_path / "dst"
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(9)
    save_hf(w, str(src_dir))

    async def run():
        dht = DHTNode()
        await dht.start()
        seeder = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        leech = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await seeder.start()
        await leech.start()
        manifest = await seed_checkpoint(
            seeder, dht, "tiny-ckpt", str(src_dir), piece_size=4096
        )
        assert any(f["name"] == "model.safetensors" for f in manifest["files"])
        await leech.connect_bootstrap(seeder.addr)
        for _ in range(200):
            if seeder.peer_id in leech.peers:
                break
            await asyncio.sleep(0.02)
        await fetch_checkpoint(leech, dht, "tiny-ckpt", str(dst_dir))
        await leech.stop()
        await seeder.stop()

    asyncio.run(run())

    # the fetched checkpoint must load to identical weights
    w2 = ModelWeights(spec, torch.device("cpu"), torch.float32).load_hf(str(dst_dir))
    assert torch.equal(w.embed, w2.embed)
    for a, b in zip(w.layers, w2.layers):
        assert torch.equal(a.wqkv, b.wqkv)
        assert torch.equal(a.w_down, b.w_down)


def test_fetch_missing_manifest(tmp_path):
    import pytest

    async def run():
        dht = DHTNode()
        await dht.start()
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        with pytest.raises(FileNotFoundError):
            await fetch_checkpoint(node, dht, "nope", str(tmp_path / "x"))
        await node.stop()

    asyncio.run(run())


def test_fetch_over_mesh_dht_then_serve(tmp_path):
    """Full distribution chain with NO shared DHT object: the seeder
    publishes through its own node.dht (mesh-replicated dht_set), the
    leech discovers + fetches through ITS own store, and the fetched
    checkpoint serves through the engine with outputs identical to the
    source weights."""
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    spec = PRESETS["tiny"]
    src_dir = tmp_path / "src"
    dst_dir = tmp_path / "dst"
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(9)
    save_hf(w, str(src_dir))

    async def run():
        seeder = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        leech = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await seeder.start()
        await leech.start()
        try:
            await leech.connect_bootstrap(seeder.addr)
            for _ in range(200):
                if seeder.peer_id in leech.peers and leech.peer_id in seeder.peers:
                    break
                await asyncio.sleep(0.02)
            # publish AFTER the link is up (one-hop replication)
            await seed_checkpoint(seeder, seeder.dht, "tiny-ckpt2",
                                  str(src_dir), piece_size=4096)
            for _ in range(100):
                if await leech.dht.get("manifest:tiny-ckpt2"):
                    break
                await asyncio.sleep(0.02)
            await fetch_checkpoint(leech, leech.dht, "tiny-ckpt2", str(dst_dir))
        finally:
            await leech.stop()
            await seeder.stop()

    asyncio.run(run())

    def greedy(model_path=None, seed=9):
        eng = InferenceEngine("tiny", device="cpu", model_path=model_path,
                              max_batch=2, max_seq_len=64, seed=seed)
        try:
            r = GenerationRequest(prompt_ids=[4, 5, 6], max_new_tokens=5,
                                  sampling=SamplingParams(greedy=True))
            eng.submit(r)
            while True:
                x = r.out_queue.get(timeout=60)
                if not isinstance(x, int):
                    break
            assert r.error is None, r.error
            return r.output_ids
        finally:
            eng.shutdown()

    assert greedy(model_path=str(dst_dir)) == greedy(seed=9)


def test_manifest_path_traversal_rejected(tmp_path):
    """A peer-poisoned manifest with '../' or absolute names must never
    reach the filesystem."""
    import json

    import pytest

    from bee2bee_amd.mesh.weightshare import (
        MAX_MANIFEST_FILES,
        _safe_dest,
        _validate_manifest,
        manifest_key,
    )

    out = tmp_path / "out"
    out.mkdir()
    for bad in ("../evil", "/etc/passwd", "a/b", "..", "", "a\\b", "x\x00y"):
        with pytest.raises(ValueError):
            _safe_dest(str(out), bad)
    # a symlink inside out_dir pointing outside must also be rejected
    (out / "link").symlink_to(tmp_path)
    with pytest.raises(ValueError):
        _safe_dest(str(out), "link")
    assert _safe_dest(str(out), "model.safetensors").startswith(str(out))

    with pytest.raises(ValueError):
        _validate_manifest({"files": [{}] * (MAX_MANIFEST_FILES + 1)})
    with pytest.raises(ValueError):
        _validate_manifest({"files": [{"piece_hashes": [], "bytes": -1}]})
    with pytest.raises(ValueError):
        _validate_manifest({"files": "nope"})

    async def run():
        dht = DHTNode()
        await dht.start()
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        evil = {"name": "x", "files": [{
            "name": "../../pwned", "bytes": 4,
            "content_hash": "00" * 32, "piece_hashes": ["00" * 32],
            "piece_size": 4,
        }]}
        await dht.set(manifest_key("x"), json.dumps(evil))
        try:
            with pytest.raises(ValueError):
                await fetch_checkpoint(node, dht, "x", str(tmp_path / "fetch"))
        finally:
            await node.stop()

    asyncio.run(run())
    assert not (tmp_path.parent / "pwned").exists()
    assert not (tmp_path / "pwned").exists()


def test_fetch_resumes_from_part_cache(tmp_path):
    """Interrupted fetch: verified pieces persisted under .parts/ are not
    re-requested on the retry (torrent resume); corrupt cached parts are
    re-fetched."""
    spec = PRESETS["tiny"]
    src_dir = tmp_path / "src"
    dst_dir = tmp_path / "dst"
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(3)
    save_hf(w, str(src_dir))

    async def run():
        dht = DHTNode()
        await dht.start()
        seeder = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        leech = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await seeder.start()
        await leech.start()
        await seed_checkpoint(seeder, dht, "ck", str(src_dir),
                              piece_size=2048)
        await leech.connect_bootstrap(seeder.addr)
        for _ in range(200):
            if seeder.peer_id in leech.peers:
                break
            await asyncio.sleep(0.02)

        requested = []
        real = leech.request_piece

        async def counting(pid, chash, i):
            requested.append((chash[:8], i))
            return await real(pid, chash, i)

        leech.request_piece = counting

        # interrupted first fetch: fail after 3 pieces
        calls = {"n": 0}

        async def failing(pid, chash, i):
            if calls["n"] >= 3:
                raise RuntimeError("simulated network drop")
            calls["n"] += 1
            return await counting(pid, chash, i)

        leech.request_piece = failing
        try:
            await fetch_checkpoint(leech, dht, "ck", str(dst_dir),
                                   keep_parts=True)
            raise AssertionError("fetch should have failed")
        except RuntimeError:
            pass
        n_first = len(requested)
        assert n_first == 3

        # corrupt ONE cached part: it must be re-fetched and replaced
        import os as _os

        from bee2bee_amd.mesh.pieces import part_path

        parts_dir = str(dst_dir / ".parts")
        cached = sorted(_os.listdir(parts_dir))
        assert len(cached) == 3
        victim = _os.path.join(parts_dir, cached[0])
        with open(victim, "wb") as f:
            f.write(b"garbage")

        # resumed fetch completes; the 2 intact parts are NOT re-requested
        leech.request_piece = counting
        await fetch_checkpoint(leech, dht, "ck", str(dst_dir))
        refetched = set(requested[n_first:])
        assert refetched  # the corrupted piece + the never-fetched rest

        def part_key(fname):
            stem = fname[:-len(".part")]
            h, idx = stem.rsplit("_", 1)
            return (h[:8], int(idx))

        intact_keys = {part_key(c) for c in cached[1:]}
        assert part_key(cached[0]) in refetched  # corrupt part re-fetched
        assert not (intact_keys & refetched)  # intact parts NOT re-fetched

        await leech.stop()
        await seeder.stop()

        # fetched checkpoint is byte-identical
        got = (dst_dir / "model.safetensors").read_bytes()
        want = (src_dir / "model.safetensors").read_bytes()
        assert got == want
        assert not (dst_dir / ".parts").exists()  # cleaned after success

    asyncio.run(run())


def test_fetch_stripes_and_fails_over_providers(tmp_path):
    """Two seeders: pieces stripe across both; killing one mid-catalog still
    completes via failover."""
    spec = PRESETS["tiny"]
    src_dir = tmp_path / "src"
    dst_dir = tmp_path / "dst"
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(5)
    save_hf(w, str(src_dir))

    async def run():
        dht = DHTNode()
        await dht.start()
        s1 = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        s2 = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        leech = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        for n in (s1, s2, leech):
            await n.start()
        await seed_checkpoint(s1, dht, "ck2", str(src_dir), piece_size=1024)
        await seed_checkpoint(s2, dht, "ck2", str(src_dir), piece_size=1024)
        for s in (s1, s2):
            await leech.connect_bootstrap(s.addr)
        for _ in range(200):
            if s1.peer_id in leech.peers and s2.peer_id in leech.peers:
                break
            await asyncio.sleep(0.02)

        served = {s1.peer_id: 0, s2.peer_id: 0}
        real = leech.request_piece

        async def counting(pid, chash, i):
            served[pid] += 1
            return await real(pid, chash, i)

        leech.request_piece = counting
        await fetch_checkpoint(leech, dht, "ck2", str(dst_dir))
        assert served[s1.peer_id] > 0 and served[s2.peer_id] > 0, served

        # failover: one seeder dies; a fresh fetch still completes
        import shutil

        shutil.rmtree(dst_dir)
        await s1.stop()
        for _ in range(200):
            if s1.peer_id not in leech.peers:
                break
            await asyncio.sleep(0.02)
        await fetch_checkpoint(leech, dht, "ck2", str(dst_dir))
        got = (dst_dir / "model.safetensors").read_bytes()
        assert got == (src_dir / "model.safetensors").read_bytes()

        await leech.stop()
        await s2.stop()

    asyncio.run(run())


def test_seeding_is_disk_backed(tmp_path):
    """seed_checkpoint must register files WITHOUT holding payload bytes in
    RAM (multi-GB shards): the node's piece table stores a path + geometry,
    and serving reads slices on demand."""
    spec = PRESETS["tiny"]
    src_dir = tmp_path / "src"
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(1)
    save_hf(w, str(src_dir))

    async def run():
        dht = DHTNode()
        await dht.start()
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        manifest = await seed_checkpoint(node, dht, "dk", str(src_dir),
                                         piece_size=4096)
        for info in node.pieces.values():
            assert "pieces" not in info, "payload held in memory"
            assert "path" in info and info["n"] >= 1
        # bytes in the manifest match the real file sizes
        import os as _os

        for entry in manifest["files"]:
            real = _os.path.getsize(_os.path.join(str(src_dir),
                                                  entry["name"]))
            assert entry["bytes"] == real
        await node.stop()

    asyncio.run(run())


def test_oversize_piece_size_rejected(tmp_path):
    """piece_size beyond the wire-frame budget is refused up front (a
    32 MiB+ base64 piece_data frame would be dropped by the WS layer and
    look like a hung fetch)."""
    import pytest

    async def run():
        dht = DHTNode()
        await dht.start()
        node = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await node.start()
        try:
            with pytest.raises(ValueError, match="frame budget"):
                await seed_checkpoint(node, dht, "big", str(tmp_path),
                                      piece_size=32 * 1024 * 1024)
        finally:
            await node.stop()

    asyncio.run(run())


def test_fetched_checkpoint_serves_identically(tmp_path):
    """The full torrent->serve loop: seed a checkpoint, fetch it over the
    mesh on another peer, and the ENGINE serving the fetched copy emits
    exactly the tokens the source copy emits."""
    spec = PRESETS["tiny"]
    src_dir = tmp_path / "src"
    dst_dir = tmp_path / "dst"
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(21)
    save_hf(w, str(src_dir))

    async def run():
        dht = DHTNode()
        await dht.start()
        seeder = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        leech = MeshNode(host="127.0.0.1", port=0, enable_nat=False)
        await seeder.start()
        await leech.start()
        await seed_checkpoint(seeder, dht, "serve-ck", str(src_dir),
                              piece_size=4096)
        await leech.connect_bootstrap(seeder.addr)
        for _ in range(200):
            if seeder.peer_id in leech.peers:
                break
            await asyncio.sleep(0.02)
        await fetch_checkpoint(leech, dht, "serve-ck", str(dst_dir))
        await leech.stop()
        await seeder.stop()

    asyncio.run(run())

    from bee2bee_amd.engine.engine import InferenceEngine

    def toks(path):
        eng = InferenceEngine("tiny", device="cpu", model_path=path,
                              max_batch=2, max_seq_len=128, seed=4)
        try:
            req = eng.generate([7, 8, 9], max_new_tokens=8, temperature=0.0,
                               repetition_penalty=1.0)
            return list(req.output_ids)
        finally:
            eng.shutdown()

    assert toks(str(src_dir)) == toks(str(dst_dir))
