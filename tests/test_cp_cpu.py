"""Context parallelism on CPU: per-rank partial attention + flash merge
must equal full-context attention exactly (fp32 reference ops); gloo
world 2 for the collective path."""
import multiprocessing as mp
import pickle

import pytest
import torch

from bee2bee_amd.ops import reference as R
from bee2bee_amd.parallel.cp import (
    local_lens,
    merge_partials,
    shard_pages,
)

B, NKV, G, HD, PAGE = 3, 2, 2, 64, 16
NQ = NKV * G
SCALE = HD ** -0.5


def _ctx(seed=7, n_pages_total=64):
    g = torch.Generator().manual_seed(seed)
    q = torch.randn(B, NQ, HD, generator=g)
    kc = torch.randn(n_pages_total, NKV, PAGE, HD, generator=g)
    vc = torch.randn(n_pages_total, NKV, PAGE, HD, generator=g)
    # ragged contexts, incl. one shorter than a page-shard
    seq_lens = torch.tensor([250, 37, 129], dtype=torch.int32)
    W = -(-int(seq_lens.max()) // PAGE)
    bt = torch.arange(B * W, dtype=torch.int32).reshape(B, W)
    return q, kc, vc, bt, seq_lens


def test_shard_pages_partition():
    for n in (0, 1, 5, 16, 17):
        for world in (1, 2, 3, 8):
            spans = [shard_pages(n, world, r) for r in range(world)]
            assert spans[0][0] == 0 and spans[-1][1] == n
            for (a, b), (c, d) in zip(spans, spans[1:]):
                assert b == c and a <= b and c <= d


def _rank_partial(q, kc, vc, bt, seq_lens, world, rank):
    """Build rank-local table/lens and run the reference lse attention."""
    lls = local_lens(seq_lens, PAGE, world, rank)
    W = bt.shape[1]
    lbt = torch.zeros_like(bt)
    for i, L in enumerate(seq_lens.tolist()):
        n_pages = -(-L // PAGE)
        lo, hi = shard_pages(n_pages, world, rank)
        lbt[i, : hi - lo] = bt[i, lo:hi]
    out, ml = R.attn_decode_lse(q, kc, vc, lbt, lls, SCALE)
    return out, ml, lls


def test_cp_merge_equals_full_context():
    q, kc, vc, bt, seq_lens = _ctx()
    full = R.attn_decode(q, kc, vc, bt, seq_lens, SCALE)
    for world in (1, 2, 3):
        outs, mls = [], []
        for r in range(world):
            out, ml, lls = _rank_partial(q, kc, vc, bt, seq_lens, world, r)
            if bool((lls == 0).any()):
                ml = ml.clone()
                ml[lls == 0, :, 0] = -1e30
                ml[lls == 0, :, 1] = 0.0
            outs.append(out)
            mls.append(ml)
        merged = merge_partials(outs, mls)
        assert torch.allclose(merged.float(), full.float(), atol=1e-5), (
            world, (merged.float() - full.float()).abs().max())


def test_lse_values_are_true_softmax_stats():
    q, kc, vc, bt, seq_lens = _ctx()
    _out, ml = R.attn_decode_lse(q, kc, vc, bt, seq_lens, SCALE)
    # softmax denominator reconstructed from (m, l) must match a direct
    # computation for sequence 0, head 0
    L = int(seq_lens[0])
    keys = kc[bt[0, : -(-L // PAGE)].long()]
    keys = keys.permute(1, 0, 2, 3).reshape(NKV, -1, HD)[:, :L]
    s = (q[0].float().view(NKV, G, HD) @ keys.transpose(1, 2)) * SCALE
    assert torch.allclose(ml[0, 0, 0], s[0, 0].max(), atol=1e-5)
    assert torch.allclose(
        ml[0, 0, 1], torch.exp(s[0, 0] - s[0, 0].max()).sum(), atol=1e-4)


def _cp_worker(rank, world, port, out_path):
    import torch.distributed as dist

    from bee2bee_amd.parallel.cp import cp_attn_decode

    dist.init_process_group(
        backend="gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world,
    )
    try:
        q, kc, vc, bt, seq_lens = _ctx()
        lls = local_lens(seq_lens, PAGE, world, rank)
        lbt = torch.zeros_like(bt)
        for i, L in enumerate(seq_lens.tolist()):
            n_pages = -(-L // PAGE)
            lo, hi = shard_pages(n_pages, world, rank)
            lbt[i, : hi - lo] = bt[i, lo:hi]
        out = cp_attn_decode(q, kc, vc, lbt, lls, SCALE)
        if rank == 0:
            with open(out_path, "wb") as f:
                pickle.dump(out, f)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_cp2_gloo_matches_single(tmp_path):
    q, kc, vc, bt, seq_lens = _ctx()
    full = R.attn_decode(q, kc, vc, bt, seq_lens, SCALE)
    out_path = str(tmp_path / "cp.pkl")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_cp_worker, args=(r, 2, 29871, out_path))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=200)
        assert p.exitcode == 0
    with open(out_path, "rb") as f:
        got = pickle.load(f)
    assert torch.allclose(got.float(), full.float(), atol=1e-5)


def _cpengine_worker(rank, world, port, out_path):
    import torch.distributed as dist

    from bee2bee_amd.parallel.cp import CPEngine

    dist.init_process_group(
        backend="gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world,
    )
    try:
        eng = CPEngine("tiny", device="cpu", max_batch=2, max_seq_len=512,
                       seed=21)
        g = torch.Generator().manual_seed(13)
        prompts = [torch.randint(4, 500, (300,), generator=g).tolist(),
                   torch.randint(4, 500, (280,), generator=g).tolist()]
        outs = eng.generate(prompts, 5)
        if rank == 0:
            with open(out_path, "wb") as f:
                pickle.dump((prompts, outs), f)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_cpengine2_matches_single(tmp_path):
    """CPEngine on gloo world 2 (each rank holding half the 256-token
    pages) must emit the same greedy tokens as the plain single-process
    engine on the same weights."""
    out_path = str(tmp_path / "cpe.pkl")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_cpengine_worker, args=(r, 2, 29873, out_path))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=260)
        assert p.exitcode == 0
    with open(out_path, "rb") as f:
        prompts, cp_outs = pickle.load(f)

    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=512,
                          seed=21)
    try:
        ref = []
        for prompt in prompts:
            req = GenerationRequest(prompt_ids=list(prompt), max_new_tokens=5,
                                    sampling=SamplingParams(greedy=True))
            eng.submit(req)
            while True:
                item = req.out_queue.get(timeout=120)
                if not isinstance(item, int):
                    break
            assert req.error is None, req.error
            ref.append(req.output_ids)
    finally:
        eng.shutdown()
    assert cp_outs == ref, (cp_outs, ref)


def _cpengine_short_worker(rank, world, port, out_path):
    import torch.distributed as dist

    from bee2bee_amd.parallel.cp import CPEngine

    dist.init_process_group(
        backend="gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world,
    )
    try:
        eng = CPEngine("tiny", device="cpu", max_batch=2, max_seq_len=512,
                       seed=23)
        prompts = [[7, 8, 9, 10, 11]]  # < one page: rank 1 owns NOTHING
        outs = eng.generate(prompts, 4)
        if rank == 0:
            with open(out_path, "wb") as f:
                pickle.dump(outs, f)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_cpengine_short_prompt_empty_shard(tmp_path):
    """A prompt shorter than one 256-token page leaves rank 1 with an
    EMPTY KV shard; the merge must still be exact."""
    out_path = str(tmp_path / "cps.pkl")
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_cpengine_short_worker,
                         args=(r, 2, 29877, out_path)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=200)
        assert p.exitcode == 0
    with open(out_path, "rb") as f:
        cp_outs = pickle.load(f)

    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine("tiny", device="cpu", max_batch=2, max_seq_len=512,
                          seed=23)
    try:
        req = GenerationRequest(prompt_ids=[7, 8, 9, 10, 11],
                                max_new_tokens=4,
                                sampling=SamplingParams(greedy=True))
        eng.submit(req)
        while True:
            item = req.out_queue.get(timeout=120)
            if not isinstance(item, int):
                break
        assert req.error is None, req.error
        ref = req.output_ids
    finally:
        eng.shutdown()
    assert cp_outs == [ref], (cp_outs, ref)


def test_cp_page_ownership_partitions_exactly():
    """Pure page-math invariants of CPEngine (no process group needed):
    for any (context length, world size), the owned positions partition
    [0, L) exactly across ranks, local positions are injective per rank,
    and the per-rank local lengths sum to L."""
    from bee2bee_amd.parallel.cp import CPEngine

    page = 256
    for world in (1, 2, 3, 5, 8):
        for L in (1, page - 1, page, page + 1, 3 * page + 17, 8 * page):
            ranks = []
            for r in range(world):
                eng = CPEngine.__new__(CPEngine)  # math only, no dist/init
                eng.rank, eng.world, eng.page = r, world, page
                ranks.append(eng)
            owners = [sum(1 for e in ranks if e._owned(p))
                      for p in range(L)]
            assert owners == [1] * L, (world, L)
            total = 0
            for e in ranks:
                mine = [p for p in range(L) if e._owned(p)]
                locs = [e._local_pos(p) for p in mine]
                assert len(set(locs)) == len(locs), "local slot collision"
                assert e._local_len(L) == len(mine), (world, L, e.rank)
                total += len(mine)
            assert total == L
