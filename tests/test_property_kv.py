"""Property-based tests for the PagedKV block allocator and DHT announce
semantics: random alloc/extend/free interleavings must conserve the block
pool exactly (no leak, no double-hand-out, no cross-sequence slot overlap),
and piece/rank announcements must be idempotent and order-insensitive.

The allocator invariants back the engine's continuous batching — the soak
tests check end-state equality; these check EVERY intermediate state on
randomized schedules.
"""
import asyncio

import pytest
import torch

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings
from hypothesis import strategies as st

from bee2bee_amd.engine.kv import PagedKV
from bee2bee_amd.models.spec import PRESETS

N_BLOCKS = 12
BLOCK = 8


def make_kv():
    return PagedKV(PRESETS["tiny"], torch.device("cpu"), torch.float32,
                   n_blocks=N_BLOCKS, block_size=BLOCK)


# ops: (kind, seq_id, amount) — schedules drawn over a small id space so
# new/free/extend collide and re-use ids aggressively
ops = st.lists(
    st.tuples(st.sampled_from(["new", "extend", "free"]),
              st.integers(min_value=0, max_value=4),
              st.integers(min_value=1, max_value=3 * BLOCK)),
    max_size=60,
)


class TestPagedKVAllocator:
    @given(schedule=ops)
    @settings(max_examples=200, deadline=None)
    def test_pool_conservation_under_random_schedule(self, schedule):
        kv = make_kv()
        live = set()
        for kind, sid, amount in schedule:
            if kind == "new":
                kv.new_seq(sid)
                live.add(sid)
            elif kind == "free":
                kv.free_seq(sid)
                live.discard(sid)
            elif kind == "extend" and sid in live:
                new_len = kv.seq_len(sid) + amount
                need = (new_len + BLOCK - 1) // BLOCK - kv.seq_n_blocks(sid)
                if need > kv.free_blocks:
                    with pytest.raises(RuntimeError, match="KV pool exhausted"):
                        kv.extend_seq(sid, new_len)
                    continue
                kv.extend_seq(sid, new_len)

            # invariant 1: every block is exactly one of {free, owned-once}
            owned = [b for s in kv._seq_blocks.values() for b in s]
            assert len(owned) == len(set(owned)), "block handed out twice"
            assert set(owned).isdisjoint(kv._free), "owned block also free"
            assert len(owned) + kv.free_blocks == N_BLOCKS, "pool leak"

            # invariant 2: block count always matches the sequence length
            for s in live:
                assert kv.seq_n_blocks(s) == (kv.seq_len(s) + BLOCK - 1) // BLOCK

        # invariant 3: freeing everything restores the full pool
        for s in list(live):
            kv.free_seq(s)
        assert kv.free_blocks == N_BLOCKS

    @given(lens=st.lists(st.integers(min_value=1, max_value=2 * BLOCK),
                         min_size=1, max_size=4))
    @settings(max_examples=100, deadline=None)
    def test_slot_mappings_disjoint_across_sequences(self, lens):
        kv = make_kv()
        slots = {}
        for sid, ln in enumerate(lens):
            kv.new_seq(sid)
            kv.extend_seq(sid, ln)
            s = kv.slot_mapping(sid, range(ln))
            assert len(s) == len(set(s)), "duplicate slot within a sequence"
            slots[sid] = set(s)
        seen = set()
        for sid, s in slots.items():
            assert seen.isdisjoint(s), "slot shared across sequences"
            seen |= s
        # every slot is inside the pool
        assert all(0 <= x < N_BLOCKS * BLOCK for x in seen)

    @given(ln=st.integers(min_value=1, max_value=3 * BLOCK))
    @settings(max_examples=50, deadline=None)
    def test_block_table_covers_slot_mapping(self, ln):
        """slot // block_size must be exactly the sequence's block table —
        the contract between host allocation and the decode kernel's
        device-side page walk."""
        kv = make_kv()
        kv.new_seq(7)
        kv.extend_seq(7, ln)
        slots = kv.slot_mapping(7, range(ln))
        table = kv.block_table([7])[0].tolist()
        n = kv.seq_n_blocks(7)
        for p, s in enumerate(slots):
            assert s // BLOCK == table[p // BLOCK]
            assert s % BLOCK == p % BLOCK
        assert table[:n] == kv._seq_blocks[7]


class TestDHTAnnounceProperties:
    @given(addrs=st.lists(st.sampled_from(["a", "b", "c", "d"]),
                          min_size=1, max_size=12))
    @settings(max_examples=50, deadline=None)
    def test_announce_idempotent_and_order_insensitive(self, addrs):
        from bee2bee_amd.mesh.dht import DHTNode, announce_piece, find_providers

        async def run(seq):
            dht = DHTNode()
            await dht.start()
            for a in seq:
                await announce_piece(dht, "h" * 8, a)
            out = await find_providers(dht, "h" * 8)
            await dht.stop()
            return out

        fwd = asyncio.run(run(addrs))
        rev = asyncio.run(run(list(reversed(addrs))))
        assert sorted(fwd) == sorted(rev) == sorted(set(addrs))

    @given(ranks=st.lists(st.integers(min_value=0, max_value=3),
                          min_size=1, max_size=8))
    @settings(max_examples=30, deadline=None)
    def test_rank_announce_last_write_wins_per_peer(self, ranks):
        from bee2bee_amd.mesh.dht import DHTNode, announce_rank, find_ranks

        async def run():
            dht = DHTNode()
            await dht.start()
            for i, r in enumerate(ranks):
                await announce_rank(
                    dht, "g", f"peer{r}",
                    {"rank": i, "endpoint": f"127.0.0.1:{5000 + i}"})
            out = await find_ranks(dht, "g")
            await dht.stop()
            return out

        table = asyncio.run(run())
        assert set(table) == {f"peer{r}" for r in ranks}
        # each peer's record is its LAST announcement
        for r in set(ranks):
            last = max(i for i, x in enumerate(ranks) if x == r)
            assert table[f"peer{r}"]["rank"] == last


class TestPlannerProperties:
    @given(
        n_layers=st.integers(min_value=1, max_value=96),
        n_stages=st.integers(min_value=1, max_value=16),
        budgets=st.one_of(
            st.none(),
            st.lists(st.integers(min_value=1, max_value=1000),
                     min_size=1, max_size=16),
        ),
    )
    @settings(max_examples=200, deadline=None)
    def test_plan_partitions_layers_exactly(self, n_layers, n_stages,
                                            budgets):
        """Any (layers, stages, budgets) combo: stages are contiguous,
        non-empty, cover [0, n_layers) exactly once; embed on the first,
        head on the last."""
        from dataclasses import replace

        from bee2bee_amd.models.spec import PRESETS
        from bee2bee_amd.parallel.planner import plan_stages

        spec = replace(PRESETS["tiny"], n_layers=n_layers)
        if budgets is not None and len(budgets) != n_stages:
            budgets = (budgets * n_stages)[:n_stages]
        if n_stages > n_layers:
            with pytest.raises(ValueError):
                plan_stages(spec, n_stages, budgets)
            return
        plans = plan_stages(spec, n_stages, budgets)
        assert len(plans) == n_stages
        lo = 0
        for r, p in enumerate(plans):
            assert p.rank == r
            assert p.layer_range[0] == lo
            assert p.layer_range[1] > p.layer_range[0], "empty stage"
            lo = p.layer_range[1]
            assert p.has_embed == (r == 0)
            assert p.has_head == (r == n_stages - 1)
            assert p.est_bytes > 0
        assert lo == n_layers


class TestDeviceSlotMapping:
    @given(lens=st.lists(st.integers(min_value=1, max_value=3 * BLOCK),
                         min_size=1, max_size=5))
    @settings(max_examples=100, deadline=None)
    def test_device_mapping_equals_host_mapping(self, lens):
        """graphs.decode_slot_mapping (the hipGraph-capturable tensor path)
        must equal PagedKV.slot_mapping (the host path) for the LAST
        position of every sequence — the decode-step contract."""
        from bee2bee_amd.engine.graphs import decode_slot_mapping

        kv = PagedKV(PRESETS["tiny"], torch.device("cpu"), torch.float32,
                     n_blocks=32, block_size=BLOCK)
        for sid, ln in enumerate(lens):
            kv.new_seq(sid)
            kv.extend_seq(sid, ln)
        table = kv.block_table(range(len(lens)))
        positions = torch.tensor([ln - 1 for ln in lens], dtype=torch.int32)
        dev_slots = decode_slot_mapping(table, positions, BLOCK).tolist()
        host_slots = [kv.slot_mapping(sid, [ln - 1])[0]
                      for sid, ln in enumerate(lens)]
        assert dev_slots == host_slots


class TestLayerRangeLoads:
    @given(data=st.data(),
           arch=st.sampled_from(["tiny", "tiny-gpt2"]))
    @settings(max_examples=20, deadline=None)
    def test_any_layer_range_loads_exact_slice(self, data, arch, tmp_path_factory):
        """Every [lo, hi) slice of a checkpoint loads exactly those layers
        (PP stage loading), with embed/pos on the first stage and final
        norm/head on the last — llama AND gpt2 formats."""
        from bee2bee_amd.models.weights import ModelWeights, save_hf

        spec = PRESETS[arch]
        n = spec.n_layers
        lo = data.draw(st.integers(min_value=0, max_value=n - 1))
        hi = data.draw(st.integers(min_value=lo + 1, max_value=n))
        d = tmp_path_factory.mktemp("ckpt")
        full = ModelWeights(spec, torch.device("cpu"),
                            torch.float32).random_init(7)
        save_hf(full, str(d))
        part = ModelWeights(spec, torch.device("cpu"), torch.float32).load_hf(
            str(d), layer_range=(lo, hi))
        for i in range(n):
            loaded = part.layers[i].wqkv is not None
            assert loaded == (lo <= i < hi), (i, lo, hi)
            if loaded:
                assert torch.equal(part.layers[i].wqkv, full.layers[i].wqkv)
        assert (part.embed is not None) == (lo == 0)
        assert (part.final_norm is not None) == (hi == n)
        if spec.pos_type == "learned":
            assert (part.pos_embed is not None) == (lo == 0)


class TestEngineSchedulingEquivalence:
    @given(data=st.data())
    @settings(max_examples=10, deadline=None)
    def test_chunked_prefill_scheduling_is_output_invariant(self, data):
        """Random request mixes: tiny chunked-prefill budget (forces chunk
        scheduling + decode interleave) must emit token-identical outputs
        to an unconstrained engine."""
        from bee2bee_amd.engine.engine import InferenceEngine

        n_req = data.draw(st.integers(min_value=1, max_value=4))
        prompts = [
            data.draw(st.lists(st.integers(min_value=5, max_value=400),
                               min_size=1, max_size=40))
            for _ in range(n_req)
        ]
        # max_new=1 is over-weighted: finish-at-prefill completions are
        # exactly the window where the _last_sampled clobber bug lived
        max_news = [data.draw(st.sampled_from([1, 1, 2, 4, 8]))
                    for _ in range(n_req)]

        spec = data.draw(st.booleans())

        def run(max_prefill_tokens, spec_decode=False):
            eng = InferenceEngine("tiny", device="cpu", max_batch=4,
                                  max_seq_len=128, seed=5,
                                  max_prefill_tokens=max_prefill_tokens,
                                  spec_decode=spec_decode)
            try:
                reqs = [eng.generate_async(p, max_new_tokens=m,
                                           temperature=0.0)
                        if hasattr(eng, "generate_async") else None
                        for p, m in zip(prompts, max_news)]
                if reqs[0] is None:
                    # no async helper: submit directly
                    from bee2bee_amd.engine.engine import GenerationRequest
                    from bee2bee_amd.engine.sampler import SamplingParams

                    reqs = [GenerationRequest(
                        prompt_ids=p, max_new_tokens=m,
                        sampling=SamplingParams(greedy=True))
                        for p, m in zip(prompts, max_news)]
                    for r in reqs:
                        eng.submit(r)
                outs = []
                for r in reqs:
                    while True:
                        item = r.out_queue.get(timeout=60)
                        if not isinstance(item, int):
                            break
                    assert r.error is None, r.error
                    outs.append(list(r.output_ids))
                return outs
            finally:
                eng.shutdown()

        base = run(max_prefill_tokens=4096)
        assert run(max_prefill_tokens=16) == base
        if spec:  # chunked prefill AND speculative decoding together
            assert run(max_prefill_tokens=16, spec_decode=True) == base


class TestSpecDecodeInvariance:
    @given(data=st.data(), penalty=st.sampled_from([1.0, 1.15]))
    @settings(max_examples=6, deadline=None)
    def test_spec_decode_output_invariant_random_mixes(self, data, penalty):
        """Speculative decoding must be token-identical to plain decode for
        ANY prompt mix, greedy and penalized-greedy both (the round-2
        penalized verification path)."""
        from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
        from bee2bee_amd.engine.sampler import SamplingParams

        n_req = data.draw(st.integers(min_value=1, max_value=3))
        # small alphabet -> repetitive prompts -> the proposer actually fires
        prompts = [
            data.draw(st.lists(st.integers(min_value=5, max_value=12),
                               min_size=2, max_size=24))
            for _ in range(n_req)
        ]

        def run(spec):
            eng = InferenceEngine("tiny", device="cpu", max_batch=4,
                                  max_seq_len=128, seed=9, spec_decode=spec)
            try:
                reqs = [GenerationRequest(
                    prompt_ids=p, max_new_tokens=10,
                    sampling=SamplingParams(greedy=True,
                                            repetition_penalty=penalty))
                    for p in prompts]
                for r in reqs:
                    eng.submit(r)
                outs = []
                for r in reqs:
                    while True:
                        item = r.out_queue.get(timeout=60)
                        if not isinstance(item, int):
                            break
                    assert r.error is None, r.error
                    outs.append(list(r.output_ids))
                return outs
            finally:
                eng.shutdown()

        assert run(True) == run(False)


class TestMixedSamplingBatch:
    @given(data=st.data())
    @settings(max_examples=8, deadline=None)
    def test_greedy_rows_invariant_to_sampled_neighbors(self, data):
        """Greedy (and penalized-greedy) requests co-batched with
        temperature-sampled neighbors must emit exactly what they emit
        alone — the sampling-group split and penalty-slot pool must never
        leak across rows."""
        from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
        from bee2bee_amd.engine.sampler import SamplingParams

        n_greedy = data.draw(st.integers(min_value=1, max_value=2))
        greedy_specs = []
        for _ in range(n_greedy):
            prompt = data.draw(st.lists(
                st.integers(min_value=5, max_value=40),
                min_size=1, max_size=12))
            pen = data.draw(st.sampled_from([1.0, 1.15]))
            greedy_specs.append((prompt, pen))
        n_sampled = data.draw(st.integers(min_value=1, max_value=2))

        def make_reqs():
            reqs = [GenerationRequest(
                prompt_ids=p, max_new_tokens=6,
                sampling=SamplingParams(greedy=True, repetition_penalty=pen))
                for p, pen in greedy_specs]
            for i in range(n_sampled):
                reqs.append(GenerationRequest(
                    prompt_ids=[20 + i, 21], max_new_tokens=6,
                    sampling=SamplingParams(temperature=0.9, top_p=0.9,
                                            top_k=8)))
            return reqs

        def drain(eng, reqs):
            for r in reqs:
                eng.submit(r)
            for r in reqs:
                while True:
                    item = r.out_queue.get(timeout=60)
                    if not isinstance(item, int):
                        break
                assert r.error is None, r.error
            return [list(r.output_ids) for r in reqs]

        eng = InferenceEngine("tiny", device="cpu", max_batch=8,
                              max_seq_len=128, seed=3)
        try:
            mixed = drain(eng, make_reqs())[:n_greedy]
        finally:
            eng.shutdown()

        eng = InferenceEngine("tiny", device="cpu", max_batch=8,
                              max_seq_len=128, seed=3)
        try:
            solo_reqs = [GenerationRequest(
                prompt_ids=p, max_new_tokens=6,
                sampling=SamplingParams(greedy=True, repetition_penalty=pen))
                for p, pen in greedy_specs]
            solo = drain(eng, solo_reqs)
        finally:
            eng.shutdown()

        assert mixed == solo


class TestArrivalTimingInvariance:
    @given(data=st.data())
    @settings(max_examples=5, deadline=None)
    def test_staggered_arrivals_match_burst(self, data):
        """Per-request greedy outputs must not depend on WHEN other
        requests arrive (mid-decode admissions vs one burst)."""
        import time as _t

        from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
        from bee2bee_amd.engine.sampler import SamplingParams

        n = data.draw(st.integers(min_value=2, max_value=4))
        prompts = [data.draw(st.lists(st.integers(min_value=5, max_value=60),
                                      min_size=1, max_size=20))
                   for _ in range(n)]

        def run(stagger):
            eng = InferenceEngine("tiny", device="cpu", max_batch=8,
                                  max_seq_len=128, seed=11)
            try:
                reqs = [GenerationRequest(
                    prompt_ids=p, max_new_tokens=5,
                    sampling=SamplingParams(greedy=True)) for p in prompts]
                for r in reqs:
                    eng.submit(r)
                    if stagger:
                        _t.sleep(0.03)  # others decode while this queues
                outs = []
                for r in reqs:
                    while True:
                        item = r.out_queue.get(timeout=60)
                        if not isinstance(item, int):
                            break
                    assert r.error is None, r.error
                    outs.append(list(r.output_ids))
                return outs
            finally:
                eng.shutdown()

        assert run(True) == run(False)
