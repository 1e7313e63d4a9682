"""Expert parallelism on CPU: gloo world_size 2, all-to-all dispatch must
reproduce the single-process MoE MLP."""
import multiprocessing as mp
import pickle

import pytest
import torch

from bee2bee_amd.models.spec import PRESETS

SEED = 33


def _single_reference():
    from bee2bee_amd.engine.kv import PagedKV
    from bee2bee_amd.engine.runner import Runner
    from bee2bee_amd.models.weights import ModelWeights

    spec = PRESETS["tiny-moe"]
    torch.manual_seed(0)
    w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(SEED)
    kv = PagedKV(spec, torch.device("cpu"), torch.float32, n_blocks=8)
    runner = Runner(spec, w, kv, torch.device("cpu"), torch.float32)
    x = torch.randn(7, spec.hidden_size, generator=torch.Generator().manual_seed(9))
    return runner._moe_mlp(w.layers[0], x), x


def _ep_worker(rank: int, world: int, port: int, out_path: str) -> None:
    import torch.distributed as dist

    from bee2bee_amd.models.weights import ModelWeights
    from bee2bee_amd.parallel.ep import ExpertParallelMoE

    dist.init_process_group(
        backend="gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world,
    )
    try:
        spec = PRESETS["tiny-moe"]
        w = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(SEED)
        lw = w.layers[0]
        ep = ExpertParallelMoE(spec.n_experts, spec.top_k_experts)
        x = torch.randn(
            7, spec.hidden_size, generator=torch.Generator().manual_seed(9)
        )
        lo, hi = ep.e_lo, ep.e_lo + ep.local_e
        out = ep.forward(x, lw.moe_gate, lw.moe_w_gate_up[lo:hi], lw.moe_w_down[lo:hi])
        if rank == 0:
            with open(out_path, "wb") as f:
                pickle.dump(out, f)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(240)
def test_ep2_matches_single(tmp_path):
    ref, _x = _single_reference()
    out_path = str(tmp_path / "ep_out.pkl")
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(target=_ep_worker, args=(r, 2, 29617, out_path))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=200)
        assert p.exitcode == 0
    with open(out_path, "rb") as f:
        ep_out = pickle.load(f)
    assert torch.allclose(ep_out, ref, atol=1e-5), (ep_out - ref).abs().max()


def _moe_engine_worker(rank, world, port, out_path):
    import torch.distributed as dist

    from bee2bee_amd.parallel.moe_engine import MoEEngine

    dist.init_process_group(
        backend="gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world,
    )
    try:
        eng = MoEEngine("tiny-moe", device="cpu", max_batch=4, max_seq_len=64,
                        seed=SEED)
        prompts = [[rank * 10 + 1, 2, 3], [rank * 10 + 4, 5]]
        outs = eng.generate(prompts, 5)
        with open(f"{out_path}.{rank}", "wb") as f:
            pickle.dump((prompts, outs), f)
    finally:
        dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_moe_engine_matches_single(tmp_path):
    """Each EP rank's generations equal the single-process engine run on
    the same prompts (fp32 exact)."""
    import pickle as pkl

    out_path = str(tmp_path / "moe_eng")
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(target=_moe_engine_worker, args=(r, 2, 29641, out_path))
        for r in range(2)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0

    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    single = InferenceEngine("tiny-moe", device="cpu", max_batch=4,
                             max_seq_len=64, seed=SEED)
    try:
        for r in range(2):
            with open(f"{out_path}.{r}", "rb") as f:
                prompts, ep_outs = pkl.load(f)
            for prompt, ep_out in zip(prompts, ep_outs):
                req = GenerationRequest(
                    prompt_ids=list(prompt), max_new_tokens=5,
                    sampling=SamplingParams(greedy=True),
                )
                single.submit(req)
                while True:
                    item = req.out_queue.get(timeout=60)
                    if not isinstance(item, int):
                        break
                assert req.output_ids == ep_out, (r, prompt)
    finally:
        single.shutdown()


@pytest.mark.timeout(300)
def test_ep4_one_expert_per_rank(tmp_path):
    """EP at world=4 on 4 experts: the degenerate one-expert-per-rank shard
    (local_e == 1) must still match the single-process reference."""
    ref, _x = _single_reference()
    out_path = str(tmp_path / "ep4_out.pkl")
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(target=_ep_worker, args=(r, 4, 29619, out_path))
        for r in range(4)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0
    with open(out_path, "rb") as f:
        ep_out = pickle.load(f)
    assert torch.allclose(ep_out, ref, atol=1e-5), (ep_out - ref).abs().max()
