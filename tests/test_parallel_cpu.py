"""Pipeline-parallel correctness on CPU: gloo backend, world_size 2, loopback
(the distributed-path test story that runs without GPUs; RCCL uses the same
code path on the GPU box)."""
import multiprocessing as mp
import os
import pickle

import pytest
import torch

from bee2bee_amd.models.spec import PRESETS
from bee2bee_amd.parallel.planner import plan_stages

PROMPTS = [[5, 6, 7, 8, 9], [100, 101], [42] * 9]
N_NEW = 6
SEED = 21


def test_planner_even_split():
    spec = PRESETS["llama3-70b"]
    plans = plan_stages(spec, 8)
    assert [p.layer_range for p in plans][0][0] == 0
    assert plans[-1].layer_range[1] == spec.n_layers
    sizes = [p.layer_range[1] - p.layer_range[0] for p in plans]
    assert sum(sizes) == 80 and max(sizes) == 10 and min(sizes) == 10
    assert plans[0].has_embed and plans[-1].has_head
    assert not plans[1].has_embed and not plans[1].has_head


def test_planner_weighted_split():
    spec = PRESETS["tiny"]  # 2 layers
    plans = plan_stages(spec, 2, mem_budgets=[1, 1])
    assert [p.layer_range for p in plans] == [(0, 1), (1, 2)]


def test_rank_order_deterministic():
    from bee2bee_amd.parallel.rendezvous import rank_order

    records = {
        "peer-bb": {"host": "10.0.0.2", "port": 1},
        "peer-aa": {"host": "10.0.0.1", "port": 29500},
    }
    peers, master = rank_order(records)
    assert peers == ["peer-aa", "peer-bb"]
    assert master == "10.0.0.1:29500"


def test_partial_random_init_matches_full():
    """A stage materializing layers [1,2) must hold the same values as the
    full init (prerequisite for PP == single-process)."""
    from bee2bee_amd.models.weights import ModelWeights

    spec = PRESETS["tiny"]
    full = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(seed=SEED)
    part = ModelWeights(spec, torch.device("cpu"), torch.float32).random_init(
        seed=SEED, layer_range=(1, 2)
    )
    assert torch.equal(full.layers[1].wqkv, part.layers[1].wqkv)
    assert torch.equal(full.layers[1].w_down, part.layers[1].w_down)
    assert torch.equal(full.lm_head, part.lm_head)


def _pp_worker(rank: int, world: int, port: int, out_path: str,
               model: str = "tiny") -> None:
    import torch.distributed as dist

    from bee2bee_amd.parallel.pp import PipelineEngine

    dist.init_process_group(
        backend="gloo",
        init_method=f"tcp://127.0.0.1:{port}",
        rank=rank,
        world_size=world,
    )
    try:
        eng = PipelineEngine(
            model, device="cpu", max_batch=4, max_seq_len=64, seed=SEED
        )
        outs = eng.generate(PROMPTS, N_NEW)
        if rank == world - 1:
            with open(out_path, "wb") as f:
                pickle.dump(outs, f)
    finally:
        dist.destroy_process_group()


def _single_process_reference(model: str = "tiny"):
    from bee2bee_amd.engine.engine import GenerationRequest, InferenceEngine
    from bee2bee_amd.engine.sampler import SamplingParams

    eng = InferenceEngine(model, device="cpu", max_batch=4, max_seq_len=64, seed=SEED)
    try:
        outs = []
        for p in PROMPTS:
            req = GenerationRequest(
                prompt_ids=list(p), max_new_tokens=N_NEW,
                sampling=SamplingParams(greedy=True),
            )
            eng.submit(req)
            while True:
                item = req.out_queue.get(timeout=60)
                if not isinstance(item, int):
                    break
            outs.append(req.output_ids)
        return outs
    finally:
        eng.shutdown()


@pytest.mark.timeout(300)
@pytest.mark.parametrize(
    "world,port,model",
    [(2, 29611, "tiny"), (3, 29612, "tiny3"),
     # gpt2 family across stages: learned positions live on the first
     # stage, LayerNorm/gelu/biases run on every stage
     (2, 29613, "tiny-gpt2")]
)
def test_pp_matches_single_process(tmp_path, world, port, model):
    """2- and 3-stage pipeline greedy decode == single-process engine
    (fp32 exact). world=3 on the 3-layer tiny3 exercises a pure-middle
    rank (no embedding, no lm_head) — the shape every interior rank of
    a llama3-70b/8 deployment has."""
    out_path = str(tmp_path / "pp_out.pkl")
    ctx = mp.get_context("spawn")
    procs = [
        ctx.Process(target=_pp_worker, args=(r, world, port, out_path, model))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0, f"pp worker failed (exit {p.exitcode})"
    with open(out_path, "rb") as f:
        pp_outs = pickle.load(f)
    ref = _single_process_reference(model)
    assert pp_outs == ref
