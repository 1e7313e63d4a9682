"""Expert-parallel MoE engine: data-parallel ranks with experts sharded
across the group (BASELINE config 5 — Mixtral 8x7B, 8 peers, expert
all-to-all on RCCL).

Every rank holds the full attention/embedding stack and E/world experts;
each rank decodes its OWN batch, and the MoE layers exchange tokens with
dist.all_to_all_single (parallel/ep.py) — the collective that drives all 7
xGMI links of each MI355X concurrently. Ranks must step in lockstep (the
all-to-all is a synchronization point per MoE layer), so the engine exposes
the same lockstep prefill/decode API as PipelineEngine.

CPU-tested (gloo, world 2) against the single-process engine in
tests/test_ep_cpu.py::test_moe_engine_matches_single.
"""
from __future__ import annotations

import logging
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from ..engine.graphs import decode_slot_mapping
from ..engine.kv import PagedKV
from ..engine import kv as kv_mod
from ..engine.runner import Runner
from ..engine.sampler import SamplingParams, sample
from ..models.spec import ModelSpec, resolve_spec
from ..models.weights import ModelWeights
from .ep import ExpertParallelMoE

logger = logging.getLogger("bee2bee_amd.parallel")


class MoEEngine:
    """One expert-parallel rank; all ranks step in lockstep over their own
    batches."""

    def __init__(
        self,
        model: str | ModelSpec,
        device: Optional[str] = None,
        dtype: Optional[torch.dtype] = None,
        model_path: Optional[str] = None,
        max_batch: int = 64,
        max_seq_len: int = 2048,
        seed: int = 0,
        group: Optional[dist.ProcessGroup] = None,
    ) -> None:
        assert dist.is_initialized(), "init the process group first"
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        self.spec = model if isinstance(model, ModelSpec) else resolve_spec(model, model_path)
        assert self.spec.is_moe, "MoEEngine is for MoE models"
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        if dtype is None:
            dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.dtype = dtype
        if self.device.type == "cuda":
            from .. import ops

            ops.require_hip()

        self.max_seq_len = min(max_seq_len, self.spec.max_seq_len)
        # expert sharding: this rank only materializes its E/world experts
        # (an 8-way Mixtral rank holds 1/8 of the expert bytes; gate and the
        # attention stack stay replicated)
        local_e = self.spec.n_experts // self.world
        e_lo = self.rank * local_e
        expert_range = (e_lo, e_lo + local_e)
        self.weights = ModelWeights(self.spec, self.device, dtype)
        if model_path:
            self.weights.load_hf(model_path, expert_range=expert_range)
        else:
            self.weights.random_init(seed=seed, expert_range=expert_range)
        blocks_per_seq = -(-self.max_seq_len // kv_mod.BLOCK_SIZE)
        self.kv = PagedKV(
            self.spec, self.device, dtype,
            n_blocks=max_batch * blocks_per_seq + 4,
        )
        self.runner = Runner(self.spec, self.weights, self.kv, self.device, dtype)
        self.runner.ep = ExpertParallelMoE(
            self.spec.n_experts, self.spec.top_k_experts, group=group
        )
        self._seqs: List[int] = []
        self._lens: List[int] = []
        logger.info(
            "EP rank %d/%d: experts [%d, %d) on %s",
            self.rank, self.world, self.runner.ep.e_lo,
            self.runner.ep.e_lo + self.runner.ep.local_e, self.device,
        )

    @torch.no_grad()
    def prefill(
        self,
        prompts: Sequence[Sequence[int]],
        sampling: Optional[SamplingParams] = None,
    ) -> torch.Tensor:
        """Prefill this rank's batch (all ranks must call together)."""
        sampling = sampling or SamplingParams(greedy=True)
        self._seqs = list(range(len(self._seqs), len(self._seqs) + len(prompts)))
        self._lens = [len(p) for p in prompts]
        ids_list, pos_list, slot_list, cu = [], [], [], [0]
        for sid, p in zip(self._seqs, prompts):
            self.kv.new_seq(sid)
            self.kv.extend_seq(sid, len(p))
            ids_list.extend(p)
            pos_list.extend(range(len(p)))
            slot_list.extend(self.kv.slot_mapping(sid, range(len(p))))
            cu.append(cu[-1] + len(p))
        dev = self.device
        hidden = self.runner.forward_prefill(
            torch.tensor(ids_list, dtype=torch.int64, device=dev),
            torch.tensor(pos_list, dtype=torch.int32, device=dev),
            torch.tensor(slot_list, dtype=torch.int32, device=dev),
            torch.tensor(cu, dtype=torch.int32, device=dev),
            max(self._lens),
        )
        last = torch.tensor([c - 1 for c in cu[1:]], dtype=torch.int64, device=dev)
        return sample(self.runner.lm_head(hidden[last]), sampling).cpu()

    @torch.no_grad()
    def decode_step(
        self,
        ids: torch.Tensor,
        sampling: Optional[SamplingParams] = None,
    ) -> torch.Tensor:
        sampling = sampling or SamplingParams(greedy=True)
        dev = self.device
        positions = torch.tensor(self._lens, dtype=torch.int32, device=dev)
        for i, sid in enumerate(self._seqs):
            self.kv.extend_seq(sid, self._lens[i] + 1)
            self._lens[i] += 1
        lens_t = torch.tensor(self._lens, dtype=torch.int32, device=dev)
        bt = self.kv.block_table(self._seqs)
        slots = decode_slot_mapping(bt, positions, self.kv.block_size)
        hidden = self.runner.forward_decode(
            ids.to(dev), positions, slots, bt, lens_t
        )
        return sample(self.runner.lm_head(hidden), sampling).cpu()

    @torch.no_grad()
    def reset(self) -> None:
        """Free the batch's KV so repeated generate() calls (serving via
        parallel/serve.py LockstepServer) never exhaust the pool — prefill
        allocates fresh sequence ids per call."""
        for sid in self._seqs:
            self.kv.free_seq(sid)
        self._seqs, self._lens = [], []

    def generate(
        self,
        prompts: Sequence[Sequence[int]],
        max_new_tokens: int,
        sampling: Optional[SamplingParams] = None,
    ) -> List[List[int]]:
        ids = self.prefill(prompts, sampling)
        outs = [[int(t)] for t in ids]
        for _ in range(max_new_tokens - 1):
            ids = self.decode_step(ids, sampling)
            for o, t in zip(outs, ids):
                o.append(int(t))
        return outs
