"""Tensor parallelism: every rank holds a head/intermediate shard of every
layer and the ranks compute the SAME batch together, synchronizing with two
RCCL all-reduces per layer (after the attention output projection and after
the MLP down projection) — the classic Megatron split, sized for xGMI:
the all-reduced tensor is [T, H] (tiny next to the sharded GEMM work), and
on an 8-GPU MI355X node RCCL rings it over the point-to-point mesh.

Sharding (world must divide n_kv_heads and intermediate_size):
  wqkv      [q+2kv, H]  -> this rank's q-head rows + kv-head rows
  wo        [H, q_size] -> columns of the owned q heads
  w_gate_up [2I, H]     -> owned rows of BOTH the gate and up halves
  w_down    [H, I]      -> columns of the owned intermediate slice
Norms, embeddings and lm_head are replicated, so after each all-reduce all
ranks hold identical hidden states -> identical logits -> identical samples
(same generator seed), no broadcast needed.

KV heads shard WITH their GQA q-groups, so the attention kernels see the
same group size G on a smaller n_kv_heads — the paged KV pool and the HIP
decode/prefill kernels run unchanged on the local spec.

The reference has no tensor parallelism (its nodes serve whole models);
this is an MI355X-native addition alongside PP (pp.py) and EP (ep.py).
CPU-tested with gloo world 2 against the single-process engine.
"""
from __future__ import annotations

import dataclasses
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

logger = logging.getLogger("bee2bee_amd.parallel")


def shard_spec(spec: ModelSpec, world: int) -> ModelSpec:
    assert spec.n_kv_heads % world == 0, "world must divide n_kv_heads"
    assert spec.intermediate_size % world == 0
    assert not spec.is_moe, "use expert parallelism (ep.py) for MoE models"
    # row-parallel projections all-reduce partial outputs; an output-side
    # bias (gpt2 family) would be added once PER RANK — scope TP to the
    # swiglu/no-out-bias families until bias handling lands
    assert spec.act_type == "swiglu" and not spec.attn_out_bias, (
        "TP supports the llama/mistral/qwen families; gpt2-family biases "
        "need rank-0-only bias handling")
    return dataclasses.replace(
        spec,
        n_heads=spec.n_heads // world,
        n_kv_heads=spec.n_kv_heads // world,
        intermediate_size=spec.intermediate_size // world,
    )


@torch.no_grad()
def shard_weights(
    full: ModelWeights, spec: ModelSpec, rank: int, world: int
) -> ModelWeights:
    """Slice a fully materialized ModelWeights into this rank's TP shard
    (values are exactly the full init's slices, so TP == single-process up
    to float summation order in the all-reduced projections)."""
    lspec = shard_spec(spec, world)
    hd = spec.head_dim
    q_lo = rank * lspec.n_heads * hd
    q_hi = q_lo + lspec.n_heads * hd
    kv_lo = rank * lspec.n_kv_heads * hd
    kv_hi = kv_lo + lspec.n_kv_heads * hd
    i_lo = rank * lspec.intermediate_size
    i_hi = i_lo + lspec.intermediate_size
    I = spec.intermediate_size

    out = ModelWeights(lspec, full.device, full.dtype)
    out.embed = full.embed
    out.final_norm = full.final_norm
    out.lm_head = full.lm_head
    for li, lw in enumerate(full.layers):
        ol = out.layers[li]
        ol.attn_norm = lw.attn_norm
        ol.mlp_norm = lw.mlp_norm
        q = lw.wqkv[q_lo:q_hi]
        k = lw.wqkv[spec.q_size + kv_lo : spec.q_size + kv_hi]
        v = lw.wqkv[spec.q_size + spec.kv_size + kv_lo :
                    spec.q_size + spec.kv_size + kv_hi]
        ol.wqkv = torch.cat([q, k, v], dim=0).contiguous()
        if lw.wqkv_bias is not None:
            ol.wqkv_bias = torch.cat([
                lw.wqkv_bias[q_lo:q_hi],
                lw.wqkv_bias[spec.q_size + kv_lo : spec.q_size + kv_hi],
                lw.wqkv_bias[spec.q_size + spec.kv_size + kv_lo :
                             spec.q_size + spec.kv_size + kv_hi],
            ], dim=0).contiguous()
        ol.wo = lw.wo[:, q_lo:q_hi].contiguous()
        ol.w_gate_up = torch.cat(
            [lw.w_gate_up[i_lo:i_hi], lw.w_gate_up[I + i_lo : I + i_hi]],
            dim=0,
        ).contiguous()
        ol.w_down = lw.w_down[:, i_lo:i_hi].contiguous()
    return out


class TPEngine:
    """One tensor-parallel rank; all ranks step in lockstep over the SAME
    batch (outputs are identical on every rank)."""

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
        spec = model if isinstance(model, ModelSpec) else resolve_spec(model, model_path)
        self.spec = spec
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        if dtype is None:
            dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.dtype = dtype
        if self.device.type == "cuda":
            from .. import ops

            ops.require_hip()

        self.max_seq_len = min(max_seq_len, spec.max_seq_len)
        full = ModelWeights(spec, self.device, dtype)
        if model_path:
            full.load_hf(model_path)
        else:
            full.random_init(seed=seed)
        self.weights = shard_weights(full, spec, self.rank, self.world)
        self.lspec = self.weights.spec
        del full  # transient: only the shard stays resident

        blocks_per_seq = -(-self.max_seq_len // kv_mod.BLOCK_SIZE)
        self.kv = PagedKV(
            self.lspec, self.device, dtype,
            n_blocks=max_batch * blocks_per_seq + 4,
        )
        self.runner = Runner(self.lspec, self.weights, self.kv, self.device, dtype)
        self.runner.tp_group = group if group is not None else dist.group.WORLD
        # non-greedy sampling must draw IDENTICAL tokens on every rank or the
        # KV caches silently diverge: a dedicated generator seeded
        # the same ranks-wide, never the global per-process RNG
        self.sample_gen = torch.Generator(device=self.device)
        self.sample_gen.manual_seed(seed + 0x5EED)
        self._seqs: List[int] = []
        self._lens: List[int] = []
        logger.info(
            "TP rank %d/%d: %d q heads, %d kv heads, I=%d on %s",
            self.rank, self.world, self.lspec.n_heads, self.lspec.n_kv_heads,
            self.lspec.intermediate_size, self.device,
        )

    @torch.no_grad()
    def prefill(
        self,
        prompts: Sequence[Sequence[int]],
        sampling: Optional[SamplingParams] = None,
    ) -> torch.Tensor:
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
        return sample(self.runner.lm_head(hidden[last]), sampling,
                      generator=self.sample_gen).cpu()

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
        return sample(self.runner.lm_head(hidden), sampling,
                      generator=self.sample_gen).cpu()

    @torch.no_grad()
    def reset(self) -> None:
        """Free the batch's KV so repeated generate() calls (serving via
        parallel/serve.py LockstepServer) never exhaust the pool."""
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
