"""Pipeline-parallel inference over torch.distributed (RCCL on MI355X).

Each rank owns a contiguous layer shard (parallel/planner.py), its own
paged-KV pool for those layers, and a Runner. Hidden states hop between
stages as bf16 [T, hidden] tensors via dist.send/recv — on an 8-GPU MI355X
node these are point-to-point xGMI transfers (one dedicated link per
pipeline edge, ~153 GB/s); a decode hop for batch 256 x 8192 hidden is
4 MB -> ~30 us per edge. The sampled token ids are broadcast from the last
stage so every rank can start the next step without a host round-trip.

The reference moved these activations as JSON float lists over WebSockets
(bee2bee/node.py:270-277) — that path is what this module replaces; the WS
mesh remains the control plane that forms the group (parallel/rendezvous).

CPU-tested with the gloo backend and world_size 2 (tests/test_parallel_cpu
.py): PP greedy decode must exactly match the single-process engine at fp32.
"""
from __future__ import annotations

import logging
from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from ..engine.graphs import decode_slot_mapping
from ..engine.kv import PagedKV
from ..engine import kv as kv_mod
from ..engine.runner import Runner
from ..engine.sampler import SamplingParams, sample
from ..models.spec import ModelSpec, resolve_spec
from ..models.weights import ModelWeights
from .planner import plan_stages

logger = logging.getLogger("bee2bee_amd.parallel")


class PipelineEngine:
    """One pipeline stage; all ranks step in lockstep.

    All ranks are constructed with the same arguments and call the same
    methods with the same control inputs (prompt lengths, step counts);
    tensor payloads flow stage-to-stage over the process group."""

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
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        if dtype is None:
            dtype = torch.bfloat16 if self.device.type == "cuda" else torch.float32
        self.dtype = dtype
        if self.device.type == "cuda":
            from .. import ops

            ops.require_hip()

        plan = plan_stages(self.spec, self.world)[self.rank]
        self.layer_range = plan.layer_range
        self.max_batch = max_batch
        self.max_seq_len = min(max_seq_len, self.spec.max_seq_len)

        self.weights = ModelWeights(self.spec, self.device, dtype)
        if model_path:
            self.weights.load_hf(model_path, layer_range=self.layer_range)
        else:
            self.weights.random_init(seed=seed, layer_range=self.layer_range)
        blocks_per_seq = -(-self.max_seq_len // kv_mod.BLOCK_SIZE)
        self.kv = PagedKV(
            self.spec, self.device, dtype,
            n_blocks=max_batch * blocks_per_seq + 4,
            layer_range=self.layer_range,
        )
        self.runner = Runner(
            self.spec, self.weights, self.kv, self.device, dtype,
            layer_range=self.layer_range,
        )
        self._seqs: List[int] = []
        self._lens: List[int] = []
        logger.info(
            "PP stage %d/%d layers %s on %s", self.rank, self.world,
            self.layer_range, self.device,
        )

    # ------------------------------------------------------------ transport
    # RCCL moves device tensors directly (xGMI); gloo (CPU tests, or a
    # single-GPU multi-process test) stages through host memory.

    @property
    def _wire_gpu(self) -> bool:
        return self.device.type == "cuda" and dist.get_backend(self.group) == "nccl"

    def _send(self, t: torch.Tensor, dst: int) -> None:
        t = t.contiguous()
        dist.send(t if self._wire_gpu else t.cpu(), dst, group=self.group)

    def _recv(self, shape, dtype, src: int) -> torch.Tensor:
        buf = torch.empty(
            shape, dtype=dtype,
            device=self.device if self._wire_gpu else "cpu",
        )
        dist.recv(buf, src, group=self.group)
        return buf.to(self.device)

    def _bcast_ids(self, ids: Optional[torch.Tensor], B: int) -> torch.Tensor:
        src = self.world - 1
        if ids is None:
            ids = torch.zeros(B, dtype=torch.int64)
        buf = ids.to(self.device) if self._wire_gpu else ids.cpu()
        dist.broadcast(buf, src, group=self.group)
        return buf

    # -------------------------------------------------------------- serving

    @torch.no_grad()
    def prefill(
        self,
        prompts: Sequence[Sequence[int]],
        sampling: Optional[SamplingParams] = None,
    ) -> torch.Tensor:
        """Prefill a batch of prompts; returns the first sampled token ids
        [B] (identical on every rank). Stage 0 needs real prompt ids; later
        stages only need the lengths (they receive hidden states)."""
        sampling = sampling or SamplingParams(greedy=True)
        B = len(prompts)
        lens = [len(p) for p in prompts]
        T = sum(lens)
        self._seqs = list(range(len(self._seqs), len(self._seqs) + B))
        self._lens = list(lens)
        ids_list, pos_list, slot_list, cu = [], [], [], [0]
        for sid, p in zip(self._seqs, prompts):
            self.kv.new_seq(sid)
            self.kv.extend_seq(sid, len(p))
            ids_list.extend(p)
            pos_list.extend(range(len(p)))
            slot_list.extend(self.kv.slot_mapping(sid, range(len(p))))
            cu.append(cu[-1] + len(p))
        dev = self.device
        positions = torch.tensor(pos_list, dtype=torch.int32, device=dev)
        slots = torch.tensor(slot_list, dtype=torch.int32, device=dev)
        cu_t = torch.tensor(cu, dtype=torch.int32, device=dev)
        max_len = max(lens)

        if self.rank == 0:
            x = torch.tensor(ids_list, dtype=torch.int64, device=dev)
        else:
            x = self._recv((T, self.spec.hidden_size), self.dtype, self.rank - 1)
        hidden = self.runner.forward_prefill(x, positions, slots, cu_t, max_len)
        next_ids = None
        if self.rank < self.world - 1:
            self._send(hidden, self.rank + 1)
        else:
            last_rows = torch.tensor([c - 1 for c in cu[1:]], dtype=torch.int64,
                                     device=dev)
            logits = self.runner.lm_head(hidden[last_rows])
            next_ids = sample(logits, sampling).cpu()
        return self._bcast_ids(next_ids, B)

    @torch.no_grad()
    def decode_step(
        self,
        ids: torch.Tensor,
        sampling: Optional[SamplingParams] = None,
    ) -> torch.Tensor:
        """One token for every active sequence. `ids` is the previous step's
        output (all ranks hold it). Returns next ids [B] on every rank."""
        sampling = sampling or SamplingParams(greedy=True)
        B = len(self._seqs)
        dev = self.device
        positions = torch.tensor(self._lens, dtype=torch.int32, device=dev)
        for i, sid in enumerate(self._seqs):
            self.kv.extend_seq(sid, self._lens[i] + 1)
            self._lens[i] += 1
        lens_t = torch.tensor(self._lens, dtype=torch.int32, device=dev)
        bt = self.kv.block_table(self._seqs)
        slots = decode_slot_mapping(bt, positions, self.kv.block_size)

        if self.rank == 0:
            x = ids.to(dev)
        else:
            x = self._recv((B, self.spec.hidden_size), self.dtype, self.rank - 1)
        hidden = self.runner.forward_decode(x, positions, slots, bt, lens_t)
        next_ids = None
        if self.rank < self.world - 1:
            self._send(hidden, self.rank + 1)
        else:
            logits = self.runner.lm_head(hidden)
            next_ids = sample(logits, sampling).cpu()
        return self._bcast_ids(next_ids, B)

    @torch.no_grad()
    def generate(
        self,
        prompts: Sequence[Sequence[int]],
        max_new_tokens: int,
        sampling: Optional[SamplingParams] = None,
    ) -> List[List[int]]:
        """Greedy/sampled generation; every rank returns the same tokens."""
        ids = self.prefill(prompts, sampling)
        outs = [[int(t)] for t in ids]
        for _ in range(max_new_tokens - 1):
            ids = self.decode_step(ids, sampling)
            for o, t in zip(outs, ids):
                o.append(int(t))
        return outs

    def reset(self) -> None:
        for sid in self._seqs:
            self.kv.free_seq(sid)
        self._seqs, self._lens = [], []
