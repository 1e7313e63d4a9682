"""Distributed serving front-end for the lockstep parallel engines.

The PP/TP/EP/CP engines (pp.py, tp.py, moe_engine.py, cp.py) share one
contract: every rank constructs the engine and calls `generate(prompts,
max_new_tokens, sampling)` in lockstep with identical arguments. This
module turns that contract into a SERVABLE deployment:

  rank 0    owns the request source (mesh node / HTTP gateway / caller)
            and broadcasts each request's arguments to the group;
  ranks 1+  run `serve_follower()` — a loop that receives the broadcast
            arguments and enters the same `generate` call.

Transport for the argument broadcast is `dist.broadcast_object_list` over
the EXISTING process group — tiny payloads (token ids + knobs), so gloo or
RCCL both do; activations keep flowing over the engines' own collectives.

Launch shape (same as scripts/bench_pp.py):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 scripts/serve_parallel.py --mode tp ...

Reference counterpart: none — the reference's only distributed execution
is the embryonic DistilBERT layer-RPC (bee2bee/node.py:236-277); this is
the MI355X-native replacement at serving level.
"""
from __future__ import annotations

import logging
from dataclasses import asdict
from typing import Any, Dict, List, Optional, Sequence

import torch.distributed as dist

from ..engine.sampler import SamplingParams

logger = logging.getLogger("bee2bee_amd.parallel")

_STOP = "__lockstep_stop__"


class LockstepServer:
    """Rank-0 handle + follower loop around one lockstep engine."""

    def __init__(self, engine: Any, group: Optional[Any] = None) -> None:
        assert dist.is_initialized(), "init the process group first"
        self.engine = engine
        self.group = group
        self.rank = dist.get_rank(group)

    # ------------------------------------------------------------- rank 0

    def generate(
        self,
        prompts: Sequence[Sequence[int]],
        max_new_tokens: int,
        sampling: Optional[SamplingParams] = None,
    ) -> List[List[int]]:
        """Broadcast the request, then enter the lockstep generate."""
        assert self.rank == 0, "generate() is the rank-0 entry point"
        payload = {
            "prompts": [list(p) for p in prompts],
            "max_new_tokens": int(max_new_tokens),
            "sampling": asdict(sampling) if sampling is not None else None,
        }
        dist.broadcast_object_list([payload], src=0, group=self.group)
        return self._run(payload)

    def shutdown(self) -> None:
        """Release the followers (their serve_follower() returns)."""
        if self.rank == 0:
            dist.broadcast_object_list([_STOP], src=0, group=self.group)

    # ------------------------------------------------------------ followers

    def serve_follower(self) -> int:
        """Ranks 1+: serve broadcasts until shutdown. Returns the number of
        requests served."""
        assert self.rank != 0, "rank 0 drives; followers follow"
        served = 0
        while True:
            box: List[Any] = [None]
            dist.broadcast_object_list(box, src=0, group=self.group)
            if box[0] == _STOP:
                return served
            self._run(box[0])
            served += 1

    # -------------------------------------------------------------- shared

    def _run(self, payload: Dict[str, Any]) -> List[List[int]]:
        sampling = (SamplingParams(**payload["sampling"])
                    if payload["sampling"] else None)
        try:
            return self.engine.generate(
                payload["prompts"], payload["max_new_tokens"], sampling)
        finally:
            reset = getattr(self.engine, "reset", None)
            if callable(reset):
                reset()


def build_engine(mode: str, model: str, device: Optional[str] = None,
                 max_batch: int = 8, max_seq_len: int = 2048,
                 model_path: Optional[str] = None, seed: int = 0) -> Any:
    """The bench_pp mode table, importable (pp/tp/ep/cp -> engine)."""
    if mode == "tp":
        from .tp import TPEngine

        return TPEngine(model, device=device, max_batch=max_batch,
                        max_seq_len=max_seq_len, model_path=model_path,
                        seed=seed)
    if mode == "pp":
        from .pp import PipelineEngine

        return PipelineEngine(model, device=device, max_batch=max_batch,
                              max_seq_len=max_seq_len, model_path=model_path,
                              seed=seed)
    if mode == "ep":
        from .moe_engine import MoEEngine

        return MoEEngine(model, device=device, max_batch=max_batch,
                         max_seq_len=max_seq_len, model_path=model_path,
                         seed=seed)
    if mode == "cp":
        from .cp import CPEngine

        return CPEngine(model, device=device, max_batch=max_batch,
                        max_seq_len=max_seq_len, model_path=model_path,
                        seed=seed)
    raise ValueError(f"unknown parallel mode: {mode}")
