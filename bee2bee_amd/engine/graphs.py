"""hipGraph-captured decode steps.

The decode step is a chain of ~100 small kernel launches per layer-stack
pass; at batch<32 launch overhead is a visible fraction of the step. We
capture the whole step (embed → layers → lm_head → slot math) per batch-size
bucket with torch.cuda.CUDAGraph (hipGraph on ROCm) into static buffers and
replay it. Buffers are updated in-place with device ops between replays —
no host sync in the loop.

Capture constraints honored: the paged-KV pool is preallocated (kv.py), all
shapes are static per bucket, the HIP kernels allocate nothing and never
sync, sampling stays outside the graph (RNG).
"""
from __future__ import annotations

import logging
from typing import Dict, Optional

import torch

logger = logging.getLogger("bee2bee_amd.engine")


def decode_slot_mapping(
    block_table: torch.Tensor, positions: torch.Tensor, block_size: int
) -> torch.Tensor:
    """slot[b] = block_table[b][pos//bs]*bs + pos%bs, computed on device."""
    blk_idx = torch.div(positions, block_size, rounding_mode="floor")
    blk = block_table.gather(1, blk_idx.unsqueeze(1).long()).squeeze(1)
    return blk * block_size + positions % block_size


class DecodeGraphs:
    """Per-batch-bucket captured decode steps over shared static buffers."""

    def __init__(self, runner, max_batch: int, max_blocks_per_seq: int) -> None:
        self.runner = runner
        self.device = runner.device
        self.max_batch = max_batch
        self.width = max_blocks_per_seq
        dev = self.device
        self.input_ids = torch.zeros(max_batch, dtype=torch.int64, device=dev)
        self.positions = torch.zeros(max_batch, dtype=torch.int32, device=dev)
        self.seq_lens = torch.ones(max_batch, dtype=torch.int32, device=dev)
        self.block_table = torch.zeros(
            max_batch, self.width, dtype=torch.int32, device=dev
        )
        self.logits: Dict[int, torch.Tensor] = {}
        self._graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._pool = None

    def buckets(self) -> list:
        out, b = [], 1
        while b < self.max_batch:
            out.append(b)
            b *= 2
        out.append(self.max_batch)
        return out

    def bucket_for(self, batch: int) -> int:
        for b in self.buckets():
            if batch <= b:
                return b
        return self.max_batch

    def _run(self, B: int) -> torch.Tensor:
        ids = self.input_ids[:B]
        pos = self.positions[:B]
        bt = self.block_table[:B]
        lens = self.seq_lens[:B]
        slots = decode_slot_mapping(bt, pos, self.runner.kv.block_size)
        hidden = self.runner.forward_decode(ids, pos, slots, bt, lens)
        return self.runner.lm_head(hidden)

    def _capture(self, B: int) -> None:
        torch.cuda.synchronize()
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._run(B)
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        if self._pool is None:
            with torch.cuda.graph(g):
                self.logits[B] = self._run(B)
            self._pool = g.pool()
        else:
            with torch.cuda.graph(g, pool=self._pool):
                self.logits[B] = self._run(B)
        self._graphs[B] = g
        logger.info("captured decode graph for batch bucket %d", B)

    def run(self, batch: int) -> torch.Tensor:
        """Replay (capturing on first use) and return logits [bucket, V].
        Caller must have filled rows [:batch] of the static buffers and
        padded rows [batch:bucket] with safe scratch values."""
        B = self.bucket_for(batch)
        if B not in self._graphs:
            self._capture(B)
        # ALWAYS consume a replay, never the capture run itself: library
        # GEMMs may pick a different algorithm while capturing, so replayed
        # steps would not be bit-identical with the capture step otherwise.
        self._graphs[B].replay()
        return self.logits[B]
