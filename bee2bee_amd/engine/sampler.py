"""Token sampling: greedy, temperature, top-k/top-p — device-side torch ops
on [B, V] logits (the per-step cost is dwarfed by the lm_head GEMM).

Matches the reference generation knobs (bee2bee/hf.py:94-103: temperature,
top_p 0.95, repetition_penalty 1.15)."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

import torch


@dataclass
class SamplingParams:
    temperature: float = 0.7
    top_p: float = 0.95
    top_k: int = 64
    repetition_penalty: float = 1.0
    greedy: bool = False

    @classmethod
    def from_request(cls, temperature: Optional[float]) -> "SamplingParams":
        t = 0.7 if temperature is None else float(temperature)
        if t <= 1e-4:
            return cls(temperature=1.0, greedy=True)
        return cls(temperature=t)


def apply_repetition_penalty(
    logits: torch.Tensor, prev_ids: torch.Tensor, penalty: float
) -> torch.Tensor:
    """prev_ids [B, L] (pad with -1); penalize already-emitted tokens."""
    if penalty == 1.0:
        return logits
    B = logits.shape[0]
    for b in range(B):
        ids = prev_ids[b]
        ids = ids[ids >= 0]
        if ids.numel() == 0:
            continue
        row = logits[b]
        vals = row[ids]
        row[ids] = torch.where(vals > 0, vals / penalty, vals * penalty)
    return logits


def sample(
    logits: torch.Tensor,
    params: SamplingParams,
    generator: Optional[torch.Generator] = None,
) -> torch.Tensor:
    """logits [B, V] -> next token ids [B] (int64, on device)."""
    if params.greedy:
        return torch.argmax(logits, dim=-1)
    logits = logits.float() / max(params.temperature, 1e-5)
    if params.top_k > 0 and params.top_k < logits.shape[-1]:
        vals, idx = torch.topk(logits, params.top_k, dim=-1)
    else:
        vals, idx = torch.sort(logits, dim=-1, descending=True)
    probs = torch.softmax(vals, dim=-1)
    if 0.0 < params.top_p < 1.0:
        cum = torch.cumsum(probs, dim=-1)
        mask = cum - probs > params.top_p  # keep first token crossing top_p
        probs = probs.masked_fill(mask, 0.0)
        probs = probs / probs.sum(dim=-1, keepdim=True)
    choice = torch.multinomial(probs, 1, generator=generator)
    return idx.gather(-1, choice).squeeze(-1)
