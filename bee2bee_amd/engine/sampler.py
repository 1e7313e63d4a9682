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
    def from_request(
        cls,
        temperature: Optional[float],
        top_p: Optional[float] = None,
        top_k: Optional[int] = None,
        repetition_penalty: Optional[float] = None,
    ) -> "SamplingParams":
        """Wire/API request knobs -> params, with the REFERENCE's generation
        defaults (bee2bee/hf.py:94-103): temperature 0.7, top_p 0.95,
        repetition_penalty 1.15. temperature <= 1e-4 selects greedy (the
        penalty still applies — matching do_sample=False + penalty in HF)."""
        import math

        t = 0.7 if temperature is None else float(temperature)
        if not math.isfinite(t):
            t = 0.7  # NaN/inf from hostile JSON: fall back to the default
        p = cls(temperature=max(t, 1e-4), greedy=t <= 1e-4)
        if top_p is not None and math.isfinite(float(top_p)):
            p.top_p = float(top_p)
        if top_k is not None:
            p.top_k = max(0, int(top_k))
        rp = (1.15 if repetition_penalty is None
              else float(repetition_penalty))
        p.repetition_penalty = rp if math.isfinite(rp) and rp > 0 else 1.15
        return p


def apply_repetition_penalty(
    logits: torch.Tensor, prev_ids: torch.Tensor, penalty: float
) -> torch.Tensor:
    """prev_ids [B, L] (pad with -1); penalize already-emitted tokens.

    Fully batched: one scatter builds a seen-token mask (pad indices land in
    a sacrificial extra column), one where applies the penalty — no host
    loop, so a large decode batch pays O(1) kernel launches."""
    if penalty == 1.0 or prev_ids.numel() == 0:
        return logits
    B, V = logits.shape
    valid = prev_ids >= 0
    idx = torch.where(valid, prev_ids, torch.full_like(prev_ids, V))
    seen = torch.zeros((B, V + 1), dtype=torch.bool, device=logits.device)
    seen.scatter_(1, idx.long(), True)
    seen = seen[:, :V]
    return torch.where(
        seen,
        torch.where(logits > 0, logits / penalty, logits * penalty),
        logits,
    )


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
