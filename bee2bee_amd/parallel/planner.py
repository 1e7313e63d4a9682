"""Layer-shard planner: split a model's layer stack into contiguous stages
sized to each peer's HBM budget.

This generalizes the reference's two sharding primitives — the
`build_distilbert_partial` [start, end) layer ranges (bee2bee/hf.py:180-205)
and pieces.py content shards — into pipeline-stage planning: the first
stage owns the embedding, the last owns the final norm + lm_head, and
middle stages get contiguous layer ranges proportional to available memory.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional, Tuple

from ..models.spec import ModelSpec


@dataclass
class StagePlan:
    rank: int
    layer_range: Tuple[int, int]
    has_embed: bool
    has_head: bool
    est_bytes: int


def _layer_bytes(spec: ModelSpec, dtype_bytes: int = 2) -> int:
    h, i = spec.hidden_size, spec.intermediate_size
    attn = h * (spec.q_size + 2 * spec.kv_size) + spec.q_size * h
    mlp = (
        spec.n_experts * 3 * h * i + h * spec.n_experts
        if spec.is_moe
        else 3 * h * i
    )
    return (attn + mlp + 2 * h) * dtype_bytes


def _embed_bytes(spec: ModelSpec, dtype_bytes: int = 2) -> int:
    return spec.vocab_size * spec.hidden_size * dtype_bytes


def plan_stages(
    spec: ModelSpec,
    n_stages: int,
    mem_budgets: Optional[List[int]] = None,
    dtype_bytes: int = 2,
) -> List[StagePlan]:
    """Contiguous layer ranges balanced by weight bytes.

    mem_budgets (bytes per peer) weight the split; equal budgets (the
    default, e.g. 8x MI355X with 288 GB each) give an even split with the
    embed/head bytes charged to the first/last stage."""
    if n_stages < 1 or n_stages > spec.n_layers:
        raise ValueError(f"n_stages must be in [1, {spec.n_layers}]")
    budgets = mem_budgets or [1] * n_stages
    if len(budgets) != n_stages:
        raise ValueError("one memory budget per stage")
    lb = _layer_bytes(spec, dtype_bytes)
    eb = _embed_bytes(spec, dtype_bytes)
    # effective capacity: subtract embed from stage 0 and head from stage -1
    eff = [float(b) for b in budgets]
    total_eff = sum(eff)
    # ideal fractional layer counts proportional to budget
    counts = [spec.n_layers * e / total_eff for e in eff]
    # round while preserving the sum
    out_counts = [max(1, int(c)) for c in counts]
    while sum(out_counts) < spec.n_layers:
        fracs = [c - oc for c, oc in zip(counts, out_counts)]
        out_counts[fracs.index(max(fracs))] += 1
    while sum(out_counts) > spec.n_layers:
        fracs = [oc - c for c, oc in zip(counts, out_counts)]
        i = fracs.index(max(fracs))
        if out_counts[i] > 1:
            out_counts[i] -= 1
        else:
            out_counts[out_counts.index(max(out_counts))] -= 1
    plans = []
    lo = 0
    for r in range(n_stages):
        hi = lo + out_counts[r]
        est = out_counts[r] * lb
        if r == 0:
            est += eb
        if r == n_stages - 1:
            est += eb if not spec.tie_embeddings else 0
            est += spec.hidden_size * dtype_bytes
        plans.append(
            StagePlan(
                rank=r,
                layer_range=(lo, hi),
                has_embed=(r == 0),
                has_head=(r == n_stages - 1),
                est_bytes=est,
            )
        )
        lo = hi
    assert lo == spec.n_layers
    return plans
