"""Context parallelism: one sequence's KV history sharded across ranks.

288 GB of HBM3E per MI355X already serves 32k contexts at flagship
batches, so CP is the BEYOND-one-GPU long-context extension: rank r holds
a contiguous page range of each sequence's history, runs the normal paged
decode-attention kernels over its local shard, and the ranks exchange
only the tiny flash-decoding merge state — per (sequence, query head):
the normalized partial output o_r [hd] plus (m_r, l_r), the max scaled
score and its sum-exp. The merge is exact (same math the decode kernel
uses chunk-to-chunk, attn_decode.hip FOLD):

    m  = max_r m_r;   w_r = exp(m_r - m);   L = sum_r w_r l_r
    O  = sum_r w_r l_r o_r / L

One all_gather of [B, nq, hd+2] per layer per step — bytes proportional
to batch x heads, NOT context length, so xGMI cost is flat while the
servable context scales with world_size x 288 GB.

The reference has no long-context story at all (sequence
length bounded by whatever HF supports); this module is MI355X-native
design, exact-match tested against single-rank attention on gloo
world 2 and on the HIP kernels via tests/test_ops_gpu.py.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import torch
import torch.distributed as dist

from .. import ops


def shard_pages(n_pages: int, world: int, rank: int) -> Tuple[int, int]:
    """Contiguous page range [lo, hi) owned by `rank` of `world`.
    Earlier ranks take the remainder so token order is preserved."""
    base = n_pages // world
    extra = n_pages % world
    lo = rank * base + min(rank, extra)
    hi = lo + base + (1 if rank < extra else 0)
    return lo, hi


def local_lens(seq_lens: torch.Tensor, page: int, world: int,
               rank: int) -> torch.Tensor:
    """Tokens of each sequence that fall inside this rank's page shard."""
    out = torch.zeros_like(seq_lens)
    for i, L in enumerate(seq_lens.tolist()):
        n_pages = -(-L // page) if L else 0
        lo, hi = shard_pages(n_pages, world, rank)
        tok_lo, tok_hi = lo * page, min(hi * page, L)
        out[i] = max(0, tok_hi - tok_lo)
    return out


def merge_partials(
    outs: Sequence[torch.Tensor],   # per rank [B, nq, hd] (normalized)
    mls: Sequence[torch.Tensor],    # per rank [B, nq, 2] = (m, l)
) -> torch.Tensor:
    """Exact flash merge of per-rank attention partials."""
    m_stack = torch.stack([ml[..., 0] for ml in mls])   # [R, B, nq]
    l_stack = torch.stack([ml[..., 1] for ml in mls])
    o_stack = torch.stack([o.float() for o in outs])    # [R, B, nq, hd]
    m = m_stack.amax(dim=0)                             # [B, nq]
    w = torch.exp(m_stack - m.unsqueeze(0)) * l_stack   # [R, B, nq]
    # ranks whose shard held zero tokens carry l=0 -> zero weight
    denom = w.sum(dim=0).clamp_min(1e-38)
    merged = (o_stack * w.unsqueeze(-1)).sum(dim=0) / denom.unsqueeze(-1)
    return merged.to(outs[0].dtype)


@torch.no_grad()
def cp_attn_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    local_block_table: torch.Tensor,
    local_seq_lens: torch.Tensor,
    scale: float,
    group: Optional[dist.ProcessGroup] = None,
) -> torch.Tensor:
    """Decode attention over a context sharded across the group's ranks.

    Every rank passes its LOCAL page table / lengths (build with
    shard_pages/local_lens); returns the full-context attention output,
    identical on all ranks."""
    assert dist.is_initialized(), "init the process group first"
    world = dist.get_world_size(group)
    out, ml = ops.attn_decode_lse(
        q, k_cache, v_cache, local_block_table, local_seq_lens, scale)
    # a rank with an empty shard produces garbage rows; zero their weight
    empty = (local_seq_lens == 0)
    if bool(empty.any()):
        ml = ml.clone()
        ml[empty, :, 0] = -1e30
        ml[empty, :, 1] = 0.0
    payload = torch.cat([out.float(), ml], dim=-1).contiguous()  # [B,nq,hd+2]
    gathered: List[torch.Tensor] = [
        torch.empty_like(payload) for _ in range(world)
    ]
    dist.all_gather(gathered, payload, group=group)
    hd = q.shape[-1]
    return merge_partials(
        [g[..., :hd] for g in gathered],
        [g[..., hd:] for g in gathered],
    ).to(q.dtype)
