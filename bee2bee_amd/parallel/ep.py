"""Expert parallelism: Mixtral-style MoE layers sharded across GPUs with
RCCL all-to-all token exchange (BASELINE config 5).

Each rank holds E/world experts. Per MoE layer:
  1. gate locally (gate weights replicated, tiny)
  2. top-k routing -> sort token-slots by destination rank
  3. dist.all_to_all_single ships the hidden states to expert owners —
     on an 8-GPU MI355X node this uses all 7 xGMI links of each GPU
     concurrently (all-to-all is the one collective that saturates the
     point-to-point mesh; ring collectives are single-link-bound)
  4. local expert SwiGLU MLPs on the received tokens
  5. all-to-all back + weighted combine

CPU-tested with gloo world_size 2 against the single-process MoE reference
(fp32 exact up to summation order).
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.distributed as dist
import torch.nn.functional as F

from .. import ops


class ExpertParallelMoE:
    """Stateless helper bound to (group, local expert weights)."""

    def __init__(
        self,
        n_experts: int,
        top_k: int,
        group: Optional[dist.ProcessGroup] = None,
    ) -> None:
        assert dist.is_initialized()
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        assert n_experts % self.world == 0, "experts must divide evenly"
        self.n_experts = n_experts
        self.top_k = top_k
        self.local_e = n_experts // self.world
        self.e_lo = self.rank * self.local_e

    @torch.no_grad()
    def forward(
        self,
        x: torch.Tensor,          # [T, H]
        gate_w: torch.Tensor,     # [E, H] (replicated)
        w_gate_up: torch.Tensor,  # [localE, 2I, H]
        w_down: torch.Tensor,     # [localE, H, I]
    ) -> torch.Tensor:
        T, H = x.shape
        logits = F.linear(x, gate_w)
        weights, idx = ops.moe_topk_gate(logits, self.top_k)  # [T,k]
        flat_expert = idx.reshape(-1).to(torch.int64)         # [T*k]
        dest_rank = flat_expert // self.local_e

        # sort slots by destination rank (stable -> deterministic combine)
        order = torch.argsort(dest_rank, stable=True)
        inv_order = torch.empty_like(order)
        inv_order[order] = torch.arange(order.numel(), device=order.device)
        send_x = x.repeat_interleave(self.top_k, dim=0)[order]
        send_e = flat_expert[order]

        send_counts = torch.bincount(dest_rank, minlength=self.world)
        recv_counts = torch.empty_like(send_counts)
        dist.all_to_all_single(recv_counts, send_counts, group=self.group)
        in_splits = send_counts.tolist()
        out_splits = recv_counts.tolist()
        n_recv = sum(out_splits)

        recv_x = torch.empty(n_recv, H, dtype=x.dtype, device=x.device)
        dist.all_to_all_single(
            recv_x, send_x.contiguous(),
            output_split_sizes=out_splits, input_split_sizes=in_splits,
            group=self.group,
        )
        recv_e = torch.empty(n_recv, dtype=send_e.dtype, device=x.device)
        dist.all_to_all_single(
            recv_e, send_e.contiguous(),
            output_split_sizes=out_splits, input_split_sizes=in_splits,
            group=self.group,
        )

        # local expert MLPs on the received tokens
        out_local = torch.zeros(n_recv, H, dtype=torch.float32, device=x.device)
        local_ids = recv_e - self.e_lo
        for le in range(self.local_e):
            mask = local_ids == le
            if not bool(mask.any()):
                continue
            rows = mask.nonzero(as_tuple=True)[0]
            xe = recv_x[rows]
            ge = F.linear(xe, w_gate_up[le])
            ye = F.linear(ops.swiglu(ge), w_down[le])
            out_local[rows] = ye.float()

        # ship results back and undo the permutation
        back = torch.empty(
            order.numel(), H, dtype=torch.float32, device=x.device
        )
        dist.all_to_all_single(
            back, out_local.contiguous(),
            output_split_sizes=in_splits, input_split_sizes=out_splits,
            group=self.group,
        )
        contrib = back[inv_order].reshape(T, self.top_k, H)
        out = (contrib * weights.unsqueeze(-1)).sum(dim=1)
        return out.to(x.dtype)
