"""Runner: executes the transformer layer stack with bee2bee_amd.ops.

This is the replacement for the reference's `transformers.generate()` hot
loop (bee2bee/hf.py:84-108): explicit prefill/decode forwards over fused
weights, paged KV, and the HIP kernel set. Projection GEMMs go through
F.linear (hipBLASLt); everything fused is ops.* (HIP on GPU).

Pipeline parallelism: a Runner can own a contiguous layer range; the first
stage embeds tokens, the last stage applies the final norm + lm_head
(parallel/pp.py moves the hidden states between stages over RCCL).
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch
import torch.nn.functional as F

from .. import ops
from ..models.spec import ModelSpec
from ..models.weights import ModelWeights
from .kv import PagedKV


class Runner:
    def __init__(
        self,
        spec: ModelSpec,
        weights: ModelWeights,
        kv: PagedKV,
        device: torch.device,
        dtype: torch.dtype = torch.bfloat16,
        layer_range: Optional[Tuple[int, int]] = None,
    ) -> None:
        self.spec = spec
        self.weights = weights
        self.kv = kv
        self.device = device
        self.dtype = dtype
        self.layer_lo, self.layer_hi = layer_range or (0, spec.n_layers)
        self.scale = spec.head_dim**-0.5
        # optional expert parallelism (parallel/ep.py); when set, MoE layers
        # dispatch tokens over RCCL all-to-all to the expert owners
        self.ep = None
        # optional tensor parallelism (parallel/tp.py): weights are a
        # head/intermediate shard and the two row-parallel projections
        # all-reduce their partial outputs over RCCL
        self.tp_group = None
        self.decode_attn_fn = None  # CP hook (parallel/cp.py)
        cos, sin = ops.rope_tables(
            spec.max_seq_len, spec.head_dim, spec.rope_theta, device,
            scaling=spec.rope_scaling,
        )
        self.rope_cos, self.rope_sin = cos, sin

    @property
    def is_first_stage(self) -> bool:
        return self.layer_lo == 0

    @property
    def is_last_stage(self) -> bool:
        return self.layer_hi == self.spec.n_layers

    # ------------------------------------------------------------- forwards

    def embed(self, input_ids: torch.Tensor,
              positions: Optional[torch.Tensor] = None) -> torch.Tensor:
        h = F.embedding(input_ids, self.weights.embed)
        if self.spec.pos_type == "learned" and positions is not None:
            h = h + F.embedding(positions.long(), self.weights.pos_embed)
        return h

    def _norm(self, x, w, b):
        if self.spec.norm_type == "layernorm":
            return ops.layernorm(x, w, b, self.spec.rms_eps)
        return ops.rmsnorm(x, w, self.spec.rms_eps)

    def _fused_add_norm(self, x, resid, w, b):
        if self.spec.norm_type == "layernorm":
            return ops.fused_add_layernorm(x, resid, w, b, self.spec.rms_eps)
        return ops.fused_add_rmsnorm(x, resid, w, self.spec.rms_eps)

    def lm_head(self, hidden: torch.Tensor) -> torch.Tensor:
        normed = self._norm(hidden, self.weights.final_norm,
                            self.weights.final_norm_bias)
        return F.linear(normed, self.weights.lm_head)

    def _layer(
        self,
        layer_idx: int,
        hidden: torch.Tensor,
        residual: Optional[torch.Tensor],
        positions: torch.Tensor,
        slot_mapping: torch.Tensor,
        attn_fn,
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        s = self.spec
        lw = self.weights.layers[layer_idx]
        if residual is None:
            residual = hidden
            normed = self._norm(hidden, lw.attn_norm, lw.attn_norm_bias)
        else:
            normed, residual = self._fused_add_norm(
                hidden, residual, lw.attn_norm, lw.attn_norm_bias
            )

        qkv = F.linear(normed, lw.wqkv)  # column-parallel under TP
        if lw.wqkv_bias is not None:  # Qwen2-style attention bias
            qkv = qkv + lw.wqkv_bias
        T = qkv.shape[0]
        q, k, v = qkv.split([s.q_size, s.kv_size, s.kv_size], dim=-1)
        q = q.view(T, s.n_heads, s.head_dim)
        k = k.view(T, s.n_kv_heads, s.head_dim)
        v = v.view(T, s.n_kv_heads, s.head_dim)
        if s.pos_type == "rope":
            ops.rope_inplace(q, k, positions, self.rope_cos, self.rope_sin)
        k_cache, v_cache = self.kv.layer(layer_idx)
        ops.kv_cache_store(k, v, k_cache, v_cache, slot_mapping)
        attn_out = attn_fn(layer_idx, q, k, v, k_cache, v_cache)
        attn_out = F.linear(attn_out.reshape(T, s.q_size), lw.wo, lw.wo_bias)
        if self.tp_group is not None:
            # row-parallel o-projection: sum the per-shard partials
            import torch.distributed as dist

            dist.all_reduce(attn_out, group=self.tp_group)

        normed, residual = self._fused_add_norm(
            attn_out, residual, lw.mlp_norm, lw.mlp_norm_bias
        )
        if s.is_moe:
            mlp_out = self._moe_mlp(lw, normed)
        elif s.act_type == "gelu":
            fc = F.linear(normed, lw.w_gate_up, lw.w_gate_up_bias)
            mlp_out = F.linear(ops.gelu(fc), lw.w_down, lw.w_down_bias)
        else:
            gate_up = F.linear(normed, lw.w_gate_up)
            mlp_out = F.linear(ops.swiglu(gate_up), lw.w_down)
            if self.tp_group is not None:
                # row-parallel down-projection
                import torch.distributed as dist

                dist.all_reduce(mlp_out, group=self.tp_group)
        return mlp_out, residual

    # below this many tokens the MoE layer runs all experts densely: at
    # decode batch sizes every expert's weights stream from HBM anyway
    # (bandwidth-bound), a batched GEMM reads them exactly once with full
    # MFMA efficiency, and every shape is static -> hipGraph-capturable
    # (the per-expert gather loop was 768 tiny launches per Mixtral step)
    MOE_DENSE_MAX_TOKENS = 256
    # above the dense threshold: sort token-slots by expert and run the
    # grouped-GEMM HIP kernel per projection — weights read once per expert,
    # flops proportional to routed tokens, segment sizes stay on device
    # (static shapes -> graph-capturable, no host sync). The kernel
    # accumulates 128 rows per W pass, so it is the default while the
    # AVERAGE segment fits one pass (T*k <= E*128); bigger batches re-read
    # W per extra pass and the padded bmm wins — those sizes only occur on
    # the (uncaptured) prefill path, where the host sync is harmless.
    MOE_GROUPED_ROWS = 128
    MOE_BMM_MAX_TOKENS = 100_000

    @classmethod
    def moe_grouped_max_tokens(cls, spec) -> int:
        return cls.MOE_GROUPED_ROWS * spec.n_experts // spec.top_k_experts

    @staticmethod
    def moe_grouped_aligned(spec) -> bool:
        """grouped_gemm kernel dim constraints (N%64, K%128) for both
        projections: gate_up [2I, H] and down [H, I]."""
        h, i = spec.hidden_size, spec.intermediate_size
        return h % 128 == 0 and i % 128 == 0

    @classmethod
    def moe_graph_capturable(cls, spec, max_tokens: int) -> bool:
        """True when every decode bucket up to max_tokens takes a
        capture-safe MoE path (dense or grouped — never padded bmm)."""
        if max_tokens <= cls.MOE_DENSE_MAX_TOKENS:
            return True
        return (cls.moe_grouped_aligned(spec)
                and max_tokens <= cls.moe_grouped_max_tokens(spec))

    def _moe_mlp(self, lw, x: torch.Tensor) -> torch.Tensor:
        """Top-k expert MLP: dense all-experts bmm for decode-sized batches,
        per-expert gather/GEMM/scatter for prefill-sized ones."""
        s = self.spec
        if self.ep is not None:
            if self.weights.expert_range is not None:
                # weights hold only this rank's expert shard (EP sharding)
                w_gu, w_dn = lw.moe_w_gate_up, lw.moe_w_down
            else:
                lo = self.ep.e_lo
                hi = lo + self.ep.local_e
                w_gu, w_dn = lw.moe_w_gate_up[lo:hi], lw.moe_w_down[lo:hi]
            return self.ep.forward(x, lw.moe_gate, w_gu, w_dn)
        logits = F.linear(x, lw.moe_gate)
        weights, idx = ops.moe_topk_gate(logits, s.top_k_experts)  # [T,k]
        T, H = x.shape
        E, I = s.n_experts, s.intermediate_size

        if T <= self.MOE_DENSE_MAX_TOKENS:
            x_e = x.unsqueeze(0).expand(E, T, H)
            gu = torch.bmm(x_e, lw.moe_w_gate_up.transpose(1, 2))  # [E,T,2I]
            act = ops.swiglu(gu.reshape(E * T, 2 * I)).view(E, T, I)
            y = torch.bmm(act, lw.moe_w_down.transpose(1, 2))  # [E,T,H]
            wfull = torch.zeros(T, E, dtype=torch.float32, device=x.device)
            wfull.scatter_(1, idx.long(), weights)
            out = torch.einsum("eth,te->th", y.float(), wfull)
            return out.to(x.dtype)

        import os as _os

        use_grouped = (
            x.device.type == "cuda"
            and self.moe_grouped_aligned(s)
            and T <= self.moe_grouped_max_tokens(s)
            and _os.environ.get("BEE2BEE_MOE_BMM") != "1"
        )
        # the padded bmm stays CPU-only (and opt-in for A/B measurement):
        # hipBLASLt's batched TN bf16 kernel memory-faults on gfx950 once
        # the padded M reaches ~1K (repro: scripts/debug_moe_t4096.py, C
        # 1049 AND 1056) — GPU sizes above the grouped range take the
        # per-expert GEMM loop below, whose plain TN GEMMs are solid
        use_bmm = (x.device.type != "cuda"
                   or _os.environ.get("BEE2BEE_MOE_BMM") == "1")
        if T <= self.MOE_BMM_MAX_TOKENS and (use_grouped or use_bmm):
            # counting sort from one-hot cumsums: every op here (one_hot,
            # cumsum, gather, index_copy) is hipGraph-capture-safe —
            # torch.bincount/argsort are not
            k = s.top_k_experts
            S = T * k
            flat_e = idx.reshape(-1).to(torch.int64)  # slot s -> expert
            oh = F.one_hot(flat_e, E)  # [S, E]
            counts = oh.sum(0)
            offs_excl = counts.cumsum(0) - counts
            rank = (oh.cumsum(0) - oh).gather(1, flat_e.unsqueeze(1)).squeeze(1)
            pos = (offs_excl[flat_e] + rank).to(torch.int64)  # dest row
            tok = torch.arange(S, device=x.device, dtype=torch.int64) // k
            if use_grouped:
                # grouped-GEMM HIP kernel (default in this range): no host
                # sync, static shapes -> the whole MoE decode step hipGraph-
                # captures; measured 3.8 TB/s of W per projection
                offsets = torch.zeros(E + 1, dtype=torch.int32, device=x.device)
                offsets[1:] = counts.cumsum(0).to(torch.int32)
                x_sorted = torch.empty(S, H, dtype=x.dtype, device=x.device)
                x_sorted.index_copy_(0, pos, x[tok])
                gu = ops.grouped_gemm(x_sorted, lw.moe_w_gate_up, offsets)
                act = ops.swiglu(gu)
                y = ops.grouped_gemm(act, lw.moe_w_down, offsets)  # [S, H]
                contrib = y[pos].float()  # back to slot order
            else:
                # padded per-expert bmm: weights read once per expert, flops
                # proportional to max routed count; counts.max() is one host
                # sync per MoE layer (so no hipGraph above the dense range).
                # M aligns up to 16: odd M (e.g. 1049) faults hipBLASLt's
                # TN bf16 batched kernel on gfx950 (repro:
                # scripts/debug_moe_t4096.py; <2% padded flops)
                C = (int(counts.max()) + 15) & ~15
                padded = torch.zeros(E, C, H, dtype=x.dtype, device=x.device)
                padded[flat_e, rank] = x[tok]
                gu = torch.bmm(padded, lw.moe_w_gate_up.transpose(1, 2))
                act = ops.swiglu(gu.reshape(E * C, 2 * I)).view(E, C, I)
                y = torch.bmm(act, lw.moe_w_down.transpose(1, 2))  # [E,C,H]
                contrib = y[flat_e, rank].float()  # [S, H] slot order
            out = torch.zeros(T, H, dtype=torch.float32, device=x.device)
            out.index_add_(
                0, tok, contrib * weights.reshape(-1).unsqueeze(-1)
            )
            return out.to(x.dtype)

        out = torch.zeros_like(x, dtype=torch.float32)
        for e in range(E):
            mask = idx == e  # [T, k]
            if not bool(mask.any()):
                continue
            tok, kslot = mask.nonzero(as_tuple=True)
            xe = x[tok]
            ge = F.linear(xe, lw.moe_w_gate_up[e])
            ye = F.linear(ops.swiglu(ge), lw.moe_w_down[e])
            out.index_add_(0, tok, ye.float() * weights[tok, kslot, None])
        return out.to(x.dtype)

    def _run_layers(
        self,
        hidden: torch.Tensor,
        positions: torch.Tensor,
        slot_mapping: torch.Tensor,
        attn_fn,
    ) -> torch.Tensor:
        residual = None
        for layer_idx in range(self.layer_lo, self.layer_hi):
            hidden, residual = self._layer(
                layer_idx, hidden, residual, positions, slot_mapping, attn_fn
            )
        return hidden + residual if residual is not None else hidden

    def forward_prefill(
        self,
        input_ids_or_hidden: torch.Tensor,
        positions: torch.Tensor,
        slot_mapping: torch.Tensor,
        cu_seqlens: torch.Tensor,
        max_seqlen: int,
        block_table: Optional[torch.Tensor] = None,
        seq_lens: Optional[torch.Tensor] = None,
        query_lens: Optional[torch.Tensor] = None,
    ) -> torch.Tensor:
        """Full- or chunked-prompt forward. Returns final hidden [T, H]
        (call lm_head on the rows you need). When seq_lens/query_lens are
        given the chunk attends to the paged history (chunked prefill)."""
        chunked = seq_lens is not None

        def attn(layer_idx, q, k, v, k_cache, v_cache):
            if chunked:
                # chunked prefill: this chunk's K/V were just stored, so the
                # kernel reads K/V exclusively from the paged cache and masks
                # causally at absolute positions (HIP attn_prefill_mfma
                # <HD, PAGED=true>; reference attn_decode_with_history on CPU)
                return ops.attn_prefill_paged(
                    q, k_cache, v_cache, block_table, seq_lens, cu_seqlens,
                    max_seqlen, self.scale,
                )
            return ops.attn_prefill(q, k, v, cu_seqlens, max_seqlen, self.scale)

        hidden = (
            self.embed(input_ids_or_hidden, positions)
            if self.is_first_stage
            else input_ids_or_hidden
        )
        return self._run_layers(hidden, positions, slot_mapping, attn)

    def forward_decode(
        self,
        input_ids_or_hidden: torch.Tensor,
        positions: torch.Tensor,
        slot_mapping: torch.Tensor,
        block_table: torch.Tensor,
        seq_lens: torch.Tensor,
    ) -> torch.Tensor:
        """One-token-per-sequence step. input [B] ids (or [B, H] hidden for
        later pipeline stages); returns final hidden [B, H]."""

        def attn(layer_idx, q, k, v, k_cache, v_cache):
            if self.decode_attn_fn is not None:
                # injected attention (context parallelism: parallel/cp.py
                # merges per-rank partials over the local KV shard)
                return self.decode_attn_fn(
                    q, k_cache, v_cache, block_table, seq_lens, self.scale
                )
            return ops.attn_decode(
                q, k_cache, v_cache, block_table, seq_lens, self.scale
            )

        hidden = (
            self.embed(input_ids_or_hidden, positions)
            if self.is_first_stage
            else input_ids_or_hidden
        )
        return self._run_layers(hidden, positions, slot_mapping, attn)
