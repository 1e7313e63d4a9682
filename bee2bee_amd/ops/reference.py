"""Pure-PyTorch fp32 reference implementations of every engine op.

These are the numerics ground truth that the HIP kernels are tested against
(tests/test_ops_gpu.py compares each CDNA4 kernel to these at fp32), and the
CPU execution path for GPU-less environments (plumbing tests, BASELINE
config 1). On a GPU box the engine REQUIRES the HIP extension — see
ops/__init__.py dispatch.

Conventions (shared with the HIP kernels):
  x        [T, H]            activations (T = tokens in batch/chunk)
  q        [T, n_heads, hd]
  k, v     [T, n_kv, hd]
  k_cache  [n_blocks, n_kv, block_size, hd]   paged KV pool
  block_table [B, max_blocks] int32
  slot_mapping [T] int32     flat slot = block_id * block_size + offset
  cos/sin  [max_len, hd/2]   fp32 RoPE tables (host-precomputed)
RoPE is llama-style rotate_half: pair (i, i + hd/2).
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch


def rmsnorm(x: torch.Tensor, weight: torch.Tensor, eps: float) -> torch.Tensor:
    xf = x.float()
    norm = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (norm * weight.float()).to(x.dtype)


def fused_add_rmsnorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor, eps: float
) -> Tuple[torch.Tensor, torch.Tensor]:
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return rmsnorm(new_residual, weight, eps), new_residual


def layernorm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
              eps: float) -> torch.Tensor:
    xf = x.float()
    mean = xf.mean(-1, keepdim=True)
    var = xf.var(-1, unbiased=False, keepdim=True)
    out = (xf - mean) * torch.rsqrt(var + eps)
    return (out * weight.float() + bias.float()).to(x.dtype)


def fused_add_layernorm(
    x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor,
    bias: torch.Tensor, eps: float,
) -> Tuple[torch.Tensor, torch.Tensor]:
    new_residual = (x.float() + residual.float()).to(x.dtype)
    return layernorm(new_residual, weight, bias, eps), new_residual


def gelu(x: torch.Tensor) -> torch.Tensor:
    """HF gelu_new (tanh approximation) — GPT-2's activation."""
    xf = x.float()
    c = 0.7978845608028654 * (xf + 0.044715 * xf.pow(3))
    return (0.5 * xf * (1.0 + torch.tanh(c))).to(x.dtype)


def rope_tables(
    max_len: int, head_dim: int, theta: float, device, dtype=torch.float32,
    scaling: "Optional[dict]" = None,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Host-precomputed cos/sin tables. `scaling` supports the llama3
    rope_type (Llama-3.1/3.2 checkpoints): frequencies below the low-freq
    wavelength divide by `factor`, above high-freq stay, in between
    interpolate smoothly — matches transformers' llama3 rope init."""
    import math

    inv_freq = 1.0 / (
        theta ** (torch.arange(0, head_dim, 2, dtype=torch.float32) / head_dim)
    )
    if scaling and scaling.get("rope_type", scaling.get("type")) == "llama3":
        factor = float(scaling["factor"])
        lo_f = float(scaling.get("low_freq_factor", 1.0))
        hi_f = float(scaling.get("high_freq_factor", 4.0))
        orig = float(scaling.get("original_max_position_embeddings", 8192))
        low_wl = orig / lo_f
        high_wl = orig / hi_f
        wavelen = 2 * math.pi / inv_freq
        scaled = torch.where(wavelen > low_wl, inv_freq / factor, inv_freq)
        smooth = (orig / wavelen - lo_f) / (hi_f - lo_f)
        mid = (1 - smooth) * inv_freq / factor + smooth * inv_freq
        is_mid = (wavelen <= low_wl) & (wavelen >= high_wl)
        inv_freq = torch.where(is_mid, mid, scaled)
    t = torch.arange(max_len, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [max_len, hd/2]
    return freqs.cos().to(device, dtype), freqs.sin().to(device, dtype)


def rope_inplace(
    q: torch.Tensor,
    k: torch.Tensor,
    positions: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
) -> None:
    hd = q.shape[-1]
    c = cos[positions].unsqueeze(1)  # [T, 1, hd/2]
    s = sin[positions].unsqueeze(1)
    for t in (q, k):
        # .float() is identity for fp32 tensors — clone so the first-half
        # write cannot alias the second-half read
        tf = t.detach().clone().float()
        x1, x2 = tf[..., : hd // 2], tf[..., hd // 2 :]
        t[..., : hd // 2] = (x1 * c - x2 * s).to(t.dtype)
        t[..., hd // 2 :] = (x2 * c + x1 * s).to(t.dtype)


def _kv_deq(cache: torch.Tensor) -> torch.Tensor:
    """Dequantize an fp8 (OCP e4m3, uint8 storage) cache to fp32; bf16/fp32
    caches pass through as fp32."""
    if cache.dtype == torch.uint8:
        return cache.view(torch.float8_e4m3fn).float()
    return cache.float()


def _kv_quant_like(x: torch.Tensor, cache: torch.Tensor) -> torch.Tensor:
    if cache.dtype == torch.uint8:
        return x.float().to(torch.float8_e4m3fn).view(torch.uint8)
    return x.to(cache.dtype)


def kv_cache_store(
    k: torch.Tensor,
    v: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    slot_mapping: torch.Tensor,
) -> None:
    n_blocks, n_kv, block_size, hd = k_cache.shape
    blk = torch.div(slot_mapping, block_size, rounding_mode="floor").long()
    off = (slot_mapping % block_size).long()
    k_cache[blk, :, off, :] = _kv_quant_like(k, k_cache)
    v_cache[blk, :, off, :] = _kv_quant_like(v, v_cache)


def attn_decode(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_table: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
    out_ml: torch.Tensor = None,
) -> torch.Tensor:
    """Paged single-token attention, GQA. q [B, nq, hd] -> out [B, nq, hd].
    When out_ml [B, nq, 2] is given, also writes the flash merge state
    (max scaled score m, sum-exp l) per query head — the cross-rank
    exchange for context parallelism."""
    B, nq, hd = q.shape
    _nb, n_kv, block_size, _ = k_cache.shape
    group = nq // n_kv
    out = torch.empty_like(q)
    for b in range(B):
        L = int(seq_lens[b])
        if L == 0:
            # empty context (a CP rank owning no pages of this sequence):
            # zero output with (m, l) = (-inf, 0) merge state
            out[b] = 0
            if out_ml is not None:
                out_ml[b, :, 0] = -1e30
                out_ml[b, :, 1] = 0.0
            continue
        nblk = (L + block_size - 1) // block_size
        blocks = block_table[b, :nblk].long()
        keys = _kv_deq(k_cache[blocks])  # [nblk, n_kv, bs, hd]
        vals = _kv_deq(v_cache[blocks])
        keys = keys.permute(1, 0, 2, 3).reshape(n_kv, nblk * block_size, hd)[:, :L]
        vals = vals.permute(1, 0, 2, 3).reshape(n_kv, nblk * block_size, hd)[:, :L]
        qb = q[b].float().view(n_kv, group, hd)
        scores = torch.einsum("kgd,kld->kgl", qb, keys) * scale
        probs = torch.softmax(scores, dim=-1)
        ob = torch.einsum("kgl,kld->kgd", probs, vals)
        out[b] = ob.reshape(nq, hd).to(q.dtype)
        if out_ml is not None:
            m = scores.amax(dim=-1)                         # [n_kv, group]
            l = torch.exp(scores - m.unsqueeze(-1)).sum(-1)  # [n_kv, group]
            out_ml[b, :, 0] = m.reshape(nq)
            out_ml[b, :, 1] = l.reshape(nq)
    return out


def attn_prefill(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    cu_seqlens: torch.Tensor,
    max_seqlen: int,
    scale: float,
    causal: bool = True,
) -> torch.Tensor:
    """Varlen full-prompt attention, GQA. q [T, nq, hd] -> [T, nq, hd]."""
    T, nq, hd = q.shape
    n_kv = k.shape[1]
    group = nq // n_kv
    out = torch.empty_like(q)
    for i in range(cu_seqlens.numel() - 1):
        s, e = int(cu_seqlens[i]), int(cu_seqlens[i + 1])
        L = e - s
        qi = q[s:e].float().view(L, n_kv, group, hd)
        ki = k[s:e].float()
        vi = v[s:e].float()
        scores = torch.einsum("qkgd,lkd->kgql", qi, ki) * scale
        if causal:
            mask = torch.triu(
                torch.ones(L, L, dtype=torch.bool, device=q.device), diagonal=1
            )
            scores.masked_fill_(mask, float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        oi = torch.einsum("kgql,lkd->qkgd", probs, vi)
        out[s:e] = oi.reshape(L, nq, hd).to(q.dtype)
    return out


def attn_decode_with_history(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_table: torch.Tensor,
    seq_lens: torch.Tensor,
    query_lens: torch.Tensor,
    scale: float,
) -> torch.Tensor:
    """Chunked-prefill attention: multiple new query tokens per sequence
    attending to the full paged history (new tokens already stored in cache).

    q [T, nq, hd] packed by sequence; seq_lens = total length incl. the new
    chunk; query_lens = chunk length per sequence."""
    T, nq, hd = q.shape
    _nb, n_kv, block_size, _ = k_cache.shape
    group = nq // n_kv
    out = torch.empty_like(q)
    t0 = 0
    for b in range(query_lens.numel()):
        QL = int(query_lens[b])
        L = int(seq_lens[b])
        nblk = (L + block_size - 1) // block_size
        blocks = block_table[b, :nblk].long()
        keys = (
            _kv_deq(k_cache[blocks])
            .permute(1, 0, 2, 3).reshape(n_kv, nblk * block_size, hd)[:, :L]
        )
        vals = (
            _kv_deq(v_cache[blocks])
            .permute(1, 0, 2, 3).reshape(n_kv, nblk * block_size, hd)[:, :L]
        )
        qb = q[t0 : t0 + QL].float().view(QL, n_kv, group, hd)
        scores = torch.einsum("qkgd,kld->kgql", qb, keys) * scale
        # causal within the chunk: query j (absolute pos L-QL+j) sees keys <= pos
        qpos = torch.arange(L - QL, L, device=q.device).view(1, 1, QL, 1)
        kpos = torch.arange(L, device=q.device).view(1, 1, 1, L)
        scores.masked_fill_(kpos > qpos, float("-inf"))
        probs = torch.softmax(scores, dim=-1)
        ob = torch.einsum("kgql,kld->qkgd", probs, vals)
        out[t0 : t0 + QL] = ob.reshape(QL, nq, hd).to(q.dtype)
        t0 += QL
    return out


def swiglu(gate_up: torch.Tensor) -> torch.Tensor:
    g, u = gate_up.float().chunk(2, dim=-1)
    return (torch.nn.functional.silu(g) * u).to(gate_up.dtype)


def moe_topk_gate(
    logits: torch.Tensor, top_k: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Softmax-after-topk gating (Mixtral). logits [T, E] ->
    (weights [T, k] fp32, indices [T, k] int32)."""
    vals, idx = torch.topk(logits.float(), top_k, dim=-1)
    weights = torch.softmax(vals, dim=-1)
    return weights, idx.to(torch.int32)


def grouped_gemm(
    x: torch.Tensor,        # [S, K] (rows sorted by expert)
    w: torch.Tensor,        # [E, N, K]
    offsets: torch.Tensor,  # [E+1] int32
) -> torch.Tensor:
    """Per-expert segment GEMM: out[offs[e]:offs[e+1]] = x_seg @ w[e].T"""
    S, K = x.shape
    E, N, _ = w.shape
    out = torch.zeros(S, N, dtype=x.dtype, device=x.device)
    for e in range(E):
        lo, hi = int(offsets[e]), int(offsets[e + 1])
        if hi > lo:
            out[lo:hi] = (x[lo:hi].float() @ w[e].float().t()).to(x.dtype)
    return out


def attn_decode_lse(
    q: torch.Tensor,
    k_cache: torch.Tensor,
    v_cache: torch.Tensor,
    block_table: torch.Tensor,
    seq_lens: torch.Tensor,
    scale: float,
):
    """attn_decode + per-head (m, l) merge state (see parallel/cp.py)."""
    out_ml = torch.empty(q.shape[0], q.shape[1], 2, dtype=torch.float32,
                         device=q.device)
    out = attn_decode(q, k_cache, v_cache, block_table, seq_lens, scale,
                      out_ml=out_ml)
    return out, out_ml
