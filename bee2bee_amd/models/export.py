"""Model export (parity with the reference's TorchScript/ONNX export,
bee2bee/hf.py:139-158).

The serving engine runs hand-written HIP kernels that do not script, so
export builds an equivalent PURE-TORCH module from the same fused weights
(full-context attention, no paged KV) — numerically the reference path the
HIP kernels are tested against — and traces that. TorchScript export works
fully offline; ONNX export is gated on the optional `onnx` dependency the
same way the reference gates on its optional imports.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F

from .spec import ModelSpec
from .weights import ModelWeights


class ExportableModel(torch.nn.Module):
    """Pure-torch full-context forward over the fused weights: logits for
    every position of `input_ids` [B, T]. Trace-friendly (no data-dependent
    control flow)."""

    def __init__(self, spec: ModelSpec, weights: ModelWeights) -> None:
        super().__init__()
        assert not spec.is_moe, "MoE export is not supported"
        self.n_layers = spec.n_layers
        self.n_heads = spec.n_heads
        self.n_kv = spec.n_kv_heads
        self.hd = spec.head_dim
        self.q_size = spec.q_size
        self.kv_size = spec.kv_size
        self.eps = spec.rms_eps
        self.register_buffer("embed", weights.embed.float().clone())
        self.register_buffer("final_norm", weights.final_norm.float())
        # clone: tied-embedding checkpoints alias embed (tracing
        # rejects shared storage between buffers)
        self.register_buffer("lm_head_w", weights.lm_head.float().clone())
        inv = 1.0 / (
            spec.rope_theta
            ** (torch.arange(0, spec.head_dim, 2).float() / spec.head_dim)
        )
        t = torch.arange(spec.max_seq_len).float()
        freqs = torch.outer(t, inv)
        self.register_buffer("rope_cos", freqs.cos())
        self.register_buffer("rope_sin", freqs.sin())
        self.qkv_bias = spec.qkv_bias
        for i, lw in enumerate(weights.layers):
            if lw.wqkv_bias is not None:
                self.register_buffer(f"wqkv_bias_{i}", lw.wqkv_bias.float())
            self.register_buffer(f"attn_norm_{i}", lw.attn_norm.float())
            self.register_buffer(f"mlp_norm_{i}", lw.mlp_norm.float())
            self.register_buffer(f"wqkv_{i}", lw.wqkv.float())
            self.register_buffer(f"wo_{i}", lw.wo.float())
            self.register_buffer(f"w_gate_up_{i}", lw.w_gate_up.float())
            self.register_buffer(f"w_down_{i}", lw.w_down.float())

    @staticmethod
    def _rms(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
        v = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + eps)
        return v * w

    def _rope(self, x: torch.Tensor, T: int) -> torch.Tensor:
        # x [B, T, H, hd]; rotate pairs (even, odd interleave = our kernel's
        # half-split layout: first hd/2 dims rotate with second hd/2)
        half = self.hd // 2
        cos = self.rope_cos[:T].view(1, T, 1, half)
        sin = self.rope_sin[:T].view(1, T, 1, half)
        x1, x2 = x[..., :half], x[..., half:]
        return torch.cat([x1 * cos - x2 * sin, x1 * sin + x2 * cos], dim=-1)

    def forward(self, input_ids: torch.Tensor) -> torch.Tensor:
        B, T = input_ids.shape
        h = F.embedding(input_ids, self.embed)
        mask = torch.triu(
            torch.full((T, T), float("-inf"), device=h.device), diagonal=1
        )
        group = self.n_heads // self.n_kv
        for i in range(self.n_layers):
            normed = self._rms(h, getattr(self, f"attn_norm_{i}"), self.eps)
            qkv = F.linear(normed, getattr(self, f"wqkv_{i}"))
            if self.qkv_bias:
                qkv = qkv + getattr(self, f"wqkv_bias_{i}")
            q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size], -1)
            q = self._rope(q.view(B, T, self.n_heads, self.hd), T)
            k = self._rope(k.view(B, T, self.n_kv, self.hd), T)
            v = v.view(B, T, self.n_kv, self.hd)
            k = k.repeat_interleave(group, dim=2)
            v = v.repeat_interleave(group, dim=2)
            att = torch.einsum("bqhd,bkhd->bhqk", q, k) / (self.hd ** 0.5)
            att = torch.softmax(att + mask, dim=-1)
            o = torch.einsum("bhqk,bkhd->bqhd", att, v).reshape(B, T, -1)
            h = h + F.linear(o, getattr(self, f"wo_{i}"))
            normed = self._rms(h, getattr(self, f"mlp_norm_{i}"), self.eps)
            gu = F.linear(normed, getattr(self, f"w_gate_up_{i}"))
            g, u = gu.chunk(2, dim=-1)
            h = h + F.linear(F.silu(g) * u, getattr(self, f"w_down_{i}"))
        h = self._rms(h, self.final_norm, self.eps)
        return F.linear(h, self.lm_head_w)


def export_torchscript(
    spec: ModelSpec, weights: ModelWeights, path: str, example_len: int = 8
) -> str:
    """Trace the pure-torch equivalent model and save TorchScript
    (reference: bee2bee/hf.py export_torchscript)."""
    model = ExportableModel(spec, weights).eval()
    example = torch.randint(0, spec.vocab_size, (1, example_len))
    with torch.no_grad():
        traced = torch.jit.trace(model, example)
    traced.save(path)
    return path


def export_onnx(
    spec: ModelSpec, weights: ModelWeights, path: str, example_len: int = 8
) -> str:
    """ONNX export, gated on the optional onnx dependency exactly like the
    reference gates its optional imports (bee2bee/hf.py:150-158)."""
    try:
        import onnx  # noqa: F401
    except Exception as e:  # pragma: no cover - depends on environment
        raise RuntimeError("onnx_support_missing") from e
    model = ExportableModel(spec, weights).eval()
    example = torch.randint(0, spec.vocab_size, (1, example_len))
    torch.onnx.export(
        model, (example,), path,
        input_names=["input_ids"], output_names=["logits"],
        dynamic_axes={"input_ids": {0: "batch", 1: "seq"},
                      "logits": {0: "batch", 1: "seq"}},
    )
    return path
