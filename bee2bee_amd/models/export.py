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
        # family knobs (python-time constants: trace-friendly branches)
        self.gpt2 = spec.norm_type == "layernorm"
        self.register_buffer("embed", weights.embed.float().clone())
        if spec.pos_type == "learned":
            self.register_buffer("pos_embed", weights.pos_embed.float())
        self.register_buffer("final_norm", weights.final_norm.float())
        if weights.final_norm_bias is not None:
            self.register_buffer("final_norm_b",
                                 weights.final_norm_bias.float())
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
            for name, t in (("attn_norm_b", lw.attn_norm_bias),
                            ("mlp_norm_b", lw.mlp_norm_bias),
                            ("wo_b", lw.wo_bias),
                            ("w_gate_up_b", lw.w_gate_up_bias),
                            ("w_down_b", lw.w_down_bias)):
                if t is not None:
                    self.register_buffer(f"{name}_{i}", t.float())

    @staticmethod
    def _rms(x: torch.Tensor, w: torch.Tensor, eps: float) -> torch.Tensor:
        v = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + eps)
        return v * w

    def _norm(self, x, i_or_final):
        if not self.gpt2:
            w = (self.final_norm if i_or_final is None
                 else getattr(self, f"attn_norm_{i_or_final}"))
            return self._rms(x, w, self.eps)
        if i_or_final is None:
            return F.layer_norm(x, x.shape[-1:], self.final_norm,
                                self.final_norm_b, self.eps)
        return F.layer_norm(x, x.shape[-1:],
                            getattr(self, f"attn_norm_{i_or_final}"),
                            getattr(self, f"attn_norm_b_{i_or_final}"),
                            self.eps)

    def _norm2(self, x, i):
        if not self.gpt2:
            return self._rms(x, getattr(self, f"mlp_norm_{i}"), self.eps)
        return F.layer_norm(x, x.shape[-1:], getattr(self, f"mlp_norm_{i}"),
                            getattr(self, f"mlp_norm_b_{i}"), self.eps)

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
        if self.gpt2:
            h = h + self.pos_embed[:T].unsqueeze(0)
        mask = torch.triu(
            torch.full((T, T), float("-inf"), device=h.device), diagonal=1
        )
        group = self.n_heads // self.n_kv
        for i in range(self.n_layers):
            normed = self._norm(h, i)
            qkv = F.linear(normed, getattr(self, f"wqkv_{i}"))
            if self.qkv_bias:
                qkv = qkv + getattr(self, f"wqkv_bias_{i}")
            q, k, v = qkv.split([self.q_size, self.kv_size, self.kv_size], -1)
            q = q.view(B, T, self.n_heads, self.hd)
            k = k.view(B, T, self.n_kv, self.hd)
            if not self.gpt2:
                q = self._rope(q, T)
                k = self._rope(k, T)
            v = v.view(B, T, self.n_kv, self.hd)
            k = k.repeat_interleave(group, dim=2)
            v = v.repeat_interleave(group, dim=2)
            att = torch.einsum("bqhd,bkhd->bhqk", q, k) / (self.hd ** 0.5)
            att = torch.softmax(att + mask, dim=-1)
            o = torch.einsum("bhqk,bkhd->bqhd", att, v).reshape(B, T, -1)
            oproj = F.linear(o, getattr(self, f"wo_{i}"))
            if hasattr(self, f"wo_b_{i}"):
                oproj = oproj + getattr(self, f"wo_b_{i}")
            h = h + oproj
            normed = self._norm2(h, i)
            if self.gpt2:
                fc = F.linear(normed, getattr(self, f"w_gate_up_{i}"),
                              getattr(self, f"w_gate_up_b_{i}"))
                act = 0.5 * fc * (1.0 + torch.tanh(
                    0.7978845608028654 * (fc + 0.044715 * fc.pow(3))))
                h = h + F.linear(act, getattr(self, f"w_down_{i}"),
                                 getattr(self, f"w_down_b_{i}"))
            else:
                gu = F.linear(normed, getattr(self, f"w_gate_up_{i}"))
                g, u = gu.chunk(2, dim=-1)
                h = h + F.linear(F.silu(g) * u, getattr(self, f"w_down_{i}"))
        h = self._norm(h, None)
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
