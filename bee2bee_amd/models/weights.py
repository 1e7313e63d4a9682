"""Model weights: HF-checkpoint-compatible loading (safetensors) and random
init — laid out the way the MI355X engine consumes them.

Engine layout decisions (MI355X-first):
  * qkv is fused into one [q+2kv, hidden] matrix and gate|up into one
    [2*inter, hidden] matrix per layer → one hipBLASLt GEMM each instead of
    three/two (fewer, larger GEMMs saturate MFMA and HBM better).
  * Weights live in bf16 on-device; 288 GB HBM3E holds Llama-3-70B (140 GB)
    on a single GPU with room for KV cache, so sharding is a choice
    (pipeline parallel), not a requirement.

HF name mapping (checkpoint compatibility, reference loads the same files via
transformers — bee2bee/hf.py:23-32):
  model.embed_tokens.weight, model.layers.{i}.self_attn.{q,k,v,o}_proj.weight,
  model.layers.{i}.mlp.{gate,up,down}_proj.weight,
  model.layers.{i}.{input,post_attention}_layernorm.weight, model.norm.weight,
  lm_head.weight; Mixtral: model.layers.{i}.block_sparse_moe.gate.weight and
  .experts.{e}.w{1,2,3}.weight.
"""
from __future__ import annotations

import glob
import json
import os
from typing import Dict, Iterator, Optional, Tuple

import torch

from .spec import ModelSpec


class LayerWeights:
    __slots__ = (
        "attn_norm", "attn_norm_bias", "wqkv", "wqkv_bias", "wo", "wo_bias",
        "mlp_norm", "mlp_norm_bias",
        "w_gate_up", "w_gate_up_bias", "w_down", "w_down_bias",  # dense mlp
        "moe_gate", "moe_w_gate_up", "moe_w_down",  # moe
    )

    def __init__(self) -> None:
        self.attn_norm = None
        self.attn_norm_bias = None  # layernorm arch (gpt2)
        self.wqkv = None
        self.wqkv_bias = None  # Qwen2/gpt2 fused [q+2kv] bias (optional)
        self.wo = None
        self.wo_bias = None
        self.mlp_norm = None
        self.mlp_norm_bias = None
        self.w_gate_up = None       # gelu arch: the single fc [I, H]
        self.w_gate_up_bias = None
        self.w_down = None
        self.w_down_bias = None
        self.moe_gate = None
        self.moe_w_gate_up = None
        self.moe_w_down = None


class ModelWeights:
    def __init__(self, spec: ModelSpec, device: torch.device, dtype: torch.dtype) -> None:
        self.spec = spec
        self.device = device
        self.dtype = dtype
        self.embed: Optional[torch.Tensor] = None  # [vocab, hidden]
        self.pos_embed: Optional[torch.Tensor] = None  # [max_seq, hidden]
        self.final_norm: Optional[torch.Tensor] = None  # [hidden]
        self.final_norm_bias: Optional[torch.Tensor] = None
        self.lm_head: Optional[torch.Tensor] = None  # [vocab, hidden]
        self.layers = [LayerWeights() for _ in range(spec.n_layers)]
        # expert-parallel shard: when set, MoE expert tensors hold only
        # experts [lo, hi) (gate stays replicated); see parallel/ep.py
        self.expert_range: Optional[Tuple[int, int]] = None

    # -------------------------------------------------------------- random

    @torch.no_grad()
    def random_init(
        self,
        seed: int = 0,
        layer_range: Optional[Tuple[int, int]] = None,
        expert_range: Optional[Tuple[int, int]] = None,
    ) -> "ModelWeights":
        """Random weights of the real architecture (synthetic serving — the
        benchmark contract requires random-init weights of the named model).

        layer_range (for pipeline stages): only materialize layers [lo, hi);
        embed only on the first stage, head/final norm only on the last.
        expert_range (for expert parallelism): only materialize that expert
        shard — drawn per expert, so a shard holds exactly the values the
        full init would (EP-vs-single equivalence tests rely on this)."""
        s = self.spec
        lo, hi = layer_range or (0, s.n_layers)
        self.expert_range = expert_range
        e_lo, e_hi = expert_range or (0, getattr(s, "n_experts", 0) or 0)
        import zlib

        # Generate directly on the target device (an 8B randn on host would
        # serialize startup). Each tensor gets its own generator seeded by
        # (seed, tensor name) so a pipeline stage materializing only layers
        # [lo, hi) draws IDENTICAL values to a full single-process init —
        # the PP-vs-single-process equivalence tests rely on this.
        def rnd(name: str, *shape: int, std: float) -> torch.Tensor:
            tseed = (seed * 1000003 + zlib.crc32(name.encode())) % (2**31)
            gen = torch.Generator(device=self.device).manual_seed(tseed)
            t = torch.randn(*shape, generator=gen, dtype=torch.float32,
                            device=self.device)
            return (t * std).to(self.dtype)

        std = 0.02
        proj_std = std / max(1.0, (2 * s.n_layers) ** 0.5)
        zeros = lambda *shape: torch.zeros(  # noqa: E731
            *shape, device=self.device, dtype=self.dtype)
        if lo == 0:
            self.embed = rnd("embed", s.vocab_size, s.hidden_size, std=std)
            if s.pos_type == "learned":
                self.pos_embed = rnd("pos_embed", s.max_seq_len,
                                     s.hidden_size, std=0.01)
        if hi == s.n_layers:
            self.final_norm = torch.ones(
                s.hidden_size, device=self.device, dtype=self.dtype
            )
            if s.norm_type == "layernorm":
                self.final_norm_bias = zeros(s.hidden_size)
            if s.tie_embeddings:
                self.lm_head = (
                    self.embed
                    if self.embed is not None
                    else rnd("embed", s.vocab_size, s.hidden_size, std=std)
                )
            else:
                self.lm_head = rnd("lm_head", s.vocab_size, s.hidden_size, std=std)
        for i in range(lo, hi):
            lw = self.layers[i]
            lw.attn_norm = torch.ones(s.hidden_size, device=self.device, dtype=self.dtype)
            lw.mlp_norm = torch.ones(s.hidden_size, device=self.device, dtype=self.dtype)
            lw.wqkv = rnd(f"l{i}.wqkv", s.q_size + 2 * s.kv_size, s.hidden_size, std=std)
            if s.qkv_bias:
                lw.wqkv_bias = rnd(
                    f"l{i}.wqkv_b", s.q_size + 2 * s.kv_size, std=std
                )
            if s.norm_type == "layernorm":
                lw.attn_norm_bias = zeros(s.hidden_size)
                lw.mlp_norm_bias = zeros(s.hidden_size)
            lw.wo = rnd(f"l{i}.wo", s.hidden_size, s.q_size, std=proj_std)
            if s.attn_out_bias:
                lw.wo_bias = zeros(s.hidden_size)
            if s.is_moe:
                lw.moe_gate = rnd(f"l{i}.gate", s.n_experts, s.hidden_size, std=std)
                lw.moe_w_gate_up = torch.stack([
                    rnd(f"l{i}.w_gu.e{e}", 2 * s.intermediate_size,
                        s.hidden_size, std=std)
                    for e in range(e_lo, e_hi)
                ])
                lw.moe_w_down = torch.stack([
                    rnd(f"l{i}.w_dn.e{e}", s.hidden_size,
                        s.intermediate_size, std=proj_std)
                    for e in range(e_lo, e_hi)
                ])
            elif s.act_type == "gelu":
                lw.w_gate_up = rnd(f"l{i}.w_fc", s.intermediate_size,
                                   s.hidden_size, std=std)
                lw.w_down = rnd(f"l{i}.w_dn", s.hidden_size,
                                s.intermediate_size, std=proj_std)
                if s.mlp_bias:
                    lw.w_gate_up_bias = zeros(s.intermediate_size)
                    lw.w_down_bias = zeros(s.hidden_size)
            else:
                lw.w_gate_up = rnd(f"l{i}.w_gu", 2 * s.intermediate_size, s.hidden_size, std=std)
                lw.w_down = rnd(f"l{i}.w_dn", s.hidden_size, s.intermediate_size, std=proj_std)
        return self

    # ---------------------------------------------------------- safetensors

    @torch.no_grad()
    def load_hf(
        self,
        model_path: str,
        layer_range: Optional[Tuple[int, int]] = None,
        expert_range: Optional[Tuple[int, int]] = None,
    ) -> "ModelWeights":
        """Load HF safetensors shards, fusing qkv / gate|up on the fly.
        expert_range: keep only that expert shard on device (EP ranks)."""
        s = self.spec
        if s.arch == "gpt2":
            return self._load_gpt2(model_path, layer_range)
        lo, hi = layer_range or (0, s.n_layers)
        self.expert_range = expert_range
        e_lo, e_hi = expert_range or (0, getattr(s, "n_experts", 0) or 0)
        n_local_e = e_hi - e_lo
        want_embed = lo == 0
        want_head = hi == s.n_layers

        pending: Dict[str, Dict[str, torch.Tensor]] = {}

        def to_dev(t: torch.Tensor) -> torch.Tensor:
            return t.to(self.device, self.dtype, non_blocking=True)

        def try_fuse(i: int) -> None:
            lw = self.layers[i]
            p = pending.get(f"attn.{i}")
            if p and len(p) == 3 and lw.wqkv is None:
                lw.wqkv = torch.cat([p["q"], p["k"], p["v"]], dim=0)
                pending.pop(f"attn.{i}")
            p = pending.get(f"attnb.{i}")
            if p and len(p) == 3 and lw.wqkv_bias is None:
                lw.wqkv_bias = torch.cat([p["q"], p["k"], p["v"]], dim=0)
                pending.pop(f"attnb.{i}")
            p = pending.get(f"mlp.{i}")
            if p and len(p) == 2 and lw.w_gate_up is None:
                lw.w_gate_up = torch.cat([p["gate"], p["up"]], dim=0)
                pending.pop(f"mlp.{i}")

        for name, tensor in iter_safetensors(model_path):
            parts = name.split(".")
            if name == "model.embed_tokens.weight":
                if want_embed:
                    self.embed = to_dev(tensor)
                elif want_head and s.tie_embeddings:
                    # PP LAST stage of a tied checkpoint: the head IS the
                    # embedding matrix, which only ships as embed_tokens
                    self.lm_head = to_dev(tensor)
                continue
            if name == "model.norm.weight":
                if want_head:
                    self.final_norm = to_dev(tensor)
                continue
            if name == "lm_head.weight":
                if want_head:
                    self.lm_head = to_dev(tensor)
                continue
            if len(parts) < 4 or parts[0] != "model" or parts[1] != "layers":
                continue
            i = int(parts[2])
            if not (lo <= i < hi):
                continue
            lw = self.layers[i]
            sub = ".".join(parts[3:])
            if sub == "input_layernorm.weight":
                lw.attn_norm = to_dev(tensor)
            elif sub == "post_attention_layernorm.weight":
                lw.mlp_norm = to_dev(tensor)
            elif sub == "self_attn.q_proj.weight":
                pending.setdefault(f"attn.{i}", {})["q"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "self_attn.k_proj.weight":
                pending.setdefault(f"attn.{i}", {})["k"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "self_attn.v_proj.weight":
                pending.setdefault(f"attn.{i}", {})["v"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "self_attn.q_proj.bias":
                pending.setdefault(f"attnb.{i}", {})["q"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "self_attn.k_proj.bias":
                pending.setdefault(f"attnb.{i}", {})["k"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "self_attn.v_proj.bias":
                pending.setdefault(f"attnb.{i}", {})["v"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "self_attn.o_proj.weight":
                lw.wo = to_dev(tensor)
            elif sub == "mlp.gate_proj.weight":
                pending.setdefault(f"mlp.{i}", {})["gate"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "mlp.up_proj.weight":
                pending.setdefault(f"mlp.{i}", {})["up"] = to_dev(tensor)
                try_fuse(i)
            elif sub == "mlp.down_proj.weight":
                lw.w_down = to_dev(tensor)
            elif sub == "block_sparse_moe.gate.weight":
                lw.moe_gate = to_dev(tensor)
            elif parts[3] == "block_sparse_moe" and parts[4] == "experts":
                e = int(parts[5])
                if not (e_lo <= e < e_hi):
                    continue  # expert owned by another EP rank
                w = parts[6]  # w1 (gate), w2 (down), w3 (up)
                key = f"moe.{i}"
                pending.setdefault(key, {})[f"{w}.{e}"] = to_dev(tensor)
                p = pending[key]
                if len(p) == 3 * n_local_e:
                    lw.moe_w_gate_up = torch.stack(
                        [
                            torch.cat([p[f"w1.{e2}"], p[f"w3.{e2}"]], dim=0)
                            for e2 in range(e_lo, e_hi)
                        ]
                    )
                    lw.moe_w_down = torch.stack(
                        [p[f"w2.{e2}"] for e2 in range(e_lo, e_hi)]
                    )
                    pending.pop(key)

        if want_head and self.lm_head is None:
            # tied embeddings checkpoints omit lm_head
            self.lm_head = self.embed
        missing = []
        if want_embed and self.embed is None:
            missing.append("embed_tokens")
        for i in range(lo, hi):
            if self.layers[i].wqkv is None:
                missing.append(f"layers.{i}.qkv")
        if missing:
            raise FileNotFoundError(
                f"checkpoint at {model_path} is missing tensors: {missing[:5]}"
            )
        return self


    @torch.no_grad()
    def _load_gpt2(self, model_path: str,
                   layer_range: Optional[Tuple[int, int]] = None
                   ) -> "ModelWeights":
        """GPT-2 family safetensors (wte/wpe/h.N.*): Conv1D weights are
        stored transposed ([in, out]) and qkv ships pre-fused as c_attn."""
        s = self.spec
        lo, hi = layer_range or (0, s.n_layers)

        def to_dev(t: torch.Tensor) -> torch.Tensor:
            return t.to(self.device, self.dtype, non_blocking=True)

        for name, tensor in iter_safetensors(model_path):
            name = name[len("transformer."):] if name.startswith(
                "transformer.") else name
            if name == "wte.weight":
                if lo == 0:
                    self.embed = to_dev(tensor)
                elif hi == s.n_layers:
                    self.lm_head = to_dev(tensor)  # tied head, PP last stage
                continue
            if name == "wpe.weight":
                if lo == 0:
                    self.pos_embed = to_dev(tensor)
                continue
            if name == "ln_f.weight":
                if hi == s.n_layers:
                    self.final_norm = to_dev(tensor)
                continue
            if name == "ln_f.bias":
                if hi == s.n_layers:
                    self.final_norm_bias = to_dev(tensor)
                continue
            parts = name.split(".")
            if parts[0] != "h":
                continue
            i = int(parts[1])
            if not (lo <= i < hi):
                continue
            lw = self.layers[i]
            rest = ".".join(parts[2:])
            if rest == "ln_1.weight":
                lw.attn_norm = to_dev(tensor)
            elif rest == "ln_1.bias":
                lw.attn_norm_bias = to_dev(tensor)
            elif rest == "ln_2.weight":
                lw.mlp_norm = to_dev(tensor)
            elif rest == "ln_2.bias":
                lw.mlp_norm_bias = to_dev(tensor)
            elif rest == "attn.c_attn.weight":
                lw.wqkv = to_dev(tensor.t().contiguous())  # Conv1D -> [3H, H]
            elif rest == "attn.c_attn.bias":
                lw.wqkv_bias = to_dev(tensor)
            elif rest == "attn.c_proj.weight":
                lw.wo = to_dev(tensor.t().contiguous())
            elif rest == "attn.c_proj.bias":
                lw.wo_bias = to_dev(tensor)
            elif rest == "mlp.c_fc.weight":
                lw.w_gate_up = to_dev(tensor.t().contiguous())  # [I, H]
            elif rest == "mlp.c_fc.bias":
                lw.w_gate_up_bias = to_dev(tensor)
            elif rest == "mlp.c_proj.weight":
                lw.w_down = to_dev(tensor.t().contiguous())  # [H, I]
            elif rest == "mlp.c_proj.bias":
                lw.w_down_bias = to_dev(tensor)
        if hi == s.n_layers and self.lm_head is None and self.embed is not None:
            self.lm_head = self.embed  # tied
        return self



def _save_gpt2(weights: "ModelWeights", out_dir: str, tensors, cpu) -> None:
    """GPT-2-format shard + config (Conv1D tensors re-transposed)."""
    from safetensors.torch import save_file

    s = weights.spec
    if weights.embed is not None:
        tensors["wte.weight"] = cpu(weights.embed)
    if weights.pos_embed is not None:
        tensors["wpe.weight"] = cpu(weights.pos_embed)
    if weights.final_norm is not None:
        tensors["ln_f.weight"] = cpu(weights.final_norm)
    if weights.final_norm_bias is not None:
        tensors["ln_f.bias"] = cpu(weights.final_norm_bias)
    for i, lw in enumerate(weights.layers):
        if lw.wqkv is None:
            continue
        p = f"h.{i}"
        tensors[f"{p}.ln_1.weight"] = cpu(lw.attn_norm)
        tensors[f"{p}.ln_1.bias"] = cpu(lw.attn_norm_bias)
        tensors[f"{p}.ln_2.weight"] = cpu(lw.mlp_norm)
        tensors[f"{p}.ln_2.bias"] = cpu(lw.mlp_norm_bias)
        tensors[f"{p}.attn.c_attn.weight"] = cpu(lw.wqkv.t())
        tensors[f"{p}.attn.c_attn.bias"] = cpu(lw.wqkv_bias)
        tensors[f"{p}.attn.c_proj.weight"] = cpu(lw.wo.t())
        tensors[f"{p}.attn.c_proj.bias"] = cpu(lw.wo_bias)
        tensors[f"{p}.mlp.c_fc.weight"] = cpu(lw.w_gate_up.t())
        tensors[f"{p}.mlp.c_fc.bias"] = cpu(lw.w_gate_up_bias)
        tensors[f"{p}.mlp.c_proj.weight"] = cpu(lw.w_down.t())
        tensors[f"{p}.mlp.c_proj.bias"] = cpu(lw.w_down_bias)
    save_file(tensors, os.path.join(out_dir, "model.safetensors"))
    cfg = {
        "architectures": ["GPT2LMHeadModel"],
        "model_type": "gpt2",
        "vocab_size": s.vocab_size,
        "n_embd": s.hidden_size,
        "n_layer": s.n_layers,
        "n_head": s.n_heads,
        "n_positions": s.max_seq_len,
        "layer_norm_epsilon": s.rms_eps,
        "tie_word_embeddings": True,
        "bos_token_id": s.bos_token_id,
        "eos_token_id": s.eos_token_id,
    }
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump(cfg, f, indent=1)


def iter_safetensors(model_path: str) -> Iterator[Tuple[str, torch.Tensor]]:
    """Yield (name, tensor) from every *.safetensors shard in a directory."""
    from safetensors import safe_open

    shards = sorted(glob.glob(os.path.join(model_path, "*.safetensors")))
    if not shards:
        raise FileNotFoundError(f"no safetensors shards in {model_path}")
    for shard in shards:
        with safe_open(shard, framework="pt", device="cpu") as f:
            for name in f.keys():
                yield name, f.get_tensor(name)


def save_hf(weights: ModelWeights, out_dir: str) -> None:
    """Write weights back out as one HF-format safetensors shard + config
    (checkpoint compatibility round-trip; used by tests)."""
    from safetensors.torch import save_file

    s = weights.spec
    os.makedirs(out_dir, exist_ok=True)
    tensors: Dict[str, torch.Tensor] = {}

    def cpu(t: torch.Tensor) -> torch.Tensor:
        return t.detach().to("cpu").contiguous()

    if s.arch == "gpt2":
        _save_gpt2(weights, out_dir, tensors, cpu)
        return

    if weights.embed is not None:
        tensors["model.embed_tokens.weight"] = cpu(weights.embed)
    if weights.final_norm is not None:
        tensors["model.norm.weight"] = cpu(weights.final_norm)
    if weights.lm_head is not None and not s.tie_embeddings:
        tensors["lm_head.weight"] = cpu(weights.lm_head)
    for i, lw in enumerate(weights.layers):
        if lw.wqkv is None:
            continue
        pfx = f"model.layers.{i}"
        q, k, v = torch.split(lw.wqkv, [s.q_size, s.kv_size, s.kv_size], dim=0)
        tensors[f"{pfx}.self_attn.q_proj.weight"] = cpu(q)
        tensors[f"{pfx}.self_attn.k_proj.weight"] = cpu(k)
        tensors[f"{pfx}.self_attn.v_proj.weight"] = cpu(v)
        if lw.wqkv_bias is not None:
            qb, kb, vb = torch.split(
                lw.wqkv_bias, [s.q_size, s.kv_size, s.kv_size], dim=0
            )
            tensors[f"{pfx}.self_attn.q_proj.bias"] = cpu(qb)
            tensors[f"{pfx}.self_attn.k_proj.bias"] = cpu(kb)
            tensors[f"{pfx}.self_attn.v_proj.bias"] = cpu(vb)
        tensors[f"{pfx}.self_attn.o_proj.weight"] = cpu(lw.wo)
        tensors[f"{pfx}.input_layernorm.weight"] = cpu(lw.attn_norm)
        tensors[f"{pfx}.post_attention_layernorm.weight"] = cpu(lw.mlp_norm)
        if s.is_moe:
            tensors[f"{pfx}.block_sparse_moe.gate.weight"] = cpu(lw.moe_gate)
            for e in range(s.n_experts):
                g, u = torch.split(
                    lw.moe_w_gate_up[e], [s.intermediate_size, s.intermediate_size], dim=0
                )
                tensors[f"{pfx}.block_sparse_moe.experts.{e}.w1.weight"] = cpu(g)
                tensors[f"{pfx}.block_sparse_moe.experts.{e}.w3.weight"] = cpu(u)
                tensors[f"{pfx}.block_sparse_moe.experts.{e}.w2.weight"] = cpu(
                    lw.moe_w_down[e]
                )
        else:
            g, u = torch.split(
                lw.w_gate_up, [s.intermediate_size, s.intermediate_size], dim=0
            )
            tensors[f"{pfx}.mlp.gate_proj.weight"] = cpu(g)
            tensors[f"{pfx}.mlp.up_proj.weight"] = cpu(u)
            tensors[f"{pfx}.mlp.down_proj.weight"] = cpu(lw.w_down)

    save_file(tensors, os.path.join(out_dir, "model.safetensors"))
    cfg = {
        "architectures": ["MixtralForCausalLM" if s.is_moe else "LlamaForCausalLM"],
        "vocab_size": s.vocab_size,
        "hidden_size": s.hidden_size,
        "intermediate_size": s.intermediate_size,
        "num_hidden_layers": s.n_layers,
        "num_attention_heads": s.n_heads,
        "num_key_value_heads": s.n_kv_heads,
        "head_dim": s.head_dim,
        "rope_theta": s.rope_theta,
        "attention_bias": s.qkv_bias,
        "rms_norm_eps": s.rms_eps,
        "max_position_embeddings": s.max_seq_len,
        "tie_word_embeddings": s.tie_embeddings,
        "bos_token_id": s.bos_token_id,
        "eos_token_id": s.eos_token_id,
    }
    if s.is_moe:
        cfg["num_local_experts"] = s.n_experts
        cfg["num_experts_per_tok"] = s.top_k_experts
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump(cfg, f, indent=2)
