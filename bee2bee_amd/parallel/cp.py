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


class CPEngine:
    """Lockstep context-parallel engine: every rank holds the FULL weights
    and computes the SAME batch; the KV history is page-sharded round-robin
    across ranks (global page p lives on rank p % world), so the servable
    context per sequence scales with world x HBM while activations stay
    replicated. Prefill compute is replicated (the dense flash prefill
    needs no cache) and only the owned pages are written; decode runs
    cp_attn_decode per layer via the Runner's decode_attn_fn hook.

    Outputs are identical on every rank (greedy or shared-seed sampling),
    mirroring TPEngine's contract. CPU-tested on gloo world 2 against the
    single-process engine (tests/test_cp_cpu.py)."""

    def __init__(
        self,
        model,
        device: Optional[str] = None,
        dtype: Optional[torch.dtype] = None,
        model_path: Optional[str] = None,
        max_batch: int = 8,
        max_seq_len: int = 4096,
        seed: int = 0,
        group: Optional[dist.ProcessGroup] = None,
    ) -> None:
        import dataclasses as _dc

        from ..engine import kv as kv_mod
        from ..engine.kv import PagedKV
        from ..engine.runner import Runner
        from ..models.spec import ModelSpec, resolve_spec
        from ..models.weights import ModelWeights

        assert dist.is_initialized(), "init the process group first"
        self.group = group
        self.rank = dist.get_rank(group)
        self.world = dist.get_world_size(group)
        spec = model if isinstance(model, ModelSpec) else resolve_spec(
            model, model_path)
        self.spec = spec
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        if dtype is None:
            dtype = (torch.bfloat16 if self.device.type == "cuda"
                     else torch.float32)
        self.dtype = dtype
        self.page = kv_mod.BLOCK_SIZE
        self.max_seq_len = min(max_seq_len, spec.max_seq_len)

        self.weights = ModelWeights(spec, self.device, dtype)
        if model_path:
            self.weights.load_hf(model_path)
        else:
            self.weights.random_init(seed=seed)

        # each rank stores ~1/world of every sequence's pages (+1 slack)
        pages_total = -(-self.max_seq_len // self.page)
        local_pages = pages_total // self.world + 1
        self.kv = PagedKV(spec, self.device, dtype,
                          n_blocks=max_batch * local_pages + 2)
        self.runner = Runner(spec, self.weights, self.kv, self.device, dtype)
        self.runner.decode_attn_fn = self._cp_attn
        # non-greedy sampling must draw IDENTICAL tokens on every rank
        # (same contract and fix as TPEngine)
        self.sample_gen = torch.Generator(device=self.device)
        self.sample_gen.manual_seed(seed + 0x5EED)
        self._seqs: List[int] = []
        self._lens: List[int] = []          # GLOBAL lengths
        self._scratch = -7
        self.kv.new_seq(self._scratch)
        self.kv.extend_seq(self._scratch, 1)

    # page p of a sequence lives on rank p % world; its LOCAL logical
    # position for PagedKV is (p // world) * page + offset
    def _owned(self, pos: int) -> bool:
        return (pos // self.page) % self.world == self.rank

    def _local_pos(self, pos: int) -> int:
        return (pos // self.page) // self.world * self.page + pos % self.page

    def _local_len(self, L: int) -> int:
        """Tokens of a length-L history stored on this rank."""
        n = 0
        for p in range(0, -(-L // self.page)):
            if p % self.world == self.rank:
                n += min(self.page, L - p * self.page)
        return n

    def _slots(self, sid: int, positions) -> List[int]:
        """Slot per global position: owned -> local slot, else scratch."""
        scratch = self.kv.slot_mapping(self._scratch, [0])[0]
        out = []
        for pos in positions:
            if self._owned(pos):
                out.append(self.kv.slot_mapping(
                    sid, [self._local_pos(pos)])[0])
            else:
                out.append(scratch)
        return out

    def _cp_attn(self, q, k_cache, v_cache, block_table, seq_lens, scale):
        return cp_attn_decode(q, k_cache, v_cache, block_table, seq_lens,
                              scale, group=self.group)

    def _local_lens_t(self) -> torch.Tensor:
        return torch.tensor([self._local_len(L) for L in self._lens],
                            dtype=torch.int32, device=self.device)

    @torch.no_grad()
    def prefill(self, prompts, sampling=None) -> torch.Tensor:
        from ..engine.sampler import SamplingParams, sample

        sampling = sampling or SamplingParams(greedy=True)
        dev = self.device
        self._seqs = list(range(len(prompts)))
        self._lens = [len(p) for p in prompts]
        ids_list, pos_list, slot_list, cu = [], [], [], [0]
        for sid, p in zip(self._seqs, prompts):
            self.kv.new_seq(sid)
            self.kv.extend_seq(sid, self._local_len(len(p)))
            ids_list.extend(p)
            pos_list.extend(range(len(p)))
            slot_list.extend(self._slots(sid, range(len(p))))
            cu.append(cu[-1] + len(p))
        hidden = self.runner.forward_prefill(
            torch.tensor(ids_list, dtype=torch.int64, device=dev),
            torch.tensor(pos_list, dtype=torch.int32, device=dev),
            torch.tensor(slot_list, dtype=torch.int32, device=dev),
            torch.tensor(cu, dtype=torch.int32, device=dev),
            max(self._lens),
        )
        last = torch.tensor([c - 1 for c in cu[1:]], dtype=torch.int64,
                            device=dev)
        return sample(self.runner.lm_head(hidden[last]), sampling,
                      generator=self.sample_gen).cpu()

    @torch.no_grad()
    def decode_step(self, ids: torch.Tensor, sampling=None) -> torch.Tensor:
        from ..engine.graphs import decode_slot_mapping  # noqa: F401
        from ..engine.sampler import SamplingParams, sample

        sampling = sampling or SamplingParams(greedy=True)
        dev = self.device
        positions = torch.tensor(self._lens, dtype=torch.int32, device=dev)
        slot_list = []
        for i, sid in enumerate(self._seqs):
            pos = self._lens[i]
            self.kv.extend_seq(sid, self._local_len(pos + 1))
            slot_list.append(self._slots(sid, [pos])[0])
            self._lens[i] = pos + 1
        lens_t = self._local_lens_t()
        bt = self.kv.block_table(self._seqs)
        slots = torch.tensor(slot_list, dtype=torch.int32, device=dev)
        hidden = self.runner.forward_decode(
            ids.to(dev), positions, slots, bt, lens_t)
        return sample(self.runner.lm_head(hidden), sampling,
                      generator=self.sample_gen).cpu()

    def reset(self) -> None:
        """Free the batch's KV so repeated generate() calls (serving via
        parallel/serve.py LockstepServer) never exhaust the local pool."""
        for sid in self._seqs:
            self.kv.free_seq(sid)
        self._seqs, self._lens = [], []

    @torch.no_grad()
    def generate(self, prompts, max_new_tokens: int, sampling=None):
        ids = self.prefill(prompts, sampling)
        outs = [[int(t)] for t in ids]
        for _ in range(max_new_tokens - 1):
            ids = self.decode_step(ids, sampling)
            for o, t in zip(outs, ids):
                o.append(int(t))
        return outs
