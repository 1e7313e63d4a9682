"""InferenceEngine: continuous-batching serving front end over the Runner.

Replaces the reference's per-request `transformers.generate()` thread
(bee2bee/hf.py:84-108) with a single engine thread that owns the GPU:
requests are admitted into a running batch (batched varlen prefill), decoded
together one token per step (hipGraph-replayed on GPU), streamed out
per-request, and retired individually. Paged KV blocks are recycled on
retirement, so the 288 GB HBM pool is shared across the whole request mix.
"""
from __future__ import annotations

import logging
import os
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import torch

from ..models.spec import ModelSpec, resolve_spec
from ..models.tokenizer import load_tokenizer
from ..models.weights import ModelWeights
from ..utils import new_id, report_engine_throughput
from .graphs import DecodeGraphs, decode_slot_mapping
from .kv import PagedKV
from . import kv as kv_mod
from .runner import Runner
from .sampler import SamplingParams, sample

logger = logging.getLogger("bee2bee_amd.engine")

_STREAM_END = object()


class TextStreamDecoder:
    """Incrementally turns a growing token list into text deltas: emits the
    first token at once (TTFT), then every `flush_every` tokens, holding
    back a possibly-incomplete multibyte char (at most 3 flushes)."""

    def __init__(self, tokenizer, flush_every: int = 4) -> None:
        self.tokenizer = tokenizer
        self.flush_every = flush_every
        self.decoded_upto = 0
        self.n_seen = 0
        # trailing-replacement hold-back state: a U+FFFD at the stream tail
        # is usually a multibyte char split across the boundary — commit the
        # valid prefix NOW, hold only the trailing replacements, and release
        # them only if the same boundary persists (genuinely invalid bytes)
        self._hold_pos = -1
        self._held_flushes = 0

    def delta(self, output_ids: List[int], final: bool = False) -> str:
        if not final:
            self.n_seen += 1
            if (self.flush_every > 1 and self.n_seen > 1
                    and self.n_seen % self.flush_every != 1):
                return ""
        text = self.tokenizer.decode(output_ids)
        d = text[self.decoded_upto:]
        if final:
            self.decoded_upto = len(text)
            return d
        if not d:
            return ""
        stripped = d.rstrip("\ufffd")
        if len(stripped) != len(d):
            cut = self.decoded_upto + len(stripped)
            if cut == self._hold_pos:
                self._held_flushes += 1
            else:
                self._hold_pos = cut
                self._held_flushes = 1
            # a real utf-8 char completes within 3 more ids, so a boundary
            # stuck longer than that is invalid data — emit it as-is
            if self._held_flushes > 3:
                stripped = d
        self.decoded_upto += len(stripped)
        return stripped


class StopStringFilter:
    """Incremental stop-string truncation for token streams.

    Emits EXACTLY the buffered truncation (cut at the EARLIEST occurrence
    of any stop) without un-saying text. Two hazards make this more than a
    tail hold-back: a stop can straddle chunk boundaries, and a LONGER
    stop completing later can start at an EARLIER position than a short
    stop already visible — so text is withheld from the earliest position
    where any stop is complete OR could still complete, and the cut is
    final only when no earlier candidate remains open. `done` flips when
    the cut is final (callers cancel generation — an improvement over the
    reference's generate-then-truncate, bee2bee/hf.py:111-136)."""

    __slots__ = ("stops", "buf", "n_emitted", "emitted", "done")

    def __init__(self, stops) -> None:
        self.stops = [s for s in (stops or []) if s]
        self.buf = ""
        self.n_emitted = 0
        self.emitted = ""
        self.done = False

    def _earliest_completed(self) -> Optional[int]:
        c = None
        for s in self.stops:
            i = self.buf.find(s, self.n_emitted)
            if i >= 0 and (c is None or i < c):
                c = i
        return c

    def _earliest_open(self) -> Optional[int]:
        """Start of the earliest tail that is a proper prefix of a stop
        (a stop that may still complete with future text)."""
        L = len(self.buf)
        p = None
        for s in self.stops:
            lo = max(self.n_emitted, L - len(s) + 1)
            for k in range(lo, L):
                if self.buf[k:] == s[: L - k]:
                    if p is None or k < p:
                        p = k
                    break
        return p

    def feed(self, delta: str) -> str:
        if self.done or not delta:
            return ""
        self.buf += delta
        c = self._earliest_completed()
        p = self._earliest_open()
        safe = len(self.buf)
        if c is not None:
            safe = min(safe, c)
        if p is not None:
            safe = min(safe, p)
        out = self.buf[self.n_emitted:safe]
        self.n_emitted = max(self.n_emitted, safe)
        self.emitted += out
        if c is not None and (p is None or p >= c):
            self.done = True
        return out

    def flush(self) -> str:
        """End of stream: open candidates are dead; finalize the cut."""
        if self.done:
            return ""
        c = self._earliest_completed()
        end = c if c is not None else len(self.buf)
        out = self.buf[self.n_emitted:end]
        self.n_emitted = end
        self.emitted += out
        if c is not None:
            self.done = True
        return out


@dataclass
class GenerationRequest:
    prompt_ids: List[int]
    max_new_tokens: int = 128
    sampling: SamplingParams = field(default_factory=SamplingParams)
    stop_token_ids: tuple = ()
    rid: str = field(default_factory=lambda: new_id("req"))
    # filled by the engine
    out_queue: "queue.Queue" = field(default_factory=queue.Queue)
    submit_ts: float = field(default_factory=time.time)
    admitted_ts: Optional[float] = None   # left the queue, prefill began
    first_token_ts: Optional[float] = None
    done_ts: Optional[float] = None
    error: Optional[str] = None
    cancelled: bool = False
    output_ids: List[int] = field(default_factory=list)
    # optional: called from the engine thread as on_emit(token_id, done);
    # when set, tokens are NOT pushed to out_queue (async consumers use this
    # to avoid one blocking thread per request)
    on_emit: Optional[Any] = None


class _Active:
    __slots__ = ("req", "seq_id", "length", "prefilled", "spec_index",
                 "pen_slot")

    def __init__(self, req: GenerationRequest, seq_id: int, length: int) -> None:
        self.req = req
        self.seq_id = seq_id
        self.length = length  # tokens currently in KV cache
        self.prefilled = 0  # prompt tokens prefetched so far (chunked prefill)
        self.spec_index = None  # lazy NGramIndex (speculative proposer)
        self.pen_slot = None    # repetition-penalty seen-mask pool row


class InferenceEngine:
    def __init__(
        self,
        model: str | ModelSpec,
        device: Optional[str] = None,
        dtype: torch.dtype = torch.bfloat16,
        model_path: Optional[str] = None,
        max_batch: int = 64,
        max_seq_len: Optional[int] = None,
        use_graphs: Optional[bool] = None,
        seed: int = 0,
        kv_margin_blocks: int = 8,
        max_prefill_tokens: int = 4096,
        spec_decode: bool = False,
        spec_k: int = 4,
        spec_ngram: int = 2,
        kv_dtype: str = "native",
    ) -> None:
        # cap on prompt tokens prefill-batched per step: bounds time-to-first
        # -token for requests behind a burst (they decode while later
        # arrivals prefill)
        self.max_prefill_tokens = max_prefill_tokens
        # speculative decoding (prompt-lookup / n-gram proposer + exact
        # greedy verification over the paged-history prefill path); output-
        # invariant: verified tokens equal plain greedy decode
        self.spec_decode = spec_decode
        self.spec_k = spec_k
        self.spec_ngram = spec_ngram
        self.spec_stats = {"steps": 0, "proposed": 0, "accepted": 0,
                           "delegated": 0}
        self.spec = model if isinstance(model, ModelSpec) else resolve_spec(model, model_path)
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        on_gpu = self.device.type == "cuda"
        if not on_gpu and dtype is torch.bfloat16:
            dtype = torch.float32  # CPU reference path runs fp32
        self.dtype = dtype
        self.max_batch = max_batch
        self.max_seq_len = min(max_seq_len or 4096, self.spec.max_seq_len)
        # the decode combine kernel stages <=128 chunk (m,l) pairs in LDS
        # (attn_decode.hip ml_s): clamp at construction instead of raising
        # mid-step and tearing down every active request
        DECODE_MAX_CTX = 128 * 256
        if self.max_seq_len > DECODE_MAX_CTX:
            logger.warning(
                "max_seq_len %d exceeds the decode kernel's %d-token context"
                " limit; clamping", self.max_seq_len, DECODE_MAX_CTX)
            self.max_seq_len = DECODE_MAX_CTX
        self.tokenizer = load_tokenizer(
            model_path, self.spec.vocab_size, self.spec.bos_token_id, self.spec.eos_token_id
        )

        if on_gpu:
            # fail loudly if the HIP extension is absent (no silent eager path)
            from .. import ops

            ops.require_hip()

        logger.info(
            "building %s (%.2fB params) on %s dtype=%s",
            self.spec.name,
            self.spec.n_params() / 1e9,
            self.device,
            dtype,
        )
        self.weights = ModelWeights(self.spec, self.device, dtype)
        if model_path:
            self.weights.load_hf(model_path)
        else:
            self.weights.random_init(seed=seed)

        blocks_per_seq = -(-self.max_seq_len // kv_mod.BLOCK_SIZE)
        n_blocks = max_batch * blocks_per_seq + kv_margin_blocks
        # "fp8": OCP e4m3 KV pool — half the decode-attention bytes; the
        # chunked-prefill and speculative paths read the cache as bf16 and
        # are not fp8-enabled yet, so they are rejected up front
        self.kv_dtype = kv_dtype
        if kv_dtype == "fp8" and spec_decode:
            raise ValueError("fp8 KV does not support spec_decode yet")
        self.kv = PagedKV(self.spec, self.device, dtype, n_blocks=n_blocks,
                          kv_dtype=kv_dtype)
        self.runner = Runner(self.spec, self.weights, self.kv, self.device, dtype)

        # MoE captures while every decode bucket takes a capture-safe path:
        # dense all-experts bmm, or the grouped-GEMM kernel (device-side
        # segment sizes, no host sync) — only the padded-bmm fallback above
        # the grouped range blocks capture.
        # spec mode keeps graphs: steps where NO sequence has a proposal
        # delegate to the captured decode path (neutral cost on workloads
        # with nothing to speculate on)
        graphs_ok = on_gpu and (
            not self.spec.is_moe
            or Runner.moe_graph_capturable(self.spec, max_batch)
        )
        self.use_graphs = graphs_ok if use_graphs is None else (use_graphs and graphs_ok)
        self.graphs: Optional[DecodeGraphs] = None
        if self.use_graphs:
            self.graphs = DecodeGraphs(self.runner, max_batch, blocks_per_seq)
        # scratch sequence backing pad rows of graph-bucket batches
        self._scratch_seq = -1
        self.kv.new_seq(self._scratch_seq)
        self.kv.extend_seq(self._scratch_seq, 1)

        self._gen = torch.Generator(device=self.device)
        self._gen.manual_seed(seed)
        # repetition-penalty seen-token masks: one pooled [slots, V] bool
        # buffer, maintained INCREMENTALLY (prompt scatter at admission, one
        # batched scatter of the sampled tokens per step) — the r2 parity
        # change made penalty 1.15 the wire default, and rebuilding a
        # [B, context] history tensor on the host every step cost ms-scale
        # GIL time on the serving path. Lazy: never allocated for
        # penalty-1.0-only workloads (bench).
        self._pen_pool: Optional[torch.Tensor] = None
        self._pen_free: List[int] = []

        self._busy_s = 0.0       # engine-thread time inside working steps
        self._busy_steps = 0
        self._born = time.time()
        self._pending: "queue.Queue[GenerationRequest]" = queue.Queue()
        self._active: List[_Active] = []
        # admitted requests whose prompt is longer than one prefill budget:
        # they prefill one chunk per step (vs the paged history) while the
        # active set keeps decoding — chunked prefill
        self._prefilling: List[_Active] = []
        self._dec_seqs = None  # device-state cache key (active seq ids)
        self._last_sampled: Optional[torch.Tensor] = None
        self._eager_state = None
        self._next_seq = 0
        self._stop = False
        self._wake = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._tok_window: List[tuple] = []  # (ts, n_tokens)
        self.total_tokens = 0

    # ------------------------------------------------------------ lifecycle

    def start(self) -> None:
        if self._thread is None:
            self._thread = threading.Thread(target=self._loop, daemon=True, name="engine")
            self._thread.start()

    def shutdown(self) -> None:
        self._stop = True
        self._wake.set()
        if self._thread is not None:
            self._thread.join(timeout=30)
            self._thread = None

    # -------------------------------------------------------------- serving

    def submit(self, req: GenerationRequest) -> GenerationRequest:
        # empty prompts become [bos] at admission (engine loop); <1
        # max_new behaves as 1 (prefill completion always emits one token)
        req.max_new_tokens = max(1, int(req.max_new_tokens))
        self.start()
        self._pending.put(req)
        self._wake.set()
        return req

    def generate(
        self,
        prompt_ids: List[int],
        max_new_tokens: int = 128,
        temperature: Optional[float] = 0.7,
        stop_token_ids: tuple = (),
        on_token=None,
        top_p: Optional[float] = None,
        top_k: Optional[int] = None,
        repetition_penalty: Optional[float] = None,
    ) -> GenerationRequest:
        """Blocking convenience wrapper: submit + drain the stream."""
        req = GenerationRequest(
            prompt_ids=list(prompt_ids),
            max_new_tokens=max_new_tokens,
            sampling=SamplingParams.from_request(
                temperature, top_p, top_k, repetition_penalty),
            stop_token_ids=tuple(stop_token_ids),
        )
        self.submit(req)
        while True:
            item = req.out_queue.get()
            if item is _STREAM_END:
                break
            if on_token is not None:
                on_token(item)
        if req.error:
            raise RuntimeError(req.error)
        return req

    def generate_text(
        self,
        prompt: str,
        max_new_tokens: int = 128,
        temperature: Optional[float] = 0.7,
        on_text=None,
        stop: Optional[List[str]] = None,
        repetition_penalty: Optional[float] = None,
        top_p: Optional[float] = None,
        top_k: Optional[int] = None,
        cancel=None,
    ) -> Dict[str, Any]:
        """Text-level wrapper used by the mesh service. `stop` strings
        truncate the output at the first occurrence (reference stop-word
        scan, bee2bee/hf.py:111-136). `cancel` (a threading.Event) stops
        generation at the next emitted token — a disconnected client must
        not keep burning decode steps."""
        t0 = time.time()
        ids = self.tokenizer.encode(prompt)
        ids = ids[-(self.max_seq_len - max_new_tokens - 1) :]
        # emit the first token at once (TTFT), then flush every few tokens:
        # per-token queue/HTTP hops dominate the serving path at high
        # concurrency; split multibyte chars are held back by the decoder,
        # split stop strings by the stop filter (which also cancels
        # generation the moment a stop completes)
        decoder = TextStreamDecoder(self.tokenizer, flush_every=4)
        stop_filter = StopStringFilter(stop)

        def _on_token(tok: int) -> None:
            if on_text is None:
                return
            out = stop_filter.feed(decoder.delta(req.output_ids))
            if out:
                on_text(out)
            if stop_filter.done:
                req.cancelled = True

        stop_ids = ()
        eos = getattr(self.tokenizer, "eos_token_id", None)
        if eos is not None:
            stop_ids = (eos,)
        sp = SamplingParams.from_request(temperature, top_p, top_k,
                                         repetition_penalty)
        req = GenerationRequest(
            prompt_ids=ids,
            max_new_tokens=max_new_tokens,
            sampling=sp,
            stop_token_ids=stop_ids,
        )
        self.submit(req)
        while True:
            item = req.out_queue.get()
            if item is _STREAM_END:
                break
            _on_token(item)
            if cancel is not None and cancel.is_set():
                req.cancelled = True  # engine stops at the next step
        if req.error:
            raise RuntimeError(req.error)
        text = self.tokenizer.decode(req.output_ids)
        if stop:
            cut = min(
                (text.find(w) for w in stop if w and text.find(w) >= 0),
                default=-1,
            )
            if cut >= 0:
                text = text[:cut]
        if on_text is not None:
            tail = stop_filter.feed(decoder.delta(req.output_ids, final=True))
            tail += stop_filter.flush()
            if tail:
                on_text(tail)
        now = time.time()
        ft = req.first_token_ts or now
        adm = req.admitted_ts or req.submit_ts
        return {
            "text": text,
            "tokens": len(req.output_ids),
            "latency_ms": int((now - t0) * 1000),
            "ttft_ms": int((ft - t0) * 1000),
            # per-stage request trace (the reference reports one opaque
            # latency_ms; SURVEY §5 calls for stage timestamps)
            "timing": {
                "queue_ms": int((adm - req.submit_ts) * 1000),
                "prefill_ms": int((ft - adm) * 1000),
                "decode_ms": int(((req.done_ts or now) - ft) * 1000),
            },
        }

    # ---------------------------------------------------------- engine loop

    def _loop(self) -> None:
        prof = None
        if os.environ.get("BEE2BEE_PROFILE_ENGINE") == "1":
            import cProfile

            prof = cProfile.Profile()
            prof.enable()
        while not self._stop:
            t0 = time.perf_counter()
            try:
                did_work = self._step()
            except Exception as e:  # engine errors fail all active requests
                logger.exception("engine step failed")
                for a in self._active + self._prefilling:
                    a.req.error = str(e)
                    if a.req.on_emit is not None:
                        try:
                            a.req.on_emit(None, True)
                        except Exception:
                            pass
                    else:
                        a.req.out_queue.put(_STREAM_END)
                    self.kv.free_seq(a.seq_id)
                    self._pen_release(a)
                self._active.clear()
                self._prefilling.clear()
                did_work = True
            if did_work:
                self._busy_s += time.perf_counter() - t0
                self._busy_steps += 1
            if not did_work:
                self._wake.wait(timeout=0.05)
                self._wake.clear()
        if prof is not None:
            import pstats

            prof.disable()
            os.makedirs("gpurun_out", exist_ok=True)
            with open("gpurun_out/engine_profile.txt", "w") as f:
                st = pstats.Stats(prof, stream=f)
                st.sort_stats("tottime").print_stats(40)
                st.sort_stats("cumulative").print_stats(40)

    def _step(self) -> bool:
        admitted = self._admit()
        worked = bool(admitted) or bool(self._prefilling)
        if worked:
            self._prefill(self._prefilling + admitted)
        if self._active:
            if self.spec_decode:
                self._decode_spec_once()
            else:
                self._decode_once()
            return True
        return worked

    def _need_blocks(self, prompt_len: int, max_new: int) -> int:
        need_len = min(prompt_len + max_new, self.max_seq_len)
        return -(-need_len // self.kv.block_size)

    def _admit(self) -> List[_Active]:
        admitted: List[_Active] = []
        admit_tokens = 0
        # blocks the current residents will still grow into: admission must
        # leave room for them or a later extend_seq blows the pool mid-step
        reserved = 0
        for a in self._active + self._prefilling:
            need = self._need_blocks(len(a.req.prompt_ids), a.req.max_new_tokens)
            reserved += max(0, need - self.kv.seq_n_blocks(a.seq_id))
        # count in-flight chunked prefills too: they join _active later and
        # must not push it past max_batch (graph buckets size to it)
        while (len(self._active) + len(self._prefilling) + len(admitted)
               < self.max_batch):
            if admitted and admit_tokens >= self.max_prefill_tokens:
                break
            try:
                req = self._pending.get_nowait()
            except queue.Empty:
                break
            blocks_needed = self._need_blocks(
                len(req.prompt_ids), req.max_new_tokens
            )
            if self.kv.free_blocks - reserved < blocks_needed:
                # out of KV memory: push back and wait for retirements
                self._pending.put(req)
                break
            reserved += blocks_needed
            seq_id = self._next_seq
            self._next_seq += 1
            self.kv.new_seq(seq_id)
            a_new = _Active(req, seq_id, 0)
            if req.sampling.repetition_penalty != 1.0:
                self._pen_assign(a_new)
            admitted.append(a_new)
            admit_tokens += len(req.prompt_ids)
        return admitted

    @torch.no_grad()
    def _prefill(self, queue_: List[_Active]) -> None:
        """Prefill up to max_prefill_tokens prompt tokens across the queued
        requests. Whole prompts take the packed varlen fast path; prompts
        longer than the budget prefill in chunks against their paged history
        (ops.attn_prefill_paged) across successive steps."""
        budget = self.max_prefill_tokens
        batch: List[tuple] = []  # (active, start, take)
        n_consumed = 0
        now_admit = time.time()
        for a in queue_:
            if budget <= 0:
                break
            if a.prefilled == 0:
                if a.req.admitted_ts is None:
                    a.req.admitted_ts = now_admit
                p = a.req.prompt_ids or [self.spec.bos_token_id]
                p = p[: self.max_seq_len - a.req.max_new_tokens - 1] or p[:1]
                a.req.prompt_ids = p
            take = min(len(a.req.prompt_ids) - a.prefilled, budget)
            batch.append((a, a.prefilled, take))
            budget -= take
            n_consumed += 1
        # keep EVERYTHING in _prefilling until this step succeeds, so a
        # mid-prefill failure reaches the error path (which fails + frees
        # _active and _prefilling) instead of orphaning admitted requests
        self._prefilling = list(queue_)

        dev = self.device
        ids_list, pos_list, slot_list, cu = [], [], [], [0]
        for a, start, take in batch:
            p = a.req.prompt_ids
            self.kv.extend_seq(a.seq_id, start + take)
            a.prefilled = a.length = start + take
            ids_list.extend(p[start : start + take])
            pos_list.extend(range(start, start + take))
            slot_list.extend(
                self.kv.slot_mapping(a.seq_id, range(start, start + take))
            )
            cu.append(cu[-1] + take)
        input_ids = torch.tensor(ids_list, dtype=torch.int64).to(dev, non_blocking=True)
        positions = torch.tensor(pos_list, dtype=torch.int32).to(dev, non_blocking=True)
        slots = torch.tensor(slot_list, dtype=torch.int32).to(dev, non_blocking=True)
        cu_seqlens = torch.tensor(cu, dtype=torch.int32).to(dev, non_blocking=True)
        max_len = max(take for _, _, take in batch)
        whole = all(
            start == 0 and take == len(a.req.prompt_ids)
            for a, start, take in batch
        )
        if whole:
            hidden = self.runner.forward_prefill(
                input_ids, positions, slots, cu_seqlens, max_len
            )
        else:
            bt = self.kv.block_table([a.seq_id for a, _, _ in batch])
            seq_lens = torch.tensor([start + take for _, start, take in batch],
                dtype=torch.int32, device=dev,
            )
            query_lens = torch.tensor(
                [take for _, _, take in batch], dtype=torch.int32).to(dev, non_blocking=True)
            hidden = self.runner.forward_prefill(
                input_ids, positions, slots, cu_seqlens, max_len,
                block_table=bt, seq_lens=seq_lens, query_lens=query_lens,
            )
        self._prefilling = [
            a
            for a, start, take in batch
            if start + take < len(a.req.prompt_ids)
        ] + [t for t in queue_[n_consumed:]]
        # sample only for requests whose whole prompt is now in the cache
        done_idx = [
            i
            for i, (a, start, take) in enumerate(batch)
            if start + take == len(a.req.prompt_ids)
        ]
        if not done_idx:
            return
        completed = [batch[i][0] for i in done_idx]
        last_rows = torch.tensor([cu[i + 1] - 1 for i in done_idx], dtype=torch.int64).to(dev, non_blocking=True)
        logits = self.runner.lm_head(hidden[last_rows])
        self._sample_and_emit(completed, logits, update_last=False)
        for a in completed:
            if a.req.done_ts is not None:
                # finished AT prefill (max_new=1 / instant stop token):
                # retire here or the KV blocks leak until shutdown
                self.kv.free_seq(a.seq_id)
                self._pen_release(a)
            else:
                self._active.append(a)

    @torch.no_grad()
    def _decode_once(self) -> None:
        """One decode step over the active set.

        Host work is incremental: the device-side ids/positions/lens/block
        tables persist across steps and are bumped in place; full rebuilds
        happen only when the active set changes membership or a sequence
        crosses a KV-block boundary (every block_size tokens)."""
        B = len(self._active)
        dev = self.device
        blocks_changed = False
        for a in self._active:
            nblk0 = (a.length + self.kv.block_size - 1) // self.kv.block_size
            self.kv.extend_seq(a.seq_id, a.length + 1)
            nblk1 = (a.length + self.kv.block_size) // self.kv.block_size
            blocks_changed |= nblk1 != nblk0

        seqs = [a.seq_id for a in self._active]
        stable = (
            seqs == self._dec_seqs
            and self._last_sampled is not None
            and self._last_sampled.shape[0] >= B
        )

        if self.graphs is not None:
            g = self.graphs
            bucket = g.bucket_for(B)
            if stable:
                g.input_ids[:B].copy_(self._last_sampled[:B])
                g.positions[:B].add_(1)
                g.seq_lens[:B].add_(1)
                if blocks_changed:
                    bt = self.kv.block_table(seqs)
                    g.block_table[:B, : bt.shape[1]].copy_(bt, non_blocking=True)
            else:
                last_ids = [
                    (a.req.output_ids[-1] if a.req.output_ids else a.req.prompt_ids[-1])
                    for a in self._active
                ]
                g.input_ids[:B].copy_(
                    torch.tensor(last_ids, dtype=torch.int64), non_blocking=True
                )
                g.positions[:B].copy_(
                    torch.tensor([a.length for a in self._active],
                                 dtype=torch.int32), non_blocking=True
                )
                g.seq_lens[:B].copy_(
                    torch.tensor([a.length + 1 for a in self._active],
                                 dtype=torch.int32), non_blocking=True
                )
                bt = self.kv.block_table(seqs)
                g.block_table[:B, : bt.shape[1]].copy_(bt, non_blocking=True)
                if bucket > B:  # pad rows -> scratch sequence, length 1
                    sb = self.kv.block_table([self._scratch_seq])[0]
                    g.block_table[B:bucket, : sb.shape[0]].copy_(sb)
                    g.positions[B:bucket].zero_()
                    g.seq_lens[B:bucket].fill_(1)
                    g.input_ids[B:bucket].zero_()
                self._dec_seqs = list(seqs)
            logits = g.run(B)[:B]
        else:
            st = self._eager_state
            if stable and st is not None:
                st["ids"].copy_(self._last_sampled[:B])
                st["pos"].add_(1)
                st["lens"].add_(1)
                if blocks_changed:
                    st["bt"] = self.kv.block_table(seqs)
            else:
                last_ids = [
                    (a.req.output_ids[-1] if a.req.output_ids else a.req.prompt_ids[-1])
                    for a in self._active
                ]
                st = {
                    "ids": torch.tensor(last_ids, dtype=torch.int64).to(dev, non_blocking=True),
                    "pos": torch.tensor([a.length for a in self._active], dtype=torch.int32).to(dev, non_blocking=True),
                    "lens": torch.tensor([a.length + 1 for a in self._active], dtype=torch.int32).to(dev, non_blocking=True),
                    "bt": self.kv.block_table(seqs),
                }
                self._eager_state = st
                self._dec_seqs = list(seqs)
            slots = decode_slot_mapping(st["bt"], st["pos"], self.kv.block_size)
            hidden = self.runner.forward_decode(
                st["ids"], st["pos"], slots, st["bt"], st["lens"]
            )
            logits = self.runner.lm_head(hidden)

        for a in self._active:
            a.length += 1
        self._sample_and_emit(self._active, logits)
        done = [a for a in self._active if a.req.done_ts is not None]
        for a in done:
            self.kv.free_seq(a.seq_id)
            self._pen_release(a)
        if done:
            self._active = [a for a in self._active if a.req.done_ts is None]
            self._dec_seqs = None  # membership changed

    # ------------------------------------------------- speculative decoding

    def _propose(self, a: "_Active", ids: List[int], k: int) -> List[int]:
        """Longest-match-first prompt-lookup via an incremental per-sequence
        n-gram index (engine/spec.py) — covers prompt AND output history at
        O(1) maintenance per token instead of the O(context) rescan of the
        naive scan; n sizes tried {spec_ngram+2, spec_ngram+1, spec_ngram}."""
        if k <= 0:
            return []
        idx = getattr(a, "spec_index", None)
        if idx is None:
            from .spec import NGramIndex

            n = self.spec_ngram
            idx = NGramIndex(ids, ns=(n + 2, n + 1, n))
            a.spec_index = idx
        else:
            idx.sync(ids)
        return idx.propose(k)

    @torch.no_grad()
    def _decode_spec_once(self) -> None:
        """One speculative step: each greedy sequence verifies its pending
        token plus up to spec_k proposed tokens in ONE forward through the
        chunked-prefill (paged-history) path; the accepted prefix is exactly
        what plain greedy decode would emit (verification is exact)."""
        dev = self.device
        acts = self._active
        # propose first: when nothing is speculatable this step, take the
        # plain (hipGraph-captured) decode path instead of the slower
        # verify-forward — spec mode then costs ~nothing on workloads
        # without self-similarity
        props: List[List[int]] = []
        for a in acts:
            r = a.req
            prop: List[int] = []
            # greedy with repetition penalty is ALSO speculatable: the
            # verify pass applies the penalty exactly (seen-mask + the
            # in-window proposal prefix) before each argmax
            if r.sampling.greedy and (
                    r.sampling.repetition_penalty == 1.0
                    or a.pen_slot is not None):
                room = self.max_seq_len - (a.length + 1) - 1
                rem = r.max_new_tokens - len(r.output_ids) - 1
                kcap = min(self.spec_k, room, rem)
                prop = self._propose(a, r.prompt_ids + r.output_ids, kcap)
            props.append(prop)
        if not any(props):
            self.spec_stats["delegated"] += 1
            self._decode_once()
            return
        ids_list, pos_list, slot_list, cu = [], [], [], [0]
        for a, prop in zip(acts, props):
            r = a.req
            ctx = r.prompt_ids + r.output_ids
            toks = [ctx[-1]] + prop
            self.kv.extend_seq(a.seq_id, a.length + len(toks))
            ids_list.extend(toks)
            pos_list.extend(range(a.length, a.length + len(toks)))
            slot_list.extend(
                self.kv.slot_mapping(a.seq_id, range(a.length, a.length + len(toks)))
            )
            cu.append(cu[-1] + len(toks))
        input_ids = torch.tensor(ids_list, dtype=torch.int64).to(dev, non_blocking=True)
        positions = torch.tensor(pos_list, dtype=torch.int32).to(dev, non_blocking=True)
        slots = torch.tensor(slot_list, dtype=torch.int32).to(dev, non_blocking=True)
        cu_t = torch.tensor(cu, dtype=torch.int32).to(dev, non_blocking=True)
        qlens = [cu[i + 1] - cu[i] for i in range(len(acts))]
        bt = self.kv.block_table([a.seq_id for a in acts])
        seq_lens = torch.tensor([a.length + q for a, q in zip(acts, qlens)],
            dtype=torch.int32, device=dev,
        )
        qlens_t = torch.tensor(qlens, dtype=torch.int32).to(dev, non_blocking=True)
        hidden = self.runner.forward_prefill(
            input_ids, positions, slots, cu_t, max(qlens),
            block_table=bt, seq_lens=seq_lens, query_lens=qlens_t,
        )
        logits = self.runner.lm_head(hidden)

        greedy_idx = [i for i, a in enumerate(acts)
                      if a.req.sampling.greedy
                      and a.req.sampling.repetition_penalty == 1.0]
        pen_idx = [i for i, a in enumerate(acts)
                   if a.req.sampling.greedy
                   and a.req.sampling.repetition_penalty != 1.0
                   and a.pen_slot is not None]
        handled = set(greedy_idx) | set(pen_idx)
        nong = [a for i, a in enumerate(acts) if i not in handled]
        if greedy_idx:
            argmax = logits.argmax(dim=-1).cpu()
        pen_argmax = {}
        if pen_idx:
            # penalized argmax under the BASE seen mask for every verify
            # row in one batch + one sync; in-window corrections happen on
            # the host walk below
            row_ids, row_act, row_p = [], [], []
            for i in pen_idx:
                q = cu[i + 1] - cu[i]
                row_ids.extend(range(cu[i], cu[i] + q))
                row_act.extend([i] * q)
                row_p.extend([acts[i].req.sampling.repetition_penalty] * q)
            rid = torch.tensor(row_ids, dtype=torch.int64).to(dev, non_blocking=True)
            slot_t = torch.tensor([acts[i].pen_slot for i in row_act], dtype=torch.int64).to(dev, non_blocking=True)
            p_t = torch.tensor(row_p, dtype=torch.float32).to(dev, non_blocking=True).unsqueeze(1)
            gl = logits[rid].float()
            seen = self._pen_pool[slot_t]
            pen_rows = torch.where(
                seen, torch.where(gl > 0, gl / p_t, gl * p_t), gl)
            pa = pen_rows.argmax(dim=-1).cpu()
            for k, row in enumerate(row_ids):
                pen_argmax[row] = int(pa[k])
        now = time.time()
        n_emitted = 0
        pen_new_slots: List[int] = []
        pen_new_toks: List[int] = []
        for i in greedy_idx + pen_idx:
            a, r, prop = acts[i], acts[i].req, props[i]
            self.spec_stats["proposed"] += len(prop)
            emitted: List[int] = []
            j = 0
            if a.req.sampling.repetition_penalty == 1.0:
                while True:
                    tok = int(argmax[cu[i] + j])
                    emitted.append(tok)
                    if (j < len(prop) and tok == prop[j]
                            and tok not in r.stop_token_ids):
                        j += 1
                        continue
                    break
            else:
                # exact penalized walk: row j's base-mask argmax is exact
                # unless it collides with a token proposed EARLIER in this
                # window (whose in-window penalty could demote it) — at a
                # collision, truncate acceptance (still exact: every
                # emitted token equals what plain penalized decode emits)
                window: set = set()
                while True:
                    tok = pen_argmax[cu[i] + j]
                    if j > 0 and tok in window:
                        break  # ambiguous under in-window penalty: stop
                    emitted.append(tok)
                    if (j < len(prop) and tok == prop[j]
                            and tok not in r.stop_token_ids):
                        window.add(tok)
                        j += 1
                        continue
                    break
            self.spec_stats["accepted"] += j
            # rewind the KV length past rejected proposals (their stored
            # keys get overwritten; attention never reads past seq_len)
            a.length += len(emitted)
            self.kv.extend_seq(a.seq_id, a.length)
            for tok in emitted:
                if r.done_ts is not None:
                    break
                if r.first_token_ts is None:
                    r.first_token_ts = now
                r.output_ids.append(tok)
                if a.pen_slot is not None:
                    pen_new_slots.append(a.pen_slot)
                    pen_new_toks.append(tok)
                n_emitted += 1
                done = (r.cancelled or tok in r.stop_token_ids
                        or len(r.output_ids) >= r.max_new_tokens)
                if done:
                    r.done_ts = now
                if r.on_emit is not None:
                    try:
                        r.on_emit(tok, done)
                    except Exception:
                        logger.exception("on_emit callback failed")
                else:
                    r.out_queue.put(tok)
                    if done:
                        r.out_queue.put(_STREAM_END)
        if n_emitted:
            self._note_throughput(n_emitted, now)
        # mark penalized-greedy emissions seen (one batched scatter)
        if pen_new_slots:
            self._pen_pool[
                torch.tensor(pen_new_slots, dtype=torch.int64).to(dev, non_blocking=True),
                torch.tensor(pen_new_toks, dtype=torch.int64).to(dev, non_blocking=True),
            ] = True
        if nong:
            # non-greedy requests take the plain one-token path
            rows = torch.tensor(
                [cu[i + 1] - 1 for i, a in enumerate(acts)
                 if i not in handled],
                dtype=torch.int64,
            ).to(logits.device, non_blocking=True)
            for a in nong:
                a.length += 1
            self._sample_and_emit(nong, logits[rows], update_last=False)
        self.spec_stats["steps"] += 1
        self._dec_seqs = None  # device decode-state caches are stale
        done_acts = [a for a in acts if a.req.done_ts is not None]
        for a in done_acts:
            self.kv.free_seq(a.seq_id)
            self._pen_release(a)
        if done_acts:
            self._active = [a for a in acts if a.req.done_ts is None]

    def _pen_slots_t(self, slots: List[int]) -> torch.Tensor:
        """Device tensor of pen-pool slots, cached against the slot list
        (the active set changes far less often than it steps)."""
        key = tuple(slots)
        cached = getattr(self, "_pen_slot_cache", None)
        if cached is not None and cached[0] == key:
            return cached[1]
        t = torch.tensor(slots, dtype=torch.int64, device=self.device)
        self._pen_slot_cache = (key, t)
        return t

    def _pen_assign(self, a: "_Active") -> None:
        """Give a penalized request a seen-mask row seeded with its prompt."""
        V = self.spec.vocab_size
        if self._pen_pool is None:
            n = self.max_batch + 8
            self._pen_pool = torch.zeros(n, V, dtype=torch.bool,
                                         device=self.device)
            self._pen_free = list(range(n))
        if not self._pen_free:
            return  # degrade: penalty falls back to the host rebuild below
        slot = self._pen_free.pop()
        a.pen_slot = slot
        row = self._pen_pool[slot]
        row.zero_()
        ids = torch.tensor(a.req.prompt_ids, dtype=torch.int64,
                           device=self.device)
        row[ids] = True

    def _pen_release(self, a: "_Active") -> None:
        if a.pen_slot is not None:
            self._pen_free.append(a.pen_slot)
            a.pen_slot = None

    def _sample_and_emit(self, acts: List[_Active], logits: torch.Tensor,
                         update_last: bool = True) -> None:
        # group rows by sampling params so each group is one sample() call
        groups: Dict[tuple, List[int]] = {}
        for i, a in enumerate(acts):
            sp = a.req.sampling
            groups.setdefault(
                (sp.greedy, sp.temperature, sp.top_p, sp.top_k,
                 sp.repetition_penalty), []
            ).append(i)
        # CAUTION (measured): torch.tensor(..., device=cuda) on this path
        # is a pageable H2D copy that synchronizes with the busy stream —
        # profiled at ~0.8 ms per call, 68% of the engine thread under
        # load. The single-group case (uniform sampling params — the
        # common serving mix) therefore indexes nothing, and the penalty
        # slot tensors are cached against the active-set signature.
        single = len(groups) == 1
        next_dev = None if single else torch.empty(
            len(acts), dtype=torch.int64, device=logits.device
        )
        for key, rows in groups.items():
            sp = acts[rows[0]].req.sampling
            gen = self._gen if self.device.type == "cuda" else None
            if single:
                idx = None
                grp_logits = logits
            else:
                idx = torch.tensor(rows, dtype=torch.int64,
                                   device=logits.device)
                grp_logits = logits[idx]
            if sp.repetition_penalty != 1.0:
                slots = [acts[r].pen_slot for r in rows]
                if self._pen_pool is not None and all(
                        sl is not None for sl in slots):
                    seen = self._pen_pool[self._pen_slots_t(slots)]
                    p = sp.repetition_penalty
                    gl = grp_logits.float()
                    grp_logits = torch.where(
                        seen, torch.where(gl > 0, gl / p, gl * p), gl)
                else:
                    # pool exhausted fallback: host-rebuilt history
                    from .sampler import apply_repetition_penalty

                    width = max(
                        (len(acts[r].req.output_ids)
                         + len(acts[r].req.prompt_ids))
                        for r in rows
                    )
                    prev = torch.full((len(rows), width), -1,
                                      dtype=torch.int64)
                    for j, r in enumerate(rows):
                        ids = acts[r].req.prompt_ids + acts[r].req.output_ids
                        prev[j, : len(ids)] = torch.tensor(
                            ids, dtype=torch.int64)
                    grp_logits = apply_repetition_penalty(
                        grp_logits.float(), prev.to(grp_logits.device),
                        sp.repetition_penalty,
                    )
            toks = sample(grp_logits, sp, generator=gen)
            if single:
                next_dev = toks
            else:
                next_dev[idx] = toks
        # mark the freshly sampled tokens seen (one batched device scatter)
        if self._pen_pool is not None:
            pen_idx = [i for i, a in enumerate(acts) if a.pen_slot is not None]
            if len(pen_idx) == len(acts):
                self._pen_pool[
                    self._pen_slots_t([a.pen_slot for a in acts]), next_dev
                ] = True
            elif pen_idx:
                slot_t = self._pen_slots_t([acts[i].pen_slot for i in pen_idx])
                idx_t = torch.tensor(pen_idx, dtype=torch.int64,
                                     device=logits.device)
                self._pen_pool[slot_t, next_dev[idx_t]] = True
        if update_last:
            # feeds the next decode step without an H2D copy. Callers whose
            # `acts` is NOT the current decode set (prefill completions,
            # spec-decode subsets) must pass update_last=False: the decode
            # stable-path copies _last_sampled[:B] as the next inputs, and a
            # clobber from another batch feeds WRONG TOKENS (a request once
            # decoded its neighbor's token — caught by the scheduling-
            # invariance property test).
            self._last_sampled = next_dev
        next_ids = next_dev.cpu()  # the one host sync per step (emission)
        now = time.time()
        n_emitted = 0
        for i, a in enumerate(acts):
            tok = int(next_ids[i])
            r = a.req
            if r.first_token_ts is None:
                r.first_token_ts = now
            r.output_ids.append(tok)
            n_emitted += 1
            done = (r.cancelled or tok in r.stop_token_ids
                    or len(r.output_ids) >= r.max_new_tokens)
            if done:
                r.done_ts = now
            if r.on_emit is not None:
                try:
                    r.on_emit(tok, done)
                except Exception:
                    logger.exception("on_emit callback failed")
            else:
                r.out_queue.put(tok)
                if done:
                    r.out_queue.put(_STREAM_END)
        self._note_throughput(n_emitted, now)

    def stats(self) -> Dict[str, Any]:
        """Live engine observability (served at the API home endpoint)."""
        now = time.time()
        span = (now - self._tok_window[0][0]) if len(self._tok_window) > 1 else 1.0
        tps = sum(x[1] for x in self._tok_window) / max(span, 1e-3)
        out = {
            "model": self.spec.name,
            "device": str(self.device),
            "kv_dtype": self.kv_dtype,
            "active_requests": len(self._active),
            "prefilling_requests": len(self._prefilling),
            "queued_requests": self._pending.qsize(),
            "kv_free_blocks": self.kv.free_blocks,
            "kv_total_blocks": self.kv.n_blocks,
            "tokens_total": self.total_tokens,
            "tokens_per_sec_10s": round(tps, 1),
            "decode_graphs": self.graphs is not None,
            "engine_busy_s": round(self._busy_s, 2),
            "engine_steps": self._busy_steps,
            "engine_ms_per_step": round(
                1e3 * self._busy_s / max(self._busy_steps, 1), 2),
        }
        if self.spec_decode:
            out["spec_decode"] = dict(self.spec_stats)
        return out

    def _note_throughput(self, n: int, now: float) -> None:
        self.total_tokens += n
        self._tok_window.append((now, n))
        cutoff = now - 10.0
        while self._tok_window and self._tok_window[0][0] < cutoff:
            self._tok_window.pop(0)
        span = now - self._tok_window[0][0] if len(self._tok_window) > 1 else 1.0
        tps = sum(x[1] for x in self._tok_window) / max(span, 1e-3)
        report_engine_throughput(tps)

    # ------------------------------------------------ direct batch API (bench)

    @torch.no_grad()
    def bench_prefill(self, batch: int, prompt_len: int, seed: int = 1234,
                      max_new: Optional[int] = None) -> List[int]:
        """Fill the KV cache with synthetic prompts; returns seq_ids.
        max_new must leave room for the prompt under max_seq_len or the
        admission truncation rule will cut the prompts short (a 10**9
        sentinel used to truncate every bench prompt to one token)."""
        if max_new is None:
            max_new = max(1, self.max_seq_len - prompt_len - 1)
        assert prompt_len + max_new < self.max_seq_len + 1, "prompt would truncate"
        gen = torch.Generator().manual_seed(seed)
        seq_ids = []
        for b in range(batch):
            seq_id = self._next_seq
            self._next_seq += 1
            self.kv.new_seq(seq_id)
            seq_ids.append(seq_id)
        ids = torch.randint(
            4, self.spec.vocab_size, (batch, prompt_len), generator=gen
        )
        acts = []
        for b, seq_id in enumerate(seq_ids):
            req = GenerationRequest(
                prompt_ids=ids[b].tolist(), max_new_tokens=max_new,
                sampling=SamplingParams(greedy=True),
            )
            acts.append(_Active(req, seq_id, 0))
        self._prefill_no_admit(acts)
        self._bench_acts = acts
        for a in acts:
            assert a.prefilled == prompt_len, "bench prompt truncated"
        return seq_ids

    @torch.no_grad()
    def _prefill_no_admit(self, acts: List[_Active]) -> None:
        """_prefill for the bench path, without touching the serving active
        set — looping until EVERY prompt is fully in the cache (_prefill
        consumes at most max_prefill_tokens per call, which also bounds the
        transient activation memory to ~max_prefill_tokens rows)."""
        saved_active = self._active
        saved_prefilling = self._prefilling
        self._active = []
        self._prefilling = []
        self._prefill(list(acts))
        while self._prefilling:
            self._prefill(list(self._prefilling))
        self._active = saved_active
        self._prefilling = saved_prefilling

    @torch.no_grad()
    def bench_setup(self, batch: int, prompt_len: int, steps_budget: int,
                    seed: int = 1234) -> None:
        """Prefill a synthetic batch and preallocate KV for the whole run so
        the timed loop is pure device work (graph replay + sampling + buffer
        bumps; zero host round-trips per step)."""
        assert batch <= self.max_batch
        seq_ids = self.bench_prefill(batch, prompt_len, seed=seed)
        for sid in seq_ids:
            self.kv.extend_seq(sid, prompt_len + steps_budget + 1)
        dev = self.device
        acts = self._bench_acts
        first = torch.tensor([a.req.output_ids[-1] for a in acts], dtype=torch.int64).to(dev, non_blocking=True)
        bt = self.kv.block_table([a.seq_id for a in acts])
        if self.graphs is not None:
            g = self.graphs
            assert g.bucket_for(batch) == batch or batch in g.buckets(), (
                "bench batch should be a graph bucket"
            )
            g.input_ids[:batch].copy_(first)
            g.positions[:batch].fill_(prompt_len)
            g.seq_lens[:batch].fill_(prompt_len + 1)
            g.block_table[:batch, : bt.shape[1]].copy_(bt)
            bucket = g.bucket_for(batch)
            if bucket > batch:
                sb = self.kv.block_table([self._scratch_seq])[0]
                g.block_table[batch:bucket, : sb.shape[0]].copy_(sb)
                g.positions[batch:bucket].zero_()
                g.seq_lens[batch:bucket].fill_(1)
                g.input_ids[batch:bucket].zero_()
            self._bench_state = ("graph", batch)
            self.graphs.run(batch)  # capture outside the timed region
            g.positions[:batch].fill_(prompt_len)
            g.seq_lens[:batch].fill_(prompt_len + 1)
            g.input_ids[:batch].copy_(first)
        else:
            self._bench_eager = {
                "ids": first.clone(),
                "pos": torch.full((batch,), prompt_len, dtype=torch.int32, device=dev),
                "lens": torch.full((batch,), prompt_len + 1, dtype=torch.int32, device=dev),
                "bt": bt,
            }
            self._bench_state = ("eager", batch)

    @torch.no_grad()
    def bench_step(self) -> None:
        """One fully device-side greedy decode step over the bench batch."""
        mode, B = self._bench_state
        if mode == "graph":
            g = self.graphs
            logits = g.run(B)[:B]
            next_ids = torch.argmax(logits, dim=-1)
            g.input_ids[:B].copy_(next_ids)
            g.positions[:B].add_(1)
            g.seq_lens[:B].add_(1)
        else:
            st = self._bench_eager
            slots = decode_slot_mapping(st["bt"], st["pos"], self.kv.block_size)
            hidden = self.runner.forward_decode(
                st["ids"], st["pos"], slots, st["bt"], st["lens"]
            )
            logits = self.runner.lm_head(hidden)
            st["ids"].copy_(torch.argmax(logits, dim=-1))
            st["pos"].add_(1)
            st["lens"].add_(1)
        self.total_tokens += B

    @torch.no_grad()
    def bench_decode_step(self) -> torch.Tensor:
        """One decode step over the bench batch; returns sampled ids [B]."""
        acts = self._bench_acts
        saved = self._active
        self._active = list(acts)
        self._decode_once()
        self._active = saved
        return torch.tensor([a.req.output_ids[-1] for a in acts])
