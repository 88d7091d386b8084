"""LLM engine: continuous batching over the paged-KV Llama model.

Implements SURVEY.md §2.11 component 3: iteration-level scheduling with
prefill/decode interleaving and per-request sampling state. Concurrent
investigations (the eval harness runs up to 32 at once) submit requests
from their own threads; a single engine thread batches whatever is
pending each iteration, so decode batches grow/shrink as agent loops
issue and await LLM calls — replacing "the provider's server does
batching" (reference src/model/llm.ts).

Grammar-constrained decoding: requests carrying a JSON schema get a
JsonFsm; every step the sampler masks logits to the FSM's allowed bytes
(plus engine-level masking to the active byte vocab), so outputs are
schema-valid by construction.
"""
from __future__ import annotations

import threading
import time
from dataclasses import dataclass, field
from typing import Any, Optional

import torch

from .. import ops
from .json_fsm import NUMBER_CLOSE_SENTINEL, _STRING_BYTES, JsonFsm
from .llama import CONFIGS, LlamaModel
from .tokenizer import (
    ACTIVE_VOCAB,
    MAX_WORD_LEN,
    WORD_BASE,
    WORD_STRINGS,
    ByteTokenizer,
    SpecialTokens,
)

MASK_REGION = 1024  # bytes + specials + word tokens, padded
assert ACTIVE_VOCAB <= MASK_REGION, "word vocab outgrew the sampler mask region"

# word ids admissible at a given string capacity (word length <= cap)
_WORD_IDS_BY_CAP: dict[int, list[int]] = {}
for _cap in range(3, MAX_WORD_LEN + 1):
    _WORD_IDS_BY_CAP[_cap] = [WORD_BASE + _i for _i, _w in enumerate(WORD_STRINGS)
                              if len(_w) <= _cap]

# Precomputed mask rows for string-content states (the overwhelmingly common
# sampler state): fancy-indexing ~700 ids per row per step costs ~0.1 ms per
# sequence; a cached row copy is ~free.
_STRING_ROW_CACHE: dict[tuple[bool, int], torch.Tensor] = {}


def _string_mask_row(has_quote: bool, cap: int) -> torch.Tensor:
    cap = min(max(cap, 0), MAX_WORD_LEN)
    key = (has_quote, cap)
    row = _STRING_ROW_CACHE.get(key)
    if row is None:
        row = torch.zeros(MASK_REGION, dtype=torch.bool)
        row[list(_STRING_BYTES)] = True
        if has_quote:
            row[0x22] = True
        if cap >= 3:
            row[_WORD_IDS_BY_CAP[cap]] = True
        _STRING_ROW_CACHE[key] = row
    return row


def _nucleus_mask(logits: torch.Tensor, top_ps: list) -> torch.Tensor:
    """Drop tokens outside each row's top-p nucleus (smallest set whose
    probability mass reaches top_p). Rows with top_p >= 1 pass through."""
    if all(p >= 1.0 for p in top_ps):
        return logits
    probs = torch.softmax(logits.float(), dim=-1)
    sp, si = probs.sort(dim=-1, descending=True)
    cum = sp.cumsum(-1)
    tp = torch.tensor([max(0.01, min(1.0, float(p))) for p in top_ps],
                      device=logits.device, dtype=cum.dtype).unsqueeze(1)
    # a sorted token is dropped when the mass BEFORE it already reaches p
    drop_sorted = (cum - sp) >= tp
    drop = torch.zeros_like(drop_sorted).scatter(1, si, drop_sorted)
    return logits.masked_fill(drop, float("-inf"))


@dataclass
class Request:
    rid: int
    prompt_ids: list[int]
    max_new_tokens: int = 512
    temperature: float = 0.0
    top_p: float = 1.0               # nucleus sampling (1.0 = disabled)
    schema: Optional[dict[str, Any]] = None
    fsm: Optional[JsonFsm] = None
    out_ids: list[int] = field(default_factory=list)
    pending_input: list[int] = field(default_factory=list)  # tokens not yet in KV
    prefill_tokens: list[int] = field(default_factory=list)  # prompt + forced prefix
    cached_len: int = 0              # prefix tokens served from the KV prefix pool
    registered: bool = False         # prompt blocks published to the prefix pool
    dedup_deferred: bool = False     # held back one round behind a same-prefix twin
    state: str = "waiting"           # waiting | running | done
    cancelled: bool = False          # swept at the next step boundary
    done_event: threading.Event = field(default_factory=threading.Event)
    prompt_len: int = 0
    pos: int = 0                     # next position to write
    error: str = ""
    submitted_at: float = field(default_factory=time.time)
    first_token_at: float = 0.0
    finished_at: float = 0.0


class LLMEngine:
    def __init__(
        self,
        model: str = "tiny",
        device: Optional[str] = None,
        tp: Optional[int] = None,
        max_prefill_tokens: int = 8192,
        max_batch: int = 64,
        kv_blocks: Optional[int] = None,
        seed: int = 1234,
        background: bool = True,
        prefix_cache: bool = True,
        checkpoint: Optional[str] = None,
    ) -> None:
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = device
        self.hf_tokenizer = None
        if checkpoint is not None:
            # trained weights: config comes from the checkpoint's config.json;
            # a tokenizer.json beside it switches the engine to full-vocab
            # free decoding (byte-FSM grammar masks only apply to the byte
            # tokenizer — see engine/bpe_tokenizer.py docstring)
            import os as _os

            from .checkpoint import load_model
            self.model = load_model(checkpoint, device=device, tp=tp,
                                    kv_blocks=kv_blocks, seed=seed)
            cfg = self.model.cfg
            tok_file = _os.path.join(checkpoint, "tokenizer.json")
            if _os.path.exists(tok_file):
                from .bpe_tokenizer import BpeTokenizer

                self.hf_tokenizer = BpeTokenizer.from_file(tok_file)
        else:
            cfg = CONFIGS[model]
            self.model = LlamaModel(cfg, device=device, tp=tp, seed=seed,
                                    kv_blocks=kv_blocks)
        self.cfg = cfg
        if self.model.tp > 1:
            # every rank must replay identical step sequences; keep eager
            # (RCCL collectives inside hipGraph capture are not validated)
            self.model.use_graphs = False
        elif device.startswith("cuda"):
            # capture every decode graph NOW, before request threads exist:
            # a capture racing concurrent GPU work (knowledge search on the
            # default stream) dies with hipErrorStreamCaptureUnsupported
            self.model.capture_decode_graphs()
        self.tokenizer = ByteTokenizer(cfg.vocab_size)
        self.max_prefill_tokens = max_prefill_tokens
        self.max_batch = max_batch
        self.prefix_cache = prefix_cache
        self._rid = 0
        self._lock = threading.Condition()
        self.waiting: list[Request] = []
        self.running: list[Request] = []
        self.stats = {"requests": 0, "prefill_tokens": 0, "decode_tokens": 0,
                      "steps": 0, "prefill_time": 0.0, "decode_time": 0.0,
                      "cached_prefix_tokens": 0}
        from collections import deque
        #: (ttft_s, e2e_s, out_tokens) of recently finished requests
        self._latency_ring: "deque" = deque(maxlen=512)
        self._stop = False
        self._ev_queue: "deque" = deque()      # (phase, ev0, ev1, ev2) pending fold
        self._step_batch: list[Request] = []   # requests in the executing step
        self._thread: Optional[threading.Thread] = None
        if background:
            self._thread = threading.Thread(target=self._loop, daemon=True,
                                            name="llm-engine")
            self._thread.start()

    # -- BPE grammar (trained checkpoints) ----------------------------------------

    _BPE_MASKER_UNSET = object()

    @property
    def bpe_masker(self):
        """Token-trie x FSM masker for the checkpoint tokenizer (None when
        absent or over the size gate) — engine/grammar_bpe.py."""
        if getattr(self, "_bpe_masker", self._BPE_MASKER_UNSET) is self._BPE_MASKER_UNSET:
            if self.hf_tokenizer is None:
                self._bpe_masker = None
            else:
                from .grammar_bpe import build_masker

                self._bpe_masker = build_masker(self.hf_tokenizer)
        return self._bpe_masker

    @property
    def supports_bpe_grammar(self) -> bool:
        return self.bpe_masker is not None

    # -- submission ---------------------------------------------------------------

    def submit(self, prompt_ids: list[int], max_new_tokens: int = 512,
               temperature: float = 0.0, schema: Optional[dict[str, Any]] = None,
               top_p: float = 1.0) -> Request:
        if schema is not None and self.hf_tokenizer is not None \
                and not self.supports_bpe_grammar:
            raise ValueError(
                "grammar-constrained decoding over this BPE vocab exceeds the "
                "token-trie masker's size gate (grammar_bpe.MAX_VOCAB); put the "
                "schema in the prompt instead (LocalEngineClient does this "
                "automatically)")
        with self._lock:
            self._rid += 1
            # clamp to the model's context window (RoPE table bound)
            room = self.cfg.max_seq_len - len(prompt_ids) - 1
            if room <= 0:
                prompt_ids = prompt_ids[-(self.cfg.max_seq_len // 2):]
                room = self.cfg.max_seq_len - len(prompt_ids) - 1
            req = Request(
                rid=self._rid,
                prompt_ids=list(prompt_ids),
                max_new_tokens=max(1, min(max_new_tokens, room)),
                temperature=temperature,
                top_p=max(0.01, min(1.0, float(top_p))),
                schema=schema,
                fsm=JsonFsm(schema) if schema else None,
            )
            req.prompt_len = len(req.prompt_ids)
            # grammar-forced prefix (e.g. '{"summary": "') enters WITH the
            # prompt prefill — zero decode steps spent on forced structure
            prefix = self._drain_forced(req)
            req.pending_input = req.prompt_ids + prefix
            req.prefill_tokens = list(req.pending_input)
            self.waiting.append(req)
            self.stats["requests"] += 1
            self._lock.notify_all()
            return req

    def generate(self, prompt_ids: list[int], max_new_tokens: int = 512,
                 temperature: float = 0.0, schema: Optional[dict[str, Any]] = None,
                 timeout_s: float = 600.0, top_p: float = 1.0) -> Request:
        req = self.submit(prompt_ids, max_new_tokens, temperature, schema,
                          top_p=top_p)
        if self._thread is None:
            self.run_until_idle()
        else:
            req.done_event.wait(timeout=timeout_s)
        if not req.done_event.is_set():
            raise TimeoutError(f"generation timed out after {timeout_s}s")
        if req.error:
            raise RuntimeError(req.error)
        return req

    # -- engine loop ----------------------------------------------------------------

    def _loop(self) -> None:
        while not self._stop:
            with self._lock:
                while not self.waiting and not self.running and not self._stop:
                    self._lock.wait(timeout=1.0)
                if self._stop:
                    return
            try:
                self.step()
            except Exception as e:  # noqa: BLE001 — fail requests, not the thread
                self._fail_step(e)

    def run_until_idle(self, max_steps: int = 100_000) -> None:
        for _ in range(max_steps):
            with self._lock:
                if not self.waiting and not self.running:
                    return
            try:
                self.step()
            except Exception as e:  # noqa: BLE001
                self._fail_step(e)

    def _fail_step(self, e: Exception) -> None:
        """Fail ONLY the requests touched by the failing step (their KV/FSM
        state is suspect); other in-flight investigations keep running. The
        step batch is in neither `waiting` nor `running` while it executes,
        so it must be failed explicitly or its callers block until timeout."""
        msg = f"{type(e).__name__}: {e}"
        self.stats["step_errors"] = self.stats.get("step_errors", 0) + 1
        self.stats["last_error"] = msg
        if self.device.startswith("cuda"):
            try:
                torch.cuda.synchronize()
            except Exception:  # noqa: BLE001 — device itself is gone
                pass
        with self._lock:
            batch = self._step_batch
            self._step_batch = []
            for req in batch:
                req.error = msg
                req.state = "done"
                self.model.kv.free(req.rid)
                req.done_event.set()
            self.running = [r for r in self.running if r not in batch]

    def cancel(self, req: Request) -> bool:
        """Cancel a request (e.g. the HTTP client disconnected). Waiting
        requests finalize immediately; running ones are swept at the next
        step boundary, where their KV blocks are freed. Returns False if
        the request had already finished."""
        with self._lock:
            if req.state == "done":
                return False
            req.cancelled = True
            if req in self.waiting:
                self.waiting.remove(req)
                self._finalize_cancel(req)
                return True
            self._lock.notify_all()
        return True

    def _finalize_cancel(self, req: Request) -> None:
        """Caller holds the lock (or the request is step-private)."""
        req.state = "done"
        req.finished_at = time.time()
        self.model.kv.free(req.rid)
        req.done_event.set()

    def shutdown(self) -> None:
        self._stop = True
        with self._lock:
            self._lock.notify_all()
        if self._thread:
            self._thread.join(timeout=5.0)
        # fail anything still in flight so callers unblock immediately
        # instead of sitting in done_event.wait until their timeout
        with self._lock:
            for req in self.running + self.waiting + self._step_batch:
                if req.state != "done":
                    req.error = req.error or "engine shut down"
                    req.state = "done"
                    req.done_event.set()
            self.running.clear()
            self.waiting.clear()
            self._step_batch = []

    # -- scheduling ------------------------------------------------------------------

    #: pending runs longer than this go through the chunked-prefill path;
    #: shorter runs (grammar-forced bytes, word-token tails) ride the
    #: graphed decode batch one token per iteration — an eager chunk call
    #: costs ~13 ms of CPU launch time (32 layers, no hipGraph) while the
    #: whole decode batch advances alongside for free (measured:
    #: profiles/PERF_LOG.md, scheduling section)
    CHUNK_THRESHOLD = 8

    def step(self) -> None:
        """One engine iteration: admit a prefill batch if any request is
        waiting (and fits); else, if any request has a LONG pending run
        (admission suffix after a prefix-cache hit, long forced run), flush
        all pending runs through the chunked-prefill path; else one decode
        step over all running sequences, each feeding the next pending
        token (sampling only where the queue drained)."""
        with self._lock:
            # sweep cancellations at the step boundary, never mid-batch
            for r in [r for r in self.running if r.cancelled and r.state != "done"]:
                self.running.remove(r)
                self._finalize_cancel(r)
            prefill_batch = self._admit_locked()
        if prefill_batch:
            self._step_batch = prefill_batch
            self._run_prefill(prefill_batch)
        else:
            with self._lock:
                pending = [r for r in self.running if r.pending_input]
                long_run = any(len(r.pending_input) > self.CHUNK_THRESHOLD
                               for r in pending)
            if long_run:
                self._step_batch = pending
                self._run_chunk(pending)
            elif pending:
                self._step_batch = pending
                self._run_decode(pending)
        self._step_batch = []
        self.stats["steps"] += 1

    def _admit_locked(self) -> list[Request]:
        batch: list[Request] = []
        tokens = 0
        kv = self.model.kv
        admitted_first: set = set()   # first full block of each admission this call
        i = 0
        while i < len(self.waiting) and len(self.running) + len(batch) < self.max_batch:
            req = self.waiting[i]
            need = len(req.pending_input) + req.max_new_tokens
            if batch and tokens + len(req.pending_input) > self.max_prefill_tokens:
                break
            if need > (kv.num_blocks - 1) * kv.block_size:
                # can NEVER fit, even with the whole pool free: fail loudly
                # instead of starving the queue forever
                self.waiting.pop(i)
                req.error = (f"request needs {need} KV tokens but the pool holds "
                             f"{(kv.num_blocks - 1) * kv.block_size}; lower "
                             f"max_new_tokens or raise kv_blocks")
                req.state = "done"
                req.done_event.set()
                continue
            if not kv.can_allocate(need):
                break
            fb = (tuple(req.pending_input[:kv.block_size])
                  if len(req.pending_input) >= kv.block_size else None)
            if (self.prefix_cache and fb is not None and fb in admitted_first
                    and not req.dedup_deferred):
                # same first block as a request admitted THIS call: its
                # prefix isn't registered yet, so admitting now would
                # re-prefill it — hold back ONE round (a cold 32-way wave
                # of identical system prompts prefills once, not 32×);
                # the flag guarantees admission next round regardless
                req.dedup_deferred = True
                i += 1
                continue
            cached = 0
            if self.prefix_cache:
                cached = kv.allocate_with_prefix(req.rid, req.pending_input, need)
            else:
                kv.allocate(req.rid, need)
            self.waiting.pop(i)
            if fb is not None:
                admitted_first.add(fb)
            if cached:
                # prefix served from the pool: skip straight to running with
                # only the suffix pending — it flows through the chunked-
                # prefill path (history attention over the shared blocks)
                kv.set_len(req.rid, cached)
                req.pos = cached
                req.cached_len = cached
                req.pending_input = req.pending_input[cached:]
                req.state = "running"
                self.running.append(req)
                self.stats["cached_prefix_tokens"] += cached
            else:
                batch.append(req)
                tokens += len(req.pending_input)
        return batch

    # -- execution --------------------------------------------------------------------

    def _tp_active(self) -> bool:
        from ..parallel.tp_serving import tp_active

        return tp_active(self.model.tp)

    def _tp_dispatch(self, payload: dict) -> torch.Tensor:
        """Broadcast + execute one model step (TP > 1 only): every rank runs
        the same collectives in the same order."""
        from ..parallel.tp_serving import broadcast_step, execute_step

        broadcast_step(payload)
        return execute_step(self.model, payload)

    def _run_prefill(self, batch: list[Request]) -> None:
        t0 = time.time()
        kv = self.model.kv
        token_ids: list[int] = []
        positions: list[int] = []
        starts = [0]
        slots: list[torch.Tensor] = []
        for req in batch:
            n = len(req.pending_input)
            token_ids.extend(req.pending_input)
            positions.extend(range(n))
            starts.append(starts[-1] + n)
            slots.append(kv.slot_mapping(req.rid, 0, n))
            kv.set_len(req.rid, n)
            req.pos = n
            req.pending_input = []
        slots_t = torch.cat(slots)
        if self._tp_active():
            logits = self._tp_dispatch({
                "op": "prefill", "token_ids": token_ids, "positions": positions,
                "seq_starts": starts, "slots": slots_t.tolist(),
            })
        else:
            logits = self.model.prefill(
                torch.tensor(token_ids, dtype=torch.int64),
                torch.tensor(positions, dtype=torch.int32),
                torch.tensor(starts, dtype=torch.int32), slots_t)
        self.stats["prefill_tokens"] += len(token_ids)
        self._maybe_register(batch)
        self._sample_and_advance(batch, logits)
        with self._lock:
            for req in batch:
                if req.state != "done":
                    req.state = "running"
                    self.running.append(req)
        self.stats["prefill_time"] += time.time() - t0

    def _run_chunk(self, batch: list[Request]) -> None:
        """Multi-token append (grammar-forced runs) via chunked prefill."""
        t0 = time.time()
        kv = self.model.kv
        token_ids: list[int] = []
        positions: list[int] = []
        starts = [0]
        slots: list[torch.Tensor] = []
        hist: list[int] = []
        for req in batch:
            n = len(req.pending_input)
            token_ids.extend(req.pending_input)
            positions.extend(range(req.pos, req.pos + n))
            starts.append(starts[-1] + n)
            kv.extend(req.rid, req.pos + n)
            slots.append(kv.slot_mapping(req.rid, req.pos, n))
            hist.append(req.pos)
            kv.set_len(req.rid, req.pos + n)
            req.pos += n
            req.pending_input = []
        bt, _lens = kv.batch_tables([r.rid for r in batch], "cpu")
        slots_t = torch.cat(slots)
        self.stats["chunk_pre_time"] = (self.stats.get("chunk_pre_time", 0.0)
                                        + time.time() - t0)
        ev0 = self._event()
        if self._tp_active():
            logits = self._tp_dispatch({
                "op": "chunk", "token_ids": token_ids, "positions": positions,
                "seq_starts": starts, "block_tables": bt.flatten().tolist(),
                "bt_shape": list(bt.shape), "hist_lens": hist,
                "slots": slots_t.tolist(),
            })
        else:
            logits = self.model.chunk_step(
                torch.tensor(token_ids, dtype=torch.int64),
                torch.tensor(positions, dtype=torch.int32),
                torch.tensor(starts, dtype=torch.int32), bt,
                torch.tensor(hist, dtype=torch.int32), slots_t)
        ev1 = self._event()
        self.stats["chunk_tokens"] = self.stats.get("chunk_tokens", 0) + len(token_ids)
        t1 = time.time()
        self.stats["chunk_launch_time"] = (self.stats.get("chunk_launch_time", 0.0)
                                           + t1 - t0)
        self.stats["chunk_steps"] = self.stats.get("chunk_steps", 0) + 1
        self._maybe_register(batch)
        self._sample_and_advance(batch, logits)
        self._fold_events("chunk", ev0, ev1)
        self.stats["sample_time"] = (self.stats.get("sample_time", 0.0)
                                     + time.time() - t1)
        with self._lock:
            self.running = [r for r in self.running if r.state != "done"]
        self.stats["decode_time"] += time.time() - t0

    def _run_decode(self, batch: list[Request]) -> None:
        t0 = time.time()
        kv = self.model.kv
        input_ids = []
        positions = []
        slot_list = []
        for req in batch:
            # feed ONE token from each request's pending queue; requests
            # mid-run (forced bytes still queued) skip sampling this step
            input_ids.append(req.pending_input[0])
            positions.append(req.pos)
            kv.extend(req.rid, req.pos + 1)
            slot_list.append(kv.slot_mapping(req.rid, req.pos, 1))
            kv.set_len(req.rid, req.pos + 1)
            req.pos += 1
            req.pending_input = req.pending_input[1:]
        bt, lens = kv.batch_tables([r.rid for r in batch], "cpu")
        slots_t = torch.cat(slot_list)
        self.stats["decode_pre_time"] = (self.stats.get("decode_pre_time", 0.0)
                                         + time.time() - t0)
        ev0 = self._event()
        if self._tp_active():
            logits = self._tp_dispatch({
                "op": "decode", "token_ids": input_ids, "positions": positions,
                "block_tables": bt.flatten().tolist(), "bt_shape": list(bt.shape),
                "seq_lens": lens.tolist(), "slots": slots_t.tolist(),
            })
        else:
            logits = self.model.decode(
                torch.tensor(input_ids, dtype=torch.int64),
                torch.tensor(positions, dtype=torch.int32),
                bt, lens, slots_t)
        self.stats["decode_tokens"] += len(batch)
        ev1 = self._event()
        t1 = time.time()
        # launch time only — the GPU-side compute lands in decode_gpu_time
        # via events (an un-synced t1 stamp here would book the whole GPU
        # decode into sample_time: round-1 verdict's measurement bug)
        self.stats["decode_launch_time"] = (
            self.stats.get("decode_launch_time", 0.0) + t1 - t0)
        self._maybe_register(batch)
        drained = [i for i, r in enumerate(batch) if not r.pending_input]
        if len(drained) == len(batch):
            self._sample_and_advance(batch, logits)
        elif drained:
            idx = torch.tensor(drained, device=logits.device)
            self._sample_and_advance([batch[i] for i in drained], logits[idx])
        self._fold_events("decode", ev0, ev1)
        self.stats["sample_time"] = (self.stats.get("sample_time", 0.0)
                                     + time.time() - t1)
        with self._lock:
            self.running = [r for r in self.running if r.state != "done"]
        self.stats["decode_time"] += time.time() - t0

    # -- phase attribution (CUDA events, folded lazily) ---------------------------

    def _event(self):
        """Timing event on the current stream (None off-GPU / under TP)."""
        if not self.device.startswith("cuda") or self.model.tp > 1:
            return None
        ev = torch.cuda.Event(enable_timing=True)
        ev.record()
        return ev

    def _fold_events(self, phase: str, ev0, ev1) -> None:
        """Record an end event AFTER sampling and fold completed triples
        into stats. The sampler's `.cpu()` sync means the PREVIOUS step's
        events are always complete by now, so folding never blocks; the
        result is an honest GPU-side split of model vs sampling time."""
        if ev0 is None:
            return
        ev2 = torch.cuda.Event(enable_timing=True)
        ev2.record()
        self._ev_queue.append((phase, ev0, ev1, ev2))
        while self._ev_queue and self._ev_queue[0][3].query():
            ph, e0, e1, e2 = self._ev_queue.popleft()
            self.stats[f"{ph}_gpu_time"] = (self.stats.get(f"{ph}_gpu_time", 0.0)
                                            + e0.elapsed_time(e1) / 1e3)
            self.stats["sample_gpu_time"] = (self.stats.get("sample_gpu_time", 0.0)
                                             + e1.elapsed_time(e2) / 1e3)

    def _maybe_register(self, batch: list[Request]) -> None:
        """Once a request's full prompt (prompt + forced prefix) is resident
        in KV, publish its whole blocks to the prefix pool — BEFORE sampling,
        which may finish the request and free its blocks."""
        if not self.prefix_cache:
            return
        kv = self.model.kv
        for req in batch:
            if not req.registered and req.pos >= len(req.prefill_tokens):
                kv.register_prefix(req.rid, req.prefill_tokens)
                req.registered = True

    # -- sampling ---------------------------------------------------------------------

    def _mask_buf(self, n: int) -> torch.Tensor:
        """Reusable (pinned on GPU hosts) CPU mask buffer: avoids a fresh
        allocation per step and makes the host->device copy async-capable."""
        buf = getattr(self, "_mask_buffer", None)
        if buf is None or buf.shape[0] < n:
            cap = max(n, self.max_batch)
            pin = self.device.startswith("cuda")
            buf = torch.zeros((cap, MASK_REGION), dtype=torch.bool,
                              pin_memory=pin)
            self._mask_buffer = buf
        out = buf[:n]
        out.fill_(False)
        return out

    def _sample_and_advance(self, batch: list[Request], logits: torch.Tensor) -> None:
        """Masked sampling per request + FSM/stop bookkeeping.

        All requests are masked to the active byte vocab (ids < MASK_REGION);
        schema'd requests are further masked to their FSM's allowed bytes.
        With a checkpoint BPE tokenizer the whole vocabulary is live and
        decoding is unmasked (schemas are rejected at submit).
        """
        if self.hf_tokenizer is not None:
            masker = self.bpe_masker
            if any(r.fsm is not None for r in batch):
                # token-trie grammar masks over the full checkpoint vocab:
                # cached-row assembly (grammar_bpe.mask_row) — the per-id
                # Python fill was the dominant per-step cost when serving
                V = logits.shape[1]
                rows = []
                for r in batch:
                    if r.fsm is None:
                        rows.append(torch.ones(V, dtype=torch.bool))
                    else:
                        rows.append(masker.mask_row(r.fsm, V))
                mask = torch.stack(rows)
                logits = logits.masked_fill(~mask.to(logits.device), float("-inf"))
            greedy = all(r.temperature <= 0.0 for r in batch)
            if greedy:
                chosen_t = logits.argmax(-1)
            else:
                temp = max(r.temperature for r in batch)
                logits = _nucleus_mask(logits, [r.top_p for r in batch])
                probs = torch.softmax(logits.float() / temp, dim=-1)
                chosen_t = torch.multinomial(probs, 1).squeeze(-1)
            now = time.time()
            chosen_l = chosen_t.cpu().tolist()
            for i, req in enumerate(batch):
                if req.first_token_at == 0.0:
                    req.first_token_at = now
                self._advance_request(req, int(chosen_l[i]))
            return
        tm0 = time.time()
        region = logits[:, :MASK_REGION]
        mask = self._mask_buf(len(batch))
        for i, req in enumerate(batch):
            if req.fsm is not None:
                allowed = req.fsm.allowed_bytes()
                if not allowed:   # FSM complete -> force EOT
                    mask[i, SpecialTokens.EOT] = True
                elif len(allowed) >= 90:   # string-content state: cached row
                    mask[i] = _string_mask_row(0x22 in allowed,
                                               req.fsm.string_capacity())
                else:
                    mask[i, allowed] = True
            else:
                mask[i, :ACTIVE_VOCAB] = True
        mask_d = mask.to(region.device, non_blocking=True)
        greedy = all(r.temperature <= 0.0 for r in batch)
        if greedy:
            chosen = ops.masked_greedy(region, mask_d)
        else:
            temp = max(r.temperature for r in batch)
            if any(r.top_p < 1.0 for r in batch):
                # nucleus over the grammar-allowed set (mass is computed
                # AFTER masking so top_p means "of the legal tokens")
                masked = region.masked_fill(~mask_d, float("-inf"))
                region = _nucleus_mask(masked, [r.top_p for r in batch])
            chosen = ops.masked_sample(region, mask_d, temperature=temp)
        self.stats["sample_mask_time"] = (self.stats.get("sample_mask_time", 0.0)
                                          + time.time() - tm0)
        tw0 = time.time()
        chosen = chosen.cpu().tolist()   # device sync: GPU wait lands HERE
        self.stats["sample_wait_time"] = (self.stats.get("sample_wait_time", 0.0)
                                          + time.time() - tw0)
        now = time.time()
        region_cpu = None
        for i, req in enumerate(batch):
            if req.first_token_at == 0.0:
                req.first_token_at = now
            tok = int(chosen[i])
            # zero-width sentinel (number close): advance FSM and re-choose
            # from the SAME logits row until a real byte appears
            while (req.fsm is not None and tok == NUMBER_CLOSE_SENTINEL
                   and not req.fsm.done):
                req.fsm.advance(NUMBER_CLOSE_SENTINEL)
                allowed = req.fsm.allowed_bytes()
                if not allowed:
                    tok = SpecialTokens.EOT
                    break
                if region_cpu is None:
                    region_cpu = region.float().cpu()
                row = region_cpu[i]
                tok = max(allowed, key=lambda b: float(row[b]))
            self._advance_request(req, tok)
        self.stats["sample_advance_time"] = (
            self.stats.get("sample_advance_time", 0.0) + time.time() - now)

    def _drain_forced(self, req: Request) -> list[int]:
        """Consume grammar-FORCED bytes (single-choice FSM states) without
        spending decode steps; they return as a chunk-prefill run."""
        run: list[int] = []
        if self.hf_tokenizer is not None:
            return run   # byte-forcing only applies to the byte tokenizer
        fsm = req.fsm
        while fsm is not None and not fsm.done:
            allowed = fsm.allowed_bytes()
            if len(allowed) != 1:
                break
            b = allowed[0]
            fsm.advance(b)
            if b != NUMBER_CLOSE_SENTINEL:
                req.out_ids.append(b)
                run.append(b)
        return run

    def _advance_request(self, req: Request, tok: int) -> None:
        finished = False
        if self.hf_tokenizer is not None and req.fsm is not None:
            # BPE grammar path: one TOKEN advances the byte FSM through its
            # whole expansion (trie admission guarantees legality)
            masker = self.bpe_masker
            token_bytes = masker.token_bytes.get(tok)
            if req.fsm.done or tok == masker.eot_id or token_bytes is None:
                finished = True
            else:
                masker.advance_token(req.fsm, token_bytes)
                req.out_ids.append(tok)
                req.pending_input = [tok]
                if req.fsm.done:
                    finished = True
            if len(req.out_ids) >= req.max_new_tokens:
                finished = True
            if finished:
                req.state = "done"
                req.finished_at = time.time()
                self._record_latency(req)
                self.model.kv.free(req.rid)
                req.done_event.set()
            return
        if req.fsm is not None:
            if req.fsm.done or tok == SpecialTokens.EOT:
                finished = True
            elif tok >= WORD_BASE:
                # word token: advance the byte automaton through the word
                for b in WORD_STRINGS[tok - WORD_BASE].encode("ascii"):
                    req.fsm.advance(b)
                req.out_ids.append(tok)
                req.pending_input = [tok]
                req.pending_input += self._drain_forced(req)
            else:
                req.fsm.advance(tok)
                if tok != NUMBER_CLOSE_SENTINEL:
                    req.out_ids.append(tok)
                    req.pending_input = [tok]
                else:
                    req.pending_input = []
                req.pending_input += self._drain_forced(req)
                if req.fsm.done:
                    finished = True
        else:
            if self.hf_tokenizer is not None:
                stops = {self.hf_tokenizer.eot_id}
            else:
                stops = {SpecialTokens.EOS, SpecialTokens.EOT}
            if tok in stops:
                finished = True
            else:
                req.out_ids.append(tok)
                req.pending_input = [tok]
        if len(req.out_ids) >= req.max_new_tokens:
            finished = True
        if not finished and not req.pending_input:
            # nothing left to feed the model (pure forced tail / FSM drained):
            # the request is complete — a running request with no pending
            # input would starve the scheduler
            finished = True
        if finished:
            req.state = "done"
            req.finished_at = time.time()
            self._record_latency(req)
            self.model.kv.free(req.rid)
            req.done_event.set()

    def _record_latency(self, req: Request) -> None:
        if req.finished_at and req.submitted_at:
            ttft = (req.first_token_at - req.submitted_at) \
                if req.first_token_at else 0.0
            self._latency_ring.append(
                (ttft, req.finished_at - req.submitted_at, len(req.out_ids)))

    # -- metrics ----------------------------------------------------------------------

    def latency_stats(self) -> dict[str, Any]:
        """TTFT / end-to-end percentiles over the last ≤512 finished
        requests (serving SLO view; exposed via /healthz and /metrics)."""
        samples = list(self._latency_ring)
        if not samples:
            return {"samples": 0}

        def pct(vals: list[float], q: float) -> float:
            vals = sorted(vals)
            return vals[min(len(vals) - 1, int(q * len(vals)))]

        ttft = [s_[0] for s_ in samples if s_[0] > 0]
        e2e = [s_[1] for s_ in samples]
        out = {"samples": len(samples),
               "e2e_p50_s": round(pct(e2e, 0.50), 4),
               "e2e_p95_s": round(pct(e2e, 0.95), 4)}
        if ttft:
            out["ttft_p50_s"] = round(pct(ttft, 0.50), 4)
            out["ttft_p95_s"] = round(pct(ttft, 0.95), 4)
        return out

    def throughput_stats(self) -> dict[str, Any]:
        s = dict(self.stats)
        if s["decode_time"] > 0:
            s["decode_tok_per_s"] = s["decode_tokens"] / s["decode_time"]
        if s["prefill_time"] > 0:
            s["prefill_tok_per_s"] = s["prefill_tokens"] / s["prefill_time"]
        return s


_engines: dict[str, LLMEngine] = {}


def get_engine(model: str = "tiny", **kwargs: Any) -> LLMEngine:
    """Process-wide engine cache (one engine per model)."""
    key = f"{model}:{kwargs.get('tp')}:{kwargs.get('device')}:{kwargs.get('checkpoint')}"
    if key not in _engines:
        _engines[key] = LLMEngine(model=model, **kwargs)
    return _engines[key]
