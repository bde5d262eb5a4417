"""Continuous batching scheduler (single-rank serving).

Requests join and leave the running decode batch between steps instead of
queueing for exclusive access (the reference serves strictly one request
at a time per job — ``ml/validator.py:642``; this exceeds parity). Built
on the paged KV cache: every slot owns a page table, the decode kernels
take per-sequence lengths, so a step runs over whichever slots are live.

Flow per scheduler iteration:
  1. admit queued requests into free slots;
  2. run ONE prefill chunk for one admitted-but-unfilled slot (chunked
     prefill: long prompts are processed ``prefill_chunk`` tokens at a
     time so running decodes are not stalled behind a long prompt —
     the attention kernel takes a ``q_off`` so chunk N attends to all
     keys written by chunks 0..N);
  3. one decode step over all running slots (ragged lengths are native to
     decode_attn_mfma via seq_lens);
  4. emitted tokens go to per-request queues (SSE streams read them);
  5. slots retire on EOS / max_new_tokens (pages return to the pool).
"""

from __future__ import annotations

import queue
import threading
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from tensorlink_amd import ops
from tensorlink_amd.models.paged import (PAGE, DynamicPagedKVCache,
                                         PagedKVCache,
                                         PrefixCachingKVCache)


@dataclass
class Request:
    input_ids: torch.Tensor              # [S] prompt
    max_new_tokens: int = 64
    temperature: float = 0.0
    top_p: float = 1.0
    top_k: int = 0
    eos_token_id: Optional[int] = None
    priority: int = 0                    # higher admits first
    presence_penalty: float = 0.0
    frequency_penalty: float = 0.0
    seed: Optional[int] = None           # per-request sampling seed
    logprobs: bool = False               # record chosen-token logprobs
    logprob_values: List[float] = field(default_factory=list)
    tokens: "queue.Queue" = field(default_factory=queue.Queue)
    done: threading.Event = field(default_factory=threading.Event)
    error: Optional[str] = None
    cancelled: bool = False

    def cancel(self):
        """Abort the request: queued requests are dropped at admission,
        running ones retire at the next scheduler step (pages return to
        the pool). The client sees the stream end."""
        self.cancelled = True

    def result(self, timeout: float = 300.0) -> List[int]:
        """Block until finished; returns the generated token ids."""
        self.done.wait(timeout)
        out = []
        while not self.tokens.empty():
            t = self.tokens.get_nowait()
            if t is not None:
                out.append(t)
        return out

    def stream(self, timeout: float = 120.0):
        """Yield token ids as they are generated."""
        while True:
            t = self.tokens.get(timeout=timeout)
            if t is None:
                return
            yield t


class _SlotView:
    """Duck-typed cache view over the live slots (k/v pools shared; table
    and seq_lens gathered per step)."""

    def __init__(self, cache: PagedKVCache, slot_ids, seq_lens):
        self.k = cache.k
        self.v = cache.v
        self.table = cache.table[slot_ids]
        self.seq_lens = seq_lens
        self.max_seq = cache.max_seq
        self.batch = len(slot_ids)
        self.pages_per_seq = cache.pages_per_seq
        self._parent = cache

    def advance(self, n):
        pass   # slot lengths are tracked by the scheduler

    def gather_contiguous(self, layer, S):
        return PagedKVCache.gather_contiguous(self, layer, S)

    def append(self, layer, k_new, v_new, positions):
        return PagedKVCache.append(self, layer, k_new, v_new, positions)


class ContinuousBatcher:
    SUPPORTS_PP = False

    def __init__(self, runner, max_slots: int = 16, max_ctx: int = 2048,
                 pool_pages: Optional[int] = None,
                 prefill_chunk: Optional[int] = None,
                 prefix_caching: bool = False,
                 speculative: bool = False, lookup_n: int = 3,
                 spec_k: int = 8):
        assert runner.world == 1 or self.SUPPORTS_PP, \
            "use PPContinuousBatcher for world > 1"
        self.runner = runner
        self.stage = runner.stage
        self.device = runner.device
        self.max_ctx = max_ctx
        # dynamic page pool: slots lease pages as sequences grow and
        # release them on retirement; pool_pages < slots*max_ctx/PAGE
        # oversubscribes memory (admission blocks when exhausted)
        if pool_pages is None:
            pool_pages = max_slots * ((max_ctx + PAGE - 1) // PAGE)
        cache_cls = (PrefixCachingKVCache if prefix_caching
                     else DynamicPagedKVCache)
        self.prefix_caching = prefix_caching
        # prompt-lookup speculation: eligible greedy slots contribute
        # 1+m ragged rows to the decode batch (rows share the slot's page
        # table with seq_lens len..len+m), so the verify runs through the
        # DECODE kernels — bitwise-identical to m+1 sequential decode
        # steps by construction, and composable with sampled slots in the
        # same batch. Exact greedy either way.
        self.speculative = speculative
        self.lookup_n = lookup_n
        self.spec_k = spec_k
        self.slot_ctx: List[Optional[list]] = [None] * max_slots
        self.spec_accepted = 0
        self.cache = cache_cls(
            self.stage.num_layers, max_slots, pool_pages, max_ctx,
            self.stage.config, runner.device, runner.dtype)
        self.max_slots = max_slots
        self.prefill_chunk = prefill_chunk    # None = whole prompt at once
        self.slots: List[Optional[Request]] = [None] * max_slots
        self.slot_len = [0] * max_slots       # tokens resident in cache
        self.slot_emitted = [0] * max_slots
        self.slot_last = [0] * max_slots      # last sampled token
        self.slot_prompt: List[Optional[torch.Tensor]] = [None] * max_slots
        self.slot_filled = [0] * max_slots    # prompt tokens prefilled
        self.slot_counts: List[dict] = [{} for _ in range(max_slots)]
        # preempted-and-swapped requests waiting to resume:
        # (req, blob, slot_len, slot_last, slot_emitted, counts, prompt)
        self._swapped: List[tuple] = []
        # admission queue ordered by (priority desc, arrival): a
        # high-priority request jumps the queue but never preempts a
        # running slot
        self._queue: "queue.PriorityQueue" = queue.PriorityQueue()
        self._seq = 0
        self._wake = threading.Event()
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self.steps = 0
        import random as _random
        self._rng = _random.Random()     # unseeded-request draw seeds
        from tensorlink_amd.utils.tracing import tracer_from_env
        self.tracer = tracer_from_env(runner.rank)  # TL_TRACE=<prefix>

    # ------------------------------------------------------------------
    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()
        return self

    def stop(self, drain: bool = False):
        """Stop the scheduler. drain=True first finishes every admitted
        request (queued ones are still cut loose); drain=False ends all
        requests at their current token (clients unblock either way —
        no request ever hangs on a stopped batcher)."""
        self._drain = drain
        self._stop.set()
        self._wake.set()
        if self._thread:
            self._thread.join(120 if drain else 30)
        # cut loose anything still attached (incl. swapped-out)
        leftovers = [r for r in self.slots if r is not None]
        leftovers += [t[0] for t in self._swapped]
        self._swapped = []
        while not self._queue.empty():
            try:
                leftovers.append(self._queue.get_nowait()[2])
            except Exception:
                break
        for req in leftovers:
            req.tokens.put(None)
            req.done.set()
        if self.tracer is not None:
            from tensorlink_amd.utils.tracing import export_from_env
            export_from_env(self.tracer)

    def submit(self, input_ids: torch.Tensor, **kw) -> Request:
        req = Request(input_ids=input_ids.reshape(-1), **kw)
        if req.input_ids.numel() == 0:
            raise ValueError("empty prompt")
        req.max_new_tokens = min(req.max_new_tokens, self.max_ctx - 1)
        if req.input_ids.numel() + req.max_new_tokens > self.max_ctx:
            keep = max(1, self.max_ctx - req.max_new_tokens)
            req.input_ids = req.input_ids[-keep:]
        self._seq += 1
        self._queue.put((-req.priority, self._seq, req))
        self._wake.set()
        return req

    # ------------------------------------------------------------------
    def _committed_pages(self) -> int:
        """Pages that admitted-but-still-prefilling slots will lease as
        their chunks land (chunked prefill leases lazily, so admission
        must count these as spoken for)."""
        tot = 0
        for i, r in enumerate(self.slots):
            if r is not None and self._prefilling(i):
                p = self.slot_prompt[i].numel()
                tot += max(0, (p + PAGE) // PAGE
                           - len(self.cache._slot_pages[i]))
        return tot

    # ---------------- preemption (swap to host) ----------------------
    def _preempt_one(self, exclude: Optional[int] = None) -> bool:
        """Swap out the youngest running slot (fewest emitted tokens —
        least work lost per page freed) to host memory; it re-enters via
        the resume path when pages free up. vLLM-style preemption keeps
        admission live when the page pool starves."""
        from tensorlink_amd.models.paged import swap_out
        cands = [i for i, r in enumerate(self.slots)
                 if r is not None and not self._prefilling(i)
                 and self.slot_filled[i] > 0 and i != exclude]
        if not cands:
            return False
        victim = min(cands, key=lambda i: self.slot_emitted[i])
        req = self.slots[victim]
        if self.tracer is not None:
            self.tracer.instant("preempt", slot=victim)
        blob = swap_out(self.cache, victim)
        self._swapped.append((req, blob, self.slot_len[victim],
                              self.slot_last[victim],
                              self.slot_emitted[victim],
                              self.slot_counts[victim],
                              self.slot_prompt[victim]))
        self.slots[victim] = None
        self.slot_prompt[victim] = None
        self.preemptions = getattr(self, "preemptions", 0) + 1
        return True

    def _resume_swapped(self) -> bool:
        from tensorlink_amd.models.paged import swap_in
        resumed = False
        while self._swapped:
            slot = self._free_slot()
            blob = self._swapped[0][1]
            if slot is None or \
                    self.cache.available_pages() < blob.n_pages + 1:
                break
            req, blob, s_len, s_last, s_emit, counts, prompt = \
                self._swapped.pop(0)
            swap_in(self.cache, slot, blob, s_len)
            self.slots[slot] = req
            self.slot_ctx[slot] = None   # resumed: plain decode
            self.slot_prompt[slot] = prompt
            self.slot_filled[slot] = prompt.numel()
            self.slot_len[slot] = s_len
            self.slot_last[slot] = s_last
            self.slot_emitted[slot] = s_emit
            self.slot_counts[slot] = counts
            resumed = True
        return resumed

    def _free_slot(self) -> Optional[int]:
        for i, s in enumerate(self.slots):
            if s is None:
                return i
        return None

    def _prefilling(self, slot: int) -> bool:
        p = self.slot_prompt[slot]
        return p is not None and self.slot_filled[slot] < p.numel()

    @torch.no_grad()
    def _prefill_chunk(self, slot: int):
        """Process the next chunk of slot's prompt; on the last chunk,
        sample the first token and enter decode."""
        req = self.slots[slot]
        prompt = self.slot_prompt[slot]
        S_total = prompt.numel()
        start = self.slot_filled[slot]
        end = (S_total if self.prefill_chunk is None
               else min(start + self.prefill_chunk, S_total))
        ids = prompt[start:end].to(self.device).unsqueeze(0)
        try:
            self.cache.ensure(slot, end + 1)
        except RuntimeError:
            # pool exhausted mid-prefill: park a decoding slot and retry
            if not self._preempt_one(exclude=slot):
                raise
            self.cache.ensure(slot, end + 1)
        pos = torch.arange(start, end, device=self.device,
                           dtype=torch.int32).unsqueeze(0)
        lens = torch.tensor([start], device=self.device, dtype=torch.int32)
        view = _SlotView(self.cache, [slot], lens)
        hidden = self.stage(ids, pos, kv_cache=view, return_logits=False)
        self.slot_filled[slot] = end
        if end < S_total:
            return
        if self.prefix_caching:
            self.cache.register_prefix(slot, prompt.tolist())
        logits = self.stage.head(hidden[:, -1:]).squeeze(1)
        tok = int(self._sample(logits, [req], slots=[slot])[0])
        self.slot_len[slot] = S_total
        self.slot_last[slot] = tok
        self.slot_emitted[slot] = 1
        if self.speculative:
            self.slot_ctx[slot] = prompt.tolist() + [tok]
        self.slot_counts[slot][tok] = self.slot_counts[slot].get(tok,
                                                                 0) + 1
        req.tokens.put(tok)
        self._maybe_finish(slot, tok)

    def _sample(self, logits, reqs, slots=None):
        """Sample one token per row. On the GPU path, penalty-free
        sampled rows go through the fused sampling kernel in ONE launch
        (ops/csrc/sampling.hip — per-request splitmix seeds keep draws
        batch-independent); greedy / penalized / logprobs rows take the
        torch reference path row-wise."""
        toks: list = [None] * len(reqs)
        fused = []
        if (logits.is_cuda and logits.dtype == torch.bfloat16
                and ops.extension_loaded()):
            fused = [i for i, r in enumerate(reqs)
                     if r.temperature > 0 and not r.logprobs
                     and r.presence_penalty == 0.0
                     and r.frequency_penalty == 0.0]
        if fused:
            dev = logits.device
            n = len(fused)
            temps = torch.tensor([reqs[i].temperature for i in fused],
                                 device=dev, dtype=torch.float32)
            tps = torch.tensor([reqs[i].top_p for i in fused],
                               device=dev, dtype=torch.float32)
            tks = torch.tensor([reqs[i].top_k for i in fused],
                               device=dev, dtype=torch.int32)
            zero = torch.zeros(n, device=dev, dtype=torch.float32)
            seeds = []
            for i in fused:
                req = reqs[i]
                req._draws = getattr(req, "_draws", 0) + 1
                base = (req.seed if req.seed is not None
                        else self._rng.getrandbits(62))
                seeds.append(ops.request_seed(base, req._draws))
            seeds_t = torch.tensor(seeds, device=dev, dtype=torch.int64)
            rows = (logits if n == len(reqs)
                    else logits[torch.tensor(fused, device=dev)])
            got = ops.sample_tokens(rows, temps=temps, top_ps=tps,
                                    top_ks=tks, pres=zero, freqs=zero,
                                    seeds=seeds_t).tolist()
            for j, i in enumerate(fused):
                toks[i] = int(got[j])
        for i, req in enumerate(reqs):
            if toks[i] is not None:
                continue
            counts = (self.slot_counts[slots[i]]
                      if slots is not None else None)
            gen = None
            if req.seed is not None:
                if not hasattr(req, "_gen"):
                    req._gen = torch.Generator(device=logits.device)
                    req._gen.manual_seed(req.seed)
                gen = req._gen
            tok = int(ops.sample_token(
                logits[i:i + 1], temperature=req.temperature,
                top_p=req.top_p, top_k=req.top_k, generator=gen,
                token_counts=counts,
                presence_penalty=req.presence_penalty,
                frequency_penalty=req.frequency_penalty)[0])
            if req.logprobs:
                lp = torch.log_softmax(logits[i].float(), -1)[tok]
                req.logprob_values.append(float(lp))
            toks[i] = tok
        return toks

    def _maybe_finish(self, slot: int, tok: int):
        req = self.slots[slot]
        if req is None:
            return
        if (req.cancelled
                or (req.eos_token_id is not None
                    and tok == req.eos_token_id)
                or self.slot_emitted[slot] >= req.max_new_tokens
                or self.slot_len[slot] + 1 >= self.max_ctx):
            req.tokens.put(None)
            req.done.set()
            self.slots[slot] = None
            self.slot_prompt[slot] = None
            self.cache.release_slot(slot)

    def _spec_eligible(self, slot: int) -> bool:
        """Greedy-only slots with no sampling state speculate; penalties,
        logprobs and sampled slots take the plain row (a greedy request
        with penalties would otherwise silently lose them in verify)."""
        req = self.slots[slot]
        return (self.speculative and self.slot_ctx[slot] is not None
                and req.temperature <= 0
                and req.presence_penalty == 0.0
                and req.frequency_penalty == 0.0
                and not req.logprobs)

    @torch.no_grad()
    def _decode_step(self):
        """One ragged decode batch over all running slots. A speculative
        slot contributes 1+m rows (its last token plus m prompt-lookup
        proposals) with per-row seq_lens len..len+m over the SAME page
        table; rope_append writes each row's K/V before attention reads
        them, so row j attends to keys 0..len+j exactly as m+1 serial
        decode steps would — same kernels, same bits. Non-speculative
        slots are single rows sampled with their penalties/seed."""
        from tensorlink_amd.parallel.pipeline import PipelineRunner
        active = [i for i, s in enumerate(self.slots)
                  if s is not None and not self._prefilling(i)
                  and self.slot_filled[i] > 0]
        if not active:
            return False
        # prompt-lookup proposals for eligible slots
        props = {}
        for i in active:
            if self._spec_eligible(i):
                req = self.slots[i]
                room = min(req.max_new_tokens - self.slot_emitted[i],
                           self.max_ctx - 1 - self.slot_len[i]) - 1
                p = PipelineRunner._lookup_propose(self.slot_ctx[i],
                                                   self.lookup_n,
                                                   self.spec_k)
                props[i] = p[:max(0, room)]
        for i in list(active):
            if self.slots[i] is None:      # preempted below this loop
                continue
            m = len(props.get(i, ()))
            try:
                self.cache.ensure(i, self.slot_len[i] + m + 2)
            except RuntimeError:
                # pool exhausted mid-decode: park another slot and retry
                if not self._preempt_one(exclude=i):
                    raise
                self.cache.ensure(i, self.slot_len[i] + m + 2)
        active = [j for j in active if self.slots[j] is not None]
        if not active:
            return False
        rows_tok, rows_len, rows_slot, seg = [], [], [], {}
        for i in active:
            prop = list(props.get(i, ()))
            seg[i] = (len(rows_tok), len(prop))
            rows_tok += [self.slot_last[i]] + prop
            rows_len += [self.slot_len[i] + j
                         for j in range(len(prop) + 1)]
            rows_slot += [i] * (len(prop) + 1)
        toks = torch.tensor(rows_tok, device=self.device, dtype=torch.int64)
        lens = torch.tensor(rows_len, device=self.device, dtype=torch.int32)
        view = _SlotView(self.cache, rows_slot, lens)
        pos = lens.unsqueeze(1)
        logits = self.stage(toks.unsqueeze(1), pos, kv_cache=view).squeeze(1)
        # sample all plain rows in ONE batched call (penalties/seeds per
        # row inside _sample); speculative slots accept below
        plain = [i for i in active if i not in props]
        plain_toks = {}
        if plain:
            rows = logits[torch.tensor([seg[i][0] for i in plain],
                                       device=logits.device)]
            got = self._sample(rows, [self.slots[i] for i in plain],
                               slots=plain)
            plain_toks = dict(zip(plain, got))
        for i in active:
            off, m = seg[i]
            req = self.slots[i]
            if i in props:
                # greedy accept: matching prefix + bonus token
                targets = logits[off:off + m + 1].argmax(-1).tolist()
                prop = props[i]
                a = 0
                while a < m and prop[a] == targets[a]:
                    a += 1
                accepted = prop[:a] + [targets[a]]
                self.spec_accepted += a
                # kv valid for the fed token + a accepted proposals
                self.slot_len[i] += 1 + a
                for tok in accepted:
                    self.slot_emitted[i] += 1
                    self.slot_counts[i][tok] = \
                        self.slot_counts[i].get(tok, 0) + 1
                    req.tokens.put(tok)
                    self.slot_ctx[i].append(tok)
                    self.slot_last[i] = tok
                    self._maybe_finish(i, tok)
                    if self.slots[i] is None:
                        break
            else:
                tok = plain_toks[i]
                self.slot_len[i] += 1
                self.slot_emitted[i] += 1
                self.slot_counts[i][tok] = \
                    self.slot_counts[i].get(tok, 0) + 1
                req.tokens.put(tok)
                self.slot_last[i] = tok
                if self.slot_ctx[i] is not None:
                    self.slot_ctx[i].append(tok)
                self._maybe_finish(i, tok)
        self.steps += 1
        return True

    def _loop(self):
        self._drain = False
        while not self._stop.is_set() or (
                self._drain and any(s is not None for s in self.slots)):
            admitted = self._resume_swapped()
            starved = 0
            while not self._queue.empty():
                slot = self._free_slot()
                if slot is None:
                    break
                # admission control: POP the head first (a concurrent
                # higher-priority submit could otherwise slip between a
                # peek and the get with an unchecked page need), check its
                # pages, and re-queue it unchanged if they don't fit —
                # (priority, seq) ordering puts it back in the same place
                try:
                    entry = self._queue.get_nowait()
                except queue.Empty:
                    break
                head = entry[2]
                # pages for the prompt plus its first generated token,
                # minus any published prefix it can adopt
                need = (head.input_ids.numel() + PAGE) // PAGE
                if self.prefix_caching:
                    need -= self.cache.match_prefix(
                        head.input_ids.tolist()) // PAGE
                if self.cache.available_pages() \
                        < need + self._committed_pages():
                    self._queue.put(entry)
                    # page starvation: swap out a young running slot so
                    # the queue keeps moving (at most one per iteration,
                    # never below two live slots, and not while others
                    # already wait to resume)
                    n_live = sum(r is not None for r in self.slots)
                    if (starved == 0 and n_live >= 2
                            and not self._swapped
                            and self._preempt_one()):
                        starved = 1
                        continue
                    break
                req = head
                if req.cancelled:
                    req.tokens.put(None)
                    req.done.set()
                    continue
                self.slots[slot] = req
                self.slot_prompt[slot] = req.input_ids
                self.slot_filled[slot] = 0
                self.slot_len[slot] = 0
                self.slot_counts[slot] = {}
                self.slot_ctx[slot] = None
                if self.prefix_caching:
                    self.slot_filled[slot] = self.cache.adopt_prefix(
                        slot, req.input_ids.tolist())
                admitted = True
            # one prefill chunk per iteration: bounds how long running
            # decodes wait behind a new long prompt
            prefilled = False
            for slot in range(self.max_slots):
                if self.slots[slot] is not None and (
                        self._prefilling(slot) or self.slot_filled[slot] == 0):
                    req = self.slots[slot]
                    if req.cancelled:          # cancelled mid-prefill
                        req.tokens.put(None)
                        req.done.set()
                        self.slots[slot] = None
                        self.slot_prompt[slot] = None
                        self.cache.release_slot(slot)
                        continue
                    try:
                        if self.tracer is not None:
                            with self.tracer.span("prefill_chunk",
                                                  slot=slot):
                                self._prefill_chunk(slot)
                        else:
                            self._prefill_chunk(slot)
                        prefilled = True
                    except Exception as e:   # pragma: no cover
                        req.error = str(e)
                        req.tokens.put(None)
                        req.done.set()
                        self.slots[slot] = None
                        self.slot_prompt[slot] = None
                        self.cache.release_slot(slot)
                    break
            if self.tracer is not None:
                with self.tracer.span("decode_step",
                                      slots=sum(x is not None
                                                for x in self.slots)):
                    busy = self._decode_step()
            else:
                busy = self._decode_step()
            if not busy and not admitted and not prefilled:
                self._wake.wait(0.05)
                self._wake.clear()


class PPContinuousBatcher(ContinuousBatcher):
    """Continuous batching across a pipeline-parallel group.

    Rank 0 owns the scheduler (queue, slots, finish decisions) and
    broadcasts ONE compact command per iteration; every rank executes its
    stage for the scheduled work — a prefill chunk and/or a decode batch —
    with activations flowing rank→rank+1 and the last rank sampling and
    isend-ing tokens back to rank 0. Follower ranks keep NO request state:
    their paged-KV allocators stay in lockstep because ensure/release are
    driven purely by the command stream (releases are deferred one
    iteration on rank 0 so every rank frees pages on the same step).

    The reference serves strictly one request at a time per job
    (``ml/validator.py:642``) and cannot batch across a pipeline at all;
    this is the PP-aware variant of the single-rank slot scheduler above.
    """

    SUPPORTS_PP = True

    def __init__(self, runner, **kw):
        super().__init__(runner, **kw)
        self.slot_samp = [None] * self.max_slots   # (temp, top_p, top_k)
        self._pending_releases: List[int] = []
        self._send_keep = []                       # isend keep-alive
        # follower-side prompt store (register_prefix needs the full
        # prompt when the last chunk lands)
        self._fol_prompt = [None] * self.max_slots

    # ------------------------- shared execution -----------------------
    def _sample_rows(self, logits, samps):
        """Sample the last rank's rows. GPU bf16 rows with sampling go
        through the fused kernel in one launch; greedy rows and the CPU
        tier use the torch path."""
        if (logits.is_cuda and logits.dtype == torch.bfloat16
                and ops.extension_loaded()
                and any(t > 0 for t, _, _ in samps)):
            dev = logits.device
            temps = torch.tensor([t for t, _, _ in samps], device=dev,
                                 dtype=torch.float32)
            tps = torch.tensor([p for _, p, _ in samps], device=dev,
                               dtype=torch.float32)
            tks = torch.tensor([k for _, _, k in samps], device=dev,
                               dtype=torch.int32)
            zero = torch.zeros(len(samps), device=dev)
            seeds = torch.tensor(
                [self._rng.getrandbits(62) for _ in samps], device=dev,
                dtype=torch.int64)
            return ops.sample_tokens(logits.contiguous(), temps=temps,
                                     top_ps=tps, top_ks=tks, pres=zero,
                                     freqs=zero, seeds=seeds).tolist()
        return [int(ops.sample_token(logits[i:i + 1], temperature=t,
                                     top_p=p, top_k=k)[0])
                for i, (t, p, k) in enumerate(samps)]

    def _isend(self, t, dst):
        t = t.contiguous()
        self._send_keep.append((self.runner.p2p.isend(t, dst), t))
        if len(self._send_keep) > 16:
            w, _ = self._send_keep.pop(0)
            w.wait()

    @torch.no_grad()
    def _exec_step(self, admissions, chunk, decodes, releases):
        """Execute one broadcast command on THIS rank (identical order
        everywhere: releases -> admissions/prefix-adopt -> prefill chunk
        -> decode groups, so the per-rank page allocators stay in
        lockstep). Returns (first_token or None, decode_tokens or None)
        on rank 0."""
        r = self.runner
        dev = self.device
        for slot in releases:
            self.cache.release_slot(slot)
        for slot, ids in admissions:
            self._fol_prompt[slot] = ids
            if self.prefix_caching:
                self.cache.adopt_prefix(slot, ids)
        chunk_tok = None
        if chunk is not None:
            slot, start, end, s_total, samp, ids = chunk
            self.cache.ensure(slot, end + 1)
            S = end - start
            pos = torch.arange(start, end, device=dev,
                               dtype=torch.int32).unsqueeze(0)
            lens = torch.tensor([start], device=dev, dtype=torch.int32)
            view = _SlotView(self.cache, [slot], lens)
            if r.is_first:
                x = torch.tensor(ids, device=dev,
                                 dtype=torch.int64).unsqueeze(0)
                hidden = self.stage(x, pos, kv_cache=view,
                                    return_logits=False)
            else:
                hidden = r.p2p.recv((1, S, r.H), r.dtype, r.prev_rank, dev)
                hidden = self.stage(hidden, pos, kv_cache=view,
                                    return_logits=False)
            if end == s_total and self.prefix_caching \
                    and self._fol_prompt[slot] is not None:
                self.cache.register_prefix(slot, self._fol_prompt[slot])
            if not r.is_last:
                self._isend(hidden, r.next_rank)
            elif end == s_total:
                logits = self.stage.head(hidden[:, -1:]).squeeze(1)
                tok = self._sample_rows(logits, [samp])
                self._isend(torch.tensor(tok, dtype=torch.int64,
                                         device=dev), 0)
        # decode: split the batch into up to `world` groups so successive
        # groups overlap across stages (rank s works on group g while
        # rank s-1 already runs group g+1 — all inter-stage transfers are
        # isends, only recv blocks). A decode entry is
        # (slot, len, tok, samp, prop): a speculating slot contributes
        # 1+len(prop) ragged rows with per-row seq_lens over its page
        # table — the same decode-kernel verify as the single-rank
        # scheduler, executed in lockstep on every rank.
        groups = []
        if decodes:
            n_groups = min(r.world, len(decodes))
            gsz = (len(decodes) + n_groups - 1) // n_groups
            groups = [decodes[i:i + gsz]
                      for i in range(0, len(decodes), gsz)]
            for d in decodes:
                self.cache.ensure(d[0], d[1] + len(d[4]) + 2)
        for sub in groups:
            rows_tok, rows_len, rows_slot = [], [], []
            for slot, ln, tok, samp, prop in sub:
                rows_tok += [tok] + list(prop)
                rows_len += [ln + j for j in range(len(prop) + 1)]
                rows_slot += [slot] * (len(prop) + 1)
            B = len(rows_tok)
            lens = torch.tensor(rows_len, device=dev, dtype=torch.int32)
            view = _SlotView(self.cache, rows_slot, lens)
            pos = lens.unsqueeze(1)
            if r.is_first:
                toks = torch.tensor(rows_tok, device=dev,
                                    dtype=torch.int64).unsqueeze(1)
                hidden = self.stage(toks, pos, kv_cache=view,
                                    return_logits=False)
            else:
                hidden = r.p2p.recv((B, 1, r.H), r.dtype, r.prev_rank, dev)
                hidden = self.stage(hidden, pos, kv_cache=view,
                                    return_logits=False)
            if not r.is_last:
                self._isend(hidden, r.next_rank)
            else:
                logits = self.stage.head(hidden).squeeze(1)
                # spec rows return raw argmax targets (rank 0 accepts);
                # plain rows are sampled here
                new = [0] * B
                off = 0
                for slot, ln, tok, samp, prop in sub:
                    m = len(prop)
                    if m > 0:
                        new[off:off + m + 1] =                             logits[off:off + m + 1].argmax(-1).tolist()
                    else:
                        new[off] = self._sample_rows(
                            logits[off:off + 1], [samp])[0]
                    off += m + 1
                self._isend(torch.tensor(new, dtype=torch.int64,
                                         device=dev), 0)
        # rank 0 collects the sampled tokens (isend on the last rank
        # breaks the send/recv cycle between pipeline passes; same-pair
        # P2P is FIFO so arrival order matches issue order)
        new_tokens = None
        if r.is_first:
            if chunk is not None and chunk[2] == chunk[3]:   # final chunk
                chunk_tok = int(r.p2p.recv((1,), torch.int64, r.world - 1,
                                           dev)[0])
            if groups:
                new_tokens = []
                for sub in groups:
                    rows = sum(1 + len(d[4]) for d in sub)
                    new_tokens += r.p2p.recv((rows,), torch.int64,
                                             r.world - 1, dev).tolist()
        return chunk_tok, new_tokens

    # ------------------------- follower (ranks 1..N-1) ------------------
    def serve_follower(self):
        """Blocking command loop for non-scheduler ranks; returns on
        batch_stop."""
        r = self.runner
        while True:
            cmd = r.p2p.broadcast_obj(None, src=0)
            if cmd[0] == "batch_stop":
                for w, _ in self._send_keep:
                    w.wait()
                return
            _, admissions, chunk, decodes, releases = cmd
            self._exec_step(admissions, chunk, decodes, releases)

    # ------------------------- scheduler (rank 0) ----------------------
    def _finish0(self, slot: int, tok: int):
        req = self.slots[slot]
        if (req.cancelled
                or (req.eos_token_id is not None
                    and tok == req.eos_token_id)
                or self.slot_emitted[slot] >= req.max_new_tokens
                or self.slot_len[slot] + 1 >= self.max_ctx):
            req.tokens.put(None)
            req.done.set()
            self.slots[slot] = None
            self.slot_prompt[slot] = None
            self.slot_ctx[slot] = None
            # defer the page release so every rank frees on the same step
            self._pending_releases.append(slot)

    def _loop(self):
        assert self.runner.rank == 0
        self._drain = False
        r = self.runner
        while not self._stop.is_set() or (
                self._drain and any(s is not None for s in self.slots)):
            admissions = []
            while not self._queue.empty():
                slot = self._free_slot()
                if slot is None:
                    break
                # pop-check-requeue (see ContinuousBatcher._loop)
                try:
                    entry = self._queue.get_nowait()
                except queue.Empty:
                    break
                head = entry[2]
                ids_list = head.input_ids.tolist()
                # pages for the prompt plus its first generated token,
                # minus any published prefix it can adopt
                need = (head.input_ids.numel() + PAGE) // PAGE
                matched = (self.cache.match_prefix(ids_list)
                           if self.prefix_caching else 0)
                if self.cache.available_pages() < (need - matched // PAGE
                                                   + self._committed_pages()):
                    self._queue.put(entry)
                    break
                req = head
                if req.cancelled:
                    req.tokens.put(None)
                    req.done.set()
                    continue
                self.slots[slot] = req
                self.slot_prompt[slot] = req.input_ids
                self.slot_filled[slot] = matched
                self.slot_len[slot] = 0
                self.slot_ctx[slot] = None
                self.slot_samp[slot] = (req.temperature, req.top_p,
                                        req.top_k)
                admissions.append((slot, ids_list))
            # one prefill chunk per iteration
            chunk = None
            chunk_slot = None
            for slot in range(self.max_slots):
                if self.slots[slot] is not None and (
                        self._prefilling(slot)
                        or self.slot_filled[slot] == 0):
                    if self.slots[slot].cancelled:
                        self.slots[slot].tokens.put(None)
                        self.slots[slot].done.set()
                        self.slots[slot] = None
                        self.slot_prompt[slot] = None
                        self._pending_releases.append(slot)
                        continue
                    prompt = self.slot_prompt[slot]
                    s_total = prompt.numel()
                    start = self.slot_filled[slot]
                    end = (s_total if self.prefill_chunk is None
                           else min(start + self.prefill_chunk, s_total))
                    chunk = (slot, start, end, s_total,
                             self.slot_samp[slot],
                             prompt[start:end].tolist())
                    chunk_slot = slot
                    break
            from tensorlink_amd.parallel.pipeline import PipelineRunner
            decodes = []
            for i in range(self.max_slots):
                if (self.slots[i] is None or self._prefilling(i)
                        or self.slot_filled[i] == 0 or i == chunk_slot):
                    continue
                prop = []
                if (self.speculative and self.slot_ctx[i] is not None
                        and self.slot_samp[i][0] <= 0):
                    req = self.slots[i]
                    room = min(req.max_new_tokens - self.slot_emitted[i],
                               self.max_ctx - 1 - self.slot_len[i]) - 1
                    prop = PipelineRunner._lookup_propose(
                        self.slot_ctx[i], self.lookup_n,
                        self.spec_k)[:max(0, room)]
                decodes.append((i, self.slot_len[i], self.slot_last[i],
                                self.slot_samp[i], prop))
            if chunk is None and not decodes and not admissions \
                    and not self._pending_releases:
                self._wake.wait(0.05)
                self._wake.clear()
                continue
            rel, self._pending_releases = self._pending_releases, []
            r.p2p.broadcast_obj(("batch_step", admissions, chunk, decodes,
                                 rel), src=0)
            chunk_tok, new_tokens = self._exec_step(admissions, chunk,
                                                    decodes, rel)
            if chunk is not None:
                slot, start, end, s_total = chunk[:4]
                self.slot_filled[slot] = end
                if chunk_tok is not None:
                    req = self.slots[slot]
                    self.slot_len[slot] = s_total
                    self.slot_last[slot] = chunk_tok
                    self.slot_emitted[slot] = 1
                    if self.speculative:
                        self.slot_ctx[slot] =                             self.slot_prompt[slot].tolist() + [chunk_tok]
                    req.tokens.put(chunk_tok)
                    self._finish0(slot, chunk_tok)
            if new_tokens:
                off = 0
                for (slot, _, _, _, prop) in decodes:
                    m = len(prop)
                    seg = new_tokens[off:off + m + 1]
                    off += m + 1
                    if self.slots[slot] is None:
                        continue
                    if m > 0:
                        a = 0
                        while a < m and prop[a] == seg[a]:
                            a += 1
                        accepted = list(prop[:a]) + [seg[a]]
                        self.spec_accepted += a
                        self.slot_len[slot] += 1 + a
                        for tok in accepted:
                            self.slot_emitted[slot] += 1
                            self.slots[slot].tokens.put(tok)
                            self.slot_ctx[slot].append(tok)
                            self.slot_last[slot] = tok
                            self._finish0(slot, tok)
                            if self.slots[slot] is None:
                                break
                    else:
                        tok = seg[0]
                        self.slot_len[slot] += 1
                        self.slot_emitted[slot] += 1
                        self.slots[slot].tokens.put(tok)
                        if self.slot_ctx[slot] is not None:
                            self.slot_ctx[slot].append(tok)
                        self.slot_last[slot] = tok
                        self._finish0(slot, tok)
                self.steps += 1
        r.p2p.broadcast_obj(("batch_stop",), src=0)
        for w, _ in self._send_keep:
            w.wait()
