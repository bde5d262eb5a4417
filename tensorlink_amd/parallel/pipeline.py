"""Pipeline-parallel runtime: SPMD stage execution over RCCL P2P.

Replaces the reference's entire L2-L5 execution machinery — the polling
worker loop (``tensorlink/ml/worker.py:1349-1445``), the forward/backward
queue tagging ``(n_iter, n_micro, module_id)`` (``p2p/torch_node.py:294``),
the micro-batch Python threads with 0.1 s stagger (``ml/module.py:374-436``)
and the 6-hop GPU→CPU→shm→TCP→shm→CPU→GPU transfer chain — with a
single-program pipeline: every rank runs the same schedule, activations move
GPU→GPU over xGMI, sampled tokens ride a thin int ring back to rank 0.

Inference uses ring-pipelined decode with one micro-batch in flight per
stage (the reference cannot pipeline decode at all — it only offloads whole
models for generation, ``ml/worker.py:359``). Training uses 1F1B.
"""

from __future__ import annotations

import statistics
import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from tensorlink_amd.models.configs import ModelConfig
from tensorlink_amd.models.dense import KVCache, StageModel, build_stage
from tensorlink_amd.models.loader import init_random_stage, load_stage_from_checkpoint
from tensorlink_amd.parallel.comm import P2P, device_for_rank
from tensorlink_amd.parallel.planner import StagePlan


@dataclass
class SamplingParams:
    temperature: float = 0.0          # 0 => greedy
    top_p: float = 1.0
    top_k: int = 0
    max_new_tokens: int = 64
    eos_token_id: Optional[int] = None
    seed: Optional[int] = None
    presence_penalty: float = 0.0     # applied on the batcher path
    frequency_penalty: float = 0.0


def _kv_slice(cache, s: int, e: int):
    from tensorlink_amd.models.paged import PagedKVCache, kv_slice_paged
    if isinstance(cache, PagedKVCache):
        return kv_slice_paged(cache, s, e)
    view = object.__new__(KVCache)
    view.k = [k[s:e] for k in cache.k]
    view.v = [v[s:e] for v in cache.v]
    view.seq_lens = cache.seq_lens[s:e]
    view.max_seq = cache.max_seq
    view.batch = e - s
    return view


class PipelineRunner:
    """One rank's stage executor. All public methods are SPMD: every rank
    in the group must call them together with the same control arguments."""

    def __init__(self, plan: StagePlan, rank: int, world: int,
                 device=None, init: str = "random", ckpt_dir: Optional[str] = None,
                 dtype=None, seed: int = 0, quantize: Optional[str] = None,
                 group=None, rank_base: int = 0,
                 kv_mode: Optional[str] = None):
        assert plan.num_stages == world, "plan stages must equal world size"
        self.group = group
        self.rank_base = rank_base
        self.plan = plan
        self.config: ModelConfig = plan.config
        self.rank = rank
        self.world = world
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        self.dtype = dtype or (torch.bfloat16 if self.device.type == "cuda"
                               else torch.float32)
        self.spec = plan.stage_for_rank(rank)
        self.stage: StageModel = build_stage(self.config, self.spec)
        import os as _os
        if init == "random" and ckpt_dir is None \
                and isinstance(plan.model, str) \
                and _os.path.isdir(plan.model) and (
                    _os.path.exists(_os.path.join(
                        plan.model, "model.safetensors"))
                    or _os.path.exists(_os.path.join(
                        plan.model, "model.safetensors.index.json"))):
            # the model "name" IS a checkpoint directory: serve its
            # weights instead of a fresh random init
            init, ckpt_dir = "checkpoint", plan.model
        if init == "random":
            init_random_stage(self.stage, device=self.device,
                              dtype=self.dtype, seed=seed + rank)
        elif init == "checkpoint":
            load_stage_from_checkpoint(self.stage, ckpt_dir,
                                       device=self.device, dtype=self.dtype)
        elif init == "empty":
            self.stage.to(device=self.device, dtype=self.dtype)
        import os as _os
        self.kv_mode = kv_mode or _os.environ.get("TL_KV_MODE",
                                                  "contiguous")
        self._no_graph = False
        if quantize == "fp8":
            from tensorlink_amd import ops as _ops
            from tensorlink_amd.models.quant import quantize_experts_fp8
            quantize_experts_fp8(self.stage)
            # fp8 experts run the grouped moe_gemm kernel (in-kernel
            # dequant, capture-safe); only without the extension would
            # they fall to torch._scaled_mm, which cannot be captured
            if not _ops.extension_loaded():
                self._no_graph = True
        elif quantize == "fp8-dense":
            from tensorlink_amd import ops as _ops
            from tensorlink_amd.models.quant import quantize_dense_fp8
            quantize_dense_fp8(self.stage)
            # with the extension, fp8 projections run the grouped-GEMM
            # in-kernel-dequant path (capture-safe); a capture attempt
            # that still hits _scaled_mm falls back to eager via the
            # existing RuntimeError catch
            if not _ops.extension_loaded():
                self._no_graph = True
        elif quantize == "fp4-dense":
            from tensorlink_amd.models.quant import quantize_dense_fp4
            quantize_dense_fp4(self.stage)
            self._no_graph = True
        self.stage.eval()
        self.p2p = P2P(rank, world, group, rank_base) if world > 1 else None
        self.is_first = rank == 0
        self.is_last = rank == world - 1
        self.next_rank = rank + 1
        self.prev_rank = rank - 1
        self.H = self.config.hidden_size
        self.kv_cache: Optional[KVCache] = None
        self._gen = None
        self._decode_graph = None       # (graph, tok_buf, pos_buf)
        from tensorlink_amd.utils.tracing import tracer_from_env
        self.tracer = tracer_from_env(rank)   # TL_TRACE=<prefix> enables

    def _trace(self, name: str, t0: float, t1: float, **args) -> None:
        if self.tracer is not None:
            self.tracer.events.append(
                {"name": name, "ph": "X", "pid": self.rank, "tid": 0,
                 "ts": (t0 - self.tracer._t0) * 1e6,
                 "dur": (t1 - t0) * 1e6, "args": args})

    # ------------------------------------------------------------------
    def _sample(self, logits: torch.Tensor, sp: SamplingParams) -> torch.Tensor:
        from tensorlink_amd import ops
        if sp.seed is not None and self._gen is None:
            self._gen = torch.Generator(device=logits.device)
            self._gen.manual_seed(sp.seed)
        return ops.sample_token(logits, temperature=sp.temperature,
                                top_p=sp.top_p, top_k=sp.top_k,
                                generator=self._gen)

    def alloc_cache(self, batch: int, max_seq: int):
        # reuse the allocation when shapes match — keeps hipGraph-captured
        # decode valid across generate() calls (pointers must not move)
        c = self.kv_cache
        if (c is not None and c.batch == batch and c.max_seq >= max_seq):
            c.reset()
            return c
        self.kv_cache = self.stage.make_kv_cache(batch, max_seq, self.device,
                                                 self.dtype,
                                                 kv_mode=self.kv_mode)
        self._decode_graph = None
        return self.kv_cache

    # ------------------------------------------------------------------
    @torch.no_grad()
    def generate(self, input_ids: Optional[torch.Tensor] = None,
                 sampling: Optional[SamplingParams] = None,
                 micro_batches: Optional[int] = None,
                 return_stats: bool = False,
                 on_token=None):
        """SPMD generate. input_ids [B, S] significant on rank 0 only.
        Returns generated tokens [B, max_new] on the FIRST rank, else None.
        """
        sp = sampling or SamplingParams()
        if sp.seed is not None:
            # a seeded call reproduces regardless of what ran before
            self._gen = None
        # agree on shapes AND micro-batch split (rank 0 is authoritative)
        if self.world > 1:
            meta = None
            if self.is_first:
                meta = (tuple(input_ids.shape), sp, micro_batches)
            meta = self.p2p.broadcast_obj(meta, src=0)
            (B, S), sp, micro_batches = meta
        else:
            B, S = input_ids.shape

        n_mb = micro_batches or self.world
        n_mb = max(1, min(n_mb, B))
        while B % n_mb != 0:
            n_mb -= 1
        b = B // n_mb
        max_seq = S + sp.max_new_tokens
        self.alloc_cache(B, max_seq)

        t_start = time.perf_counter()
        t_first: Optional[float] = None
        t_firsts: List[float] = []    # per micro-batch (per-request TTFT)
        use_events = self.device.type == "cuda"
        if use_events:
            ev_start = torch.cuda.Event(enable_timing=True)
            ev_start.record()
            ev_firsts: List[torch.cuda.Event] = []

        if self.is_first:
            input_ids = input_ids.to(self.device)
        pos_row = torch.arange(S, device=self.device, dtype=torch.int32)

        # ---------------- prefill (per micro-batch, pipelined) ----------
        first_tokens: List[torch.Tensor] = []
        send_handles = []  # (work, tensor) keep-alive
        for mb in range(n_mb):
            s, e = mb * b, (mb + 1) * b
            cache = _kv_slice(self.kv_cache, s, e)
            pos = pos_row.unsqueeze(0).expand(b, -1).contiguous()
            if self.is_first:
                hidden = self.stage(input_ids[s:e], pos, kv_cache=cache,
                                    return_logits=False)
            else:
                hidden = self.p2p.recv((b, S, self.H), self.dtype,
                                       self.prev_rank, self.device)
                hidden = self.stage(hidden, pos, kv_cache=cache,
                                    return_logits=False)
            if not self.is_last:
                hidden = hidden.contiguous()
                send_handles.append((self.p2p.isend(hidden, self.next_rank),
                                     hidden))
            else:
                logits = self.stage.head(hidden[:, -1:]).squeeze(1)
                tok = self._sample(logits, sp).to(torch.int64)
                if use_events:
                    # event timestamps: no per-chunk host sync (a sync
                    # here exposed kernel-launch gaps and cost ~5%
                    # throughput at 8 prefill chunks)
                    ev = torch.cuda.Event(enable_timing=True)
                    ev.record()
                    ev_firsts.append(ev)
                else:
                    t_firsts.append(time.perf_counter())
                    if t_first is None:
                        t_first = t_firsts[-1]
                first_tokens.append(tok)

        if self.is_last and use_events and ev_firsts:
            # resolve event times once (synchronizes on the recorded
            # events only, which have completed by first-token time)
            torch.cuda.synchronize(self.device)
            t_firsts = [t_start + ev_start.elapsed_time(e) / 1e3
                        for e in ev_firsts]
            t_first = t_firsts[0]
        self._trace("prefill", t_start, time.perf_counter(), batch=B,
                    seq=S, micro_batches=n_mb)
        t_dec0 = time.perf_counter()

        # ---------------- ring-pipelined decode -------------------------
        T = sp.max_new_tokens
        positions = torch.full((B,), S, device=self.device,
                               dtype=torch.int32)
        if self.is_last:
            out_tokens = torch.empty(B, T, device=self.device,
                                     dtype=torch.int64)
            for mb in range(n_mb):
                out_tokens[mb * b:(mb + 1) * b, 0] = first_tokens[mb]

        if self.world == 1:
            # end of the WHOLE prefill (all micro-batches), not the
            # first chunk — keeps the prefill/decode phase split honest
            t_prefill_end = (t_firsts[-1] if t_firsts else t_first)
            cur = first_tokens[0] if n_mb == 1 else torch.cat(first_tokens)
            if on_token is not None:
                on_token(0, cur)
            eos = sp.eos_token_id
            finished = (cur == eos) if eos is not None else None
            n_out = T
            import os as _os
            from tensorlink_amd import ops as _ops
            use_graph = (self.device.type == "cuda"
                         and (sp.temperature <= 0
                              or _ops.extension_loaded())
                         and T > 4 and not _os.environ.get("TL_NO_GRAPH")
                         and not self._no_graph
                         and self.kv_mode == "contiguous"
                         and on_token is None and eos is None)
            if use_graph:
                # hipGraph-captured decode: the whole per-token step (all
                # layers + head + sample) replays as one graph launch,
                # eliminating ~10 kernel-launch gaps per layer. Sampled
                # decode uses the fused sampling kernel with a device RNG
                # counter bumped inside the capture (graph-safe draws).
                try:
                    self._graph_decode(cur, positions, out_tokens, T, sp)
                except RuntimeError:
                    # capture-unsupported op somewhere in the stage: fall
                    # back to the eager loop permanently for this runner
                    self._no_graph = True
                    self._decode_graph = None
                    use_graph = False
            if not use_graph:
                for t in range(1, T):
                    pos = positions.unsqueeze(1)
                    logits = self.stage(cur.unsqueeze(1), pos.int(),
                                        kv_cache=self.kv_cache).squeeze(1)
                    cur = self._sample(logits, sp).to(torch.int64)
                    out_tokens[:, t] = cur
                    positions += 1
                    if on_token is not None:
                        on_token(t, cur)
                    if finished is not None:
                        finished |= (cur == eos)
                        if bool(finished.all()):
                            n_out = t + 1
                            break
            out = out_tokens[:, :n_out]
            self._trace("decode", t_dec0, time.perf_counter(), batch=B,
                        new_tokens=n_out)
            stats = self._finish_stats(B, S, n_out, t_start, t_first,
                                       t_firsts)
            if t_prefill_end is not None and stats is not None:
                dec = stats["total_s"] - (t_prefill_end - t_start)
                stats["prefill_s"] = t_prefill_end - t_start
                stats["decode_s"] = dec
                stats["decode_tokens_per_s"] = B * (n_out - 1) / dec if dec > 0 else None
            return (out, stats) if return_stats else out

        # world > 1: step-major ring. In round `step`, rank 0 feeds each
        # micro-batch its token (mb, step); the hidden flows down the
        # pipeline; the last rank samples token (mb, step+1) and has already
        # sent token (mb, step) back to rank 0 at the top of the round —
        # with n_mb >= world every stage has a micro-batch in flight every
        # tick (1-deep decode pipelining the reference cannot do).
        inflight: List = []  # (work, tensor) keep-alive for isends

        def _isend(t, dst):
            t = t.contiguous()
            inflight.append((self.p2p.isend(t, dst), t))
            if len(inflight) > 2 * n_mb + 2:
                w, _ = inflight.pop(0)
                w.wait()

        # hoist per-micro-batch cache views out of the tick loop: the
        # contiguous views alias stable parent storage (seq_lens slices
        # advance through the parent), and building them per tick was
        # measurable host overhead at world>1 eager decode
        from tensorlink_amd.models.paged import PagedKVCache
        if isinstance(self.kv_cache, PagedKVCache):
            mb_caches = None
        else:
            mb_caches = [_kv_slice(self.kv_cache, mb * b, (mb + 1) * b)
                         for mb in range(n_mb)]
        for step in range(T - 1):
            for mb in range(n_mb):
                s, e = mb * b, (mb + 1) * b
                cache = (mb_caches[mb] if mb_caches is not None
                         else _kv_slice(self.kv_cache, s, e))
                pos = positions[s:e].unsqueeze(1)
                if self.is_first:
                    tok = self.p2p.recv((b,), torch.int64, self.world - 1,
                                        self.device)
                    if on_token is not None:
                        on_token(step, tok)
                    hidden = self.stage(tok.unsqueeze(1), pos, kv_cache=cache,
                                        return_logits=False)
                    _isend(hidden, self.next_rank)
                elif not self.is_last:
                    hidden = self.p2p.recv((b, 1, self.H), self.dtype,
                                           self.prev_rank, self.device)
                    hidden = self.stage(hidden, pos, kv_cache=cache,
                                        return_logits=False)
                    _isend(hidden, self.next_rank)
                else:
                    _isend(out_tokens[s:e, step], 0)
                    hidden = self.p2p.recv((b, 1, self.H), self.dtype,
                                           self.prev_rank, self.device)
                    hidden = self.stage(hidden, pos, kv_cache=cache,
                                        return_logits=False)
                    logits = self.stage.head(hidden).squeeze(1)
                    out_tokens[s:e, step + 1] = self._sample(logits, sp)
                positions[s:e] += 1
        for w, _ in inflight + send_handles:
            w.wait()

        # ship results to rank 0
        result = None
        if self.is_last:
            self.p2p.send(out_tokens, 0)
        if self.is_first:
            result = self.p2p.recv((B, T), torch.int64, self.world - 1,
                                   self.device)
        self._trace("decode", t_dec0, time.perf_counter(), batch=B,
                    new_tokens=T)
        stats = self._finish_stats(B, S, T, t_start, t_first,
                                   t_firsts)
        if return_stats:
            return result, stats
        return result

    @torch.no_grad()
    def generate_speculative(self, input_ids: torch.Tensor,
                             max_new_tokens: int = 64,
                             lookup_n: int = 3, k: int = 8,
                             eos_token_id: Optional[int] = None):
        """Prompt-lookup speculative decoding (world==1, greedy, B==1):
        propose the k tokens that followed the most recent occurrence of
        the current lookup_n-gram earlier in the context, verify all of
        them in ONE cached forward (the chunked-prefill q_off attention
        path scores m+1 positions at once), and accept the longest
        matching prefix plus the bonus token. Output is EXACTLY plain
        greedy decoding — rejected positions simply roll seq_lens back
        and are overwritten. No draft model needed; repetitive text
        (code, extraction, chat history echoes) decodes several tokens
        per forward. Returns (tokens [1, n], accepted_via_spec count).
        """
        assert self.world == 1, "speculative decode is single-rank"
        B, S = input_ids.shape
        assert B == 1, "prompt-lookup speculation is per-request (B=1)"
        ids = input_ids.to(self.device)
        cache = self.alloc_cache(1, S + max_new_tokens + k + 1)
        pos = torch.arange(S, device=self.device,
                           dtype=torch.int32).unsqueeze(0)
        hidden = self.stage(ids, pos, kv_cache=cache, return_logits=False)
        cur = int(self.stage.head(hidden[:, -1:]).squeeze(1).argmax(-1))
        context = input_ids[0].tolist() + [cur]
        out = [cur]
        n_spec = 0
        position = S                      # cache holds kv for 0..S-1
        while len(out) < max_new_tokens and (eos_token_id is None
                                             or cur != eos_token_id):
            proposal = self._lookup_propose(context, lookup_n, k)
            m = len(proposal)
            feed = torch.tensor([[cur] + proposal, ], device=self.device,
                                dtype=torch.int64)
            p = (position + torch.arange(m + 1, device=self.device,
                                         dtype=torch.int32)).unsqueeze(0)
            logits = self.stage(feed, p, kv_cache=cache)     # [1, m+1, V]
            targets = logits.argmax(-1)[0].tolist()
            a = 0
            while a < m and proposal[a] == targets[a]:
                a += 1
            accepted = proposal[:a] + [targets[a]]           # + bonus
            n_spec += a
            if eos_token_id is not None and eos_token_id in accepted:
                accepted = accepted[:accepted.index(eos_token_id) + 1]
            room = max_new_tokens - len(out)
            accepted = accepted[:room]
            out.extend(accepted)
            context.extend(accepted)
            cur = accepted[-1] if accepted else targets[0]
            # roll back: kv is valid for cur + the a accepted proposals
            position += 1 + a
            cache.seq_lens.fill_(position)
        return (torch.tensor([out], device=self.device,
                             dtype=torch.int64), n_spec)

    @staticmethod
    def _lookup_propose(context, n, k):
        """Most recent earlier occurrence of the trailing n-gram; the
        tokens that followed it are the proposal."""
        if len(context) <= n:
            return []
        tail = context[-n:]
        # scan right-to-left, excluding the trailing occurrence itself
        for i in range(len(context) - n - 1, -1, -1):
            if context[i:i + n] == tail:
                return context[i + n:i + n + k]
        return []

    @torch.no_grad()
    def generate_beam(self, input_ids: torch.Tensor,
                      max_new_tokens: int = 64, num_beams: int = 4,
                      length_penalty: float = 1.0,
                      eos_token_id: Optional[int] = None) -> torch.Tensor:
        """Beam-search decode (world==1). The reference exposes
        ``num_beams`` through HF ``generate`` for whole-model jobs
        (``ml/formatter.py:94-99`` normalizes the beam/sample conflict,
        ``ml/worker.py:359`` runs generate); this is the native
        equivalent: beams share the batch dimension of one KV cache and
        the cache is index-reordered to each beam's parent between
        steps. Returns [B, <=max_new_tokens] (best beam per sequence).
        """
        assert self.world == 1, "beam search is single-rank"
        nb = num_beams
        B, S = input_ids.shape
        T = max_new_tokens
        dev = self.device
        ids = input_ids.to(dev).repeat_interleave(nb, 0)     # [B*nb, S]
        # fresh cache (beam reorder swaps slabs; don't disturb the
        # graph-captured cache reuse)
        cache = self.stage.make_kv_cache(B * nb, S + T, dev, self.dtype,
                                         kv_mode="contiguous")
        pos = torch.arange(S, device=dev, dtype=torch.int32) \
            .unsqueeze(0).expand(B * nb, -1).contiguous()
        hidden = self.stage(ids, pos, kv_cache=cache, return_logits=False)
        logp = torch.log_softmax(
            self.stage.head(hidden[:, -1:]).squeeze(1).float(), -1)
        V = logp.shape[-1]
        # beam 0 seeds each sequence (identical prefills => mask others)
        scores = torch.full((B, nb), float("-inf"), device=dev)
        scores[:, 0] = 0.0
        tokens = torch.empty(B * nb, 0, device=dev, dtype=torch.int64)
        alive = torch.ones(B * nb, dtype=torch.bool, device=dev)
        positions = torch.full((B * nb,), S, device=dev, dtype=torch.int32)
        for t in range(T):
            cand = scores.view(-1, 1) + logp                 # [B*nb, V]
            if eos_token_id is not None and t > 0:
                # a finished beam only extends with eos at no cost
                frozen = ~alive
                cand[frozen] = float("-inf")
                cand[frozen, eos_token_id] = scores.view(-1)[frozen]
            top_val, top_idx = cand.view(B, nb * V).topk(nb, -1)
            parent = top_idx // V                            # [B, nb]
            tok = (top_idx % V).to(torch.int64)
            gather = (torch.arange(B, device=dev).unsqueeze(1) * nb
                      + parent).view(-1)                     # [B*nb]
            # reorder beam state to each survivor's parent
            tokens = torch.cat([tokens[gather], tok.view(-1, 1)], 1)
            scores = top_val
            alive = alive[gather]
            if eos_token_id is not None:
                alive &= tok.view(-1) != eos_token_id
            if t == T - 1 or not bool(alive.any()):
                break
            for li in range(len(cache.k)):
                cache.k[li] = cache.k[li][gather].contiguous()
                cache.v[li] = cache.v[li][gather].contiguous()
            lg = self.stage(tokens[:, -1:],
                            positions.unsqueeze(1).int(), kv_cache=cache)
            logp = torch.log_softmax(lg.squeeze(1).float(), -1)
            positions += 1
        # pick the best beam per sequence under the GNMT length penalty
        lengths = tokens.shape[1] - (tokens == eos_token_id).sum(1).float() \
            if eos_token_id is not None else \
            torch.full((B * nb,), float(tokens.shape[1]), device=dev)
        lengths = lengths.clamp(min=1.0).view(B, nb)
        final = scores / lengths.pow(length_penalty)
        best = final.argmax(-1)                              # [B]
        idx = torch.arange(B, device=dev) * nb + best
        return tokens[idx]

    def _graph_decode(self, cur: torch.Tensor, positions: torch.Tensor,
                      out_tokens: torch.Tensor, T: int,
                      sp: "SamplingParams") -> None:
        """Decode via a captured hipGraph (world==1), greedy or sampled.

        The KV cache must already hold the prefill; `positions` holds the
        next position per sequence. State mutated by the captured step:
        tok/pos static buffers, cache contents, seq_lens and the RNG
        counter (all by device pointer, so replays see fresh values).
        Sampled capture: the fused sampling kernel draws from
        splitmix(seed_base, counter, row) — the counter advances inside
        the graph, so every replay gets fresh randomness with no host RNG.
        """
        from tensorlink_amd import ops
        B = cur.shape[0]
        skey = (float(sp.temperature), float(sp.top_p), int(sp.top_k),
                sp.seed)
        if self._decode_graph is not None and self._decode_graph[3] != skey:
            self._decode_graph = None
        if self._decode_graph is None:
            tok_buf = cur.clone()
            pos_buf = positions.clone()
            cache = self.kv_cache
            saved_lens = cache.seq_lens.clone()
            dev = self.device
            if sp.temperature > 0:
                C = ops._require_ext()
                self._sample_ctr = torch.zeros(1, device=dev,
                                               dtype=torch.int64)
                ctr = self._sample_ctr
                temps = torch.full((B,), float(sp.temperature), device=dev)
                tps = torch.full((B,), float(sp.top_p), device=dev)
                tks = torch.full((B,), int(sp.top_k), device=dev,
                                 dtype=torch.int32)
                zero = torch.zeros(B, device=dev)
                # the captured kernel reads these tensors BY ADDRESS on
                # every replay: they must outlive this call or the
                # allocator recycles them under the graph (the capture
                # call then samples correctly and every later call reads
                # recycled bytes — a very fun bug to find)
                self._graph_keep = (temps, tps, tks, zero)
                import random as _random
                seed_base = (sp.seed if sp.seed is not None
                             else _random.getrandbits(62))

                def pick(logits):
                    C.bump_sample_counter(ctr)
                    return ops.sample_tokens(
                        logits, temps=temps, top_ps=tps, top_ks=tks,
                        pres=zero, freqs=zero, counter=ctr,
                        seed_base=seed_base)
            else:
                def pick(logits):
                    return logits.argmax(-1)

            def step():
                logits = self.stage(tok_buf.unsqueeze(1),
                                    pos_buf.unsqueeze(1), kv_cache=cache)
                tok_buf.copy_(pick(logits.squeeze(1)))
                pos_buf.add_(1)

            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                step()  # warmup (writes garbage at pos S; replay 1 rewrites)
            torch.cuda.current_stream().wait_stream(s)
            cache.seq_lens.copy_(saved_lens)
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                step()
            cache.seq_lens.copy_(saved_lens)
            # one throwaway replay: the very first replay after capture
            # observed once-off divergence on the sampled path (capture
            # pool warm-up); flush it so the capture call's real replays
            # see the same state as every later call
            graph.replay()
            cache.seq_lens.copy_(saved_lens)
            self._decode_graph = (graph, tok_buf, pos_buf, skey)
        graph, tok_buf, pos_buf = self._decode_graph[:3]
        if (sp.temperature > 0 and sp.seed is not None
                and getattr(self, "_sample_ctr", None) is not None):
            # reset the RNG counter so a SEEDED generate() reproduces;
            # unseeded calls keep advancing it (fresh draws per call)
            self._sample_ctr.zero_()
        tok_buf.copy_(cur)
        pos_buf.copy_(positions)
        for t in range(1, T):
            graph.replay()
            out_tokens[:, t].copy_(tok_buf)
        positions.copy_(pos_buf)

    def _finish_stats(self, B, S, T, t_start, t_first,
                      t_firsts=None):
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t_end = time.perf_counter()
        stats = {
            "batch": B, "prompt_len": S, "new_tokens": T,
            "total_s": t_end - t_start,
            "ttft_s": (t_first - t_start) if t_first is not None else None,
            # honest per-REQUEST TTFT: micro-batch mb's requests see their
            # first token after mb+1 prefill chunks; the median over
            # requests is the middle micro-batch's timestamp
            "ttft_p50_s": (statistics.median(t_firsts) - t_start
                           if t_firsts else
                           (t_first - t_start) if t_first is not None
                           else None),
            "output_tokens_per_s": B * T / (t_end - t_start),
        }
        if self.world > 1:
            stats = self.p2p.broadcast_obj(
                stats if self.is_last else None, src=self.world - 1)
        return stats


class PipelineTrainer:
    """1F1B pipeline-parallel trainer (SPMD).

    Replaces the reference's thread-per-micro-batch forward/backward fan-out
    (``ml/module.py:374-436``) and per-worker optimizer IPC
    (``ml/optim.py:131-203``). Schedule: non-interleaved 1F1B — warmup
    forwards, steady fwd+bwd pairs, cooldown backwards; activations and
    gradients move as RCCL P2P between adjacent ranks.
    """

    def __init__(self, plan: StagePlan, rank: int, world: int, device=None,
                 init: str = "random", ckpt_dir: Optional[str] = None,
                 dtype=None, seed: int = 0, lr: float = 1e-4, group=None,
                 rank_base: int = 0, max_grad_norm: Optional[float] = None,
                 grad_checkpointing: bool = False,
                 lora_r: Optional[int] = None, lora_alpha: float = 16.0,
                 **opt_kwargs):
        self.runner = PipelineRunner(plan, rank, world, device=device,
                                     init=init, ckpt_dir=ckpt_dir,
                                     dtype=dtype, seed=seed, group=group,
                                     rank_base=rank_base)
        # called with (self) right before optimizer.step(): the DP grad
        # all-reduce hook attaches here (parallel/dp.py)
        self.grad_hook = None
        self.stage = self.runner.stage
        self.stage.train()
        self.rank, self.world = rank, world
        self.device = self.runner.device
        self.dtype = self.runner.dtype
        self.p2p = self.runner.p2p
        self.H = self.runner.H
        from tensorlink_amd.optim import FusedAdamW
        if lora_r:
            # LoRA fine-tuning: freeze the stage, train only adapters
            from tensorlink_amd.models.lora import (apply_lora,
                                                    lora_parameters)
            apply_lora(self.stage, r=lora_r, alpha=lora_alpha)
            self.optimizer = FusedAdamW(lora_parameters(self.stage),
                                        lr=lr, **opt_kwargs)
        else:
            self.optimizer = FusedAdamW(self.stage.parameters(), lr=lr,
                                        **opt_kwargs)
        self.max_grad_norm = max_grad_norm
        self.lr_scheduler = None        # optional WarmupCosineLR
        # trade activation memory for a recompute pass per layer
        self.stage.grad_checkpointing = grad_checkpointing
        # chunked CE for big vocabularies (memory); plain CE for tiny
        # test configs (keeps exact parity with existing expectations)
        self._chunked_ce = (self.stage.has_head
                            and plan.config.vocab_size >= 32000
                            and hasattr(self.stage, "norm")
                            and hasattr(self.stage, "lm_head"))
        # (v0, group) when the LM head is vocab-sharded across a TP
        # group (set by TPPPTrainer(vocab_parallel=True))
        self._vp = None

    def train_step(self, input_ids: Optional[torch.Tensor] = None,
                   labels: Optional[torch.Tensor] = None,
                   n_micro: int = None) -> float:
        """One optimizer step over a global batch. input_ids/labels are
        significant on rank 0; returns the mean loss on every rank."""
        P, r = self.world, self.rank
        is_first = r == 0
        is_last = r == P - 1

        if P > 1:
            meta = (tuple(input_ids.shape), n_micro) if is_first else None
            meta = self.p2p.broadcast_obj(meta, src=0)
            (B, S), n_micro = meta
        else:
            B, S = input_ids.shape
        M = n_micro or max(1, P)
        M = min(M, B)
        while B % M != 0:
            M -= 1
        b = B // M

        if is_first:
            input_ids = input_ids.to(self.device)
            labels = labels.to(self.device)
        if P > 1:
            # labels go to the last rank (blocking send: once per step,
            # off the hot path)
            if is_first and not is_last:
                self.p2p.send(labels.to(torch.int64), P - 1)
                labels_last = None
            elif is_last and not is_first:
                labels_last = self.p2p.recv((B, S), torch.int64, 0,
                                            self.device)
            else:
                labels_last = labels if is_last else None
        else:
            labels_last = labels

        pos = torch.arange(S, device=self.device,
                           dtype=torch.int32).unsqueeze(0).expand(b, -1).contiguous()
        from tensorlink_amd import ops as tl_ops

        stash = {}          # micro -> (inp, out) or (None, loss)
        losses = []
        inflight = []

        def _isend(t, dst):
            t = t.contiguous()
            inflight.append((self.p2p.isend(t, dst), t))

        def fwd(i):
            s, e = i * b, (i + 1) * b
            want_logits = is_last and not self._chunked_ce \
                and self._vp is None
            if is_first:
                inp = None
                out = self.stage(input_ids[s:e], pos, training=True,
                                 return_logits=want_logits)
            else:
                inp = self.p2p.recv((b, S, self.H), self.dtype, r - 1,
                                    self.device).requires_grad_(True)
                out = self.stage(inp, pos, training=True,
                                 return_logits=want_logits)
            if is_last:
                loss = self._loss(out, labels_last[s:e]) / M
                losses.append(loss.detach())
                stash[i] = (inp, loss)
            else:
                _isend(out.detach(), r + 1)
                stash[i] = (inp, out)

        def bwd(i):
            inp, out = stash.pop(i)
            if is_last:
                out.backward()          # out is the loss
            else:
                g = self.p2p.recv((b, S, self.H), self.dtype, r + 1,
                                  self.device)
                out.backward(g)
            if not is_first:
                _isend(inp.grad, r - 1)

        warmup = min(M, P - 1 - r)
        f = bw = 0
        for _ in range(warmup):
            fwd(f); f += 1
        for _ in range(M - warmup):
            fwd(f); f += 1
            bwd(bw); bw += 1
        while bw < M:
            bwd(bw); bw += 1
        for w, _ in inflight:
            w.wait()

        if self.grad_hook is not None:
            self.grad_hook(self)
        if self.max_grad_norm is not None:
            # global norm across pipeline stages: each rank reduces its
            # flat buffer, then one scalar all-reduce
            ns = self.optimizer.grad_norm_sq()
            if P > 1:
                import torch.distributed as dist
                dist.all_reduce(ns, group=self.runner.group)
            self.optimizer.clip_grad_norm_(self.max_grad_norm, norm_sq=ns)
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        self.optimizer.step()
        self.optimizer.zero_grad()

        loss_val = float(torch.stack(losses).sum()) if is_last else None
        if P > 1:
            loss_val = self.p2p.broadcast_obj(loss_val, src=P - 1)
        return loss_val

    # ------------------------------------------------------------------
    # Split forward/backward for the DistributedModel user API: the user
    # computes an arbitrary loss on rank 0's returned logits and calls
    # loss.backward(), which routes here (reference CustomAutogradRouter,
    # ml/module.py:126-144).
    # ------------------------------------------------------------------
    def _loss(self, out, labels):
        """CE over one micro-batch. Large vocabularies use the chunked
        head-GEMM+CE (no [B,S,V] logits materialization — saves
        ~0.6 MB/token at Qwen vocab); small ones take the plain path.
        Both are exactly F.cross_entropy(head(hidden), labels)."""
        from tensorlink_amd import ops as tl_ops
        if self._vp is not None:
            from tensorlink_amd.parallel.tp import vocab_parallel_ce
            v0, group = self._vp
            h = tl_ops.rmsnorm(out, self.stage.norm.to(out.dtype),
                               self.stage.config.rms_norm_eps)
            lg = self.stage.lm_head(h)[:, :-1]
            lb = labels[:, 1:].reshape(-1)
            return vocab_parallel_ce(
                lg.reshape(-1, lg.shape[-1]), lb, v0, group)
        if self._chunked_ce:
            # apply the final norm here (stage.head fuses norm+GEMM;
            # the chunked CE takes the raw head weight)
            h = tl_ops.rmsnorm(out, self.stage.norm.to(out.dtype),
                               self.stage.config.rms_norm_eps)
            w = (self.stage.embed_tokens.weight
                 if self.stage.config.tie_word_embeddings
                 and self.stage.has_embedding
                 else self.stage.lm_head.weight)
            return tl_ops.chunked_causal_lm_loss(h, w, labels)
        return tl_ops.causal_lm_loss(out, labels)

    # ------------------------------------------------------------------
    # training resume: weights + optimizer moments + step count
    # (the reference checkpoints weights only — parameter retrieval,
    #  ml/module.py:577-670; moment/step resume exceeds parity)
    # ------------------------------------------------------------------
    def save_checkpoint(self, out_dir: str) -> str:
        import os

        from tensorlink_amd.models.loader import save_stage_to_safetensors
        save_stage_to_safetensors(self.stage, out_dir, self.rank)
        torch.save(self.optimizer.state_dict(),
                   os.path.join(out_dir, f"optim_{self.rank}.pt"))
        return out_dir

    def load_checkpoint(self, out_dir: str) -> None:
        """Restore weights + optimizer state saved by save_checkpoint
        under the SAME pipeline partitioning. load_state_dict copies
        in-place, so the flat-buffer views FusedAdamW re-pointed the
        parameters into stay valid."""
        import os

        from tensorlink_amd.models.loader import load_stage_from_safetensors
        load_stage_from_safetensors(self.stage, out_dir, self.rank)
        sd = torch.load(os.path.join(out_dir, f"optim_{self.rank}.pt"),
                        weights_only=False)
        self.optimizer.load_state_dict(sd)

    def spmd_forward(self, input_ids: Optional[torch.Tensor] = None):
        """Single-micro-batch pipeline forward with autograd stashes.
        Returns logits [B,S,V] on rank 0 (detached), None elsewhere."""
        P, r = self.world, self.rank
        is_first, is_last = r == 0, r == P - 1
        if P > 1:
            meta = tuple(input_ids.shape) if is_first else None
            B, S = self.p2p.broadcast_obj(meta, src=0)
        else:
            B, S = input_ids.shape
        pos = torch.arange(S, device=self.device, dtype=torch.int32) \
            .unsqueeze(0).expand(B, -1).contiguous()
        if is_first:
            input_ids = input_ids.to(self.device)
            inp = None
            out = self.stage(input_ids, pos, training=True,
                             return_logits=is_last)
        else:
            inp = self.p2p.recv((B, S, self.H), self.dtype, r - 1,
                                self.device).requires_grad_(True)
            out = self.stage(inp, pos, training=True, return_logits=is_last)
        self._fb_stash = (inp, out, (B, S))
        if P == 1:
            return out          # local graph intact: user autograd works
        if is_last:
            self.p2p.send(out.detach().to(self.dtype), 0)
            return None
        self.p2p.send(out.detach(), r + 1)
        if is_first:
            V = self.runner.config.vocab_size
            logits = self.p2p.recv((B, S, V), self.dtype, P - 1, self.device)
            return logits
        return None

    def spmd_backward(self, grad_logits: Optional[torch.Tensor] = None):
        """Backward for the stashed spmd_forward."""
        P, r = self.world, self.rank
        is_first, is_last = r == 0, r == P - 1
        inp, out, (B, S) = self._fb_stash
        if P == 1:
            raise RuntimeError("PP=1 uses local autograd directly")
        if is_last:
            V = self.runner.config.vocab_size
            g = self.p2p.recv((B, S, V), self.dtype, 0, self.device)
            out.backward(g)
            self.p2p.send(inp.grad, r - 1)
        elif not is_first:
            g = self.p2p.recv((B, S, self.H), self.dtype, r + 1, self.device)
            out.backward(g)
            self.p2p.send(inp.grad, r - 1)
        else:
            if grad_logits is not None:
                self.p2p.send(grad_logits.to(self.device, self.dtype), P - 1)
            g = self.p2p.recv((B, S, self.H), self.dtype, r + 1, self.device)
            out.backward(g)
        self._fb_stash = None


def _runner_forward_logits(self, input_ids=None):
    """SPMD eval forward: full logits [B,S,V] on rank 0 (no grad, no KV
    cache). The reference's remote-inference forward path
    (tests/test_distributed_model.py:27-38)."""
    with torch.no_grad():
        if self.world == 1:
            B, S = input_ids.shape
            pos = torch.arange(S, device=self.device, dtype=torch.int32) \
                .unsqueeze(0).expand(B, -1).contiguous()
            return self.stage(input_ids.to(self.device), pos)
        meta = tuple(input_ids.shape) if self.is_first else None
        B, S = self.p2p.broadcast_obj(meta, src=0)
        pos = torch.arange(S, device=self.device, dtype=torch.int32) \
            .unsqueeze(0).expand(B, -1).contiguous()
        if self.is_first:
            h = self.stage(input_ids.to(self.device), pos,
                           return_logits=False)
            self.p2p.send(h.to(self.dtype), self.next_rank)
            V = self.config.vocab_size
            return self.p2p.recv((B, S, V), self.dtype, self.world - 1,
                                 self.device)
        h = self.p2p.recv((B, S, self.H), self.dtype, self.prev_rank,
                          self.device)
        out = self.stage(h, pos, return_logits=self.is_last)
        if self.is_last:
            self.p2p.send(out.to(self.dtype), 0)
        else:
            self.p2p.send(out, self.next_rank)
        return None


PipelineRunner.forward_logits = _runner_forward_logits
