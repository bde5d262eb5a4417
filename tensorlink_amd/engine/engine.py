"""Serving engine: hosted-model table, request execution, streaming,
demand tracking and metrics.

This is the MI355X collapse of the reference's validator serving machinery
(``tensorlink/ml/validator.py`` — ``_handle_generate_request`` 642,
``_prepare_generation`` 579, ``_generate_streaming`` 724, hosted-job
lifecycle 901-1103, demand tracking 181-260) plus the job table that
replaces the DHT. Each generated token goes decode-loop -> queue -> SSE
(the reference routes it GPU -> IPC -> TCP -> net thread -> IPC poll ->
format -> IPC -> asyncio queue).

SPMD: rank 0 owns the engine API; ranks 1..N-1 sit in
:meth:`InferenceEngine.worker_loop` executing broadcast commands.
"""

from __future__ import annotations

import collections
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, Iterator, List, Optional

import torch

from tensorlink_amd.engine.formatter import (ResponseFormatter,
                                             format_chat_prompt,
                                             normalize_generate_args)
from tensorlink_amd.engine.tokenizer import load_tokenizer
from tensorlink_amd.models.configs import get_config
from tensorlink_amd.parallel.comm import P2P, device_for_rank
from tensorlink_amd.parallel.pipeline import PipelineRunner, SamplingParams
from tensorlink_amd.parallel.planner import plan_for_world


@dataclass
class ModelJob:
    name: str
    runner: PipelineRunner
    tokenizer: object
    state: str = "ready"            # loading | ready | failed
    created: float = field(default_factory=time.time)
    requests_served: int = 0
    tokens_generated: int = 0
    last_request: float = 0.0
    batcher: object = None          # ContinuousBatcher when enabled


class EngineMetrics:
    """Keeper-lite: rolling serving statistics (reference
    ``nodes/keeper.py:259-501`` network-history rollups)."""

    def __init__(self, window: int = 3600):
        self.start_time = time.time()
        self.requests_total = 0
        self.tokens_total = 0
        self.errors_total = 0
        self.latencies: collections.deque = collections.deque(maxlen=512)
        self.history: collections.deque = collections.deque(maxlen=1440)
        self._lock = threading.Lock()
        self._auto_lock = threading.Lock()
        self._auto_loaded: set = set()
        self._auto_demand_mark: Dict[str, int] = {}
        self._auto_cold: Dict[str, int] = {}   # unloaded-at demand level
        self._last_roll = time.time()

    def record(self, tokens: int, latency_s: float, error: bool = False):
        with self._lock:
            self.requests_total += 1
            self.tokens_total += tokens
            if error:
                self.errors_total += 1
            self.latencies.append(latency_s)
            now = time.time()
            if now - self._last_roll >= 60:
                self.history.append({
                    "t": int(now), "requests": self.requests_total,
                    "tokens": self.tokens_total})
                self._last_roll = now

    def snapshot(self) -> Dict:
        with self._lock:
            lat = sorted(self.latencies)
            p50 = lat[len(lat) // 2] if lat else None
            return {
                "uptime_s": time.time() - self.start_time,
                "requests_total": self.requests_total,
                "tokens_total": self.tokens_total,
                "errors_total": self.errors_total,
                "latency_p50_s": p50,
                "history": list(self.history)[-60:],
            }


class InferenceEngine:
    """Rank-0 serving front end + SPMD worker protocol."""

    def __init__(self, rank: int = 0, world: int = 1, device=None,
                 default_init: str = "random"):
        self.rank = rank
        self.world = world
        self.device = (torch.device(device) if device is not None
                       else device_for_rank())
        self.default_init = default_init
        self.jobs: Dict[str, ModelJob] = {}
        self.metrics = EngineMetrics()
        self.demand: Dict[str, int] = collections.Counter()
        self.p2p = P2P(rank, world) if world > 1 else None
        self._lock = threading.Lock()
        self._auto_lock = threading.Lock()
        self._auto_loaded: set = set()
        self._auto_demand_mark: Dict[str, int] = {}
        self._auto_cold: Dict[str, int] = {}   # unloaded-at demand level
        # serializes SPMD runner use: one in-flight generation per engine
        # (continuous batching is a later milestone)
        self._serve_lock = threading.Lock()

    # ---------------- worker side ----------------
    def worker_loop(self):
        """Ranks 1..N-1: execute commands broadcast from rank 0."""
        assert self.rank != 0
        while True:
            cmd = self.p2p.broadcast_obj(None, src=0)
            op = cmd[0]
            if op == "shutdown":
                return
            if op == "load":
                _, name, init, ckpt_dir, kwargs = cmd
                self._build_runner(name, init, ckpt_dir, **kwargs)
            elif op == "unload":
                self.jobs.pop(cmd[1], None)
            elif op == "generate":
                _, name = cmd
                job = self.jobs[name]
                job.runner.generate(None, None)

    def _bcast(self, cmd):
        if self.world > 1:
            self.p2p.broadcast_obj(cmd, src=0)

    def shutdown(self):
        if self.rank == 0:
            self._bcast(("shutdown",))

    # ---------------- model lifecycle ----------------
    def _build_runner(self, name: str, init: str, ckpt_dir: Optional[str],
                      **kwargs) -> ModelJob:
        config = get_config(name)
        quantize = kwargs.pop("quantize", None)
        continuous = kwargs.pop("continuous", False)
        max_slots = kwargs.pop("max_slots", 16)
        max_ctx = kwargs.pop("max_ctx", 2048)
        pool_pages = kwargs.pop("pool_pages", None)
        prefill_chunk = kwargs.pop("prefill_chunk", None)
        prefix_caching = kwargs.pop("prefix_caching", False)
        speculative = kwargs.pop("speculative", False)
        lookup_n = kwargs.pop("lookup_n", 3)
        spec_k = kwargs.pop("spec_k", 8)
        plan = plan_for_world(config, self.world, **kwargs)
        import os as _os
        if init == "random" and ckpt_dir is None and _os.path.isdir(name) \
                and (_os.path.exists(_os.path.join(
                    name, "model.safetensors"))
                     or _os.path.exists(_os.path.join(
                         name, "model.safetensors.index.json"))):
            # serving a checkpoint DIRECTORY: load its weights, don't
            # random-init over them
            init, ckpt_dir = "checkpoint", name
        runner = PipelineRunner(plan, self.rank, self.world,
                                device=self.device, init=init,
                                ckpt_dir=ckpt_dir, quantize=quantize)
        tok = load_tokenizer(ckpt_dir or name)
        job = ModelJob(name=name, runner=runner, tokenizer=tok)
        if continuous and self.world == 1 and self.rank == 0:
            from tensorlink_amd.engine.batcher import ContinuousBatcher
            job.batcher = ContinuousBatcher(
                runner, max_slots=max_slots, max_ctx=max_ctx,
                pool_pages=pool_pages, prefill_chunk=prefill_chunk,
                prefix_caching=prefix_caching,
                speculative=speculative, lookup_n=lookup_n,
                spec_k=spec_k).start()
        elif continuous and self.world > 1:
            from tensorlink_amd.engine.batcher import PPContinuousBatcher
            b = PPContinuousBatcher(runner, max_slots=max_slots,
                                    max_ctx=max_ctx, pool_pages=pool_pages,
                                    prefill_chunk=prefill_chunk,
                                    prefix_caching=prefix_caching)
            if self.rank == 0:
                job.batcher = b.start()
                self.jobs[name] = job
            else:
                self.jobs[name] = job
                b.serve_follower()      # blocks until rank 0 stops it
            return job
        self.jobs[name] = job
        return job

    def load_model(self, name: str, init: Optional[str] = None,
                   ckpt_dir: Optional[str] = None, **kwargs) -> ModelJob:
        """/request-model analog (reference api/node.py:223 →
        validator job pipeline §3.5 — here a direct call)."""
        with self._lock:
            if name in self.jobs:
                return self.jobs[name]
            if (kwargs.get("continuous") and self.world > 1
                    and any(j.batcher is not None
                            for j in self.jobs.values())):
                # follower ranks serve ONE PP batcher command stream at a
                # time (single collective channel per group)
                raise ValueError(
                    "one continuous PP job per engine; unload the "
                    "existing one first")
            init = init or self.default_init
            self._bcast(("load", name, init, ckpt_dir, kwargs))
            try:
                return self._build_runner(name, init, ckpt_dir, **kwargs)
            except Exception:
                self.jobs[name] = ModelJob(name=name, runner=None,
                                           tokenizer=None, state="failed")
                raise

    def unload_model(self, name: str):
        with self._lock:
            job = self.jobs.pop(name, None)
            # stop the batcher FIRST: follower ranks sit in its command
            # loop and must see batch_stop before the unload broadcast
            if job is not None and job.batcher is not None:
                job.batcher.stop()
            self._bcast(("unload", name))

    def model_status(self, name: str) -> Dict:
        job = self.jobs.get(name)
        if job is None:
            return {"model": name, "status": "not_loaded"}
        b = job.batcher
        extra = {}
        if b is not None:
            extra = {"batcher": {
                "steps": b.steps,
                "preemptions": getattr(b, "preemptions", 0),
                "swapped_waiting": len(getattr(b, "_swapped", [])),
                "prefix_cache_hit_tokens": getattr(b.cache, "hits", 0),
                "pages_free": b.cache.available_pages(),
                "pages_total": b.cache.allocator.n_pages}}
        return {"model": name, "status": job.state, **extra,
                "requests_served": job.requests_served,
                "tokens_generated": job.tokens_generated}

    def models(self) -> List[Dict]:
        return [self.model_status(n) for n in self.jobs]

    # ---------------- generation ----------------
    def _prepare(self, request: Dict):
        name = request.get("hf_name") or request.get("model")
        if not name:
            raise ValueError("request needs 'hf_name' or 'model'")
        # reference aliases (README.md request table): `prompt` is an
        # alternative to `message`; `is_chat_completion` selects the
        # OpenAI response shape
        if not request.get("message") and request.get("prompt"):
            request = dict(request, message=request["prompt"])
        if request.get("is_chat_completion") and \
                "output_format" not in request:
            request = dict(request, output_format="openai")
        self.demand[name] += 1
        job = self.jobs.get(name)
        if job is None or job.state != "ready":
            raise KeyError(f"model {name!r} not loaded (POST /request-model)")

        if request.get("input_format", "chat") == "chat" and \
                (request.get("history") or request.get("message")):
            messages = list(request.get("history") or [])
            if request.get("message"):
                messages.append({"role": "user",
                                 "content": request["message"]})
            prompt = format_chat_prompt(messages, job.tokenizer, job.name)
        else:
            prompt = request.get("message", "")

        args = normalize_generate_args(request, job.tokenizer)
        ids = job.tokenizer(prompt, return_tensors="pt")["input_ids"]
        max_ctx = job.runner.config.max_position_embeddings
        # clamp max_new_tokens first (formatter caps can exceed a small
        # model's context, e.g. GPT-2 1024 vs the 2048 default) and keep
        # at least one prompt token — as batcher.submit does
        args["max_new_tokens"] = max(1, min(args["max_new_tokens"],
                                            max_ctx - 1))
        if ids.shape[1] + args["max_new_tokens"] > max_ctx:
            keep = max(1, max_ctx - args["max_new_tokens"])
            ids = ids[:, -keep:]
        sp = SamplingParams(
            temperature=args["temperature"] if args["do_sample"] else 0.0,
            top_p=args["top_p"], top_k=int(args.get("top_k", 0) or 0),
            max_new_tokens=args["max_new_tokens"],
            eos_token_id=args.get("eos_token_id"),
            presence_penalty=args.get("presence_penalty", 0.0),
            frequency_penalty=args.get("frequency_penalty", 0.0),
            seed=request.get("seed"))
        stop = request.get("stop") or []
        if isinstance(stop, str):
            stop = [stop]
        self._num_beams = int(args.get("num_beams", 1) or 1)
        return job, prompt, ids, sp, list(stop)

    @staticmethod
    def _apply_stop(text: str, stop) -> str:
        """Truncate at the earliest stop string (OpenAI `stop` param)."""
        for s in stop:
            idx = text.find(s)
            if idx >= 0:
                text = text[:idx]
        return text

    def generate(self, request: Dict) -> Dict:
        """Non-streaming completion; returns a formatted response dict."""
        t0 = time.time()
        fmt = ResponseFormatter(request.get("hf_name") or
                                request.get("model", ""),
                                request.get("output_format", "openai"))
        try:
            job, prompt, ids, sp, stop = self._prepare(request)
            n = max(1, int(request.get("n", 1) or 1))
            if job.batcher is not None:
                kw = dict(max_new_tokens=sp.max_new_tokens,
                          temperature=sp.temperature, top_p=sp.top_p,
                          top_k=sp.top_k, eos_token_id=sp.eos_token_id,
                          presence_penalty=sp.presence_penalty,
                          frequency_penalty=sp.frequency_penalty,
                          seed=sp.seed,
                          logprobs=bool(request.get("logprobs")))
                reqs = [job.batcher.submit(ids[0], **kw)
                        for _ in range(n)]
                results = [rq.result() for rq in reqs]
                out_tokens = results[0]
                n_new = sum(len(t) for t in results)
                texts = [self._apply_stop(
                    job.tokenizer.decode(t, skip_special_tokens=True),
                    stop) for t in results]
                text = texts[0]
                extra = texts[1:] or None
                job.requests_served += 1
                job.tokens_generated += n_new
                self.metrics.record(n_new, time.time() - t0)
                resp = fmt.format_response(
                    text, prompt_tokens=ids.shape[1],
                    completion_tokens=n_new,
                    reasoning=bool(request.get("reasoning")),
                    processing_time=time.time() - t0,
                    extra_texts=extra)
                if request.get("logprobs"):
                    lp = reqs[0].logprob_values
                    if "choices" in resp:
                        resp["choices"][0]["logprobs"] = {
                            "token_logprobs": lp}
                    else:
                        resp["logprobs"] = lp
                return resp
            nb = getattr(self, "_num_beams", 1)
            extra = None
            if nb > 1 and self.world == 1:
                with self._serve_lock:
                    out = job.runner.generate_beam(
                        ids, max_new_tokens=sp.max_new_tokens,
                        num_beams=nb, eos_token_id=sp.eos_token_id)
            elif (request.get("speculative") and self.world == 1
                    and sp.temperature <= 0):
                with self._serve_lock:
                    out, _ = job.runner.generate_speculative(
                        ids, max_new_tokens=sp.max_new_tokens,
                        eos_token_id=sp.eos_token_id)
            else:
                with self._serve_lock:
                    self._bcast(("generate", job.name))
                    out, stats = job.runner.generate(ids, sp,
                                                     return_stats=True)
                if n > 1 and sp.temperature > 0:
                    # serial path: additional sampled completions
                    extra = []
                    for _ in range(n - 1):
                        with self._serve_lock:
                            self._bcast(("generate", job.name))
                            o2 = job.runner.generate(ids, sp)
                        extra.append(self._apply_stop(
                            job.tokenizer.decode(o2[0].tolist(),
                                                 skip_special_tokens=True),
                            stop))
            text = job.tokenizer.decode(out[0].tolist(),
                                        skip_special_tokens=True)
            text = self._apply_stop(text, stop)
            n_new = out.shape[1]
            job.requests_served += 1
            job.tokens_generated += n_new
            job.last_request = time.time()
            self.metrics.record(n_new, time.time() - t0)
            return fmt.format_response(
                text, prompt_tokens=ids.shape[1], completion_tokens=n_new,
                reasoning=bool(request.get("reasoning")),
                processing_time=time.time() - t0, extra_texts=extra)
        except Exception as e:
            self.metrics.record(0, time.time() - t0, error=True)
            return fmt.format_error(str(e))

    def generate_stream(self, request: Dict) -> Iterator[str]:
        """SSE generator: decode-loop -> queue -> yield."""
        t0 = time.time()
        fmt = ResponseFormatter(request.get("hf_name") or
                                request.get("model", ""),
                                request.get("output_format", "openai"))
        try:
            job, prompt, ids, sp, stop = self._prepare(request)
        except Exception as e:
            yield f"data: {__import__('json').dumps(fmt.format_error(str(e)))}\n\n"
            yield "data: [DONE]\n\n"
            return

        q: queue.Queue = queue.Queue()
        emitted = 0

        def on_token(step, toks):
            q.put(int(toks[0]))

        breq = {}                       # batcher Request (for cancel)

        def run():
            try:
                if job.batcher is not None:
                    req = job.batcher.submit(
                        ids[0], max_new_tokens=sp.max_new_tokens,
                        temperature=sp.temperature, top_p=sp.top_p,
                        top_k=sp.top_k, eos_token_id=sp.eos_token_id,
                        presence_penalty=sp.presence_penalty,
                        frequency_penalty=sp.frequency_penalty,
                        seed=sp.seed)
                    breq["req"] = req
                    for tok in req.stream():
                        q.put(tok)
                else:
                    with self._serve_lock:
                        self._bcast(("generate", job.name))
                        out = job.runner.generate(ids, sp,
                                                  on_token=on_token)
                    if out is not None:
                        q.put(("final", out[0].tolist()))
            except Exception as e:  # pragma: no cover
                q.put(("error", str(e)))
            q.put(None)

        thread = threading.Thread(target=run, daemon=True)
        thread.start()

        sent: List[int] = []
        first = True
        prev_text = ""
        try:
            yield from self._stream_body(q, fmt, job, sp, sent, t0,
                                         stop, request)
        finally:
            # client disconnected (GeneratorExit) or stream finished:
            # stop the batcher request so its slot frees immediately
            if "req" in breq and not breq["req"].done.is_set():
                breq["req"].cancel()

    def _stream_body(self, q, fmt, job, sp, sent, t0, stop, request):
        emitted = 0
        first = True
        prev_text = ""
        while True:
            item = q.get(timeout=120)
            if item is None:
                break
            if isinstance(item, tuple):
                kind, payload = item
                if kind == "error":
                    yield f"data: {__import__('json').dumps(fmt.format_error(payload))}\n\n"
                    break
                # final: emit any tokens the ring didn't stream back
                sent_n = len(sent)
                for tok in payload[sent_n:]:
                    text = job.tokenizer.decode(sent + [tok],
                                                skip_special_tokens=True)
                    delta = text[len(prev_text):]
                    prev_text = text
                    sent.append(tok)
                    emitted += 1
                    yield fmt.format_stream_chunk(delta, first=first)
                    first = False
                continue
            sent.append(item)
            text = job.tokenizer.decode(sent, skip_special_tokens=True)
            delta = text[len(prev_text):]
            prev_text = text
            emitted += 1
            if delta:
                yield fmt.format_stream_chunk(delta, first=first)
                first = False
        job.requests_served += 1
        job.tokens_generated += emitted
        self.metrics.record(emitted, time.time() - t0)
        yield fmt.format_final_chunk()

    # ---------------- status ----------------
    def stats(self) -> Dict:
        s = self.metrics.snapshot()
        s["models"] = self.models()
        s["world_size"] = self.world
        if torch.cuda.is_available():
            free, total = torch.cuda.mem_get_info(self.device)
            s["gpu_memory"] = {"free": free, "total": total}
        return s

    def manage_auto_loaded_models(self, max_auto: int = 2,
                                  min_demand: int = 1) -> None:
        """Demand-driven auto load/unload of the configured default
        models (reference ``ml/validator.py:278``:
        ``_manage_auto_loaded_models`` keeps DEFAULT_MODELS hot and drops
        cold ones). Called periodically — e.g. from a Watchdog tick.

        - every configured default model not yet loaded is loaded (up to
          ``max_auto`` auto-loaded models);
        - an auto-loaded model whose demand since the last sweep is
          below ``min_demand`` is unloaded (explicitly-loaded models are
          never touched).
        """
        from tensorlink_amd.config import DEFAULT_MODELS
        with self._auto_lock:
            # unload cold auto-loaded models
            for name in list(self._auto_loaded):
                idle = (self.demand.get(name, 0)
                        - self._auto_demand_mark.get(name, 0))
                if idle < min_demand and name in self.jobs:
                    self.unload_model(name)
                    self._auto_loaded.discard(name)
                    # cooled: only reload once demand grows again
                    self._auto_cold[name] = self.demand.get(name, 0)
            # load configured defaults
            for name in DEFAULT_MODELS:
                if len(self._auto_loaded) >= max_auto:
                    break
                if name not in self.jobs:
                    if name in self._auto_cold and \
                            self.demand.get(name, 0) <= \
                            self._auto_cold[name]:
                        continue
                    try:
                        self.load_model(name)
                        self._auto_loaded.add(name)
                    except Exception:   # unknown config etc. — skip
                        continue
            for name in self._auto_loaded:
                self._auto_demand_mark[name] = self.demand.get(name, 0)

    def model_demand(self) -> Dict[str, int]:
        return dict(self.demand)
